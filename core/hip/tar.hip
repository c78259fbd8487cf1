// CDNA4 tar kernels: index a tar archive resident in HBM and scatter file
// payloads to per-file device buffers.
//
// Replaces the reference's CPU UnTGZ extract loop (pkg/client/helper.go:
// 55-79) for archives landed in HBM by the pinned-ring engine. The header
// walk is inherently sequential (each header's size field locates the next),
// so one device thread builds the entry table (hundreds of 512 B headers —
// microseconds); the payload scatter is the bulk work and runs as a grid of
// 256-thread workgroups, one per ≤4 MiB segment, uint4-vectorized.
#include <hip/hip_runtime.h>

#include <cstdint>

namespace {

struct TarEntry {
  uint64_t header_off;   // offset of the 512 B header block
  uint64_t payload_off;  // offset of file data
  uint64_t size;         // file size in bytes
  uint32_t typeflag;     // '0'/'\0' regular, '5' dir, 'L' GNU longname, ...
  uint32_t mode;         // octal-decoded
};

__device__ uint64_t parse_octal(const uint8_t* p, int n) {
  uint64_t v = 0;
  for (int i = 0; i < n; i++) {
    uint8_t c = p[i];
    if (c == ' ' || c == 0) {
      if (v) break;
      continue;
    }
    if (c < '0' || c > '7') break;
    v = (v << 3) | (c - '0');
  }
  return v;
}

// Single-thread sequential header walk. entries/count sized by caller.
__global__ void tar_index_kernel(const uint8_t* __restrict__ tar, uint64_t tar_len,
                                 TarEntry* __restrict__ entries, uint32_t max_entries,
                                 uint32_t* __restrict__ count, uint32_t* __restrict__ error) {
  if (blockIdx.x != 0 || threadIdx.x != 0) return;
  uint64_t off = 0;
  uint32_t n = 0;
  *error = 0;
  while (off + 512 <= tar_len) {
    const uint8_t* hdr = tar + off;
    // empty block = end of archive
    bool empty = true;
    for (int i = 0; i < 512 && empty; i++) empty = hdr[i] == 0;
    if (empty) break;
    uint64_t size = parse_octal(hdr + 124, 12);
    uint32_t mode = static_cast<uint32_t>(parse_octal(hdr + 100, 8));
    uint32_t typeflag = hdr[156];
    if (n < max_entries) {
      entries[n] = TarEntry{off, off + 512, size, typeflag, mode};
      n++;
    } else {
      *error = 2;  // table overflow
      break;
    }
    uint64_t padded = (size + 511) & ~511ull;
    off += 512 + padded;
  }
  *count = n;
}

// Scatter segments: seg i copies src[src_off..src_off+len) -> dst_ptr.
struct CopySeg {
  uint64_t src_off;
  uint64_t dst_ptr;  // device address
  uint64_t len;
};

constexpr int kScatterThreads = 256;

__global__ __launch_bounds__(kScatterThreads) void tar_scatter_kernel(
    const uint8_t* __restrict__ tar, const CopySeg* __restrict__ segs, uint32_t nsegs) {
  uint32_t seg_idx = blockIdx.x;
  if (seg_idx >= nsegs) return;
  CopySeg seg = segs[seg_idx];
  const uint8_t* src = tar + seg.src_off;
  uint8_t* dst = reinterpret_cast<uint8_t*>(seg.dst_ptr);
  uint64_t len = seg.len;
  // vector main body when both sides share 16-byte phase
  uint64_t i = threadIdx.x;
  if (((reinterpret_cast<uintptr_t>(src) ^ reinterpret_cast<uintptr_t>(dst)) & 15u) == 0) {
    uintptr_t mis = reinterpret_cast<uintptr_t>(src) & 15u;
    uint64_t head = mis ? (16 - mis) : 0;
    if (head > len) head = len;
    for (uint64_t k = threadIdx.x; k < head; k += kScatterThreads) dst[k] = src[k];
    uint64_t body = (len - head) / 16;
    const uint4* vs = reinterpret_cast<const uint4*>(src + head);
    uint4* vd = reinterpret_cast<uint4*>(dst + head);
    for (uint64_t k = threadIdx.x; k < body; k += kScatterThreads) vd[k] = vs[k];
    for (uint64_t k = head + body * 16 + threadIdx.x; k < len; k += kScatterThreads)
      dst[k] = src[k];
  } else {
    for (uint64_t k = i; k < len; k += kScatterThreads) dst[k] = src[k];
  }
}

}  // namespace

extern "C" {

hipError_t modelx_tar_index(const void* tar, uint64_t tar_len, void* entries,
                            uint32_t max_entries, uint32_t* count_dev, uint32_t* error_dev,
                            hipStream_t stream) {
  hipLaunchKernelGGL(tar_index_kernel, dim3(1), dim3(64), 0, stream,
                     static_cast<const uint8_t*>(tar), tar_len,
                     static_cast<TarEntry*>(entries), max_entries, count_dev, error_dev);
  return hipGetLastError();
}

hipError_t modelx_tar_scatter(const void* tar, const void* segs, uint32_t nsegs,
                              hipStream_t stream) {
  if (nsegs == 0) return hipSuccess;
  hipLaunchKernelGGL(tar_scatter_kernel, dim3(nsegs), dim3(kScatterThreads), 0, stream,
                     static_cast<const uint8_t*>(tar), static_cast<const CopySeg*>(segs), nsegs);
  return hipGetLastError();
}

}  // extern "C"
