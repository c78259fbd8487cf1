// CDNA4 chunk-dedup kernels (SURVEY.md §2.2 chunk_verify_dedup.hip):
// an open-addressing hash table of HBM-resident chunks keyed by their
// SHA-256 leaf digest, probed and gathered entirely on-device.
//
// Replaces (reference): HEAD-based whole-blob dedup only
// (pkg/client/push.go:169-177) — the reference has no chunk-level dedup.
//
// insert:  one thread per chunk — linear probe, atomicCAS on the 8-byte
//          key prefix, then the full 32-byte digest + chunk address.
// probe:   one thread per expected chunk — writes the resident source
//          address (or 0) per chunk.
// gather:  one 256-thread workgroup per hit chunk — D2D copy at HBM rate.
//
// Insert and probe/gather phases are serialized by the engine (host mutex +
// stream ordering), so entries are never read while half-written.
#include <hip/hip_runtime.h>

#include <cstdint>

namespace {

typedef uint8_t u8;
typedef uint32_t u32;
typedef uint64_t u64;

struct DedupEntry {
  u64 key;        // first 8 bytes of the digest; 0 = empty
  u64 addr;       // device address of the chunk
  u64 len;        // chunk length (tail chunks differ)
  u64 digest_hi[3];  // remaining 24 digest bytes
};

__device__ inline u64 load_key(const u8* d) {
  u64 v = 0;
  for (int i = 0; i < 8; i++) v |= (u64)d[i] << (8 * i);
  return v;
}

constexpr u32 kMaxProbes = 64;

__global__ void dedup_insert_kernel(const u8* __restrict__ leaves, u32 nchunks, u64 base,
                                    u64 chunk_size, u64 total, DedupEntry* __restrict__ tab,
                                    u64 cap_mask, u32* __restrict__ dropped) {
  u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= nchunks) return;
  const u8* d = leaves + (u64)i * 32;
  u64 key = load_key(d);
  if (key == 0) return;  // sentinel collision: skip (dedup is best-effort)
  u64 off = (u64)i * chunk_size;
  u64 len = total - off < chunk_size ? total - off : chunk_size;
  u64 h = key & cap_mask;
  for (u32 p = 0; p < kMaxProbes; p++) {
    u64 slot = (h + p) & cap_mask;
    u64 prev = atomicCAS(reinterpret_cast<unsigned long long*>(&tab[slot].key), 0ull,
                         (unsigned long long)key);
    if (prev == 0) {
      tab[slot].addr = base + off;
      tab[slot].len = len;
      for (int k = 0; k < 3; k++) tab[slot].digest_hi[k] = load_key(d + 8 + 8 * k);
      __threadfence();
      return;
    }
    if (prev == key) {
      // possible duplicate content — keep the existing entry (first wins)
      bool same = true;
      for (int k = 0; k < 3 && same; k++)
        same = tab[slot].digest_hi[k] == load_key(d + 8 + 8 * k);
      if (same) return;
      // prefix collision with different digest: keep probing
    }
  }
  atomicAdd(dropped, 1);  // table loaded: entry skipped
}

__global__ void dedup_probe_kernel(const u8* __restrict__ leaves, u32 nchunks, u64 chunk_size,
                                   u64 total, const DedupEntry* __restrict__ tab, u64 cap_mask,
                                   u64* __restrict__ src_addr, u64* __restrict__ src_len) {
  u32 i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= nchunks) return;
  const u8* d = leaves + (u64)i * 32;
  u64 key = load_key(d);
  u64 off = (u64)i * chunk_size;
  u64 want = total - off < chunk_size ? total - off : chunk_size;
  src_addr[i] = 0;
  src_len[i] = 0;
  if (key == 0) return;
  u64 h = key & cap_mask;
  for (u32 p = 0; p < kMaxProbes; p++) {
    u64 slot = (h + p) & cap_mask;
    u64 k = tab[slot].key;
    if (k == 0) return;  // empty -> not present
    if (k == key) {
      bool same = tab[slot].len == want;
      for (int q = 0; q < 3 && same; q++) same = tab[slot].digest_hi[q] == load_key(d + 8 + 8 * q);
      if (same) {
        src_addr[i] = tab[slot].addr;
        src_len[i] = tab[slot].len;
        return;
      }
    }
  }
}

constexpr int kGatherThreads = 256;

// one workgroup per chunk; skips misses (src_addr == 0)
__global__ __launch_bounds__(kGatherThreads) void dedup_gather_kernel(
    const u64* __restrict__ src_addr, const u64* __restrict__ src_len, u32 nchunks,
    u64 dst_base, u64 chunk_size) {
  u32 c = blockIdx.x;
  if (c >= nchunks) return;
  u64 sa = src_addr[c];
  if (sa == 0) return;
  u64 len = src_len[c];
  const u8* src = reinterpret_cast<const u8*>(sa);
  u8* dst = reinterpret_cast<u8*>(dst_base + (u64)c * chunk_size);
  // chunk bases are chunk_size-aligned relative to their tensors; use
  // uint4 when the 16-byte phases agree
  if (((reinterpret_cast<uintptr_t>(src) ^ reinterpret_cast<uintptr_t>(dst)) & 15u) == 0) {
    uintptr_t mis = reinterpret_cast<uintptr_t>(src) & 15u;
    u64 head = mis ? (16 - mis) : 0;
    if (head > len) head = len;
    for (u64 k = threadIdx.x; k < head; k += kGatherThreads) dst[k] = src[k];
    u64 body = (len - head) / 16;
    const uint4* vs = reinterpret_cast<const uint4*>(src + head);
    uint4* vd = reinterpret_cast<uint4*>(dst + head);
    for (u64 k = threadIdx.x; k < body; k += kGatherThreads) vd[k] = vs[k];
    for (u64 k = head + body * 16 + threadIdx.x; k < len; k += kGatherThreads)
      dst[k] = src[k];
  } else {
    for (u64 k = threadIdx.x; k < len; k += kGatherThreads) dst[k] = src[k];
  }
}

}  // namespace

extern "C" {

hipError_t modelx_dedup_insert(const void* leaves_dev, uint32_t nchunks, uint64_t base,
                               uint64_t chunk_size, uint64_t total, void* table, uint64_t cap,
                               uint32_t* dropped_dev, hipStream_t stream) {
  if (nchunks == 0) return hipSuccess;
  u32 blocks = (nchunks + 255) / 256;
  hipLaunchKernelGGL(dedup_insert_kernel, dim3(blocks), dim3(256), 0, stream,
                     static_cast<const u8*>(leaves_dev), nchunks, base, chunk_size, total,
                     static_cast<DedupEntry*>(table), cap - 1, dropped_dev);
  return hipGetLastError();
}

hipError_t modelx_dedup_probe(const void* leaves_dev, uint32_t nchunks, uint64_t chunk_size,
                              uint64_t total, const void* table, uint64_t cap,
                              uint64_t* src_addr_dev, uint64_t* src_len_dev,
                              hipStream_t stream) {
  if (nchunks == 0) return hipSuccess;
  u32 blocks = (nchunks + 255) / 256;
  hipLaunchKernelGGL(dedup_probe_kernel, dim3(blocks), dim3(256), 0, stream,
                     static_cast<const u8*>(leaves_dev), nchunks, chunk_size, total,
                     static_cast<const DedupEntry*>(table), cap - 1, src_addr_dev,
                     src_len_dev);
  return hipGetLastError();
}

hipError_t modelx_dedup_gather(const uint64_t* src_addr_dev, const uint64_t* src_len_dev,
                               uint32_t nchunks, uint64_t dst_base, uint64_t chunk_size,
                               hipStream_t stream) {
  if (nchunks == 0) return hipSuccess;
  hipLaunchKernelGGL(dedup_gather_kernel, dim3(nchunks), dim3(kGatherThreads), 0, stream,
                     src_addr_dev, src_len_dev, nchunks, dst_base, chunk_size);
  return hipGetLastError();
}

}  // extern "C"
