// CDNA4 (gfx950) SHA-256 chunk-digest kernels.
//
// Replaces the reference's CPU streaming sha256 on the data plane
// (reference: pkg/client/push.go:149-161 digest(), pull.go:115-124 verify).
// SHA-256 is strictly sequential per stream, so single-stream hashing is
// CPU-bound at ~2 GB/s anywhere. The chunked digest (modelx_amd/wire/
// digest.py: leaf_i = SHA256(chunk_i), root over leaves) turns one blob into
// thousands of independent chains; this kernel runs ONE CHAIN PER LANE
// (multi-buffer style), 64 chains per wave64:
//
//  - message blocks enter via 16-byte vector loads (4 × dwordx4 per 64 B
//    block per lane); each lane streams 64 consecutive bytes per iteration,
//    so the 128 B cache lines are fully consumed every two iterations
//  - the 16-word message schedule lives in VGPRs as a ring (W[t&15]),
//    unrolled 64 rounds
//  - the per-lane tail/padding scratch (up to two blocks) is staged in LDS
//    (128 B/lane, padded layout) instead of burning 32 VGPRs per lane, which
//    keeps occupancy at the 4-waves/SIMD the latency hiding wants
//
// Throughput is VALU-bound (~64 rounds × ~10 int-ops per 64 B per lane);
// with >16k chunks in flight the chip hashes at hundreds of GB/s — far above
// the S3/NIC feed rate, which is the point: digest verification stays off
// the pinned-ring pipeline's critical path.
#include <hip/hip_runtime.h>

#include <cstdint>

#define DEV __device__ __forceinline__

namespace {

__constant__ uint32_t K256[64] = {
    0x428a2f98, 0x71374491, 0xb5c0fbcf, 0xe9b5dba5, 0x3956c25b, 0x59f111f1, 0x923f82a4,
    0xab1c5ed5, 0xd807aa98, 0x12835b01, 0x243185be, 0x550c7dc3, 0x72be5d74, 0x80deb1fe,
    0x9bdc06a7, 0xc19bf174, 0xe49b69c1, 0xefbe4786, 0x0fc19dc6, 0x240ca1cc, 0x2de92c6f,
    0x4a7484aa, 0x5cb0a9dc, 0x76f988da, 0x983e5152, 0xa831c66d, 0xb00327c8, 0xbf597fc7,
    0xc6e00bf3, 0xd5a79147, 0x06ca6351, 0x14292967, 0x27b70a85, 0x2e1b2138, 0x4d2c6dfc,
    0x53380d13, 0x650a7354, 0x766a0abb, 0x81c2c92e, 0x92722c85, 0xa2bfe8a1, 0xa81a664b,
    0xc24b8b70, 0xc76c51a3, 0xd192e819, 0xd6990624, 0xf40e3585, 0x106aa070, 0x19a4c116,
    0x1e376c08, 0x2748774c, 0x34b0bcb5, 0x391c0cb3, 0x4ed8aa4a, 0x5b9cca4f, 0x682e6ff3,
    0x748f82ee, 0x78a5636f, 0x84c87814, 0x8cc70208, 0x90befffa, 0xa4506ceb, 0xbef9a3f7,
    0xc67178f2};

DEV uint32_t rotr(uint32_t x, int n) { return __builtin_rotateright32(x, n); }
DEV uint32_t bswap(uint32_t x) { return __builtin_bswap32(x); }

struct Sha256State {
  uint32_t h[8];
  DEV void init() {
    h[0] = 0x6a09e667; h[1] = 0xbb67ae85; h[2] = 0x3c6ef372; h[3] = 0xa54ff53a;
    h[4] = 0x510e527f; h[5] = 0x9b05688c; h[6] = 0x1f83d9ab; h[7] = 0x5be0cd19;
  }
};

// 64 rounds over a 16-word block (big-endian words in W, mutated as a ring).
DEV void sha256_compress(Sha256State& st, uint32_t W[16]) {
  uint32_t a = st.h[0], b = st.h[1], c = st.h[2], d = st.h[3];
  uint32_t e = st.h[4], f = st.h[5], g = st.h[6], hh = st.h[7];
#pragma unroll
  for (int t = 0; t < 64; t++) {
    uint32_t w;
    if (t < 16) {
      w = W[t];
    } else {
      uint32_t w15 = W[(t - 15) & 15], w2 = W[(t - 2) & 15];
      uint32_t s0 = rotr(w15, 7) ^ rotr(w15, 18) ^ (w15 >> 3);
      uint32_t s1 = rotr(w2, 17) ^ rotr(w2, 19) ^ (w2 >> 10);
      w = W[t & 15] + s0 + W[(t - 7) & 15] + s1;
      W[t & 15] = w;
    }
    uint32_t S1 = rotr(e, 6) ^ rotr(e, 11) ^ rotr(e, 25);
    uint32_t ch = (e & f) ^ (~e & g);
    uint32_t t1 = hh + S1 + ch + K256[t] + w;
    uint32_t S0 = rotr(a, 2) ^ rotr(a, 13) ^ rotr(a, 22);
    uint32_t maj = (a & b) ^ (a & c) ^ (b & c);
    uint32_t t2 = S0 + maj;
    hh = g; g = f; f = e; e = d + t1;
    d = c; c = b; b = a; a = t1 + t2;
  }
  st.h[0] += a; st.h[1] += b; st.h[2] += c; st.h[3] += d;
  st.h[4] += e; st.h[5] += f; st.h[6] += g; st.h[7] += hh;
}

// Tail handling: stage the final <64 B + padding + bit-length into an
// LDS-resident two-block scratch (132 B/lane incl. 4 B pad against bank
// camping), then compress 1-2 final blocks from it.
constexpr int kTailDwords = 33;  // 32 data dwords + 1 pad

DEV void sha256_tail(Sha256State& st, const uint8_t* tail_src, uint64_t rem, uint64_t total_len,
                     uint32_t* lds_tail) {
#pragma unroll
  for (int i = 0; i < 32; i++) lds_tail[i] = 0;
  uint8_t* tb = reinterpret_cast<uint8_t*>(lds_tail);
  for (uint64_t i = 0; i < rem; i++) tb[i] = tail_src[i];
  tb[rem] = 0x80;
  const int nblocks = (rem + 1 + 8 <= 64) ? 1 : 2;
  const uint64_t bitlen = total_len << 3;
  uint8_t* lenp = tb + nblocks * 64 - 8;
  for (int i = 0; i < 8; i++) lenp[i] = static_cast<uint8_t>(bitlen >> (56 - 8 * i));
  for (int b = 0; b < nblocks; b++) {
    uint32_t Wr[16];
#pragma unroll
    for (int i = 0; i < 16; i++) Wr[i] = bswap(lds_tail[b * 16 + i]);
    sha256_compress(st, Wr);
  }
}

DEV void store_digest(const Sha256State& st, uint8_t* out32) {
  uint32_t* out = reinterpret_cast<uint32_t*>(out32);
#pragma unroll
  for (int i = 0; i < 8; i++) out[i] = bswap(st.h[i]);
}

constexpr int kBlockThreads = 256;  // 4 wavefronts

// Hash chunk `chunk` of one buffer (one lane) — shared by the single- and
// many-buffer chunk-leaf kernels.
DEV void hash_one_chunk(const uint8_t* __restrict__ data, uint64_t total,
                        uint64_t chunk_size, uint32_t chunk,
                        uint8_t* __restrict__ leaves, uint32_t* lds_tail_slot) {
  const uint64_t begin = static_cast<uint64_t>(chunk) * chunk_size;
  const uint64_t len = (begin + chunk_size <= total) ? chunk_size : (total - begin);
  const uint8_t* p = data + begin;

  Sha256State st;
  st.init();

  const uint64_t nfull = len >> 6;
  if ((reinterpret_cast<uintptr_t>(p) & 15u) == 0) {
    const uint4* v = reinterpret_cast<const uint4*>(p);
    for (uint64_t blk = 0; blk < nfull; blk++) {
      uint32_t W[16];
#pragma unroll
      for (int i = 0; i < 4; i++) {
        uint4 q = v[blk * 4 + i];
        W[i * 4 + 0] = bswap(q.x);
        W[i * 4 + 1] = bswap(q.y);
        W[i * 4 + 2] = bswap(q.z);
        W[i * 4 + 3] = bswap(q.w);
      }
      sha256_compress(st, W);
    }
  } else {
    const uint32_t* p32 = reinterpret_cast<const uint32_t*>(p);  // 4-aligned (chunk offsets)
    for (uint64_t blk = 0; blk < nfull; blk++) {
      uint32_t W[16];
#pragma unroll
      for (int i = 0; i < 16; i++) W[i] = bswap(p32[blk * 16 + i]);
      sha256_compress(st, W);
    }
  }
  sha256_tail(st, p + (nfull << 6), len - (nfull << 6), len, lds_tail_slot);
  store_digest(st, leaves + static_cast<uint64_t>(chunk) * 32);
}

// One chunk per lane. chunk c covers [c*chunk_size, min((c+1)*chunk_size,
// total)); leaves: 32 B per chunk.
__global__ __launch_bounds__(kBlockThreads) void sha256_chunk_leaves_kernel(
    const uint8_t* __restrict__ data, uint64_t total, uint64_t chunk_size,
    uint32_t nchunks, uint8_t* __restrict__ leaves) {
  __shared__ uint32_t lds_tail[kBlockThreads * kTailDwords];
  const uint32_t chunk = blockIdx.x * kBlockThreads + threadIdx.x;
  if (chunk >= nchunks) return;
  hash_one_chunk(data, total, chunk_size, chunk, leaves,
                 &lds_tail[threadIdx.x * kTailDwords]);
}

// Chunk leaves of MANY buffers in ONE launch: a 64 MiB blob alone is 512
// chains (a near-idle chip, ~2 ms per launch × N blobs serialized on one
// stream — the measured config-5 verify wall); the combined grid puts every
// blob's chunks in flight at once. Buffers are located by binary search
// over the chunk-prefix array in the descriptors.
struct ChunkLeavesDesc {
  const uint8_t* data;
  uint8_t* leaves;
  uint64_t total;
  uint64_t chunk_size;
  uint32_t chunk_base;  // chunks in all preceding buffers
  uint32_t pad_;
};

__global__ __launch_bounds__(kBlockThreads) void sha256_chunk_leaves_many_kernel(
    const ChunkLeavesDesc* __restrict__ descs, uint32_t nbuf, uint32_t total_chunks) {
  __shared__ uint32_t lds_tail[kBlockThreads * kTailDwords];
  const uint32_t g = blockIdx.x * kBlockThreads + threadIdx.x;
  if (g >= total_chunks) return;
  uint32_t lo = 0, hi = nbuf - 1;
  while (lo < hi) {
    uint32_t mid = (lo + hi + 1) >> 1;
    if (descs[mid].chunk_base <= g) lo = mid;
    else hi = mid - 1;
  }
  const ChunkLeavesDesc d = descs[lo];
  hash_one_chunk(d.data, d.total, d.chunk_size, g - d.chunk_base, d.leaves,
                 &lds_tail[threadIdx.x * kTailDwords]);
}

// Canonical single-stream SHA-256 of N independent buffers, one per lane:
// batched canonical digests for many-blob pushes (BASELINE config 3 shape).
__global__ __launch_bounds__(kBlockThreads) void sha256_multibuf_kernel(
    const uint8_t* const* __restrict__ buffers, const uint64_t* __restrict__ lengths,
    uint32_t nbuf, uint8_t* __restrict__ digests) {
  __shared__ uint32_t lds_tail[kBlockThreads * kTailDwords];
  const uint32_t idx = blockIdx.x * kBlockThreads + threadIdx.x;
  if (idx >= nbuf) return;
  const uint8_t* p = buffers[idx];
  const uint64_t len = lengths[idx];
  Sha256State st;
  st.init();
  const uint64_t nfull = len >> 6;
  if ((reinterpret_cast<uintptr_t>(p) & 15u) == 0) {
    const uint4* v = reinterpret_cast<const uint4*>(p);
    for (uint64_t blk = 0; blk < nfull; blk++) {
      uint32_t W[16];
#pragma unroll
      for (int i = 0; i < 4; i++) {
        uint4 q = v[blk * 4 + i];
        W[i * 4 + 0] = bswap(q.x);
        W[i * 4 + 1] = bswap(q.y);
        W[i * 4 + 2] = bswap(q.z);
        W[i * 4 + 3] = bswap(q.w);
      }
      sha256_compress(st, W);
    }
  } else {
    for (uint64_t blk = 0; blk < nfull; blk++) {
      uint32_t W[16];
#pragma unroll
      for (int i = 0; i < 16; i++) {
        uint32_t w;
        __builtin_memcpy(&w, p + blk * 64 + i * 4, 4);
        W[i] = bswap(w);
      }
      sha256_compress(st, W);
    }
  }
  sha256_tail(st, p + (nfull << 6), len - (nfull << 6), len,
              &lds_tail[threadIdx.x * kTailDwords]);
  store_digest(st, digests + static_cast<uint64_t>(idx) * 32);
}

}  // namespace

// ------------------------------------------------------------------ C API --

extern "C" {

hipError_t modelx_sha256_chunk_leaves(const void* data, uint64_t total, uint64_t chunk_size,
                                      void* leaves, uint32_t nchunks, hipStream_t stream) {
  if (nchunks == 0) return hipSuccess;
  // NOTE: an in-lane ILP4 variant (4 interleaved chains per lane, meant to
  // hide the round dependency at few-wave grids) was A/B-tested on
  // hardware and LOST badly: 160 GiB/s on 8 GiB (vs 1017 plain) and
  // 1.3 GiB/s on 64 MiB (vs 9) — W[4][16] + 4 states under the unrolled
  // round loop spills to scratch and every round pays a memory round
  // trip. One chain per lane stays (docs/engineering-notes.md #13).
  dim3 grid((nchunks + kBlockThreads - 1) / kBlockThreads);
  hipLaunchKernelGGL(sha256_chunk_leaves_kernel, grid, dim3(kBlockThreads), 0, stream,
                     static_cast<const uint8_t*>(data), total, chunk_size, nchunks,
                     static_cast<uint8_t*>(leaves));
  return hipGetLastError();
}

hipError_t modelx_sha256_chunk_leaves_many(const void* descs_dev, uint32_t nbuf,
                                           uint32_t total_chunks, hipStream_t stream) {
  if (total_chunks == 0 || nbuf == 0) return hipSuccess;
  dim3 grid((total_chunks + kBlockThreads - 1) / kBlockThreads);
  hipLaunchKernelGGL(sha256_chunk_leaves_many_kernel, grid, dim3(kBlockThreads), 0, stream,
                     static_cast<const ChunkLeavesDesc*>(descs_dev), nbuf, total_chunks);
  return hipGetLastError();
}

hipError_t modelx_sha256_multibuf(const void* const* buffers, const uint64_t* lengths,
                                  uint32_t nbuf, void* digests, hipStream_t stream) {
  if (nbuf == 0) return hipSuccess;
  dim3 grid((nbuf + kBlockThreads - 1) / kBlockThreads);
  hipLaunchKernelGGL(sha256_multibuf_kernel, grid, dim3(kBlockThreads), 0, stream,
                     reinterpret_cast<const uint8_t* const*>(buffers), lengths, nbuf,
                     static_cast<uint8_t*>(digests));
  return hipGetLastError();
}

}  // extern "C"
