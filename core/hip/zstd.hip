// CDNA4 zstd kernels (gfx950): frame-parallel decompress + compress.
//
// The codec itself is the shared header-only core (modelx/zstd_core.hpp,
// zstd_enc.hpp) — the same code the CPU path runs, here executed one frame
// per 64-lane workgroup under the redundant-wavefront model: scalar control
// flow runs identically on all lanes, bulk byte moves fan out lane-strided
// (mx_par_copy / mx_match_copy). All tables (huffman, FSE, build scratch,
// LZ hash) live in LDS so nothing spills 64 private copies to scratch.
//
// Parallelism comes from frame count: the seekable container stores one
// independent frame per `frame_raw` bytes (default 128 KiB), so a 2 GiB
// blob is 16 Ki workgroups — far above the 256-CU fill point.
//
// Replaces (reference): pkg/client/helper.go:19-22 (archiver gzip),
// pull.go:145-204 (directory decompress) — the reference has no parallel or
// GPU decompression at all.
#include <hip/hip_runtime.h>

#include "modelx/zstd_core.hpp"
#include "modelx/zstd_enc.hpp"

using namespace modelx::zstd;

struct MxzFrameC {
  uint64_t c_off, c_size, d_off, d_size;
};

// ----------------------------------------------------------- decompress ----

// 256 threads (4 waves): every wave redundantly runs the scalar decode so
// barrier counts line up, and the lane-strided copies go 4x wider — the
// win on raw-literal frames (our encoder's output), which are copy-bound.
__global__ __launch_bounds__(256) void zstd_decompress_frames_kernel(
    const u8* __restrict__ src, const MxzFrameC* __restrict__ frames, u32 nframes,
    u8* __restrict__ dst, u8* __restrict__ lit_scratch, i64* __restrict__ rc, u32 flags) {
  __shared__ DecCtx ctx;
  u32 f = blockIdx.x;
  if (f >= nframes) return;
  ctx.flags = flags;
  ctx.lit_scratch = lit_scratch + (u64)f * kBlockMax;
  MxzFrameC fr = frames[f];
  i64 n = decode_frame(src + fr.c_off, fr.c_size, dst + fr.d_off, fr.d_size, &ctx, nullptr);
  if (threadIdx.x == 0) rc[f] = (n == (i64)fr.d_size) ? 0 : (n < 0 ? n : MXZ_ERR_CORRUPT);
}

extern "C" hipError_t modelx_zstd_decompress_frames(const void* src, const void* frames_dev,
                                                    uint32_t nframes, void* dst,
                                                    void* lit_scratch, int64_t* rc_dev,
                                                    uint32_t flags, hipStream_t stream) {
  if (nframes == 0) return hipSuccess;
  hipLaunchKernelGGL(zstd_decompress_frames_kernel, dim3(nframes), dim3(256), 0, stream,
                     static_cast<const u8*>(src), static_cast<const MxzFrameC*>(frames_dev),
                     nframes, static_cast<u8*>(dst), static_cast<u8*>(lit_scratch), rc_dev,
                     flags);
  return hipGetLastError();
}

// ------------------------------------------------------------- compress ----

__global__ __launch_bounds__(64) void zstd_compress_frames_kernel(
    const u8* __restrict__ src, u64 srclen, u32 frame_raw, u32 first_frame, u32 nframes,
    u8* __restrict__ dst_scratch, u64 stride, Seq* __restrict__ seq_scratch, u32 max_seqs,
    i64* __restrict__ out_sizes, u32 flags) {
  __shared__ u32 hash[1u << kHashLog];  // 32 KiB
  __shared__ EncTables et;
  et.flags = flags;
  u32 b = blockIdx.x;
  if (b >= nframes) return;
  u32 f = first_frame + b;
  u64 off = (u64)f * frame_raw;
  if (off >= srclen && !(srclen == 0 && f == 0)) {
    if (threadIdx.x == 0) out_sizes[b] = 0;
    return;
  }
  u64 flen = srclen - off < frame_raw ? srclen - off : frame_raw;
  int rc = enc_tables_init(&et);
  i64 n = rc < 0 ? rc
                 : encode_frame(src + off, flen, dst_scratch + (u64)b * stride, stride, hash,
                                seq_scratch + (u64)b * max_seqs, &et);
  if (threadIdx.x == 0) out_sizes[b] = n;
}

extern "C" hipError_t modelx_zstd_compress_frames(const void* src, uint64_t srclen,
                                                  uint32_t frame_raw, uint32_t first_frame,
                                                  uint32_t nframes, void* dst_scratch,
                                                  uint64_t stride, void* seq_scratch,
                                                  uint32_t max_seqs, int64_t* out_sizes_dev,
                                                  uint32_t flags, hipStream_t stream) {
  if (nframes == 0) return hipSuccess;
  hipLaunchKernelGGL(zstd_compress_frames_kernel, dim3(nframes), dim3(64), 0, stream,
                     static_cast<const u8*>(src), srclen, frame_raw, first_frame, nframes,
                     static_cast<u8*>(dst_scratch), stride, static_cast<Seq*>(seq_scratch),
                     max_seqs, out_sizes_dev, flags);
  return hipGetLastError();
}
