// zstd codec core — shared between the CDNA4 HIP kernels (core/hip/zstd.hip,
// one workgroup per frame) and the CPU fallback/test path (core/src/zstd_cpu.cpp).
//
// MI355X-native replacement for the reference's gzip path
// (reference: pkg/client/helper.go:19-22 archiver.Gz, pull.go:145-204): gzip is
// a single sequential stream, so the GPU-native blob format is a sequence of
// INDEPENDENT standard zstd frames (RFC 8878) + a zstd-seekable-format seek
// table in a trailing skippable frame. Every frame decodes in its own
// workgroup; `zstd -d` on any machine still reads the same bytes.
//
// Decoder: full RFC 8878 single-dictionary-less frames — raw/RLE/compressed
// blocks, raw/RLE/Huffman(1&4-stream)/treeless literals, predefined/RLE/
// FSE-compressed/repeat sequence tables, repeat offsets. Interop-tested
// against libzstd output (tests/test_zstd.py).
// Encoder: standard frames with greedy LZ77 matches, raw literals and
// predefined FSE sequence tables — always decodable by stock zstd.
//
// Device execution model: the whole per-frame codec runs redundantly on all
// 64 lanes of one wavefront (identical control flow, zero divergence); only
// the bulk byte moves fan out across lanes (mx_par_copy / mx_match_copy).
#pragma once

#include <stddef.h>
#include <stdint.h>

#if defined(__HIPCC__) || defined(__HIP__)
#include <hip/hip_runtime.h>
#define MX_HD __host__ __device__
#else
#define MX_HD
#endif

namespace modelx {
namespace zstd {

typedef uint8_t u8;
typedef uint16_t u16;
typedef uint32_t u32;
typedef uint64_t u64;
typedef int16_t i16;
typedef int32_t i32;
typedef int64_t i64;

static const u32 kMagic = 0xFD2FB528u;
static const u32 kMagicSkippableSeek = 0x184D2A5Eu;  // seekable-format skippable frame
static const u32 kSeekTableMagic = 0x8F92EAB1u;
static const u32 kBlockMax = 128 * 1024;

// error codes (returned negative from decode/encode entry points)
enum {
  MXZ_OK = 0,
  MXZ_ERR_MAGIC = -1,
  MXZ_ERR_HEADER = -2,
  MXZ_ERR_BLOCK = -3,
  MXZ_ERR_LITERALS = -4,
  MXZ_ERR_HUFFMAN = -5,
  MXZ_ERR_FSE = -6,
  MXZ_ERR_SEQUENCES = -7,
  MXZ_ERR_OFFSET = -8,
  MXZ_ERR_DST_SMALL = -9,
  MXZ_ERR_SRC_TRUNC = -10,
  MXZ_ERR_CORRUPT = -11,
  MXZ_ERR_UNSUPPORTED = -12,
};

// ---------------------------------------------------------------- helpers --

MX_HD static inline u32 mx_highbit(u32 v) {
  // floor(log2(v)), v != 0
#if defined(__HIP_DEVICE_COMPILE__)
  return 31 - __clz((int)v);
#elif defined(__GNUC__)
  return 31 - __builtin_clz(v);
#else
  u32 r = 0;
  while (v >>= 1) r++;
  return r;
#endif
}

MX_HD static inline u32 mx_read_le32(const u8* p) {
  return (u32)p[0] | ((u32)p[1] << 8) | ((u32)p[2] << 16) | ((u32)p[3] << 24);
}
MX_HD static inline u64 mx_read_le64(const u8* p) {
  return (u64)mx_read_le32(p) | ((u64)mx_read_le32(p + 4) << 32);
}
MX_HD static inline void mx_write_le32(u8* p, u32 v) {
  p[0] = (u8)v;
  p[1] = (u8)(v >> 8);
  p[2] = (u8)(v >> 16);
  p[3] = (u8)(v >> 24);
}

// lane id / width under the redundant-wavefront model
MX_HD static inline u32 mx_lane() {
#if defined(__HIP_DEVICE_COMPILE__)
  return threadIdx.x;
#else
  return 0;
#endif
}
MX_HD static inline u32 mx_width() {
#if defined(__HIP_DEVICE_COMPILE__)
  return blockDim.x;
#else
  return 1;
#endif
}
MX_HD static inline void mx_sync() {
#if defined(__HIP_DEVICE_COMPILE__)
  __syncthreads();
#endif
}

// parallel byte copy (non-overlapping)
MX_HD static inline void mx_par_copy(u8* dst, const u8* src, u64 n) {
  u32 lane = mx_lane(), w = mx_width();
  // wide path when both pointers share alignment
  if (n >= 64 && (((uintptr_t)dst ^ (uintptr_t)src) & 7) == 0) {
    u64 head = (8 - ((uintptr_t)dst & 7)) & 7;
    for (u64 i = lane; i < head; i += w) dst[i] = src[i];
    u64 body = (n - head) / 8;
    const u64* s8 = (const u64*)(src + head);
    u64* d8 = (u64*)(dst + head);
    for (u64 i = lane; i < body; i += w) d8[i] = s8[i];
    for (u64 i = head + body * 8 + lane; i < n; i += w) dst[i] = src[i];
  } else {
    for (u64 i = lane; i < n; i += w) dst[i] = src[i];
  }
  mx_sync();
}

// match copy: dst[i] = dst[i - offset] with RLE-extension semantics
// (offset may be < n). Parallel-safe via modulo into the pre-existing window.
MX_HD static inline void mx_match_copy(u8* dst, u64 offset, u64 n) {
  u32 lane = mx_lane(), w = mx_width();
  const u8* src = dst - offset;
  if (offset >= n) {
    for (u64 i = lane; i < n; i += w) dst[i] = src[i];
  } else {
    for (u64 i = lane; i < n; i += w) dst[i] = src[i % offset];
  }
  mx_sync();
}

MX_HD static inline void mx_par_set(u8* dst, u8 v, u64 n) {
  u32 lane = mx_lane(), w = mx_width();
  for (u64 i = lane; i < n; i += w) dst[i] = v;
  mx_sync();
}

// ------------------------------------------------------------ bit streams --

// Forward LSB-first bit reader (FSE table headers, huffman weight headers).
struct FwdBits {
  const u8* src;
  u64 len;       // bytes
  u64 bitpos;    // consumed bits

  MX_HD void init(const u8* s, u64 l) {
    src = s;
    len = l;
    bitpos = 0;
  }
  MX_HD u32 peek(u32 nbits) {
    u64 byte = bitpos >> 3;
    u32 shift = (u32)(bitpos & 7);
    u64 v = 0;
    for (u32 i = 0; i < 8 && byte + i < len; i++) v |= (u64)src[byte + i] << (8 * i);
    return (u32)((v >> shift) & ((nbits < 32 ? (1u << nbits) : 0) - 1u));
  }
  MX_HD u32 read(u32 nbits) {
    u32 v = peek(nbits);
    bitpos += nbits;
    return v;
  }
};

// Backward bit reader (huffman streams, FSE weight stream, sequences).
// The stream was written forward; reading starts at the LAST byte whose
// highest set bit is the padding marker. offset_bits counts readable bits
// remaining; reads past the start return the remaining bits left-shifted
// (zero-padded low bits) — matching zstd's overflow semantics for final
// state updates.
struct BackBits {
  const u8* src;   // first byte of the stream
  i64 bits;        // readable bits remaining (excluding marker)

  // returns false on empty/invalid stream (no marker)
  MX_HD bool init(const u8* s, u64 len) {
    src = s;
    if (len == 0 || s[len - 1] == 0) {
      bits = 0;
      return false;
    }
    bits = (i64)(len * 8) - 1 - (i64)(7 - mx_highbit(s[len - 1]));
    return true;
  }
  MX_HD u32 read(u32 nbits) {
    if (nbits == 0) return 0;
    i64 newbits = bits - (i64)nbits;
    i64 lo = newbits < 0 ? 0 : newbits;
    // gather bits [lo, bits) MSB-first relative to stream end
    u64 byte0 = (u64)(lo >> 3);
    u32 avail = (u32)(bits - lo);
    u64 v = 0;
    for (u32 i = 0; i < 9 && byte0 + i <= (u64)((bits - 1) >> 3); i++)
      v |= (u64)src[byte0 + i] << (8 * i);
    v >>= (lo & 7);
    u32 out = (u32)(v & ((avail < 32 ? ((u64)1 << avail) : 0x100000000ull) - 1));
    if (newbits < 0) out <<= (u32)(-newbits);  // zero-pad low side on overflow
    bits = newbits;
    return out;
  }
  MX_HD bool overflowed() const { return bits < 0; }
  MX_HD bool finished() const { return bits == 0; }
};

// ------------------------------------------------------------------ FSE ----

static const u32 kMaxFseLog = 9;          // LL/ML max 9, OF max 8, weights 6
static const u32 kMaxFseSize = 1 << kMaxFseLog;

struct FseTable {
  u8 symbol[kMaxFseSize];
  u8 nbits[kMaxFseSize];
  u16 base[kMaxFseSize];  // newState base
  u32 log;                // accuracy log (table size = 1<<log)
};

// Table-construction temporaries. On the GPU these live in LDS (inside
// DecCtx) so the redundant-wavefront execution doesn't spill 64 private
// copies to scratch memory; on the CPU they're just part of the context.
// INVARIANT (multi-wave workgroups): every write to these shared tables
// must be a deterministic same-value write — all threads compute the same
// value for the same cell — because the 4 waves of a 256-thread decompress
// workgroup are NOT in lockstep between barriers. Read-modify-write state
// (occurrence counters, repeat offsets) therefore lives in per-thread
// locals, never in this struct.
struct BuildScratch {
  u16 spread[kMaxFseSize];  // cell -> symbol during spread
  u8 weights[256];          // huffman weights
  i16 counts[256];          // normalized counts from fse_read_ncount
  FseTable wtab;            // huffman-weight FSE table
};

// Build a decoding table from normalized counts (-1 == "less than 1").
MX_HD static inline int fse_build_dtable(FseTable* t, const i16* counts, u32 nsym, u32 log,
                                         BuildScratch* bs) {
  // Barrier before overwriting shared scratch/tables: __syncthreads only
  // bounds wave drift to ONE barrier interval, and a whole block's table
  // builds share an interval — without this, a leading wave overwrites
  // bs->spread / the target table while a trailing wave still reads the
  // previous section's content (observed ~1/50k frames as rc=-7 or wrong
  // output under multi-wave decode).
  mx_sync();
  if (log > kMaxFseLog) return MXZ_ERR_FSE;
  u32 size = 1u << log;
  u32 high = size - 1;
  u16* pos_syms = bs->spread;
  // low-prob (-1) symbols get one cell from the top, in symbol order
  for (u32 s = 0; s < nsym; s++)
    if (counts[s] == -1) pos_syms[high--] = (u16)s;
  u32 step = (size >> 1) + (size >> 3) + 3;
  u32 mask = size - 1;
  u32 pos = 0;
  for (u32 s = 0; s < nsym; s++) {
    if (counts[s] <= 0) continue;
    for (i32 i = 0; i < counts[s]; i++) {
      pos_syms[pos] = (u16)s;
      do {
        pos = (pos + step) & mask;
      } while (pos > high);
    }
  }
  if (pos != 0) return MXZ_ERR_FSE;
  // per-symbol occurrence counters start at the normalized count.
  // Thread-LOCAL: counter[s]++ is a read-modify-write and must not be
  // shared across the workgroup's waves (see BuildScratch invariant).
  u16 counter[256];
  for (u32 s = 0; s < nsym; s++) counter[s] = (u16)(counts[s] == -1 ? 1 : (counts[s] < 0 ? 0 : counts[s]));
  for (u32 c = 0; c < size; c++) {
    u32 s = pos_syms[c];
    u16 x = counter[s]++;
    u32 nb = log - mx_highbit(x);
    t->symbol[c] = (u8)s;
    t->nbits[c] = (u8)nb;
    t->base[c] = (u16)((x << nb) - size);
  }
  t->log = log;
  return MXZ_OK;
}

// Read an FSE table description (forward bitstream) per RFC 8878 §4.1.1.
// Returns bytes consumed (>=0) or error (<0). max_log bounds accuracy.
MX_HD static inline int fse_read_ncount(i16* counts, u32* nsym_out, u32* log_out, u32 max_sym,
                                        u32 max_log, const u8* src, u64 srclen) {
  if (srclen < 1) return MXZ_ERR_FSE;
  FwdBits bits;
  bits.init(src, srclen);
  u32 log = bits.read(4) + 5;
  if (log > max_log) return MXZ_ERR_FSE;
  i32 remaining = (i32)(1u << log) + 1;
  u32 threshold = 1u << log;
  u32 nbits = log + 1;
  u32 sym = 0;
  bool prev0 = false;
  for (u32 s = 0; s <= max_sym; s++) counts[s] = 0;
  while (remaining > 1 && sym <= max_sym) {
    if (prev0) {
      // runs of zero-prob symbols: 2-bit repeat codes, value 3 = continue
      // (fast path: 8 consecutive 3s = 24 zeros in 16 bits)
      while (bits.peek(16) == 0xFFFF) {
        bits.read(16);
        sym += 24;
        if (sym > max_sym + 1) return MXZ_ERR_FSE;
      }
      while (bits.peek(2) == 3) {
        bits.read(2);
        sym += 3;
        if (sym > max_sym + 1) return MXZ_ERR_FSE;
      }
      sym += bits.read(2);
      if (sym > max_sym) break;
      prev0 = false;
    }
    u32 max = (2 * threshold - 1) - (u32)remaining;
    i32 count;
    // threshold coding: values < max fit in nbits-1 bits
    {
      u32 v = bits.peek(nbits);
      u32 small = v & (threshold - 1);
      if (small < max) {
        count = (i32)small;
        bits.bitpos += nbits - 1;
      } else {
        u32 full = v & (2 * threshold - 1);
        if (full >= threshold) full -= max;
        count = (i32)full;
        bits.bitpos += nbits;
      }
    }
    count--;  // -1 means "less than 1"
    remaining -= count < 0 ? -count : count;
    counts[sym++] = (i16)count;
    prev0 = (count == 0);
    while ((u32)remaining < threshold) {
      nbits--;
      threshold >>= 1;
    }
    if ((u64)(bits.bitpos >> 3) > srclen) return MXZ_ERR_FSE;
  }
  if (remaining != 1) return MXZ_ERR_FSE;
  *nsym_out = sym;
  *log_out = log;
  return (int)((bits.bitpos + 7) >> 3);
}

// -------------------------------------------------------------- Huffman ----

static const u32 kMaxHufBits = 11;

struct HufTable {
  u8 symbol[1 << kMaxHufBits];
  u8 nbits[1 << kMaxHufBits];
  u32 maxbits;
  bool valid;
};

// Build the decode table from weights[0..nsym-1] (last weight already derived).
MX_HD static inline int huf_build(HufTable* t, const u8* weights, u32 nsym, u32 maxbits) {
  if (maxbits > kMaxHufBits) return MXZ_ERR_HUFFMAN;
  u32 size = 1u << maxbits;
  u32 pos = 0;
  for (u32 w = 1; w <= maxbits; w++) {
    u32 run = 1u << (w - 1);
    for (u32 s = 0; s < nsym; s++) {
      if (weights[s] != w) continue;
      u32 nb = maxbits + 1 - w;
      if (pos + run > size) return MXZ_ERR_HUFFMAN;
      for (u32 i = 0; i < run; i++) {
        t->symbol[pos + i] = (u8)s;
        t->nbits[pos + i] = (u8)nb;
      }
      pos += run;
    }
  }
  if (pos != size) return MXZ_ERR_HUFFMAN;
  t->maxbits = maxbits;
  t->valid = true;
  return MXZ_OK;
}

// Parse a Huffman_Tree_Description; returns bytes consumed or <0.
MX_HD static inline int huf_read_table(HufTable* t, const u8* src, u64 srclen,
                                       BuildScratch* bs) {
  mx_sync();  // see fse_build_dtable: shared-scratch overwrite boundary
  if (srclen < 1) return MXZ_ERR_HUFFMAN;
  u8* weights = bs->weights;
  u32 nsym = 0;
  u64 consumed;
  u8 hb = src[0];
  if (hb >= 128) {
    // direct 4-bit weights
    nsym = hb - 127;
    u64 wbytes = (nsym + 1) / 2;
    if (1 + wbytes > srclen) return MXZ_ERR_HUFFMAN;
    for (u32 i = 0; i < nsym; i++) {
      u8 b = src[1 + i / 2];
      weights[i] = (i & 1) ? (b & 0xF) : (b >> 4);
    }
    consumed = 1 + wbytes;
  } else {
    // FSE-compressed weights, compressed size = hb
    if (1 + (u64)hb > srclen) return MXZ_ERR_HUFFMAN;
    u32 fns, flog;
    int hdr = fse_read_ncount(bs->counts, &fns, &flog, 255, 6, src + 1, hb);
    if (hdr < 0) return hdr;
    FseTable& ft = bs->wtab;
    int rc = fse_build_dtable(&ft, bs->counts, fns, flog, bs);
    if (rc < 0) return rc;
    BackBits bb;
    if (!bb.init(src + 1 + hdr, hb - hdr)) return MXZ_ERR_HUFFMAN;
    u32 st1 = bb.read(ft.log), st2 = bb.read(ft.log);
    if (bb.overflowed()) return MXZ_ERR_HUFFMAN;
    // two interleaved states; on overflow emit the other state's symbol
    while (true) {
      if (nsym >= 255) return MXZ_ERR_HUFFMAN;
      weights[nsym++] = ft.symbol[st1];
      st1 = ft.base[st1] + bb.read(ft.nbits[st1]);
      if (bb.overflowed()) {
        if (nsym >= 255) return MXZ_ERR_HUFFMAN;
        weights[nsym++] = ft.symbol[st2];
        break;
      }
      if (nsym >= 255) return MXZ_ERR_HUFFMAN;
      weights[nsym++] = ft.symbol[st2];
      st2 = ft.base[st2] + bb.read(ft.nbits[st2]);
      if (bb.overflowed()) {
        if (nsym >= 255) return MXZ_ERR_HUFFMAN;
        weights[nsym++] = ft.symbol[st1];
        break;
      }
    }
    consumed = 1 + (u64)hb;
  }
  // derive the implicit last weight
  u64 sum = 0;
  for (u32 i = 0; i < nsym; i++) {
    if (weights[i] > kMaxHufBits) return MXZ_ERR_HUFFMAN;
    if (weights[i]) sum += (u64)1 << (weights[i] - 1);
  }
  if (sum == 0) return MXZ_ERR_HUFFMAN;
  u32 maxbits = mx_highbit((u32)sum) + 1;
  u64 left = ((u64)1 << maxbits) - sum;
  if (left == 0 || (left & (left - 1)) != 0) return MXZ_ERR_HUFFMAN;
  weights[nsym++] = (u8)(mx_highbit((u32)left) + 1);
  int rc = huf_build(t, weights, nsym, maxbits);
  if (rc < 0) return rc;
  return (int)consumed;
}

// Decode one backward Huffman stream into dst (exactly dstlen symbols).
MX_HD static inline int huf_decode_stream(const HufTable* t, const u8* src, u64 srclen, u8* dst,
                                          u64 dstlen) {
  BackBits bb;
  if (!bb.init(src, srclen)) return dstlen == 0 ? MXZ_OK : MXZ_ERR_HUFFMAN;
  u64 produced = 0;
  u32 mb = t->maxbits;
  i64 bits = bb.bits;
  const u8* sp = bb.src;
  const u32 mask = (1u << mb) - 1;
  // fast path: one unaligned 64-bit window load serves many symbols — the
  // per-symbol 9-byte bounded gather below was the measured serial tail of
  // literal-heavy (bf16 tensor) frames. The window is anchored at bit `lo`
  // (LSB), holding bits [lo, lo+64); symbols read the top `mb` bits below
  // the frontier. Requires 8 loadable bytes at byte(lo) — the last few
  // symbols fall through to the bounded loop.
  while (produced < dstlen) {
    i64 lo = bits - 57;
    if (lo < 0) break;  // near the stream start: bounded path
    u64 byte0 = (u64)(lo >> 3);
    if (byte0 + 8 > srclen) {
      // shift the window down so the 8-byte load stays inside the stream
      byte0 = srclen - 8;  // srclen >= 8 guaranteed: bits >= 57 here
      lo = (i64)(byte0 << 3);
    }
    u64 v;
    __builtin_memcpy(&v, sp + byte0, 8);
    v >>= (lo & 7);
    u32 have = (u32)(bits - lo);  // <= 57 + 7
    if (have > 57) {              // keep idx math in-range after the shift-down
      v >>= (have - 57);
      lo += have - 57;
      have = 57;
    }
    while (have >= mb && produced < dstlen) {
      u32 idx = (u32)(v >> (have - mb)) & mask;
      u32 nb = t->nbits[idx];
      dst[produced++] = t->symbol[idx];
      bits -= nb;
      have -= nb;
    }
    if (have >= mb) break;  // dstlen reached
  }
  // bounded tail (original semantics, incl. zero-padded final reads)
  while (produced < dstlen) {
    i64 lo = bits - (i64)mb;
    u32 idx;
    {
      i64 l = lo < 0 ? 0 : lo;
      u64 byte0 = (u64)(l >> 3);
      u64 v = 0;
      for (u32 i = 0; i < 9 && byte0 + i <= (u64)((bits - 1) >> 3); i++)
        v |= (u64)sp[byte0 + i] << (8 * i);
      v >>= (l & 7);
      u32 avail = (u32)(bits - l);
      idx = (u32)(v & (((u64)1 << avail) - 1));
      if (lo < 0) idx <<= (u32)(-lo);
    }
    u32 nb = t->nbits[idx];
    if ((i64)nb > bits) return MXZ_ERR_HUFFMAN;  // ran out of bits
    dst[produced++] = t->symbol[idx];
    bits -= nb;
  }
  return MXZ_OK;
}

// --------------------------------------------- sequence code tables --------

// literals-length codes (RFC 8878 table)
MX_HD static inline void ll_code_info(u32 code, u32* nbits, u32* base) {
  static const u8 kBits[36] = {0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
                               1, 1, 1, 1, 2, 2, 3, 3, 4, 6, 7, 8, 9, 10, 11, 12,
                               13, 14, 15, 16};
  static const u32 kBase[36] = {0,  1,  2,  3,  4,  5,  6,  7,  8,   9,   10,  11,
                                12, 13, 14, 15, 16, 18, 20, 22, 24,  28,  32,  40,
                                48, 64, 128, 256, 512, 1024, 2048, 4096, 8192, 16384,
                                32768, 65536};
  *nbits = kBits[code];
  *base = kBase[code];
}

MX_HD static inline void ml_code_info(u32 code, u32* nbits, u32* base) {
  static const u8 kBits[53] = {0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
                               0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 1, 1, 1, 1,
                               2, 2, 3, 3, 4, 4, 5, 7, 8, 9, 10, 11, 12, 13, 14, 15, 16};
  static const u32 kBase[53] = {3,  4,  5,  6,  7,  8,  9,  10, 11, 12, 13, 14, 15, 16,
                                17, 18, 19, 20, 21, 22, 23, 24, 25, 26, 27, 28, 29, 30,
                                31, 32, 33, 34, 35, 37, 39, 41, 43, 47, 51, 59, 67, 83,
                                99, 131, 259, 515, 1027, 2051, 4099, 8195, 16387, 32771,
                                65539};
  *nbits = kBits[code];
  *base = kBase[code];
}

// predefined distributions (RFC 8878 §3.1.1.3.2.2)
MX_HD static inline const i16* ll_default_dist(u32* nsym, u32* log) {
  static const i16 d[36] = {4, 3, 2, 2, 2, 2, 2, 2, 2, 2, 2, 2, 2, 1, 1, 1, 2, 2,
                            2, 2, 2, 2, 2, 2, 2, 3, 2, 1, 1, 1, 1, 1, -1, -1, -1, -1};
  *nsym = 36;
  *log = 6;
  return d;
}
MX_HD static inline const i16* ml_default_dist(u32* nsym, u32* log) {
  static const i16 d[53] = {1, 4, 3, 2, 2, 2, 2, 2, 2, 1, 1, 1, 1, 1, 1, 1, 1, 1,
                            1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1,
                            1, 1, 1, 1, 1, 1, 1, 1, 1, 1, -1, -1, -1, -1, -1, -1, -1};
  *nsym = 53;
  *log = 6;
  return d;
}
MX_HD static inline const i16* of_default_dist(u32* nsym, u32* log) {
  static const i16 d[29] = {1, 1, 1, 1, 1, 1, 2, 2, 2, 1, 1, 1, 1, 1, 1,
                            1, 1, 1, 1, 1, 1, 1, 1, 1, -1, -1, -1, -1, -1};
  *nsym = 29;
  *log = 5;
  return d;
}

// -------------------------------------------------------- decode context ---

struct DecCtx {
  u32 flags;            // bit1: disable wave-split literal-stream decode
  HufTable huf;         // persists across blocks (treeless literals)
  FseTable ll, of, ml;  // persist across blocks (repeat mode)
  BuildScratch bs;      // table-construction temporaries
  bool ll_valid, of_valid, ml_valid;
  int stream_rc;        // per-wave huffman stream results (multi-wave decode)
  u8* lit_scratch;      // >= kBlockMax bytes
};

MX_HD static inline void dec_ctx_init(DecCtx* c, u8* lit_scratch) {
  c->huf.valid = false;
  c->ll_valid = c->of_valid = c->ml_valid = false;
  c->stream_rc = MXZ_OK;
  c->lit_scratch = lit_scratch;
}

// Read one sequence-table description per the 2-bit mode. Returns bytes
// consumed or <0.  mode: 0 predefined, 1 RLE, 2 FSE, 3 repeat.
MX_HD static inline int seq_table_load(FseTable* t, bool* valid, u32 mode, u32 kind,
                                       const u8* src, u64 srclen, BuildScratch* bs) {
  mx_sync();  // see fse_build_dtable: shared-scratch overwrite boundary
  // kind: 0=LL 1=OF 2=ML
  if (mode == 0) {
    u32 nsym, log;
    const i16* d = kind == 0 ? ll_default_dist(&nsym, &log)
                             : (kind == 1 ? of_default_dist(&nsym, &log)
                                          : ml_default_dist(&nsym, &log));
    int rc = fse_build_dtable(t, d, nsym, log, bs);
    if (rc < 0) return rc;
    *valid = true;
    return 0;
  }
  if (mode == 1) {  // RLE: single symbol, 1 byte
    if (srclen < 1) return MXZ_ERR_SEQUENCES;
    u8 sym = src[0];
    u32 maxs = kind == 0 ? 35 : (kind == 1 ? 31 : 52);
    if (sym > maxs) return MXZ_ERR_SEQUENCES;
    t->log = 0;
    t->symbol[0] = sym;
    t->nbits[0] = 0;
    t->base[0] = 0;
    *valid = true;
    return 1;
  }
  if (mode == 2) {
    u32 nsym, log;
    u32 maxs = kind == 0 ? 35 : (kind == 1 ? 31 : 52);
    u32 maxlog = kind == 0 ? 9 : (kind == 1 ? 8 : 9);
    int hdr = fse_read_ncount(bs->counts, &nsym, &log, maxs, maxlog, src, srclen);
    if (hdr < 0) return hdr;
    int rc = fse_build_dtable(t, bs->counts, nsym, log, bs);
    if (rc < 0) return rc;
    *valid = true;
    return hdr;
  }
  // repeat
  if (!*valid) return MXZ_ERR_SEQUENCES;
  return 0;
}

// Decode one compressed block into dst (history = bytes already written to
// the frame buffer before dst). Returns regenerated size or <0.
MX_HD static inline i64 decode_block(DecCtx* c, u32* rep, const u8* src, u64 srclen, u8* dst,
                                     u64 dstcap, u64 history) {
  // block boundary: the previous block's sequence loop read lit_scratch and
  // the FSE/huffman tables; make every wave finish those reads before this
  // block overwrites them (see fse_build_dtable note)
  mx_sync();
  // ---- literals section ----
  if (srclen < 1) return MXZ_ERR_LITERALS;
  u32 b0 = src[0];
  u32 lit_type = b0 & 3;
  u32 size_fmt = (b0 >> 2) & 3;
  u64 lit_regen = 0, lit_comp = 0, lit_hdr = 0;
  u32 nstreams = 1;
  if (lit_type == 0 || lit_type == 1) {  // Raw / RLE
    if (size_fmt == 0 || size_fmt == 2) {
      lit_regen = b0 >> 3;
      lit_hdr = 1;
    } else if (size_fmt == 1) {
      if (srclen < 2) return MXZ_ERR_LITERALS;
      lit_regen = (b0 >> 4) | ((u64)src[1] << 4);
      lit_hdr = 2;
    } else {
      if (srclen < 3) return MXZ_ERR_LITERALS;
      lit_regen = (b0 >> 4) | ((u64)src[1] << 4) | ((u64)src[2] << 12);
      lit_hdr = 3;
    }
  } else {  // Compressed / Treeless
    u64 h;
    if (size_fmt == 0 || size_fmt == 1) {
      if (srclen < 3) return MXZ_ERR_LITERALS;
      h = (u64)b0 | ((u64)src[1] << 8) | ((u64)src[2] << 16);
      lit_regen = (h >> 4) & 0x3FF;
      lit_comp = (h >> 14) & 0x3FF;
      lit_hdr = 3;
      nstreams = size_fmt == 0 ? 1 : 4;
    } else if (size_fmt == 2) {
      if (srclen < 4) return MXZ_ERR_LITERALS;
      h = (u64)b0 | ((u64)src[1] << 8) | ((u64)src[2] << 16) | ((u64)src[3] << 24);
      lit_regen = (h >> 4) & 0x3FFF;
      lit_comp = (h >> 18) & 0x3FFF;
      lit_hdr = 4;
      nstreams = 4;
    } else {
      if (srclen < 5) return MXZ_ERR_LITERALS;
      h = (u64)b0 | ((u64)src[1] << 8) | ((u64)src[2] << 16) | ((u64)src[3] << 24) |
          ((u64)src[4] << 32);
      lit_regen = (h >> 4) & 0x3FFFF;
      lit_comp = (h >> 22) & 0x3FFFF;
      lit_hdr = 5;
      nstreams = 4;
    }
  }
  if (lit_regen > kBlockMax) return MXZ_ERR_LITERALS;
  const u8* lits = nullptr;  // where literals live for sequence execution
  u8 rle_byte = 0;
  bool lits_rle = false;
  u64 pos = lit_hdr;
  if (lit_type == 0) {  // raw
    if (pos + lit_regen > srclen) return MXZ_ERR_LITERALS;
    lits = src + pos;
    pos += lit_regen;
  } else if (lit_type == 1) {  // RLE
    if (pos + 1 > srclen) return MXZ_ERR_LITERALS;
    rle_byte = src[pos];
    lits_rle = true;
    pos += 1;
  } else {
    if (pos + lit_comp > srclen) return MXZ_ERR_LITERALS;
    const u8* cl = src + pos;
    u64 cl_len = lit_comp;
    if (lit_type == 2) {  // fresh huffman table
      int consumed = huf_read_table(&c->huf, cl, cl_len, &c->bs);
      if (consumed < 0) return consumed;
      cl += consumed;
      cl_len -= consumed;
    } else if (!c->huf.valid) {
      return MXZ_ERR_HUFFMAN;  // treeless with no previous table
    }
    if (nstreams == 1) {
      int rc = huf_decode_stream(&c->huf, cl, cl_len, c->lit_scratch, lit_regen);
      if (rc < 0) return rc;
    } else {
      if (cl_len < 6) return MXZ_ERR_LITERALS;
      u64 s1 = (u64)cl[0] | ((u64)cl[1] << 8);
      u64 s2 = (u64)cl[2] | ((u64)cl[3] << 8);
      u64 s3 = (u64)cl[4] | ((u64)cl[5] << 8);
      if (6 + s1 + s2 + s3 > cl_len) return MXZ_ERR_LITERALS;
      u64 s4 = cl_len - 6 - s1 - s2 - s3;
      u64 r123 = (lit_regen + 3) / 4;
      if (r123 * 3 > lit_regen) return MXZ_ERR_LITERALS;
      u64 r4 = lit_regen - 3 * r123;
      const u8* sp = cl + 6;
      const u8* srcs[4] = {sp, sp + s1, sp + s1 + s2, sp + s1 + s2 + s3};
      u64 slens[4] = {s1, s2, s3, s4};
      u64 rlens[4] = {r123, r123, r123, r4};
      int rc = MXZ_OK;
#if defined(__HIP_DEVICE_COMPILE__)
      if (c->flags & 2u) {
        for (u32 k = 0; k < 4 && rc == MXZ_OK; k++)
          rc = huf_decode_stream(&c->huf, srcs[k], slens[k],
                                 c->lit_scratch + (u64)k * r123, rlens[k]);
      } else {
        // one wave per stream: the four backward streams are independent,
        // and huffman decode is the serial tail of literal-heavy frames.
        // Divergent across waves (no barriers inside), re-converged below.
        u32 wave = mx_lane() / 64;
        u32 nwaves = (mx_width() + 63) / 64;
        int my_rc = MXZ_OK;
        for (u32 k = wave; k < 4; k += nwaves)
          if (my_rc == MXZ_OK)
            my_rc = huf_decode_stream(&c->huf, srcs[k], slens[k],
                                      c->lit_scratch + (u64)k * r123, rlens[k]);
        // publish failures (same-value or benign-differing error codes)
        if (my_rc != MXZ_OK) c->stream_rc = my_rc;
        mx_sync();
        rc = c->stream_rc;
        mx_sync();
        c->stream_rc = MXZ_OK;  // reset for the next block (all waves write)
        mx_sync();
      }
#else
      for (u32 k = 0; k < 4 && rc == MXZ_OK; k++)
        rc = huf_decode_stream(&c->huf, srcs[k], slens[k],
                               c->lit_scratch + (u64)k * r123, rlens[k]);
#endif
      if (rc < 0) return rc;
    }
    lits = c->lit_scratch;
    pos += lit_comp;
  }

  // ---- sequences section ----
  if (pos >= srclen) return MXZ_ERR_SEQUENCES;
  u32 nseq;
  u32 sb0 = src[pos];
  if (sb0 == 0) {
    nseq = 0;
    pos += 1;
  } else if (sb0 < 128) {
    nseq = sb0;
    pos += 1;
  } else if (sb0 < 255) {
    if (pos + 2 > srclen) return MXZ_ERR_SEQUENCES;
    nseq = ((sb0 - 128) << 8) + src[pos + 1];
    pos += 2;
  } else {
    if (pos + 3 > srclen) return MXZ_ERR_SEQUENCES;
    nseq = src[pos + 1] + ((u32)src[pos + 2] << 8) + 0x7F00;
    pos += 3;
  }

  u64 out = 0;
  if (nseq == 0) {
    // all-literals block
    if (lit_regen > dstcap) return MXZ_ERR_DST_SMALL;
    if (lits_rle)
      mx_par_set(dst, rle_byte, lit_regen);
    else
      mx_par_copy(dst, lits, lit_regen);
    return (i64)lit_regen;
  }

  if (pos >= srclen) return MXZ_ERR_SEQUENCES;
  u32 modes = src[pos++];
  if (modes & 3) return MXZ_ERR_SEQUENCES;  // reserved bits must be 0
  int n = seq_table_load(&c->ll, &c->ll_valid, (modes >> 6) & 3, 0, src + pos, srclen - pos, &c->bs);
  if (n < 0) return n;
  pos += n;
  n = seq_table_load(&c->of, &c->of_valid, (modes >> 4) & 3, 1, src + pos, srclen - pos, &c->bs);
  if (n < 0) return n;
  pos += n;
  n = seq_table_load(&c->ml, &c->ml_valid, (modes >> 2) & 3, 2, src + pos, srclen - pos, &c->bs);
  if (n < 0) return n;
  pos += n;

  BackBits bb;
  if (!bb.init(src + pos, srclen - pos)) return MXZ_ERR_SEQUENCES;
  u32 stLL = bb.read(c->ll.log);
  u32 stOF = bb.read(c->of.log);
  u32 stML = bb.read(c->ml.log);
  if (bb.overflowed()) return MXZ_ERR_SEQUENCES;

  u64 lit_used = 0;
  for (u32 s = 0; s < nseq; s++) {
    // bound every FSE state before it indexes a table: a corrupt table
    // description can break the base+nbits invariant, and one
    // slightly-out-of-range state reads garbage base[] that catapults the
    // next state far outside the struct (found by ASAN replay of the
    // decoder mutation fuzz — intermittent SEGV). Applies to the GPU
    // decoder identically (device OOB reads crash the box).
    if (stLL >= (1u << c->ll.log) || stOF >= (1u << c->of.log) ||
        stML >= (1u << c->ml.log))
      return MXZ_ERR_SEQUENCES;
    u32 ofCode = c->of.symbol[stOF];
    u32 mlCode = c->ml.symbol[stML];
    u32 llCode = c->ll.symbol[stLL];
    if (ofCode > 31 || mlCode > 52 || llCode > 35) return MXZ_ERR_SEQUENCES;
    u64 ofValue = ((u64)1 << ofCode) + bb.read(ofCode);  // ofCode == nbits
    u32 mlBits, mlBase, llBits, llBase;
    ml_code_info(mlCode, &mlBits, &mlBase);
    u64 ml = mlBase + bb.read(mlBits);
    ll_code_info(llCode, &llBits, &llBase);
    u64 ll = llBase + bb.read(llBits);
    if (bb.overflowed() && s + 1 < nseq) return MXZ_ERR_SEQUENCES;
    if (s + 1 < nseq) {
      stLL = c->ll.base[stLL] + bb.read(c->ll.nbits[stLL]);
      stML = c->ml.base[stML] + bb.read(c->ml.nbits[stML]);
      stOF = c->of.base[stOF] + bb.read(c->of.nbits[stOF]);
    }
    // resolve offset (repeat offsets)
    u64 offset;
    if (ofValue <= 3) {
      u32 idx = (u32)ofValue - 1 + (ll == 0 ? 1 : 0);
      if (idx == 0) {
        offset = rep[0];
      } else {
        offset = idx < 3 ? rep[idx] : (u64)rep[0] - 1;
        if (offset == 0) return MXZ_ERR_OFFSET;
        if (idx > 1) rep[2] = rep[1];
        rep[1] = rep[0];
        rep[0] = (u32)offset;
      }
    } else {
      offset = ofValue - 3;
      rep[2] = rep[1];
      rep[1] = rep[0];
      rep[0] = (u32)offset;
    }
    // copy literals
    if (lit_used + ll > lit_regen) return MXZ_ERR_SEQUENCES;
    if (out + ll + ml > dstcap) return MXZ_ERR_DST_SMALL;
    if (ll) {
      if (lits_rle)
        mx_par_set(dst + out, rle_byte, ll);
      else
        mx_par_copy(dst + out, lits + lit_used, ll);
      lit_used += ll;
      out += ll;
    }
    // copy match
    if (ml) {
      if (offset > history + out) return MXZ_ERR_OFFSET;
      mx_match_copy(dst + out, offset, ml);
      out += ml;
    }
  }
  // trailing literals
  u64 rest = lit_regen - lit_used;
  if (out + rest > dstcap) return MXZ_ERR_DST_SMALL;
  if (rest) {
    if (lits_rle)
      mx_par_set(dst + out, rle_byte, rest);
    else
      mx_par_copy(dst + out, lits + lit_used, rest);
    out += rest;
  }
  if (!bb.finished() && !bb.overflowed()) return MXZ_ERR_SEQUENCES;  // leftover bits
  return (i64)out;
}

// Decode one complete frame from src into dst. Returns decompressed size or
// <0. `ctx` is caller-allocated (LDS on the GPU); ctx->lit_scratch must
// point at >= kBlockMax bytes.
MX_HD static inline i64 decode_frame(const u8* src, u64 srclen, u8* dst, u64 dstcap,
                                     DecCtx* ctx, u64* consumed_out) {
  if (srclen < 4) return MXZ_ERR_SRC_TRUNC;
  u32 magic = mx_read_le32(src);
  if (magic != kMagic) return MXZ_ERR_MAGIC;
  u64 pos = 4;
  if (pos >= srclen) return MXZ_ERR_SRC_TRUNC;
  u8 fhd = src[pos++];
  u32 fcs_flag = fhd >> 6;
  bool single_seg = (fhd >> 5) & 1;
  bool checksum = (fhd >> 2) & 1;
  u32 dict_flag = fhd & 3;
  if (fhd & 0x8) return MXZ_ERR_HEADER;  // reserved bit
  if (!single_seg) {
    if (pos >= srclen) return MXZ_ERR_SRC_TRUNC;
    pos++;  // window descriptor (we rely on dst being the whole frame)
  }
  if (dict_flag) {
    u32 nb = dict_flag == 3 ? 4 : dict_flag;  // 1,2,4 bytes
    u32 did = 0;
    for (u32 i = 0; i < nb && pos < srclen; i++) did |= (u32)src[pos++] << (8 * i);
    if (did != 0) return MXZ_ERR_UNSUPPORTED;  // dictionaries unsupported
  }
  u64 fcs = 0;
  bool have_fcs = true;
  switch (fcs_flag) {
    case 0:
      if (single_seg) {
        if (pos >= srclen) return MXZ_ERR_SRC_TRUNC;
        fcs = src[pos++];
      } else {
        have_fcs = false;
      }
      break;
    case 1:
      if (pos + 2 > srclen) return MXZ_ERR_SRC_TRUNC;
      fcs = ((u64)src[pos] | ((u64)src[pos + 1] << 8)) + 256;
      pos += 2;
      break;
    case 2:
      if (pos + 4 > srclen) return MXZ_ERR_SRC_TRUNC;
      fcs = mx_read_le32(src + pos);
      pos += 4;
      break;
    default:
      if (pos + 8 > srclen) return MXZ_ERR_SRC_TRUNC;
      fcs = mx_read_le64(src + pos);
      pos += 8;
      break;
  }
  if (have_fcs && fcs > dstcap) return MXZ_ERR_DST_SMALL;

  dec_ctx_init(ctx, ctx->lit_scratch);
  u32 rep[3] = {1, 4, 8};  // thread-local: RMW state (see BuildScratch note)
  u64 out = 0;
  while (true) {
    if (pos + 3 > srclen) return MXZ_ERR_SRC_TRUNC;
    u32 bh = (u32)src[pos] | ((u32)src[pos + 1] << 8) | ((u32)src[pos + 2] << 16);
    pos += 3;
    bool last = bh & 1;
    u32 btype = (bh >> 1) & 3;
    u32 bsize = bh >> 3;
    if (btype == 0) {  // raw
      if (pos + bsize > srclen) return MXZ_ERR_SRC_TRUNC;
      if (out + bsize > dstcap) return MXZ_ERR_DST_SMALL;
      mx_par_copy(dst + out, src + pos, bsize);
      out += bsize;
      pos += bsize;
    } else if (btype == 1) {  // RLE
      if (pos + 1 > srclen) return MXZ_ERR_SRC_TRUNC;
      if (out + bsize > dstcap) return MXZ_ERR_DST_SMALL;
      mx_par_set(dst + out, src[pos], bsize);
      out += bsize;
      pos += 1;
    } else if (btype == 2) {
      if (bsize > kBlockMax + 32) return MXZ_ERR_BLOCK;
      if (pos + bsize > srclen) return MXZ_ERR_SRC_TRUNC;
      i64 n = decode_block(ctx, rep, src + pos, bsize, dst + out, dstcap - out, out);
      if (n < 0) return n;
      out += (u64)n;
      pos += bsize;
    } else {
      return MXZ_ERR_BLOCK;
    }
    if (last) break;
  }
  if (checksum) {
    if (pos + 4 > srclen) return MXZ_ERR_SRC_TRUNC;
    pos += 4;  // XXH64 low-32 — integrity is covered by the registry digests
  }
  if (have_fcs && out != fcs) return MXZ_ERR_CORRUPT;
  if (consumed_out) *consumed_out = pos;
  return (i64)out;
}

}  // namespace zstd
}  // namespace modelx
