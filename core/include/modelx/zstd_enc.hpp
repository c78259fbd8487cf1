// zstd encoder core — standard RFC 8878 frames from greedy LZ77 matches,
// raw literals and the predefined FSE sequence tables. Always decodable by
// stock zstd; shared between the CDNA4 compress kernel (core/hip/zstd.hip)
// and the CPU path (core/src/zstd_cpu.cpp).
//
// Same redundant-wavefront execution model as zstd_core.hpp: control flow
// runs identically on all 64 lanes; literal byte moves go through
// mx_par_copy.
#pragma once

#include "modelx/zstd_core.hpp"

namespace modelx {
namespace zstd {

// ---------------------------------------------------------- bit writer -----

// Forward byte emission, backward-readable (zstd bitstream convention):
// bits accumulate LSB-first; close() appends the 1-marker then pads to a
// byte boundary.
struct BitW {
  u8* dst;
  u64 cap;
  u64 pos;     // bytes written
  u64 acc;
  u32 nacc;    // bits in acc
  bool overflow;

  MX_HD void init(u8* d, u64 c) {
    dst = d;
    cap = c;
    pos = 0;
    acc = 0;
    nacc = 0;
    overflow = false;
  }
  MX_HD void add(u64 v, u32 nbits) {
    if (nbits == 0) return;
    acc |= (v & (((u64)1 << nbits) - 1)) << nacc;
    nacc += nbits;
    while (nacc >= 8) {
      if (pos >= cap) {
        overflow = true;
        return;
      }
      dst[pos++] = (u8)acc;
      acc >>= 8;
      nacc -= 8;
    }
  }
  MX_HD void close() {
    add(1, 1);  // padding marker
    if (nacc) {
      if (pos >= cap) {
        overflow = true;
        return;
      }
      dst[pos++] = (u8)acc;
      acc = 0;
      nacc = 0;
    }
  }
};

// ------------------------------------------------------- FSE encoder -------

struct FseEnc {
  u16 state_table[1 << 6];   // predefined tables: log <= 6
  i32 delta_find[53];
  u32 delta_nbits[53];
  u32 log;
};

MX_HD static inline int fse_build_ctable(FseEnc* e, const i16* counts, u32 nsym, u32 log) {
  u32 size = 1u << log;
  if (size > (1u << 6)) return MXZ_ERR_FSE;
  u16 spread[1 << 6];
  u32 high = size - 1;
  for (u32 s = 0; s < nsym; s++)
    if (counts[s] == -1) spread[high--] = (u16)s;
  u32 step = (size >> 1) + (size >> 3) + 3;
  u32 mask = size - 1;
  u32 pos = 0;
  for (u32 s = 0; s < nsym; s++) {
    if (counts[s] <= 0) continue;
    for (i32 i = 0; i < counts[s]; i++) {
      spread[pos] = (u16)s;
      do {
        pos = (pos + step) & mask;
      } while (pos > high);
    }
  }
  if (pos != 0) return MXZ_ERR_FSE;
  u32 cumul[54];
  cumul[0] = 0;
  for (u32 s = 0; s < nsym; s++)
    cumul[s + 1] = cumul[s] + (u32)(counts[s] == -1 ? 1 : (counts[s] < 0 ? 0 : counts[s]));
  u32 fill[54];
  for (u32 s = 0; s <= nsym; s++) fill[s] = cumul[s];
  for (u32 u = 0; u < size; u++) {
    u32 s = spread[u];
    e->state_table[fill[s]++] = (u16)(size + u);
  }
  u32 total = 0;
  for (u32 s = 0; s < nsym; s++) {
    i32 c = counts[s];
    if (c == 0) {
      e->delta_nbits[s] = ((log + 1) << 16) - size;
      e->delta_find[s] = 0;
    } else if (c == -1 || c == 1) {
      e->delta_nbits[s] = (log << 16) - size;
      e->delta_find[s] = (i32)total - 1;
      total += 1;
    } else {
      u32 max_bits = log - mx_highbit((u32)c - 1);
      e->delta_nbits[s] = (max_bits << 16) - ((u32)c << max_bits);
      e->delta_find[s] = (i32)total - c;
      total += (u32)c;
    }
  }
  e->log = log;
  return MXZ_OK;
}

struct FseState {
  u32 value;
};

MX_HD static inline void fse_enc_init(const FseEnc* e, FseState* st, u32 sym) {
  u32 nbits = (e->delta_nbits[sym] + (1u << 15)) >> 16;
  u32 v = (nbits << 16) - e->delta_nbits[sym];
  st->value = e->state_table[(v >> nbits) + (u32)((i32)e->delta_find[sym])];
}

MX_HD static inline void fse_enc_symbol(const FseEnc* e, FseState* st, BitW* bw, u32 sym) {
  u32 nbits = (st->value + e->delta_nbits[sym]) >> 16;
  bw->add(st->value, nbits);
  st->value = e->state_table[(st->value >> nbits) + (u32)((i32)e->delta_find[sym])];
}

MX_HD static inline void fse_enc_flush(const FseEnc* e, FseState* st, BitW* bw) {
  bw->add(st->value, e->log);
}

// ---------------------------------------- FSE-compressed huffman weights --
// The direct 4-bit weight table covers alphabets whose last symbol is
// < 128; full-byte alphabets (bf16/fp32 tensor bytes have the sign bit set
// half the time) need the RFC 8878 FSE-compressed representation or they
// fall all the way back to RAW blocks — measured 1.000 ratio on bf16
// weights vs 0.78 for zstd -3, the whole gap being this serialization.

// Normalize the weight histogram to sum 1<<log. Every count is capped at
// size/2 so every FSE state consumes >=1 bit — the overflow-terminated
// weight decoder (huf_read_table) needs the final transition read to
// actually overflow.
MX_HD static inline bool fse_normalize_weights(const u32* hist, u32 nsym, u32 total,
                                               u32 log, i16* counts) {
  u32 size = 1u << log;
  u32 cap = size / 2;
  i32 assigned = 0;
  u32 nz = 0;
  for (u32 s = 0; s < nsym; s++) counts[s] = 0;
  for (u32 s = 0; s < nsym; s++) {
    if (!hist[s]) continue;
    nz++;
    u32 c = (u32)((u64)hist[s] * size / total);
    if (c == 0) {
      counts[s] = -1;
      assigned += 1;
    } else {
      if (c > cap) c = cap;
      counts[s] = (i16)c;
      assigned += (i32)c;
    }
  }
  if (nz < 2) return false;  // degenerate alphabet (caller falls back)
  while (assigned < (i32)size) {
    u32 pick = nsym;
    u32 best = 0;
    for (u32 s = 0; s < nsym; s++)
      if (counts[s] >= 1 && (u32)counts[s] < cap && hist[s] >= best) {
        best = hist[s];
        pick = s;
      }
    if (pick == nsym) return false;
    counts[pick]++;
    assigned++;
  }
  while (assigned > (i32)size) {
    u32 pick = nsym;
    u32 best = 0xFFFFFFFFu;
    for (u32 s = 0; s < nsym; s++)
      if (counts[s] > 1 && hist[s] < best) {
        best = hist[s];
        pick = s;
      }
    if (pick == nsym) return false;
    counts[pick]--;
    assigned--;
  }
  return true;
}

// Serialize normalized counts — the exact inverse of fse_read_ncount
// (forward LSB-first bitstream, threshold coding, 2-bit zero-run codes).
// Returns bytes written or <0.
MX_HD static inline i64 fse_write_ncount(const i16* counts, u32 last_sym, u32 log, u8* dst,
                                         u64 cap) {
  BitW bw;
  bw.init(dst, cap);
  bw.add(log - 5, 4);
  i32 remaining = (i32)(1u << log) + 1;
  u32 threshold = 1u << log;
  u32 nbits = log + 1;
  u32 s = 0;
  while (remaining > 1 && s <= last_sym) {
    i32 count = counts[s];
    u32 value = (u32)(count + 1);  // reader does count--
    u32 max = (2 * threshold - 1) - (u32)remaining;
    if (value < max)
      bw.add(value, nbits - 1);
    else
      bw.add(value < threshold ? value : value + max, nbits);
    if (bw.overflow) return MXZ_ERR_DST_SMALL;
    remaining -= count < 0 ? -count : count;
    s++;
    while ((u32)remaining < threshold) {
      nbits--;
      threshold >>= 1;
    }
    if (count == 0 && remaining > 1) {
      // reader consumes 2-bit zero-run codes right after a zero count
      u32 zrun = 0;
      while (s + zrun <= last_sym && counts[s + zrun] == 0) zrun++;
      u32 z = zrun;
      while (z >= 3) {
        bw.add(3, 2);
        z -= 3;
      }
      bw.add(z, 2);
      if (bw.overflow) return MXZ_ERR_DST_SMALL;
      s += zrun;
    }
  }
  if (remaining != 1) return MXZ_ERR_FSE;
  // byte-align (the reader consumes whole bytes)
  if (bw.nacc) {
    if (bw.pos >= bw.cap) return MXZ_ERR_DST_SMALL;
    bw.dst[bw.pos++] = (u8)bw.acc;
  }
  return (i64)bw.pos;
}

// Huffman_Tree_Description in the FSE-compressed form: header byte =
// compressed size (<128), FSE ncount table, then the two-state backward
// bitstream (classic zstd order: input consumed from the end, states
// flushed last so the decoder reads them first). Returns total bytes
// or <0 (caller falls back to raw literals).
MX_HD static inline i64 emit_fse_weights(const u8* wgts, u32 n, u8* dst, u64 cap,
                                         FseEnc* fe) {
  if (n < 2 || cap < 4) return MXZ_ERR_HUFFMAN;
  u32 hist[16] = {};
  u32 maxw = 0;
  for (u32 i = 0; i < n; i++) {
    u32 w = wgts[i];
    if (w > 12) return MXZ_ERR_HUFFMAN;
    hist[w]++;
    if (w > maxw) maxw = w;
  }
  const u32 log = 6;  // weights max accuracy log (RFC 8878)
  i16 counts[16];
  if (!fse_normalize_weights(hist, maxw + 1, n, log, counts)) return MXZ_ERR_HUFFMAN;
  i64 hdr = fse_write_ncount(counts, maxw, log, dst + 1, cap - 1);
  if (hdr < 0) return hdr;
  if (fse_build_ctable(fe, counts, maxw + 1, log) < 0) return MXZ_ERR_FSE;
  BitW bw;
  bw.init(dst + 1 + hdr, cap - 1 - (u64)hdr);
  FseState s1, s2;
  i64 ip = n;
  if (n & 1) {
    fse_enc_init(fe, &s1, wgts[--ip]);
    fse_enc_init(fe, &s2, wgts[--ip]);
    fse_enc_symbol(fe, &s1, &bw, wgts[--ip]);
  } else {
    fse_enc_init(fe, &s2, wgts[--ip]);
    fse_enc_init(fe, &s1, wgts[--ip]);
  }
  while (ip > 0) {
    fse_enc_symbol(fe, &s2, &bw, wgts[--ip]);
    fse_enc_symbol(fe, &s1, &bw, wgts[--ip]);
  }
  fse_enc_flush(fe, &s2, &bw);
  fse_enc_flush(fe, &s1, &bw);
  bw.close();
  if (bw.overflow) return MXZ_ERR_DST_SMALL;
  u64 csize = (u64)hdr + bw.pos;
  if (csize >= 128) return MXZ_ERR_HUFFMAN;  // header byte form requires <128
  dst[0] = (u8)csize;
  return (i64)(1 + csize);
}

// ------------------------------------------------------ code mapping -------

MX_HD static inline u32 ll_code_of(u32 ll) {
  static const u8 kTab[64] = {0,  1,  2,  3,  4,  5,  6,  7,  8,  9,  10, 11, 12, 13, 14, 15,
                              16, 16, 17, 17, 18, 18, 19, 19, 20, 20, 20, 20, 21, 21, 21, 21,
                              22, 22, 22, 22, 22, 22, 22, 22, 23, 23, 23, 23, 23, 23, 23, 23,
                              24, 24, 24, 24, 24, 24, 24, 24, 24, 24, 24, 24, 24, 24, 24, 24};
  if (ll < 64) return kTab[ll];
  u32 hb = mx_highbit(ll);
  return hb + 19;  // 64..127 -> 25, 128.. -> 26, ... (2^h -> h+19)
}

MX_HD static inline u32 ml_code_of(u32 ml) {
  // ml >= 3; mlBase = ml - 3
  static const u8 kTab[128] = {
      0,  1,  2,  3,  4,  5,  6,  7,  8,  9,  10, 11, 12, 13, 14, 15, 16, 17, 18, 19, 20, 21,
      22, 23, 24, 25, 26, 27, 28, 29, 30, 31, 32, 32, 33, 33, 34, 34, 35, 35, 36, 36, 36, 36,
      37, 37, 37, 37, 38, 38, 38, 38, 38, 38, 38, 38, 39, 39, 39, 39, 39, 39, 39, 39, 40, 40,
      40, 40, 40, 40, 40, 40, 40, 40, 40, 40, 40, 40, 40, 40, 41, 41, 41, 41, 41, 41, 41, 41,
      41, 41, 41, 41, 41, 41, 41, 41, 42, 42, 42, 42, 42, 42, 42, 42, 42, 42, 42, 42, 42, 42,
      42, 42, 42, 42, 42, 42, 42, 42, 42, 42, 42, 42, 42, 42, 42, 42, 42, 42};
  u32 v = ml - 3;
  if (v < 128) return kTab[v];
  u32 hb = mx_highbit(v);
  return hb + 36;  // 128.. -> 43, 256.. -> 44, ...
}

// ------------------------------------------------------------- LZ parse ----

static const u32 kHashLog = 13;  // 8192-entry table, 32 KiB as u32

MX_HD static inline u32 lz_hash(const u8* p) {
  u32 v = mx_read_le32(p);
  return (v * 2654435761u) >> (32 - kHashLog);
}

struct Seq {
  u32 ll, ml, off;  // literal run length, match length, raw offset (>=1)
};

// Greedy parse of block[0..len) with matches back into base[0..block_end).
// `hash` is a (1<<kHashLog) u32 table of position+1 into base, shared across
// blocks of one frame (zeroed at frame start). Returns number of sequences.
// seqs capacity must be >= len/4 + 1.
//
// Device build: wave-parallel speculation — each round, the 64 lanes probe
// the hash table at positions p..p+63 simultaneously (racy same-round LDS
// inserts are fine: the table is a heuristic, every candidate is verified
// against the actual bytes), a ballot picks the first real match, and match
// extension compares 64 bytes per step. Output differs from the CPU parse
// (insertion races) but is always a valid zstd stream; tests verify
// roundtrips, not byte equality.
#if defined(__HIP_DEVICE_COMPILE__)
MX_HD static inline u32 lz_parse(const u8* base, u64 block_off, u64 block_len, u32* hash,
                                 Seq* seqs, u32 max_seqs, u64* lit_total) {
  u64 end = block_off + block_len;
  u64 p = block_off;
  u64 lit_start = p;
  u32 nseq = 0;
  u64 limit = end >= 8 ? end - 8 : 0;
  u32 lane = threadIdx.x;
  while (p < limit && nseq < max_seqs) {
    u64 q = p + lane;
    bool in_range = q < limit;
    u64 cpos = 0;
    bool found = false;
    if (in_range) {
      u32 h = lz_hash(base + q);
      u32 cand = hash[h];
      hash[h] = (u32)(q + 1);  // racy across lanes — heuristic table
      if (cand != 0) {
        cpos = (u64)cand - 1;
        u64 dist = q - cpos;  // wraps huge if cand is a later same-round pos
        found = dist > 0 && dist <= ((u64)1 << 27) &&
                mx_read_le32(base + cpos) == mx_read_le32(base + q);
      }
    }
    unsigned long long ballot = __ballot(found);
    if (ballot == 0) {
      p += 64;
      continue;
    }
    u32 l0 = (u32)__builtin_ctzll(ballot);
    u64 q0 = p + l0;
    u64 c0 = __shfl((unsigned long long)cpos, (int)l0);
    // parallel extension: 64 bytes per step
    u64 m = 4;
    while (true) {
      u64 t = q0 + m + lane;
      bool diff = t >= end || base[c0 + m + lane] != base[t];
      unsigned long long bb = __ballot(diff);
      if (bb) {
        m += __builtin_ctzll(bb);
        break;
      }
      m += 64;
    }
    seqs[nseq].ll = (u32)(q0 - lit_start);
    seqs[nseq].ml = (u32)m;
    seqs[nseq].off = (u32)(q0 - c0);
    nseq++;
    p = q0 + m;
    lit_start = p;
  }
  __syncthreads();
  u64 t = end - lit_start;
  for (u32 i = 0; i < nseq; i++) t += seqs[i].ll;
  *lit_total = t;
  return nseq;
}
#else
MX_HD static inline u32 lz_parse(const u8* base, u64 block_off, u64 block_len, u32* hash,
                                 Seq* seqs, u32 max_seqs, u64* lit_total) {
  u64 end = block_off + block_len;
  u64 p = block_off;
  u64 lit_start = p;
  u32 nseq = 0;
  u64 limit = end >= 8 ? end - 8 : 0;  // room for u32 loads + extension
  while (p < limit && nseq < max_seqs) {
    u32 h = lz_hash(base + p);
    u32 cand = hash[h];
    hash[h] = (u32)(p + 1);
    if (cand != 0) {
      u64 cpos = cand - 1;
      u64 dist = p - cpos;
      if (dist > 0 && dist <= (u64)1 << 27 && mx_read_le32(base + cpos) == mx_read_le32(base + p)) {
        // extend match
        u64 m = 4;
        while (p + m < end && base[cpos + m] == base[p + m]) m++;
        seqs[nseq].ll = (u32)(p - lit_start);
        seqs[nseq].ml = (u32)m;
        seqs[nseq].off = (u32)(p - cpos);
        nseq++;
        // sparse hash inserts inside the match body
        u64 stop = p + m < limit ? p + m : limit;
        for (u64 q = p + 1; q < stop; q += 7) hash[lz_hash(base + q)] = (u32)(q + 1);
        p += m;
        lit_start = p;
        continue;
      }
    }
    p++;
  }
  u64 t = end - lit_start;  // trailing literal run
  for (u32 i = 0; i < nseq; i++) t += seqs[i].ll;
  *lit_total = t;
  return nseq;
}
#endif

// ------------------------------------------------- huffman literals --------

// Byte histogram. Device: lane-strided with LDS atomics (single-wave
// kernel; all lanes see the final counts after the barrier).
MX_HD static inline void lit_histogram(const u8* lit, u64 n, u32* hist) {
#if defined(__HIP_DEVICE_COMPILE__)
  u32 lane = mx_lane(), w = mx_width();
  for (u32 i = lane; i < 256; i += w) hist[i] = 0;
  mx_sync();
  for (u64 i = lane; i < n; i += w) atomicAdd(&hist[lit[i]], 1u);
  mx_sync();
#else
  for (u32 i = 0; i < 256; i++) hist[i] = 0;
  for (u64 i = 0; i < n; i++) hist[lit[i]]++;
#endif
}

struct HufEnc {
  u16 code[256];
  u8 len[256];  // 0 = absent
  u32 maxbits;
  u32 last_sym;
  u64 est_bits;
};

// Kraft-exact length-limited (<=11) code construction + canonical code
// assignment matching the decoder's table fill (huf_build: weight
// ascending == length descending, symbol ascending). Returns false when
// huffman can't apply (degenerate alphabet). Alphabets whose last symbol
// is past 127 serialize their weights FSE-compressed (emit_fse_weights);
// smaller ones use the direct 4-bit form.
MX_HD static inline bool huf_build_enc(const u32* hist, u64 total, HufEnc* e) {
  u32 nsym = 0, last = 0;
  for (u32 s = 0; s < 256; s++)
    if (hist[s]) {
      nsym++;
      last = s;
    }
  if (nsym < 2 || total < 64) return false;
  for (u32 s = 0; s < 256; s++) e->len[s] = 0;
  i64 K = 0;  // kraft sum in units of 2^-11
  for (u32 s = 0; s <= last; s++) {
    if (!hist[s]) continue;
    u32 ratio = (u32)(total / hist[s]);
    u32 l = ratio <= 1 ? 1 : mx_highbit(ratio) + ((ratio & (ratio - 1)) ? 1 : 0);
    if (l < 1) l = 1;
    if (l > 11) l = 11;
    e->len[s] = (u8)l;
    K += (i64)1 << (11 - l);
  }
  // shrink shares of the rarest symbols until the sum fits
  for (int guard = 0; K > 2048 && guard < 4096; guard++) {
    u32 pick = 256;
    u32 best = 0xFFFFFFFFu;
    for (u32 s = 0; s <= last; s++)
      if (e->len[s] && e->len[s] < 11 && hist[s] < best) {
        best = hist[s];
        pick = s;
      }
    if (pick == 256) return false;
    K -= (i64)1 << (11 - e->len[pick] - 1);
    e->len[pick]++;
  }
  // grow shares of the most frequent symbols to land exactly on 2^11
  for (int guard = 0; K < 2048 && guard < 4096; guard++) {
    u32 pick = 256;
    u32 best = 0;
    i64 gap = 2048 - K;
    for (u32 s = 0; s <= last; s++) {
      if (!e->len[s] || e->len[s] <= 1) continue;
      i64 delta = (i64)1 << (11 - e->len[s]);
      if (delta <= gap && hist[s] >= best) {
        best = hist[s];
        pick = s;
      }
    }
    if (pick == 256) return false;
    K += (i64)1 << (11 - e->len[pick]);
    e->len[pick]--;
  }
  if (K != 2048) return false;
  u32 maxbits = 0;
  for (u32 s = 0; s <= last; s++)
    if (e->len[s] > maxbits) maxbits = e->len[s];
  // canonical codes in the decoder's fill order
  u32 size = 1u << maxbits;
  u32 pos = 0;
  for (u32 l = maxbits; l >= 1; l--) {
    u32 run = 1u << (maxbits - l);
    for (u32 s = 0; s <= last; s++) {
      if (e->len[s] != l) continue;
      e->code[s] = (u16)(pos >> (maxbits - l));
      pos += run;
    }
  }
  if (pos != size) return false;
  u64 bits = 0;
  for (u32 s = 0; s <= last; s++) bits += (u64)hist[s] * e->len[s];
  e->maxbits = maxbits;
  e->last_sym = last;
  e->est_bits = bits;
  return true;
}

// Encode lit[a..b) as one backward huffman stream; returns bytes or <0.
//
// Device path: the bit position of every code is a prefix sum of code
// lengths, so the whole stream is written lane-parallel — scan 64 lengths
// per step (wave shuffles), then each lane ORs its code into the output
// words (atomicOr; codes are <= 11 bits so they span at most two u32s).
// The serial BitW path remains for the CPU and for short runs.
MX_HD static inline i64 huf_encode_stream(const HufEnc* e, const u8* lit, u64 a, u64 b,
                                          u8* dst, u64 cap, u32 flags, u32* scan_tmp) {
#if defined(__HIP_DEVICE_COMPILE__)
  u64 n = b - a;
  if (n >= 256 && !(flags & 1u)) {
    u32 lane = threadIdx.x & 63;
    // total bits: per-lane partials exchanged through LDS (explicit
    // barriers — no reliance on wave-shuffle out-of-range semantics)
    u32 my_bits = 0;
    for (u64 k = lane; k < n; k += 64) my_bits += e->len[lit[b - 1 - k]];
    scan_tmp[lane] = my_bits;
    mx_sync();
    u64 T = 0;
    for (u32 i = 0; i < 64; i++) T += scan_tmp[i];
    mx_sync();
    u64 bytes = (T + 1 + 7) / 8;
    if (bytes > cap) return MXZ_ERR_DST_SMALL;
    // word-aligned view; preserve the bytes before dst in the first word
    uintptr_t addr = reinterpret_cast<uintptr_t>(dst);
    u32 misal = (u32)(addr & 3);
    u32* words = reinterpret_cast<u32*>(addr - misal);
    u64 bit0 = (u64)misal * 8;  // stream bit i lives at word bit (bit0 + i)
    u64 nwords = (bit0 + T + 1 + 31) / 32;
    u32 keep = 0;
    if (lane == 0 && misal) {
      // NB: index arithmetic kept in signed/pointer space — mixing a
      // negated value with an unsigned index promotes to u32 and wraps to
      // a ~4 GiB offset (page-faulted on hardware)
      const u8* before = dst - misal;
      for (u32 i = 0; i < misal; i++) keep |= (u32)before[i] << (8 * i);
    }
    for (u64 wdi = lane; wdi < nwords; wdi += 64) words[wdi] = 0;
    mx_sync();
    if (lane == 0 && misal) atomicOr(&words[0], keep);
    // chunked scan + scatter: lengths exchanged through LDS, every lane
    // computes its own exclusive prefix
    u64 running = 0;
    for (u64 base = 0; base < n; base += 64) {
      u64 k = base + lane;
      u32 sym = k < n ? lit[b - 1 - k] : 0;
      u32 nk = k < n ? e->len[sym] : 0;
      scan_tmp[lane] = nk;
      mx_sync();
      u32 excl = 0;
      u32 chunk_total = 0;
      for (u32 i = 0; i < 64; i++) {
        if (i < lane) excl += scan_tmp[i];
        chunk_total += scan_tmp[i];
      }
      mx_sync();
      u64 S = running + excl + bit0;
      if (k < n && nk) {
        u64 wide = (u64)e->code[sym] << (S & 31);
        atomicOr(&words[S >> 5], (u32)wide);
        u32 hi = (u32)(wide >> 32);
        if (hi) atomicOr(&words[(S >> 5) + 1], hi);
      }
      running += chunk_total;
    }
    if (lane == 0) atomicOr(&words[(bit0 + T) >> 5], 1u << ((bit0 + T) & 31));
    mx_sync();
    return (i64)bytes;
  }
#endif
  BitW bw;
  bw.init(dst, cap);
  // written back-to-front so the backward reader produces them in order
  for (i64 i = (i64)b - 1; i >= (i64)a; i--) {
    u8 s = lit[i];
    bw.add(e->code[s], e->len[s]);
    if (bw.overflow) return MXZ_ERR_DST_SMALL;
  }
  bw.close();
  if (bw.overflow) return MXZ_ERR_DST_SMALL;
  return (i64)bw.pos;
}

// Emit a Compressed_Literals_Block (4-stream huffman, direct 4-bit weight
// table). Returns total section bytes (header included) or <0 when raw is
// better / capacity exceeded.
MX_HD static inline i64 emit_huf_literals(const HufEnc* e, const u8* lit, u64 n, u8* out,
                                          u64 cap, u32 flags, u32* scan_tmp,
                                          FseEnc* wfse) {
  if (n < 256) return MXZ_ERR_DST_SMALL;  // not worth the table
  u32 nweights = e->last_sym;  // last symbol's weight is implied
  u64 est_tbl = nweights < 128 ? 1 + (nweights + 1) / 2 : 1 + 128;
  u64 est_total = 5 + est_tbl + 6 + (e->est_bits + 7) / 8 + 8;
  if (est_total >= n) return MXZ_ERR_DST_SMALL;
  // header needs the compressed size — assemble body first at a safe
  // offset (max header 5 bytes), then write the header knowing sizes
  u64 hmax = 5;
  if (hmax + est_tbl + 6 >= cap) return MXZ_ERR_DST_SMALL;
  u8* body = out + hmax;
  u64 bcap = cap - hmax;
  u64 tbl;
  if (nweights < 128) {
    // direct 4-bit weight table
    body[0] = (u8)(127 + nweights);
    for (u32 i = 0; i < nweights; i++) {
      u32 l = e->len[i];
      u32 wgt = l ? (e->maxbits + 1 - l) : 0;
      if (i & 1)
        body[1 + i / 2] |= (u8)wgt;
      else
        body[1 + i / 2] = (u8)(wgt << 4);
    }
    tbl = 1 + (nweights + 1) / 2;
  } else {
    // full-byte alphabet: FSE-compressed weights
    u8 wgts[256];
    for (u32 i = 0; i < nweights; i++) {
      u32 l = e->len[i];
      wgts[i] = (u8)(l ? (e->maxbits + 1 - l) : 0);
    }
    i64 t = emit_fse_weights(wgts, nweights, body, bcap, wfse);
    if (t < 0) return t;
    tbl = (u64)t;
  }
  u64 bpos = tbl;
  u8* jump = body + bpos;
  bpos += 6;
  u64 r123 = (n + 3) / 4;
  u64 r4 = n - 3 * r123;
  if (r4 == 0) return MXZ_ERR_DST_SMALL;  // stream 4 must be non-empty
  u64 sizes[4];
  u64 offs[4] = {0, r123, 2 * r123, 3 * r123};
  u64 lens[4] = {r123, r123, r123, r4};
  for (int k = 0; k < 4; k++) {
    i64 m = huf_encode_stream(e, lit, offs[k], offs[k] + lens[k], body + bpos, bcap - bpos,
                              flags, scan_tmp);
    if (m < 0) return m;
    if (k < 3 && m > 0xFFFF) return MXZ_ERR_DST_SMALL;
    sizes[k] = (u64)m;
    bpos += (u64)m;
    if (bpos + 64 > bcap) return MXZ_ERR_DST_SMALL;
  }
  jump[0] = (u8)sizes[0];
  jump[1] = (u8)(sizes[0] >> 8);
  jump[2] = (u8)sizes[1];
  jump[3] = (u8)(sizes[1] >> 8);
  jump[4] = (u8)sizes[2];
  jump[5] = (u8)(sizes[2] >> 8);
  u64 comp = bpos;  // table + jump + streams
  if (comp >= n) return MXZ_ERR_DST_SMALL;
  // literals header: type=2 (Compressed), 4 streams, size-format 11
  // (18-bit fields — always large enough, and a fixed 5-byte header means
  // the body was assembled at its final offset)
  u64 h = 2u | (3u << 2);
  h |= n << 4;
  h |= comp << 22;
  for (u64 i = 0; i < 5; i++) out[i] = (u8)(h >> (8 * i));
  return (i64)(5 + comp);
}

// ----------------------------------------------------------- block emit ----

// Predefined-table FSE encoders + literal histogram (caller-allocated:
// LDS on the GPU so the redundant-wavefront execution doesn't spill
// per-lane copies; the histogram is filled with LDS atomics).
struct EncTables {
  u32 flags;  // bit0: disable the lane-parallel huffman stream encoder
  FseEnc ell, eof, eml;
  FseEnc wfse;  // huffman-weight FSE encoder (full-byte alphabets)
  HufEnc he;
  u32 lit_hist[256];
  u32 scan_tmp[64];  // cross-lane exchange for the parallel huffman encoder
};

MX_HD static inline int enc_tables_init(EncTables* et) {
  u32 nsym, log;
  const i16* d = ll_default_dist(&nsym, &log);
  if (fse_build_ctable(&et->ell, d, nsym, log) < 0) return MXZ_ERR_FSE;
  d = of_default_dist(&nsym, &log);
  if (fse_build_ctable(&et->eof, d, nsym, log) < 0) return MXZ_ERR_FSE;
  d = ml_default_dist(&nsym, &log);
  if (fse_build_ctable(&et->eml, d, nsym, log) < 0) return MXZ_ERR_FSE;
  return MXZ_OK;
}

// Emit one zstd block (compressed if it wins, raw otherwise) for
// src[block_off..block_off+block_len). Literals pick the cheapest of
// RLE / 4-stream huffman / raw; a whole-block single byte becomes an RLE
// block. Returns bytes written or <0.
MX_HD static inline i64 encode_block(const u8* src, u64 block_off, u64 block_len, bool last,
                                     u8* dst, u64 dstcap, u32* hash, Seq* seqs, u32 max_seqs,
                                     EncTables* et) {
  const u8* block = src + block_off;
  u64 raw_total = 3 + block_len;
  if (dstcap < raw_total) return MXZ_ERR_DST_SMALL;

  u64 lit_total = 0;
  u32 nseq = lz_parse(src, block_off, block_len, hash, seqs, max_seqs, &lit_total);

  u64 csize = 0;
  bool use_raw = false;
  if (nseq == 0) {
    // no matches: RLE block for a single repeated byte, else try a
    // huffman-literals-only compressed block, else raw
    lit_histogram(block, block_len, et->lit_hist);
    u32 nsym = 0;
    for (u32 i = 0; i < 256; i++)
      if (et->lit_hist[i]) nsym++;
    if (nsym == 1 && block_len > 0 && block_len < (1u << 21)) {
      u32 bh = ((u32)block_len << 3) | (1u << 1) | (last ? 1 : 0);
      dst[0] = (u8)bh;
      dst[1] = (u8)(bh >> 8);
      dst[2] = (u8)(bh >> 16);
      dst[3] = block[0];
      return 4;
    }
    if (block_len >= 512 && huf_build_enc(et->lit_hist, block_len, &et->he)) {
      i64 n = emit_huf_literals(&et->he, block, block_len, dst + 3, dstcap - 5, et->flags,
                                et->scan_tmp, &et->wfse);
      if (n > 0 && (u64)n + 1 < block_len) {
        dst[3 + n] = 0;  // zero sequences
        csize = (u64)n + 1;
        u32 bh = ((u32)csize << 3) | (2u << 1) | (last ? 1 : 0);
        dst[0] = (u8)bh;
        dst[1] = (u8)(bh >> 8);
        dst[2] = (u8)(bh >> 16);
        return (i64)(3 + csize);
      }
    }
    use_raw = true;
  }

  if (!use_raw && nseq > 0) {
    u8* out = dst + 3;
    u64 cap = dstcap - 3;
    // assemble the literal bytes into the tail of the seqs scratch
    // (capacity proof: lit_total <= block_len - 4*nseq and the free tail
    // is 12*(max_seqs - nseq) bytes, which always dominates)
    u8* lit_buf = reinterpret_cast<u8*>(seqs + nseq);
    {
      u64 p = block_off;
      u64 lw = 0;
      for (u32 i = 0; i < nseq; i++) {
        mx_par_copy(lit_buf + lw, src + p, seqs[i].ll);
        lw += seqs[i].ll;
        p += seqs[i].ll + (u64)seqs[i].ml;
      }
      mx_par_copy(lit_buf + lw, src + p, (block_off + block_len) - p);
    }
    lit_histogram(lit_buf, lit_total, et->lit_hist);
    u32 nsym = 0;
    for (u32 i = 0; i < 256; i++)
      if (et->lit_hist[i]) nsym++;

    // ---- literals section: RLE / huffman / raw ----
    u64 w = 0;
    bool lit_done = false;
    if (lit_total > 0 && nsym == 1) {  // RLE literals
      if (lit_total <= 31) {
        out[w++] = (u8)(1 | (0 << 2) | (lit_total << 3));
      } else if (lit_total <= 4095) {
        out[w++] = (u8)(1 | (1 << 2) | ((lit_total & 0xF) << 4));
        out[w++] = (u8)(lit_total >> 4);
      } else {
        out[w++] = (u8)(1 | (3 << 2) | ((lit_total & 0xF) << 4));
        out[w++] = (u8)((lit_total >> 4) & 0xFF);
        out[w++] = (u8)(lit_total >> 12);
      }
      out[w++] = lit_buf[0];
      lit_done = true;
    }
    if (!lit_done && lit_total >= 512 && nsym > 1 &&
        huf_build_enc(et->lit_hist, lit_total, &et->he)) {
      i64 n = emit_huf_literals(&et->he, lit_buf, lit_total, out + w, cap - w - 32,
                                et->flags, et->scan_tmp, &et->wfse);
      if (n > 0) {
        w += (u64)n;
        lit_done = true;
      }
    }
    if (!lit_done) {  // raw literals
      u64 lp = w;
      if (lit_total <= 31) {
        out[lp++] = (u8)(0 | (0 << 2) | (lit_total << 3));
      } else if (lit_total <= 4095) {
        out[lp++] = (u8)(0 | (1 << 2) | ((lit_total & 0xF) << 4));
        out[lp++] = (u8)(lit_total >> 4);
      } else {
        out[lp++] = (u8)(0 | (3 << 2) | ((lit_total & 0xF) << 4));
        out[lp++] = (u8)((lit_total >> 4) & 0xFF);
        out[lp++] = (u8)(lit_total >> 12);
      }
      if (lp + lit_total + 16 > cap) {
        use_raw = true;
      } else {
        mx_par_copy(out + lp, lit_buf, lit_total);
        w = lp + lit_total;
      }
    }

    if (!use_raw) {
      // ---- sequences section ----
      if (nseq < 128) {
        out[w++] = (u8)nseq;
      } else if (nseq < 0x7F00) {
        out[w++] = (u8)((nseq >> 8) + 128);
        out[w++] = (u8)(nseq & 0xFF);
      } else {
        out[w++] = 255;
        out[w++] = (u8)((nseq - 0x7F00) & 0xFF);
        out[w++] = (u8)((nseq - 0x7F00) >> 8);
      }
      out[w++] = 0;  // modes: all predefined
      const FseEnc* ell = &et->ell;
      const FseEnc* eof = &et->eof;
      const FseEnc* eml = &et->eml;
      BitW bw;
      bw.init(out + w, cap - w > block_len ? block_len : cap - w);  // must beat raw
      const Seq& lastq = seqs[nseq - 1];
      u32 ll_c = ll_code_of(lastq.ll);
      u32 ml_c = ml_code_of(lastq.ml);
      u64 of_v = (u64)lastq.off + 3;
      u32 of_c = mx_highbit((u32)of_v);
      FseState sll, sof, sml;
      fse_enc_init(eml, &sml, ml_c);
      fse_enc_init(eof, &sof, of_c);
      fse_enc_init(ell, &sll, ll_c);
      u32 llb, llbase, mlb, mlbase;
      ll_code_info(ll_c, &llb, &llbase);
      ml_code_info(ml_c, &mlb, &mlbase);
      bw.add(lastq.ll - llbase, llb);
      bw.add(lastq.ml - mlbase, mlb);
      bw.add(of_v - ((u64)1 << of_c), of_c);
      for (i32 i = (i32)nseq - 2; i >= 0; i--) {
        const Seq& q = seqs[i];
        u32 lc = ll_code_of(q.ll);
        u32 mc = ml_code_of(q.ml);
        u64 ov = (u64)q.off + 3;
        u32 oc = mx_highbit((u32)ov);
        fse_enc_symbol(eof, &sof, &bw, oc);
        fse_enc_symbol(eml, &sml, &bw, mc);
        fse_enc_symbol(ell, &sll, &bw, lc);
        ll_code_info(lc, &llb, &llbase);
        ml_code_info(mc, &mlb, &mlbase);
        bw.add(q.ll - llbase, llb);
        bw.add(q.ml - mlbase, mlb);
        bw.add(ov - ((u64)1 << oc), oc);
        if (bw.overflow) break;
      }
      fse_enc_flush(eml, &sml, &bw);
      fse_enc_flush(eof, &sof, &bw);
      fse_enc_flush(ell, &sll, &bw);
      bw.close();
      if (bw.overflow) {
        use_raw = true;
      } else {
        csize = w + bw.pos;
        if (csize >= block_len) use_raw = true;
      }
    }
  }

  if (use_raw) {
    u32 bh = ((u32)block_len << 3) | (0 << 1) | (last ? 1 : 0);
    dst[0] = (u8)bh;
    dst[1] = (u8)(bh >> 8);
    dst[2] = (u8)(bh >> 16);
    mx_par_copy(dst + 3, block, block_len);
    return (i64)(3 + block_len);
  }
  u32 bh = ((u32)csize << 3) | (2 << 1) | (last ? 1 : 0);
  dst[0] = (u8)bh;
  dst[1] = (u8)(bh >> 8);
  dst[2] = (u8)(bh >> 16);
  return (i64)(3 + csize);
}


// Encode src[0..len) as ONE standard zstd frame into dst. `hash` is a
// (1<<kHashLog) u32 scratch (will be zeroed), `seqs` holds >= kBlockMax/4+1
// entries, `et` holds initialized predefined tables (enc_tables_init).
// Returns frame size or <0.
MX_HD static inline i64 encode_frame(const u8* src, u64 len, u8* dst, u64 dstcap, u32* hash,
                                     Seq* seqs, EncTables* et) {

  {
    u32 lane = mx_lane(), w = mx_width();
    for (u32 i = lane; i < (1u << kHashLog); i += w) hash[i] = 0;
    mx_sync();
  }

  u64 pos = 0;
  if (dstcap < 16) return MXZ_ERR_DST_SMALL;
  mx_write_le32(dst, kMagic);
  pos = 4;
  // frame header: single-segment, FCS sized to len
  u32 fcs_flag;
  if (len < 256)
    fcs_flag = 0;
  else if (len < 65536 + 256)
    fcs_flag = 1;
  else if (len <= 0xFFFFFFFFull)
    fcs_flag = 2;
  else
    fcs_flag = 3;
  dst[pos++] = (u8)((fcs_flag << 6) | (1u << 5));
  switch (fcs_flag) {
    case 0:
      dst[pos++] = (u8)len;
      break;
    case 1: {
      u64 v = len - 256;
      dst[pos++] = (u8)v;
      dst[pos++] = (u8)(v >> 8);
      break;
    }
    case 2:
      mx_write_le32(dst + pos, (u32)len);
      pos += 4;
      break;
    default:
      for (u32 i = 0; i < 8; i++) dst[pos++] = (u8)(len >> (8 * i));
      break;
  }
  if (len == 0) {
    // single empty raw block
    if (pos + 3 > dstcap) return MXZ_ERR_DST_SMALL;
    dst[pos] = 1;  // last=1, type=raw, size=0
    dst[pos + 1] = 0;
    dst[pos + 2] = 0;
    return (i64)(pos + 3);
  }
  u64 off = 0;
  while (off < len) {
    u64 blen = len - off < kBlockMax ? len - off : kBlockMax;
    bool last = off + blen >= len;
    i64 n = encode_block(src, off, blen, last, dst + pos, dstcap - pos, hash, seqs,
                         (u32)(kBlockMax / 4 + 1), et);
    if (n < 0) return n;
    pos += (u64)n;
    off += blen;
  }
  return (i64)pos;
}

}  // namespace zstd
}  // namespace modelx
