// CPU path of the zstd codec: the same header-only core the CDNA4 kernels
// use (modelx/zstd_core.hpp, zstd_enc.hpp), run on host threads — used by
// the CPU client (pull of +zstd blobs without a GPU), by tests as the
// CPU-vs-libzstd oracle harness, and by the seekable-container helpers.
//
// Replaces the reference's CPU gzip path (pkg/client/helper.go:19-22).
#include "modelx/zstd_host.hpp"

#include <atomic>
#include <memory>
#include <cstring>
#include <stdexcept>
#include <string>
#include <thread>

#include "modelx/zstd_core.hpp"
#include "modelx/zstd_enc.hpp"

namespace modelx {
namespace zstdhost {

using namespace modelx::zstd;

static const u32 kSkippableMagicMin = 0x184D2A50u;
static const u32 kSkippableMagicMax = 0x184D2A5Fu;

std::vector<SeekEntry> parse_seek_table(const uint8_t* blob, size_t len) {
  std::vector<SeekEntry> out;
  if (len < 17) return out;
  // footer: u32 num_frames, u8 descriptor, u32 seekable magic
  if (mx_read_le32(blob + len - 4) != kSeekTableMagic) return out;
  uint8_t desc = blob[len - 5];
  bool checksums = desc & 0x80;
  uint32_t nframes = mx_read_le32(blob + len - 9);
  size_t entry_sz = checksums ? 12 : 8;
  uint64_t tbl_payload = (uint64_t)nframes * entry_sz + 9;
  if (tbl_payload + 8 > len) return out;
  const uint8_t* skf = blob + len - tbl_payload - 8;
  if (mx_read_le32(skf) != kMagicSkippableSeek) return out;
  if (mx_read_le32(skf + 4) != tbl_payload) return out;
  const uint8_t* e = skf + 8;
  uint64_t c_off = 0, d_off = 0;
  out.reserve(nframes);
  for (uint32_t i = 0; i < nframes; i++) {
    uint32_t cs = mx_read_le32(e);
    uint32_t ds = mx_read_le32(e + 4);
    out.push_back({c_off, cs, d_off, ds});
    c_off += cs;
    d_off += ds;
    e += entry_sz;
  }
  if (c_off != (uint64_t)(skf - blob)) return {};  // table doesn't cover the frames
  return out;
}

// Skip over one frame without decoding payloads; returns (csize, dsize) or
// false on malformed input.
static bool skim_frame(const uint8_t* p, size_t len, uint64_t* csize, uint64_t* dsize) {
  if (len < 4) return false;
  u32 magic = mx_read_le32(p);
  if (magic >= kSkippableMagicMin && magic <= kSkippableMagicMax) {
    if (len < 8) return false;
    uint32_t sz = mx_read_le32(p + 4);
    if (8ull + sz > len) return false;
    *csize = 8ull + sz;
    *dsize = 0;
    return true;
  }
  if (magic != kMagic) return false;
  size_t pos = 4;
  uint8_t fhd = p[pos++];
  u32 fcs_flag = fhd >> 6;
  bool single_seg = (fhd >> 5) & 1;
  bool checksum = (fhd >> 2) & 1;
  u32 dict_flag = fhd & 3;
  if (!single_seg) pos++;
  if (dict_flag) pos += dict_flag == 3 ? 4 : dict_flag;
  uint64_t fcs = 0;
  bool have_fcs = true;
  switch (fcs_flag) {
    case 0:
      if (single_seg) fcs = p[pos++];
      else have_fcs = false;
      break;
    case 1:
      fcs = ((uint64_t)p[pos] | ((uint64_t)p[pos + 1] << 8)) + 256;
      pos += 2;
      break;
    case 2:
      fcs = mx_read_le32(p + pos);
      pos += 4;
      break;
    default:
      fcs = mx_read_le64(p + pos);
      pos += 8;
      break;
  }
  uint64_t dtot = 0;
  while (true) {
    if (pos + 3 > len) return false;
    u32 bh = (u32)p[pos] | ((u32)p[pos + 1] << 8) | ((u32)p[pos + 2] << 16);
    pos += 3;
    bool last = bh & 1;
    u32 btype = (bh >> 1) & 3;
    u32 bsize = bh >> 3;
    if (btype == 0) {
      pos += bsize;
      dtot += bsize;
    } else if (btype == 1) {
      pos += 1;
      dtot += bsize;
    } else if (btype == 2) {
      pos += bsize;
      // compressed block: regenerated size is unknowable without decoding —
      // without a Frame_Content_Size the frame walk cannot report d_size,
      // so reject instead of returning a wrong size
      if (!have_fcs) return false;
    } else {
      return false;
    }
    if (pos > len) return false;
    if (last) break;
  }
  if (checksum) pos += 4;
  if (pos > len) return false;
  *csize = pos;
  *dsize = have_fcs ? fcs : dtot;  // dtot only exact for raw/RLE-only frames
  return true;
}

std::vector<SeekEntry> walk_frames(const uint8_t* blob, size_t len) {
  std::vector<SeekEntry> out;
  uint64_t c_off = 0, d_off = 0;
  while (c_off < len) {
    uint64_t cs = 0, ds = 0;
    if (!skim_frame(blob + c_off, len - c_off, &cs, &ds)) return {};
    u32 magic = mx_read_le32(blob + c_off);
    if (magic == kMagic) {
      out.push_back({c_off, cs, d_off, ds});
      d_off += ds;
    }
    c_off += cs;
  }
  return out;
}

std::vector<uint8_t> build_seek_table(const std::vector<SeekEntry>& entries) {
  std::vector<uint8_t> out(8 + entries.size() * 8 + 9);
  uint8_t* p = out.data();
  mx_write_le32(p, kMagicSkippableSeek);
  mx_write_le32(p + 4, (uint32_t)(entries.size() * 8 + 9));
  p += 8;
  for (const auto& e : entries) {
    mx_write_le32(p, (uint32_t)e.c_size);
    mx_write_le32(p + 4, (uint32_t)e.d_size);
    p += 8;
  }
  mx_write_le32(p, (uint32_t)entries.size());
  p[4] = 0;  // no per-frame checksums (registry digests cover the blob)
  mx_write_le32(p + 5, kSeekTableMagic);
  return out;
}

std::vector<uint8_t> compress_seekable(const uint8_t* src, size_t len, uint32_t frame_raw) {
  if (frame_raw == 0) frame_raw = 128 * 1024;
  size_t nframes = len ? (len + frame_raw - 1) / frame_raw : 1;
  // worst case: raw blocks + frame overhead
  size_t stride = (size_t)frame_raw + 64 + 3 * ((frame_raw + kBlockMax - 1) / kBlockMax);
  std::vector<uint8_t> scratch(nframes * stride);
  std::vector<int64_t> sizes(nframes, 0);

  unsigned nthreads = std::thread::hardware_concurrency();
  if (nthreads == 0) nthreads = 4;
  if (nthreads > nframes) nthreads = (unsigned)nframes;
  std::vector<std::thread> pool;
  std::atomic<size_t> next{0};
  for (unsigned t = 0; t < nthreads; t++) {
    pool.emplace_back([&] {
      std::vector<u32> hash(1u << kHashLog);
      std::vector<Seq> seqs(kBlockMax / 4 + 1);
      EncTables et;
      et.flags = 0;
      enc_tables_init(&et);
      size_t i;
      while ((i = next.fetch_add(1)) < nframes) {
        uint64_t off = (uint64_t)i * frame_raw;
        uint64_t flen = len - off < frame_raw ? len - off : frame_raw;
        sizes[i] = encode_frame(src + off, flen, scratch.data() + i * stride, stride,
                                hash.data(), seqs.data(), &et);
      }
    });
  }
  for (auto& t : pool) t.join();

  std::vector<SeekEntry> entries(nframes);
  uint64_t c_off = 0;
  for (size_t i = 0; i < nframes; i++) {
    if (sizes[i] < 0) throw std::runtime_error("zstd encode failed: frame " + std::to_string(i) +
                                               " rc=" + std::to_string(sizes[i]));
    uint64_t off = (uint64_t)i * frame_raw;
    entries[i] = {c_off, (uint64_t)sizes[i], off,
                  len - off < frame_raw ? len - off : (uint64_t)frame_raw};
    c_off += (uint64_t)sizes[i];
  }
  std::vector<uint8_t> table = build_seek_table(entries);
  std::vector<uint8_t> out(c_off + table.size());
  for (size_t i = 0; i < nframes; i++)
    memcpy(out.data() + entries[i].c_off, scratch.data() + i * stride, entries[i].c_size);
  memcpy(out.data() + c_off, table.data(), table.size());
  return out;
}

uint64_t content_size(const uint8_t* blob, size_t len) {
  auto t = parse_seek_table(blob, len);
  if (t.empty()) t = walk_frames(blob, len);
  uint64_t n = 0;
  for (const auto& e : t) n += e.d_size;
  return n;
}

std::vector<uint8_t> decompress(const uint8_t* blob, size_t len) {
  auto table = parse_seek_table(blob, len);
  if (table.empty()) table = walk_frames(blob, len);
  if (table.empty() && len > 0) throw std::runtime_error("zstd: no frames found");
  uint64_t total = 0;
  for (const auto& e : table) total += e.d_size;
  std::vector<uint8_t> out(total);

  unsigned nthreads = std::thread::hardware_concurrency();
  if (nthreads == 0) nthreads = 4;
  if (nthreads > table.size()) nthreads = (unsigned)table.size();
  std::vector<std::thread> pool;
  std::atomic<size_t> next{0};
  std::atomic<long> err{0};
  for (unsigned t = 0; t < nthreads && t < 256; t++) {
    pool.emplace_back([&] {
      std::vector<u8> lit(kBlockMax);
      auto ctx = std::make_unique<DecCtx>();
      ctx->flags = 0;
      ctx->lit_scratch = lit.data();
      size_t i;
      while ((i = next.fetch_add(1)) < table.size()) {
        const auto& e = table[i];
        i64 n = decode_frame(blob + e.c_off, e.c_size, out.data() + e.d_off, e.d_size,
                             ctx.get(), nullptr);
        if (n < 0 || (uint64_t)n != e.d_size) err.store(n < 0 ? n : MXZ_ERR_CORRUPT);
      }
    });
  }
  for (auto& t : pool) t.join();
  if (err.load()) throw std::runtime_error("zstd decode failed rc=" + std::to_string(err.load()));
  return out;
}

}  // namespace zstdhost
}  // namespace modelx
