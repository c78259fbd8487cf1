// Host-side zstd container helpers: seekable multi-frame blobs.
//
// Blob layout = N standard zstd frames (each <= frame_raw decompressed bytes)
// concatenated, followed by one skippable frame holding the
// zstd-seekable-format seek table. Stock `zstd -d` decodes the blob (it
// ignores skippable frames); the GPU kernels use the table to decode all
// frames in parallel (one workgroup per frame).
#pragma once

#include <cstddef>
#include <cstdint>
#include <vector>

namespace modelx {
namespace zstdhost {

struct SeekEntry {
  uint64_t c_off, c_size;  // compressed span within the blob
  uint64_t d_off, d_size;  // decompressed span
};

// Parse the seek table from a full blob. Returns empty vector if the blob
// has no seekable footer (caller falls back to sequential frame walking).
std::vector<SeekEntry> parse_seek_table(const uint8_t* blob, size_t len);

// Walk frames sequentially (no seek table needed); returns entries or empty
// on malformed input. Skips skippable frames.
std::vector<SeekEntry> walk_frames(const uint8_t* blob, size_t len);

// Serialize the seekable skippable frame for the given entries.
std::vector<uint8_t> build_seek_table(const std::vector<SeekEntry>& entries);

// CPU compress src into a seekable multi-frame blob (threads over frames).
std::vector<uint8_t> compress_seekable(const uint8_t* src, size_t len, uint32_t frame_raw);

// CPU decompress a seekable (or plain multi-frame) blob.
// Returns decompressed bytes; throws std::runtime_error on corrupt input.
std::vector<uint8_t> decompress(const uint8_t* blob, size_t len);

// Total decompressed size from the table (or walk).
uint64_t content_size(const uint8_t* blob, size_t len);

}  // namespace zstdhost
}  // namespace modelx
