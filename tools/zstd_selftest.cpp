// Dev-loop self-test for the zstd codec core (CPU path) against libzstd
// (dlopen'd — no headers in this image). Exercised properly from
// tests/test_zstd.py; this binary exists for fast compile-run iteration:
//
//   g++ -O2 -std=c++17 -Icore/include core/src/zstd_cpu.cpp \
//       tools/zstd_selftest.cpp -o /tmp/zstd_selftest -ldl -pthread
#include <dlfcn.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <random>
#include <vector>

#include "modelx/zstd_host.hpp"

typedef size_t (*fn_bound)(size_t);
typedef size_t (*fn_compress)(void*, size_t, const void*, size_t, int);
typedef size_t (*fn_decompress)(void*, size_t, const void*, size_t);
typedef unsigned (*fn_iserr)(size_t);

struct LibZstd {
  fn_bound bound;
  fn_compress compress;
  fn_decompress decompress;
  fn_iserr iserr;
  LibZstd() {
    void* h = dlopen("libzstd.so.1", RTLD_NOW);
    if (!h) {
      fprintf(stderr, "no libzstd\n");
      exit(2);
    }
    bound = (fn_bound)dlsym(h, "ZSTD_compressBound");
    compress = (fn_compress)dlsym(h, "ZSTD_compress");
    decompress = (fn_decompress)dlsym(h, "ZSTD_decompress");
    iserr = (fn_iserr)dlsym(h, "ZSTD_isError");
  }
};

static std::vector<uint8_t> gen(size_t n, int mode, std::mt19937& rng) {
  std::vector<uint8_t> v(n);
  switch (mode) {
    case 0:  // random (incompressible)
      for (auto& b : v) b = (uint8_t)rng();
      break;
    case 1:  // zeros
      break;
    case 2:  // repeated text
      for (size_t i = 0; i < n; i++) v[i] = "the quick brown fox jumps over "[i % 31];
      break;
    case 3: {  // random with repeats
      size_t i = 0;
      while (i < n) {
        if (rng() % 3 && i > 64) {
          size_t off = 1 + rng() % std::min<size_t>(i, 60000);
          size_t len = 4 + rng() % 200;
          for (size_t k = 0; k < len && i < n; k++, i++) v[i] = v[i - off];
        } else {
          size_t len = 1 + rng() % 50;
          for (size_t k = 0; k < len && i < n; k++, i++) v[i] = (uint8_t)rng();
        }
      }
      break;
    }
    case 4:  // low entropy bytes (exercises huffman in libzstd output)
      for (auto& b : v) b = "aab"[rng() % 3];
      break;
  }
  return v;
}

int main(int argc, char** argv) {
  LibZstd z;
  std::mt19937 rng(42);
  int fails = 0;

  // A) our encoder -> libzstd decoder
  for (int mode = 0; mode <= 4; mode++) {
    for (size_t n : {0ul, 1ul, 5ul, 100ul, 4096ul, 131072ul, 131073ul, 400000ul, 1500000ul}) {
      auto data = gen(n, mode, rng);
      auto blob = modelx::zstdhost::compress_seekable(data.data(), n, 128 * 1024);
      std::vector<uint8_t> back(n + 16);
      size_t m = z.decompress(back.data(), back.size(), blob.data(), blob.size());
      if (z.iserr(m) || m != n || memcmp(back.data(), data.data(), n) != 0) {
        printf("FAIL enc mode=%d n=%zu libzstd rc err=%u m=%zu\n", mode, n, z.iserr(m), m);
        fails++;
        continue;
      }
      // B) our encoder -> our decoder
      auto ours = modelx::zstdhost::decompress(blob.data(), blob.size());
      if (ours.size() != n || memcmp(ours.data(), data.data(), n) != 0) {
        printf("FAIL roundtrip mode=%d n=%zu (got %zu)\n", mode, n, ours.size());
        fails++;
        continue;
      }
      if (mode == 2 && n == 400000)
        printf("ratio mode=%d n=%zu: %zu -> %zu\n", mode, n, n, blob.size());
    }
  }

  // C) libzstd encoder -> our decoder (per-frame, all levels incl. huffman)
  for (int mode = 0; mode <= 4; mode++) {
    for (int level : {1, 3, 9, 19}) {
      for (size_t n : {1ul, 100ul, 5000ul, 65536ul, 131072ul}) {
        auto data = gen(n, mode, rng);
        std::vector<uint8_t> comp(z.bound(n));
        size_t cn = z.compress(comp.data(), comp.size(), data.data(), n, level);
        if (z.iserr(cn)) {
          printf("libzstd compress err\n");
          fails++;
          continue;
        }
        comp.resize(cn);
        try {
          auto ours = modelx::zstdhost::decompress(comp.data(), comp.size());
          if (ours.size() != n || memcmp(ours.data(), data.data(), n) != 0) {
            printf("FAIL dec mode=%d level=%d n=%zu (got %zu)\n", mode, level, n, ours.size());
            fails++;
          }
        } catch (const std::exception& e) {
          printf("FAIL dec mode=%d level=%d n=%zu: %s\n", mode, level, n, e.what());
          fails++;
        }
      }
    }
  }

  printf(fails ? "SELFTEST FAILED: %d\n" : "selftest ok (%d)\n", fails);
  return fails ? 1 : 0;
}
