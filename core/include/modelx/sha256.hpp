// CPU SHA-256 via OpenSSL EVP (SHA-NI accelerated, ~2 GB/s/core).
// The GPU path (core/hip/sha256.hip) replaces this on the data plane;
// this is the control-plane / fallback / test-reference implementation.
#pragma once

#include <cstdint>
#include <string>

#include <openssl/evp.h>
#include <openssl/hmac.h>

namespace modelx {

inline std::string hex_encode(const unsigned char* data, size_t n) {
  static const char* hexd = "0123456789abcdef";
  std::string out(n * 2, '0');
  for (size_t i = 0; i < n; i++) {
    out[2 * i] = hexd[data[i] >> 4];
    out[2 * i + 1] = hexd[data[i] & 15];
  }
  return out;
}

class Sha256 {
 public:
  Sha256() : ctx_(EVP_MD_CTX_new()) { EVP_DigestInit_ex(ctx_, EVP_sha256(), nullptr); }
  ~Sha256() { EVP_MD_CTX_free(ctx_); }
  Sha256(const Sha256&) = delete;
  Sha256& operator=(const Sha256&) = delete;

  void update(const void* data, size_t n) { EVP_DigestUpdate(ctx_, data, n); }
  void final(unsigned char out[32]) {
    unsigned int len = 32;
    EVP_DigestFinal_ex(ctx_, out, &len);
  }
  std::string final_hex() {
    unsigned char d[32];
    final(d);
    return hex_encode(d, 32);
  }
  void reset() { EVP_DigestInit_ex(ctx_, EVP_sha256(), nullptr); }

 private:
  EVP_MD_CTX* ctx_;
};

inline std::string sha256_hex(const void* data, size_t n) {
  Sha256 h;
  h.update(data, n);
  return h.final_hex();
}

inline std::string sha256_hex(const std::string& s) { return sha256_hex(s.data(), s.size()); }

inline void hmac_sha256(const void* key, size_t keylen, const void* data, size_t datalen,
                        unsigned char out[32]) {
  unsigned int outlen = 32;
  HMAC(EVP_sha256(), key, static_cast<int>(keylen), static_cast<const unsigned char*>(data),
       datalen, out, &outlen);
}

}  // namespace modelx
