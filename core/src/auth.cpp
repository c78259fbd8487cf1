// Offline token verification for modelxd.
//
// The reference uses go-oidc against a live issuer (pkg/registry/helper.go:
// 63-96) — this environment has no egress, so the equivalent here is HS256
// JWT verification against a shared secret (--jwt-hs256-secret) plus static
// bearer tokens (--auth-tokens). The reference's context-drop defect
// (helper.go:93, username never reached handlers) is NOT replicated: the
// subject is returned to the caller.
#include <cstring>
#include <ctime>
#include <string>

#include "modelx/json.hpp"
#include "modelx/sha256.hpp"

namespace modelx {
namespace registry {

static bool b64url_decode(const std::string& in, std::string* out) {
  static int8_t table[256];
  static bool init = false;
  if (!init) {
    memset(table, -1, sizeof table);
    const char* alpha = "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789-_";
    for (int i = 0; i < 64; i++) table[static_cast<unsigned char>(alpha[i])] = static_cast<int8_t>(i);
    table[static_cast<unsigned char>('+')] = 62;
    table[static_cast<unsigned char>('/')] = 63;
    init = true;
  }
  out->clear();
  uint32_t acc = 0;
  int bits = 0;
  for (char c : in) {
    if (c == '=') break;
    int8_t v = table[static_cast<unsigned char>(c)];
    if (v < 0) return false;
    acc = (acc << 6) | static_cast<uint32_t>(v);
    bits += 6;
    if (bits >= 8) {
      bits -= 8;
      out->push_back(static_cast<char>((acc >> bits) & 0xFF));
    }
  }
  return true;
}

bool verify_jwt_hs256(const std::string& token, const std::string& secret, std::string* subject) {
  size_t d1 = token.find('.');
  if (d1 == std::string::npos) return false;
  size_t d2 = token.find('.', d1 + 1);
  if (d2 == std::string::npos) return false;
  std::string signing_input = token.substr(0, d2);
  std::string sig_b64 = token.substr(d2 + 1);
  std::string sig;
  if (!b64url_decode(sig_b64, &sig) || sig.size() != 32) return false;
  unsigned char expect[32];
  hmac_sha256(secret.data(), secret.size(), signing_input.data(), signing_input.size(), expect);
  // constant-time compare
  unsigned char diff = 0;
  for (int i = 0; i < 32; i++) diff |= expect[i] ^ static_cast<unsigned char>(sig[i]);
  if (diff != 0) return false;
  std::string header_json, payload_json;
  if (!b64url_decode(token.substr(0, d1), &header_json)) return false;
  if (!b64url_decode(token.substr(d1 + 1, d2 - d1 - 1), &payload_json)) return false;
  try {
    auto header = json::parse(header_json);
    if (header["alg"].as_string() != "HS256") return false;
    auto payload = json::parse(payload_json);
    int64_t exp = payload["exp"].as_int(0);
    if (exp > 0 && time(nullptr) > exp) return false;
    if (subject) *subject = payload["sub"].as_string();
  } catch (...) {
    return false;
  }
  return true;
}

}  // namespace registry
}  // namespace modelx
