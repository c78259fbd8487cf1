// Offline token verification for modelxd.
//
// The reference uses go-oidc against a live issuer (pkg/registry/helper.go:
// 63-96). Issuer *discovery* needs egress, but signature verification does
// not: RS256 ID tokens are verified against a configured JWKS document
// (--oidc-jwks, the same RSA keys the issuer's jwks_uri would serve), with
// optional iss/aud claim checks — this is what go-oidc does after discovery,
// done fully offline. HS256 shared-secret JWTs (--jwt-hs256-secret) and
// static bearer tokens (--auth-tokens) remain. The reference's context-drop
// defect (helper.go:93, username never reached handlers) is NOT replicated:
// the subject is returned to the caller.
#include <openssl/bn.h>
#include <openssl/evp.h>
#include <openssl/param_build.h>

#include <cstring>
#include <ctime>
#include <fstream>
#include <sstream>
#include <string>
#include <vector>

#include "modelx/json.hpp"
#include "modelx/registry.hpp"
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

// ---------------------------------------------------------------- RS256 --

// Load a JWKS document ({"keys":[{"kty":"RSA","kid":...,"n":...,"e":...}]})
// from disk; n/e are base64url big-endian integers (RFC 7518 §6.3).
bool load_jwks_file(const std::string& path, std::vector<JwksKey>* out, std::string* err) {
  std::ifstream f(path, std::ios::binary);
  if (!f) {
    if (err) *err = "cannot read " + path;
    return false;
  }
  std::stringstream ss;
  ss << f.rdbuf();
  try {
    auto doc = json::parse(ss.str());
    for (const auto& k : doc["keys"].items()) {
      if (k["kty"].as_string() != "RSA") continue;
      const std::string& alg = k["alg"].as_string();
      if (!alg.empty() && alg != "RS256") continue;
      JwksKey key;
      key.kid = k["kid"].as_string();
      if (!b64url_decode(k["n"].as_string(), &key.n) ||
          !b64url_decode(k["e"].as_string(), &key.e) || key.n.empty() || key.e.empty()) {
        if (err) *err = "bad n/e in JWKS key " + key.kid;
        return false;
      }
      out->push_back(std::move(key));
    }
  } catch (const std::exception& e) {
    if (err) *err = std::string("JWKS parse: ") + e.what();
    return false;
  }
  if (out->empty()) {
    if (err) *err = "no usable RSA keys in " + path;
    return false;
  }
  return true;
}

// RSASSA-PKCS1-v1_5 / SHA-256 verification of `sig` over `data` with (n, e).
static bool rsa_verify_sha256(const std::string& n, const std::string& e,
                              const std::string& data, const std::string& sig) {
  BIGNUM* bn_n = BN_bin2bn(reinterpret_cast<const unsigned char*>(n.data()),
                           static_cast<int>(n.size()), nullptr);
  BIGNUM* bn_e = BN_bin2bn(reinterpret_cast<const unsigned char*>(e.data()),
                           static_cast<int>(e.size()), nullptr);
  bool ok = false;
  OSSL_PARAM_BLD* bld = OSSL_PARAM_BLD_new();
  OSSL_PARAM* params = nullptr;
  EVP_PKEY_CTX* kctx = nullptr;
  EVP_PKEY* pkey = nullptr;
  EVP_MD_CTX* mctx = nullptr;
  do {
    if (!bn_n || !bn_e || !bld) break;
    if (OSSL_PARAM_BLD_push_BN(bld, "n", bn_n) != 1 ||
        OSSL_PARAM_BLD_push_BN(bld, "e", bn_e) != 1)
      break;
    params = OSSL_PARAM_BLD_to_param(bld);
    if (!params) break;
    kctx = EVP_PKEY_CTX_new_from_name(nullptr, "RSA", nullptr);
    if (!kctx || EVP_PKEY_fromdata_init(kctx) != 1 ||
        EVP_PKEY_fromdata(kctx, &pkey, EVP_PKEY_PUBLIC_KEY, params) != 1)
      break;
    mctx = EVP_MD_CTX_new();
    if (!mctx ||
        EVP_DigestVerifyInit(mctx, nullptr, EVP_sha256(), nullptr, pkey) != 1)
      break;
    ok = EVP_DigestVerify(mctx, reinterpret_cast<const unsigned char*>(sig.data()),
                          sig.size(), reinterpret_cast<const unsigned char*>(data.data()),
                          data.size()) == 1;
  } while (false);
  if (mctx) EVP_MD_CTX_free(mctx);
  if (pkey) EVP_PKEY_free(pkey);
  if (kctx) EVP_PKEY_CTX_free(kctx);
  if (params) OSSL_PARAM_free(params);
  if (bld) OSSL_PARAM_BLD_free(bld);
  if (bn_n) BN_free(bn_n);
  if (bn_e) BN_free(bn_e);
  return ok;
}

bool verify_jwt_rs256(const std::string& token, const std::vector<JwksKey>& keys,
                      const std::string& issuer, const std::string& audience,
                      std::string* subject) {
  size_t d1 = token.find('.');
  if (d1 == std::string::npos) return false;
  size_t d2 = token.find('.', d1 + 1);
  if (d2 == std::string::npos) return false;
  std::string signing_input = token.substr(0, d2);
  std::string sig;
  if (!b64url_decode(token.substr(d2 + 1), &sig) || sig.empty()) return false;
  std::string header_json, payload_json;
  if (!b64url_decode(token.substr(0, d1), &header_json)) return false;
  if (!b64url_decode(token.substr(d1 + 1, d2 - d1 - 1), &payload_json)) return false;
  try {
    auto header = json::parse(header_json);
    if (header["alg"].as_string() != "RS256") return false;
    const std::string& kid = header["kid"].as_string();
    bool sig_ok = false;
    for (const auto& k : keys) {
      if (!kid.empty() && !k.kid.empty() && k.kid != kid) continue;
      if (rsa_verify_sha256(k.n, k.e, signing_input, sig)) {
        sig_ok = true;
        break;
      }
    }
    if (!sig_ok) return false;
    auto payload = json::parse(payload_json);
    // exp is REQUIRED (OIDC Core §2 mandates it; accepting tokens without
    // one would make any leaked token eternal)
    int64_t exp = payload["exp"].as_int(0);
    int64_t now = static_cast<int64_t>(time(nullptr));
    if (exp <= 0 || now > exp) return false;
    int64_t nbf = payload["nbf"].as_int(0);
    if (nbf > 0 && now + 60 < nbf) return false;  // 60 s clock-skew allowance
    if (!issuer.empty() && payload["iss"].as_string() != issuer) return false;
    if (!audience.empty()) {
      const auto& aud = payload["aud"];
      bool aud_ok = false;
      if (aud.is_string()) {
        aud_ok = aud.as_string() == audience;
      } else if (aud.is_array()) {
        for (const auto& a : aud.items())
          if (a.as_string() == audience) aud_ok = true;
      }
      if (!aud_ok) return false;
    }
    if (subject) *subject = payload["sub"].as_string();
  } catch (...) {
    return false;
  }
  return true;
}

}  // namespace registry
}  // namespace modelx
