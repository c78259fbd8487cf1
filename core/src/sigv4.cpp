#include "modelx/sigv4.hpp"

#include <cstdio>
#include <cstring>
#include <ctime>

#include "modelx/http.hpp"
#include "modelx/sha256.hpp"

namespace modelx {
namespace sigv4 {

std::string amz_date_now() {
  time_t now = time(nullptr);
  struct tm tmv;
  gmtime_r(&now, &tmv);
  char buf[32];
  strftime(buf, sizeof buf, "%Y%m%dT%H%M%SZ", &tmv);
  return buf;
}

static std::string scope_date(const std::string& amz_date) { return amz_date.substr(0, 8); }

static std::string credential_scope(const Credentials& cred, const std::string& amz_date) {
  return scope_date(amz_date) + "/" + cred.region + "/" + cred.service + "/aws4_request";
}

static void derive_key(const Credentials& cred, const std::string& amz_date,
                       unsigned char out[32]) {
  std::string k0 = "AWS4" + cred.secret_key;
  unsigned char k1[32], k2[32], k3[32];
  std::string date = scope_date(amz_date);
  hmac_sha256(k0.data(), k0.size(), date.data(), date.size(), k1);
  hmac_sha256(k1, 32, cred.region.data(), cred.region.size(), k2);
  hmac_sha256(k2, 32, cred.service.data(), cred.service.size(), k3);
  static const char* terminator = "aws4_request";
  hmac_sha256(k3, 32, terminator, strlen(terminator), out);
}

static std::string canonical_query(const std::map<std::string, std::string>& query) {
  // keys already encoded; std::map sorts them
  std::string out;
  for (auto& kv : query) {
    if (!out.empty()) out.push_back('&');
    out += kv.first + "=" + kv.second;
  }
  return out;
}

static std::string build_string_to_sign(const std::string& canonical_request,
                                        const Credentials& cred, const std::string& amz_date) {
  return "AWS4-HMAC-SHA256\n" + amz_date + "\n" + credential_scope(cred, amz_date) + "\n" +
         sha256_hex(canonical_request);
}

static std::string hmac_hex(const Credentials& cred, const std::string& amz_date,
                            const std::string& string_to_sign) {
  unsigned char key[32], sig[32];
  derive_key(cred, amz_date, key);
  hmac_sha256(key, 32, string_to_sign.data(), string_to_sign.size(), sig);
  return hex_encode(sig, 32);
}

std::string sign_authorization(RequestToSign& req, const Credentials& cred,
                               const std::string& amz_date) {
  req.headers["x-amz-date"] = amz_date;
  req.headers["x-amz-content-sha256"] = req.payload_hash;
  std::string signed_headers, canonical_headers;
  for (auto& kv : req.headers) {  // lowercase keys expected; map sorts
    if (!signed_headers.empty()) signed_headers.push_back(';');
    signed_headers += kv.first;
    canonical_headers += kv.first + ":" + kv.second + "\n";
  }
  std::string canonical = req.method + "\n" + req.path + "\n" + canonical_query(req.query) + "\n" +
                          canonical_headers + "\n" + signed_headers + "\n" + req.payload_hash;
  std::string sts = build_string_to_sign(canonical, cred, amz_date);
  std::string signature = hmac_hex(cred, amz_date, sts);
  return "AWS4-HMAC-SHA256 Credential=" + cred.access_key + "/" + credential_scope(cred, amz_date) +
         ", SignedHeaders=" + signed_headers + ", Signature=" + signature;
}

std::string presign_query(const RequestToSign& req, const Credentials& cred,
                          const std::string& amz_date, int expires_seconds) {
  std::map<std::string, std::string> q = req.query;
  q["X-Amz-Algorithm"] = "AWS4-HMAC-SHA256";
  q["X-Amz-Credential"] =
      http::url_encode_query(cred.access_key + "/" + credential_scope(cred, amz_date));
  q["X-Amz-Date"] = amz_date;
  q["X-Amz-Expires"] = std::to_string(expires_seconds);
  q["X-Amz-SignedHeaders"] = "host";
  auto host_it = req.headers.find("host");
  std::string host = host_it != req.headers.end() ? host_it->second : "";
  std::string canonical = req.method + "\n" + req.path + "\n" + canonical_query(q) + "\n" +
                          "host:" + host + "\n" + "\nhost\nUNSIGNED-PAYLOAD";
  std::string sts = build_string_to_sign(canonical, cred, amz_date);
  std::string signature = hmac_hex(cred, amz_date, sts);
  q["X-Amz-Signature"] = signature;
  return canonical_query(q);
}

bool verify_presigned(const std::string& method, const std::string& raw_path,
                      const std::map<std::string, std::string>& raw_query,
                      const std::string& host_header, const Credentials& cred, long now_epoch,
                      std::string* error) {
  auto get = [&](const char* k) -> std::string {
    auto it = raw_query.find(k);
    return it == raw_query.end() ? "" : it->second;
  };
  std::string given_sig = get("X-Amz-Signature");
  std::string amz_date = get("X-Amz-Date");
  std::string expires = get("X-Amz-Expires");
  if (given_sig.empty() || amz_date.empty()) {
    if (error) *error = "missing X-Amz-Signature/X-Amz-Date";
    return false;
  }
  // expiry check
  struct tm tmv{};
  if (strptime(amz_date.c_str(), "%Y%m%dT%H%M%SZ", &tmv) != nullptr) {
    long ts = timegm(&tmv);
    long exp = atol(expires.c_str());
    if (exp > 0 && now_epoch > ts + exp) {
      if (error) *error = "presigned URL expired";
      return false;
    }
  }
  // recompute signature over the query minus X-Amz-Signature
  std::map<std::string, std::string> q;
  for (auto& kv : raw_query) {
    if (kv.first == "X-Amz-Signature") continue;
    q[http::url_encode_query(kv.first)] = http::url_encode_query(kv.second);
  }
  std::string canonical = method + "\n" + raw_path + "\n" + canonical_query(q) + "\n" +
                          "host:" + host_header + "\n" + "\nhost\nUNSIGNED-PAYLOAD";
  std::string sts = build_string_to_sign(canonical, cred, amz_date);
  std::string expect = hmac_hex(cred, amz_date, sts);
  if (expect != given_sig) {
    if (error) *error = "signature mismatch";
    return false;
  }
  return true;
}

}  // namespace sigv4
}  // namespace modelx
