// HTTP registry server — route table + handlers
// (reference: pkg/registry/route.go:15-51, registry.go:18-271).
#pragma once

#include <atomic>
#include <memory>
#include <string>
#include <vector>

#include "modelx/http.hpp"
#include "modelx/json.hpp"
#include "modelx/store.hpp"

namespace modelx {
namespace registry {

// One RSA public key from a JWKS document (RFC 7517); n/e raw big-endian.
struct JwksKey {
  std::string kid;
  std::string n;
  std::string e;
};

struct AuthConfig {
  // Static bearer tokens (comma-separated via --auth-tokens); empty = open.
  std::vector<std::string> tokens;
  // HS256 JWT shared secret (offline OIDC-style verification); empty = off.
  std::string jwt_hs256_secret;
  // Offline OIDC: RS256 ID tokens verified against these JWKS keys
  // (--oidc-jwks FILE; reference helper.go:63-96 minus the egress-needing
  // issuer discovery). Empty = off.
  std::vector<JwksKey> jwks;
  std::string oidc_issuer;    // checked against `iss` when non-empty
  std::string oidc_audience;  // checked against `aud` when non-empty
  bool enabled() const {
    return !tokens.empty() || !jwt_hs256_secret.empty() || !jwks.empty();
  }
};

bool load_jwks_file(const std::string& path, std::vector<JwksKey>* out, std::string* err);
bool verify_jwt_hs256(const std::string& token, const std::string& secret,
                      std::string* subject);
bool verify_jwt_rs256(const std::string& token, const std::vector<JwksKey>& keys,
                      const std::string& issuer, const std::string& audience,
                      std::string* subject);

// Prometheus-style counters (reference has no metrics endpoint — SURVEY.md
// §5 observability gap; this is the /metrics the new framework adds).
struct Metrics {
  std::atomic<uint64_t> requests_total{0};
  std::atomic<uint64_t> blob_bytes_in{0};
  std::atomic<uint64_t> blob_bytes_out{0};
  std::atomic<uint64_t> presign_upload_total{0};
  std::atomic<uint64_t> presign_download_total{0};
  std::atomic<uint64_t> manifests_put_total{0};
  std::atomic<uint64_t> gc_blobs_removed_total{0};
  std::atomic<uint64_t> errors_total{0};
  std::string render() const;
};

class Registry {
 public:
  Registry(std::shared_ptr<store::RegistryStore> s, AuthConfig auth = {})
      : store_(std::move(s)), auth_(std::move(auth)) {}

  // the single mux entry point (LoggingFilter wraps this in main)
  void handle(http::Request& req, http::ResponseWriter& w);

 private:
  bool authorize(http::Request& req, http::ResponseWriter& w);

  void get_global_index(http::Request&, http::ResponseWriter&);
  void get_index(http::Request&, http::ResponseWriter&, const std::string& name);
  void delete_index(http::Request&, http::ResponseWriter&, const std::string& name);
  void get_manifest(http::Request&, http::ResponseWriter&, const std::string& name,
                    const std::string& ref);
  void put_manifest(http::Request&, http::ResponseWriter&, const std::string& name,
                    const std::string& ref);
  void delete_manifest(http::Request&, http::ResponseWriter&, const std::string& name,
                       const std::string& ref);
  void head_blob(http::Request&, http::ResponseWriter&, const std::string& name,
                 const std::string& digest);
  void get_blob(http::Request&, http::ResponseWriter&, const std::string& name,
                const std::string& digest);
  void put_blob(http::Request&, http::ResponseWriter&, const std::string& name,
                const std::string& digest);
  void blob_location(http::Request&, http::ResponseWriter&, const std::string& name,
                     const std::string& digest, const std::string& purpose);
  void pull_plans(http::Request&, http::ResponseWriter&, const std::string& name);
  bool build_pull_plan(const std::string& name, const std::string& ref, json::Value* out);
  void pull_plan(http::Request&, http::ResponseWriter&, const std::string& name,
                 const std::string& ref);
  void garbage_collect(http::Request&, http::ResponseWriter&, const std::string& name);

  std::shared_ptr<store::RegistryStore> store_;
  AuthConfig auth_;
  Metrics metrics_;
};

void response_error(http::ResponseWriter& w, const wire::ErrorInfo& e);

}  // namespace registry
}  // namespace modelx
