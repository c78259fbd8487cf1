#include "modelx/registry.hpp"

#include <cstring>
#include <vector>

namespace modelx {
namespace registry {

static constexpr int64_t kMaxManifestBytes = 1 << 20;  // helper.go:19 MaxBytesRead

void response_error(http::ResponseWriter& w, const wire::ErrorInfo& e) {
  w.write_json(e.http_status, e.to_json_body());
}

std::string Metrics::render() const {
  auto line = [](const char* name, uint64_t v) {
    return "# TYPE " + std::string(name) + " counter\n" + name + " " + std::to_string(v) + "\n";
  };
  std::string out;
  out += line("modelx_requests_total", requests_total.load());
  out += line("modelx_blob_bytes_in_total", blob_bytes_in.load());
  out += line("modelx_blob_bytes_out_total", blob_bytes_out.load());
  out += line("modelx_presign_upload_total", presign_upload_total.load());
  out += line("modelx_presign_download_total", presign_download_total.load());
  out += line("modelx_manifests_put_total", manifests_put_total.load());
  out += line("modelx_gc_blobs_removed_total", gc_blobs_removed_total.load());
  out += line("modelx_errors_total", errors_total.load());
  return out;
}

static void response_ok(http::ResponseWriter& w, const json::Value& v) {
  // reference ResponseOK uses json.Encoder → trailing newline
  w.write_json(200, v.dump() + "\n");
}

// split path into segments (already url-decoded)
static std::vector<std::string> segments(const std::string& path) {
  std::vector<std::string> out;
  size_t pos = 1;  // skip leading /
  while (pos <= path.size()) {
    size_t slash = path.find('/', pos);
    if (slash == std::string::npos) slash = path.size();
    if (slash > pos) out.push_back(path.substr(pos, slash - pos));
    pos = slash + 1;
  }
  return out;
}

// NameRegexp component check (route.go:10): alnum groups joined by [._-]
static bool valid_name_component(const std::string& s) {
  if (s.empty()) return false;
  bool prev_sep = true;
  for (char c : s) {
    bool alnum = (c >= 'a' && c <= 'z') || (c >= 'A' && c <= 'Z') || (c >= '0' && c <= '9');
    if (alnum) {
      prev_sep = false;
    } else if (c == '.' || c == '_' || c == '-') {
      if (prev_sep) return false;
      prev_sep = true;
    } else {
      return false;
    }
  }
  return !prev_sep;
}

// ReferenceRegexp (route.go:11)
static bool valid_reference(const std::string& s) {
  if (s.empty() || s.size() > 128) return false;
  char c0 = s[0];
  bool ok0 = (c0 >= 'a' && c0 <= 'z') || (c0 >= 'A' && c0 <= 'Z') || (c0 >= '0' && c0 <= '9') ||
             c0 == '_';
  if (!ok0) return false;
  for (char c : s) {
    bool ok = (c >= 'a' && c <= 'z') || (c >= 'A' && c <= 'Z') || (c >= '0' && c <= '9') ||
              c == '.' || c == '_' || c == '-';
    if (!ok) return false;
  }
  return true;
}

bool Registry::authorize(http::Request& req, http::ResponseWriter& w) {
  if (!auth_.enabled()) return true;
  std::string token;
  auto it = req.headers.find("Authorization");
  if (it != req.headers.end() && it->second.rfind("Bearer ", 0) == 0)
    token = it->second.substr(7);
  if (token.empty()) {
    // `?token=` fallback (reference: pkg/registry/helper.go:69-74)
    auto q = req.query.find("token");
    if (q != req.query.end()) token = q->second;
  }
  if (token.empty()) {
    response_error(w, wire::ErrorInfo{401, "UNAUTHORIZED", "missing bearer token", ""});
    return false;
  }
  for (auto& t : auth_.tokens)
    if (t == token) return true;
  if (!auth_.jwt_hs256_secret.empty()) {
    std::string subject;
    if (verify_jwt_hs256(token, auth_.jwt_hs256_secret, &subject)) return true;
  }
  if (!auth_.jwks.empty()) {
    std::string subject;
    if (verify_jwt_rs256(token, auth_.jwks, auth_.oidc_issuer, auth_.oidc_audience,
                         &subject))
      return true;
  }
  response_error(w, wire::ErrorInfo{401, "UNAUTHORIZED", "invalid token", ""});
  return false;
}

void Registry::handle(http::Request& req, http::ResponseWriter& w) {
  const std::string& m = req.method;
  metrics_.requests_total.fetch_add(1);
  if (req.path == "/healthz" && m == "GET") {
    w.write_all(200, "ok");
    return;
  }
  if (req.path == "/metrics" && m == "GET") {
    w.write_all(200, metrics_.render(), "text/plain; version=0.0.4");
    return;
  }
  if (req.path == "/" || req.path.empty()) {
    if (m == "GET") {
      if (!authorize(req, w)) return;
      get_global_index(req, w);
      return;
    }
    w.write_all(405, "method not allowed");
    return;
  }
  auto seg = segments(req.path);
  // all repository routes need name = <project>/<name> (route.go:10 NameRegexp)
  if (seg.size() < 3 || !valid_name_component(seg[0]) || !valid_name_component(seg[1])) {
    response_error(w, wire::ErrorInfo{404, "NAME_INVALID", "invalid repository path", req.path});
    return;
  }
  if (!authorize(req, w)) return;
  std::string name = seg[0] + "/" + seg[1];

  if (seg.size() == 3 && seg[2] == "garbage-collect" && m == "POST") {
    garbage_collect(req, w, name);
    return;
  }
  if (seg.size() == 3 && seg[2] == "pull-plans" && m == "POST") {
    pull_plans(req, w, name);
    return;
  }
  if (seg.size() == 3 && seg[2] == "index") {
    if (m == "GET") return get_index(req, w, name);
    if (m == "DELETE") return delete_index(req, w, name);
    w.write_all(405, "method not allowed");
    return;
  }
  if (seg.size() == 5 && seg[2] == "manifests" && seg[4] == "pull-plan" && m == "GET") {
    const std::string& ref = seg[3];
    if (!valid_reference(ref)) {
      response_error(w, wire::ErrorInfo{404, "NAME_INVALID", "invalid reference", ref});
      return;
    }
    return pull_plan(req, w, name, ref);
  }
  if (seg.size() == 4 && seg[2] == "manifests") {
    const std::string& ref = seg[3];
    if (!valid_reference(ref)) {
      response_error(w, wire::ErrorInfo{404, "NAME_INVALID", "invalid reference", ref});
      return;
    }
    if (m == "GET") return get_manifest(req, w, name, ref);
    if (m == "PUT") return put_manifest(req, w, name, ref);
    if (m == "DELETE") return delete_manifest(req, w, name, ref);
    w.write_all(405, "method not allowed");
    return;
  }
  if (seg.size() == 4 && seg[2] == "blobs") {
    const std::string& digest = seg[3];
    if (!wire::digest_valid(digest)) {
      response_error(w, wire::ErrorInfo{400, "DIGEST_INVALID", "digest invalid: " + digest, ""});
      return;
    }
    if (m == "HEAD") return head_blob(req, w, name, digest);
    if (m == "GET") return get_blob(req, w, name, digest);
    if (m == "PUT") return put_blob(req, w, name, digest);
    w.write_all(405, "method not allowed");
    return;
  }
  if (seg.size() == 6 && seg[2] == "blobs" && seg[4] == "locations" && m == "GET") {
    const std::string& digest = seg[3];
    if (!wire::digest_valid(digest)) {
      response_error(w, wire::ErrorInfo{400, "DIGEST_INVALID", "digest invalid: " + digest, ""});
      return;
    }
    return blob_location(req, w, name, digest, seg[5]);
  }
  response_error(w, wire::ErrorInfo{404, "UNKNOWN", "no such route", req.path});
}

void Registry::get_global_index(http::Request& req, http::ResponseWriter& w) {
  wire::Index index;
  std::string search;
  auto it = req.query.find("search");
  if (it != req.query.end()) search = it->second;
  if (!store_->GetGlobalIndex(search, &index)) {
    response_ok(w, wire::Index{}.to_json());
    return;
  }
  response_ok(w, index.to_json());
}

void Registry::get_index(http::Request& req, http::ResponseWriter& w, const std::string& name) {
  wire::Index index;
  std::string search;
  auto it = req.query.find("search");
  if (it != req.query.end()) search = it->second;
  if (!store_->GetIndex(name, search, &index)) {
    response_error(w, wire::ErrorInfo{404, "INDEX_UNKNOWN", "index: " + name + " not found", ""});
    return;
  }
  response_ok(w, index.to_json());
}

void Registry::delete_index(http::Request& req, http::ResponseWriter& w,
                            const std::string& name) {
  if (!store_->RemoveIndex(name)) {
    response_error(w, wire::ErrorInfo{404, "INDEX_UNKNOWN", "index: " + name + " not found", ""});
    return;
  }
  response_ok(w, json::Value("ok"));
}

void Registry::get_manifest(http::Request& req, http::ResponseWriter& w, const std::string& name,
                            const std::string& ref) {
  wire::Manifest manifest;
  if (!store_->GetManifest(name, ref, &manifest)) {
    response_error(w,
                   wire::ErrorInfo{404, "MANIFEST_UNKNOWN", "manifest: " + ref + " not found", ""});
    return;
  }
  response_ok(w, manifest.to_json());
}

void Registry::put_manifest(http::Request& req, http::ResponseWriter& w, const std::string& name,
                            const std::string& ref) {
  if (req.content_length > kMaxManifestBytes) {
    response_error(w, wire::ErrorInfo{413, "MANIFEST_INVALID", "manifest too large", ""});
    return;
  }
  std::string body;
  try {
    body = req.read_body_all(kMaxManifestBytes);
  } catch (const std::exception& e) {
    response_error(w, wire::ErrorInfo{400, "MANIFEST_INVALID", e.what(), ""});
    return;
  }
  wire::Manifest manifest;
  try {
    manifest = wire::Manifest::from_json(json::parse(body));
  } catch (const std::exception& e) {
    response_error(w, wire::ErrorInfo{400, "MANIFEST_INVALID", e.what(), ""});
    return;
  }
  std::string content_type;
  auto it = req.headers.find("Content-Type");
  if (it != req.headers.end()) content_type = it->second;
  std::string err;
  if (!store_->PutManifest(name, ref, content_type, manifest, &err)) {
    metrics_.errors_total.fetch_add(1);
    response_error(w, wire::ErrorInfo{500, "INTERNAL", err, ""});
    return;
  }
  metrics_.manifests_put_total.fetch_add(1);
  w.write_all(201, "");  // registry.go:106 StatusCreated
}

void Registry::delete_manifest(http::Request& req, http::ResponseWriter& w,
                               const std::string& name, const std::string& ref) {
  if (!store_->ExistsManifest(name, ref)) {
    response_error(w,
                   wire::ErrorInfo{404, "MANIFEST_UNKNOWN", "manifest: " + ref + " not found", ""});
    return;
  }
  if (!store_->DeleteManifest(name, ref)) {
    response_error(w, wire::ErrorInfo{500, "INTERNAL", "delete failed", ""});
    return;
  }
  w.write_all(202, "");  // registry.go:119 StatusAccepted
}

void Registry::head_blob(http::Request& req, http::ResponseWriter& w, const std::string& name,
                         const std::string& digest) {
  if (store_->ExistsBlob(name, digest))
    w.write_all(200, "");
  else
    w.write_all(404, "");
}

void Registry::get_blob(http::Request& req, http::ResponseWriter& w, const std::string& name,
                        const std::string& digest) {
  store::FileMeta meta;
  auto reader = store_->GetBlob(name, digest, &meta);
  if (!reader) {
    response_error(w, wire::ErrorInfo{404, "BLOB_UNKNOWN", "blob: " + digest + " not found", ""});
    return;
  }
  w.set_header("Content-Type",
               meta.content_type.empty() ? "application/octet-stream" : meta.content_type);
  metrics_.blob_bytes_out.fetch_add(static_cast<uint64_t>(meta.size));
  w.begin(200, meta.size);
  int fd = reader->sendfile_fd();
  if (fd >= 0) {
    w.sendfile(fd, 0, meta.size);
    return;
  }
  std::vector<char> buf(1 << 20);
  while (true) {
    ssize_t r = reader->read(buf.data(), buf.size());
    if (r <= 0) break;
    w.write(buf.data(), static_cast<size_t>(r));
  }
}

void Registry::put_blob(http::Request& req, http::ResponseWriter& w, const std::string& name,
                        const std::string& digest) {
  std::string content_type;
  auto it = req.headers.find("Content-Type");
  if (it != req.headers.end()) content_type = it->second;
  if (content_type.empty()) {
    response_error(w, wire::ErrorInfo{400, "INVALID_PARAMETER", "content type invalid: empty", ""});
    return;
  }
  metrics_.blob_bytes_in.fetch_add(static_cast<uint64_t>(req.content_length));
  bool ok = store_->PutBlob(name, digest, content_type, req.content_length,
                            [&](char* buf, size_t n) { return req.read_body(buf, n); });
  if (!ok) {
    response_error(w, wire::ErrorInfo{500, "INTERNAL", "store put blob failed", ""});
    return;
  }
  w.write_all(201, "");  // registry.go:162 StatusCreated
}

void Registry::blob_location(http::Request& req, http::ResponseWriter& w, const std::string& name,
                             const std::string& digest, const std::string& purpose) {
  std::map<std::string, std::string> properties;
  for (auto& kv : req.query) properties[kv.first] = kv.second;
  auto loc = store_->GetBlobLocation(name, digest, purpose, properties);
  if (loc.supported) {
    if (purpose == "upload") metrics_.presign_upload_total.fetch_add(1);
    if (purpose == "download") metrics_.presign_download_total.fetch_add(1);
  }
  if (!loc.supported) {
    response_error(w, wire::ErrorInfo{501, "UNSUPPORTED", "blob location not supported", ""});
    return;
  }
  json::Object o;
  o["provider"] = json::Value(loc.provider);
  o["purpose"] = json::Value(loc.purpose);
  o["properties"] = loc.properties;
  response_ok(w, json::Value(std::move(o)));
}

static const char kB64[] = "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789+/";

static std::string base64_encode(const std::string& in) {
  std::string out;
  out.reserve((in.size() + 2) / 3 * 4);
  size_t i = 0;
  for (; i + 3 <= in.size(); i += 3) {
    uint32_t v = (uint8_t)in[i] << 16 | (uint8_t)in[i + 1] << 8 | (uint8_t)in[i + 2];
    out += kB64[v >> 18];
    out += kB64[(v >> 12) & 63];
    out += kB64[(v >> 6) & 63];
    out += kB64[v & 63];
  }
  size_t rem = in.size() - i;
  if (rem == 1) {
    uint32_t v = (uint8_t)in[i] << 16;
    out += kB64[v >> 18];
    out += kB64[(v >> 12) & 63];
    out += "==";
  } else if (rem == 2) {
    uint32_t v = (uint8_t)in[i] << 16 | (uint8_t)in[i + 1] << 8;
    out += kB64[v >> 18];
    out += kB64[(v >> 12) & 63];
    out += kB64[(v >> 6) & 63];
    out += '=';
  }
  return out;
}

// GET /{name}/manifests/{ref}/pull-plan — one-round-trip pull metadata
// (no reference counterpart; clients fall back to per-blob calls on 404).
// Returns the manifest plus, per blob digest: the presigned download
// location and — when small enough — the inlined leaves sidecar (clients
// verify it against the annotation digest, so it is not trusted).
void Registry::pull_plan(http::Request& req, http::ResponseWriter& w, const std::string& name,
                         const std::string& ref) {
  json::Value plan;
  if (!build_pull_plan(name, ref, &plan)) {
    response_error(w,
                   wire::ErrorInfo{404, "MANIFEST_UNKNOWN", "manifest: " + ref + " not found", ""});
    return;
  }
  response_ok(w, plan);
}

// POST /{name}/pull-plans  body {"refs": [...]} → {"plans": {ref: plan}}.
// One control-plane round trip for MANY versions — the per-version GET
// still left config-5-shaped indexes (one small blob per version, hundreds
// of versions) paying one round trip per version. Unknown refs are simply
// omitted; clients fall back per-ref.
void Registry::pull_plans(http::Request& req, http::ResponseWriter& w,
                          const std::string& name) {
  constexpr size_t kMaxRefs = 4096;
  std::string body = req.read_body_all(kMaxManifestBytes);
  json::Object plans;
  try {
    auto doc = json::parse(body);
    const auto& refs = doc["refs"].items();
    if (refs.size() > kMaxRefs) {
      response_error(w, wire::ErrorInfo{400, "UNKNOWN", "too many refs", ""});
      return;
    }
    for (const auto& r : refs) {
      const std::string& ref = r.as_string();
      if (ref.empty() || plans.contains(ref)) continue;
      json::Value plan;
      if (build_pull_plan(name, ref, &plan)) plans[ref] = std::move(plan);
    }
  } catch (const std::exception& e) {
    response_error(w, wire::ErrorInfo{400, "UNKNOWN", std::string("bad body: ") + e.what(), ""});
    return;
  }
  json::Object o;
  o["plans"] = json::Value(std::move(plans));
  response_ok(w, json::Value(std::move(o)));
}

bool Registry::build_pull_plan(const std::string& name, const std::string& ref,
                               json::Value* out) {
  constexpr int64_t kInlineLeavesMax = 256 << 10;  // blobs <= 1 GiB at 128 KiB chunks
  wire::Manifest manifest;
  if (!store_->GetManifest(name, ref, &manifest)) return false;
  json::Object blobs;
  std::vector<const wire::Descriptor*> descs;
  descs.push_back(&manifest.config);
  for (auto& d : manifest.blobs) descs.push_back(&d);
  for (const wire::Descriptor* d : descs) {
    if (d->size == 0 || d->digest.empty()) continue;
    if (blobs.contains(d->digest)) continue;
    json::Object e;
    std::map<std::string, std::string> props;
    props["size"] = std::to_string(d->size);
    props["name"] = d->name;
    auto loc = store_->GetBlobLocation(name, d->digest, "download", props);
    if (loc.supported) {
      metrics_.presign_download_total.fetch_add(1);
      json::Object lo;
      lo["provider"] = json::Value(loc.provider);
      lo["purpose"] = json::Value(loc.purpose);
      lo["properties"] = loc.properties;
      e["location"] = json::Value(std::move(lo));
    }
    std::string leaves_digest;
    for (auto& kv : d->annotations)
      if (kv.first == "modelx.amd/leaves-blob") leaves_digest = kv.second;
    if (!leaves_digest.empty()) {
      store::FileMeta meta;
      auto reader = store_->GetBlob(name, leaves_digest, &meta);
      if (reader && meta.size > 0 && meta.size <= kInlineLeavesMax) {
        std::string lv(static_cast<size_t>(meta.size), '\0');
        size_t got = 0;
        while (got < lv.size()) {
          ssize_t r = reader->read(&lv[got], lv.size() - got);
          if (r <= 0) break;
          got += static_cast<size_t>(r);
        }
        if (got == lv.size()) e["leaves64"] = json::Value(base64_encode(lv));
      }
    }
    blobs[d->digest] = json::Value(std::move(e));
  }
  json::Object o;
  o["manifest"] = manifest.to_json();
  o["blobs"] = json::Value(std::move(blobs));
  *out = json::Value(std::move(o));
  return true;
}

void Registry::garbage_collect(http::Request& req, http::ResponseWriter& w,
                               const std::string& name) {
  int removed = store_->GCBlobs(name);
  metrics_.gc_blobs_removed_total.fetch_add(static_cast<uint64_t>(removed));
  json::Object o;
  o["blobs"] = json::Value(removed);
  response_ok(w, json::Value(std::move(o)));
}

}  // namespace registry
}  // namespace modelx
