#include "modelx/s3.hpp"

#include <cstring>

#include "modelx/sha256.hpp"

namespace modelx {
namespace store {

// ---- tiny XML helpers (S3 control responses only — MinIO/modelx-s3d) ------

static std::string xml_unescape(const std::string& s) {
  std::string out;
  out.reserve(s.size());
  for (size_t i = 0; i < s.size(); i++) {
    if (s[i] == '&') {
      if (s.compare(i, 4, "&lt;") == 0) { out += '<'; i += 3; continue; }
      if (s.compare(i, 4, "&gt;") == 0) { out += '>'; i += 3; continue; }
      if (s.compare(i, 5, "&amp;") == 0) { out += '&'; i += 4; continue; }
      if (s.compare(i, 6, "&quot;") == 0) { out += '"'; i += 5; continue; }
      if (s.compare(i, 6, "&apos;") == 0) { out += '\''; i += 5; continue; }
    }
    out += s[i];
  }
  return out;
}

static std::string xml_escape(const std::string& s) {
  std::string out;
  for (char c : s) {
    switch (c) {
      case '<': out += "&lt;"; break;
      case '>': out += "&gt;"; break;
      case '&': out += "&amp;"; break;
      default: out += c;
    }
  }
  return out;
}

// find all <tag>...</tag> values inside body (flat extraction)
static std::vector<std::string> xml_all(const std::string& body, const std::string& tag) {
  std::vector<std::string> out;
  std::string open = "<" + tag + ">";
  std::string close = "</" + tag + ">";
  size_t pos = 0;
  while ((pos = body.find(open, pos)) != std::string::npos) {
    size_t start = pos + open.size();
    size_t end = body.find(close, start);
    if (end == std::string::npos) break;
    out.push_back(xml_unescape(body.substr(start, end - start)));
    pos = end + close.size();
  }
  return out;
}

static std::string xml_first(const std::string& body, const std::string& tag) {
  auto v = xml_all(body, tag);
  return v.empty() ? "" : v[0];
}

// --------------------------------------------------------------- provider --

S3FSProvider::S3FSProvider(S3Options opts) : opts_(std::move(opts)) {
  endpoint_ = http::Url::parse(opts_.endpoint);
  if (opts_.public_endpoint.empty()) opts_.public_endpoint = opts_.endpoint;
}

sigv4::Credentials S3FSProvider::creds() const {
  sigv4::Credentials c;
  c.access_key = opts_.access_key;
  c.secret_key = opts_.secret_key;
  c.region = opts_.region;
  return c;
}

std::string S3FSProvider::prefixed_key(const std::string& path) const {
  if (opts_.prefix.empty()) return path;
  return opts_.prefix + "/" + path;
}

std::string S3FSProvider::host_header() const {
  return endpoint_.host + ":" + std::to_string(endpoint_.port);
}

std::string S3FSProvider::public_base() const {
  std::string b = opts_.public_endpoint;
  while (!b.empty() && b.back() == '/') b.pop_back();
  return b;
}

http::ClientResponse S3FSProvider::call(const std::string& method, const std::string& key,
                                        const std::map<std::string, std::string>& query,
                                        const std::string& body, const std::string& content_type) {
  std::string path = "/" + opts_.bucket;
  if (!key.empty()) path += "/" + http::url_encode_path(key);
  sigv4::RequestToSign rts;
  rts.method = method;
  rts.path = path;
  for (auto& kv : query)
    rts.query[http::url_encode_query(kv.first)] = http::url_encode_query(kv.second);
  rts.headers["host"] = host_header();
  rts.payload_hash = body.empty() ? sha256_hex("", 0) : sha256_hex(body);
  std::string auth = sigv4::sign_authorization(rts, creds(), sigv4::amz_date_now());
  http::Headers headers;
  headers["Authorization"] = auth;
  headers["x-amz-date"] = rts.headers["x-amz-date"];
  headers["x-amz-content-sha256"] = rts.headers["x-amz-content-sha256"];
  if (!content_type.empty()) headers["Content-Type"] = content_type;
  std::string target = path;
  std::string qs;
  for (auto& kv : rts.query) {
    if (!qs.empty()) qs += "&";
    qs += kv.first;
    if (!kv.second.empty()) qs += "=" + kv.second;
  }
  if (!qs.empty()) target += "?" + qs;
  http::ClientConn conn(endpoint_.host, endpoint_.port, endpoint_.scheme == "https");
  http::ClientResponse resp;
  if (!conn.do_request(method, target, headers, body, &resp))
    throw std::runtime_error("s3 call failed: " + method + " " + target);
  return resp;
}

bool S3FSProvider::Put(const std::string& path, const std::string& content_type, int64_t length,
                       const ReadFn& read) {
  // buffered: server-side Put is metadata-sized (manifests/indexes); big blob
  // bytes flow client→S3 directly via presign
  std::string body;
  if (length >= 0) body.reserve(static_cast<size_t>(length));
  char buf[65536];
  while (true) {
    ssize_t r = read(buf, sizeof buf);
    if (r < 0) return false;
    if (r == 0) break;
    body.append(buf, static_cast<size_t>(r));
    if (length >= 0 && static_cast<int64_t>(body.size()) >= length) break;
  }
  auto resp = call("PUT", prefixed_key(path), {}, body, content_type);
  return resp.status >= 200 && resp.status < 300;
}

namespace {
class StringReader : public BlobReader {
 public:
  explicit StringReader(std::string data) : data_(std::move(data)) {}
  ssize_t read(char* buf, size_t n) override {
    size_t take = std::min(n, data_.size() - pos_);
    memcpy(buf, data_.data() + pos_, take);
    pos_ += take;
    return static_cast<ssize_t>(take);
  }

 private:
  std::string data_;
  size_t pos_ = 0;
};
}  // namespace

std::unique_ptr<BlobReader> S3FSProvider::Get(const std::string& path, FileMeta* meta) {
  auto resp = call("GET", prefixed_key(path), {}, "");
  if (resp.status != 200) return nullptr;
  meta->name = path;
  meta->size = static_cast<int64_t>(resp.body.size());
  auto it = resp.headers.find("Content-Type");
  meta->content_type = it != resp.headers.end() ? it->second : "application/octet-stream";
  auto lm = resp.headers.find("Last-Modified");
  if (lm != resp.headers.end()) meta->last_modified = lm->second;
  return std::make_unique<StringReader>(std::move(resp.body));
}

bool S3FSProvider::Stat(const std::string& path, FileMeta* meta) {
  auto resp = call("HEAD", prefixed_key(path), {}, "");
  if (resp.status != 200) return false;
  meta->name = path;
  auto cl = resp.headers.find("Content-Length");
  meta->size = cl != resp.headers.end() ? atoll(cl->second.c_str()) : 0;
  auto it = resp.headers.find("Content-Type");
  meta->content_type = it != resp.headers.end() ? it->second : "application/octet-stream";
  return true;
}

bool S3FSProvider::Exists(const std::string& path) {
  FileMeta m;
  return Stat(path, &m);
}

bool S3FSProvider::Remove(const std::string& path, bool recursive) {
  if (!recursive) {
    auto resp = call("DELETE", prefixed_key(path), {}, "");
    return resp.status == 204 || resp.status == 200 || resp.status == 404;
  }
  // recursive: list + delete (reference: fs_s3.go:97-134 ListObjects+DeleteObjects)
  for (auto& m : List(path, true)) {
    std::string child = path.empty() ? m.name : path + "/" + m.name;
    call("DELETE", prefixed_key(child), {}, "");
  }
  return true;
}

std::vector<FileMeta> S3FSProvider::List(const std::string& prefix, bool recursive) {
  std::string full_prefix = prefixed_key(prefix);
  if (!full_prefix.empty() && full_prefix.back() != '/') full_prefix += "/";
  std::vector<FileMeta> out;
  std::string token;
  while (true) {
    std::map<std::string, std::string> q{{"list-type", "2"}, {"prefix", full_prefix},
                                         {"max-keys", "1000"}};
    if (!recursive) q["delimiter"] = "/";
    if (!token.empty()) q["continuation-token"] = token;
    auto resp = call("GET", "", q, "");
    if (resp.status != 200) break;
    auto keys = xml_all(resp.body, "Key");
    auto sizes = xml_all(resp.body, "Size");
    auto mods = xml_all(resp.body, "LastModified");
    for (size_t i = 0; i < keys.size(); i++) {
      FileMeta m;
      std::string key = keys[i];
      if (key.size() < full_prefix.size()) continue;
      m.name = key.substr(full_prefix.size());
      if (m.name.empty()) continue;
      if (i < sizes.size()) m.size = atoll(sizes[i].c_str());
      if (i < mods.size()) m.last_modified = mods[i];
      out.push_back(std::move(m));
    }
    if (xml_first(resp.body, "IsTruncated") != "true") break;
    token = xml_first(resp.body, "NextContinuationToken");
    if (token.empty()) break;
  }
  return out;
}

// ----------------------------------------------------------------- store ---

std::string S3RegistryStore::presign(const std::string& method, const std::string& key,
                                     const std::map<std::string, std::string>& extra_query) {
  const auto& opts = s3_->options();
  std::string path = "/" + opts.bucket + "/" + http::url_encode_path(s3_->prefixed_key(key));
  sigv4::RequestToSign rts;
  rts.method = method;
  rts.path = path;
  for (auto& kv : extra_query)
    rts.query[http::url_encode_query(kv.first)] = http::url_encode_query(kv.second);
  // presigned host must match what the CLIENT will connect to
  http::Url pub = http::Url::parse(s3_->public_base());
  rts.headers["host"] = pub.host + ":" + std::to_string(pub.port);
  std::string qs = sigv4::presign_query(rts, s3_->creds(), sigv4::amz_date_now(),
                                        opts.presign_expire_seconds);
  return s3_->public_base() + path + "?" + qs;
}

std::string S3RegistryStore::get_upload_id(const std::string& key, bool with_create) {
  // reuse pending upload (store_s3.go:235-264)
  auto resp = s3_->call("GET", "",
                        {{"uploads", ""}, {"prefix", s3_->prefixed_key(key)}, {"delimiter", "/"}},
                        "");
  if (resp.status == 200) {
    auto ids = xml_all(resp.body, "UploadId");
    if (!ids.empty()) return ids[0];
  }
  if (!with_create) return "";
  auto create = s3_->call("POST", s3_->prefixed_key(key), {{"uploads", ""}}, "");
  if (create.status != 200) return "";
  return xml_first(create.body, "UploadId");
}

BlobLocationResult S3RegistryStore::upload_location(
    const std::string& key, const std::map<std::string, std::string>& properties) {
  int64_t size = 0;
  bool multipart = false;
  std::string name;
  auto it = properties.find("size");
  if (it != properties.end()) size = atoll(it->second.c_str());
  it = properties.find("multipart");
  if (it != properties.end()) multipart = it->second == "true" || it->second == "1";
  it = properties.find("name");
  if (it != properties.end()) name = it->second;

  BlobLocationResult out;
  out.supported = true;
  out.provider = "s3";
  out.purpose = "upload";

  if (!multipart && size <= kMultiPartUploadThreshold) {
    // single presigned PUT (store_s3.go:192-226)
    json::Object part;
    part["url"] = json::Value(presign("PUT", key, {}));
    part["method"] = json::Value("PUT");
    json::Array parts;
    parts.push_back(json::Value(std::move(part)));
    json::Object props;
    props["parts"] = json::Value(std::move(parts));
    out.properties = json::Value(std::move(props));
    return out;
  }
  // multipart (store_s3.go:266-309)
  std::string upload_id = get_upload_id(key, true);
  int64_t parts_count = kDefaultPartCount;
  if (size / kMultiPartUploadThreshold != 0) {
    parts_count = size / kMultiPartUploadThreshold;
    if (size % kMultiPartUploadThreshold != 0) parts_count++;
  }
  // MI355X improvement: allow the client to ask for more parallelism than
  // ceil(size/5GiB) via part-count property (reference hardcodes the minimum)
  it = properties.find("part-count");
  if (it != properties.end()) {
    int64_t want = atoll(it->second.c_str());
    if (want > 0 && want <= 10000) parts_count = want;
  }
  json::Array parts;
  for (int64_t i = 0; i < parts_count; i++) {
    json::Object part;
    part["url"] = json::Value(presign("PUT", key,
                                      {{"partNumber", std::to_string(i + 1)},
                                       {"uploadId", upload_id}}));
    part["method"] = json::Value("PUT");
    part["partNumber"] = json::Value(i + 1);
    parts.push_back(json::Value(std::move(part)));
  }
  json::Object props;
  props["multipart"] = json::Value(true);
  props["uploadId"] = json::Value(upload_id);
  props["parts"] = json::Value(std::move(parts));
  out.properties = json::Value(std::move(props));
  return out;
}

BlobLocationResult S3RegistryStore::download_location(const std::string& key, int64_t size) {
  // single presigned GET; Range is not part of the signature (SignedHeaders=
  // host), so the client's pinned-ring engine issues parallel ranged GETs
  // against this one URL (the reference downloads single-stream —
  // extension_s3.go:24-37 — which SURVEY.md flags as the perf gap)
  BlobLocationResult out;
  out.supported = true;
  out.provider = "s3";
  out.purpose = "download";
  json::Object part;
  part["url"] = json::Value(presign("GET", key, {}));
  part["method"] = json::Value("GET");
  json::Array parts;
  parts.push_back(json::Value(std::move(part)));
  json::Object props;
  props["parts"] = json::Value(std::move(parts));
  if (size > 0) props["size"] = json::Value(size);
  out.properties = json::Value(std::move(props));
  return out;
}

BlobLocationResult S3RegistryStore::GetBlobLocation(
    const std::string& repository, const std::string& digest, const std::string& purpose,
    const std::map<std::string, std::string>& properties) {
  std::string key = wire::blob_digest_path(repository, digest);
  if (purpose == "upload") return upload_location(key, properties);
  if (purpose == "download") {
    store::FileMeta meta;
    int64_t size = 0;
    if (s3_->Stat(key, &meta)) size = meta.size;
    return download_location(key, size);
  }
  return BlobLocationResult{};
}

bool S3RegistryStore::complete_multipart(const std::string& key, const std::string& upload_id,
                                         std::string* err) {
  // ListParts → CompleteMultipartUpload (store_s3.go:136-190)
  auto list = s3_->call("GET", s3_->prefixed_key(key), {{"uploadId", upload_id}}, "");
  if (list.status != 200) {
    if (err) *err = "list parts failed";
    return false;
  }
  auto numbers = xml_all(list.body, "PartNumber");
  auto etags = xml_all(list.body, "ETag");
  if (numbers.empty()) {
    if (err) *err = "no parts uploaded";
    return false;
  }
  std::string xml = "<CompleteMultipartUpload>";
  for (size_t i = 0; i < numbers.size(); i++) {
    xml += "<Part><PartNumber>" + numbers[i] + "</PartNumber><ETag>" + xml_escape(etags[i]) +
           "</ETag></Part>";
  }
  xml += "</CompleteMultipartUpload>";
  auto resp = s3_->call("POST", s3_->prefixed_key(key), {{"uploadId", upload_id}}, xml,
                        "application/xml");
  if (resp.status != 200 || resp.body.find("<Error>") != std::string::npos) {
    if (err) *err = "complete multipart failed";
    return false;
  }
  return true;
}

bool S3RegistryStore::PutManifest(const std::string& repository, const std::string& reference,
                                  const std::string& content_type, const wire::Manifest& manifest,
                                  std::string* err) {
  // manifest PUT is the commit point: adopt + complete pending multipart
  // uploads, verify sizes, delete mismatches (store_s3.go:68-92)
  std::vector<wire::Descriptor> all = manifest.blobs;
  all.push_back(manifest.config);
  for (auto& blob : all) {
    if (blob.digest.empty()) continue;
    std::string key = wire::blob_digest_path(repository, blob.digest);
    std::string upload_id = get_upload_id(key, false);
    if (!upload_id.empty()) {
      std::string cerr;
      if (!complete_multipart(key, upload_id, &cerr)) {
        if (err) *err = "blob " + blob.digest + ": " + cerr;
        return false;
      }
    }
    store::FileMeta meta;
    if (!s3_->Stat(key, &meta)) {
      if (err) *err = "blob " + blob.digest + " not uploaded";
      return false;
    }
    if (blob.size != 0 && meta.size != blob.size) {
      s3_->Remove(key, false);
      if (err)
        *err = "blob " + blob.digest + " size mismatch: manifest " + std::to_string(blob.size) +
               " != stored " + std::to_string(meta.size);
      return false;
    }
  }
  return RegistryStore::PutManifest(repository, reference, content_type, manifest, err);
}

}  // namespace store
}  // namespace modelx
