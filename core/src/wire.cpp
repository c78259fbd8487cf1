#include "modelx/wire.hpp"

namespace modelx {
namespace wire {

const char* kMediaTypeIndex = "application/vnd.modelx.model.index.v1.json";
const char* kMediaTypeManifest = "application/vnd.modelx.model.manifest.v1.json";
const char* kMediaTypeConfig = "application/vnd.modelx.model.config.v1.yaml";
const char* kMediaTypeFile = "application/vnd.modelx.model.file.v1";
const char* kMediaTypeDirTarGz = "application/vnd.modelx.model.directory.v1.tar+gz";
const char* kGoZeroTime = "0001-01-01T00:00:00Z";

static json::Value annotations_json(const std::vector<std::pair<std::string, std::string>>& a) {
  json::Object o;
  for (auto& kv : a) o[kv.first] = json::Value(kv.second);
  return json::Value(std::move(o));
}

static std::vector<std::pair<std::string, std::string>> annotations_from(const json::Value& v) {
  std::vector<std::pair<std::string, std::string>> out;
  for (auto& kv : v.object()) out.emplace_back(kv.first, kv.second.as_string());
  return out;
}

json::Value Descriptor::to_json() const {
  json::Object o;
  o["name"] = json::Value(name);
  if (!media_type.empty()) o["mediaType"] = json::Value(media_type);
  if (!digest.empty()) o["digest"] = json::Value(digest);
  if (size) o["size"] = json::Value(size);
  if (mode) o["mode"] = json::Value(static_cast<int64_t>(mode));
  // Go always serializes modified (omitempty no-op on time.Time)
  o["modified"] = json::Value(modified.empty() ? kGoZeroTime : modified);
  if (!annotations.empty()) o["annotations"] = annotations_json(annotations);
  return json::Value(std::move(o));
}

Descriptor Descriptor::from_json(const json::Value& v) {
  Descriptor d;
  d.name = v["name"].as_string();
  d.media_type = v["mediaType"].as_string();
  d.digest = v["digest"].as_string();
  d.size = v["size"].as_int();
  d.mode = static_cast<uint32_t>(v["mode"].as_int());
  d.modified = v["modified"].as_string();
  if (d.modified.empty()) d.modified = kGoZeroTime;
  d.annotations = annotations_from(v["annotations"]);
  return d;
}

const std::string* Descriptor::annotation(const std::string& key) const {
  for (auto& kv : annotations)
    if (kv.first == key) return &kv.second;
  return nullptr;
}

json::Value Index::to_json() const {
  json::Object o;
  o["schemaVersion"] = json::Value(schema_version);
  if (!media_type.empty()) o["mediaType"] = json::Value(media_type);
  json::Array arr;
  for (auto& m : manifests) arr.push_back(m.to_json());
  o["manifests"] = json::Value(std::move(arr));
  if (!annotations.empty()) o["annotations"] = annotations_json(annotations);
  return json::Value(std::move(o));
}

Index Index::from_json(const json::Value& v) {
  Index idx;
  idx.schema_version = static_cast<int>(v["schemaVersion"].as_int());
  idx.media_type = v["mediaType"].as_string();
  for (auto& m : v["manifests"].items()) idx.manifests.push_back(Descriptor::from_json(m));
  idx.annotations = annotations_from(v["annotations"]);
  return idx;
}

json::Value Manifest::to_json() const {
  json::Object o;
  o["schemaVersion"] = json::Value(schema_version);
  if (!media_type.empty()) o["mediaType"] = json::Value(media_type);
  o["config"] = config.to_json();
  json::Array arr;
  for (auto& b : blobs) arr.push_back(b.to_json());
  o["blobs"] = json::Value(std::move(arr));
  if (!annotations.empty()) o["annotations"] = annotations_json(annotations);
  return json::Value(std::move(o));
}

Manifest Manifest::from_json(const json::Value& v) {
  Manifest m;
  m.schema_version = static_cast<int>(v["schemaVersion"].as_int());
  m.media_type = v["mediaType"].as_string();
  m.config = Descriptor::from_json(v["config"]);
  for (auto& b : v["blobs"].items()) m.blobs.push_back(Descriptor::from_json(b));
  m.annotations = annotations_from(v["annotations"]);
  return m;
}

std::string ErrorInfo::to_json_body() const {
  json::Object o;
  o["code"] = json::Value(code);
  o["message"] = json::Value(message);
  o["detail"] = json::Value(detail);
  return json::Value(std::move(o)).dump();
}

bool digest_split(const std::string& digest, std::string* algo, std::string* hex) {
  size_t colon = digest.find(':');
  if (colon == std::string::npos || colon == 0 || colon + 32 > digest.size()) return false;
  for (size_t i = 0; i < colon; i++) {
    char c = digest[i];
    bool ok = (c >= 'a' && c <= 'z') || (c >= 'A' && c <= 'Z') || (c >= '0' && c <= '9') ||
              c == '-' || c == '_' || c == '+' || c == '.';
    if (!ok) return false;
  }
  for (size_t i = colon + 1; i < digest.size(); i++) {
    char c = digest[i];
    bool ok = (c >= '0' && c <= '9') || (c >= 'a' && c <= 'f') || (c >= 'A' && c <= 'F');
    if (!ok) return false;
  }
  if (digest.size() - colon - 1 < 32) return false;
  if (algo) *algo = digest.substr(0, colon);
  if (hex) {
    *hex = digest.substr(colon + 1);
    for (auto& c : *hex)
      if (c >= 'A' && c <= 'F') c = static_cast<char>(c + 32);
  }
  return true;
}

bool digest_valid(const std::string& digest) { return digest_split(digest, nullptr, nullptr); }

std::string blob_digest_path(const std::string& repository, const std::string& digest) {
  std::string algo, hex;
  digest_split(digest, &algo, &hex);
  return repository + "/blobs/" + algo + "/" + hex;
}

std::string index_path(const std::string& repository) { return repository + "/index.json"; }

std::string manifest_path(const std::string& repository, const std::string& reference) {
  return repository + "/manifests/" + reference;
}

}  // namespace wire
}  // namespace modelx
