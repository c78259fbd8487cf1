// C++ wire structs — byte-compatible with the reference Go JSON
// (reference: pkg/types/types.go:20-66, pkg/errors/errors.go:11-44).
// Mirrored by the Python dataclasses in modelx_amd/wire/types.py.
#pragma once

#include <cstdint>
#include <map>
#include <string>
#include <vector>

#include "modelx/json.hpp"

namespace modelx {
namespace wire {

extern const char* kMediaTypeIndex;      // application/vnd.modelx.model.index.v1.json
extern const char* kMediaTypeManifest;   // application/vnd.modelx.model.manifest.v1.json
extern const char* kMediaTypeConfig;     // application/vnd.modelx.model.config.v1.yaml
extern const char* kMediaTypeFile;       // application/vnd.modelx.model.file.v1
extern const char* kMediaTypeDirTarGz;   // application/vnd.modelx.model.directory.v1.tar+gz
extern const char* kGoZeroTime;          // 0001-01-01T00:00:00Z

struct Descriptor {
  std::string name;
  std::string media_type;
  std::string digest;
  int64_t size = 0;
  uint32_t mode = 0;
  std::string modified = kGoZeroTime;  // RFC3339; kept as string (opaque passthrough)
  std::vector<std::pair<std::string, std::string>> annotations;

  json::Value to_json() const;
  static Descriptor from_json(const json::Value& v);
  const std::string* annotation(const std::string& key) const;
};

struct Index {
  int schema_version = 1;
  std::string media_type;
  std::vector<Descriptor> manifests;
  std::vector<std::pair<std::string, std::string>> annotations;

  json::Value to_json() const;
  static Index from_json(const json::Value& v);
};

struct Manifest {
  int schema_version = 1;
  std::string media_type;
  Descriptor config;
  std::vector<Descriptor> blobs;
  std::vector<std::pair<std::string, std::string>> annotations;

  json::Value to_json() const;
  static Manifest from_json(const json::Value& v);
};

struct ErrorInfo {
  int http_status = 400;
  std::string code = "UNKNOWN";
  std::string message;
  std::string detail;

  std::string to_json_body() const;
};

// digest helpers
bool digest_valid(const std::string& digest);
bool digest_split(const std::string& digest, std::string* algo, std::string* hex);

// key layout (reference: pkg/registry/store.go:56-74)
std::string blob_digest_path(const std::string& repository, const std::string& digest);
std::string index_path(const std::string& repository);
std::string manifest_path(const std::string& repository, const std::string& reference);

}  // namespace wire
}  // namespace modelx
