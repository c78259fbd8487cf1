// Storage layer: FSProvider (L1) + RegistryStore (L2).
// MI355X-native equivalents of reference pkg/registry/{fs.go,fs_local.go,
// store.go,store_fs.go,store_s3.go}. Same key layout + sidecar-meta scheme so
// an on-disk/S3 registry written by the reference server is readable here.
#pragma once

#include <cstdint>
#include <functional>
#include <memory>
#include <mutex>
#include <string>
#include <vector>

#include "modelx/sigv4.hpp"
#include "modelx/wire.hpp"

namespace modelx {
namespace store {

struct FileMeta {
  std::string name;           // relative key
  int64_t size = 0;
  std::string content_type;
  std::string last_modified;  // RFC3339
};

class BlobReader {
 public:
  virtual ~BlobReader() = default;
  virtual ssize_t read(char* buf, size_t n) = 0;
  // when >= 0, caller may sendfile() directly from this fd at current offset
  virtual int sendfile_fd() { return -1; }
};

using ReadFn = std::function<ssize_t(char*, size_t)>;  // pull bytes from request body

// L1: 6-method blob-store abstraction (reference: pkg/registry/fs.go:15-22)
class FSProvider {
 public:
  virtual ~FSProvider() = default;
  virtual bool Put(const std::string& path, const std::string& content_type, int64_t length,
                   const ReadFn& read) = 0;
  virtual std::unique_ptr<BlobReader> Get(const std::string& path, FileMeta* meta) = 0;
  virtual bool Stat(const std::string& path, FileMeta* meta) = 0;
  virtual bool Remove(const std::string& path, bool recursive) = 0;
  virtual bool Exists(const std::string& path) = 0;
  virtual std::vector<FileMeta> List(const std::string& prefix, bool recursive) = 0;

  bool PutBytes(const std::string& path, const std::string& content_type,
                const std::string& data);
  bool GetBytes(const std::string& path, std::string* out, std::string* content_type = nullptr);
};

// Local disk provider; data file + "<path>.meta" JSON sidecar
// (reference: pkg/registry/fs_local.go:41-44,155-169)
class LocalFSProvider : public FSProvider {
 public:
  explicit LocalFSProvider(std::string basepath);
  bool Put(const std::string& path, const std::string& content_type, int64_t length,
           const ReadFn& read) override;
  std::unique_ptr<BlobReader> Get(const std::string& path, FileMeta* meta) override;
  bool Stat(const std::string& path, FileMeta* meta) override;
  bool Remove(const std::string& path, bool recursive) override;
  bool Exists(const std::string& path) override;
  std::vector<FileMeta> List(const std::string& prefix, bool recursive) override;

 private:
  std::string abs(const std::string& rel) const;
  std::string basepath_;
};

struct BlobLocationResult {
  bool supported = false;
  std::string provider;  // "s3"
  std::string purpose;
  json::Value properties;  // presign schema (store_s3.go:228-308)
};

// L2: semantic store (reference: pkg/registry/store.go:34-54)
class RegistryStore {
 public:
  virtual ~RegistryStore() = default;

  virtual bool GetGlobalIndex(const std::string& search, wire::Index* out);
  virtual bool GetIndex(const std::string& repository, const std::string& search,
                        wire::Index* out);
  virtual bool RemoveIndex(const std::string& repository);

  virtual bool ExistsManifest(const std::string& repository, const std::string& reference);
  virtual bool GetManifest(const std::string& repository, const std::string& reference,
                           wire::Manifest* out);
  virtual bool PutManifest(const std::string& repository, const std::string& reference,
                           const std::string& content_type, const wire::Manifest& manifest,
                           std::string* err);
  virtual bool DeleteManifest(const std::string& repository, const std::string& reference);

  virtual std::vector<std::string> ListBlobs(const std::string& repository);
  virtual std::unique_ptr<BlobReader> GetBlob(const std::string& repository,
                                              const std::string& digest, FileMeta* meta);
  virtual bool DeleteBlob(const std::string& repository, const std::string& digest);
  virtual bool PutBlob(const std::string& repository, const std::string& digest,
                       const std::string& content_type, int64_t length, const ReadFn& read);
  virtual bool ExistsBlob(const std::string& repository, const std::string& digest);

  virtual BlobLocationResult GetBlobLocation(const std::string& repository,
                                             const std::string& digest, const std::string& purpose,
                                             const std::map<std::string, std::string>& properties);

  // mark-and-sweep GC (reference: pkg/registry/gc.go — with ListBlobs FIXED;
  // the reference's returns nil and its GC is a no-op, SURVEY.md §6 defects)
  int GCBlobs(const std::string& repository);
  int GCBlobsAll();

  explicit RegistryStore(std::shared_ptr<FSProvider> fs) : fs_(std::move(fs)) {}
  FSProvider* fs() { return fs_.get(); }

  bool RefreshIndex(const std::string& repository);
  bool RefreshGlobalIndex();

 protected:
  bool RefreshGlobalIndexLocked();
  bool PutIndex(const std::string& repository, wire::Index index);
  std::shared_ptr<FSProvider> fs_;
  std::mutex index_mu_;  // serialize index rebuilds
};

}  // namespace store
}  // namespace modelx
