// S3 backend: FSProvider over the S3 REST API (SigV4) + S3RegistryStore with
// presigned-URL blob locations and server-side multipart completion.
// MI355X-native equivalents of reference pkg/registry/{fs_s3.go,store_s3.go}
// without aws-sdk-go — hand-rolled SigV4 + HTTP against any S3-compatible
// endpoint (MinIO; the bundled modelx-s3d test server).
#pragma once

#include <memory>
#include <string>

#include "modelx/http.hpp"
#include "modelx/sigv4.hpp"
#include "modelx/store.hpp"

namespace modelx {
namespace store {

struct S3Options {
  std::string endpoint;  // http://host:port
  std::string bucket;
  std::string access_key;
  std::string secret_key;
  std::string region = "us-east-1";
  std::string prefix = "registry";  // fs_s3.go:77
  int presign_expire_seconds = 3600;  // fs_s3.go:37 default 1h
  // URL the *client* should use to reach S3 (presigned host); defaults to
  // endpoint. Lets modelxd talk to S3 on an internal address while presigning
  // public ones.
  std::string public_endpoint;
};

class S3FSProvider : public FSProvider {
 public:
  explicit S3FSProvider(S3Options opts);
  bool Put(const std::string& path, const std::string& content_type, int64_t length,
           const ReadFn& read) override;
  std::unique_ptr<BlobReader> Get(const std::string& path, FileMeta* meta) override;
  bool Stat(const std::string& path, FileMeta* meta) override;
  bool Remove(const std::string& path, bool recursive) override;
  bool Exists(const std::string& path) override;
  std::vector<FileMeta> List(const std::string& prefix, bool recursive) override;

  const S3Options& options() const { return opts_; }
  sigv4::Credentials creds() const;
  // raw signed S3 call; target path is /<bucket>/<prefixed key>
  http::ClientResponse call(const std::string& method, const std::string& key,
                            const std::map<std::string, std::string>& query,
                            const std::string& body, const std::string& content_type = "");
  std::string prefixed_key(const std::string& path) const;
  std::string host_header() const;
  std::string public_base() const;

 private:
  S3Options opts_;
  http::Url endpoint_;
};

// reference: pkg/registry/store_s3.go:19-22
constexpr int64_t kMultiPartUploadThreshold = 5LL << 30;  // 5 GiB
constexpr int kDefaultPartCount = 3;

class S3RegistryStore : public RegistryStore {
 public:
  explicit S3RegistryStore(std::shared_ptr<S3FSProvider> fs)
      : RegistryStore(fs), s3_(fs.get()) {}

  BlobLocationResult GetBlobLocation(const std::string& repository, const std::string& digest,
                                     const std::string& purpose,
                                     const std::map<std::string, std::string>& properties) override;

  // completes pending multipart uploads + verifies sizes before committing
  // (store_s3.go:68-92)
  bool PutManifest(const std::string& repository, const std::string& reference,
                   const std::string& content_type, const wire::Manifest& manifest,
                   std::string* err) override;

 private:
  std::string get_upload_id(const std::string& key, bool with_create);
  bool complete_multipart(const std::string& key, const std::string& upload_id, std::string* err);
  BlobLocationResult upload_location(const std::string& key,
                                     const std::map<std::string, std::string>& properties);
  BlobLocationResult download_location(const std::string& key, int64_t size);
  std::string presign(const std::string& method, const std::string& key,
                      const std::map<std::string, std::string>& extra_query);

  S3FSProvider* s3_;
};

}  // namespace store
}  // namespace modelx
