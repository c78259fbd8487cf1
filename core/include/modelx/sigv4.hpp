// AWS Signature V4 — signing (Authorization header), presigning (query auth)
// and verification (used by the bundled mini-S3 server to validate presigned
// URLs the way MinIO would).
//
// MI355X-native replacement for the reference's aws-sdk-go-v2 presign usage
// (reference: pkg/registry/store_s3.go:192-226, fs_s3.go:53-80).
#pragma once

#include <map>
#include <string>

namespace modelx {
namespace sigv4 {

struct Credentials {
  std::string access_key;
  std::string secret_key;
  std::string region = "us-east-1";
  std::string service = "s3";
};

// Canonical pieces given to the signer. `query` must contain RAW-encoded
// key/value pairs (already URI-encoded as they will appear on the wire).
struct RequestToSign {
  std::string method;
  std::string path;  // raw path as on the wire (will be used as-is)
  std::map<std::string, std::string> query;       // encoded k -> encoded v
  std::map<std::string, std::string> headers;     // lowercase k -> v (must include host)
  std::string payload_hash = "UNSIGNED-PAYLOAD";  // or hex sha256
};

// RFC3339-basic timestamp "20230501T123015Z" for now (or fixed for tests).
std::string amz_date_now();

// Returns the Authorization header value; adds x-amz-date/x-amz-content-sha256
// to req.headers (caller sends them).
std::string sign_authorization(RequestToSign& req, const Credentials& cred,
                               const std::string& amz_date);

// Returns the full query string (encoded, '&'-joined, including
// X-Amz-Signature) for a presigned URL valid for expires seconds.
std::string presign_query(const RequestToSign& req, const Credentials& cred,
                          const std::string& amz_date, int expires_seconds);

// Verify a presigned-URL request (query auth). Returns true when the
// signature matches and the URL is not expired (now_epoch seconds).
bool verify_presigned(const std::string& method, const std::string& raw_path,
                      const std::map<std::string, std::string>& raw_query,
                      const std::string& host_header, const Credentials& cred, long now_epoch,
                      std::string* error);

}  // namespace sigv4
}  // namespace modelx
