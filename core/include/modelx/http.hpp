// Minimal HTTP/1.1 server + client over POSIX sockets.
//
// MI355X-native replacement for the reference's net/http + gorilla/mux stack
// (reference: pkg/registry/server.go, route.go). The server is the CONTROL
// plane only — blob bytes flow client<->S3 via presigned redirect — so a
// thread-per-connection keep-alive model is the right complexity point.
// The data-plane client (pinned-ring range-GET engine) lives in engine.cpp
// and reuses HttpClientConn.
#pragma once

#include <atomic>
#include <cstdint>
#include <functional>
#include <map>
#include <memory>
#include <string>
#include <thread>
#include <vector>

namespace modelx {
namespace http {

// case-insensitive header map
struct CiLess {
  bool operator()(const std::string& a, const std::string& b) const;
};
using Headers = std::map<std::string, std::string, CiLess>;

std::string url_decode(const std::string& s);
std::string url_encode_path(const std::string& s);   // keeps '/'
std::string url_encode_query(const std::string& s);  // encodes all reserved
std::map<std::string, std::string> parse_query(const std::string& q);

struct Url {
  std::string scheme, host, path, query;  // path is raw (undecoded)
  int port = 80;
  static Url parse(const std::string& url);
  std::string target() const { return query.empty() ? path : path + "?" + query; }
};

// ---------------------------------------------------------------- server ---

class Conn;

struct Request {
  std::string method;
  std::string target;      // raw request-target
  std::string path;        // decoded path
  std::string query_raw;   // raw query string
  std::map<std::string, std::string> query;  // decoded
  Headers headers;
  int64_t content_length = 0;
  std::string client_addr;

  // body streaming
  Conn* conn = nullptr;
  int64_t body_remaining = 0;
  ssize_t read_body(char* buf, size_t n);          // returns 0 at end
  // raw socket fd when the connection is plaintext AND the conn buffer is
  // fully drained (returns -1 otherwise) — enables splice()-based sinks
  int raw_fd_if_plain() const;
  std::string read_body_all(size_t max_bytes);      // throws if over max
  void drain_body();
};

class ResponseWriter {
 public:
  explicit ResponseWriter(Conn* c, bool is_head) : conn_(c), head_(is_head) {}
  Headers& headers() { return headers_; }
  void set_header(const std::string& k, const std::string& v) { headers_[k] = v; }
  // Fixed-length response paths (always Content-Length; no chunked TE).
  void write_all(int status, const std::string& body, const std::string& content_type = "");
  void write_json(int status, const std::string& body);
  // Streaming: declare length first, then stream exactly that many bytes.
  void begin(int status, int64_t content_length);
  void write(const char* data, size_t n);
  bool sendfile(int fd, int64_t offset, int64_t count);  // after begin()
  // Hard-fail a partially-sent response: marks the writer failed and shuts
  // the socket down so the peer sees EOF mid-body instead of a desynced
  // stream (next response bytes read as body).
  void abort_connection();
  bool sent() const { return sent_; }
  bool failed() const { return failed_; }

 private:
  Conn* conn_;
  bool head_;
  bool sent_ = false;
  bool failed_ = false;
  Headers headers_;
};

using Handler = std::function<void(Request&, ResponseWriter&)>;

// TLS serving (reference: pkg/registry/server.go:37-43 ListenAndServeTLS).
struct TlsConfig {
  std::string cert_file, key_file;
  bool enabled() const { return !cert_file.empty(); }
};

class Server {
 public:
  Server(std::string listen_addr, Handler handler, TlsConfig tls = {});
  ~Server();
  // binds + starts accept thread; returns bound port (for :0)
  int start();
  void stop();
  int port() const { return port_; }

 private:
  void accept_loop();
  void serve_conn(int fd, std::string peer);

  std::string listen_addr_;
  Handler handler_;
  TlsConfig tls_;
  void* ssl_ctx_ = nullptr;  // SSL_CTX* when TLS is enabled
  int listen_fd_ = -1;
  int port_ = 0;
  std::atomic<bool> stopping_{false};
  std::thread accept_thread_;
  std::atomic<int> live_conns_{0};
};

// ---------------------------------------------------------------- client ---

struct ClientResponse {
  int status = 0;
  Headers headers;
  std::string body;
};

// One keep-alive connection to host:port (plain TCP; control-plane use).
class ClientConn {
 public:
  // tls: speak TLS (https presigned URLs / TLS object stores). Certificate
  // verification uses the system CA paths; MODELX_TLS_INSECURE=1 skips it
  // (self-signed MinIO/s3d, mirrors the CLI --insecure).
  ClientConn(std::string host, int port, bool tls = false)
      : host_(std::move(host)), port_(port), tls_(tls) {}
  ~ClientConn() { close_fd(); }
  bool connected() const { return fd_ >= 0; }
  bool tls() const { return tls_; }

  // Buffered full-body request. body may be empty. Returns false on socket error.
  bool do_request(const std::string& method, const std::string& target, const Headers& headers,
                  const std::string& body, ClientResponse* out, size_t max_body = (64u << 20));

  // Streaming request: send headers (+optional body via body_cb writes), then
  // read status+headers; body is then read with read_body into caller buffers.
  bool send_request(const std::string& method, const std::string& target, const Headers& headers,
                    int64_t content_length);
  bool send_body(const char* data, size_t n);
  bool read_response_head(int* status, Headers* headers);
  // reads up to n body bytes; 0 = body complete; <0 = error
  ssize_t read_body(char* buf, size_t n);
  // reads exactly n body bytes (MSG_WAITALL bulk recv); false on error/EOF
  bool read_body_exact(char* buf, size_t n);
  void close_fd();

  const std::string& host() const { return host_; }
  int port() const { return port_; }

 private:
  bool ensure_connected();
  bool write_full(const char* data, size_t n);
  bool read_line(std::string* line);
  // one recv: plain socket or SSL; waitall loops until n bytes (or error)
  ssize_t conn_recv(char* buf, size_t n, bool waitall);

  std::string host_;
  int port_;
  bool tls_ = false;
  void* ssl_ = nullptr;  // SSL* when tls_ and connected
  int fd_ = -1;
  std::string rbuf_;
  size_t rpos_ = 0;
  int64_t body_remaining_ = 0;
  bool body_eof_ = false;      // close-delimited body
  bool keep_alive_ = true;
};

// Convenience one-shot (opens its own connection).
ClientResponse fetch(const std::string& method, const std::string& url, const Headers& headers = {},
                     const std::string& body = "", size_t max_body = (256u << 20));

}  // namespace http
}  // namespace modelx
