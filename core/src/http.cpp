#include "modelx/http.hpp"

#include <openssl/err.h>
#include <openssl/ssl.h>

#include <vector>

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/sendfile.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstring>
#include <stdexcept>

namespace modelx {
namespace http {

static int lower(int c) { return (c >= 'A' && c <= 'Z') ? c + 32 : c; }

bool CiLess::operator()(const std::string& a, const std::string& b) const {
  size_t n = std::min(a.size(), b.size());
  for (size_t i = 0; i < n; i++) {
    int ca = lower((unsigned char)a[i]), cb = lower((unsigned char)b[i]);
    if (ca != cb) return ca < cb;
  }
  return a.size() < b.size();
}

static bool is_unreserved(unsigned char c) {
  return (c >= 'A' && c <= 'Z') || (c >= 'a' && c <= 'z') || (c >= '0' && c <= '9') || c == '-' ||
         c == '_' || c == '.' || c == '~';
}

std::string url_decode(const std::string& s) {
  std::string out;
  out.reserve(s.size());
  for (size_t i = 0; i < s.size(); i++) {
    if (s[i] == '%' && i + 2 < s.size()) {
      auto hex = [](char c) -> int {
        if (c >= '0' && c <= '9') return c - '0';
        if (c >= 'a' && c <= 'f') return c - 'a' + 10;
        if (c >= 'A' && c <= 'F') return c - 'A' + 10;
        return -1;
      };
      int h = hex(s[i + 1]), l = hex(s[i + 2]);
      if (h >= 0 && l >= 0) {
        out.push_back(static_cast<char>(h * 16 + l));
        i += 2;
        continue;
      }
    }
    if (s[i] == '+') {
      out.push_back(' ');
      continue;
    }
    out.push_back(s[i]);
  }
  return out;
}

static std::string encode_impl(const std::string& s, bool keep_slash) {
  static const char* hexd = "0123456789ABCDEF";
  std::string out;
  out.reserve(s.size());
  for (unsigned char c : s) {
    if (is_unreserved(c) || (keep_slash && c == '/')) {
      out.push_back(static_cast<char>(c));
    } else {
      out.push_back('%');
      out.push_back(hexd[c >> 4]);
      out.push_back(hexd[c & 15]);
    }
  }
  return out;
}

std::string url_encode_path(const std::string& s) { return encode_impl(s, true); }
std::string url_encode_query(const std::string& s) { return encode_impl(s, false); }

std::map<std::string, std::string> parse_query(const std::string& q) {
  std::map<std::string, std::string> out;
  size_t pos = 0;
  while (pos < q.size()) {
    size_t amp = q.find('&', pos);
    if (amp == std::string::npos) amp = q.size();
    std::string kv = q.substr(pos, amp - pos);
    size_t eq = kv.find('=');
    if (eq == std::string::npos)
      out[url_decode(kv)] = "";
    else
      out[url_decode(kv.substr(0, eq))] = url_decode(kv.substr(eq + 1));
    pos = amp + 1;
  }
  return out;
}

Url Url::parse(const std::string& url) {
  Url u;
  std::string rest = url;
  size_t sep = rest.find("://");
  if (sep != std::string::npos) {
    u.scheme = rest.substr(0, sep);
    rest = rest.substr(sep + 3);
  } else {
    u.scheme = "http";
  }
  size_t slash = rest.find('/');
  std::string hostport = slash == std::string::npos ? rest : rest.substr(0, slash);
  std::string target = slash == std::string::npos ? "/" : rest.substr(slash);
  size_t colon = hostport.rfind(':');
  if (colon != std::string::npos) {
    u.host = hostport.substr(0, colon);
    u.port = atoi(hostport.c_str() + colon + 1);
  } else {
    u.host = hostport;
    u.port = (u.scheme == "https") ? 443 : 80;
  }
  size_t qm = target.find('?');
  if (qm != std::string::npos) {
    u.path = target.substr(0, qm);
    u.query = target.substr(qm + 1);
  } else {
    u.path = target;
  }
  return u;
}

// ---------------------------------------------------------------- conn -----

class Conn {
 public:
  explicit Conn(int fd) : fd_(fd) {}
  ~Conn() {
    if (ssl_) {
      SSL_shutdown(static_cast<SSL*>(ssl_));
      SSL_free(static_cast<SSL*>(ssl_));
    }
    if (fd_ >= 0) ::close(fd_);
  }
  int fd() const { return fd_; }
  void set_ssl(void* ssl) { ssl_ = ssl; }  // takes ownership

  // buffered read of one line (ending \n); returns false on EOF/error
  bool read_line(std::string* line, size_t max = 65536) {
    line->clear();
    while (line->size() < max) {
      if (rpos_ >= rbuf_.size()) {
        if (!fill()) return false;
      }
      char c = rbuf_[rpos_++];
      if (c == '\n') {
        if (!line->empty() && line->back() == '\r') line->pop_back();
        return true;
      }
      line->push_back(c);
    }
    return false;
  }

  ssize_t read_some(char* buf, size_t n) {
    if (rpos_ < rbuf_.size()) {
      size_t take = std::min(n, rbuf_.size() - rpos_);
      memcpy(buf, rbuf_.data() + rpos_, take);
      rpos_ += take;
      return static_cast<ssize_t>(take);
    }
    return raw_recv(buf, n);
  }

  void shutdown_now() {
    if (fd_ >= 0) ::shutdown(fd_, SHUT_RDWR);
  }

  // raw fd for splice()-style consumers; -1 when TLS or bytes are buffered
  int raw_fd_if_plain() const { return (!ssl_ && rpos_ >= rbuf_.size()) ? fd_ : -1; }

  bool write_full(const char* data, size_t n) {
    if (ssl_) {
      SSL* ssl = static_cast<SSL*>(ssl_);
      while (n > 0) {
        int w = SSL_write(ssl, data, static_cast<int>(std::min<size_t>(n, 1u << 30)));
        if (w <= 0) return false;
        data += w;
        n -= static_cast<size_t>(w);
      }
      return true;
    }
    while (n > 0) {
      ssize_t w = ::send(fd_, data, n, MSG_NOSIGNAL);
      if (w < 0) {
        if (errno == EINTR) continue;
        return false;
      }
      data += w;
      n -= static_cast<size_t>(w);
    }
    return true;
  }

  bool sendfile_full(int in_fd, int64_t offset, int64_t count) {
    if (ssl_) {
      // no zero-copy under TLS: pread + SSL_write
      std::vector<char> buf(256 << 10);
      while (count > 0) {
        ssize_t r = ::pread(in_fd, buf.data(),
                            static_cast<size_t>(std::min<int64_t>(count, buf.size())), offset);
        if (r <= 0) return false;
        if (!write_full(buf.data(), static_cast<size_t>(r))) return false;
        offset += r;
        count -= r;
      }
      return true;
    }
    off_t off = offset;
    while (count > 0) {
      ssize_t w = ::sendfile(fd_, in_fd, &off, static_cast<size_t>(count));
      if (w < 0) {
        if (errno == EINTR) continue;
        return false;
      }
      if (w == 0) return false;
      count -= w;
    }
    return true;
  }

 private:
  ssize_t raw_recv(char* buf, size_t n) {
    if (ssl_) {
      int r = SSL_read(static_cast<SSL*>(ssl_), buf,
                       static_cast<int>(std::min<size_t>(n, 1u << 30)));
      return r <= 0 ? (SSL_get_error(static_cast<SSL*>(ssl_), r) == SSL_ERROR_ZERO_RETURN
                           ? 0
                           : -1)
                    : r;
    }
    ssize_t r;
    do {
      r = ::recv(fd_, buf, n, 0);
    } while (r < 0 && errno == EINTR);
    return r;
  }

  bool fill() {
    char buf[16384];
    ssize_t r = raw_recv(buf, sizeof buf);
    if (r <= 0) return false;
    rbuf_.assign(buf, static_cast<size_t>(r));
    rpos_ = 0;
    return true;
  }
  int fd_;
  void* ssl_ = nullptr;  // SSL*
  std::string rbuf_;
  size_t rpos_ = 0;
};

// -------------------------------------------------------------- request ----

int Request::raw_fd_if_plain() const { return conn ? conn->raw_fd_if_plain() : -1; }

ssize_t Request::read_body(char* buf, size_t n) {
  if (body_remaining <= 0) return 0;
  size_t want = std::min<int64_t>(static_cast<int64_t>(n), body_remaining);
  ssize_t r = conn->read_some(buf, want);
  if (r > 0) body_remaining -= r;
  return r;
}

std::string Request::read_body_all(size_t max_bytes) {
  if (content_length > static_cast<int64_t>(max_bytes))
    throw std::runtime_error("body too large");
  std::string out;
  out.resize(static_cast<size_t>(content_length));
  size_t got = 0;
  while (got < out.size()) {
    ssize_t r = read_body(&out[got], out.size() - got);
    if (r <= 0) throw std::runtime_error("short body");
    got += static_cast<size_t>(r);
  }
  return out;
}

void Request::drain_body() {
  char buf[65536];
  while (body_remaining > 0) {
    if (read_body(buf, sizeof buf) <= 0) break;
  }
}

// ------------------------------------------------------- response writer ---

static const char* status_text(int code) {
  switch (code) {
    case 200: return "OK";
    case 201: return "Created";
    case 202: return "Accepted";
    case 204: return "No Content";
    case 206: return "Partial Content";
    case 301: return "Moved Permanently";
    case 302: return "Found";
    case 304: return "Not Modified";
    case 400: return "Bad Request";
    case 401: return "Unauthorized";
    case 403: return "Forbidden";
    case 404: return "Not Found";
    case 405: return "Method Not Allowed";
    case 409: return "Conflict";
    case 411: return "Length Required";
    case 413: return "Payload Too Large";
    case 416: return "Range Not Satisfiable";
    case 429: return "Too Many Requests";
    case 500: return "Internal Server Error";
    case 501: return "Not Implemented";
    default: return "Status";
  }
}

void ResponseWriter::begin(int status, int64_t content_length) {
  std::string head = "HTTP/1.1 " + std::to_string(status) + " " + status_text(status) + "\r\n";
  for (auto& kv : headers_) head += kv.first + ": " + kv.second + "\r\n";
  if (!headers_.count("Content-Length"))
    head += "Content-Length: " + std::to_string(content_length) + "\r\n";
  head += "\r\n";
  sent_ = true;
  if (!conn_->write_full(head.data(), head.size())) failed_ = true;
}

void ResponseWriter::write(const char* data, size_t n) {
  if (head_ || failed_) return;
  if (!conn_->write_full(data, n)) failed_ = true;
}

bool ResponseWriter::sendfile(int fd, int64_t offset, int64_t count) {
  if (head_ || failed_) return true;
  if (!conn_->sendfile_full(fd, offset, count)) {
    failed_ = true;
    return false;
  }
  return true;
}

void ResponseWriter::abort_connection() {
  failed_ = true;
  conn_->shutdown_now();
}

void ResponseWriter::write_all(int status, const std::string& body, const std::string& ct) {
  if (!ct.empty()) headers_["Content-Type"] = ct;
  begin(status, static_cast<int64_t>(body.size()));
  write(body.data(), body.size());
}

void ResponseWriter::write_json(int status, const std::string& body) {
  write_all(status, body, "application/json");
}

// --------------------------------------------------------------- server ----

Server::Server(std::string listen_addr, Handler handler, TlsConfig tls)
    : listen_addr_(std::move(listen_addr)), handler_(std::move(handler)), tls_(std::move(tls)) {}

Server::~Server() {
  stop();
  if (ssl_ctx_) SSL_CTX_free(static_cast<SSL_CTX*>(ssl_ctx_));
}

int Server::start() {
  std::string host = "0.0.0.0";
  int port = 8080;
  size_t colon = listen_addr_.rfind(':');
  if (colon != std::string::npos) {
    std::string h = listen_addr_.substr(0, colon);
    if (!h.empty()) host = h;
    port = atoi(listen_addr_.c_str() + colon + 1);
  }
  if (tls_.enabled() && !ssl_ctx_) {
    SSL_library_init();
    SSL_CTX* ctx = SSL_CTX_new(TLS_server_method());
    if (!ctx) throw std::runtime_error("SSL_CTX_new failed");
    if (SSL_CTX_use_certificate_chain_file(ctx, tls_.cert_file.c_str()) != 1 ||
        SSL_CTX_use_PrivateKey_file(ctx, tls_.key_file.c_str(), SSL_FILETYPE_PEM) != 1 ||
        SSL_CTX_check_private_key(ctx) != 1) {
      SSL_CTX_free(ctx);
      throw std::runtime_error("TLS cert/key load failed: " + tls_.cert_file);
    }
    ssl_ctx_ = ctx;
  }
  listen_fd_ = ::socket(AF_INET, SOCK_STREAM | SOCK_CLOEXEC, 0);
  if (listen_fd_ < 0) throw std::runtime_error("socket() failed");
  int one = 1;
  setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof one);
  sockaddr_in addr{};
  addr.sin_family = AF_INET;
  addr.sin_port = htons(static_cast<uint16_t>(port));
  if (host == "0.0.0.0" || host.empty()) {
    addr.sin_addr.s_addr = INADDR_ANY;
  } else if (inet_pton(AF_INET, host.c_str(), &addr.sin_addr) != 1) {
    addr.sin_addr.s_addr = INADDR_ANY;
  }
  if (::bind(listen_fd_, reinterpret_cast<sockaddr*>(&addr), sizeof addr) != 0)
    throw std::runtime_error("bind(" + listen_addr_ + ") failed: " + strerror(errno));
  if (::listen(listen_fd_, 256) != 0) throw std::runtime_error("listen() failed");
  socklen_t alen = sizeof addr;
  getsockname(listen_fd_, reinterpret_cast<sockaddr*>(&addr), &alen);
  port_ = ntohs(addr.sin_port);
  accept_thread_ = std::thread([this] { accept_loop(); });
  return port_;
}

void Server::stop() {
  if (stopping_.exchange(true)) return;
  if (listen_fd_ >= 0) {
    ::shutdown(listen_fd_, SHUT_RDWR);
    ::close(listen_fd_);
    listen_fd_ = -1;
  }
  if (accept_thread_.joinable()) accept_thread_.join();
}

void Server::accept_loop() {
  while (!stopping_.load()) {
    sockaddr_in peer{};
    socklen_t plen = sizeof peer;
    int fd = ::accept4(listen_fd_, reinterpret_cast<sockaddr*>(&peer), &plen, SOCK_CLOEXEC);
    if (fd < 0) {
      if (stopping_.load()) break;
      if (errno == EINTR || errno == ECONNABORTED) continue;
      break;
    }
    char ip[64];
    inet_ntop(AF_INET, &peer.sin_addr, ip, sizeof ip);
    std::string addr = std::string(ip) + ":" + std::to_string(ntohs(peer.sin_port));
    live_conns_.fetch_add(1);
    std::thread([this, fd, addr] {
      serve_conn(fd, addr);
      live_conns_.fetch_sub(1);
    }).detach();
  }
}

void Server::serve_conn(int fd, std::string peer) {
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
  int buf = 8 << 20;
  setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &buf, sizeof buf);
  setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &buf, sizeof buf);
  Conn conn(fd);
  if (ssl_ctx_) {
    SSL* ssl = SSL_new(static_cast<SSL_CTX*>(ssl_ctx_));
    if (!ssl) return;
    SSL_set_fd(ssl, fd);
    if (SSL_accept(ssl) != 1) {
      SSL_free(ssl);
      return;
    }
    conn.set_ssl(ssl);
  }
  while (!stopping_.load()) {
    std::string line;
    if (!conn.read_line(&line)) return;
    if (line.empty()) continue;
    Request req;
    req.client_addr = peer;
    // request line: METHOD SP target SP HTTP/1.1
    size_t sp1 = line.find(' ');
    size_t sp2 = line.rfind(' ');
    if (sp1 == std::string::npos || sp2 == sp1) return;
    req.method = line.substr(0, sp1);
    req.target = line.substr(sp1 + 1, sp2 - sp1 - 1);
    bool http11 = line.compare(sp2 + 1, std::string::npos, "HTTP/1.1") == 0;
    size_t qm = req.target.find('?');
    if (qm != std::string::npos) {
      req.path = url_decode(req.target.substr(0, qm));
      req.query_raw = req.target.substr(qm + 1);
      req.query = parse_query(req.query_raw);
    } else {
      req.path = url_decode(req.target);
    }
    // headers
    while (true) {
      std::string h;
      if (!conn.read_line(&h)) return;
      if (h.empty()) break;
      size_t c = h.find(':');
      if (c == std::string::npos) continue;
      size_t v = c + 1;
      while (v < h.size() && h[v] == ' ') v++;
      req.headers[h.substr(0, c)] = h.substr(v);
    }
    auto it = req.headers.find("Content-Length");
    req.content_length = it != req.headers.end() ? atoll(it->second.c_str()) : 0;
    req.body_remaining = req.content_length;
    req.conn = &conn;
    bool client_close = false;
    auto ch = req.headers.find("Connection");
    if (ch != req.headers.end() && (ch->second == "close" || ch->second == "Close"))
      client_close = true;
    auto te = req.headers.find("Transfer-Encoding");
    ResponseWriter w(&conn, req.method == "HEAD");
    if (te != req.headers.end()) {
      w.write_all(411, "chunked transfer encoding not supported");
      return;
    }
    try {
      handler_(req, w);
    } catch (const std::exception& e) {
      if (!w.sent()) w.write_all(500, std::string("internal error: ") + e.what());
      return;  // state unknown; drop connection
    }
    if (!w.sent()) w.write_all(500, "handler sent no response");
    if (w.failed()) return;
    req.drain_body();  // leftover body would desync keep-alive
    if (client_close || !http11) return;
  }
}

// --------------------------------------------------------------- client ----

// Shared client-side TLS context: system CA paths; MODELX_TLS_INSECURE=1
// disables verification (self-signed stores).
static SSL_CTX* client_tls_ctx() {
  static SSL_CTX* ctx = [] {
    SSL_CTX* c = SSL_CTX_new(TLS_client_method());
    if (!c) return (SSL_CTX*)nullptr;
    SSL_CTX_set_default_verify_paths(c);
    const char* insecure = getenv("MODELX_TLS_INSECURE");
    SSL_CTX_set_verify(c, (insecure && *insecure == '1') ? SSL_VERIFY_NONE : SSL_VERIFY_PEER,
                       nullptr);
    return c;
  }();
  return ctx;
}

bool ClientConn::ensure_connected() {
  if (fd_ >= 0) return true;
  addrinfo hints{};
  hints.ai_family = AF_INET;
  hints.ai_socktype = SOCK_STREAM;
  addrinfo* res = nullptr;
  if (getaddrinfo(host_.c_str(), std::to_string(port_).c_str(), &hints, &res) != 0) return false;
  int fd = -1;
  for (addrinfo* ai = res; ai; ai = ai->ai_next) {
    fd = ::socket(ai->ai_family, ai->ai_socktype | SOCK_CLOEXEC, ai->ai_protocol);
    if (fd < 0) continue;
    if (::connect(fd, ai->ai_addr, ai->ai_addrlen) == 0) break;
    ::close(fd);
    fd = -1;
  }
  freeaddrinfo(res);
  if (fd < 0) return false;
  int one = 1;
  setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof one);
  // large buffers: the data plane moves multi-GiB objects over few
  // sockets; default loopback buffers throttle per-conn throughput
  int buf = 8 << 20;
  setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &buf, sizeof buf);
  setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &buf, sizeof buf);
  fd_ = fd;
  rbuf_.clear();
  rpos_ = 0;
  if (tls_) {
    SSL_CTX* ctx = client_tls_ctx();
    SSL* ssl = ctx ? SSL_new(ctx) : nullptr;
    if (!ssl) {
      close_fd();
      return false;
    }
    SSL_set_fd(ssl, fd_);
    SSL_set_tlsext_host_name(ssl, host_.c_str());  // SNI
    const char* insecure = getenv("MODELX_TLS_INSECURE");
    if (!(insecure && *insecure == '1')) SSL_set1_host(ssl, host_.c_str());
    int rc;
    do {
      rc = SSL_connect(ssl);
    } while (rc <= 0 && SSL_get_error(ssl, rc) == SSL_ERROR_WANT_READ);
    if (rc != 1) {
      SSL_free(ssl);
      close_fd();
      return false;
    }
    ssl_ = ssl;
  }
  return true;
}

void ClientConn::close_fd() {
  if (ssl_) {
    SSL_shutdown(static_cast<SSL*>(ssl_));
    SSL_free(static_cast<SSL*>(ssl_));
    ssl_ = nullptr;
  }
  if (fd_ >= 0) {
    ::close(fd_);
    fd_ = -1;
  }
  rbuf_.clear();
  rpos_ = 0;
}

ssize_t ClientConn::conn_recv(char* buf, size_t n, bool waitall) {
  if (ssl_) {
    SSL* ssl = static_cast<SSL*>(ssl_);
    size_t got = 0;
    while (got < n) {
      int r = SSL_read(ssl, buf + got, static_cast<int>(std::min<size_t>(n - got, 1u << 30)));
      if (r <= 0) {
        int err = SSL_get_error(ssl, r);
        if (err == SSL_ERROR_WANT_READ || err == SSL_ERROR_WANT_WRITE) continue;
        return got ? static_cast<ssize_t>(got) : (err == SSL_ERROR_ZERO_RETURN ? 0 : -1);
      }
      got += static_cast<size_t>(r);
      if (!waitall) break;
    }
    return static_cast<ssize_t>(got);
  }
  ssize_t r;
  do {
    r = ::recv(fd_, buf, n, waitall ? MSG_WAITALL : 0);
  } while (r < 0 && errno == EINTR);
  return r;
}

bool ClientConn::write_full(const char* data, size_t n) {
  if (ssl_) {
    SSL* ssl = static_cast<SSL*>(ssl_);
    while (n > 0) {
      int w = SSL_write(ssl, data, static_cast<int>(std::min<size_t>(n, 1u << 30)));
      if (w <= 0) {
        int err = SSL_get_error(ssl, w);
        if (err == SSL_ERROR_WANT_WRITE || err == SSL_ERROR_WANT_READ) continue;
        return false;
      }
      data += w;
      n -= static_cast<size_t>(w);
    }
    return true;
  }
  while (n > 0) {
    ssize_t w = ::send(fd_, data, n, MSG_NOSIGNAL);
    if (w < 0) {
      if (errno == EINTR) continue;
      return false;
    }
    data += w;
    n -= static_cast<size_t>(w);
  }
  return true;
}

bool ClientConn::send_request(const std::string& method, const std::string& target,
                              const Headers& headers, int64_t content_length) {
  for (int attempt = 0; attempt < 2; attempt++) {
    if (!ensure_connected()) return false;
    std::string head = method + " " + target + " HTTP/1.1\r\n";
    if (!headers.count("Host")) head += "Host: " + host_ + ":" + std::to_string(port_) + "\r\n";
    for (auto& kv : headers) head += kv.first + ": " + kv.second + "\r\n";
    if (content_length >= 0 &&
        (content_length > 0 || method == "PUT" || method == "POST"))
      head += "Content-Length: " + std::to_string(content_length) + "\r\n";
    head += "\r\n";
    if (write_full(head.data(), head.size())) return true;
    close_fd();  // stale keep-alive; retry once with a fresh connection
  }
  return false;
}

bool ClientConn::send_body(const char* data, size_t n) {
  if (!write_full(data, n)) {
    close_fd();
    return false;
  }
  return true;
}

bool ClientConn::read_line(std::string* line) {
  line->clear();
  while (true) {
    if (rpos_ >= rbuf_.size()) {
      char buf[16384];
      ssize_t r = conn_recv(buf, sizeof buf, false);
      if (r <= 0) return false;
      rbuf_.assign(buf, static_cast<size_t>(r));
      rpos_ = 0;
    }
    char c = rbuf_[rpos_++];
    if (c == '\n') {
      if (!line->empty() && line->back() == '\r') line->pop_back();
      return true;
    }
    line->push_back(c);
  }
}

bool ClientConn::read_response_head(int* status, Headers* headers) {
  std::string line;
  if (!read_line(&line)) {
    close_fd();
    return false;
  }
  // HTTP/1.1 200 OK
  if (line.size() < 12 || line.compare(0, 5, "HTTP/") != 0) {
    close_fd();
    return false;
  }
  *status = atoi(line.c_str() + 9);
  keep_alive_ = line.compare(5, 3, "1.1") == 0;
  headers->clear();
  while (true) {
    if (!read_line(&line)) {
      close_fd();
      return false;
    }
    if (line.empty()) break;
    size_t c = line.find(':');
    if (c == std::string::npos) continue;
    size_t v = c + 1;
    while (v < line.size() && line[v] == ' ') v++;
    (*headers)[line.substr(0, c)] = line.substr(v);
  }
  auto it = headers->find("Content-Length");
  if (it != headers->end()) {
    body_remaining_ = atoll(it->second.c_str());
    body_eof_ = false;
  } else {
    auto te = headers->find("Transfer-Encoding");
    if (te != headers->end()) {
      // chunked responses not supported by this control-plane client
      close_fd();
      return false;
    }
    body_remaining_ = -1;  // close-delimited
    body_eof_ = false;
  }
  auto ch = headers->find("Connection");
  if (ch != headers->end() && (ch->second == "close" || ch->second == "Close")) keep_alive_ = false;
  return true;
}

ssize_t ClientConn::read_body(char* buf, size_t n) {
  if (body_remaining_ == 0 || body_eof_) return 0;
  size_t want = n;
  if (body_remaining_ > 0) want = std::min<int64_t>(static_cast<int64_t>(n), body_remaining_);
  ssize_t r;
  if (rpos_ < rbuf_.size()) {
    size_t take = std::min(want, rbuf_.size() - rpos_);
    memcpy(buf, rbuf_.data() + rpos_, take);
    rpos_ += take;
    r = static_cast<ssize_t>(take);
  } else {
    r = conn_recv(buf, want, false);
    if (r < 0) {
      close_fd();
      return -1;
    }
    if (r == 0) {
      if (body_remaining_ < 0) {
        body_eof_ = true;
        close_fd();
        return 0;
      }
      close_fd();
      return -1;  // premature close
    }
  }
  if (body_remaining_ > 0) {
    body_remaining_ -= r;
    if (body_remaining_ == 0 && !keep_alive_) close_fd();
  }
  return r;
}

bool ClientConn::read_body_exact(char* buf, size_t n) {
  size_t got = 0;
  // drain buffered bytes first
  while (got < n && rpos_ < rbuf_.size()) {
    ssize_t r = read_body(buf + got, n - got);
    if (r <= 0) return false;
    got += static_cast<size_t>(r);
  }
  while (got < n) {
    size_t want = n - got;
    if (body_remaining_ >= 0 && static_cast<int64_t>(want) > body_remaining_)
      want = static_cast<size_t>(body_remaining_);
    if (want == 0) return false;
    ssize_t r = conn_recv(buf + got, want, true);
    if (r <= 0) {  // partial returns retry via the outer loop
      close_fd();
      return false;
    }
    got += static_cast<size_t>(r);
    if (body_remaining_ > 0) {
      body_remaining_ -= r;
      if (body_remaining_ == 0 && !keep_alive_) close_fd();
    }
  }
  return true;
}

bool ClientConn::do_request(const std::string& method, const std::string& target,
                            const Headers& headers, const std::string& body, ClientResponse* out,
                            size_t max_body) {
  int64_t clen = static_cast<int64_t>(body.size());
  if (method == "GET" || method == "HEAD" || method == "DELETE") {
    if (body.empty()) clen = -1;
  }
  if (!send_request(method, target, headers, clen)) return false;
  if (!body.empty() && !send_body(body.data(), body.size())) return false;
  if (!read_response_head(&out->status, &out->headers)) return false;
  out->body.clear();
  if (method == "HEAD") return true;
  char buf[65536];
  while (true) {
    ssize_t r = read_body(buf, sizeof buf);
    if (r < 0) return false;
    if (r == 0) break;
    if (out->body.size() + static_cast<size_t>(r) > max_body) return false;
    out->body.append(buf, static_cast<size_t>(r));
  }
  return true;
}

ClientResponse fetch(const std::string& method, const std::string& url, const Headers& headers,
                     const std::string& body, size_t max_body) {
  Url u = Url::parse(url);
  ClientConn conn(u.host, u.port);
  ClientResponse resp;
  if (!conn.do_request(method, u.target(), headers, body, &resp, max_body))
    throw std::runtime_error("http request failed: " + method + " " + url);
  return resp;
}

}  // namespace http
}  // namespace modelx
