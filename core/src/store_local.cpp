#include <dirent.h>
#include <fcntl.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cstring>
#include <ctime>

#include "modelx/store.hpp"

namespace modelx {
namespace store {

bool FSProvider::PutBytes(const std::string& path, const std::string& content_type,
                          const std::string& data) {
  size_t off = 0;
  return Put(path, content_type, static_cast<int64_t>(data.size()),
             [&](char* buf, size_t n) -> ssize_t {
               size_t take = std::min(n, data.size() - off);
               memcpy(buf, data.data() + off, take);
               off += take;
               return static_cast<ssize_t>(take);
             });
}

bool FSProvider::GetBytes(const std::string& path, std::string* out, std::string* content_type) {
  FileMeta meta;
  auto r = Get(path, &meta);
  if (!r) return false;
  if (content_type) *content_type = meta.content_type;
  out->clear();
  char buf[65536];
  while (true) {
    ssize_t n = r->read(buf, sizeof buf);
    if (n < 0) return false;
    if (n == 0) break;
    out->append(buf, static_cast<size_t>(n));
  }
  return true;
}

namespace {

std::string rfc3339_from_time(time_t t) {
  struct tm tmv;
  gmtime_r(&t, &tmv);
  char buf[32];
  strftime(buf, sizeof buf, "%Y-%m-%dT%H:%M:%SZ", &tmv);
  return buf;
}

bool mkdirs_for(const std::string& filepath) {
  size_t pos = 0;
  while ((pos = filepath.find('/', pos + 1)) != std::string::npos) {
    std::string dir = filepath.substr(0, pos);
    if (mkdir(dir.c_str(), 0755) != 0 && errno != EEXIST) return false;
  }
  return true;
}

class FdReader : public BlobReader {
 public:
  explicit FdReader(int fd) : fd_(fd) {}
  ~FdReader() override {
    if (fd_ >= 0) ::close(fd_);
  }
  ssize_t read(char* buf, size_t n) override {
    ssize_t r;
    do {
      r = ::read(fd_, buf, n);
    } while (r < 0 && errno == EINTR);
    return r;
  }
  int sendfile_fd() override { return fd_; }

 private:
  int fd_;
};

constexpr const char* kMetaSuffix = ".meta";

}  // namespace

LocalFSProvider::LocalFSProvider(std::string basepath) : basepath_(std::move(basepath)) {
  if (basepath_.empty()) basepath_ = "data/registry";  // fs_local.go:22-26 default
  mkdirs_for(basepath_ + "/x");
}

std::string LocalFSProvider::abs(const std::string& rel) const {
  if (rel.empty()) return basepath_;
  return basepath_ + "/" + rel;
}

bool LocalFSProvider::Put(const std::string& path, const std::string& content_type, int64_t length,
                          const ReadFn& read) {
  std::string full = abs(path);
  if (!mkdirs_for(full)) return false;
  std::string tmp = full + ".tmp";
  int fd = ::open(tmp.c_str(), O_WRONLY | O_CREAT | O_TRUNC | O_CLOEXEC, 0644);
  if (fd < 0) return false;
  int64_t written = 0;
  std::vector<char> buf(1 << 20);
  bool ok = true;
  while (length < 0 || written < length) {
    size_t want = buf.size();
    if (length >= 0) want = std::min<int64_t>(static_cast<int64_t>(want), length - written);
    ssize_t r = read(buf.data(), want);
    if (r < 0) {
      ok = false;
      break;
    }
    if (r == 0) break;
    ssize_t off = 0;
    while (off < r) {
      ssize_t w = ::write(fd, buf.data() + off, static_cast<size_t>(r - off));
      if (w < 0) {
        if (errno == EINTR) continue;
        ok = false;
        break;
      }
      off += w;
    }
    if (!ok) break;
    written += r;
  }
  ::close(fd);
  if (!ok || (length >= 0 && written != length)) {
    ::unlink(tmp.c_str());
    return false;
  }
  if (::rename(tmp.c_str(), full.c_str()) != 0) {
    ::unlink(tmp.c_str());
    return false;
  }
  // sidecar meta (fs_local.go:41-44)
  json::Object meta;
  meta["contentType"] = json::Value(content_type);
  meta["contentLength"] = json::Value(written);
  std::string metastr = json::Value(std::move(meta)).dump();
  int mfd = ::open((full + kMetaSuffix).c_str(), O_WRONLY | O_CREAT | O_TRUNC | O_CLOEXEC, 0644);
  if (mfd >= 0) {
    ssize_t unused = ::write(mfd, metastr.data(), metastr.size());
    (void)unused;
    ::close(mfd);
  }
  return true;
}

bool LocalFSProvider::Stat(const std::string& path, FileMeta* meta) {
  std::string full = abs(path);
  struct stat st;
  if (::stat(full.c_str(), &st) != 0 || !S_ISREG(st.st_mode)) return false;
  meta->name = path;
  meta->size = st.st_size;
  meta->last_modified = rfc3339_from_time(st.st_mtime);
  meta->content_type = "application/octet-stream";
  // sidecar
  int mfd = ::open((full + kMetaSuffix).c_str(), O_RDONLY | O_CLOEXEC);
  if (mfd >= 0) {
    char buf[4096];
    ssize_t r = ::read(mfd, buf, sizeof buf);
    ::close(mfd);
    if (r > 0) {
      try {
        auto v = json::parse(buf, static_cast<size_t>(r));
        if (v["contentType"].is_string()) meta->content_type = v["contentType"].as_string();
      } catch (...) {
      }
    }
  }
  return true;
}

std::unique_ptr<BlobReader> LocalFSProvider::Get(const std::string& path, FileMeta* meta) {
  if (!Stat(path, meta)) return nullptr;
  int fd = ::open(abs(path).c_str(), O_RDONLY | O_CLOEXEC);
  if (fd < 0) return nullptr;
  return std::make_unique<FdReader>(fd);
}

bool LocalFSProvider::Exists(const std::string& path) {
  struct stat st;
  return ::stat(abs(path).c_str(), &st) == 0 && S_ISREG(st.st_mode);
}

static bool remove_tree(const std::string& dir) {
  DIR* d = opendir(dir.c_str());
  if (!d) return ::unlink(dir.c_str()) == 0 || errno == ENOENT;
  struct dirent* e;
  while ((e = readdir(d)) != nullptr) {
    if (strcmp(e->d_name, ".") == 0 || strcmp(e->d_name, "..") == 0) continue;
    std::string child = dir + "/" + e->d_name;
    struct stat st;
    if (::lstat(child.c_str(), &st) == 0 && S_ISDIR(st.st_mode))
      remove_tree(child);
    else
      ::unlink(child.c_str());
  }
  closedir(d);
  return ::rmdir(dir.c_str()) == 0;
}

bool LocalFSProvider::Remove(const std::string& path, bool recursive) {
  std::string full = abs(path);
  if (recursive) {
    struct stat st;
    if (::stat(full.c_str(), &st) != 0) return true;  // already gone
    if (S_ISDIR(st.st_mode)) return remove_tree(full);
  }
  ::unlink((full + kMetaSuffix).c_str());
  return ::unlink(full.c_str()) == 0 || errno == ENOENT;
}

static void list_dir(const std::string& base, const std::string& rel, bool recursive,
                     LocalFSProvider* self, std::vector<FileMeta>* out) {
  std::string dir = rel.empty() ? base : base + "/" + rel;
  DIR* d = opendir(dir.c_str());
  if (!d) return;
  struct dirent* e;
  while ((e = readdir(d)) != nullptr) {
    if (strcmp(e->d_name, ".") == 0 || strcmp(e->d_name, "..") == 0) continue;
    std::string name = e->d_name;
    // skip sidecars and tmp files (fs_local.go:108,135)
    if (name.size() > 5 && name.compare(name.size() - 5, 5, ".meta") == 0) continue;
    if (name.size() > 4 && name.compare(name.size() - 4, 4, ".tmp") == 0) continue;
    std::string child_rel = rel.empty() ? name : rel + "/" + name;
    struct stat st;
    std::string child_abs = dir + "/" + name;
    if (::stat(child_abs.c_str(), &st) != 0) continue;
    if (S_ISDIR(st.st_mode)) {
      if (recursive) list_dir(base, child_rel, recursive, self, out);
    } else if (S_ISREG(st.st_mode)) {
      FileMeta m;
      self->Stat(child_rel, &m);
      m.name = child_rel;
      out->push_back(std::move(m));
    }
  }
  closedir(d);
}

std::vector<FileMeta> LocalFSProvider::List(const std::string& prefix, bool recursive) {
  std::vector<FileMeta> out;
  // prefix is a directory-ish key; entries are returned relative to prefix
  std::string p = prefix;
  while (!p.empty() && p.back() == '/') p.pop_back();
  std::vector<FileMeta> all;
  list_dir(basepath_, p, recursive, this, &all);
  for (auto& m : all) {
    if (!p.empty()) {
      if (m.name.size() <= p.size() + 1) continue;
      m.name = m.name.substr(p.size() + 1);
    }
    out.push_back(std::move(m));
  }
  return out;
}

}  // namespace store
}  // namespace modelx
