// FSRegistryStore — semantic store on any FSProvider
// (reference: pkg/registry/store_fs.go, gc.go). Index-refresh, sorting and
// annotation-adoption rules are identical; ListBlobs is implemented for real
// (the reference returns nil → its GC is a no-op, SURVEY.md §6 defects).
#include <regex>
#include <set>

#include "modelx/store.hpp"

namespace modelx {
namespace store {

static bool filter_index(wire::Index* idx, const std::string& search) {
  if (search.empty()) return true;
  try {
    std::regex re(search);
    std::vector<wire::Descriptor> kept;
    for (auto& m : idx->manifests)
      if (std::regex_search(m.name, re)) kept.push_back(m);
    idx->manifests = std::move(kept);
    return true;
  } catch (const std::regex_error&) {
    return false;
  }
}

bool RegistryStore::GetGlobalIndex(const std::string& search, wire::Index* out) {
  std::string data;
  if (!fs_->GetBytes(wire::index_path(""), &data)) {
    // empty registry → empty index (first boot builds it)
    RefreshGlobalIndex();
    if (!fs_->GetBytes(wire::index_path(""), &data)) {
      *out = wire::Index{};
      return true;
    }
  }
  try {
    *out = wire::Index::from_json(json::parse(data));
  } catch (...) {
    return false;
  }
  return filter_index(out, search);
}

bool RegistryStore::GetIndex(const std::string& repository, const std::string& search,
                             wire::Index* out) {
  std::string data;
  if (!fs_->GetBytes(wire::index_path(repository), &data)) return false;
  try {
    *out = wire::Index::from_json(json::parse(data));
  } catch (...) {
    return false;
  }
  return filter_index(out, search);
}

bool RegistryStore::RemoveIndex(const std::string& repository) {
  if (!fs_->Remove(repository, true)) return false;
  return RefreshIndex(repository);
}

bool RegistryStore::ExistsManifest(const std::string& repository, const std::string& reference) {
  return fs_->Exists(wire::manifest_path(repository, reference));
}

bool RegistryStore::GetManifest(const std::string& repository, const std::string& reference,
                                wire::Manifest* out) {
  std::string data;
  if (!fs_->GetBytes(wire::manifest_path(repository, reference), &data)) return false;
  try {
    *out = wire::Manifest::from_json(json::parse(data));
  } catch (...) {
    return false;
  }
  return true;
}

bool RegistryStore::PutManifest(const std::string& repository, const std::string& reference,
                                const std::string& content_type, const wire::Manifest& manifest,
                                std::string* err) {
  std::string body = manifest.to_json().dump();
  if (!fs_->PutBytes(wire::manifest_path(repository, reference), content_type, body)) {
    if (err) *err = "failed to store manifest";
    return false;
  }
  if (!RefreshIndex(repository)) {
    if (err) *err = "failed to refresh index";
    return false;
  }
  return true;
}

bool RegistryStore::DeleteManifest(const std::string& repository, const std::string& reference) {
  if (!fs_->Remove(wire::manifest_path(repository, reference), false)) return false;
  return RefreshIndex(repository);
}

std::vector<std::string> RegistryStore::ListBlobs(const std::string& repository) {
  std::vector<std::string> out;
  for (auto& m : fs_->List(repository + "/blobs", true)) {
    // name is "<algo>/<hex>"
    size_t slash = m.name.find('/');
    if (slash == std::string::npos) continue;
    out.push_back(m.name.substr(0, slash) + ":" + m.name.substr(slash + 1));
  }
  return out;
}

std::unique_ptr<BlobReader> RegistryStore::GetBlob(const std::string& repository,
                                                   const std::string& digest, FileMeta* meta) {
  return fs_->Get(wire::blob_digest_path(repository, digest), meta);
}

bool RegistryStore::DeleteBlob(const std::string& repository, const std::string& digest) {
  return fs_->Remove(wire::blob_digest_path(repository, digest), false);
}

bool RegistryStore::PutBlob(const std::string& repository, const std::string& digest,
                            const std::string& content_type, int64_t length, const ReadFn& read) {
  return fs_->Put(wire::blob_digest_path(repository, digest), content_type, length, read);
}

bool RegistryStore::ExistsBlob(const std::string& repository, const std::string& digest) {
  return fs_->Exists(wire::blob_digest_path(repository, digest));
}

BlobLocationResult RegistryStore::GetBlobLocation(
    const std::string&, const std::string&, const std::string&,
    const std::map<std::string, std::string>&) {
  return BlobLocationResult{};  // local FS: unsupported (store_fs.go:391-395)
}

bool RegistryStore::PutIndex(const std::string& repository, wire::Index index) {
  std::sort(index.manifests.begin(), index.manifests.end(),
            [](const wire::Descriptor& a, const wire::Descriptor& b) { return a.name < b.name; });
  // adopt first manifest's annotations (store_fs.go:150-157)
  for (auto& m : index.manifests) {
    if (!m.annotations.empty()) {
      index.annotations = m.annotations;
      break;
    }
  }
  return fs_->PutBytes(wire::index_path(repository), wire::kMediaTypeIndex,
                       index.to_json().dump());
}

bool RegistryStore::RefreshIndex(const std::string& repository) {
  std::lock_guard<std::mutex> lock(index_mu_);
  auto metas = fs_->List(wire::manifest_path(repository, ""), false);
  wire::Index index;
  for (auto& meta : metas) {
    wire::Manifest manifest;
    if (!GetManifest(repository, meta.name, &manifest)) continue;
    wire::Descriptor desc;
    desc.name = meta.name;
    desc.modified = meta.last_modified;
    desc.annotations = manifest.annotations;
    desc.size = manifest.config.size;
    for (auto& b : manifest.blobs) desc.size += b.size;
    index.manifests.push_back(std::move(desc));
  }
  if (!index.manifests.empty()) {
    if (!PutIndex(repository, std::move(index))) return false;
  } else {
    // all manifests gone → drop stale index so the repo disappears
    fs_->Remove(wire::index_path(repository), false);
  }
  return RefreshGlobalIndexLocked();
}

bool RegistryStore::RefreshGlobalIndex() {
  std::lock_guard<std::mutex> lock(index_mu_);
  return RefreshGlobalIndexLocked();
}

bool RegistryStore::RefreshGlobalIndexLocked() {
  auto metas = fs_->List("", true);
  wire::Index global;
  for (auto& meta : metas) {
    if (meta.name == "index.json") continue;
    size_t slash = meta.name.rfind('/');
    if (slash == std::string::npos) continue;
    if (meta.name.substr(slash + 1) != "index.json") continue;
    std::string repository = meta.name.substr(0, slash);
    wire::Index idx;
    if (!GetIndex(repository, "", &idx)) continue;
    wire::Descriptor desc;
    desc.name = repository;
    desc.media_type = wire::kMediaTypeIndex;
    desc.annotations = idx.annotations;
    global.manifests.push_back(std::move(desc));
  }
  std::sort(global.manifests.begin(), global.manifests.end(),
            [](const wire::Descriptor& a, const wire::Descriptor& b) { return a.name < b.name; });
  return fs_->PutBytes(wire::index_path(""), wire::kMediaTypeIndex, global.to_json().dump());
}

// mark-and-sweep (reference: pkg/registry/gc.go:10-68, functional here)
int RegistryStore::GCBlobs(const std::string& repository) {
  wire::Index index;
  if (!GetIndex(repository, "", &index)) return 0;
  std::set<std::string> marked;
  for (auto& mdesc : index.manifests) {
    wire::Manifest manifest;
    if (!GetManifest(repository, mdesc.name, &manifest)) continue;
    marked.insert(manifest.config.digest);
    for (auto& b : manifest.blobs) marked.insert(b.digest);
  }
  int removed = 0;
  for (auto& d : ListBlobs(repository)) {
    if (marked.count(d)) continue;
    if (DeleteBlob(repository, d)) removed++;
  }
  return removed;
}

int RegistryStore::GCBlobsAll() {
  wire::Index global;
  if (!GetGlobalIndex("", &global)) return 0;
  int removed = 0;
  for (auto& repo : global.manifests) removed += GCBlobs(repo.name);
  return removed;
}

}  // namespace store
}  // namespace modelx
