"""Push engine (reference: pkg/client/push.go:29-207).

Same protocol: ParseManifest (dir scan) → per-blob HEAD dedup → presigned
upload (fallback: direct PUT through the registry) → PutManifest last (the
commit point). The reference's pushBlob fallback nil-deref (push.go:196-207,
SURVEY.md defects) is fixed: the fallback RETURNS after success.

Digests: canonical sha256 + the GPU-verifiable chunked digest as an
annotation (see modelx_amd/wire/digest.py). digest_mode="chunked" makes the
chunked digest the descriptor digest itself (fast path for huge blobs).
"""
from __future__ import annotations

import os
import stat
from datetime import datetime, timezone
from typing import List, Optional

from ..wire import digest as dg
from ..wire import types
from . import extension as ext
from .progress import Bar, MultiBar

PULL_PUSH_CONCURRENCY = 3  # push.go:27
MODELX_CACHE_DIR = ".modelx"


def _file_descriptor(path: str, name: str, chunk_size: int, digest_mode: str) -> types.Descriptor:
    st = os.stat(path)
    d = dg.StreamingDigester(chunk_size=chunk_size)
    with open(path, "rb", buffering=0) as f:
        while True:
            b = f.read(4 << 20)
            if not b:
                break
            d.update(b)
    canonical = d.canonical_digest()
    chunked = d.chunk_digest()
    desc = types.Descriptor(
        name=name,
        media_type=types.MEDIA_TYPE_MODEL_FILE,
        digest=chunked if digest_mode == "chunked" else canonical,
        size=st.st_size,
        mode=stat.S_IMODE(st.st_mode),
        modified=datetime.fromtimestamp(st.st_mtime, tz=timezone.utc),
        annotations={
            types.ANNOTATION_CHUNK_DIGEST: chunked,
            types.ANNOTATION_CHUNK_SIZE: str(chunk_size),
        },
    )
    return desc


def _zstd_file_descriptor(path: str, name: str, chunk_size: int,
                          digest_mode: str, cache_dir: str) -> types.Descriptor:
    """Compress a file into the seekable multi-frame zstd format (CPU path
    of core/hip/zstd.hip's codec) and describe the COMPRESSED blob; the
    uncompressed identity travels in raw-digest/raw-size annotations."""
    from modelx_amd import _core

    st = os.stat(path)
    with open(path, "rb") as f:
        raw = f.read()
    blob = _core.zstd_compress_cpu(raw, 128 << 10)
    comp_path = os.path.join(cache_dir, name + ".zst")
    with open(comp_path, "wb") as f:
        f.write(blob)
    raw_chunked = dg.chunked_digest(raw, chunk_size)
    canonical = dg.sha256_digest(blob)
    chunked = dg.chunked_digest(blob, chunk_size)
    return types.Descriptor(
        name=name,
        media_type=types.MEDIA_TYPE_MODEL_FILE_ZSTD,
        digest=chunked if digest_mode == "chunked" else canonical,
        size=len(blob),
        mode=stat.S_IMODE(st.st_mode),
        modified=datetime.fromtimestamp(st.st_mtime, tz=timezone.utc),
        annotations={
            types.ANNOTATION_CHUNK_DIGEST: chunked,
            types.ANNOTATION_CHUNK_SIZE: str(chunk_size),
            types.ANNOTATION_RAW_DIGEST: raw_chunked,
            types.ANNOTATION_RAW_SIZE: str(len(raw)),
        },
    )


def parse_manifest(basedir: str, configfile: str = "modelx.yaml",
                   chunk_size: int = dg.DEFAULT_CHUNK_SIZE,
                   digest_mode: str = "sha256",
                   dir_format: str = "tar+gz", compress: str = "") -> types.Manifest:
    """Scan basedir into a Manifest (reference: push.go:67-100).
    Dot-files skipped; directories become tar.gz descriptors.
    compress="zstd" stores file blobs in the seekable +zstd format."""
    manifest = types.Manifest(media_type=types.MEDIA_TYPE_MODEL_MANIFEST_JSON)
    cache_dir = os.path.join(basedir, MODELX_CACHE_DIR)
    os.makedirs(cache_dir, exist_ok=True)
    entries = sorted(os.listdir(basedir))
    for entry in entries:
        if entry.startswith("."):  # push.go:76-78
            continue
        path = os.path.join(basedir, entry)
        if entry == configfile:
            desc = _file_descriptor(path, entry, chunk_size, digest_mode)
            desc.media_type = types.MEDIA_TYPE_MODEL_CONFIG_YAML
            manifest.config = desc
            continue
        if os.path.isdir(path):
            # dirs → tar[.gz] blob (push.go:86-92,102-118), archived to cache
            from .helper import tgz

            compressed = dir_format != "tar"
            suffix = ".tar.gz" if compressed else ".tar"
            tgz_path = os.path.join(cache_dir, entry + suffix)
            canonical, chunked, size = tgz(path, tgz_path, chunk_size, compress=compressed)
            desc = types.Descriptor(
                name=entry,
                media_type=types.MEDIA_TYPE_MODEL_DIRECTORY_TARGZ if compressed
                else types.MEDIA_TYPE_MODEL_DIRECTORY_TAR,
                digest=chunked if digest_mode == "chunked" else canonical,
                size=size,
                modified=datetime.fromtimestamp(os.stat(path).st_mtime, tz=timezone.utc),
                annotations={
                    types.ANNOTATION_CHUNK_DIGEST: chunked,
                    types.ANNOTATION_CHUNK_SIZE: str(chunk_size),
                },
            )
            manifest.blobs.append(desc)
        elif compress == "zstd":
            manifest.blobs.append(
                _zstd_file_descriptor(path, entry, chunk_size, digest_mode, cache_dir))
        else:
            manifest.blobs.append(_file_descriptor(path, entry, chunk_size, digest_mode))
    manifest.blobs = types.sort_descriptors_by_name(manifest.blobs)  # push.go:98
    if not manifest.config.name:
        raise FileNotFoundError(f"config file {configfile!r} not found in {basedir}")
    return manifest


class Pusher:
    def __init__(self, remote, concurrency: int = PULL_PUSH_CONCURRENCY):
        self.remote = remote
        self.concurrency = concurrency

    def push(self, repository: str, version: str, basedir: str,
             configfile: str = "modelx.yaml", digest_mode: str = "sha256",
             chunk_size: int = dg.DEFAULT_CHUNK_SIZE, quiet: Optional[bool] = None,
             dir_format: str = "tar+gz", compress: str = "") -> types.Manifest:
        manifest = parse_manifest(basedir, configfile, chunk_size, digest_mode, dir_format,
                                  compress)
        with MultiBar(f"push {repository}@{version}", self.concurrency, quiet=quiet) as mb:
            descs: List[types.Descriptor] = [manifest.config] + list(manifest.blobs)
            for desc in descs:
                src_path = self._blob_source(basedir, desc)
                mb.go(desc.name, desc.size,
                      lambda bar, d=desc, p=src_path: self.push_blob(repository, d, p, bar))
            mb.wait()
        # PutManifest LAST — the commit point (push.go:29-65; docs/api.md 约定)
        self.remote.put_manifest(repository, version, manifest)
        return manifest

    @staticmethod
    def _blob_source(basedir: str, desc: types.Descriptor) -> str:
        if desc.media_type == types.MEDIA_TYPE_MODEL_DIRECTORY_TARGZ:
            return os.path.join(basedir, MODELX_CACHE_DIR, desc.name + ".tar.gz")
        if desc.media_type == types.MEDIA_TYPE_MODEL_DIRECTORY_TAR:
            return os.path.join(basedir, MODELX_CACHE_DIR, desc.name + ".tar")
        if desc.media_type == types.MEDIA_TYPE_MODEL_FILE_ZSTD:
            return os.path.join(basedir, MODELX_CACHE_DIR, desc.name + ".zst")
        return os.path.join(basedir, desc.name)

    def push_blob(self, repository: str, desc: types.Descriptor, src_path: str,
                  bar: Optional[Bar] = None) -> None:
        # skip empty (push.go:166-168)
        if desc.size == 0 or desc.digest == dg.EMPTY_SHA256:
            if bar:
                bar.set_status("empty", complete=True)
            return
        # HEAD dedup (push.go:169-177) — resume-after-interrupt comes free
        if self.remote.head_blob(repository, desc.digest):
            if bar:
                bar.set_status("exists", complete=True)
            return
        src = ext.ContentSource(path=src_path)
        location = self.remote.get_blob_location(repository, desc, "upload")
        if location is not None:
            extension = ext.get(location.provider)
            if extension is None:
                raise ValueError(f"no extension for provider {location.provider!r}")
            extension.upload(desc, location, src, bar)
            return
        # fallback: direct PUT through the registry (push.go:196-207 — with the
        # missing-return defect fixed)
        with open(src_path, "rb") as f:
            self.remote.upload_blob_content(repository, desc, f)
        if bar:
            bar.advance(desc.size)
