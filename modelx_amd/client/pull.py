"""Pull engine (reference: pkg/client/pull.go:19-223).

Same resume/dedup semantics: pre-hash local files and skip when the digest
matches; presigned parallel ranged download (fallback: stream through the
registry); directories re-archived locally for digest compare, then
downloaded to the .modelx cache and extracted.
"""
from __future__ import annotations

import os
from typing import List, Optional

from ..wire import digest as dg
from ..wire import errors as er
from ..wire import types
from . import extension as ext
from .progress import Bar, MultiBar
from .push import MODELX_CACHE_DIR, PULL_PUSH_CONCURRENCY


def _verify_digest_of_file(path: str, digest: str) -> bool:
    try:
        algo, _ = dg.parse(digest)
    except ValueError:
        return False
    if algo == "sha256":
        return dg.sha256_file(path) == digest
    cs = dg.algo_chunk_size(algo)
    if cs:
        return dg.chunked_digest_file(path, cs) == digest
    return False


class Puller:
    def __init__(self, remote, concurrency: int = PULL_PUSH_CONCURRENCY):
        self.remote = remote
        self.concurrency = concurrency

    def pull(self, repository: str, version: str, into_dir: str,
             blob_filter=None, quiet: Optional[bool] = None) -> types.Manifest:
        os.makedirs(into_dir, exist_ok=True)
        # one round trip for manifest + presigned locations where the
        # server supports pull plans (falls back transparently)
        plan = None
        try:
            plan = self.remote.get_pull_plan(repository, version)
        except Exception:
            plan = None
        if plan and plan.get("manifest"):
            manifest = types.Manifest.from_dict(plan["manifest"])
            plan_blobs = plan.get("blobs") or {}
        else:
            manifest = self.remote.get_manifest(repository, version)
            plan_blobs = {}
        descs: List[types.Descriptor] = [manifest.config] + list(manifest.blobs)
        if blob_filter is not None:
            descs = [d for d in descs if blob_filter(d)]
        self.pull_blobs(repository, descs, into_dir, quiet=quiet,
                        plan_blobs=plan_blobs)
        return manifest

    def pull_blobs(self, repository: str, descs: List[types.Descriptor], into_dir: str,
                   quiet: Optional[bool] = None, plan_blobs: Optional[dict] = None) -> None:
        # the pull plan is threaded through as a parameter (not instance
        # state) so concurrent pull() calls on one Puller can't cross plans
        with MultiBar("pull", self.concurrency, quiet=quiet) as mb:
            for desc in descs:
                mb.go(desc.name, desc.size,
                      lambda bar, d=desc: self._pull_one(repository, d, into_dir, bar,
                                                         plan_blobs))
            mb.wait()

    def _pull_one(self, repository: str, desc: types.Descriptor, into_dir: str,
                  bar: Optional[Bar], plan_blobs: Optional[dict] = None) -> None:
        if desc.media_type in (types.MEDIA_TYPE_MODEL_DIRECTORY_TARGZ,
                               types.MEDIA_TYPE_MODEL_DIRECTORY_TAR):
            self._pull_directory(repository, desc, into_dir, bar, plan_blobs)
        elif desc.media_type == types.MEDIA_TYPE_MODEL_LEAVES:
            if bar:
                bar.set_status("sidecar", complete=True)
        elif desc.media_type == types.MEDIA_TYPE_MODEL_FILE_ZSTD:
            self._pull_zstd_file(repository, desc, into_dir, bar, plan_blobs)
        else:
            self._pull_file(repository, desc, into_dir, bar, plan_blobs)

    # ------------------------------------------------------------- files --

    def _pull_file(self, repository: str, desc: types.Descriptor, into_dir: str,
                   bar: Optional[Bar], plan_blobs: Optional[dict] = None) -> None:
        dest = os.path.join(into_dir, desc.name)
        # skip when the local file already matches (pull.go:115-124)
        if os.path.isfile(dest) and _verify_digest_of_file(dest, desc.digest):
            if bar:
                bar.set_status("up to date", complete=True)
            return
        self.pull_blob(repository, desc, dest, bar, plan_blobs=plan_blobs)
        if desc.mode:
            os.chmod(dest, desc.mode & 0o7777)

    def _pull_zstd_file(self, repository: str, desc: types.Descriptor, into_dir: str,
                        bar: Optional[Bar], plan_blobs: Optional[dict] = None) -> None:
        """CPU pull of a +zstd blob (GPU path: GpuClient.pull_zstd_blob_to_device).
        The stored bytes are verified by pull_blob against desc.digest; the
        decompressed output is verified against the raw-digest annotation."""
        from modelx_amd import _core
        from ..wire import digest as dg

        dest = os.path.join(into_dir, desc.name)
        raw_digest = desc.annotations.get(types.ANNOTATION_RAW_DIGEST, "")
        if os.path.isfile(dest) and raw_digest and _verify_digest_of_file(dest, raw_digest):
            if bar:
                bar.set_status("up to date", complete=True)
            return
        cache_dir = os.path.join(into_dir, MODELX_CACHE_DIR)
        os.makedirs(cache_dir, exist_ok=True)
        comp_path = os.path.join(cache_dir, desc.name + ".zst")
        if not (os.path.isfile(comp_path) and _verify_digest_of_file(comp_path, desc.digest)):
            self.pull_blob(repository, desc, comp_path, bar, plan_blobs=plan_blobs)
        with open(comp_path, "rb") as f:
            raw = _core.zstd_decompress_cpu(f.read())
        if raw_digest and not dg.verify_bytes(raw, raw_digest):
            raise ValueError(f"uncompressed digest mismatch for {desc.name}")
        tmp = dest + ".part"
        with open(tmp, "wb") as f:
            f.write(raw)
        os.replace(tmp, dest)
        if desc.mode:
            os.chmod(dest, desc.mode & 0o7777)
        os.remove(comp_path)

    # ------------------------------------------------------- directories --

    def _pull_directory(self, repository: str, desc: types.Descriptor, into_dir: str,
                        bar: Optional[Bar], plan_blobs: Optional[dict] = None) -> None:
        from .helper import tgz, untgz

        compressed = desc.media_type == types.MEDIA_TYPE_MODEL_DIRECTORY_TARGZ
        target = os.path.join(into_dir, desc.name)
        if os.path.isdir(target):
            # re-archive locally and compare digests (pull.go:148-154)
            canonical, chunked, _ = tgz(target, None, compress=compressed)
            if desc.digest in (canonical, chunked):
                if bar:
                    bar.set_status("up to date", complete=True)
                return
        cache_dir = os.path.join(into_dir, MODELX_CACHE_DIR)
        os.makedirs(cache_dir, exist_ok=True)
        suffix = ".tar.gz" if compressed else ".tar"
        tgz_path = os.path.join(cache_dir, desc.name + suffix)
        # cached two-phase: download then extract (pull.go:158-182)
        if not (os.path.isfile(tgz_path) and _verify_digest_of_file(tgz_path, desc.digest)):
            self.pull_blob(repository, desc, tgz_path, bar, plan_blobs=plan_blobs)
        untgz(tgz_path, target, compressed=compressed)

    # -------------------------------------------------------------- blobs --

    @staticmethod
    def _planned_location(desc: types.Descriptor,
                          plan_blobs: Optional[dict]) -> Optional[types.BlobLocation]:
        entry = plan_blobs.get(desc.digest) if plan_blobs else None
        loc = (entry or {}).get("location")
        if not loc:
            return None
        return types.BlobLocation.from_dict(loc)

    def pull_blob(self, repository: str, desc: types.Descriptor, dest_path: str,
                  bar: Optional[Bar] = None, verify: bool = True,
                  plan_blobs: Optional[dict] = None) -> None:
        """Presigned-location download with registry-stream fallback
        (pull.go:206-215), then digest verification (the reference never
        verifies after download — we do, and re-fetch once on mismatch)."""
        for attempt in range(2):
            location = self._planned_location(desc, plan_blobs) if attempt == 0 else None
            if location is None:
                location = self.remote.get_blob_location(repository, desc, "download")
            if location is not None:
                extension = ext.get(location.provider)
                if extension is None:
                    raise ValueError(f"no extension for provider {location.provider!r}")
                extension.download(desc, location, dest_path, bar)
            else:
                os.makedirs(os.path.dirname(os.path.abspath(dest_path)), exist_ok=True)
                tmp = dest_path + ".part"
                with open(tmp, "wb") as f:
                    for chunk in self.remote.get_blob_content(repository, desc.digest):
                        f.write(chunk)
                        if bar:
                            bar.advance(len(chunk))
                os.replace(tmp, dest_path)
            if not verify or not desc.digest:
                return
            chunk_note = desc.annotations.get(types.ANNOTATION_CHUNK_DIGEST, "")
            ok = _verify_digest_of_file(dest_path, desc.digest) or (
                bool(chunk_note) and _verify_digest_of_file(dest_path, chunk_note))
            if ok:
                return
            if bar:
                bar.set_status("digest mismatch, refetching", failed=False)
        raise er.ModelxError(er.ErrCode.DIGEST_INVALID,
                             f"digest mismatch after refetch: {desc.name}")
