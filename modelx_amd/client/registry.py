"""Registry HTTP client (reference: pkg/client/registry.go:28-191).
Same endpoints, headers and error decoding; User-Agent ``modelx/<ver>``."""
from __future__ import annotations

import json
import threading
from typing import Any, Dict, Iterator, Optional

import requests

from .._version import __version__
from ..wire import errors as er
from ..wire import types


class _InsecureSession(requests.Session):
    """requests >= 2.32 stopped honoring session-level ``verify=False`` in
    some paths — inject it per request (the --insecure TLS skip,
    reference: cmd/modelx/modelx.go:29-36)."""

    def request(self, *args, **kwargs):  # noqa: D102
        kwargs.setdefault("verify", False)
        return super().request(*args, **kwargs)


class _SessionPerThread:
    """requests.Session is not thread-safe (psf/requests#1871): a shared
    session under concurrent use (GpuClient.pull_many) can cross responses
    between threads — observed on hardware as a pull landing ANOTHER blob's
    presigned content (self-consistent wrong bytes, digest mismatch on
    every chunk). One session per thread, same API surface."""

    def __init__(self, insecure: bool):
        self._insecure = insecure
        self._local = threading.local()

    def _s(self) -> requests.Session:
        s = getattr(self._local, "s", None)
        if s is None:
            s = _InsecureSession() if self._insecure else requests.Session()
            s.verify = not self._insecure
            self._local.s = s
        return s

    def __getattr__(self, name):
        return getattr(self._s(), name)


class RegistryClient:
    def __init__(self, registry: str, authorization: str = "", insecure: bool = False):
        self.registry = registry.rstrip("/")
        self.authorization = authorization
        self.session = _SessionPerThread(insecure)
        self.user_agent = f"modelx/{__version__}"

    # --- internals -------------------------------------------------------

    def _headers(self, extra: Optional[Dict[str, str]] = None) -> Dict[str, str]:
        h = {"User-Agent": self.user_agent}
        if self.authorization:
            h["Authorization"] = self.authorization
        if extra:
            h.update(extra)
        return h

    def _raise_for(self, resp: requests.Response) -> None:
        """Decode the JSON error body (registry.go:146-191 request())."""
        if resp.status_code < 400:
            return
        try:
            body = resp.json()
            raise er.ModelxError.from_dict(body, http_status=resp.status_code)
        except (ValueError, KeyError):
            raise er.ModelxError(
                er.ErrCode.UNKNOWN,
                f"HTTP {resp.status_code}: {resp.text[:200]}",
                http_status=resp.status_code,
            )

    def _url(self, *parts: str) -> str:
        return self.registry + "/" + "/".join(parts)

    # --- API (reference: pkg/client/registry.go) -------------------------

    def get_global_index(self, search: str = "") -> types.Index:
        params = {"search": search} if search else {}
        r = self.session.get(self.registry + "/", headers=self._headers(), params=params)
        self._raise_for(r)
        return types.Index.from_dict(r.json())

    def get_index(self, repository: str, search: str = "") -> types.Index:
        params = {"search": search} if search else {}
        r = self.session.get(self._url(repository, "index"), headers=self._headers(), params=params)
        self._raise_for(r)
        return types.Index.from_dict(r.json())

    def delete_index(self, repository: str) -> None:
        r = self.session.delete(self._url(repository, "index"), headers=self._headers())
        self._raise_for(r)

    def get_manifest(self, repository: str, version: str = "") -> types.Manifest:
        if not version:
            version = "latest"  # registry.go:34-36
        r = self.session.get(self._url(repository, "manifests", version), headers=self._headers())
        self._raise_for(r)
        return types.Manifest.from_dict(r.json())

    def put_manifest(self, repository: str, version: str, manifest: types.Manifest) -> None:
        if not version:
            version = "latest"
        r = self.session.put(
            self._url(repository, "manifests", version),
            headers=self._headers({"Content-Type": types.MEDIA_TYPE_MODEL_MANIFEST_JSON}),
            data=types.dumps(manifest),
        )
        self._raise_for(r)

    def delete_manifest(self, repository: str, version: str) -> None:
        r = self.session.delete(self._url(repository, "manifests", version), headers=self._headers())
        self._raise_for(r)

    def head_blob(self, repository: str, digest: str) -> bool:
        r = self.session.head(self._url(repository, "blobs", digest), headers=self._headers())
        if r.status_code == 404:
            return False
        if r.status_code == 200:
            return True
        self._raise_for(r)
        return False

    def get_blob_content(self, repository: str, digest: str, chunk_size: int = 1 << 20
                         ) -> Iterator[bytes]:
        """Stream blob bytes via the registry (fallback when no location)."""
        r = self.session.get(self._url(repository, "blobs", digest), headers=self._headers(),
                             stream=True)
        self._raise_for(r)
        return r.iter_content(chunk_size=chunk_size)

    def upload_blob_content(self, repository: str, desc: types.Descriptor, data) -> None:
        """Direct PUT through the registry (fallback; data: bytes or file-like)."""
        r = self.session.put(
            self._url(repository, "blobs", desc.digest),
            headers=self._headers({"Content-Type": desc.media_type or "application/octet-stream"}),
            data=data,
        )
        self._raise_for(r)

    def get_blob_location(self, repository: str, desc: types.Descriptor, purpose: str,
                          extra: Optional[Dict[str, str]] = None) -> Optional[types.BlobLocation]:
        """Query params mirror registry.go:95-100 (size/name/media-type) plus
        our part-count hint. Returns None when the server says UNSUPPORTED."""
        params: Dict[str, Any] = {
            "size": str(desc.size),
            "name": desc.name,
            "media-type": desc.media_type,
        }
        if extra:
            params.update(extra)
        r = self.session.get(
            self._url(repository, "blobs", desc.digest, "locations", purpose),
            headers=self._headers(),
            params=params,
        )
        if r.status_code == 404 or r.status_code == 501:
            return None
        self._raise_for(r)
        return types.BlobLocation.from_dict(r.json())

    def get_pull_plan(self, repository: str, version: str = "") -> Optional[Dict[str, Any]]:
        """One-round-trip pull metadata: manifest + per-blob presigned
        download locations + (small) inlined leaves sidecars. Returns None
        when the server doesn't support the endpoint (MI355X-native
        addition; reference servers 404 here and the client falls back to
        per-blob calls). Inlined leaves are verified by the caller against
        the annotation digest — the plan is an optimization, not a trust
        root."""
        r = self.session.get(
            self._url(repository, "manifests", version or "latest", "pull-plan"),
            headers=self._headers())
        if r.status_code in (404, 405, 501):
            return None
        self._raise_for(r)
        return r.json()

    def get_pull_plans(self, repository: str, refs) -> Optional[Dict[str, Any]]:
        """Batched pull plans: ONE POST for many versions (config-5-shaped
        indexes store one small blob per version, so the per-version GET
        still paid a round trip each). Returns {ref: plan} or None when the
        server lacks the endpoint; unknown refs are omitted (the caller
        falls back per-ref)."""
        refs = list(refs)
        if not refs:
            return {}
        r = self.session.post(self._url(repository, "pull-plans"),
                              headers=self._headers(), json={"refs": refs})
        if r.status_code in (404, 405, 501):
            return None
        self._raise_for(r)
        return (r.json() or {}).get("plans") or {}

    def garbage_collect(self, repository: str) -> Dict[str, Any]:
        r = self.session.post(self._url(repository, "garbage-collect"), headers=self._headers())
        self._raise_for(r)
        return r.json()

    def healthz(self) -> bool:
        try:
            r = self.session.get(self.registry + "/healthz", timeout=5)
            return r.status_code == 200
        except requests.RequestException:
            return False
