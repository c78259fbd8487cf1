"""Transfer-extension framework — the provider-keyed data plane
(reference: pkg/client/extension.go:14-52, extension_s3.go, extension_http.go).

The "s3" extension consumes the presign properties schema
``{multipart, uploadId, parts:[{url, method, signedHeader, partNumber}]}``
(store_s3.go:228-308). Downloads are PARALLEL ranged GETs against the
presigned URL — the reference reads only parts[0] single-stream
(extension_s3.go:24-37), which SURVEY.md flags as the throughput gap.
Uploads split the blob into exactly ``len(parts)`` ranges like the reference
(calcParts, extension_s3.go:99-112) with per-part retry ×3.

This module is the CPU path (requests). The GPU pinned-ring engine
(modelx_amd.client.gpu) replaces `download` when the destination is HBM.
"""
from __future__ import annotations

import os
import threading
from concurrent.futures import ThreadPoolExecutor
from typing import BinaryIO, Callable, Dict, List, Optional

import requests

from ..wire import types
from .progress import Bar

DOWNLOAD_PART_CONCURRENCY = 8   # reference: 3 (extension_s3.go:17-20); raised for ranged GETs
UPLOAD_PART_CONCURRENCY = 4
PART_RETRIES = 3                # extension_s3.go:70,133-148
RANGE_SPLIT_MIN = 8 << 20       # don't range-split below 8 MiB


class Extension:
    def download(self, desc: types.Descriptor, location: types.BlobLocation,
                 dest_path: str, bar: Optional[Bar] = None) -> None:
        raise NotImplementedError

    def upload(self, desc: types.Descriptor, location: types.BlobLocation,
               src: "ContentSource", bar: Optional[Bar] = None) -> None:
        raise NotImplementedError


class ContentSource:
    """Random-access content (file or bytes) for ranged part uploads
    (reference: DescriptorWithContent, helper.go:14-17)."""

    def __init__(self, path: Optional[str] = None, data: Optional[bytes] = None):
        assert (path is None) != (data is None)
        self.path = path
        self.data = data

    @property
    def size(self) -> int:
        if self.data is not None:
            return len(self.data)
        return os.path.getsize(self.path)

    def read_range(self, offset: int, length: int) -> bytes:
        if self.data is not None:
            return self.data[offset : offset + length]
        with open(self.path, "rb") as f:
            f.seek(offset)
            return f.read(length)


GLOBAL_EXTENSIONS: Dict[str, Extension] = {}


def register(provider: str, ext: Extension) -> None:
    GLOBAL_EXTENSIONS[provider] = ext


def get(provider: str) -> Optional[Extension]:
    return GLOBAL_EXTENSIONS.get(provider)


def _signed_headers(part: Dict) -> Dict[str, str]:
    out = {}
    for k, v in (part.get("signedHeader") or {}).items():
        out[k] = ",".join(v) if isinstance(v, list) else str(v)
    return out


def calc_parts(size: int, count: int) -> List[tuple]:
    """(offset, length) per part; exactly `count` parts covering size
    (reference: extension_s3.go:99-112)."""
    if count <= 0:
        count = 1
    base = size // count
    parts = []
    off = 0
    for i in range(count):
        ln = base if i < count - 1 else size - off
        parts.append((off, ln))
        off += ln
    return parts


class S3Extension(Extension):
    """Presigned-URL direct-to-storage transfers (registers provider "s3")."""

    def __init__(self):
        self._local = threading.local()

    def _session(self) -> requests.Session:
        s = getattr(self._local, "session", None)
        if s is None:
            s = requests.Session()
            adapter = requests.adapters.HTTPAdapter(pool_connections=32, pool_maxsize=32)
            s.mount("http://", adapter)
            s.mount("https://", adapter)
            # self-signed TLS object stores: same switch the native engine
            # honors for https presigned URLs
            if os.environ.get("MODELX_TLS_INSECURE") == "1":
                s.verify = False
                s.trust_env = False  # CURL_CA_BUNDLE would override verify
                import urllib3

                urllib3.disable_warnings()
            self._local.session = s
        return s

    # ---------------------------------------------------------- download --

    def download(self, desc: types.Descriptor, location: types.BlobLocation,
                 dest_path: str, bar: Optional[Bar] = None) -> None:
        parts = location.properties.get("parts") or []
        if not parts:
            raise ValueError("no parts in blob location")
        url = parts[0]["url"]
        headers = _signed_headers(parts[0])
        size = desc.size or int(location.properties.get("size") or 0)

        os.makedirs(os.path.dirname(os.path.abspath(dest_path)), exist_ok=True)
        tmp = dest_path + ".part"

        if size < RANGE_SPLIT_MIN or size == 0:
            self._download_single(url, headers, tmp, bar)
        else:
            nranges = min(DOWNLOAD_PART_CONCURRENCY * 2, max(1, size // RANGE_SPLIT_MIN))
            ranges = calc_parts(size, nranges)
            with open(tmp, "wb") as f:
                f.truncate(size)
            errors: List[BaseException] = []
            with ThreadPoolExecutor(max_workers=DOWNLOAD_PART_CONCURRENCY) as pool:
                futs = [
                    pool.submit(self._download_range, url, headers, tmp, off, ln, bar)
                    for off, ln in ranges
                ]
                for fu in futs:
                    try:
                        fu.result()
                    except BaseException as e:
                        errors.append(e)
            if errors:
                raise errors[0]
        os.replace(tmp, dest_path)

    def _download_single(self, url: str, headers: Dict[str, str], dest: str,
                         bar: Optional[Bar]) -> None:
        last = None
        for _ in range(PART_RETRIES):
            try:
                with self._session().get(url, headers=headers, stream=True, timeout=300) as r:
                    r.raise_for_status()
                    with open(dest, "wb") as f:
                        for chunk in r.iter_content(chunk_size=1 << 20):
                            f.write(chunk)
                            if bar:
                                bar.advance(len(chunk))
                return
            except requests.RequestException as e:
                last = e
        raise last

    def _download_range(self, url: str, headers: Dict[str, str], dest: str, offset: int,
                        length: int, bar: Optional[Bar]) -> None:
        h = dict(headers)
        h["Range"] = f"bytes={offset}-{offset + length - 1}"
        frag = bar.add_fragment(offset, length) if bar else None
        last = None
        for _ in range(PART_RETRIES):
            try:
                if frag:
                    frag.done = 0
                with self._session().get(url, headers=h, stream=True, timeout=300) as r:
                    r.raise_for_status()
                    with open(dest, "r+b") as f:
                        f.seek(offset)
                        for chunk in r.iter_content(chunk_size=1 << 20):
                            f.write(chunk)
                            if bar:
                                bar.advance(len(chunk), frag)
                return
            except requests.RequestException as e:
                last = e
        raise last

    # ------------------------------------------------------------ upload --

    def upload(self, desc: types.Descriptor, location: types.BlobLocation,
               src: ContentSource, bar: Optional[Bar] = None) -> None:
        parts = location.properties.get("parts") or []
        if not parts:
            raise ValueError("no parts in blob location")
        size = src.size
        ranges = calc_parts(size, len(parts))
        errors: List[BaseException] = []
        with ThreadPoolExecutor(max_workers=UPLOAD_PART_CONCURRENCY) as pool:
            futs = []
            for part, (off, ln) in zip(parts, ranges):
                futs.append(pool.submit(self._upload_part, part, src, off, ln, bar))
            for fu in futs:
                try:
                    fu.result()
                except BaseException as e:
                    errors.append(e)
        if errors:
            raise errors[0]

    def _upload_part(self, part: Dict, src: ContentSource, offset: int, length: int,
                     bar: Optional[Bar]) -> None:
        url = part["url"]
        method = part.get("method") or ("PUT" if "X-Amz-Credential" in url else "POST")
        headers = _signed_headers(part)
        frag = bar.add_fragment(offset, length) if bar else None
        data = src.read_range(offset, length)
        last: Optional[BaseException] = None
        for _ in range(PART_RETRIES):  # extension_s3.go:133-148
            try:
                r = self._session().request(method, url, headers=headers, data=data, timeout=600)
                r.raise_for_status()
                if frag:
                    frag.done = length
                elif bar:
                    bar.advance(length)
                return
            except requests.RequestException as e:
                last = e
        raise last


register("s3", S3Extension())
