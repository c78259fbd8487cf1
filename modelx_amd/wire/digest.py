"""Content digests.

Two digest algorithms:

1. ``sha256`` — canonical, byte-stream SHA-256, identical to the reference
   (pkg/client/push.go:149-161 uses go-digest's sha256). Required for full wire
   interop: a reference Go client can verify blobs we push in this mode.
   SHA-256 is strictly sequential at 64-byte granularity, so a single stream
   is CPU-bound at ~1.5-2 GB/s (SHA-NI) no matter the hardware — this is the
   reference's hidden bottleneck.

2. ``sha256c<N>`` — the MI355X-native *chunked* digest ("c" = chunked, N = chunk
   size, e.g. ``sha256c1m`` for 1 MiB chunks). Defined as::

       leaf_i = SHA256(chunk_i)                       # independent per chunk
       root   = SHA256(b"modelx-chunk-v1" || u64le(chunk_size) ||
                       u64le(total_len)   || leaf_0 || leaf_1 || ...)

   Every leaf is an independent SHA-256 chain, so a GPU hashes thousands of
   chunks concurrently (one lane per chunk, LDS-staged message blocks —
   core/hip/sha256.hip). This is what makes digest-verified pulls run at
   HBM-class bandwidth instead of 2 GB/s.

   Interop: when the Descriptor's main ``digest`` is canonical sha256, the
   chunked digest travels in the ``modelx.amd/chunk-digest`` annotation
   (reference clients ignore unknown annotations). A client may also be run
   with digest_mode="chunked" where the main digest itself is ``sha256c1m:...``
   — reference servers accept it (digest is an opaque path token server-side,
   pkg/registry/store.go:56-61), reference clients lose only their
   skip-if-present optimization.
"""
from __future__ import annotations

import hashlib
import re
from typing import Iterable, List, Optional, Tuple

DEFAULT_CHUNK_SIZE = 1 << 20  # 1 MiB
CHUNK_MAGIC = b"modelx-chunk-v1"

# go-digest grammar (also route.go:12 DigestRegexp)
_DIGEST_RE = re.compile(
    r"^(?P<algo>[A-Za-z][A-Za-z0-9]*(?:[-_+.][A-Za-z][A-Za-z0-9]*)*)"
    r":(?P<hex>[0-9a-fA-F]{32,})$"
)

_SIZE_SUFFIX = {"": 1, "k": 1 << 10, "m": 1 << 20, "g": 1 << 30}

EMPTY_SHA256 = "sha256:" + hashlib.sha256(b"").hexdigest()  # push.go:25 EmptyFileDigiest


def parse(digest: str) -> Tuple[str, str]:
    """Return (algorithm, hex); raises ValueError on bad grammar."""
    m = _DIGEST_RE.match(digest)
    if not m:
        raise ValueError(f"invalid digest: {digest!r}")
    return m.group("algo"), m.group("hex").lower()


def is_valid(digest: str) -> bool:
    return bool(_DIGEST_RE.match(digest))


def algo_chunk_size(algo: str) -> Optional[int]:
    """``sha256c1m`` -> 1 MiB; None if algo is not a chunked-sha256 algo."""
    m = re.match(r"^sha256c(\d+)([kmg]?)$", algo)
    if not m:
        return None
    return int(m.group(1)) * _SIZE_SUFFIX[m.group(2)]


def chunked_algo_name(chunk_size: int) -> str:
    for suf in ("g", "m", "k"):
        mult = _SIZE_SUFFIX[suf]
        if chunk_size % mult == 0:
            return f"sha256c{chunk_size // mult}{suf}"
    return f"sha256c{chunk_size}"


def sha256_hex(data: bytes) -> str:
    return hashlib.sha256(data).hexdigest()


def sha256_digest(data: bytes) -> str:
    return "sha256:" + hashlib.sha256(data).hexdigest()


def sha256_file(path: str, bufsize: int = 4 << 20) -> str:
    h = hashlib.sha256()
    with open(path, "rb", buffering=0) as f:
        while True:
            b = f.read(bufsize)
            if not b:
                break
            h.update(b)
    return "sha256:" + h.hexdigest()


def chunk_leaves(data: bytes, chunk_size: int = DEFAULT_CHUNK_SIZE) -> List[bytes]:
    """Per-chunk SHA-256 leaves (the CPU reference for the HIP kernel)."""
    return [
        hashlib.sha256(data[off : off + chunk_size]).digest()
        for off in range(0, max(len(data), 1), chunk_size)
    ] if data else [hashlib.sha256(b"").digest()]


def chunked_root(leaves: Iterable[bytes], chunk_size: int, total_len: int) -> str:
    h = hashlib.sha256()
    h.update(CHUNK_MAGIC)
    h.update(chunk_size.to_bytes(8, "little"))
    h.update(total_len.to_bytes(8, "little"))
    for leaf in leaves:
        h.update(leaf)
    return h.hexdigest()


def chunked_digest(data: bytes, chunk_size: int = DEFAULT_CHUNK_SIZE) -> str:
    """Full chunked digest string, e.g. ``sha256c1m:<hex>``."""
    leaves = chunk_leaves(data, chunk_size)
    return f"{chunked_algo_name(chunk_size)}:{chunked_root(leaves, chunk_size, len(data))}"


def chunked_digest_file(path: str, chunk_size: int = DEFAULT_CHUNK_SIZE) -> str:
    h = hashlib.sha256()
    h.update(CHUNK_MAGIC)
    h.update(chunk_size.to_bytes(8, "little"))
    leaves = []
    total = 0
    with open(path, "rb", buffering=0) as f:
        while True:
            b = f.read(chunk_size)
            if not b:
                break
            total += len(b)
            leaves.append(hashlib.sha256(b).digest())
    if not leaves:
        leaves = [hashlib.sha256(b"").digest()]
    h.update(total.to_bytes(8, "little"))
    for leaf in leaves:
        h.update(leaf)
    return f"{chunked_algo_name(chunk_size)}:{h.hexdigest()}"


def root_from_leaf_bytes(leaves: bytes, chunk_size: int, total_len: int) -> str:
    """Chunked digest string from a packed 32 B/leaf array (what the HIP
    kernel returns)."""
    n = len(leaves) // 32
    parts = [leaves[i * 32 : (i + 1) * 32] for i in range(n)] or [hashlib.sha256(b"").digest()]
    return f"{chunked_algo_name(chunk_size)}:{chunked_root(parts, chunk_size, total_len)}"


def verify_bytes(data: bytes, digest: str) -> bool:
    """Verify bytes against a digest string of either algorithm."""
    try:
        algo, _ = parse(digest)
    except ValueError:
        return False
    cs = algo_chunk_size(algo)
    if cs:
        return chunked_digest(data, cs) == digest
    if algo == "sha256":
        return sha256_digest(data) == digest
    return False


class StreamingDigester:
    """Incremental digest over a byte stream, computing BOTH the canonical
    sha256 and the chunked digest in one pass (used by the CPU push path;
    the GPU path computes leaves in the HIP kernel instead)."""

    def __init__(self, chunk_size: int = DEFAULT_CHUNK_SIZE, canonical: bool = True):
        self.chunk_size = chunk_size
        self._canon = hashlib.sha256() if canonical else None
        self._leaf = hashlib.sha256()
        self._leaf_fill = 0
        self.leaves: List[bytes] = []
        self.total = 0

    def update(self, data: bytes) -> None:
        if self._canon is not None:
            self._canon.update(data)
        self.total += len(data)
        view = memoryview(data)
        while view:
            take = min(len(view), self.chunk_size - self._leaf_fill)
            self._leaf.update(view[:take])
            self._leaf_fill += take
            view = view[take:]
            if self._leaf_fill == self.chunk_size:
                self.leaves.append(self._leaf.digest())
                self._leaf = hashlib.sha256()
                self._leaf_fill = 0

    def canonical_digest(self) -> Optional[str]:
        if self._canon is None:
            return None
        return "sha256:" + self._canon.hexdigest()

    def chunk_digest(self) -> str:
        leaves = list(self.leaves)
        if self._leaf_fill or not leaves:
            leaves.append(self._leaf.digest())
        return f"{chunked_algo_name(self.chunk_size)}:{chunked_root(leaves, self.chunk_size, self.total)}"
