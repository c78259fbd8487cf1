"""Storage key layout — identical to the reference so on-disk/S3 registries
interop (reference: pkg/registry/store.go:56-74, fs_s3.go:77 prefix)."""
from __future__ import annotations

import posixpath
from typing import Tuple

from . import digest as digestmod

REGISTRY_INDEX_FILENAME = "index.json"
# key prefix inside the S3 bucket (reference: pkg/registry/fs_s3.go:77)
S3_KEY_PREFIX = "registry"


def blob_digest_path(repository: str, digest: str) -> str:
    """``<repo>/blobs/<algo>/<hex>`` (store.go:56-61)."""
    if not digest:
        algo, hexpart = "", ""
    else:
        algo, hexpart = digestmod.parse(digest)
    return posixpath.join(repository, "blobs", algo, hexpart)


def index_path(repository: str) -> str:
    """``<repo>/index.json`` (store.go:63-65)."""
    return posixpath.join(repository, REGISTRY_INDEX_FILENAME)


def manifest_path(repository: str, reference: str) -> str:
    """``<repo>/manifests/<ref>`` (store.go:67-69)."""
    return posixpath.join(repository, "manifests", reference)


def split_manifest_path(p: str) -> Tuple[str, str]:
    """(store.go:71-74)"""
    if p.startswith("manifests"):
        p = p[len("manifests"):]
    head, tail = posixpath.split(p)
    if head and not head.endswith("/"):
        head += "/"
    return head, tail
