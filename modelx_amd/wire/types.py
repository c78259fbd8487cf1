"""Wire types — the modelx JSON contract.

Wire-compatible with the reference Go structs (reference: pkg/types/types.go:20-66).
Field names, omitempty semantics and the always-serialized ``schemaVersion`` /
``modified`` quirks are reproduced exactly:

- Go's ``json:",omitempty"`` drops zero strings/ints/maps/slices but does NOT drop
  a zero ``time.Time`` — so ``modified`` is always present on a Descriptor and
  serializes as RFC3339(Nano), zero value ``0001-01-01T00:00:00Z``.
- ``schemaVersion`` has no omitempty → always serialized (types.go:54,61).
- ``manifests`` / ``blobs`` / ``config`` have no omitempty → always serialized.
"""
from __future__ import annotations

import json
from dataclasses import dataclass, field
from datetime import datetime, timezone
from typing import Any, Dict, List, Optional

# reference: pkg/types/types.go:11-13
ANNOTATION_FILE_MODE = "filemode"

# reference: pkg/types/types.go:15-18
BLOB_LOCATION_PURPOSE_UPLOAD = "upload"
BLOB_LOCATION_PURPOSE_DOWNLOAD = "download"

# Media types — reference: pkg/client/push.go:18-22, pkg/registry/helper.go:17
MEDIA_TYPE_MODEL_INDEX_JSON = "application/vnd.modelx.model.index.v1.json"
MEDIA_TYPE_MODEL_MANIFEST_JSON = "application/vnd.modelx.model.manifest.v1.json"
MEDIA_TYPE_MODEL_CONFIG_YAML = "application/vnd.modelx.model.config.v1.yaml"
MEDIA_TYPE_MODEL_FILE = "application/vnd.modelx.model.file.v1"
MEDIA_TYPE_MODEL_DIRECTORY_TARGZ = "application/vnd.modelx.model.directory.v1.tar+gz"

# MI355X-native additions (no reference counterpart; reference clients ignore
# unknown annotations on a Descriptor, so these are interop-safe):
#   chunked GPU-verifiable digest (see modelx_amd/wire/digest.py)
ANNOTATION_CHUNK_DIGEST = "modelx.amd/chunk-digest"
#   chunk size used for the chunked digest, bytes (decimal string)
ANNOTATION_CHUNK_SIZE = "modelx.amd/chunk-size"
#   digest of the companion leaves blob (packed 32 B per-chunk SHA-256 array)
#   enabling chunk-level resume/refetch/dedup
ANNOTATION_LEAVES_BLOB = "modelx.amd/leaves-blob"
# media type of a leaves blob (listed in the manifest so GC keeps it)
MEDIA_TYPE_MODEL_LEAVES = "application/vnd.modelx.amd.leaves.v1"
# uncompressed tar directory blob — GPU-scatterable without a CPU inflate
# (optional alternative to ...directory.v1.tar+gz; reference clients treat it
# as an opaque file)
MEDIA_TYPE_MODEL_DIRECTORY_TAR = "application/vnd.modelx.amd.directory.v1.tar"
# zstd-compressed file blob: standard multi-frame zstd stream (`zstd -d`
# decodes it) + zstd-seekable-format seek table, one independent frame per
# 128 KiB of raw data so the CDNA4 kernels (core/hip/zstd.hip) decode all
# frames in parallel. The descriptor digest covers the COMPRESSED bytes
# (content addressing = stored bytes); the annotations below carry the
# uncompressed identity for post-decompress GPU verification.
MEDIA_TYPE_MODEL_FILE_ZSTD = "application/vnd.modelx.amd.file.v1+zstd"
#   chunked digest of the UNCOMPRESSED content (same scheme as chunk-digest)
ANNOTATION_RAW_DIGEST = "modelx.amd/raw-digest"
#   uncompressed size, bytes (decimal string)
ANNOTATION_RAW_SIZE = "modelx.amd/raw-size"

GO_ZERO_TIME = "0001-01-01T00:00:00Z"


def _rfc3339(dt: Optional[datetime]) -> str:
    """Format like Go time.Time MarshalJSON (RFC3339 with nanoseconds, Z suffix)."""
    if dt is None:
        return GO_ZERO_TIME
    if dt.tzinfo is None:
        dt = dt.replace(tzinfo=timezone.utc)
    dt = dt.astimezone(timezone.utc)
    base = dt.strftime("%Y-%m-%dT%H:%M:%S")
    if dt.microsecond:
        frac = f".{dt.microsecond:06d}".rstrip("0")
        return f"{base}{frac}Z"
    return f"{base}Z"


def _parse_rfc3339(s: str) -> Optional[datetime]:
    if not s or s == GO_ZERO_TIME:
        return None
    # tolerate nanosecond precision Go emits
    if s.endswith("Z"):
        s2 = s[:-1]
        if "." in s2:
            head, frac = s2.split(".", 1)
            frac = (frac + "000000")[:6]
            s2 = f"{head}.{frac}"
        return datetime.fromisoformat(s2).replace(tzinfo=timezone.utc)
    return datetime.fromisoformat(s)


@dataclass
class Descriptor:
    """reference: pkg/types/types.go:28-37"""

    name: str = ""
    media_type: str = ""
    digest: str = ""
    size: int = 0
    mode: int = 0
    urls: List[str] = field(default_factory=list)
    modified: Optional[datetime] = None
    annotations: Dict[str, str] = field(default_factory=dict)

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"name": self.name}
        if self.media_type:
            d["mediaType"] = self.media_type
        if self.digest:
            d["digest"] = self.digest
        if self.size:
            d["size"] = self.size
        if self.mode:
            d["mode"] = self.mode
        if self.urls:
            d["urls"] = self.urls
        # Go never omits modified (omitempty is a no-op on time.Time)
        d["modified"] = _rfc3339(self.modified)
        if self.annotations:
            d["annotations"] = self.annotations
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "Descriptor":
        return cls(
            name=d.get("name", "") or "",
            media_type=d.get("mediaType", "") or "",
            digest=d.get("digest", "") or "",
            size=int(d.get("size", 0) or 0),
            mode=int(d.get("mode", 0) or 0),
            urls=list(d.get("urls") or []),
            modified=_parse_rfc3339(d.get("modified", "") or ""),
            annotations=dict(d.get("annotations") or {}),
        )


@dataclass
class Index:
    """reference: pkg/types/types.go:53-58"""

    schema_version: int = 1
    media_type: str = ""
    manifests: List[Descriptor] = field(default_factory=list)
    annotations: Dict[str, str] = field(default_factory=dict)

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"schemaVersion": self.schema_version}
        if self.media_type:
            d["mediaType"] = self.media_type
        d["manifests"] = [m.to_dict() for m in self.manifests]
        if self.annotations:
            d["annotations"] = self.annotations
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "Index":
        return cls(
            schema_version=int(d.get("schemaVersion", 0) or 0),
            media_type=d.get("mediaType", "") or "",
            manifests=[Descriptor.from_dict(m) for m in (d.get("manifests") or [])],
            annotations=dict(d.get("annotations") or {}),
        )


@dataclass
class Manifest:
    """reference: pkg/types/types.go:60-66"""

    schema_version: int = 1
    media_type: str = ""
    config: Descriptor = field(default_factory=Descriptor)
    blobs: List[Descriptor] = field(default_factory=list)
    annotations: Dict[str, str] = field(default_factory=dict)

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"schemaVersion": self.schema_version}
        if self.media_type:
            d["mediaType"] = self.media_type
        d["config"] = self.config.to_dict()
        d["blobs"] = [b.to_dict() for b in self.blobs]
        if self.annotations:
            d["annotations"] = self.annotations
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "Manifest":
        return cls(
            schema_version=int(d.get("schemaVersion", 0) or 0),
            media_type=d.get("mediaType", "") or "",
            config=Descriptor.from_dict(d.get("config") or {}),
            blobs=[Descriptor.from_dict(b) for b in (d.get("blobs") or [])],
            annotations=dict(d.get("annotations") or {}),
        )

    def all_descriptors(self) -> List[Descriptor]:
        return [self.config] + list(self.blobs)


@dataclass
class BlobLocation:
    """reference: pkg/types/types.go:20-26"""

    provider: str = ""
    purpose: str = ""
    properties: Dict[str, Any] = field(default_factory=dict)

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {}
        if self.provider:
            d["provider"] = self.provider
        if self.purpose:
            d["purpose"] = self.purpose
        if self.properties:
            d["properties"] = self.properties
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "BlobLocation":
        return cls(
            provider=d.get("provider", "") or "",
            purpose=d.get("purpose", "") or "",
            properties=dict(d.get("properties") or {}),
        )


def dumps(obj) -> str:
    """Serialize a wire object compactly (Go json.Marshal style, no spaces)."""
    d = obj.to_dict() if hasattr(obj, "to_dict") else obj
    return json.dumps(d, separators=(",", ":"))


def sort_descriptors_by_name(descs: List[Descriptor]) -> List[Descriptor]:
    """reference: pkg/types/types.go:48-50 (SortDescriptorName)"""
    return sorted(descs, key=lambda d: d.name)
