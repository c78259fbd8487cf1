"""tar.gz archive helpers with digest tee
(reference: pkg/client/helper.go:19-79 TGZ/UnTGZ)."""
from __future__ import annotations

import gzip
import io
import os
import tarfile
from typing import Optional, Tuple

from ..wire.digest import StreamingDigester


class _TeeWriter(io.RawIOBase):
    def __init__(self, sink, digester: StreamingDigester):
        self.sink = sink
        self.digester = digester

    def writable(self):
        return True

    def write(self, b):
        self.digester.update(bytes(b))
        if self.sink is not None:
            self.sink.write(b)
        return len(b)


def tgz(src_dir: str, out_file: Optional[str], chunk_size: int = 1 << 20,
        compress: bool = True) -> Tuple[str, str, int]:
    """Archive src_dir to tar[.gz] (optionally writing out_file), digesting
    the output stream. Returns (sha256_digest, chunk_digest, size).
    Deterministic: entries sorted, gzip mtime zeroed → digests stable.
    compress=False emits plain tar (GPU-scatterable, see core/hip/tar.hip)."""
    digester = StreamingDigester(chunk_size=chunk_size)
    sink = open(out_file, "wb") if out_file else None
    try:
        tee = _TeeWriter(sink, digester)
        ctx = gzip.GzipFile(fileobj=tee, mode="wb", mtime=0) if compress else None
        stream = ctx if compress else tee
        try:
            with tarfile.open(fileobj=stream, mode="w|") as tar:
                base = os.path.basename(src_dir.rstrip("/"))
                entries = []
                for root, dirs, files in os.walk(src_dir):
                    dirs.sort()
                    for f in sorted(files):
                        entries.append(os.path.join(root, f))
                for path in entries:
                    arcname = os.path.join(base, os.path.relpath(path, src_dir))
                    tar.add(path, arcname=arcname, recursive=False)
        finally:
            if ctx is not None:
                ctx.close()
    finally:
        if sink:
            sink.close()
    return digester.canonical_digest(), digester.chunk_digest(), digester.total


def tar_plain(src_dir: str, out_file: Optional[str], chunk_size: int = 1 << 20):
    return tgz(src_dir, out_file, chunk_size, compress=False)


def untgz(archive_path: str, dest_dir: str, compressed: bool = True) -> None:
    """Extract tar[.gz] stripping the top-level directory component
    (reference: helper.go:55-79 extracts into dest)."""
    os.makedirs(dest_dir, exist_ok=True)
    with tarfile.open(archive_path, mode="r:gz" if compressed else "r:") as tar:
        for member in tar.getmembers():
            parts = member.name.split("/", 1)
            member.name = parts[1] if len(parts) == 2 else parts[0]
            if not member.name:
                continue
            if member.name.startswith("/") or ".." in member.name.split("/"):
                raise ValueError(f"unsafe tar member: {member.name}")
            tar.extract(member, dest_dir)


def digest_tgz_of_dir(src_dir: str, chunk_size: int = 1 << 20) -> Tuple[str, str, int]:
    """Digest the tgz of a local dir without writing it (pull-side compare,
    reference: pull.go:148-154)."""
    return tgz(src_dir, None, chunk_size)
