"""Client facade (reference: pkg/client/client.go:9-26)."""
from __future__ import annotations

from typing import Optional

from ..wire import types
from .extension import GLOBAL_EXTENSIONS, ContentSource, S3Extension  # noqa: F401
from .pull import Puller
from .push import Pusher, parse_manifest  # noqa: F401
from .registry import RegistryClient


class Client:
    def __init__(self, registry: str, authorization: str = "", insecure: bool = False,
                 concurrency: int = 3):
        self.remote = RegistryClient(registry, authorization, insecure=insecure)
        self.pusher = Pusher(self.remote, concurrency=concurrency)
        self.puller = Puller(self.remote, concurrency=concurrency)

    def ping(self) -> types.Index:
        """reference: client.go:21-26 (Ping = GetGlobalIndex)"""
        return self.remote.get_global_index()

    def push(self, repository: str, version: str, basedir: str,
             configfile: str = "modelx.yaml", digest_mode: str = "sha256",
             quiet: Optional[bool] = None, dir_format: str = "tar+gz",
             compress: str = "") -> types.Manifest:
        return self.pusher.push(repository, version or "latest", basedir, configfile,
                                digest_mode=digest_mode, quiet=quiet, dir_format=dir_format,
                                compress=compress)

    def pull(self, repository: str, version: str, into_dir: str,
             quiet: Optional[bool] = None) -> types.Manifest:
        return self.puller.pull(repository, version or "latest", into_dir, quiet=quiet)

    def get_manifest(self, repository: str, version: str = "") -> types.Manifest:
        return self.remote.get_manifest(repository, version)

    def get_index(self, repository: str, search: str = "") -> types.Index:
        return self.remote.get_index(repository, search)

    def get_global_index(self, search: str = "") -> types.Index:
        return self.remote.get_global_index(search)

    def get_config_content(self, repository: str, version: str = "") -> bytes:
        manifest = self.remote.get_manifest(repository, version)
        return b"".join(self.remote.get_blob_content(repository, manifest.config.digest))
