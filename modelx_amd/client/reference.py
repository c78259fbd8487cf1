"""Model reference grammar: ``<repo-alias>/<project>/<name>@<version>`` or a
full URL ``http(s)://host/project/name@version[?token=...]``.

Semantics match the reference (cmd/modelx/model/reference.go:33-86):
- no ``://`` → first path element is a repo alias resolved via
  ``~/.modelx/repos.json``; its token becomes ``Bearer <token>`` auth
- ``MODELX_AUTH`` env overrides the stored token
- ``?token=`` in the URI overrides both
- bare ``name`` (no ``/``) → ``library/<name>``
- missing ``@version`` → empty version (client defaults to ``latest`` at
  request time, reference: pkg/client/registry.go:34-36)
"""
from __future__ import annotations

import os
from dataclasses import dataclass, field
from urllib.parse import parse_qs, urlsplit

from .repos import RepoManager, default_repo_manager

MODELX_AUTH_ENV = "MODELX_AUTH"
SPLITOR_REPO = "/"
SPLITOR_VERSION = "@"


@dataclass
class Reference:
    registry: str = ""
    repository: str = ""
    version: str = ""
    authorization: str = ""

    def __str__(self) -> str:
        if not self.version:
            return f"{self.registry}/{self.repository}"
        return f"{self.registry}/{self.repository}@{self.version}"

    def client(self, **kw):
        from . import Client

        return Client(self.registry, self.authorization, **kw)


def parse_reference(raw: str, repo_manager: RepoManager = None) -> Reference:
    auth = os.environ.get(MODELX_AUTH_ENV, "")
    if "://" not in raw:
        splits = raw.split(SPLITOR_REPO, 1)
        mgr = repo_manager or default_repo_manager()
        details = mgr.get(splits[0])  # raises if alias unknown
        if not auth and details.token:
            auth = "Bearer " + details.token
        # the stored URL may carry its own query (?token=...): splice the
        # repository path in BEFORE the query, not after it
        base, _, query = details.url.partition("?")
        raw = base.rstrip("/") + ("/" + splits[1] if len(splits) == 2 else "")
        if query:
            raw += "?" + query

    if not raw.startswith(("http://", "https://")):
        raw = "https://" + raw
    u = urlsplit(raw)
    if not u.netloc:
        raise ValueError("invalid reference: missing host")
    token = (parse_qs(u.query).get("token") or [""])[0]
    if token:
        auth = "Bearer " + token

    path = u.path or ""
    repository, version = "", ""
    splits = path.split(SPLITOR_VERSION, 1)
    if len(splits) == 2 and splits[1]:
        version = splits[1]
    if splits[0]:
        repository = splits[0][1:]  # strip leading /

    if repository and "/" not in repository:
        repository = "library/" + repository

    return Reference(
        registry=f"{u.scheme}://{u.netloc}",
        repository=repository,
        version=version,
        authorization=auth,
    )
