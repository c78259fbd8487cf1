"""Local repo-alias/credential store ``~/.modelx/repos.json``
(reference: cmd/modelx/repo/repo.go:28-131). File format is identical."""
from __future__ import annotations

import json
import os
from dataclasses import dataclass
from typing import List, Optional
from urllib.parse import urlsplit


@dataclass
class RepoDetails:
    name: str = ""
    url: str = ""
    token: str = ""

    def to_dict(self):
        d = {}
        if self.name:
            d["name"] = self.name
        if self.url:
            d["url"] = self.url
        if self.token:
            d["token"] = self.token
        return d


class RepoManager:
    def __init__(self, path: Optional[str] = None):
        self.path = path or os.path.join(os.path.expanduser("~"), ".modelx", "repos.json")

    def _load(self) -> List[RepoDetails]:
        try:
            with open(self.path, "r") as f:
                data = json.load(f)
        except (FileNotFoundError, json.JSONDecodeError):
            return []
        return [
            RepoDetails(r.get("name", ""), r.get("url", ""), r.get("token", ""))
            for r in data.get("repos") or []
        ]

    def _save(self, repos: List[RepoDetails]) -> None:
        os.makedirs(os.path.dirname(self.path), exist_ok=True)
        with open(self.path, "w") as f:
            json.dump({"repos": [r.to_dict() for r in repos]} if repos else {}, f, indent=2)

    def list(self) -> List[RepoDetails]:
        return self._load()

    def get(self, name: str) -> RepoDetails:
        for r in self._load():
            if r.name == name:
                return r
        raise KeyError(f"repo {name!r} not found; add it with `modelx repo add {name} <url>`")

    def set(self, item: RepoDetails) -> None:
        u = urlsplit(item.url)
        if not u.scheme or not u.netloc:
            raise ValueError(f"invalid url: {item.url}")
        repos = self._load()
        for i, r in enumerate(repos):
            if r.name == item.name:
                repos[i] = item
                break
        else:
            repos.append(item)
        self._save(repos)

    def remove(self, name: str) -> bool:
        repos = self._load()
        kept = [r for r in repos if r.name != name]
        if len(kept) == len(repos):
            return False
        self._save(kept)
        return True


def default_repo_manager() -> RepoManager:
    return RepoManager()
