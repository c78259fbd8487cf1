"""Reference-string grammar tests — the reference's sole unit test
(cmd/modelx/model/reference_test.go) ported and FIXED (its third case
contradicted the implementation)."""
import json

import pytest

from modelx_amd.client.reference import Reference, parse_reference
from modelx_amd.client.repos import RepoDetails, RepoManager


@pytest.fixture
def mgr(tmp_path):
    m = RepoManager(path=str(tmp_path / "repos.json"))
    m.set(RepoDetails(name="myrepo", url="https://registry.example.com", token="tok123"))
    return m


class TestParseReference:
    def test_full_url(self, monkeypatch):
        monkeypatch.delenv("MODELX_AUTH", raising=False)
        ref = parse_reference("https://registry.example.com/project/name@v1")
        assert ref.registry == "https://registry.example.com"
        assert ref.repository == "project/name"
        assert ref.version == "v1"

    def test_url_no_version(self, monkeypatch):
        monkeypatch.delenv("MODELX_AUTH", raising=False)
        ref = parse_reference("https://registry.example.com/project/name")
        assert ref.version == ""

    def test_bare_name_gets_library_project(self, monkeypatch):
        monkeypatch.delenv("MODELX_AUTH", raising=False)
        ref = parse_reference("https://registry.example.com/name@v2")
        assert ref.repository == "library/name"
        assert ref.version == "v2"

    def test_alias_resolution(self, mgr, monkeypatch):
        monkeypatch.delenv("MODELX_AUTH", raising=False)
        ref = parse_reference("myrepo/project/name@v1", repo_manager=mgr)
        assert ref.registry == "https://registry.example.com"
        assert ref.repository == "project/name"
        assert ref.authorization == "Bearer tok123"

    def test_unknown_alias_raises(self, mgr):
        with pytest.raises(KeyError):
            parse_reference("nosuch/project/name", repo_manager=mgr)

    def test_env_auth_overrides_stored_token(self, mgr, monkeypatch):
        monkeypatch.setenv("MODELX_AUTH", "Bearer envtok")
        ref = parse_reference("myrepo/project/name@v1", repo_manager=mgr)
        assert ref.authorization == "Bearer envtok"

    def test_uri_token_overrides_all(self, mgr, monkeypatch):
        monkeypatch.setenv("MODELX_AUTH", "Bearer envtok")
        ref = parse_reference(
            "https://registry.example.com/project/name@v1?token=urltok"
        )
        assert ref.authorization == "Bearer urltok"

    def test_http_scheme_preserved(self, monkeypatch):
        monkeypatch.delenv("MODELX_AUTH", raising=False)
        ref = parse_reference("http://127.0.0.1:8080/p/n@v")
        assert ref.registry == "http://127.0.0.1:8080"

    def test_missing_host_rejected(self, monkeypatch):
        monkeypatch.delenv("MODELX_AUTH", raising=False)
        with pytest.raises(ValueError):
            parse_reference("https:///p/n@v")

    def test_str_roundtrip(self):
        r = Reference(registry="https://h", repository="p/n", version="v1")
        assert str(r) == "https://h/p/n@v1"
        assert str(Reference(registry="https://h", repository="p/n")) == "https://h/p/n"


class TestRepoManager:
    def test_crud(self, tmp_path):
        m = RepoManager(path=str(tmp_path / "repos.json"))
        m.set(RepoDetails(name="a", url="https://a.example", token="t"))
        m.set(RepoDetails(name="b", url="https://b.example", token=""))
        assert {r.name for r in m.list()} == {"a", "b"}
        m.set(RepoDetails(name="a", url="https://a2.example", token="t2"))
        assert m.get("a").url == "https://a2.example"
        assert m.remove("b") is True
        assert m.remove("b") is False
        # file format matches reference repos.json
        data = json.load(open(tmp_path / "repos.json"))
        assert data["repos"][0]["name"] == "a"

    def test_invalid_url_rejected(self, tmp_path):
        m = RepoManager(path=str(tmp_path / "repos.json"))
        with pytest.raises(ValueError):
            m.set(RepoDetails(name="x", url="not-a-url"))
