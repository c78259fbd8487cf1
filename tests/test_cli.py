"""CLI verb tests against a live modelxd (reference: cmd/modelx verbs)."""
import json
import os

import pytest

from modelx_amd.cli.main import main as cli_main
from util_servers import start_modelxd_local


@pytest.fixture(scope="module")
def server(tmp_path_factory):
    p = start_modelxd_local(str(tmp_path_factory.mktemp("cli-reg")))
    yield p
    p.stop()


@pytest.fixture()
def home(tmp_path, monkeypatch):
    monkeypatch.setenv("HOME", str(tmp_path))
    monkeypatch.delenv("MODELX_AUTH", raising=False)
    return tmp_path


def run(*argv):
    return cli_main(list(argv))


def test_init_creates_scaffold(tmp_path, home):
    d = tmp_path / "mymodel"
    assert run("init", str(d)) == 0
    assert (d / "modelx.yaml").exists()
    assert (d / "README.md").exists()
    assert run("init", str(d)) == 1  # refuses without --force
    assert run("init", str(d), "--force") == 0


def test_repo_add_list_remove(home, capsys):
    assert run("repo", "add", "myhub", "https://hub.example", "--token", "tok") == 0
    assert run("repo", "list") == 0
    out = capsys.readouterr().out
    assert "myhub" in out and "***" in out
    assert run("repo", "remove", "myhub") == 0
    assert run("repo", "remove", "myhub") == 1


def test_push_list_info_pull_roundtrip(server, tmp_path, home, capsys):
    d = tmp_path / "model"
    run("init", str(d))
    (d / "weights.bin").write_bytes(os.urandom(1024))
    url_ref = f"{server.url}/proj/cli@v1"
    assert run("push", url_ref, str(d)) == 0
    # list versions
    assert run("list", f"{server.url}/proj/cli") == 0
    out = capsys.readouterr().out
    assert "v1" in out
    # list files
    assert run("list", url_ref) == 0
    out = capsys.readouterr().out
    assert "weights.bin" in out
    # info prints yaml
    assert run("info", url_ref) == 0
    out = capsys.readouterr().out
    assert "description:" in out
    # global list
    assert run("list", server.url) == 0
    out = capsys.readouterr().out
    assert "proj/cli" in out
    # pull
    dest = tmp_path / "out"
    assert run("pull", url_ref, str(dest)) == 0
    assert (dest / "weights.bin").read_bytes() == (d / "weights.bin").read_bytes()
    # gc
    assert run("gc", url_ref) == 0
    # version
    assert run("version") == 0
    assert json.loads(capsys.readouterr().out.split("}\n")[-2] + "}")


def test_login_stores_repo(server, home):
    assert run("login", f"{server.url}/", "--token", "tok123", "--name", "local") == 0
    data = json.load(open(home / ".modelx" / "repos.json"))
    assert data["repos"][0]["name"] == "local"
    assert data["repos"][0]["token"] == "tok123"


def test_completion_emits_script(home, capsys):
    assert run("completion", "bash") == 0
    assert "_modelx_completions" in capsys.readouterr().out


def test_modelxdl_filters_by_modelfiles(server, tmp_path, home):
    import yaml

    from modelx_amd.cli.dl import main as dl_main

    d = tmp_path / "dlmodel"
    run("init", str(d))
    cfg = yaml.safe_load((d / "modelx.yaml").read_text())
    cfg["modelFiles"] = ["wanted.bin"]
    (d / "modelx.yaml").write_text(yaml.safe_dump(cfg))
    (d / "wanted.bin").write_bytes(os.urandom(512))
    (d / "ignored.bin").write_bytes(os.urandom(512))
    assert run("push", f"{server.url}/proj/dl@v1", str(d)) == 0
    dest = tmp_path / "dl-out"
    assert dl_main([f"{server.url}/proj/dl@v1", str(dest)]) == 0
    assert (dest / "wanted.bin").exists()
    assert not (dest / "ignored.bin").exists()


class TestTokenAliasFlow:
    """End-to-end CLI against an auth-enabled server through a repo alias
    whose URL carried ?token= (reference.go:61-63 URI auth). Regression:
    the alias query used to be spliced AFTER the path, mangling both the
    repository path and the token."""

    def test_full_flow(self, tmp_path, monkeypatch, capsys):
        from util_servers import MODELXD, ServerProc, _build_servers, free_port, wait_http

        from modelx_amd.cli.main import main as cli

        _build_servers()
        monkeypatch.setenv("HOME", str(tmp_path))
        monkeypatch.delenv("MODELX_AUTH", raising=False)
        port = free_port()
        srv = ServerProc([MODELXD, "--listen", f"127.0.0.1:{port}", "--local-data",
                          str(tmp_path / "data"), "--auth-tokens", "tok123"], port)
        try:
            wait_http(port)
            assert cli(["repo", "add", "local", f"http://127.0.0.1:{port}?token=tok123"]) == 0
            d = tmp_path / "model"
            d.mkdir()
            assert cli(["init", str(d)]) == 0
            (d / "w.bin").write_bytes(os.urandom(64 * 1024))
            assert cli(["push", "local/proj/demo@v1", str(d)]) == 0
            assert cli(["list", "local"]) == 0
            assert "proj/demo" in capsys.readouterr().out
            assert cli(["list", "local/proj/demo@v1"]) == 0
            assert "w.bin" in capsys.readouterr().out
            out = tmp_path / "out"
            assert cli(["pull", "local/proj/demo@v1", str(out)]) == 0
            assert (out / "w.bin").read_bytes() == (d / "w.bin").read_bytes()
            # wrong token is rejected
            monkeypatch.setenv("MODELX_AUTH", "Bearer nope")
            assert cli(["list", f"http://127.0.0.1:{port}/proj/demo"]) != 0
        finally:
            srv.stop()
