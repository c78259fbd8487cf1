"""Auth filter + /metrics tests (reference: pkg/registry/helper.go:63-113 —
with the context-drop defect fixed; OIDC is offline here so the equivalent is
static bearer tokens + HS256 JWT against a shared secret)."""
import base64
import hashlib
import hmac
import json
import os
import time

import pytest
import requests

from util_servers import MODELXD, ServerProc, free_port, wait_http, _build_servers


def _b64url(b: bytes) -> str:
    return base64.urlsafe_b64encode(b).decode().rstrip("=")


def make_jwt(secret: str, sub: str = "alice", exp_delta: int = 3600) -> str:
    header = _b64url(json.dumps({"alg": "HS256", "typ": "JWT"}).encode())
    payload = _b64url(json.dumps({"sub": sub, "exp": int(time.time()) + exp_delta}).encode())
    signing = f"{header}.{payload}".encode()
    sig = _b64url(hmac.new(secret.encode(), signing, hashlib.sha256).digest())
    return f"{header}.{payload}.{sig}"


@pytest.fixture(scope="module")
def auth_server(tmp_path_factory):
    _build_servers()
    data = tmp_path_factory.mktemp("auth-reg")
    port = free_port()
    p = ServerProc([MODELXD, "--listen", f"127.0.0.1:{port}", "--local-data", str(data),
                    "--auth-tokens", "statictok1,statictok2",
                    "--jwt-hs256-secret", "topsecret"], port)
    wait_http(port)
    yield p
    p.stop()


class TestAuth:
    def test_no_token_rejected(self, auth_server):
        r = requests.get(auth_server.url + "/")
        assert r.status_code == 401
        assert r.json()["code"] == "UNAUTHORIZED"

    def test_healthz_open(self, auth_server):
        assert requests.get(auth_server.url + "/healthz").status_code == 200

    def test_static_token_accepted(self, auth_server):
        r = requests.get(auth_server.url + "/",
                         headers={"Authorization": "Bearer statictok2"})
        assert r.status_code == 200

    def test_query_token_fallback(self, auth_server):
        # ?token= fallback (helper.go:69-74)
        r = requests.get(auth_server.url + "/?token=statictok1")
        assert r.status_code == 200

    def test_bad_token_rejected(self, auth_server):
        r = requests.get(auth_server.url + "/",
                         headers={"Authorization": "Bearer wrong"})
        assert r.status_code == 401

    def test_valid_jwt_accepted(self, auth_server):
        tok = make_jwt("topsecret")
        r = requests.get(auth_server.url + "/", headers={"Authorization": f"Bearer {tok}"})
        assert r.status_code == 200

    def test_expired_jwt_rejected(self, auth_server):
        tok = make_jwt("topsecret", exp_delta=-100)
        r = requests.get(auth_server.url + "/", headers={"Authorization": f"Bearer {tok}"})
        assert r.status_code == 401

    def test_wrong_secret_jwt_rejected(self, auth_server):
        tok = make_jwt("othersecret")
        r = requests.get(auth_server.url + "/", headers={"Authorization": f"Bearer {tok}"})
        assert r.status_code == 401

    def test_authed_push_pull(self, auth_server, tmp_path):
        from modelx_amd.client import Client
        from modelx_amd.config import ModelConfig

        d = tmp_path / "m"
        d.mkdir()
        (d / "modelx.yaml").write_text(ModelConfig(description="auth").to_yaml())
        (d / "w.bin").write_bytes(os.urandom(1024))
        c = Client(auth_server.url, authorization="Bearer statictok1")
        c.push("proj/authed", "v1", str(d), quiet=True)
        out = tmp_path / "out"
        c.pull("proj/authed", "v1", str(out), quiet=True)
        assert (out / "w.bin").read_bytes() == (d / "w.bin").read_bytes()


class TestMetrics:
    def test_metrics_counters(self, tmp_path):
        from util_servers import start_modelxd_local

        from modelx_amd.client import Client
        from modelx_amd.config import ModelConfig

        srv = start_modelxd_local(str(tmp_path / "reg"))
        try:
            d = tmp_path / "m"
            d.mkdir()
            (d / "modelx.yaml").write_text(ModelConfig(description="x").to_yaml())
            (d / "w.bin").write_bytes(os.urandom(2048))
            c = Client(srv.url)
            c.push("proj/met", "v1", str(d), quiet=True)
            c.pull("proj/met", "v1", str(tmp_path / "out"), quiet=True)
            text = requests.get(srv.url + "/metrics").text
            metrics = {}
            for line in text.splitlines():
                if line and not line.startswith("#"):
                    k, v = line.split()
                    metrics[k] = int(v)
            assert metrics["modelx_requests_total"] > 5
            assert metrics["modelx_blob_bytes_in_total"] >= 2048
            assert metrics["modelx_blob_bytes_out_total"] >= 2048
            assert metrics["modelx_manifests_put_total"] == 1
        finally:
            srv.stop()
