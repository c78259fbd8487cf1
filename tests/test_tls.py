"""TLS serving (reference: pkg/registry/server.go:37-43 ListenAndServeTLS;
client --insecure = cmd/modelx/modelx.go:29-36 InsecureSkipVerify)."""
import os
import subprocess

import pytest

from modelx_amd.client import Client
from modelx_amd.config import ModelConfig

from util_servers import MODELXD, ServerProc, _build_servers, free_port, wait_http


@pytest.fixture(scope="module")
def certs(tmp_path_factory):
    d = tmp_path_factory.mktemp("tls")
    cert, key = str(d / "cert.pem"), str(d / "key.pem")
    r = subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes", "-keyout", key,
         "-out", cert, "-days", "1", "-subj", "/CN=127.0.0.1",
         "-addext", "subjectAltName=IP:127.0.0.1"],
        capture_output=True, text=True)
    if r.returncode != 0:
        pytest.skip(f"openssl cert generation failed: {r.stderr[-200:]}")
    return cert, key


def test_https_push_pull(certs, tmp_path):
    _build_servers()
    cert, key = certs
    port = free_port()
    srv = ServerProc([MODELXD, "--listen", f"127.0.0.1:{port}", "--local-data",
                      str(tmp_path / "data"), "--tls-cert", cert, "--tls-key", key], port)
    try:
        import requests
        import urllib3

        urllib3.disable_warnings()
        wait_https = False
        import time

        for _ in range(100):
            try:
                if requests.get(f"https://127.0.0.1:{port}/healthz", verify=False,
                                timeout=1).status_code == 200:
                    wait_https = True
                    break
            except requests.RequestException:
                time.sleep(0.05)
        assert wait_https, "TLS server did not come up"

        # plain http against the TLS port must fail (not silently work)
        with pytest.raises(Exception):
            requests.get(f"http://127.0.0.1:{port}/healthz", timeout=2).raise_for_status()

        d = tmp_path / "model"
        d.mkdir()
        (d / "modelx.yaml").write_text(ModelConfig(description="tls").to_yaml())
        (d / "weights.bin").write_bytes(os.urandom(512 * 1024))
        c = Client(f"https://127.0.0.1:{port}", insecure=True)
        c.push("tls/model", "v1", str(d), quiet=True)
        out = tmp_path / "out"
        c.pull("tls/model", "v1", str(out), quiet=True)
        assert (out / "weights.bin").read_bytes() == (d / "weights.bin").read_bytes()
    finally:
        srv.stop()


def test_native_client_tls(certs, tmp_path, monkeypatch):
    """The NATIVE http client (the engine's ranged-GET/push path) speaks
    TLS: https presigned URLs from a TLS object store work on the GPU data
    plane. Verified here on CPU via _core.http_get against a TLS modelxd;
    certificate verification is on by default (self-signed fails) and
    MODELX_TLS_INSECURE=1 opts out, mirroring the CLI --insecure."""
    from modelx_amd import _core

    _build_servers()
    cert, key = certs
    port = free_port()
    p = ServerProc([MODELXD, "--listen", f"127.0.0.1:{port}", "--local-data",
                    str(tmp_path / "d"), "--tls-cert", cert, "--tls-key", key], port)
    try:
        import time

        deadline = time.time() + 10
        last = None
        monkeypatch.setenv("MODELX_TLS_INSECURE", "1")
        while time.time() < deadline:
            try:
                status, body = _core.http_get(f"https://127.0.0.1:{port}/healthz")
                last = (status, body)
                break
            except RuntimeError as e:
                last = e
                time.sleep(0.2)
        assert last == (200, b"ok"), last
        # env is latched at first TLS use in this process (static ctx), so
        # the strict-verification negative needs a fresh process
        import subprocess
        import sys

        r = subprocess.run(
            [sys.executable, "-c",
             "from modelx_amd import _core;"
             f"_core.http_get('https://127.0.0.1:{port}/healthz')"],
            capture_output=True, text=True, timeout=30,
            env={k: v for k, v in os.environ.items() if k != "MODELX_TLS_INSECURE"},
            cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
        assert r.returncode != 0 and "request failed" in r.stderr
    finally:
        p.stop()


def test_tls_s3d_presigned_data_plane(certs, tmp_path, monkeypatch):
    """End-to-end TLS object store: s3d serves HTTPS, modelxd talks SigV4
    to it over TLS and hands out https presigned URLs, and the CPU client
    pulls through them (the GPU engine uses the same native TLS client —
    test_gpu_kernels covers it on hardware)."""
    import os as _os

    from modelx_amd.client import Client
    from modelx_amd.config import ModelConfig
    from util_servers import ACCESS_KEY, BUCKET, S3D, SECRET_KEY

    _build_servers()
    cert, key = certs
    monkeypatch.setenv("MODELX_TLS_INSECURE", "1")
    s3_port = free_port()
    s3 = ServerProc([S3D, "--listen", f"127.0.0.1:{s3_port}", "--root",
                     str(tmp_path / "s3"), "--access-key", ACCESS_KEY,
                     "--secret-key", SECRET_KEY, "--tls-cert", cert,
                     "--tls-key", key], s3_port)
    _os.makedirs(tmp_path / "s3" / BUCKET, exist_ok=True)
    # wait_http can't probe a TLS port — poll through the native TLS client
    # (modelxd's startup index refresh dies if the store isn't up yet)
    import time as _time

    from modelx_amd import _core

    deadline = _time.time() + 10
    while _time.time() < deadline:
        try:
            if _core.http_get(f"https://127.0.0.1:{s3_port}/healthz")[0] == 200:
                break
        except RuntimeError:
            _time.sleep(0.1)
    mdx_port = free_port()
    mdx = ServerProc([MODELXD, "--listen", f"127.0.0.1:{mdx_port}", "--s3-url",
                      f"https://127.0.0.1:{s3_port}", "--s3-bucket", BUCKET,
                      "--s3-access-key", ACCESS_KEY, "--s3-secret-key", SECRET_KEY,
                      "--enable-redirect"], mdx_port)
    wait_http(mdx_port)
    try:
        d = tmp_path / "m"
        d.mkdir()
        (d / "modelx.yaml").write_text(ModelConfig(description="tls").to_yaml())
        payload = _os.urandom(3 << 20)
        (d / "w.bin").write_bytes(payload)
        c = Client(f"http://127.0.0.1:{mdx_port}")
        c.push("tls/model", "v1", str(d), quiet=True)
        loc = c.remote.get_blob_location(
            "tls/model",
            next(b for b in c.get_manifest("tls/model", "v1").blobs
                 if b.name == "w.bin"), "download")
        url = loc.properties["parts"][0]["url"]
        assert url.startswith("https://")
        # the NATIVE client fetches the https presigned URL
        from modelx_amd import _core

        status, body = _core.http_get(url)
        assert status == 200 and body == payload
        out = tmp_path / "out"
        c.pull("tls/model", "v1", str(out), quiet=True)
        assert (out / "w.bin").read_bytes() == payload
    finally:
        mdx.stop()
        s3.stop()

