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
