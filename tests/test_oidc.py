"""Offline OIDC: RS256 ID-token verification against a configured JWKS
document (reference: pkg/registry/helper.go:63-96 — go-oidc verifies RS256
tokens after fetching the issuer's jwks_uri; with no egress the JWKS is
configured via --oidc-jwks and verification itself is identical and fully
offline). Keys are generated and tokens signed with the openssl CLI."""
import base64
import json
import os
import subprocess
import time

import pytest
import requests

from util_servers import MODELXD, ServerProc, _build_servers, free_port, wait_http

ISSUER = "https://issuer.test/realm"
AUDIENCE = "modelx"


def _b64url(b: bytes) -> str:
    return base64.urlsafe_b64encode(b).decode().rstrip("=")


@pytest.fixture(scope="module")
def rsa(tmp_path_factory):
    """(private key PEM path, JWKS file path) via openssl CLI."""
    d = tmp_path_factory.mktemp("oidc")
    key = str(d / "key.pem")
    subprocess.run(["openssl", "genrsa", "-out", key, "2048"], check=True,
                   capture_output=True)
    # modulus: openssl rsa -noout -modulus -> "Modulus=ABCD..."
    out = subprocess.run(["openssl", "rsa", "-in", key, "-noout", "-modulus"],
                         check=True, capture_output=True, text=True).stdout
    n_hex = out.strip().split("=", 1)[1]
    n = bytes.fromhex(n_hex)
    e = (65537).to_bytes(3, "big")
    jwks = {"keys": [{"kty": "RSA", "alg": "RS256", "use": "sig", "kid": "k1",
                      "n": _b64url(n), "e": _b64url(e)}]}
    jwks_path = str(d / "jwks.json")
    with open(jwks_path, "w") as f:
        json.dump(jwks, f)
    return key, jwks_path, str(d)


def sign_rs256(key_pem: str, workdir: str, claims: dict, kid: str = "k1") -> str:
    header = _b64url(json.dumps({"alg": "RS256", "typ": "JWT", "kid": kid}).encode())
    payload = _b64url(json.dumps(claims).encode())
    signing = f"{header}.{payload}".encode()
    inp = os.path.join(workdir, "signing.bin")
    with open(inp, "wb") as f:
        f.write(signing)
    sig = subprocess.run(["openssl", "dgst", "-sha256", "-sign", key_pem, inp],
                         check=True, capture_output=True).stdout
    return f"{header}.{payload}.{_b64url(sig)}"


def claims(exp_delta=3600, iss=ISSUER, aud=AUDIENCE, sub="alice", **extra):
    c = {"sub": sub, "exp": int(time.time()) + exp_delta}
    if iss is not None:
        c["iss"] = iss
    if aud is not None:
        c["aud"] = aud
    c.update(extra)
    return c


@pytest.fixture(scope="module")
def oidc_server(rsa, tmp_path_factory):
    _build_servers()
    _, jwks_path, _ = rsa
    data = tmp_path_factory.mktemp("oidc-reg")
    port = free_port()
    p = ServerProc([MODELXD, "--listen", f"127.0.0.1:{port}", "--local-data", str(data),
                    "--oidc-jwks", jwks_path, "--oidc-issuer", ISSUER,
                    "--oidc-audience", AUDIENCE], port)
    wait_http(port)
    yield p
    p.stop()


def _get(server, token):
    return requests.get(server.url + "/", headers={"Authorization": f"Bearer {token}"})


class TestOidcRs256:
    def test_valid_token_accepted(self, oidc_server, rsa):
        key, _, d = rsa
        assert _get(oidc_server, sign_rs256(key, d, claims())).status_code == 200

    def test_aud_as_array_accepted(self, oidc_server, rsa):
        key, _, d = rsa
        tok = sign_rs256(key, d, claims(aud=["other", AUDIENCE]))
        assert _get(oidc_server, tok).status_code == 200

    def test_query_token_fallback(self, oidc_server, rsa):
        key, _, d = rsa
        tok = sign_rs256(key, d, claims())
        assert requests.get(oidc_server.url + f"/?token={tok}").status_code == 200

    def test_no_token_rejected(self, oidc_server):
        assert requests.get(oidc_server.url + "/").status_code == 401

    def test_expired_rejected(self, oidc_server, rsa):
        key, _, d = rsa
        tok = sign_rs256(key, d, claims(exp_delta=-30))
        assert _get(oidc_server, tok).status_code == 401

    def test_missing_exp_rejected(self, oidc_server, rsa):
        key, _, d = rsa
        c = claims()
        del c["exp"]
        assert _get(oidc_server, sign_rs256(key, d, c)).status_code == 401

    def test_wrong_issuer_rejected(self, oidc_server, rsa):
        key, _, d = rsa
        tok = sign_rs256(key, d, claims(iss="https://evil.test"))
        assert _get(oidc_server, tok).status_code == 401

    def test_wrong_audience_rejected(self, oidc_server, rsa):
        key, _, d = rsa
        tok = sign_rs256(key, d, claims(aud="another-service"))
        assert _get(oidc_server, tok).status_code == 401

    def test_tampered_signature_rejected(self, oidc_server, rsa):
        key, _, d = rsa
        tok = sign_rs256(key, d, claims())
        head, payload, sig = tok.split(".")
        bad = sig[:-2] + ("AA" if sig[-2:] != "AA" else "BB")
        assert _get(oidc_server, f"{head}.{payload}.{bad}").status_code == 401

    def test_tampered_payload_rejected(self, oidc_server, rsa):
        key, _, d = rsa
        tok = sign_rs256(key, d, claims(sub="alice"))
        head, _, sig = tok.split(".")
        forged = _b64url(json.dumps(claims(sub="mallory")).encode())
        assert _get(oidc_server, f"{head}.{forged}.{sig}").status_code == 401

    def test_hs256_alg_confusion_rejected(self, oidc_server, rsa):
        """A token claiming alg=HS256 signed with the JWKS modulus as HMAC
        key must not pass (classic JWT alg-confusion attack)."""
        import hashlib
        import hmac as hmac_mod

        _, jwks_path, _ = rsa
        with open(jwks_path) as f:
            n_b64 = json.load(f)["keys"][0]["n"]
        header = _b64url(json.dumps({"alg": "HS256", "typ": "JWT"}).encode())
        payload = _b64url(json.dumps(claims()).encode())
        signing = f"{header}.{payload}".encode()
        sig = _b64url(hmac_mod.new(n_b64.encode(), signing, hashlib.sha256).digest())
        assert _get(oidc_server, f"{header}.{payload}.{sig}").status_code == 401

    def test_unknown_kid_still_verifies_by_trying_keys(self, rsa, tmp_path):
        """A token without kid verifies against any configured key (JWKS
        with one key and no kid hints are common)."""
        key, jwks_path, d = rsa
        _build_servers()
        port = free_port()
        p = ServerProc([MODELXD, "--listen", f"127.0.0.1:{port}",
                        "--local-data", str(tmp_path), "--oidc-jwks", jwks_path], port)
        wait_http(port)
        try:
            header = _b64url(json.dumps({"alg": "RS256", "typ": "JWT"}).encode())
            payload = _b64url(json.dumps(claims()).encode())
            inp = os.path.join(d, "nokid.bin")
            with open(inp, "wb") as f:
                f.write(f"{header}.{payload}".encode())
            sig = subprocess.run(["openssl", "dgst", "-sha256", "-sign", key, inp],
                                 check=True, capture_output=True).stdout
            tok = f"{header}.{payload}.{_b64url(sig)}"
            assert _get(p, tok).status_code == 200
        finally:
            p.stop()

    def test_bad_jwks_file_fails_startup(self, tmp_path):
        _build_servers()
        bad = tmp_path / "bad.json"
        bad.write_text("{\"keys\": []}")
        r = subprocess.run([MODELXD, "--listen", "127.0.0.1:0", "--oidc-jwks", str(bad)],
                           capture_output=True, text=True, timeout=10)
        assert r.returncode == 1
        assert "no usable RSA keys" in r.stderr
