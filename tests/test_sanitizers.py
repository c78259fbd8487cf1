"""Sanitizer builds of the C++ servers under the real integration flow
(the reference configures no race detection at all — no
`-race` in its Makefile/CI).

ASAN: full push→pull→GC loop against modelxd-asan (local FS) and the
modelxd-asan + modelx-s3d-asan presign path. TSAN: concurrent pulls hammer
the s3d thread pool. A sanitizer report aborts the server mid-request →
the client errors → the test fails.
"""
import concurrent.futures
import os
import subprocess

import pytest

from modelx_amd.client import Client
from modelx_amd.config import ModelConfig

from util_servers import (ACCESS_KEY, BUCKET, REPO_ROOT, SECRET_KEY, ServerProc, free_port,
                          wait_http)

BIN = os.path.join(REPO_ROOT, "bin")


def _build(target: str):
    r = subprocess.run(["make", target], cwd=REPO_ROOT, capture_output=True, text=True)
    if r.returncode != 0:
        pytest.skip(f"make {target} failed (no sanitizer toolchain?): {r.stderr[-300:]}")


def _model_dir(tmp_path, nbytes=1 << 20):
    d = tmp_path / "model"
    d.mkdir()
    (d / "modelx.yaml").write_text(ModelConfig(description="san").to_yaml())
    (d / "weights.bin").write_bytes(os.urandom(nbytes))
    return d


@pytest.mark.parametrize("flavor", ["asan"])
def test_local_flow_under_sanitizer(flavor, tmp_path):
    _build(f"servers-{flavor}")
    port = free_port()
    srv = ServerProc([os.path.join(BIN, f"modelxd-{flavor}"), "--listen",
                      f"127.0.0.1:{port}", "--local-data", str(tmp_path / "data")], port)
    try:
        wait_http(port, timeout=30)
        c = Client(srv.url)
        d = _model_dir(tmp_path)
        c.push("san/model", "v1", str(d), quiet=True)
        c.pull("san/model", "v1", str(tmp_path / "out"), quiet=True)
        assert (tmp_path / "out" / "weights.bin").read_bytes() == \
            (d / "weights.bin").read_bytes()
        assert srv.proc.poll() is None, "server died (sanitizer report?)"
    finally:
        srv.stop()


@pytest.mark.parametrize("flavor", ["tsan"])
def test_s3_concurrent_pulls_under_sanitizer(flavor, tmp_path):
    _build(f"servers-{flavor}")
    s3_port = free_port()
    s3 = ServerProc([os.path.join(BIN, f"modelx-s3d-{flavor}"), "--listen",
                     f"127.0.0.1:{s3_port}", "--root", str(tmp_path / "s3"),
                     "--access-key", ACCESS_KEY, "--secret-key", SECRET_KEY], s3_port)
    os.makedirs(tmp_path / "s3" / BUCKET, exist_ok=True)
    mdx_port = free_port()
    mdx = ServerProc([os.path.join(BIN, f"modelxd-{flavor}"), "--listen",
                      f"127.0.0.1:{mdx_port}", "--s3-url", s3.url, "--s3-bucket", BUCKET,
                      "--s3-access-key", ACCESS_KEY, "--s3-secret-key", SECRET_KEY,
                      "--enable-redirect"], mdx_port)
    try:
        wait_http(s3_port, timeout=30)
        wait_http(mdx_port, timeout=30)
        c = Client(mdx.url)
        d = _model_dir(tmp_path, nbytes=4 << 20)
        c.push("san/s3model", "v1", str(d), quiet=True)

        def pull(i):
            out = tmp_path / f"out{i}"
            Client(mdx.url).pull("san/s3model", "v1", str(out), quiet=True)
            return (out / "weights.bin").stat().st_size

        with concurrent.futures.ThreadPoolExecutor(max_workers=6) as pool:
            sizes = list(pool.map(pull, range(6)))
        assert all(s == 4 << 20 for s in sizes)
        assert s3.proc.poll() is None and mdx.proc.poll() is None
    finally:
        mdx.stop()
        s3.stop()
