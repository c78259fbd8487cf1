"""CPU integration: full push→list→info→pull→GC loop against the C++ modelxd
with the local-FS backend (reference semantics: §3.1-3.5 call stacks)."""
import json
import os

import pytest

from modelx_amd.client import Client
from modelx_amd.config import ModelConfig
from modelx_amd.wire import errors as er
from modelx_amd.wire import types

from util_servers import start_modelxd_local


@pytest.fixture(scope="module")
def server(tmp_path_factory):
    data = tmp_path_factory.mktemp("registry-data")
    p = start_modelxd_local(str(data))
    yield p
    p.stop()


@pytest.fixture()
def model_dir(tmp_path):
    d = tmp_path / "model"
    d.mkdir()
    cfg = ModelConfig(description="test model", framework="pytorch", task="x",
                      model_files=["weights.bin"])
    (d / "modelx.yaml").write_text(cfg.to_yaml())
    (d / "weights.bin").write_bytes(os.urandom(256 * 1024))
    (d / "README.md").write_text("# test\n")
    sub = d / "tokenizer"
    sub.mkdir()
    (sub / "vocab.txt").write_text("a\nb\nc\n")
    (sub / "merges.txt").write_text("a b\n")
    return d


def test_push_pull_roundtrip(server, model_dir, tmp_path):
    c = Client(server.url, concurrency=3)
    manifest = c.push("proj/demo", "v1", str(model_dir), quiet=True)
    assert manifest.config.name == "modelx.yaml"
    names = [b.name for b in manifest.blobs]
    assert names == sorted(names)
    assert "weights.bin" in names and "tokenizer" in names

    # index + global index
    idx = c.get_index("proj/demo")
    assert [m.name for m in idx.manifests] == ["v1"]
    assert idx.manifests[0].size > 0
    gidx = c.get_global_index()
    assert any(m.name == "proj/demo" for m in gidx.manifests)

    # pull to a fresh dir and compare bytes
    out = tmp_path / "out"
    c.pull("proj/demo", "v1", str(out), quiet=True)
    assert (out / "weights.bin").read_bytes() == (model_dir / "weights.bin").read_bytes()
    assert (out / "modelx.yaml").read_text() == (model_dir / "modelx.yaml").read_text()
    assert (out / "tokenizer" / "vocab.txt").read_text() == "a\nb\nc\n"

    # pull again → everything up to date (resume/skip path)
    c.pull("proj/demo", "v1", str(out), quiet=True)


def test_pull_latest_default(server, model_dir, tmp_path):
    c = Client(server.url)
    c.push("proj/latestdemo", "latest", str(model_dir), quiet=True)
    m = c.get_manifest("proj/latestdemo")  # no version → latest
    assert m.config.name == "modelx.yaml"


def test_manifest_unknown_error(server):
    c = Client(server.url)
    with pytest.raises(er.ModelxError) as exc:
        c.get_manifest("proj/noexist", "v9")
    assert exc.value.code == er.ErrCode.MANIFEST_UNKNOWN
    assert exc.value.http_status == 404


def test_search_filters_index(server, model_dir):
    c = Client(server.url)
    c.push("proj/searchme", "v1", str(model_dir), quiet=True)
    c.push("proj/searchme", "v2", str(model_dir), quiet=True)
    idx = c.get_index("proj/searchme", search="v1")
    assert [m.name for m in idx.manifests] == ["v1"]
    gidx = c.get_global_index(search="searchme")
    assert [m.name for m in gidx.manifests] == ["proj/searchme"]


def test_delete_manifest_and_gc(server, model_dir):
    c = Client(server.url)
    c.push("proj/gcdemo", "v1", str(model_dir), quiet=True)
    c.push("proj/gcdemo", "v2", str(model_dir), quiet=True)
    c.remote.delete_manifest("proj/gcdemo", "v2")
    idx = c.get_index("proj/gcdemo")
    assert [m.name for m in idx.manifests] == ["v1"]
    # GC: v1 still references the blobs → nothing collected
    result = c.remote.garbage_collect("proj/gcdemo")
    assert result["blobs"] == 0
    # delete the whole index → repo gone from the global index
    c.remote.delete_index("proj/gcdemo")
    gidx = c.get_global_index()
    assert not any(m.name == "proj/gcdemo" for m in gidx.manifests)


def test_gc_collects_orphans(server, model_dir, tmp_path):
    c = Client(server.url)
    c.push("proj/gcorphan", "v1", str(model_dir), quiet=True)
    # orphan a blob by replacing the manifest with one that drops weights.bin
    m = c.get_manifest("proj/gcorphan", "v1")
    dropped = [b for b in m.blobs if b.name != "weights.bin"]
    orphan_digest = next(b.digest for b in m.blobs if b.name == "weights.bin")
    m.blobs = dropped
    c.remote.put_manifest("proj/gcorphan", "v1", m)
    result = c.remote.garbage_collect("proj/gcorphan")
    assert result["blobs"] >= 1
    assert not c.remote.head_blob("proj/gcorphan", orphan_digest)


def test_push_resume_dedup(server, model_dir):
    """Interrupted pushes resume: 2nd push of identical content skips blobs
    via HEAD dedup (push.go:169-177)."""
    c = Client(server.url)
    c.push("proj/dedup", "v1", str(model_dir), quiet=True)
    m = c.get_manifest("proj/dedup", "v1")
    for b in m.blobs:
        assert c.remote.head_blob("proj/dedup", b.digest)
    c.push("proj/dedup", "v2", str(model_dir), quiet=True)  # all blobs skipped
    assert c.get_manifest("proj/dedup", "v2").blobs[0].digest == m.blobs[0].digest


def test_digest_verified_on_pull(server, model_dir, tmp_path):
    """A corrupted blob on the server must be detected (the reference never
    verifies after download — we do)."""
    c = Client(server.url)
    manifest = c.push("proj/corrupt", "v1", str(model_dir), quiet=True)
    weights = next(b for b in manifest.blobs if b.name == "weights.bin")
    # corrupt the stored blob behind the server's back
    from modelx_amd.wire import paths as pathsmod

    data_root = None
    # find the data dir the fixture created (single registry-data dir)
    base = server  # ServerProc
    # walk the local-data dir from the process args
    args = base.proc.args
    data_root = args[args.index("--local-data") + 1]
    blob_path = os.path.join(data_root, pathsmod.blob_digest_path("proj/corrupt", weights.digest))
    with open(blob_path, "r+b") as f:
        f.seek(0)
        b0 = f.read(1)
        f.seek(0)
        f.write(bytes([b0[0] ^ 0xFF]))
    out = tmp_path / "corrupt-out"
    with pytest.raises(er.ModelxError) as exc:
        c.pull("proj/corrupt", "v1", str(out), quiet=True)
    assert exc.value.code == er.ErrCode.DIGEST_INVALID


def test_chunked_digest_mode(server, model_dir, tmp_path):
    """digest_mode=chunked: descriptor digest is sha256c1m:... (GPU-fast
    path); pull verifies with the chunked algorithm."""
    c = Client(server.url)
    manifest = c.push("proj/chunked", "v1", str(model_dir), digest_mode="chunked", quiet=True)
    weights = next(b for b in manifest.blobs if b.name == "weights.bin")
    assert weights.digest.startswith("sha256c1m:")
    out = tmp_path / "chunked-out"
    c.pull("proj/chunked", "v1", str(out), quiet=True)
    assert (out / "weights.bin").read_bytes() == (model_dir / "weights.bin").read_bytes()


def test_scheduled_gc_sweeps_orphans(model_dir, tmp_path):
    """--gc-interval runs the mark-sweep periodically (the reference has
    only the manual POST endpoint)."""
    import time

    from util_servers import MODELXD, ServerProc, _build_servers, free_port, wait_http

    _build_servers()
    port = free_port()
    p = ServerProc([MODELXD, "--listen", f"127.0.0.1:{port}", "--local-data",
                    str(tmp_path / "reg"), "--gc-interval", "1"], port)
    wait_http(port)
    try:
        c = Client(p.url)
        c.push("proj/gcsched", "v1", str(model_dir), quiet=True)
        m = c.get_manifest("proj/gcsched", "v1")
        orphan = next(b.digest for b in m.blobs if b.name == "weights.bin")
        m.blobs = [b for b in m.blobs if b.name != "weights.bin"]
        c.remote.put_manifest("proj/gcsched", "v1", m)
        deadline = time.time() + 10
        while time.time() < deadline and c.remote.head_blob("proj/gcsched", orphan):
            time.sleep(0.3)
        assert not c.remote.head_blob("proj/gcsched", orphan)
    finally:
        p.stop()
