"""CPU integration over the presigned-redirect data plane: modelxd (S3
backend, --enable-redirect) + modelx-s3d (MinIO stand-in). This is BASELINE
config 1: init + push/pull a 10 MiB random blob against local modelxd+S3."""
import os

import pytest

from modelx_amd.client import Client
from modelx_amd.config import ModelConfig
from modelx_amd.wire import types

from util_servers import start_modelxd_s3, start_s3d


@pytest.fixture(scope="module")
def stack(tmp_path_factory):
    s3_root = tmp_path_factory.mktemp("s3-data")
    s3d = start_s3d(str(s3_root))
    mdx = start_modelxd_s3(s3d.url, redirect=True)
    yield mdx, s3d
    mdx.stop()
    s3d.stop()


@pytest.fixture()
def model_dir(tmp_path):
    d = tmp_path / "model"
    d.mkdir()
    cfg = ModelConfig(description="baseline config 1", framework="pytorch",
                      model_files=["blob.bin"])
    (d / "modelx.yaml").write_text(cfg.to_yaml())
    (d / "blob.bin").write_bytes(os.urandom(10 * 1024 * 1024))  # 10 MiB
    return d


def test_presigned_push_pull_10mib(stack, model_dir, tmp_path):
    mdx, _ = stack
    c = Client(mdx.url)
    manifest = c.push("proj/cfg1", "v1", str(model_dir), quiet=True)
    blob = next(b for b in manifest.blobs if b.name == "blob.bin")
    assert blob.size == 10 * 1024 * 1024

    # the upload must have gone via a presigned location (provider s3)
    loc = c.remote.get_blob_location("proj/cfg1", blob, "download")
    assert loc is not None and loc.provider == "s3"
    assert loc.properties["parts"][0]["url"].startswith("http")

    out = tmp_path / "out"
    c.pull("proj/cfg1", "v1", str(out), quiet=True)
    assert (out / "blob.bin").read_bytes() == (model_dir / "blob.bin").read_bytes()


def test_presigned_url_is_ranged(stack, model_dir):
    """The pinned-ring engine depends on ranged GETs against one presigned
    URL; prove the signature stays valid with a Range header."""
    import requests

    mdx, _ = stack
    c = Client(mdx.url)
    manifest = c.push("proj/cfg1b", "v1", str(model_dir), quiet=True)
    blob = next(b for b in manifest.blobs if b.name == "blob.bin")
    loc = c.remote.get_blob_location("proj/cfg1b", blob, "download")
    url = loc.properties["parts"][0]["url"]
    r = requests.get(url, headers={"Range": "bytes=1024-2047"})
    assert r.status_code == 206
    assert len(r.content) == 1024
    assert r.content == (model_dir / "blob.bin").read_bytes()[1024:2048]


def test_multipart_upload_completes_on_manifest_put(stack, tmp_path):
    """Force multipart (part-count hint) and verify the server completes the
    pending upload at manifest PUT (store_s3.go:68-92 semantics)."""
    mdx, _ = stack
    c = Client(mdx.url)
    d = tmp_path / "mpmodel"
    d.mkdir()
    cfg = ModelConfig(description="mp", model_files=["big.bin"])
    (d / "modelx.yaml").write_text(cfg.to_yaml())
    (d / "big.bin").write_bytes(os.urandom(6 * 1024 * 1024))

    from modelx_amd.client import extension as ext
    from modelx_amd.client.push import parse_manifest

    manifest = parse_manifest(str(d))
    blob = next(b for b in manifest.blobs if b.name == "big.bin")
    # upload the config blob normally (manifest PUT verifies it too)
    cfg_loc = c.remote.get_blob_location("proj/mp", manifest.config, "upload")
    ext.get("s3").upload(manifest.config, cfg_loc,
                         ext.ContentSource(path=str(d / "modelx.yaml")))
    # ask for a multipart location explicitly
    loc = c.remote.get_blob_location("proj/mp", blob, "upload",
                                     extra={"multipart": "true", "part-count": "3"})
    assert loc.properties.get("multipart") is True
    assert len(loc.properties["parts"]) == 3
    s3ext = ext.get("s3")
    s3ext.upload(blob, loc, ext.ContentSource(path=str(d / "big.bin")))
    # blob is NOT visible yet (multipart pending)...
    # ...until manifest PUT completes it
    c.remote.put_manifest("proj/mp", "v1", manifest)
    assert c.remote.head_blob("proj/mp", blob.digest)

    out = tmp_path / "mp-out"
    c.pull("proj/mp", "v1", str(out), quiet=True)
    assert (out / "big.bin").read_bytes() == (d / "big.bin").read_bytes()


def test_size_mismatch_rejected_at_manifest_put(stack, tmp_path):
    """Manifest PUT verifies stored sizes and deletes mismatches
    (store_s3.go:77-88)."""
    from modelx_amd.client import extension as ext
    from modelx_amd.client.push import parse_manifest
    from modelx_amd.wire import errors as er

    mdx, _ = stack
    c = Client(mdx.url)
    d = tmp_path / "szmodel"
    d.mkdir()
    (d / "modelx.yaml").write_text(ModelConfig(description="sz").to_yaml())
    (d / "data.bin").write_bytes(os.urandom(128 * 1024))
    manifest = parse_manifest(str(d))
    blob = next(b for b in manifest.blobs if b.name == "data.bin")
    loc = c.remote.get_blob_location("proj/szbad", blob, "upload")
    # upload TRUNCATED content
    s3ext = ext.get("s3")
    s3ext.upload(blob, loc, ext.ContentSource(data=(d / "data.bin").read_bytes()[:1000]))
    with pytest.raises(er.ModelxError):
        c.remote.put_manifest("proj/szbad", "v1", manifest)
    # mismatched blob was deleted server-side
    assert not c.remote.head_blob("proj/szbad", blob.digest)


def test_presigned_url_expiry_enforced(tmp_path):
    """Expired presigned URLs are rejected by the object server
    (reference: --s3-presign-expire, pkg/registry/fs_s3.go:37; sigv4
    expiry check is ours — MinIO does the same)."""
    import time

    import requests

    from util_servers import MODELXD, ServerProc, free_port, start_s3d, wait_http

    s3d = start_s3d(str(tmp_path / "s3"))
    port = free_port()
    mdx = ServerProc([MODELXD, "--listen", f"127.0.0.1:{port}", "--s3-url", s3d.url,
                      "--s3-bucket", "modelx", "--s3-access-key", "modelx",
                      "--s3-secret-key", "modelx123", "--enable-redirect",
                      "--s3-presign-expire", "1"], port)
    try:
        wait_http(port)
        d = tmp_path / "m"
        d.mkdir()
        (d / "modelx.yaml").write_text(ModelConfig(description="exp").to_yaml())
        (d / "w.bin").write_bytes(os.urandom(256 * 1024))
        c = Client(mdx.url)
        c.push("exp/model", "v1", str(d), quiet=True)
        manifest = c.get_manifest("exp/model", "v1")
        (desc,) = [b for b in manifest.blobs if b.name == "w.bin"]
        loc = c.remote.get_blob_location("exp/model", desc, "download")
        url = loc.properties["parts"][0]["url"]
        assert requests.get(url, timeout=10).status_code == 200  # fresh: works
        time.sleep(2.5)
        r = requests.get(url, timeout=10)
        assert r.status_code == 403, f"expired presign must 403, got {r.status_code}"
    finally:
        mdx.stop()
        s3d.stop()


def test_pull_plan_endpoint(stack, tmp_path):
    """Pull-plan: one response carrying manifest + presigned locations +
    verified-inlineable leaves (MI355X addition; measured to remove the
    per-blob control-plane round trips that dominate small-blob indexes)."""
    import hashlib

    import requests

    from modelx_amd.wire import digest as dg

    mdx, _ = stack
    c = Client(mdx.url)
    payload = os.urandom(300 * 1024)
    leaves = b"".join(hashlib.sha256(payload[o : o + (128 << 10)]).digest()
                      for o in range(0, len(payload), 128 << 10))
    pd = dg.sha256_digest(payload)
    ld = dg.sha256_digest(leaves)
    c.remote.upload_blob_content("plan/model", types.Descriptor(
        name="w.bin", digest=pd, size=len(payload)), payload)
    c.remote.upload_blob_content("plan/model", types.Descriptor(
        name="w.bin.leaves", digest=ld, size=len(leaves)), leaves)
    cfg = b"description: plan\n"
    cd = dg.sha256_digest(cfg)
    c.remote.upload_blob_content("plan/model", types.Descriptor(
        name="modelx.yaml", digest=cd, size=len(cfg)), cfg)
    from datetime import datetime, timezone

    m = types.Manifest(media_type=types.MEDIA_TYPE_MODEL_MANIFEST_JSON)
    m.config = types.Descriptor(name="modelx.yaml", digest=cd, size=len(cfg),
                                media_type=types.MEDIA_TYPE_MODEL_CONFIG_YAML,
                                modified=datetime.now(timezone.utc))
    m.blobs = [
        types.Descriptor(name="w.bin", digest=pd, size=len(payload),
                         media_type=types.MEDIA_TYPE_MODEL_FILE,
                         modified=datetime.now(timezone.utc),
                         annotations={types.ANNOTATION_LEAVES_BLOB: ld,
                                      types.ANNOTATION_CHUNK_SIZE: str(128 << 10)}),
        types.Descriptor(name="w.bin.leaves", digest=ld, size=len(leaves),
                         media_type=types.MEDIA_TYPE_MODEL_LEAVES,
                         modified=datetime.now(timezone.utc)),
    ]
    c.remote.put_manifest("plan/model", "v1", m)

    plan = c.remote.get_pull_plan("plan/model", "v1")
    assert plan is not None
    got_m = types.Manifest.from_dict(plan["manifest"])
    assert [b.name for b in got_m.blobs] == [b.name for b in m.blobs]
    entry = plan["blobs"][pd]
    # presigned location is directly fetchable
    part = entry["location"]["properties"]["parts"][0]
    r = requests.get(part["url"], headers={"Range": "bytes=0-99"}, timeout=10)
    assert r.status_code == 206 and r.content == payload[:100]
    # inlined leaves round-trip and verify
    import base64

    assert base64.b64decode(entry["leaves64"]) == leaves

    # client-side verification rejects tampered plan leaves
    from modelx_amd.client.gpu import GpuClient

    desc = m.blobs[0]
    good = GpuClient._plan_leaves({"leaves64": entry["leaves64"]}, desc)
    assert good == leaves
    bad64 = base64.b64encode(b"x" + leaves[1:]).decode()
    assert GpuClient._plan_leaves({"leaves64": bad64}, desc) is None

    # unknown manifest -> None (client falls back and 404s properly)
    assert c.remote.get_pull_plan("plan/model", "nope") is None


def test_pull_plans_batch_endpoint(stack, tmp_path):
    """POST /{name}/pull-plans: ONE round trip for many versions' plans
    (config-5-shaped indexes store one blob per version). Unknown refs are
    omitted; each returned plan matches the per-version GET."""
    mdx, _ = stack
    c = Client(mdx.url)
    d = tmp_path / "bp"
    d.mkdir()
    from modelx_amd.config import ModelConfig

    (d / "modelx.yaml").write_text(ModelConfig(description="bp").to_yaml())
    for v in ("v1", "v2", "v3"):
        (d / "w.bin").write_bytes(os.urandom(64 * 1024) + v.encode())
        c.push("plan/batch", v, str(d), quiet=True)

    plans = c.remote.get_pull_plans("plan/batch", ["v1", "v2", "v3", "missing"])
    assert plans is not None
    assert set(plans) == {"v1", "v2", "v3"}
    for v in ("v1", "v2", "v3"):
        single = c.remote.get_pull_plan("plan/batch", v)
        assert plans[v]["manifest"] == single["manifest"]
        assert set(plans[v]["blobs"]) == set(single["blobs"])
    assert c.remote.get_pull_plans("plan/batch", []) == {}
    # oversized ref lists are rejected, not served
    import requests

    r = requests.post(mdx.url + "/plan/batch/pull-plans",
                      json={"refs": ["x"] * 5000}, timeout=10)
    assert r.status_code == 400
