"""GPU fan-out path tests (single-GPU degenerate forms; the 8-GPU scaling
run is the driver's). Exercises RCCL init (world 1) + ranged base_offset
pulls used by the pipelined broadcast."""
import os

import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def stack(tmp_path_factory):
    from util_servers import start_modelxd_s3, start_s3d

    s3_root = tmp_path_factory.mktemp("s3-fan")
    s3d = start_s3d(str(s3_root))
    mdx = start_modelxd_s3(s3d.url, redirect=True)
    yield mdx, s3d
    mdx.stop()
    s3d.stop()


def test_pull_with_base_offset(stack):
    from modelx_amd.client.gpu import GpuClient

    mdx, _ = stack
    g = GpuClient(mdx.url, device=0)
    src = torch.randint(0, 256, (32 << 20,), dtype=torch.uint8, device="cuda:0")
    g.push_from_gpu("fan/off", "v1", {"w.bin": src})
    desc = next(b for b in g.remote.get_manifest("fan/off", "v1").blobs if b.name == "w.bin")
    url, headers = g._download_url("fan/off", desc)
    dst = torch.zeros(8 << 20, dtype=torch.uint8, device="cuda:0")
    g.engine.pull_to_device(url, headers, 8 << 20, dst.data_ptr(), 4, 1 << 20)
    assert torch.equal(dst, src[1 << 20 : 9 << 20])


def test_fanout_broadcast_rccl_world1(stack):
    import torch.distributed as dist

    from modelx_amd.client.fanout import fanout_pull_broadcast
    from modelx_amd.client.gpu import GpuClient

    mdx, _ = stack
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29771")
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        g = GpuClient(mdx.url, device=0)
        src = torch.randint(0, 256, (24 << 20,), dtype=torch.uint8, device="cuda:0")
        g.push_from_gpu("fan/bcast", "v1", {"w.bin": src})
        out = fanout_pull_broadcast(dist, g, "fan/bcast", "v1", device=0,
                                    chunk=4 << 20)
        assert torch.equal(out["w.bin"], src)
    finally:
        dist.destroy_process_group()


def test_fanout_sharded_world1(stack):
    import torch.distributed as dist

    from modelx_amd.client.fanout import fanout_pull_sharded
    from modelx_amd.client.gpu import GpuClient

    mdx, _ = stack
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29772")
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        g = GpuClient(mdx.url, device=0)
        tensors = {f"shard{i}.bin": torch.randint(0, 256, (4 << 20,), dtype=torch.uint8,
                                                  device="cuda:0") for i in range(3)}
        g.push_from_gpu("fan/shard", "v1", tensors)
        out = fanout_pull_sharded(dist, g, "fan/shard", "v1", device=0, replicate=True)
        for name, t in tensors.items():
            assert torch.equal(out[name], t)
    finally:
        dist.destroy_process_group()


def test_chunk_level_resume_fetches_only_bad_chunks(stack):
    """Corrupt 2 MiB of a 32 MiB resident copy → resume fetches only the bad
    chunks, not the blob (SURVEY.md §5 fault-injection requirement)."""
    from modelx_amd.client.gpu import GpuClient

    mdx, _ = stack
    g = GpuClient(mdx.url, device=0)
    src = torch.randint(0, 256, (32 << 20,), dtype=torch.uint8, device="cuda:0")
    g.push_from_gpu("fan/resume", "v1", {"w.bin": src})
    desc = next(b for b in g.remote.get_manifest("fan/resume", "v1").blobs
                if b.name == "w.bin")
    # local copy with a 2 MiB hole
    local = src.clone()
    local[5 << 20 : 7 << 20] = 0
    g.last_stats.clear()
    out = g.pull_blob_to_device("fan/resume", desc, tensor=local, resume=True)
    assert torch.equal(out, src)
    st = next(s for s in g.last_stats if s.get("phase") == "pull-resume")
    assert st["bytes"] <= 3 << 20  # only the corrupted region (+ boundary chunks)
    assert st["skipped"] >= 29 << 20


def test_leaves_sidecar_in_manifest(stack):
    from modelx_amd.client.gpu import GpuClient
    from modelx_amd.wire import types as wt

    mdx, _ = stack
    g = GpuClient(mdx.url, device=0)
    src = torch.randint(0, 256, (4 << 20,), dtype=torch.uint8, device="cuda:0")
    m = g.push_from_gpu("fan/leaves", "v1", {"w.bin": src})
    names = {b.name: b for b in m.blobs}
    assert "w.bin.leaves" in names
    assert names["w.bin.leaves"].media_type == wt.MEDIA_TYPE_MODEL_LEAVES
    assert names["w.bin"].annotations[wt.ANNOTATION_LEAVES_BLOB] == names["w.bin.leaves"].digest
    # pull_to_gpu must skip the sidecar
    out = g.pull_to_gpu("fan/leaves", "v1")
    assert set(out) == {"w.bin"}


def test_tar_scatter_directory_to_gpu(stack, tmp_path):
    """Plain-tar directory blob → GPU tar_index + tar_scatter; files land as
    per-file HBM tensors matching the originals byte-for-byte."""
    import os as _os

    from modelx_amd.client import Client
    from modelx_amd.client.gpu import GpuClient
    from modelx_amd.config import ModelConfig
    from modelx_amd.wire import types as wt

    mdx, _ = stack
    d = tmp_path / "dirmodel"
    d.mkdir()
    (d / "modelx.yaml").write_text(ModelConfig(description="tar").to_yaml())
    sub = d / "shards"
    sub.mkdir()
    payloads = {}
    for i, size in enumerate([100, 512, 4096, 1 << 20, (1 << 20) + 777]):
        payloads[f"part{i}.bin"] = _os.urandom(size)
        (sub / f"part{i}.bin").write_bytes(payloads[f"part{i}.bin"])
    deep = sub / "nested" / "deeper"
    deep.mkdir(parents=True)
    payloads["nested/deeper/x-" + "l" * 120 + ".bin"] = _os.urandom(2048)  # long name
    (deep / ("x-" + "l" * 120 + ".bin")).write_bytes(
        payloads["nested/deeper/x-" + "l" * 120 + ".bin"])

    c = Client(mdx.url)
    c.push("gpu/tardir", "v1", str(d), dir_format="tar", quiet=True)
    manifest = c.get_manifest("gpu/tardir", "v1")
    dirblob = next(b for b in manifest.blobs if b.name == "shards")
    assert dirblob.media_type == wt.MEDIA_TYPE_MODEL_DIRECTORY_TAR

    g = GpuClient(mdx.url, device=0)
    files = g.pull_dir_to_gpu("gpu/tardir", dirblob)
    # names inside the archive carry the top-level dir prefix
    got = {name.split("/", 1)[1]: t for name, t in files.items()}
    assert set(got) == set(payloads)
    for name, data in payloads.items():
        assert bytes(got[name].cpu().numpy().tobytes()) == data


def test_plain_tar_cpu_pull_roundtrip(stack, tmp_path):
    """dir_format=tar round-trips through the normal CPU pull too."""
    import os as _os

    from modelx_amd.client import Client
    from modelx_amd.config import ModelConfig

    mdx, _ = stack
    d = tmp_path / "dm2"
    d.mkdir()
    (d / "modelx.yaml").write_text(ModelConfig(description="t2").to_yaml())
    sub = d / "data"
    sub.mkdir()
    (sub / "a.bin").write_bytes(_os.urandom(10000))
    c = Client(mdx.url)
    c.push("gpu/tardir2", "v1", str(d), dir_format="tar", quiet=True)
    out = tmp_path / "out2"
    c.pull("gpu/tardir2", "v1", str(out), quiet=True)
    assert (out / "data" / "a.bin").read_bytes() == (sub / "a.bin").read_bytes()


def test_modelxdl_gpus_flag(stack, tmp_path, capsys):
    """modelxdl --gpus lands selected shards straight into HBM
    (cmd/modelxdl contract + fanout_pull_single_process)."""
    import torch

    from modelx_amd.cli.dl import main as dl_main
    from modelx_amd.client.gpu import GpuClient
    from modelx_amd.config import ModelConfig

    mdx, _ = stack
    g = GpuClient(mdx.url, device=0, num_slots=4, slot_bytes=8 << 20)
    w = torch.randint(0, 256, (8 << 20,), dtype=torch.uint8, device="cuda:0")
    cfg = ModelConfig(description="dl", model_files=["weights.bin"])
    g.push_from_gpu("dl/model", "v1", {"weights.bin": w, "skipme.bin": w[: 1 << 20]},
                    config_yaml=cfg.to_yaml())
    rc = dl_main([f"{mdx.url}/dl/model@v1", "--gpus", "0"])
    assert rc == 0
    out = capsys.readouterr().out
    assert "weights.bin" in out and "skipme.bin" not in out


def test_fanout_broadcast_rccl_world2():
    """World-2 fan-out with TWO processes on real device tensors. RCCL
    rejects two ranks on one device ("Duplicate GPU detected",
    init.cc:1108 — verified on hardware), so the helper picks the
    transport by hardware: real RCCL when a GPU per rank exists (the
    driver's 8-GPU box runs this as true RCCL world-2), gloo transport
    for the CUDA tensors on a 1-GPU box — either way the full multi-rank
    choreography (pipelined broadcast, ShardPlan replicate,
    digest-after-collective on every rank) executes beyond world-1."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", NCCL_DEBUG="WARN")
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    log_dir = os.path.join(repo, "gpurun_out", "rccl2-logs")
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", "29881", "--log-dir", log_dir,
           "--redirects", "3", "--tee", "3",
           os.path.join(repo, "tests", "rccl2_helper.py")]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                       cwd=repo, env=env)
    if r.returncode != 0:
        # surface each rank's own stderr tail (the launcher interleaves and
        # truncates)
        import glob

        tails = []
        for f in sorted(glob.glob(os.path.join(log_dir, "**", "*"), recursive=True)):
            if os.path.isfile(f):
                with open(f, errors="replace") as fh:
                    tails.append(f"== {f}\n" + fh.read()[-1500:])
        raise AssertionError("\n".join(tails)[-8000:] or r.stderr[-3000:])
    assert "RCCL2 OK" in r.stdout


def test_targz_compat_streamed_to_gpu(stack, tmp_path):
    """tar.gz compat directory blob lands via the STREAMED path (presigned
    GET → zlib inflate → bounded pinned staging → HBM tar scatter) with the
    stored bytes digest-verified on the fly — host memory stays bounded
    instead of buffering the whole gzip blob (reference pull.go:184-203
    pipes download∥extract)."""
    import os as _os

    from modelx_amd.client import Client
    from modelx_amd.client.gpu import GpuClient
    from modelx_amd.config import ModelConfig
    from modelx_amd.wire import types as wt

    mdx, _ = stack
    d = tmp_path / "gzmodel"
    d.mkdir()
    (d / "modelx.yaml").write_text(ModelConfig(description="gz").to_yaml())
    sub = d / "weights"
    sub.mkdir()
    payloads = {}
    for i in range(4):
        payloads[f"w{i}.bin"] = _os.urandom(3 << 20)
        (sub / f"w{i}.bin").write_bytes(payloads[f"w{i}.bin"])
    c = Client(mdx.url)
    c.push("gpu/gzdir", "v1", str(d), quiet=True)  # default dir_format=tar.gz
    manifest = c.get_manifest("gpu/gzdir", "v1")
    dirblob = next(b for b in manifest.blobs if b.name == "weights")
    assert dirblob.media_type == wt.MEDIA_TYPE_MODEL_DIRECTORY_TARGZ

    # small slots force many staging flushes (the bounded-memory claim)
    g = GpuClient(mdx.url, device=0, num_slots=4, slot_bytes=1 << 20)
    files = g.pull_dir_to_gpu("gpu/gzdir", dirblob)
    got = {name.split("/", 1)[1]: t for name, t in files.items()}
    assert set(got) == set(payloads)
    for name, data in payloads.items():
        assert bytes(got[name].cpu().numpy().tobytes()) == data
    assert any(s.get("phase") == "pull-targz-stream" for s in g.last_stats)


def test_targz_streamed_detects_corruption(stack, tmp_path):
    import os as _os

    from modelx_amd.client import Client
    from modelx_amd.client.gpu import GpuClient
    from modelx_amd.config import ModelConfig
    from modelx_amd.wire import errors as er
    from modelx_amd.wire import paths as pm
    from modelx_amd.wire import types as wt

    mdx, s3d = stack
    d = tmp_path / "gzbad"
    d.mkdir()
    (d / "modelx.yaml").write_text(ModelConfig(description="gzb").to_yaml())
    sub = d / "data"
    sub.mkdir()
    (sub / "a.bin").write_bytes(_os.urandom(2 << 20))
    c = Client(mdx.url)
    c.push("gpu/gzbad", "v1", str(d), quiet=True)
    manifest = c.get_manifest("gpu/gzbad", "v1")
    dirblob = next(b for b in manifest.blobs if b.name == "data")
    root = s3d.proc.args[s3d.proc.args.index("--root") + 1]
    obj = os.path.join(root, "modelx", "registry",
                       pm.blob_digest_path("gpu/gzbad", dirblob.digest))
    with open(obj, "r+b") as f:
        f.seek(os.path.getsize(obj) // 2)
        b0 = f.read(1)
        f.seek(os.path.getsize(obj) // 2)
        f.write(bytes([b0[0] ^ 0x10]))
    g = GpuClient(mdx.url, device=0)
    with pytest.raises((er.ModelxError, Exception)):
        g.pull_dir_to_gpu("gpu/gzbad", dirblob)
