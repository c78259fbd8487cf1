"""Concurrent-pull stress on real hardware: the engine's pull path is
reentrant (shared pinned-slot pool, per-call range state) — these tests
hammer it from thread pools and verify CONTENT equality, not just digests,
isolating each layer: plain pulls, zstd pulls, multipart-stitched huge
blobs, and the dedup gather path."""
import os

import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def stack(tmp_path_factory):
    from util_servers import start_modelxd_s3, start_s3d

    root = tmp_path_factory.mktemp("conc")
    s3d = start_s3d(str(root / "s3"))
    mdx = start_modelxd_s3(s3d.url, redirect=True)
    yield mdx
    mdx.stop()
    s3d.stop()


@pytest.fixture(scope="module")
def client(stack):
    from modelx_amd.client.gpu import GpuClient

    return GpuClient(stack.url, device=0, num_slots=16, slot_bytes=16 << 20)


def _push_random(g, repo, versions, nbytes, compress=""):
    ref = {}
    for v in versions:
        t = torch.randint(0, 256, (nbytes,), dtype=torch.uint8, device="cuda:0")
        g.push_from_gpu(repo, v, {"blob.bin": t}, compress=compress)
        ref[v] = t
    return ref


class TestConcurrentPulls:
    def test_plain_parallel(self, client):
        vs = [f"p{i}" for i in range(8)]
        ref = _push_random(client, "conc/plain", vs, 48 << 20)
        for it in range(3):
            outs = client.pull_many("conc/plain", vs, parallel=8)
            for v in vs:
                assert torch.equal(outs[v]["blob.bin"], ref[v]), (it, v)

    def test_zstd_parallel(self, client):
        vs = [f"z{i}" for i in range(8)]
        base = torch.randint(0, 256, (1 << 20,), dtype=torch.uint8, device="cuda:0")
        ref = {}
        for v in vs:
            t = base.repeat(32).contiguous()  # compressible
            t[: 1 << 20] = torch.randint(0, 256, (1 << 20,), dtype=torch.uint8,
                                         device="cuda:0")
            client.push_from_gpu("conc/z", v, {"blob.bin": t}, compress="zstd")
            ref[v] = t
        for it in range(3):
            outs = client.pull_many("conc/z", vs, parallel=8)
            for v in vs:
                assert torch.equal(outs[v]["blob.bin"], ref[v]), (it, v)

    def test_multipart_stitched_parallel(self, client):
        # > part_bytes so the object is stored as a part manifest and every
        # ranged GET stitches across part files
        vs = [f"m{i}" for i in range(3)]
        ref = _push_random(client, "conc/mp", vs, 600 << 20)
        # small parts to force many stitch boundaries
        for v in vs:
            t = torch.randint(0, 256, (600 << 20,), dtype=torch.uint8, device="cuda:0")
            client.push_from_gpu("conc/mp2", v, {"blob.bin": t}, part_bytes=64 << 20)
            ref[v] = t
        outs = client.pull_many("conc/mp2", vs, parallel=3)
        for v in vs:
            assert torch.equal(outs[v]["blob.bin"], ref[v]), v

    def test_dedup_parallel(self, client):
        client.dedup = True
        client.clear_chunk_index()
        try:
            page = torch.randint(0, 256, (1 << 20,), dtype=torch.uint8, device="cuda:0")
            ref = {}
            for i in range(6):
                t = page.repeat(48).contiguous()  # 48 MiB of shared tiles
                t[: 2 << 20] = torch.randint(0, 256, (2 << 20,), dtype=torch.uint8,
                                             device="cuda:0")
                client.push_from_gpu("conc/dd", f"d{i}", {"blob.bin": t})
                ref[f"d{i}"] = t
            for it in range(2):
                client.clear_chunk_index()
                # prime the index with one pull, then hit it from 5 threads
                prime = client.pull_to_gpu("conc/dd", "d0")
                assert torch.equal(prime["blob.bin"], ref["d0"]), it
                outs = client.pull_many("conc/dd", [f"d{i}" for i in range(1, 6)],
                                        parallel=5)
                for v, t in ref.items():
                    if v == "d0":
                        continue
                    assert torch.equal(outs[v]["blob.bin"], t), (it, v)
            dd = [s for s in client.last_stats if s.get("phase") == "pull-dedup"]
            assert dd, "dedup path never taken"
        finally:
            client.dedup = False
            client.clear_chunk_index()
