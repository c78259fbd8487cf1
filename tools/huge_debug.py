"""Isolate the huge-multipart corruption: push one config5-shaped 2.4 GiB
blob, then (a) compare a GPU pull against the source tensor, and (b)
compare the SERVER-side object bytes against the source via independent
CPU ranged GETs — distinguishing push-time corruption (stored bytes wrong)
from pull-time corruption (stored right, landed wrong). Prints bad chunk
positions relative to part boundaries."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                                "tests"))

import hashlib
import tempfile

import requests
import torch

from modelx_amd.client.gpu import GpuClient
from util_servers import start_modelxd_s3, start_s3d

CS = 128 << 10
SIZE = int(2.4 * (1 << 30))


def leaves_of_tensor(g, t):
    return g.engine.sha256_chunk_leaves(t.data_ptr(), t.numel(), CS)


def main():
    work = tempfile.mkdtemp(prefix="huge-dbg-")
    s3d = start_s3d(os.path.join(work, "s3"))
    mdx = start_modelxd_s3(s3d.url, redirect=True)
    try:
        g = GpuClient(mdx.url, device=0)
        page = torch.randint(0, 256, (1 << 20,), dtype=torch.uint8, device="cuda:0")
        src = page.repeat(SIZE // page.numel() + 1)[:SIZE].contiguous()
        torch.manual_seed(2000)
        src[: 4 << 20] = torch.randint(0, 256, (4 << 20,), dtype=torch.uint8, device="cuda:0")
        src_leaves = leaves_of_tensor(g, src)
        for attempt in range(3):
            repo = f"dbg/huge{attempt}"
            g.push_from_gpu(repo, "v1", {"blob.bin": src})
            man = g.remote.get_manifest(repo, "v1")
            (desc,) = [d for d in man.blobs if d.name == "blob.bin"]
            nparts = 10
            psize = desc.size // nparts

            # (b) server-side content check via independent CPU ranged GETs
            loc = g.remote.get_blob_location(repo, desc, "download")
            url = loc.properties["parts"][0]["url"]
            bad_server = []
            step = 64 << 20
            for off in range(0, desc.size, step):
                ln = min(step, desc.size - off)
                r = requests.get(url, headers={
                    "Range": f"bytes={off}-{off + ln - 1}"}, timeout=120)
                assert r.status_code == 206, r.status_code
                data = r.content
                assert len(data) == ln, (len(data), ln)
                for c0 in range(0, ln, CS):
                    ci = (off + c0) // CS
                    cl = min(CS, ln - c0)
                    h = hashlib.sha256(data[c0 : c0 + cl]).digest()
                    if h != src_leaves[ci * 32 : (ci + 1) * 32]:
                        bad_server.append(ci)
            print(f"attempt {attempt}: server-side bad chunks: {len(bad_server)}")
            for ci in bad_server[:8]:
                off = ci * CS
                print(f"  chunk {ci} at {off} (part {off // psize}, "
                      f"{off % psize} into part, {psize - off % psize} before boundary)")

            # (a) GPU pull comparison
            got = g.pull_blob_to_device(repo, desc, verify=False)
            eq = torch.equal(got[:SIZE], src)
            if not eq:
                gl = leaves_of_tensor(g, got[:SIZE].contiguous())
                bad = [i for i in range(len(src_leaves) // 32)
                       if gl[i * 32 : (i + 1) * 32] != src_leaves[i * 32 : (i + 1) * 32]]
                print(f"  gpu pull bad chunks: {len(bad)} first={bad[:6]}")
            else:
                print("  gpu pull matches source")
            del got
            torch.cuda.empty_cache()
            if bad_server:
                break
    finally:
        mdx.stop()
        s3d.stop()


if __name__ == "__main__":
    main()
