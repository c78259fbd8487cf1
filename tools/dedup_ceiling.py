"""Dedup ceiling: pull a blob whose chunks are ALL HBM-resident — the
transfer degenerates to the on-device probe + D2D gather (dedup.hip), so
the measured rate is the chunk-dedup path's ceiling, not the network's."""
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                                "tests"))

import torch

from modelx_amd.client.gpu import GpuClient
from util_servers import start_modelxd_s3, start_s3d


def main():
    gib = float(sys.argv[1]) if len(sys.argv) > 1 else 16.0
    n = int(gib * (1 << 30))
    work = tempfile.mkdtemp(prefix="dedup-ceil-")
    s3d = start_s3d(os.path.join(work, "s3"))
    mdx = start_modelxd_s3(s3d.url, redirect=True)
    try:
        g = GpuClient(mdx.url, device=0, dedup=True)
        src = torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda:0")
        g.push_from_gpu("ceil/a", "v1", {"blob.bin": src})  # registers chunks
        # same content under a different name: every chunk resident
        g.push_from_gpu("ceil/b", "v1", {"blob.bin": src})
        torch.cuda.synchronize()
        t0 = time.monotonic()
        out = g.pull_to_gpu("ceil/b", "v1")
        torch.cuda.synchronize()
        dt = time.monotonic() - t0
        assert torch.equal(out["blob.bin"], src)
        dd = [s for s in g.last_stats if s.get("phase") == "pull-dedup"]
        print(f"fully-resident pull: {gib:.0f} GiB in {dt * 1e3:.0f} ms = "
              f"{gib / dt:.1f} GiB/s (dedup_bytes={dd[-1]['dedup_bytes'] if dd else 0}, "
              f"fetched={dd[-1]['bytes'] if dd else '?'})")
    finally:
        mdx.stop()
        s3d.stop()


if __name__ == "__main__":
    main()
