#!/usr/bin/env python3
"""Is the loopback S3 path full-duplex? Measures push-only, pull-only, and
concurrent push+pull aggregate on one GpuClient — decides whether a
pipelined bench step (pull step k-1 while pushing step k) would pay."""
import os
import sys
import threading
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))

import torch

from modelx_amd.client.gpu import GpuClient
from util_servers import start_modelxd_s3, start_s3d

GIB = float(os.environ.get("DUPLEX_GIB", "4"))
N = int(GIB * (1 << 30))

s3d = start_s3d("/dev/shm/duplex-s3")
mdx = start_modelxd_s3(s3d.url, redirect=True)
try:
    g = GpuClient(mdx.url, device=0)
    a = torch.randint(0, 256, (N,), dtype=torch.uint8, device="cuda:0")
    b = torch.randint(0, 256, (N,), dtype=torch.uint8, device="cuda:0")
    torch.cuda.synchronize()
    g.push_from_gpu("dx/m", "va", {"a.bin": a})

    t0 = time.monotonic()
    g.push_from_gpu("dx/m", "vb", {"b.bin": b})
    t_push = time.monotonic() - t0

    t0 = time.monotonic()
    g.pull_to_gpu("dx/m", "va")
    t_pull = time.monotonic() - t0

    # concurrent: push a fresh version while pulling another
    c = torch.randint(0, 256, (N,), dtype=torch.uint8, device="cuda:0")
    torch.cuda.synchronize()
    t0 = time.monotonic()
    th = threading.Thread(target=lambda: g.push_from_gpu("dx/m", "vc", {"c.bin": c}))
    th.start()
    g.pull_to_gpu("dx/m", "vb")
    th.join()
    t_both = time.monotonic() - t0

    print(f"push-only : {GIB / t_push:6.2f} GiB/s")
    print(f"pull-only : {GIB / t_pull:6.2f} GiB/s")
    print(f"concurrent: {2 * GIB / t_both:6.2f} GiB/s aggregate "
          f"(speedup vs serial {(t_push + t_pull) / t_both:4.2f}x)")
finally:
    mdx.stop()
    s3d.stop()
