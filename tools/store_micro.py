#!/usr/bin/env python3
"""A/B the object-store backing, PUT-then-immediately-GET (the bench
pattern). Mount-free variants (gpurun boxes refuse mounts): existing
/dev/shm (tmpfs) vs the scratch disk. Pass a mount-opts string as a third
variant only where mounting is permitted."""
import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))

import torch

from modelx_amd import _core
from util_servers import S3D, free_port, wait_http

SIZE = 8 << 30


def bench_store(tag, root_base=None, mount_opts=None):
    if mount_opts:
        root = f"/mnt/modelx-{tag}"
        os.makedirs(root, exist_ok=True)
        subprocess.run(["umount", root], capture_output=True)
        r = subprocess.run(["mount", "-t", "tmpfs", "-o", mount_opts, "none", root],
                           capture_output=True, text=True)
        if r.returncode != 0:
            print(f"{tag}: mount failed ({r.stderr.strip()}) — skipping")
            return
    else:
        root = os.path.join(root_base, f"modelx-{tag}")
    os.makedirs(f"{root}/modelx", exist_ok=True)
    port = free_port()
    proc = subprocess.Popen([S3D, "--listen", f"127.0.0.1:{port}", "--root", root, "--no-auth"],
                            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    wait_http(port)
    try:
        eng = _core.GpuEngine(device=0, num_slots=16, slot_bytes=64 << 20, num_streams=4)
        buf = torch.empty(SIZE, dtype=torch.uint8, device="cuda:0")
        buf.random_(0, 256)
        torch.cuda.synchronize()
        base = f"http://127.0.0.1:{port}/modelx"
        for rep in range(2):
            # push as 8 parallel 1 GiB parts (objects o0..o7)
            from concurrent.futures import ThreadPoolExecutor

            part = SIZE // 8
            t0 = time.monotonic()
            with ThreadPoolExecutor(max_workers=8) as pool:
                futs = [pool.submit(eng.push_part_from_device, f"{base}/o{i}", "PUT", {},
                                    buf.data_ptr() + i * part, part) for i in range(8)]
                [f.result() for f in futs]
            t_push = time.monotonic() - t0
            # immediately pull each object back (fresh pages, bench pattern)
            t0 = time.monotonic()
            for i in range(8):
                eng.pull_to_device(f"{base}/o{i}", {}, part, buf.data_ptr() + i * part, 8)
            t_pull = time.monotonic() - t0
            print(f"{tag} rep{rep}: push {SIZE / t_push / (1 << 30):.1f} GiB/s, "
                  f"pull {SIZE / t_pull / (1 << 30):.1f} GiB/s")
            for i in range(8):
                os.unlink(f"{root}/modelx/o{i}")
    finally:
        proc.terminate()
        proc.wait()
        if mount_opts:
            subprocess.run(["umount", root], capture_output=True)
        else:
            import shutil

            shutil.rmtree(root, ignore_errors=True)


if __name__ == "__main__":
    bench_store("shm", root_base="/dev/shm")
    bench_store("disk", root_base="/tmp")
    bench_store("hugetmpfs", mount_opts="size=48g,huge=always")
