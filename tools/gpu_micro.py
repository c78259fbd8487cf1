#!/usr/bin/env python3
"""GPU micro-benchmarks: digest-kernel rate vs chunk size, pull-transfer
rate vs connection count, push parallelism. Run on an MI355X box."""
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))

import torch

from modelx_amd import _core


def main():
    print("nproc:", os.cpu_count())
    os.system("df -h /dev/shm | tail -1")
    eng = _core.GpuEngine(device=0, num_slots=10, slot_bytes=64 << 20, num_streams=4)

    size = 8 << 30
    buf = torch.empty(size, dtype=torch.uint8, device="cuda:0")
    buf.random_(0, 256)
    torch.cuda.synchronize()

    print("\n-- sha256_chunk_leaves rate vs chunk size (8 GiB buffer) --")
    for cs in [64 << 10, 128 << 10, 256 << 10, 512 << 10, 1 << 20]:
        eng.sha256_chunk_leaves(buf.data_ptr(), 1 << 20, cs)  # warm
        t0 = time.monotonic()
        eng.sha256_chunk_leaves(buf.data_ptr(), size, cs)
        dt = time.monotonic() - t0
        print(f"chunk={cs >> 10}KiB: {size / dt / (1 << 30):.1f} GiB/s "
              f"({size // cs} chunks, {dt * 1e3:.0f}ms)")

    # transfer tests against a local s3d (no-auth mode, sparse object)
    import subprocess

    from util_servers import S3D, free_port, wait_http

    s3root = "/dev/shm/modelx-micro" if os.path.isdir("/dev/shm") else "/tmp/modelx-micro"
    os.makedirs(os.path.join(s3root, "modelx"), exist_ok=True)
    obj = os.path.join(s3root, "modelx", "big.bin")
    with open(obj, "wb") as f:
        f.truncate(size)  # sparse: measures the transfer path, not disk
    objr = os.path.join(s3root, "modelx", "real.bin")
    os.system(f"dd if=/dev/zero of={objr} bs=64M count={size >> 26} status=none")
    port = free_port()
    proc = subprocess.Popen([S3D, "--listen", f"127.0.0.1:{port}", "--root", s3root,
                             "--no-auth"], stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    wait_http(port)
    url = f"http://127.0.0.1:{port}/modelx/big.bin"
    try:
        print("\n-- pull_to_device rate vs conns (8 GiB sparse object) --")
        for conns in [4, 8, 16, 24, 32]:
            stats = eng.pull_to_device(url, {}, size, buf.data_ptr(), conns)
            print(f"conns={conns}: {stats['gib_per_s']:.1f} GiB/s")

        print("\n-- pull_to_device REAL pages (8 GiB, 16 conns) --")
        urlr = f"http://127.0.0.1:{port}/modelx/real.bin"
        for rep in range(2):
            stats = eng.pull_to_device(urlr, {}, size, buf.data_ptr(), 16)
            print(f"real rep{rep}: {stats['gib_per_s']:.1f} GiB/s")

        print("\n-- pull rate vs slot size (16 conns) --")
        for slot_mib in [16, 32, 64, 128]:
            e2 = _core.GpuEngine(device=0, num_slots=max(8, 2 * 16), slot_bytes=slot_mib << 20,
                                 num_streams=4)
            stats = e2.pull_to_device(url, {}, size, buf.data_ptr(), 16)
            print(f"slot={slot_mib}MiB: {stats['gib_per_s']:.1f} GiB/s")
            del e2

        print("\n-- push_part_from_device rate vs parallel parts (4 GiB) --")
        import glob
        from concurrent.futures import ThreadPoolExecutor

        psize = 4 << 30
        for npar in [2, 4, 8, 12]:
            part = psize // npar
            t0 = time.monotonic()
            with ThreadPoolExecutor(max_workers=npar) as pool:
                futs = [pool.submit(eng.push_part_from_device,
                                    f"http://127.0.0.1:{port}/modelx/up-{npar}-{i}", "PUT", {},
                                    buf.data_ptr() + i * part, part)
                        for i in range(npar)]
                [f.result() for f in futs]
            dt = time.monotonic() - t0
            print(f"parallel={npar}: {psize / dt / (1 << 30):.1f} GiB/s")
            for f in glob.glob(os.path.join(s3root, "modelx", "up-*")):
                os.unlink(f)
    finally:
        proc.terminate()


if __name__ == "__main__":
    main()
