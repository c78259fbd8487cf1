"""zstd kernel throughput on MI355X (run via gpurun).

    python tools/zstd_prof.py [--gib 2]

Measures GPU compress + decompress GiB/s on three payload shapes
(compressible tiled pages, text, incompressible random) and verifies the
roundtrip. rocprofv3 over this script gives per-kernel evidence
(zstd_compress_frames_kernel / zstd_decompress_frames_kernel).
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from modelx_amd import _core


def payload(kind: str, nbytes: int) -> torch.Tensor:
    if kind == "random":
        return torch.randint(0, 256, (nbytes,), dtype=torch.uint8, device="cuda")
    if kind == "low-entropy":  # 64 symbols uniform: pure huffman-literals work
        return torch.randint(0, 64, (nbytes,), dtype=torch.uint8, device="cuda")
    if kind == "tiled4k":
        page = torch.randint(0, 256, (4096,), dtype=torch.uint8, device="cuda")
        return page.repeat(nbytes // 4096 + 1)[:nbytes].contiguous()
    if kind == "text":
        s = (b"The quick brown fox jumps over the lazy dog. " * 100)[:4096]
        page = torch.frombuffer(bytearray(s), dtype=torch.uint8).cuda()
        return page.repeat(nbytes // 4096 + 1)[:nbytes].contiguous()
    raise ValueError(kind)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gib", type=float, default=2.0)
    ap.add_argument("--frame-kib", type=int, default=128)
    args = ap.parse_args()
    n = int(args.gib * (1 << 30))
    eng = _core.GpuEngine(device=0, num_slots=4, slot_bytes=8 << 20, num_streams=2)
    frame_raw = args.frame_kib << 10

    for kind in ("tiled4k", "text", "low-entropy", "random"):
        src = payload(kind, n)
        bound = _core.zstd_compress_bound(n, frame_raw)
        comp = torch.empty(bound, dtype=torch.uint8, device="cuda")
        torch.cuda.synchronize()
        # warmup (small)
        eng.zstd_compress_device(src.data_ptr(), 64 << 20, frame_raw, comp.data_ptr(), bound)
        torch.cuda.synchronize()
        t0 = time.monotonic()
        csize = eng.zstd_compress_device(src.data_ptr(), n, frame_raw, comp.data_ptr(), bound)
        torch.cuda.synchronize()
        t1 = time.monotonic()
        back = torch.empty(n, dtype=torch.uint8, device="cuda")
        eng.zstd_decompress_device(comp.data_ptr(), csize, back.data_ptr(), n)  # warmup
        torch.cuda.synchronize()
        t2 = time.monotonic()
        m = eng.zstd_decompress_device(comp.data_ptr(), csize, back.data_ptr(), n)
        torch.cuda.synchronize()
        t3 = time.monotonic()
        assert m == n, (m, n)
        assert torch.equal(back, src), kind
        comp_rate = n / (t1 - t0) / (1 << 30)
        dec_rate = n / (t3 - t2) / (1 << 30)
        print(f"{kind:8s} {args.gib:5.1f} GiB: compress {comp_rate:8.2f} GiB/s "
              f"ratio {csize / n:6.4f} | decompress {dec_rate:8.2f} GiB/s", flush=True)


if __name__ == "__main__":
    main()
