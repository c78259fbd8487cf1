"""Repro: GPU-compress the config5-shaped tiled payload, decode with BOTH
the CPU codec and libzstd per frame; dump any frame that fails to
gpurun_out/ for offline bitstream analysis."""
import ctypes
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from modelx_amd import _core


def main():
    eng = _core.GpuEngine(device=0, num_slots=4, slot_bytes=8 << 20, num_streams=2)
    z = ctypes.CDLL("libzstd.so.1")
    z.ZSTD_decompress.restype = ctypes.c_size_t
    z.ZSTD_isError.restype = ctypes.c_uint

    import sys as _sys

    kind = _sys.argv[1] if len(_sys.argv) > 1 else "tiled"
    n = 64 << 20
    if kind == "lowent":
        src = torch.randint(0, 64, (n,), dtype=torch.uint8, device="cuda:0")
    else:
        page = torch.randint(0, 256, (64 << 10,), dtype=torch.uint8, device="cuda:0")
        src = page.repeat(n // page.numel())[:n].contiguous()
        torch.manual_seed(1000)
        src[: 4 << 20] = torch.randint(0, 256, (4 << 20,), dtype=torch.uint8,
                                       device="cuda:0")
    torch.cuda.synchronize()

    bound = _core.zstd_compress_bound(n)
    comp = torch.empty(bound, dtype=torch.uint8, device="cuda:0")
    csize = eng.zstd_compress_device(src.data_ptr(), n, 128 << 10, comp.data_ptr(), bound)
    blob = bytes(comp[:csize].cpu().numpy().tobytes())
    print(f"gpu-compressed {n} -> {csize} (ratio {csize / n:.4f})")
    frames = _core.zstd_frames(blob)
    print(f"{len(frames)} frames")

    src_host = bytes(src.cpu().numpy().tobytes())
    os.makedirs("gpurun_out", exist_ok=True)
    bad = 0
    for i, (c_off, c_size, d_off, d_size) in enumerate(frames):
        fb = blob[c_off : c_off + c_size]
        expect = src_host[d_off : d_off + d_size]
        # libzstd oracle on the single frame
        out = ctypes.create_string_buffer(d_size)
        m = z.ZSTD_decompress(out, d_size, fb, len(fb))
        lib_ok = (not z.ZSTD_isError(m)) and m == d_size and out.raw == expect
        # CPU codec
        try:
            ours = _core.zstd_decompress_cpu(fb)
            cpu_ok = ours == expect
        except Exception as e:
            cpu_ok = False
        if not (lib_ok and cpu_ok):
            bad += 1
            if bad <= 3:
                path = f"gpurun_out/badframe_{i}.bin"
                with open(path, "wb") as f:
                    f.write(fb)
                with open(f"gpurun_out/badframe_{i}.raw", "wb") as f:
                    f.write(expect)
                print(f"frame {i}: lib_ok={lib_ok} cpu_ok={cpu_ok} "
                      f"c_size={c_size} d_size={d_size} -> dumped {path}")
    print(f"bad frames: {bad} / {len(frames)}")
    # CPU-compress the same content and GPU-decode it (isolates decoder)
    cpu_blob = _core.zstd_compress_cpu(src_host, 128 << 10)
    csrc = torch.frombuffer(bytearray(cpu_blob), dtype=torch.uint8).cuda()
    cback = torch.empty(n, dtype=torch.uint8, device="cuda:0")
    try:
        m2 = eng.zstd_decompress_device(csrc.data_ptr(), len(cpu_blob), cback.data_ptr(), n)
        ok2 = m2 == n and bytes(cback.cpu().numpy().tobytes()) == src_host
        print("gpu decode of CPU blob:", "OK" if ok2 else "MISMATCH")
    except Exception as e:
        print("gpu decode of CPU blob FAILED:", e)
    # also exercise GPU decode end-to-end
    back = torch.empty(n, dtype=torch.uint8, device="cuda:0")
    try:
        m = eng.zstd_decompress_device(comp.data_ptr(), csize, back.data_ptr(), n)
        print("gpu decode:", "OK" if (m == n and torch.equal(back, src)) else "MISMATCH")
    except Exception as e:
        print("gpu decode FAILED:", e)


if __name__ == "__main__":
    main()
