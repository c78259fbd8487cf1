"""PMC workload for the round-2 kernels (run under rocprofv3 --pmc via
scripts/prof_pmc_r2.sh): one many-buffer leaf-hash launch + one
huffman-window bf16 decode."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from modelx_amd import _core

eng = _core.GpuEngine(device=0, num_slots=4, slot_bytes=32 << 20, num_streams=4)
torch.manual_seed(1)
bufs = [torch.randint(0, 256, (64 << 20,), dtype=torch.uint8, device="cuda")
        for _ in range(32)]
torch.cuda.synchronize()
eng.sha256_chunk_leaves_many([(b.data_ptr(), b.numel(), 128 << 10) for b in bufs])
del bufs
data = (torch.randn(1 << 29, device="cuda") * 0.02).to(torch.bfloat16).view(torch.uint8)
torch.cuda.synchronize()
n = data.numel()
bound = _core.zstd_compress_bound(n)
dst = torch.empty(bound, dtype=torch.uint8, device="cuda")
sz = eng.zstd_compress_device(data.data_ptr(), n, 128 << 10, dst.data_ptr(), bound)
back = torch.empty(n, dtype=torch.uint8, device="cuda")
m = eng.zstd_decompress_device(dst.data_ptr(), sz, back.data_ptr(), n)
assert m == n and torch.equal(back, data)
print("workload ok")
