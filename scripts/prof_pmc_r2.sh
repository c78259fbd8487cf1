#!/bin/bash
# PMC capture for the round-2 kernels: the one-launch many-buffer leaf
# hasher and the huffman-window zstd decode on a bf16 payload.
# rocprofv3 --pmc must NOT be combined with trace domains (pool rule).
set -x
REPO="$(cd "$(dirname "$0")/.." && pwd)"
export TMPDIR=/tmp
mkdir -p "$REPO/gpurun_out"
cd /tmp
rocprofv3 --pmc must NOT be combined with trace domains (pool rule).
set -x
REPO="$(cd "$(dirname "$0")/.." && pwd)"
export TMPDIR=/tmp
mkdir -p "$REPO/gpurun_out"
cd /tmp
cat > /tmp/pmc_workload.py <<'EOF'
import torch
from modelx_amd import _core
eng = _core.GpuEngine(device=0, num_slots=4, slot_bytes=32 << 20, num_streams=4)
torch.manual_seed(1)
# many-buffer leaves: 32 x 64 MiB blobs in one launch
bufs = [torch.randint(0, 256, (64 << 20,), dtype=torch.uint8, device="cuda")
        for _ in range(32)]
torch.cuda.synchronize()
eng.sha256_chunk_leaves_many([(b.data_ptr(), b.numel(), 128 << 10) for b in bufs])
del bufs
# huffman-window decode: 1 GiB bf16
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
EOF
rocprofv3 --pmc SQ_INSTS_VALU SQ_LDS_BANK_CONFLICT -d "$REPO/gpurun_out/pmc_r2" -o r2 -- \
  bash -c "cd '$REPO' && python tools/pmc_workload_r2.py > gpurun_out/pmc_workload.log 2>&1"
rc=$?
echo "pmc_rc=$rc"
ls "$REPO/gpurun_out/pmc_r2" 2>/dev/null
cat "$REPO/gpurun_out/pmc_workload.log" 2>/dev/null | tail -3
exit $rc
