#!/bin/bash
# PMC capture for the round-2 kernels: the one-launch many-buffer leaf
# hasher and the huffman-window zstd decode on a bf16 payload.
# NOTE: rocprofv3 --pmc must not be combined with trace domains.
set -x
REPO="$(cd "$(dirname "$0")/.." && pwd)"
export TMPDIR=/tmp
mkdir -p "$REPO/gpurun_out"
cd /tmp
rocprofv3 --pmc SQ_INSTS_VALU SQ_LDS_BANK_CONFLICT -d "$REPO/gpurun_out/pmc_r2" -o r2 -- \
  bash -c "cd '$REPO' && python tools/pmc_workload_r2.py > gpurun_out/pmc_workload.log 2>&1"
rc=$?
echo "pmc_rc=$rc"
ls "$REPO/gpurun_out/pmc_r2" 2>/dev/null
cat "$REPO/gpurun_out/pmc_workload.log" 2>/dev/null | tail -3
exit $rc
