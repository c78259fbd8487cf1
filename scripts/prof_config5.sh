#!/bin/bash
# rocprofv3 kernel-stats capture of the config-5 shaped bench (run on a GPU
# box via gpurun). rocprofv3 wants a writable cwd (TMPDIR=/tmp, cd /tmp);
# the workload itself runs from the repo root.
set -x
REPO="$(cd "$(dirname "$0")/.." && pwd)"
export TMPDIR=/tmp
mkdir -p "$REPO/gpurun_out"
cd /tmp
rocprofv3 --kernel-trace --stats -d "$REPO/gpurun_out/prof5" -o c5 -- \
  bash -c "cd '$REPO' && python tools/bench_shapes.py config5 --scale 0.02 --steps 1 > gpurun_out/c5prof.json 2>&1"
rc=$?
echo "prof_rc=$rc"
ls "$REPO/gpurun_out/prof5" 2>/dev/null
head -c 300 "$REPO/gpurun_out/c5prof.json" 2>/dev/null
exit $rc
