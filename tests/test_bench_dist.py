"""Dry-run of bench.py's multi-rank orchestration under the EXACT torchrun
invocation the driver uses at round end (one rank per GPU) — on CPU via
MODELX_BENCH_CPU=1 (gloo, fake transfer client). Validates rank/port/store
layout, barrier flow, MAX-reduce and the one-line JSON contract for
world_size 2 without needing 8 GPUs."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


import pytest


def _run(tmp_path, world, extra, port):
    env = dict(os.environ, MODELX_BENCH_CPU="1", MASTER_ADDR="127.0.0.1")
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", str(world), "--master-addr", "127.0.0.1",
           "--master-port", str(port), os.path.join(REPO, "bench.py"),
           "--gpus", str(world), "--steps", "2", "--warmup", "1",
           "--blob-gib", "0.002", "--store", str(tmp_path / "store")] + extra
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=300, cwd=REPO, env=env)
    assert r.returncode == 0, r.stdout[-1500:] + r.stderr[-1500:]
    lines = [ln for ln in r.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, f"exactly one JSON line expected, got {lines}"
    return json.loads(lines[0])


@pytest.mark.parametrize("world", [2, 4])
def test_torchrun_cpu_dry_run_fanout_default(tmp_path, world):
    """The driver's exact multi-rank invocation lands in fan-out mode:
    shared store, per-rank shard push, every shard broadcast from its owner
    (config-4 semantics — what SCALE measures on an 8-GPU node)."""
    out = _run(tmp_path, world, [], 29671 + world * 40)
    assert out["n_gpus"] == world
    assert out["steps"] == 2
    assert out["scaling"] == "weak"
    assert out["value"] > 0
    assert out["metric"].startswith("push+pull GiB/s")
    assert out["config"]["parallelism"] == f"fanout{world}-rccl-xgmi-broadcast-pipelined"
    assert out["config"]["model"].startswith("config4-")


def test_torchrun_cpu_dry_run_independent_mode(tmp_path):
    """--mode independent keeps the per-rank-stack shape (no collectives in
    the step; A/B comparator for the fan-out)."""
    out = _run(tmp_path, 2, ["--mode", "independent"], 29891)
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "dp2-presigned-s3"
