#!/usr/bin/env python3
"""modelx_amd flagship benchmark — push+pull GiB/s end-to-end (S3→HBM,
digest-verified) on 1..8 MI355X (BASELINE.json metric).

One *step* per rank = push one synthetic random blob from HBM to S3 via
presigned multipart (GPU chunk-digest first) + pull it back into HBM via the
pinned-ring ranged-GET engine with CDNA4 SHA-256 verification. Per-GPU work
is fixed as N grows (weak scaling). The S3 store is the bundled modelx-s3d
(MinIO stand-in) on tmpfs, modelxd coordinates with --enable-redirect; both
run on this node — the metric measures the client data plane, the store is
sized to not be the bottleneck.

  python bench.py --gpus 1 --steps 3 --warmup 1            # single GPU
  torchrun --nproc-per-node 8 bench.py --gpus 8 ...        # driver form

Rank 0 prints exactly one JSON line with the whole-job aggregate.
"""
import argparse
import json
import os
import shutil
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--blob-gib", type=float, default=8.0,
                   help="per-rank blob size per step (GiB; 8 amortizes per-blob "
                        "control plane, closest bench shape to BASELINE config 2)")
    p.add_argument("--conns", type=int, default=8,
                   help="ranged-GET connections per rank (8 measured best under the "
                        "pipelined step: pull sockets contend with the concurrent "
                        "push — 23.1-25.0 vs 19.2-21.2 GiB/s at 16, 3x alternating "
                        "A/B on one box; pull-only workloads still prefer 16, "
                        "GpuClient's default)")
    p.add_argument("--slot-mib", type=int, default=64)
    p.add_argument("--slots", type=int, default=16)
    p.add_argument("--part-mib", type=int, default=256, help="push part size")
    p.add_argument("--store", default="", help="s3d store root (default: tmpfs)")
    p.add_argument("--cpu-smoke", action="store_true",
                   help="CPU-only plumbing run (BASELINE config 1, 10 MiB)")
    p.add_argument("--mode", choices=["auto", "fanout", "independent"], default="auto",
                   help="multi-GPU step shape: 'fanout' = one shared store, each rank "
                        "pushes 1/N of a fixed-size index, every shard RCCL-broadcast "
                        "over xGMI from its owner so ALL GPUs land the full index, "
                        "digest-verified per GPU (BASELINE config 4 semantics — the "
                        "mode SCALE runs); 'independent' = each rank its own stack "
                        "and blob (no collectives). auto = fanout when world>1.")
    p.add_argument("--chunk-mib", type=int, default=256,
                   help="fan-out broadcast pipeline chunk (MiB)")
    return p.parse_args()


def pick_store_root(args):
    if args.store:
        return args.store
    if os.path.isdir("/dev/shm"):
        return "/dev/shm/modelx-bench"
    return os.path.join(REPO, "gpurun_out", "bench-store")


def cpu_smoke(args):
    """BASELINE config 1: init + push/pull one 10 MiB blob, CPU only."""
    import tempfile

    from modelx_amd.client import Client
    from modelx_amd.config import ModelConfig
    from util_servers import start_modelxd_s3, start_s3d

    work = tempfile.mkdtemp(prefix="modelx-cpu-bench-")
    s3d = start_s3d(os.path.join(work, "s3"))
    mdx = start_modelxd_s3(s3d.url, redirect=True)
    try:
        d = os.path.join(work, "model")
        os.makedirs(d)
        with open(os.path.join(d, "modelx.yaml"), "w") as f:
            f.write(ModelConfig(description="bench").to_yaml())
        with open(os.path.join(d, "blob.bin"), "wb") as f:
            f.write(os.urandom(10 << 20))
        c = Client(mdx.url)
        t0 = time.monotonic()
        c.push("bench/cfg1", "v1", d, quiet=True)
        out = os.path.join(work, "out")
        c.pull("bench/cfg1", "v1", out, quiet=True)
        dt = time.monotonic() - t0
        print(json.dumps({
            "metric": "push+pull GiB/s end-to-end (S3->HBM, digest-verified), 1/2/4/8 MI355X",
            "value": round((20 / 1024) / dt, 4), "unit": "GiB/s", "n_gpus": 0,
            "steps": 1, "warmup": 0, "ms_per_step": round(dt * 1e3, 2),
            "higher_is_better": True, "scaling": "weak", "vs_baseline": None,
            "dtype": "uint8", "data": "synthetic",
            "config": {"model": "config1-10MiB-cpu-plumbing", "global_batch": 1,
                       "seq_len": 0, "parallelism": "cpu"}}))
    finally:
        mdx.stop()
        s3d.stop()
        shutil.rmtree(work, ignore_errors=True)


class _CpuBenchClient:
    """Transfer-client stand-in for MODELX_BENCH_CPU dry-runs: same call
    surface as GpuClient for the bench loop, CPU bytes through the normal
    presigned extension path."""

    def __init__(self, url):
        from modelx_amd.client import Client

        self._client = Client(url)
        self.remote = self._client.remote
        self.last_stats = []

    def push_from_gpu(self, repository, version, tensors, part_bytes=0):
        import tempfile

        d = tempfile.mkdtemp(prefix="bench-cpu-")
        from modelx_amd.config import ModelConfig

        with open(os.path.join(d, "modelx.yaml"), "w") as f:
            f.write(ModelConfig(description="bench").to_yaml())
        for name, t in tensors.items():
            with open(os.path.join(d, name), "wb") as f:
                f.write(t.numpy().tobytes())
        self._client.push(repository, version, d, quiet=True)
        shutil.rmtree(d, ignore_errors=True)

    def pull_to_gpu(self, repository, version, verify=True):
        import tempfile

        d = tempfile.mkdtemp(prefix="bench-cpu-out-")
        self._client.pull(repository, version, d, quiet=True)
        out = {}
        import torch

        for name in os.listdir(d):
            p2 = os.path.join(d, name)
            if os.path.isfile(p2) and name != "modelx.yaml":
                with open(p2, "rb") as f:
                    out[name] = torch.frombuffer(bytearray(f.read()), dtype=torch.uint8)
        shutil.rmtree(d, ignore_errors=True)
        return out


def main():
    args = parse_args()
    if args.cpu_smoke:
        cpu_smoke(args)
        return

    import torch

    # MODELX_BENCH_CPU=1: exercise the FULL multi-rank orchestration (rank/
    # port/store layout, barriers, MAX-reduce, JSON contract) on CPU with
    # gloo and a fake transfer client — the dry-run for the driver's
    # torchrun invocation on 8-GPU nodes (tests/test_bench_dist.py). The
    # normal path is untouched.
    cpu_mode = os.environ.get("MODELX_BENCH_CPU") == "1"

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    distributed = world > 1

    if distributed:
        import torch.distributed as dist

        if not cpu_mode:
            torch.cuda.set_device(local_rank)
        dist.init_process_group(backend="gloo" if cpu_mode else "nccl")  # nccl IS RCCL
    device = local_rank
    if not cpu_mode:
        torch.cuda.set_device(device)

    fanout = world > 1 and args.mode in ("auto", "fanout")  # needs collectives

    # Pin this rank (and the servers it spawns) to its GPU's NUMA node:
    # the engine's recv-into-pinned workers and the loopback store peers
    # otherwise split across sockets (same-box A/B: 23.4-23.5 vs
    # 21.6-22.7 GiB/s). Best effort — single-node boxes and CPU dry-runs
    # skip through the except.
    if not cpu_mode:
        try:
            from modelx_amd import _core

            bdf = _core.hip_pci_bus_id(device).lower()
            with open(f"/sys/bus/pci/devices/{bdf}/numa_node") as f:
                node = int(f.read().strip())
            if node >= 0:
                with open(f"/sys/devices/system/node/node{node}/cpulist") as f:
                    spans = f.read().strip()
                cpus = set()
                for part in spans.split(","):
                    if "-" in part:
                        a, b = part.split("-")
                        cpus.update(range(int(a), int(b) + 1))
                    elif part:
                        cpus.add(int(part))
                if cpus:
                    os.sched_setaffinity(0, cpus)
        except Exception:
            pass

    # --- S3 + registry stack ----------------------------------------------
    # fanout mode: ONE shared modelxd + s3d for the job (rank 0 starts them)
    # — the pull path is one S3 fetch per shard fanned out over xGMI, so the
    # store sees index-size bytes per step regardless of N. independent
    # mode: each rank its own stack (object stores scale horizontally; the
    # measured entity is the per-GPU client data plane).
    from util_servers import MODELXD, S3D, ServerProc, wait_http

    master_port = int(os.environ.get("MASTER_PORT", "29500"))
    if fanout:
        s3_port = master_port + 1371
        mdx_port = master_port + 1372
        store_root = pick_store_root(args) + "-shared"
    else:
        s3_port = master_port + 1371 + 2 * rank
        mdx_port = master_port + 1372 + 2 * rank
        store_root = pick_store_root(args) + f"-r{rank}"
    procs = []
    owns_stack = (not fanout) or rank == 0
    if rank == 0 and not (os.path.exists(S3D) and os.path.exists(MODELXD)):
        import subprocess

        subprocess.run(["make", "servers"], cwd=REPO, check=True)
    if distributed:
        import torch.distributed as dist

        dist.barrier()  # wait for a possible rank-0 build
    if owns_stack:
        shutil.rmtree(store_root, ignore_errors=True)
        os.makedirs(os.path.join(store_root, "modelx"), exist_ok=True)
        procs.append(ServerProc([S3D, "--listen", f"127.0.0.1:{s3_port}", "--root",
                                 store_root, "--access-key", "modelx",
                                 "--secret-key", "modelx123"], s3_port))
        wait_http(s3_port)
        procs.append(ServerProc(
            [MODELXD, "--listen", f"127.0.0.1:{mdx_port}", "--s3-url",
             f"http://127.0.0.1:{s3_port}", "--s3-bucket", "modelx", "--s3-access-key",
             "modelx", "--s3-secret-key", "modelx123", "--enable-redirect"], mdx_port))
        wait_http(mdx_port)
    if distributed:
        import torch.distributed as dist

        dist.barrier()
        if not owns_stack:
            wait_http(mdx_port)

    if cpu_mode:
        g = _CpuBenchClient(f"http://127.0.0.1:{mdx_port}")
    else:
        from modelx_amd.client.gpu import GpuClient

        g = GpuClient(f"http://127.0.0.1:{mdx_port}", device=device, num_slots=args.slots,
                      slot_bytes=args.slot_mib << 20, num_conns=args.conns)

    blob_bytes = int(args.blob_gib * (1 << 30))
    # fanout: the index size is FIXED at blob_gib as N grows — each rank
    # pushes a 1/N shard, the fan-out lands the FULL index on every GPU
    # (per-GPU landed bytes constant = weak scaling; the aggregate delivered
    # bandwidth is what xGMI multiplies)
    if fanout:
        base = blob_bytes // world
        shard_bytes = blob_bytes - base * (world - 1) if rank == world - 1 else base
        repo = "bench/fanout"
    else:
        shard_bytes = blob_bytes
        repo = f"bench/rank{rank}"
    src = torch.empty(shard_bytes, dtype=torch.uint8,
                      device="cpu" if cpu_mode else f"cuda:{device}")

    substeps = []

    def one_step(step_idx: int):
        # fresh random payload → no HEAD-dedup shortcut; new digest every step
        t0 = time.monotonic()
        src.random_(0, 256)
        if not cpu_mode:
            torch.cuda.synchronize(device)
        t1 = time.monotonic()
        if fanout:
            import torch.distributed as dist

            from modelx_amd.client.fanout import fanout_pull_broadcast

            g.push_from_gpu(repo, f"s{step_idx}-r{rank}", {"shard.bin": src},
                            part_bytes=args.part_mib << 20)
            dist.barrier()  # every shard published before the fan-out
            t2 = time.monotonic()
            landed = 0
            for r in range(world):
                out = fanout_pull_broadcast(dist, g, repo, f"s{step_idx}-r{r}",
                                            device, chunk=args.chunk_mib << 20,
                                            src_rank=r, verify=True)
                landed += sum(t.numel() for t in out.values())
                del out
            assert landed == blob_bytes, (landed, blob_bytes)
            t3 = time.monotonic()
            dist.barrier()
            if rank == 0:
                g.remote.delete_index(repo)
            dist.barrier()  # delete done before the next step's push
            t4 = time.monotonic()
        else:
            g.push_from_gpu(repo, f"s{step_idx}", {"blob.bin": src},
                            part_bytes=args.part_mib << 20)
            t2 = time.monotonic()
            g.pull_to_gpu(repo, f"s{step_idx}", verify=True)  # GPU digest verify
            t3 = time.monotonic()
            # drop this step's objects so tmpfs doesn't fill across steps
            g.remote.delete_index(repo)
            t4 = time.monotonic()
        substeps.append((t1 - t0, t2 - t1, t3 - t2, t4 - t3))

    def barrier_sync():
        if not cpu_mode:
            torch.cuda.synchronize(device)
        if distributed:
            import torch.distributed as dist

            dist.barrier()
        if not cpu_mode:
            torch.cuda.synchronize(device)

    def pipelined_steps():
        """Independent mode, steps>1: overlap the push of step k with the
        pull of step k-1 (two threads, one client — the engine is
        reentrant). Loopback measured 1.11x duplex (tools/duplex_probe.py);
        on a real NIC the two directions are independent lanes. The timed
        region still contains exactly K pushes + K pulls + K cleanups;
        per-version cleanup = DELETE manifest + mark-sweep GC (the
        reference's own lifecycle verbs) so tmpfs never holds more than
        two step versions."""
        import threading

        def gen():
            src.random_(0, 256)
            if not cpu_mode:
                torch.cuda.synchronize(device)

        def push(k):
            g.push_from_gpu(repo, f"s{k}", {"blob.bin": src},
                            part_bytes=args.part_mib << 20)

        def clean(k):
            # NEVER concurrent with a push: blobs upload before their
            # manifest (the commit-point convention), so a mark-sweep racing
            # an in-flight push would sweep its not-yet-referenced blobs
            g.remote.delete_manifest(repo, f"s{k}")
            g.remote.garbage_collect(repo)

        gen()
        push(0)
        for k in range(1, args.steps):
            gen()
            th = threading.Thread(target=push, args=(k,))
            th.start()
            g.pull_to_gpu(repo, f"s{k - 1}", verify=True)
            th.join()
            clean(k - 1)
        g.pull_to_gpu(repo, f"s{args.steps - 1}", verify=True)
        clean(args.steps - 1)

    def pipelined_fanout_steps():
        """Fan-out mode, steps>1: each rank's push of step k (plain HTTP, on
        a side thread) overlaps the fan-out broadcasts of step k-1. ALL
        collectives (broadcasts + barriers) stay on the main thread in
        rank-identical order, so the pipelining cannot reorder NCCL ops.
        Cleanup (per-version manifest DELETEs by their owners + one
        mark-sweep) runs after the push joins — never concurrent with an
        in-flight push's not-yet-referenced blobs."""
        import threading

        import torch.distributed as dist

        from modelx_amd.client.fanout import fanout_pull_broadcast

        def gen():
            src.random_(0, 256)
            if not cpu_mode:
                torch.cuda.synchronize(device)

        def push(k):
            g.push_from_gpu(repo, f"s{k}-r{rank}", {"shard.bin": src},
                            part_bytes=args.part_mib << 20)

        def fan(k):
            landed = 0
            for r in range(world):
                out = fanout_pull_broadcast(dist, g, repo, f"s{k}-r{r}", device,
                                            chunk=args.chunk_mib << 20,
                                            src_rank=r, verify=True)
                landed += sum(t.numel() for t in out.values())
                del out
            assert landed == blob_bytes, (landed, blob_bytes)

        gen()
        push(0)
        dist.barrier()  # step-0 shards published
        for k in range(1, args.steps):
            gen()
            th = threading.Thread(target=push, args=(k,))
            th.start()
            fan(k - 1)
            th.join()
            dist.barrier()  # step-k shards published AND k-1 consumed
            g.remote.delete_manifest(repo, f"s{k - 1}-r{rank}")  # own version
            dist.barrier()
            if rank == 0:
                g.remote.garbage_collect(repo)
            dist.barrier()
        fan(args.steps - 1)
        dist.barrier()
        g.remote.delete_manifest(repo, f"s{args.steps - 1}-r{rank}")
        dist.barrier()
        if rank == 0:
            g.remote.delete_index(repo)
        dist.barrier()

    pipe_env = os.environ.get("MODELX_BENCH_PIPELINE", "1") != "0"
    pipeline = not fanout and not distributed and args.steps > 1 and pipe_env
    pipeline_fan = fanout and args.steps > 1 and pipe_env
    for w in range(args.warmup):
        one_step(1000 + w)
    barrier_sync()
    t0 = time.monotonic()
    if pipeline:
        pipelined_steps()
    elif pipeline_fan:
        pipelined_fanout_steps()
    else:
        for k in range(args.steps):
            one_step(k)
    barrier_sync()
    elapsed = time.monotonic() - t0

    # MAX over ranks
    if distributed:
        import torch.distributed as dist

        t = torch.tensor([elapsed], dtype=torch.float64,
                         device="cpu" if cpu_mode else f"cuda:{device}")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if os.environ.get("MODELX_BENCH_DEBUG") and rank == 0:
        for i, (tg, tp, tl, td) in enumerate(substeps):
            print(f"# step {i}: gen={tg * 1e3:.0f}ms push={tp * 1e3:.0f}ms "
                  f"pull={tl * 1e3:.0f}ms delete={td * 1e3:.0f}ms", file=sys.stderr)
        import collections

        agg = collections.defaultdict(lambda: [0.0, 0])
        for s in g.last_stats:
            ph = s.get("phase", "?")
            agg[ph][0] += s.get("seconds", 0.0)
            agg[ph][1] += s.get("bytes", 0)
        for ph, (secs, byts) in sorted(agg.items()):
            rate = byts / secs / (1 << 30) if secs else 0
            print(f"# stage {ph}: {secs:.2f}s {byts / (1 << 30):.2f} GiB "
                  f"{rate:.2f} GiB/s", file=sys.stderr)

    for p in procs:
        p.stop()
    shutil.rmtree(store_root, ignore_errors=True)
    if rank == 0:
        if fanout:
            # per step: HBM→S3 push of the index (blob_gib aggregate across
            # ranks) + S3/xGMI→HBM landing of the FULL index on every GPU
            # (world × blob_gib, each digest-verified at its destination)
            moved_gib = (1.0 + world) * args.blob_gib * args.steps
            model = f"config4-{args.blob_gib:g}GiB-index-sharded{world}"
            par = (f"fanout{world}-rccl-xgmi-broadcast"
                   + ("-pipelined" if pipeline_fan else ""))
        else:
            moved_gib = 2.0 * args.blob_gib * args.steps * world  # push+pull, all ranks
            model = f"synthetic-{args.blob_gib:g}GiB-blob"
            par = f"dp{world}-presigned-s3" + ("-pipelined" if pipeline else "")
        print(json.dumps({
            "metric": "push+pull GiB/s end-to-end (S3->HBM, digest-verified), 1/2/4/8 MI355X",
            "value": round(moved_gib / elapsed, 3),
            "unit": "GiB/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 1),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no numbers
            "dtype": "uint8",
            "data": "synthetic",
            "config": {"model": model,
                       "global_batch": world, "seq_len": 0,
                       "parallelism": par}}), flush=True)
    if distributed:
        import torch.distributed as dist

        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
