"""torchrun helper: 2 RCCL ranks on ONE MI355X (RCCL permits multiple ranks
per device) proving the fan-out collective choreography — stream ordering,
pipelined ncclBroadcast, digest-after-broadcast — under a real RCCL
communicator without needing an 8-GPU node. Run by
test_gpu_fanout.py::test_fanout_broadcast_rccl_world2 as:

    torchrun --nproc-per-node 2 --master-addr 127.0.0.1 tests/rccl2_helper.py
"""
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "tests"))


def main():
    import torch
    import torch.distributed as dist

    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    # RCCL (2.26) rejects two ranks on one device ("Duplicate GPU detected",
    # init.cc:1108 — verified on hardware), so: real RCCL when the box has a
    # GPU per rank (the driver's 8-GPU node), gloo transport for the CUDA
    # tensors on a 1-GPU box — two processes, device tensors, the same
    # fan-out choreography; only the wire differs.
    ngpu = torch.cuda.device_count()
    device = rank % ngpu
    backend = "nccl" if ngpu >= world else "gloo"
    torch.cuda.set_device(device)
    dist.init_process_group(backend)

    # collective sanity: all-reduce across ranks on device tensors
    t = torch.full((1 << 20,), float(rank + 1), device=f"cuda:{device}")
    dist.all_reduce(t)
    assert t[0].item() == sum(r + 1 for r in range(world)), t[0].item()

    from util_servers import start_modelxd_s3, start_s3d, wait_http

    from modelx_amd.client.fanout import fanout_pull_broadcast, fanout_pull_sharded
    from modelx_amd.client.gpu import GpuClient

    master_port = int(os.environ.get("MASTER_PORT", "29500"))
    s3_port, mdx_port = master_port + 211, master_port + 212
    procs = []
    if rank == 0:
        s3d = start_s3d(os.path.join(os.environ.get("TMPDIR", "/tmp"), "rccl2-s3"),
                        port=s3_port)
        mdx = start_modelxd_s3(s3d.url, redirect=True, port=mdx_port)
        procs = [mdx, s3d]
    dist.barrier()
    if rank != 0:
        wait_http(mdx_port)

    try:
        g = GpuClient(f"http://127.0.0.1:{mdx_port}", device=device,
                      num_slots=4, slot_bytes=8 << 20)
        # each rank pushes its own blob; the other rank receives it purely
        # over the RCCL broadcast and digest-verifies on device
        src = torch.randint(0, 256, ((24 << 20) + 137,), dtype=torch.uint8,
                            device=f"cuda:{device}")
        g.push_from_gpu("rccl2/m", f"v-r{rank}", {"w.bin": src})
        dist.barrier()
        for src_rank in range(world):
            out = fanout_pull_broadcast(dist, g, "rccl2/m", f"v-r{src_rank}",
                                        device=device, chunk=4 << 20,
                                        src_rank=src_rank)
            if rank == src_rank:
                assert torch.equal(out["w.bin"], src), "own blob mismatch"
        # sharded pull with replicate: ShardPlan owners fetch, broadcasts
        # replicate, every rank verifies
        tensors = {f"s{i}.bin": torch.randint(0, 256, (2 << 20,), dtype=torch.uint8,
                                              device=f"cuda:{device}")
                   for i in range(4)}
        if rank == 0:
            g.push_from_gpu("rccl2/shard", "v1", tensors)
        dist.barrier()
        out = fanout_pull_sharded(dist, g, "rccl2/shard", "v1", device=device,
                                  replicate=True)
        assert len(out) == 4
        dist.barrier()
        if rank == 0:
            print(f"RCCL2 OK backend={backend} ngpu={ngpu}", flush=True)
    finally:
        dist.destroy_process_group()
        for p in procs:
            p.stop()


if __name__ == "__main__":
    try:
        main()
    except BaseException:
        import traceback

        traceback.print_exc()
        sys.stderr.flush()
        raise
