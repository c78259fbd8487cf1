"""Fan-out choreography tests — CPU, gloo backend, world_size=2
("distributed" tests are single-node multi-rank; the RCCL/xGMI
path shares this exact choreography with backend swapped to nccl)."""
import os

import pytest
import torch
import torch.multiprocessing as mp

from modelx_amd.client.fanout import (ShardPlan, broadcast_blob_pipelined,
                                      chunk_schedule)
from modelx_amd.wire import types


class TestShardPlan:
    def test_balanced_assignment(self):
        descs = [types.Descriptor(name=f"b{i}", size=s)
                 for i, s in enumerate([100, 90, 50, 40, 10, 10])]
        plan = ShardPlan.build(descs, 2)
        assert set(plan.owners.values()) <= {0, 1}
        assert abs(plan.rank_bytes[0] - plan.rank_bytes[1]) <= 50
        assert sum(plan.rank_bytes) == 300

    def test_more_ranks_than_blobs(self):
        descs = [types.Descriptor(name="only", size=5)]
        plan = ShardPlan.build(descs, 8)
        assert plan.owners["only"] in range(8)
        assert sum(plan.rank_bytes) == 5

    def test_deterministic_across_ranks(self):
        descs = [types.Descriptor(name=f"b{i}", size=i * 7 % 13 + 1) for i in range(20)]
        assert ShardPlan.build(descs, 4).owners == ShardPlan.build(list(descs), 4).owners


class TestChunkSchedule:
    def test_exact_cover(self):
        sched = chunk_schedule(1000, 300)
        assert sched == [(0, 300), (300, 300), (600, 300), (900, 100)]

    def test_zero(self):
        assert chunk_schedule(0, 300) == [(0, 0)]


def _bcast_worker(rank, world, port, results):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    size = 1 << 20
    tensor = torch.zeros(size, dtype=torch.uint8)
    fetched = []

    fetch = None
    if rank == 0:
        src_data = torch.arange(size, dtype=torch.int64).remainder(251).to(torch.uint8)

        def fetch(off, ln):
            tensor[off : off + ln] = src_data[off : off + ln]
            fetched.append((off, ln))

    broadcast_blob_pipelined(dist, tensor, size, 0, fetch, chunk=200_000)
    expect = torch.arange(size, dtype=torch.int64).remainder(251).to(torch.uint8)
    results[rank] = bool(torch.equal(tensor, expect)) and (
        rank != 0 or len(fetched) == len(chunk_schedule(size, 200_000)))
    dist.destroy_process_group()


def test_pipelined_broadcast_gloo_two_ranks():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    with mp.Manager() as mgr:
        results = mgr.dict()
        mp.spawn(_bcast_worker, args=(2, port, results), nprocs=2, join=True)
        assert results[0] and results[1]


def _shard_worker(rank, world, port, results):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    descs = [types.Descriptor(name=f"shard{i}", size=1000 + i) for i in range(6)]
    plan = ShardPlan.build(descs, world)
    # owners pull "their" blobs (simulated), then replicate via broadcast
    tensors = {}
    for d in descs:
        t = torch.zeros(d.size, dtype=torch.uint8)
        if plan.owners[d.name] == rank:
            t.fill_(ord(d.name[-1]) % 251)
        tensors[d.name] = t
    for d in descs:
        dist.broadcast(tensors[d.name], src=plan.owners[d.name])
    ok = all(bool((tensors[d.name] == ord(d.name[-1]) % 251).all()) for d in descs)
    results[rank] = ok
    dist.destroy_process_group()


def test_sharded_replicate_gloo_two_ranks():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    with mp.Manager() as mgr:
        results = mgr.dict()
        mp.spawn(_shard_worker, args=(2, port, results), nprocs=2, join=True)
        assert results[0] and results[1]
