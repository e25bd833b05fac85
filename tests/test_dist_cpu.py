"""Multi-process (gloo, world_size=2) CPU tests of the distributed path:
bucket sharding is a partition, and the bench's barrier + max-over-ranks
timing reduction works — the same collective calls bench.py issues over RCCL
on the GPU box."""

import os

import pytest
import torch
import torch.multiprocessing as mp

from paimon_amd.dist import shard_buckets, aggregate_rows_per_sec


def test_shard_buckets_partition():
    for world in (1, 2, 4, 8):
        for n in (1, 7, 8, 64):
            seen = []
            for r in range(world):
                seen += shard_buckets(n, world, r)
            assert sorted(seen) == list(range(n))
    assert shard_buckets(64, 8, 3) == list(range(3, 64, 8))


def test_aggregate():
    assert aggregate_rows_per_sec([100, 200], 2.0) == 150.0


def _worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    # the bench's collective sequence: barrier, work, barrier, max-reduce
    dist.barrier()
    elapsed = torch.tensor([1.0 + rank], dtype=torch.float64)
    dist.all_reduce(elapsed, op=dist.ReduceOp.MAX)
    assert elapsed.item() == world  # max over ranks
    rows = torch.tensor([1000.0 * (rank + 1)])
    dist.all_reduce(rows, op=dist.ReduceOp.SUM)
    assert rows.item() == sum(1000.0 * (r + 1) for r in range(world))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_gloo_world2_barrier_and_reduce():
    world = 2
    port = 29811
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, world, port))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(100)
        assert p.exitcode == 0
