"""Multi-process (gloo, world_size=2) CPU coverage of the distributed
path bench.py uses at N>1: process-group init, the max-over-ranks timing
reduction, and disjoint-covering work-unit sharding (SURVEY.md §8(e) —
no data-path collective exists by design, so this IS the whole
distributed surface)."""

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from bigstitcher_spark_amd import host


def _worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    # max-over-ranks reduction as bench.py does it
    t = torch.tensor([1.0 + rank], dtype=torch.float64)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    my_units = host.shard(100, world, rank)
    gathered = [None] * world
    dist.all_gather_object(gathered, my_units.tolist())
    dist.barrier()
    if rank == 0:
        q.put((float(t.item()), gathered))
    dist.destroy_process_group()


def test_gloo_world2_reduction_and_shard():
    ctxm = mp.get_context("spawn")
    q = ctxm.Queue()
    port = 29511
    ps = [
        ctxm.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)
    ]
    for p in ps:
        p.start()
    maxval, gathered = q.get(timeout=120)
    for p in ps:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert maxval == 2.0  # MAX over ranks
    allu = sorted(gathered[0] + gathered[1])
    assert allu == list(range(100))  # covering
    assert not (set(gathered[0]) & set(gathered[1]))  # disjoint
    # both ranks get a nontrivial share
    assert 20 < len(gathered[0]) < 80


def test_shard_deterministic():
    a = host.shard(1000, 8, 3)
    b = host.shard(1000, 8, 3)
    assert np.array_equal(a, b)
    total = sum(len(host.shard(1000, 8, r)) for r in range(8))
    assert total == 1000
