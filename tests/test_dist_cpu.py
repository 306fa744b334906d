"""Multi-process (gloo, world_size=2) CPU tests for the N>1 path.

The data path has NO collective for configs 1-3,5 (partitions are independent —
SURVEY.md §8e); what the bench does across ranks is (a) per-rank shard
generation, (b) MAX-reduce of wall time, (c) SUM-reduce of row counts, and for
config 4 (d) SUM-reduce of the aggregate partial (RCCL over xGMI on the GPU
box; gloo here). These tests run the same reduce logic on CPU."""
import multiprocessing as mp
import os

import pytest


def _worker(rank, world, q):
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29517",
        "RANK": str(rank), "WORLD_SIZE": str(world),
    })
    import torch
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        # per-rank shard (what bench.py's make_input does per rank)
        from tests.zillow_data import make_zillow_csv_bytes
        data, rows = make_zillow_csv_bytes(200, seed=42 + rank)
        n_rows = torch.tensor([float(len(rows))])
        dist.all_reduce(n_rows, op=dist.ReduceOp.SUM)

        wall = torch.tensor([0.5 + 0.25 * rank])
        dist.all_reduce(wall, op=dist.ReduceOp.MAX)

        # config-4 aggregate partial combine (ncclReduce analog)
        partial = torch.tensor([float(sum(range(rank * 10, rank * 10 + 10)))])
        dist.all_reduce(partial, op=dist.ReduceOp.SUM)

        q.put((rank, float(n_rows.item()), float(wall.item()),
               float(partial.item()), len(data) > 0))
    finally:
        dist.destroy_process_group()


def test_world2_reduce_logic():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(2)]
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    for rank, n_rows, wall, partial, ok in results:
        assert n_rows == 400.0          # SUM of both ranks' shards
        assert wall == 0.75             # MAX over ranks
        assert partial == sum(range(20))  # combined aggregate
        assert ok


def test_rank_shards_differ():
    from tests.zillow_data import make_zillow_csv_bytes
    d0, _ = make_zillow_csv_bytes(100, seed=42)
    d1, _ = make_zillow_csv_bytes(100, seed=43)
    assert d0 != d1  # weak scaling: each rank gets its own synthetic shard
