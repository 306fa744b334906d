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


# ---- full-pipeline world-2 engine test ----------------------
#
# Runs the ACTUAL engine (Context.csv -> map/filter -> collect / tocsv /
# aggregate) across 2 ranks: csvio shards the input rows across ranks and the
# final merge is the collective path (object gather + the tensor all_reduce
# scalar-aggregate combine — same code that runs RCCL on the GPU box, gloo
# here). Output must be byte/value-identical to a 1-rank run.

def _sortfirst(x):
    # sorted() is outside the compiled vocabulary -> interpreter path on CPU
    return (sorted([x["zipcode"], x["zipcode"]])[0], x["price"])


def _dist_pipeline(ctx_conf, inp, outdir):
    import tuplex_amd
    ctx = tuplex_amd.Context(ctx_conf)
    ds = (ctx.csv(inp)
          .map(_sortfirst)
          .filter(lambda x: x[1] % 3 != 0))
    rows = ds.collect()
    out_csv = os.path.join(outdir, "out.csv")
    (tuplex_amd.Context(ctx_conf).csv(inp)
     .map(_sortfirst)
     .filter(lambda x: x[1] % 3 != 0)
     .tocsv(out_csv))
    agg = (tuplex_amd.Context(ctx_conf).csv(inp)
           .map(_sortfirst)
           .aggregate(lambda a, b: a + b, lambda a, x: a + x[1], 0)
           .collect())
    with open(out_csv, "rb") as f:
        return rows, f.read(), agg


def _pipe_worker(rank, world, tmpdir, q):
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29531",
        "RANK": str(rank), "WORLD_SIZE": str(world),
    })
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        # shared output dir: rank 0 writes the merged csv, the sink barrier
        # in csvio guarantees it exists before rank 1 reads it
        res = _dist_pipeline({}, os.path.join(tmpdir, "in.csv"),
                             os.path.join(tmpdir, "shared"))
        q.put((rank, res))
    finally:
        dist.destroy_process_group()


def test_world2_full_pipeline_matches_single_rank(tmp_path):
    tmpdir = str(tmp_path)
    inp = os.path.join(tmpdir, "in.csv")
    lines = ["zipcode,price\n"]
    for i in range(500):
        lines.append("%05d,%d\n" % (10000 + i * 7 % 900, (i * 37) % 10000))
    with open(inp, "w") as f:
        f.write("".join(lines))

    # 1-rank reference (no dist initialized)
    ref_rows, ref_csv, ref_agg = _dist_pipeline(
        {}, inp, os.path.join(tmpdir, "ref"))

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_pipe_worker, args=(r, 2, tmpdir, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, res = q.get(timeout=300)
        results[rank] = res
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0

    for rank in (0, 1):
        rows, _, agg = results[rank]
        assert rows == ref_rows, "rank %d collect() diverged" % rank
        assert agg == ref_agg, "rank %d aggregate diverged" % rank
    # tocsv: rank 0 wrote the file; bytes identical to the 1-rank run
    assert results[0][1] == ref_csv
