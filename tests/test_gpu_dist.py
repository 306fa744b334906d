"""GPU world-2 engine test: two ranks SHARING device 0 (gloo merge backend),
running the compiled GPU pipeline with the engine's own rank sharding
(csvio.run_csv dist path). Asserts collect()/tocsv/aggregate outputs are
identical to a 1-rank GPU run — the product multi-GPU orchestration
(LocalBackend.cc:491 task fan-out analog), minus only the physical second
device (gpurun boxes have one GPU; the chunk->rank assignment and collective
merge are device-count independent)."""
import multiprocessing as mp
import os

import pytest

pytestmark = pytest.mark.gpu

N_ROWS = 40000


def _make_input(path):
    lines = ["a,b,name\n"]
    for i in range(N_ROWS):
        # a few quoted cells and dirty rows so replay + quote parity matter
        if i % 997 == 0:
            lines.append('%d,notanint,"x,y%d"\n' % (i, i))
        elif i % 31 == 0:
            lines.append('%d,%d,"q""%d"\n' % (i, i * 3, i))
        else:
            lines.append("%d,%d,n%d\n" % (i, i * 3, i))
    with open(path, "w") as f:
        f.write("".join(lines))


def _du_map(x):
    return (x["a"] * 2, x["b"] + 1, x["name"])


def _du_keep(x):
    return x[0] % 5 != 0


def _du_ab(x):
    return (x["a"], x["b"])


def _comb(a, b):
    return a + b


def _sumb(a, x):
    return a + x[1]


def _pipeline(inp, out_csv):
    import tuplex_amd
    conf = {"tuplex.inputSplitSize": "64KB"}  # force many chunks
    ds = tuplex_amd.Context(conf).csv(inp).map(_du_map).filter(_du_keep)
    rows = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    tuplex_amd.Context(conf).csv(inp).map(_du_map).filter(_du_keep) \
        .tocsv(out_csv)
    agg_ds = tuplex_amd.Context(conf).csv(inp).map(_du_ab) \
        .aggregate(_comb, _sumb, 0)
    agg = agg_ds.collect()
    with open(out_csv, "rb") as f:
        return rows, f.read(), agg


def _worker(rank, world, tmpdir, q):
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29533",
        "RANK": str(rank), "WORLD_SIZE": str(world),
    })
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        res = _pipeline(os.path.join(tmpdir, "in.csv"),
                        os.path.join(tmpdir, "shared", "out.csv"))
        q.put((rank, res))
    finally:
        dist.destroy_process_group()


def test_gpu_world2_engine_matches_single_rank(tmp_path):
    tmpdir = str(tmp_path)
    inp = os.path.join(tmpdir, "in.csv")
    _make_input(inp)

    ref_rows, ref_csv, ref_agg = _pipeline(
        inp, os.path.join(tmpdir, "ref", "out.csv"))

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, tmpdir, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, res = q.get(timeout=600)
        results[rank] = res
    for p in procs:
        p.join(timeout=600)
        assert p.exitcode == 0

    for rank in (0, 1):
        rows, _, agg = results[rank]
        assert rows == ref_rows, "rank %d collect() diverged" % rank
        assert agg == ref_agg, "rank %d aggregate diverged" % rank
    assert results[0][1] == ref_csv
