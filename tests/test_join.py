"""Hash join (SURVEY.md §8f-3 MVP): right side = unique-key build table
embedded in the generated stage source; probe on device. Output layout per
logical/JoinOperator.cc:164: | left cols except key | key | right cols except
key |; leftJoin nulls the right columns; the key keeps the left name."""
import pytest

import tuplex_amd
from oracle import pyoracle
from tests.pipelines import apply_ops

RIGHT = [(1, "one", 1.5), (2, "two", 2.5), (3, "three", None)]
RCOLS = ["k", "label", "w"]


def _jop(how="inner", pre=("", ""), suf=("", "")):
    return ("join", RIGHT, RCOLS, "key", "k", how, pre[0], suf[0], pre[1],
            suf[1])


def test_join_plan_and_compile():
    from tuplex_amd import codegen, plan
    from tuplex_amd import ttypes as T
    sp = plan.build_stage([T.I64, T.STR], ["key", "val"], [_jop()])
    assert sp.compilable, sp.why_not_compilable
    assert sp.output_columns == ["val", "key", "label", "w"]
    assert sp.output_types == [T.STR, T.I64, T.STR, ("opt", T.F64)]
    src, desc = codegen.generate_stage(sp, source="mem", sink="mem")
    from tuplex_amd.engine import GpuLib
    glib = GpuLib.get()
    st = glib.lib.tpx_stage_compile(src.encode(), desc.encode(), b"", 1)
    assert st, glib.err()
    # left join: right columns become nullable
    spl = plan.build_stage([T.I64, T.STR], ["key", "val"], [_jop("left")])
    assert spl.output_types == [T.STR, T.I64, ("opt", T.STR), ("opt", T.F64)]


def test_join_string_key_compile():
    from tuplex_amd import codegen, plan
    from tuplex_amd import ttypes as T
    right = [("BOS", "Boston"), ("NYC", "New York")]
    op = ("join", right, ["code", "city"], "ap", "code", "inner",
          "", "", "", "")
    sp = plan.build_stage([T.STR, T.I64], ["ap", "n"], [op])
    assert sp.compilable, sp.why_not_compilable
    src, desc = codegen.generate_stage(sp, source="mem", sink="mem")
    assert "tpx_jhash_bytes" in src


def test_join_duplicate_keys_fallback_semantics():
    """Duplicate build keys: 1:N expansion — interpreter path this round
    (plan falls back); every match emits a row in BUILD order and each joined
    row runs the remaining ops independently."""
    from tuplex_amd import plan
    from tuplex_amd import ttypes as T
    dup = [(1, "a"), (1, "b"), (2, "c")]
    jop = ("join", dup, ["k", "x"], "key", "k", "inner", "", "", "", "")
    # TERMINAL dup join: GPU-capable (write kernel loops the bucket)
    sp = plan.build_stage([T.I64, T.STR], ["key", "val"], [jop])
    assert sp.compilable and getattr(sp.ops[-1], "join_dup", False)
    # mid-pipeline dup join: interpreter path

    def postop(x):
        return (x[0], x[2])

    spm = plan.build_stage([T.I64, T.STR], ["key", "val"],
                           [jop, ("map", postop)])
    assert not spm.compilable
    assert "duplicate" in spm.why_not_compilable

    rows = [(1, "L"), (2, "M"), (9, "N")]
    r = pyoracle.run_pipeline(rows, [jop], columns=["key", "val"])
    assert r["output"] == [("L", 1, "a"), ("L", 1, "b"), ("M", 2, "c")]

    def wide(x):
        return (x["val"], x["x"].upper())

    r2 = pyoracle.run_pipeline(rows, [jop, ("map", wide)],
                               columns=["key", "val"])
    assert r2["output"] == [("L", "A"), ("L", "B"), ("M", "C")]

    # per-duplicate exceptions: a post-join op failing on ONE duplicate
    # leaves the other outputs intact
    def picky(x):
        if x["x"] == "b":
            raise ValueError("b")
        return (x["val"], x["x"])

    r3 = pyoracle.run_pipeline(rows, [jop, ("map", picky)],
                               columns=["key", "val"])
    assert r3["output"] == [("L", "a"), ("M", "c")]
    assert r3["exception_counts"] == {"ValueError": 1}


def test_join_oracle_semantics():
    rows = [(1, "a"), (2, "b"), (5, "c")]
    r = pyoracle.run_pipeline(rows, [_jop()], columns=["key", "val"])
    assert r["output"] == [("a", 1, "one", 1.5), ("b", 2, "two", 2.5)]
    r = pyoracle.run_pipeline(rows, [_jop("left")], columns=["key", "val"])
    assert r["output"][2] == ("c", 5, None, None)
    # prefixes/suffixes rename like the reference
    r = pyoracle.run_pipeline(rows, [_jop(pre=("l_", "r_"))],
                              columns=["key", "val"])
    assert r["output"][0] == ("a", 1, "one", 1.5)


def test_join_pushdown_lineage():
    from tuplex_amd import plan
    from tuplex_amd import ttypes as T
    ops = [_jop(), ("selectColumns", ["label"])]
    sp = plan.build_stage([T.I64, T.STR, T.STR], ["key", "val", "junk"], ops)
    assert sp.compilable
    # only the probe key is read from the source
    assert sorted(sp.used_source_cols) == [0]


@pytest.mark.gpu
def test_join_gpu_mem_parity():
    import random
    rng = random.Random(11)
    rows = [(rng.randint(0, 6), "v%d" % i) for i in range(20000)]
    ctx = tuplex_amd.Context()
    ds = apply_ops(ctx.parallelize(rows, columns=["key", "val"]), [_jop()])
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle.run_pipeline(rows, [_jop()], columns=["key", "val"])
    assert got == ref["output"]
    assert len(got) < len(rows)  # keys 0,4,5,6 miss

    dsl = apply_ops(ctx.parallelize(rows, columns=["key", "val"]),
                    [_jop("left")])
    gotl = dsl.collect()
    assert dsl._last_outcome.mode == "gpu"
    refl = pyoracle.run_pipeline(rows, [_jop("left")], columns=["key", "val"])
    assert gotl == refl["output"]
    assert len(gotl) == len(rows)


@pytest.mark.gpu
def test_join_gpu_csv_string_key(tmp_path):
    """flights-style: csv fact rows joined to a string-keyed dimension."""
    import os
    codes = ["BOS", "JFK", "LAX", "ORD", "SEA", "SFO"]
    dim = [(c, "City-%s" % c, i * 10) for i, c in enumerate(codes)]
    lines = [b"ap,delay"]
    import random
    rng = random.Random(3)
    for i in range(30000):
        ap = rng.choice(codes + ["XXX"])  # XXX misses the dimension
        lines.append(b"%s,%d" % (ap.encode(), rng.randint(-10, 500)))
    data = b"\n".join(lines) + b"\n"
    p = os.path.join(str(tmp_path), "fl.csv")
    with open(p, "wb") as f:
        f.write(data)

    jop = ("join", dim, ["code", "city", "rank"], "ap", "code", "inner",
           "", "", "", "")

    def late(x):
        return x["delay"] > 60

    ops = [("filter", late), jop, ("selectColumns", ["ap", "city", "delay"])]
    ctx = tuplex_amd.Context()
    ds = apply_ops(ctx.csv(p), ops)
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    from oracle import pyoracle_csv
    ref = pyoracle_csv.run_csv_pipeline(data, ops)
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]
    assert all(c.startswith("City-") for _, c, _d in got)


# module-level so tests/pipelines.precompile_tests can prebuild the exact
# stage (this 110-col + 2-join stage costs ~13 min of hipRTC on a cold box)
FL_CARRIERS = [("aa", "American"), ("dl", "Delta"), ("ua", "United")]
FL_CITIES = [("City-%d Airport" % i, i * 3) for i in range(1, 150)]
FL_J1 = ("join", FL_CARRIERS, ["code_c", "carrier_name"], "code", "code_c",
         "inner", "", "", "", "")
FL_J2 = ("join", FL_CITIES, ["city_a", "tz"], "c3", "city_a",
         "left", "", "", "", "")


@pytest.mark.gpu
def test_join_gpu_flights_pipeline(tmp_path):
    """The flights benchmark shape (runtuplex.py:210+): parse the wide CSV,
    derive columns, then join TWO string-keyed dimension tables (airports,
    carriers) — the full pipeline the reference's HashJoinStage serves."""
    import os
    from tests import extra_data as X
    from oracle import pyoracle_csv

    data = X.make_flights_csv(4000, seed=9, bad_frac=0.01)
    p = os.path.join(str(tmp_path), "fl.csv")
    with open(p, "wb") as f:
        f.write(data)

    ops = X.flights_ops() + [FL_J1, FL_J2]

    ctx = tuplex_amd.Context()
    ds = apply_ops(ctx.csv(p), ops)
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle_csv.run_csv_pipeline(data, ops)
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]
    assert len(got) > 0
    assert any(r[-1] is None for r in got) or all(r[-1] for r in got)


def test_join_duplicate_keys_product_fallback():
    """End-to-end product path: dup-key join -> whole-stage interpreter
    fallback (CPU-runnable), output matches the oracle incl. 1:N expansion."""
    dup = [(1, "a"), (1, "b"), (2, "c")]
    rows = [(1, "L"), (2, "M"), (9, "N")]
    def tail(x):
        return (x["val"], x["key"], x["x"])

    ctx = tuplex_amd.Context()
    rds = ctx.parallelize([r for r in dup], columns=["k", "x"])
    ds = ctx.parallelize(rows, columns=["key", "val"]).join(
        rds, "key", "k").map(tail)  # post-join op -> interpreter path
    got = ds.collect()
    assert ds._last_outcome.mode == "fallback"
    assert "duplicate" in ds._last_outcome.fallback_reason
    assert got == [("L", 1, "a"), ("L", 1, "b"), ("M", 2, "c")]


@pytest.mark.gpu
def test_join_gpu_duplicate_keys():
    """GPU 1:N: terminal dup-key join — keep01 carries bucket counts, the
    write kernel loops the bucket; output order = input row order x build
    order, exactly the oracle's fork order."""
    import random
    dup = [(1, "a", 10), (1, "b", None), (2, "c", 30), (7, "z", 5),
           (1, "d", 40)]
    rng = random.Random(5)
    rows = []
    for i in range(30000):
        v = "v%d" % i
        if i % 411 == 0:
            v = "vé%d" % i  # non-ASCII -> divert -> multi-row replay fork
        rows.append((rng.randint(0, 8), v))
    jop = ("join", dup, ["k", "x", "w"], "key", "k", "inner",
           "", "", "", "")
    ctx = tuplex_amd.Context()
    ds = apply_ops(ctx.parallelize(rows, columns=["key", "val"]), [jop])
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle.run_pipeline(rows, [jop], columns=["key", "val"])
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]
    per_key1 = sum(1 for r in rows if r[0] == 1)
    assert sum(1 for g in got if g[1] == 1) == 3 * per_key1

    jopl = ("join", dup, ["k", "x", "w"], "key", "k", "left",
            "", "", "", "")
    dsl = apply_ops(ctx.parallelize(rows, columns=["key", "val"]), [jopl])
    gotl = dsl.collect()
    assert dsl._last_outcome.mode == "gpu"
    refl = pyoracle.run_pipeline(rows, [jopl], columns=["key", "val"])
    assert gotl == refl["output"]


@pytest.mark.gpu
def test_join_gpu_duplicate_keys_csv(tmp_path):
    """Dup-key join fed from the CSV source (string keys)."""
    import os
    import random
    dim = [("aa", "A1"), ("aa", "A2"), ("bb", "B1"), ("cc", "C1")]
    rng = random.Random(6)
    lines = [b"code,n"]
    for i in range(20000):
        lines.append(b"%s,%d" % (rng.choice([b"aa", b"bb", b"cc", b"xx"]), i))
    data = b"\n".join(lines) + b"\n"
    p = os.path.join(str(tmp_path), "d.csv")
    with open(p, "wb") as f:
        f.write(data)
    jop = ("join", dim, ["k", "label"], "code", "k", "inner", "", "", "", "")
    ctx = tuplex_amd.Context()
    ds = apply_ops(ctx.csv(p, header=True), [jop])
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    from oracle import pyoracle_csv
    ref = pyoracle_csv.run_csv_pipeline(data, [jop], header=True)
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]


@pytest.mark.gpu
def test_join_gpu_duplicate_keys_tocsv(tmp_path):
    """Dup-key join straight to the csv sink (file->file, 1:N rows in the
    device-formatted output)."""
    import os
    import random
    dim = [("aa", "A1", 1), ("aa", "A2", 2), ("bb", "B1", 3)]
    rng = random.Random(8)
    lines = [b"code,n"]
    for i in range(15000):
        lines.append(b"%s,%d" % (rng.choice([b"aa", b"bb", b"xx"]), i))
    data = b"\n".join(lines) + b"\n"
    p = os.path.join(str(tmp_path), "in.csv")
    with open(p, "wb") as f:
        f.write(data)
    outp = os.path.join(str(tmp_path), "out.csv")
    jop = ("join", dim, ["k", "label", "r"], "code", "k", "inner",
           "", "", "", "")
    ctx = tuplex_amd.Context()
    ds = apply_ops(ctx.csv(p, header=True), [jop])
    ds.tocsv(outp)
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    from oracle import pyoracle_csv
    ref = pyoracle_csv.run_csv_pipeline(data, [jop], header=True, sink="csv")
    with open(outp, "rb") as f:
        got = f.read()
    assert got == ref["csv_text"]


def _post_join_map(x):
    return (x["key"], x["x"],
            (x["w"] if x["w"] is not None else 0) + 1)


def _post_join_keep(x):
    return x[2] != 31


@pytest.mark.gpu
def test_join_gpu_duplicate_keys_mid_pipeline():
    """Mid-pipeline dup-key join (HashJoinStage.cc:473 case): the engine
    splits the pipeline at the join (find_dup_join_split) so BOTH halves run
    compiled on the GPU — stage 1 ends in the terminal 1:N expansion, stage 2
    maps/filters the expanded rows from the materialized partitions. Output
    identical to the oracle's forked replay."""
    import random
    dup = [(1, "a", 10), (1, "b", None), (2, "c", 30), (7, "z", 5),
           (1, "d", 40)]
    rng = random.Random(5)
    rows = [(rng.randint(0, 8), "v%d" % i) for i in range(30000)]
    jop = ("join", dup, ["k", "x", "w"], "key", "k", "inner",
           "", "", "", "")
    ops = [jop, ("map", _post_join_map), ("filter", _post_join_keep)]
    ctx = tuplex_amd.Context()
    ds = apply_ops(ctx.parallelize(rows, columns=["key", "val"]), ops)
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle.run_pipeline(rows, ops, columns=["key", "val"])
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]
    assert len(got) > 10000
