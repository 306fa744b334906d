"""CPU tests for host-side aggregate finalization on the interpreter fallback
paths (ADVICE round-1 findings): aggregateByKey/unique must be applied after
the ordered merge in every fallback path, with the agg UDF seeing the
PIPELINE's output column names — mirroring the GPU merge the engine does for
reducible aggregates (LocalBackend.cc:1180-1207; :2219 createFinalHashmap)."""
import os

import pytest

import tuplex_amd
from tuplex_amd.engine import finalize_merged, output_columns_of


def _noncompilable(x):
    # sorted() is outside the compiled vocabulary -> whole-stage fallback
    return sorted([x["k"], x["k"]])[0]


def _write_csv(tmp_path, text):
    p = os.path.join(str(tmp_path), "in.csv")
    with open(p, "w") as f:
        f.write(text)
    return p


def test_parallelize_fallback_aggregate_by_key():
    """run_collect fallback path: aggregateByKey folds per key host-side."""
    ctx = tuplex_amd.Context()
    rows = [(1, 10), (2, 5), (1, 7), (2, 1)]
    ds = ctx.parallelize(rows, columns=["k", "v"]) \
        .withColumn("k2", lambda x: sorted([x["k"]])[0]) \
        .selectColumns(["k2", "v"]) \
        .aggregateByKey(lambda a, b: a + b,
                        lambda a, x: a + x["v"], 0, ["k2"])
    got = ds.collect()
    assert ds._last_outcome.mode == "fallback"
    assert sorted(got) == [(1, 17), (2, 6)]


def test_parallelize_fallback_unique():
    ctx = tuplex_amd.Context()
    ds = ctx.parallelize([3, 1, 3, 2, 1]).map(_noncompilable_scalar).unique()
    got = ds.collect()
    assert ds._last_outcome.mode == "fallback"
    assert sorted(got) == [1, 2, 3]


def _noncompilable_scalar(x):
    return sorted([x, x])[0]


def test_csv_fallback_aggregate_by_key(tmp_path):
    p = _write_csv(tmp_path, "k,v\n1,10\n2,5\n1,7\n2,1\n")
    ctx = tuplex_amd.Context()
    ds = ctx.csv(p).mapColumn("k", _noncompilable_scalar) \
        .aggregateByKey(lambda a, b: a + b,
                        lambda a, x: a + x["v"], 0, ["k"])
    got = ds.collect()
    assert ds._last_outcome.mode == "fallback"
    assert sorted(got) == [(1, 17), (2, 6)]


def test_csv_fallback_unique(tmp_path):
    p = _write_csv(tmp_path, "v\n3\n1\n3\n2\n1\n")
    ctx = tuplex_amd.Context()
    ds = ctx.csv(p).mapColumn("v", _noncompilable_scalar).unique()
    got = ds.collect()
    assert ds._last_outcome.mode == "fallback"
    assert sorted(got) == [1, 2, 3]


def test_csv_fallback_aggregate_uses_output_columns(tmp_path):
    """ADVICE medium: the fallback fold must hand the agg UDF the pipeline's
    OUTPUT columns — after renameColumn the agg fn keys on the new name."""
    p = _write_csv(tmp_path, "a,b\n1,10\n2,20\n")
    ctx = tuplex_amd.Context()
    ds = ctx.csv(p).mapColumn("b", _noncompilable_scalar) \
        .renameColumn("b", "price") \
        .aggregate(lambda a, b: a + b, lambda a, x: a + x["price"], 0)
    got = ds.collect()
    assert ds._last_outcome.mode == "fallback"
    assert got == [30]


def test_csv_fallback_tocsv(tmp_path):
    """tocsv through the whole-stage fallback (was NotImplementedError)."""
    p = _write_csv(tmp_path, "a,b\n1,x\n2,y\n")
    out = os.path.join(str(tmp_path), "out.csv")
    ctx = tuplex_amd.Context()
    ds = ctx.csv(p).mapColumn("a", _noncompilable_scalar)
    ds.tocsv(out)
    with open(out) as f:
        assert f.read() == "a,b\n1,x\n2,y\n"


def test_orc_fallback_trailing_aggregate(tmp_path):
    """ADVICE medium: ORC pipeline ending in a non-reducible aggregate must
    still fold host-side (agg fn is max-shaped, not `a + expr(x)`)."""
    pa = pytest.importorskip("pyarrow")
    import pyarrow.orc as paorc
    tab = pa.table({"a": [3, 9, 4]})
    p = os.path.join(str(tmp_path), "t.orc")
    paorc.write_table(tab, p)
    ctx = tuplex_amd.Context()
    # single-column rows are SCALARS in UDFs (reference row semantics)
    ds = ctx.orc(p).mapColumn("a", _noncompilable_scalar) \
        .aggregate(lambda a, b: max(a, b), lambda a, x: max(a, x), 0)
    got = ds.collect()
    assert ds._last_outcome.mode == "fallback"
    assert got == [9]


def test_orc_fallback_aggregate_by_key(tmp_path):
    pa = pytest.importorskip("pyarrow")
    import pyarrow.orc as paorc
    import decimal
    # decimal column forces the no-StageProgram fallback path
    tab = pa.table({"k": [1, 2, 1],
                    "v": [10, 5, 7],
                    "d": pa.array([decimal.Decimal("1.25")] * 3,
                                  type=pa.decimal128(10, 2))})
    p = os.path.join(str(tmp_path), "t.orc")
    paorc.write_table(tab, p)
    ctx = tuplex_amd.Context()
    ds = ctx.orc(p).selectColumns(["k", "v"]) \
        .aggregateByKey(lambda a, b: a + b, lambda a, x: a + x["v"], 0, ["k"])
    got = ds.collect()
    assert ds._last_outcome.mode == "fallback"
    assert sorted(got) == [(1, 17), (2, 5)]


def test_replay_positional_subscript():
    """ADVICE low: a 1-param UDF using x[0] on a named multi-column row must
    replay identically to the compiled mapping (x[0] -> column 0)."""
    ctx = tuplex_amd.Context()
    rows = [(1, "a"), (2, "b")]
    ds = ctx.parallelize(rows, columns=["n", "s"]) \
        .map(lambda x: sorted([x[0], x[0]])[0])  # noncompilable + positional
    got = ds.collect()
    assert ds._last_outcome.mode == "fallback"
    assert got == [1, 2]


def test_rowview_mixed_access():
    from tuplex_amd.resolve import RowView
    r = RowView((5, "z"), ["num", "s"])
    assert r["num"] == 5 and r[0] == 5 and r[1] == "z" and r["s"] == "z"
    assert r[0:2] == (5, "z")
    assert len(r) == 2 and set(r.keys()) == {"num", "s"}


def test_finalize_merged_helpers():
    rows = [(1, 10), (2, 5), (1, 7)]
    out = finalize_merged(rows, [("aggregateByKey", None,
                                  lambda a, x: a + x["v"], 0, ["k"])],
                          ["k", "v"])
    assert sorted(out) == [(1, 17), (2, 5)]
    assert output_columns_of(["a", "b"],
                             [("renameColumn", "b", "c")]) == ["a", "c"]
