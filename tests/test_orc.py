"""ORC columnar ingest: Arrow buffers -> HBM, typed loads,
no parse. CPU tests cover schema mapping / codegen compile / fails-loudly /
interpreter fallback; GPU tests are full parity vs the mem-semantics oracle."""
import os

import pytest

import tuplex_amd
from oracle import pyoracle

pa = pytest.importorskip("pyarrow")
import pyarrow.orc as paorc  # noqa: E402


def _mk_table(n=4000, with_unicode=True):
    import random
    rng = random.Random(7)
    a, b, name, price, active = [], [], [], [], []
    for i in range(n):
        a.append(i)
        b.append(None if i % 50 == 0 else i * 3)
        s = "item-%d" % i
        if with_unicode and i % 97 == 0:
            s = "itém-%d" % i  # non-ASCII -> NCV divert -> host replay
        name.append(s)
        price.append(rng.random() * 100.0)
        active.append(i % 3 == 0)
    return pa.table({"a": a, "b": b, "name": name, "price": price,
                     "active": active})


def _write_orc(tmp_path, tab, fname="t.orc"):
    p = os.path.join(str(tmp_path), fname)
    paorc.write_table(tab, p)
    return p


def _rows(tab):
    cols = [tab.column(n).to_pylist() for n in tab.schema.names]
    return list(zip(*cols))


def orc_ops():
    def use(x):
        return (x["a"], x["name"].upper(), x["b"], x["price"] * 2.0)

    def keep(x):
        return x[0] % 2 == 0

    return [("map", use), ("filter", keep)]


def test_orc_schema_and_codegen_compile(tmp_path):
    from tuplex_amd import codegen, plan
    from tuplex_amd import ttypes as T
    sp = plan.build_stage([T.I64, ("opt", T.I64), T.STR, T.F64, T.BOOL],
                          ["a", "b", "name", "price", "active"], orc_ops())
    assert sp.compilable, sp.why_not_compilable
    src, desc = codegen.generate_stage(sp, source="col", sink="mem")
    assert "tpx_stage_execute" not in src  # kernels only
    assert "const void* const* ct" in src
    assert "source=col" in desc
    # compile-only (hiprtc works without a GPU)
    from tuplex_amd.engine import GpuLib
    glib = GpuLib.get()
    st = glib.lib.tpx_stage_compile(src.encode(), desc.encode(), b"", 1)
    assert st, glib.err()


def test_orc_fails_loudly_without_gpu(tmp_path):
    from tuplex_amd.engine import GpuLib
    try:
        has_gpu = GpuLib.get().device_count() > 0
    except RuntimeError:
        has_gpu = False
    if has_gpu:
        pytest.skip("only meaningful without a GPU")
    tab = _mk_table(100)
    p = _write_orc(tmp_path, tab)
    from tests.pipelines import apply_ops
    ds = apply_ops(tuplex_amd.Context().orc(p), orc_ops())
    with pytest.raises(RuntimeError, match="HIP|GPU"):
        ds.collect()


def test_orc_unsupported_type_falls_back(tmp_path):
    """decimal column -> whole-stage interpreter fallback (CPU-runnable)."""
    import decimal
    tab = pa.table({"a": [1, 2, 3],
                    "d": pa.array([decimal.Decimal("1.25")] * 3,
                                  type=pa.decimal128(10, 2))})
    p = _write_orc(tmp_path, tab)

    def probe(x):
        return x["a"] * 10

    ds = tuplex_amd.Context().orc(p).map(probe)
    got = ds.collect()
    assert ds._last_outcome.mode == "fallback"
    assert got == [10, 20, 30]


@pytest.mark.gpu
def test_orc_gpu_parity(tmp_path):
    tab = _mk_table(4000)
    p = _write_orc(tmp_path, tab)
    from tests.pipelines import apply_ops
    ds = apply_ops(tuplex_amd.Context().orc(p), orc_ops())
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle.run_pipeline(_rows(tab), orc_ops(),
                                columns=list(tab.schema.names))
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]
    assert any("ITÉM" in r[1] for r in got)  # replayed non-ASCII rows merged


@pytest.mark.gpu
def test_orc_gpu_aggregate(tmp_path):
    tab = _mk_table(10000, with_unicode=False)
    p = _write_orc(tmp_path, tab)

    def comb(a, b):
        return a + b

    def addup(a, x):
        return a + x["a"]

    ds = tuplex_amd.Context().orc(p).aggregate(comb, addup, 0)
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    assert got == [sum(range(10000))]


@pytest.mark.gpu
def test_orc_multi_file_and_tocsv(tmp_path):
    t1 = _mk_table(1500, with_unicode=False)
    t2 = _mk_table(700, with_unicode=False)
    _write_orc(tmp_path, t1, "part-0.orc")
    _write_orc(tmp_path, t2, "part-1.orc")
    pattern = os.path.join(str(tmp_path), "part-*.orc")
    from tests.pipelines import apply_ops
    ds = apply_ops(tuplex_amd.Context().orc(pattern), orc_ops())
    out = os.path.join(str(tmp_path), "out.csv")
    ds.tocsv(out)
    assert ds._last_outcome.mode == "gpu"
    rows = _rows(t1) + _rows(t2)
    ref = pyoracle.run_pipeline(rows, orc_ops(),
                                columns=list(t1.schema.names))
    with open(out, "rb") as f:
        lines = f.read().decode().strip().split("\n")
    assert len(lines) == 1 + len(ref["output"])  # header + rows, in order


@pytest.mark.gpu
def test_cache_and_toorc_roundtrip(tmp_path):
    """cache() materializes (CacheOperator analog); toorc -> orc() round trip."""
    tab = _mk_table(2000, with_unicode=False)
    p = _write_orc(tmp_path, tab)
    from tests.pipelines import apply_ops
    ds = apply_ops(tuplex_amd.Context().orc(p), orc_ops())
    cached = ds.cache()
    assert cached._last_outcome.mode == "gpu"

    def second(x):
        return (x[0] * 10, x[1])

    got = cached.map(second).collect()
    ref0 = pyoracle.run_pipeline(_rows(tab), orc_ops(),
                                 columns=list(tab.schema.names))
    ref = pyoracle.run_pipeline(ref0["output"], [("map", second)])
    assert got == ref["output"]

    out = os.path.join(str(tmp_path), "o.orc")
    cached.toorc(out)
    back = paorc.read_table(out)
    assert back.num_rows == len(ref0["output"])
    assert _rows(back) == ref0["output"]
