"""GPU CSV parity: ctx.csv(...) through the GPU scan+parse kernels vs the oracle
CSV restatement, on clean and dirty synthetic Zillow data (BASELINE.json
configs[1]/[2] shape)."""
import os

import pytest

import tuplex_amd
from oracle import pyoracle_csv
from tests.pipelines import apply_ops
from tests.test_codegen_compile import zillow_ops
from tests.zillow_data import make_zillow_csv_bytes

pytestmark = pytest.mark.gpu


def _write(tmp_path, data):
    p = os.path.join(str(tmp_path), "z.csv")
    with open(p, "wb") as f:
        f.write(data)
    return p


@pytest.mark.parametrize("dirty", [0.0, 0.03], ids=["clean", "dirty"])
def test_zillow_csv_collect(tmp_path, dirty):
    data, _rows = make_zillow_csv_bytes(3000, seed=42, dirty_frac=dirty)
    path = _write(tmp_path, data)
    ctx = tuplex_amd.Context()
    ds = apply_ops(ctx.csv(path), zillow_ops())
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle_csv.run_csv_pipeline(data, zillow_ops())
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]
    assert len(got) > 100


def test_zillow_tocsv(tmp_path):
    data, _rows = make_zillow_csv_bytes(2000, seed=9, dirty_frac=0.02)
    path = _write(tmp_path, data)
    outp = os.path.join(str(tmp_path), "out.csv")
    ctx = tuplex_amd.Context()
    ds = apply_ops(ctx.csv(path), zillow_ops())
    ds.tocsv(outp)
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    got = open(outp, "rb").read()
    ref = pyoracle_csv.run_csv_pipeline(data, zillow_ops(), sink="csv")
    assert got == ref["csv_text"]


def test_csv_quoted_newlines_and_escapes(tmp_path):
    rows = [b"a,b,c",
            b'1,"x\ny",3',
            b'2,"say ""hi""",4',
            b"3,plain,5",
            b'4,"q,comma",6']
    data = b"\n".join(rows) + b"\n"
    path = _write(tmp_path, data)
    ctx = tuplex_amd.Context()
    from tests.pipelines import sq_map  # any trivial compilable op

    def ident3(x):
        return (x["a"], x["b"], x["c"])

    ds = ctx.csv(path).map(ident3)
    got = ds.collect()
    ref = pyoracle_csv.run_csv_pipeline(data, [("map", ident3)])
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]


def test_csv_numeric_sniff_and_parse(tmp_path):
    lines = [b"id,val,flag"] + \
        [b"%d,%d.5,true" % (i, i * 3) for i in range(500)] + \
        [b"bogus,1.5,false"]  # one bad i64 row -> badparse replay
    data = b"\n".join(lines) + b"\n"
    path = _write(tmp_path, data)

    def add(x):
        return x["id"] + x["val"]

    ctx = tuplex_amd.Context()
    ds = ctx.csv(path).map(add)
    got = ds.collect()
    ref = pyoracle_csv.run_csv_pipeline(data, [("map", add)])
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]
