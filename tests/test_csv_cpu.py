"""CPU-side CSV tests: oracle CSV restatement self-consistency, sniffing, row/cell
split rules, and hipRTC compile of the CSV-source Zillow stage."""
import ctypes
import os

import pytest

from oracle import pyoracle, pyoracle_csv
from tuplex_amd import codegen, csvio, plan
from tuplex_amd import ttypes as T
from tests.zillow_data import ZILLOW_COLS, make_zillow_csv_bytes
from tests.test_codegen_compile import zillow_ops, _compile_only


def test_split_rows_quoted_newline():
    data = b'a,"x\ny",b\nc,d,e\n'
    rows = pyoracle_csv.split_rows(data)
    assert len(rows) == 2
    assert rows[0] == b'a,"x\ny",b\n'


def test_split_cells_quotes_and_escapes():
    cells, flags = pyoracle_csv.split_cells(b'plain,"quoted, cell",tail\n')
    assert [c.decode() for c in cells] == ["plain", "quoted, cell", "tail"]
    assert flags == 0
    cells, flags = pyoracle_csv.split_cells(b'a,"he said ""hi""",b\n')
    assert flags & 2  # escape -> divert
    cells, flags = pyoracle_csv.split_cells(b'a,"unterminated\n')
    assert flags & 4


def test_product_split_matches_oracle():
    import random
    rng = random.Random(3)
    pool = 'abc,"\n'
    for _ in range(200):
        s = "".join(rng.choice(pool) for _ in range(rng.randint(0, 40))) + "\n"
        b = s.encode()
        assert csvio.split_rows(b) == pyoracle_csv.split_rows(b)
        assert csvio.split_cells(b) == pyoracle_csv.split_cells(b)


def test_sniff_zillow_schema():
    data, _rows = make_zillow_csv_bytes(500, seed=42)
    has_header, names, types = pyoracle_csv.sniff(data, [""], 0.9, None,
                                                  None)
    assert has_header
    assert names == ZILLOW_COLS
    assert types[4] == "f64"  # postal_code
    assert all(t == "str" for i, t in enumerate(types) if i != 4)
    # product sniffer agrees
    h2, n2, t2 = csvio.sniff(data, [""], 0.9, None, None)
    assert (h2, n2) == (has_header, names)
    assert [T.deopt(t) if not T.is_opt(t) else t for t in t2] == \
        ["str", "str", "str", "str", "f64", "str", "str", "str", "str", "str"]


def test_oracle_csv_matches_oracle_mem_on_zillow():
    """Internal consistency: the oracle's CSV path over serialized rows must equal
    the oracle's mem path over the same typed rows."""
    data, rows = make_zillow_csv_bytes(800, seed=11, dirty_frac=0.03)
    r_csv = pyoracle_csv.run_csv_pipeline(data, zillow_ops(), columns=None,
                                          header=None, null_values=[""])
    r_mem = pyoracle.run_pipeline(rows, zillow_ops(), columns=ZILLOW_COLS)
    assert r_csv["output"] == r_mem["output"]
    assert r_csv["exception_counts"] == r_mem["exception_counts"]
    assert len(r_csv["output"]) > 20


def test_oracle_csv_tocsv_text():
    data, rows = make_zillow_csv_bytes(200, seed=5, dirty_frac=0.0)
    r = pyoracle_csv.run_csv_pipeline(data, zillow_ops(), sink="csv")
    text = r["csv_text"].decode()
    lines = text.strip().split("\n")
    assert lines[0].startswith("url,zipcode,address,city,state,bedrooms")
    assert len(lines) == len(r["output"]) + 1


def test_compile_zillow_csv_source():
    sp = plan.build_stage(
        [T.STR, T.STR, T.STR, T.STR, T.F64, T.STR, T.STR, T.STR, T.STR, T.STR],
        ZILLOW_COLS, zillow_ops())
    assert sp.compilable, sp.why_not_compilable
    _compile_only(sp, source="csv", sink="csv")
    _compile_only(sp, source="csv", sink="mem")


def test_try_f64_quirks():
    assert csvio.try_f64("-")      # fast_atod quirk: '-' -> -0.0
    assert csvio.try_f64(".")
    assert csvio.try_f64("e5")
    assert csvio.try_f64("nan") and csvio.try_f64("INF")
    assert not csvio.try_f64("-nan")   # sign consumes -> nan match disabled
    assert not csvio.try_f64("abc")
    assert not csvio.try_f64("")
    assert csvio.try_f64("1801.0") and csvio.try_f64("-1e3")


def _nc_pair(x):
    # sorted() -> interpreter path (CPU-runnable)
    return (sorted([x["a"], x["a"]])[0], x["b"] * 2)


def test_multifile_headers_and_missing_trailing_newline(tmp_path):
    """fmeta/_load_data path: glob over files that each carry a header and
    whose last file lacks a trailing newline — concatenation must strip every
    header and synthesize the final newline (reference concatenates parts the
    same way)."""
    import os
    want = []
    for fi in range(3):
        p = os.path.join(str(tmp_path), "part%d.csv" % fi)
        body = ""
        for i in range(5):
            v = fi * 100 + i
            body += "%d,%d\n" % (v, v * 3)
            want.append((v, v * 6))
        if fi == 2:
            body = body[:-1]  # no trailing newline on the last file
        with open(p, "w") as f:
            f.write("a,b\n" + body)
    import tuplex_amd
    ds = (tuplex_amd.Context()
          .csv(os.path.join(str(tmp_path), "part*.csv")).map(_nc_pair))
    got = ds.collect()
    assert ds._last_outcome.mode == "fallback"
    assert got == want


def test_single_file_no_header_numeric(tmp_path):
    import os
    p = os.path.join(str(tmp_path), "in.csv")
    with open(p, "w") as f:
        for i in range(20):
            f.write("%d,%d\n" % (i, i + 1))
    import tuplex_amd
    ds = tuplex_amd.Context().csv(p, header=False).map(_nc_pair_cols)
    got = ds.collect()
    assert ds._last_outcome.mode == "fallback"
    assert got == [(i, (i + 1) * 2) for i in range(20)]


def _nc_pair_cols(x):
    return (sorted([x[0], x[0]])[0], x[1] * 2)
