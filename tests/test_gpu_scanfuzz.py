"""GPU fuzz parity for the fused multi-needle scan machinery (codegen
_emit_scan_group): random haystacks stressing 8/32-byte window boundaries,
case-insensitive matches, needles longer than the 8-byte register window, and
unaligned cell starts — results must equal CPython's str.find/in exactly.

Also covers the round-2 finalize paths that need a device: unique() over a csv
source (crash fix), and ORC pipelines ending in a non-reducible aggregate."""
import os
import random

import pytest

import tuplex_amd

pytestmark = pytest.mark.gpu


def scan_udf(x):
    s = x["h"]
    return (s.find(" bd"), s.find(" ba"), s.find(" sqft"),
            s.find("Price/sqft:"), 1 if "sale" in s.lower() else 0,
            1 if "foreclose" in s.lower() else 0, s.rfind(","))


def _mk_haystacks(n=4000, seed=11):
    rng = random.Random(seed)
    frags = [" bd", " ba", " sqft", " b", "bd", "sqf", "Price/sqft:",
             "Price/sqft", "SALE", "sale", "SaLe", "foreclose", "FORECLOSE",
             "sal", "x", " ", ",", "q", "s", "3", "ba ", "t:", "price"]
    out = []
    for _ in range(n):
        parts = [rng.choice(frags) for _ in range(rng.randint(0, 14))]
        s = "".join(parts)
        # pad to hit specific lengths around the 8/32-B loop edges
        tgt = rng.choice([0, 1, 7, 8, 9, 15, 31, 32, 33, 40, 63, 64, 65, 100])
        if len(s) < tgt:
            s += "".join(rng.choice("abQZ ,9") for _ in range(tgt - len(s)))
        out.append(s.replace('"', "").replace("\n", ""))
    return out


def test_scan_fuzz_parity():
    data = _mk_haystacks()
    ctx = tuplex_amd.Context()
    ds = ctx.parallelize(data, columns=["h"]).map(scan_udf)
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    exp = [scan_udf({"h": s}) for s in data]
    assert got == exp


def test_scan_fuzz_csv_source(tmp_path):
    """Same scans via the csv source (LDS-staged haystacks at arbitrary
    in-row offsets)."""
    data = _mk_haystacks(2000, seed=12)
    p = os.path.join(str(tmp_path), "scan.csv")
    with open(p, "w") as f:
        f.write("pre,h,post\n")
        for i, s in enumerate(data):
            # RFC-4180-quote cells containing the delimiter (the GPU parses
            # unescaped quoted cells natively; bare commas would split the cell)
            cell = '"%s"' % s if "," in s else s
            f.write("p%d,%s,%d\n" % (i, cell, i))
    ctx = tuplex_amd.Context()
    ds = ctx.csv(p).map(scan_udf)
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    exp = [scan_udf({"h": s}) for s in data]
    assert got == exp


def test_csv_unique_gpu(tmp_path):
    """unique() on a csv source (round-1 ADVICE: the GPU merge crashed)."""
    p = os.path.join(str(tmp_path), "u.csv")
    with open(p, "w") as f:
        f.write("v\n")
        for i in range(5000):
            f.write("%d\n" % (i % 7))
    ctx = tuplex_amd.Context()
    ds = ctx.csv(p).unique()
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    assert sorted(got) == list(range(7))


def test_orc_trailing_nonreducible_aggregate(tmp_path):
    """ORC + GPU stage + host-side fold of a max-shaped aggregate (round-1
    ADVICE: run_orc returned raw rows)."""
    pa = pytest.importorskip("pyarrow")
    import pyarrow.orc as paorc
    tab = pa.table({"a": list(range(100)) + [7]})
    p = os.path.join(str(tmp_path), "t.orc")
    paorc.write_table(tab, p)
    ctx = tuplex_amd.Context()
    # single-column rows are SCALARS in UDFs (reference row semantics)
    ds = ctx.orc(p).aggregate(lambda a, b: max(a, b),
                              lambda a, x: max(a, x), 0)
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    assert got == [99]
