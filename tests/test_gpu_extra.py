"""GPU parity for the remaining BASELINE configs: TPC-H Q6 aggregate (configs[3]),
flights-shaped wide schema w/ ~1% malformed rows (configs[2]), weblog string-split
(configs[4]), plus span-cap fallback (long rows) and the GPU-reduced aggregates."""
import os

import pytest

import tuplex_amd
from oracle import pyoracle, pyoracle_csv
from tests.pipelines import apply_ops
from tests import extra_data as X

pytestmark = pytest.mark.gpu


def _write(tmp_path, data, name="d.csv"):
    p = os.path.join(str(tmp_path), name)
    with open(p, "wb") as f:
        f.write(data)
    return p


def test_q6_aggregate_csv(tmp_path):
    data = make = X.make_lineitem_csv(200000, seed=42)
    path = _write(tmp_path, data)
    ctx = tuplex_amd.Context()
    ds = apply_ops(ctx.csv(path, columns=X.LINEITEM_COLS, header=False,
                           delimiter="|"), X.q6_ops())
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle_csv.run_csv_pipeline(data, X.q6_ops(),
                                        columns=X.LINEITEM_COLS, header=False,
                                        delimiter="|")
    assert len(got) == 1
    r, g = ref["output"][0], got[0]
    assert abs(g - r) <= 1e-9 * max(1.0, abs(r)), (g, r)
    assert ds.exception_counts == ref["exception_counts"]
    assert r != 0.0


def test_agg_sum_count_exact_gpu():
    """i64 aggregates are exact regardless of reduction order."""
    ctx = tuplex_amd.Context()
    data = list(range(1, 100001))
    from tests.pipelines import agg_combine, agg_sum
    ds = ctx.parallelize(data).aggregate(agg_combine, agg_sum, 0)
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu"
    assert got == [sum(data)]


def test_flights_wide_csv(tmp_path):
    data = X.make_flights_csv(3000, seed=7, bad_frac=0.01)
    path = _write(tmp_path, data)
    ctx = tuplex_amd.Context()
    ds = apply_ops(ctx.csv(path), X.flights_ops())
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    from tests.pipelines import csv_used_cols
    ref = pyoracle_csv.run_csv_pipeline(
        data, X.flights_ops(),
        used_cols=csv_used_cols(data, X.flights_ops()))
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]
    assert len(got) > 100


def test_logs_split_csv(tmp_path):
    data = X.make_weblog_lines(20000, seed=3, bad_frac=0.01)
    path = _write(tmp_path, data, "access.log")
    ctx = tuplex_amd.Context()
    ds = apply_ops(ctx.csv(path, header=False), X.logs_ops())
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle_csv.run_csv_pipeline(data, X.logs_ops(), header=False)
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]
    assert len(got) > 1000


def test_long_rows_span_fallback():
    """Rows far beyond the 16 KiB/wave LDS staging cap take the global-parse
    path; results must be identical."""
    data = X.make_long_rows(300, seed=9)

    def probe(x):
        return (x.find("zzz"), len(x), x[:16])

    ctx = tuplex_amd.Context()
    ds = ctx.parallelize(data).map(probe)
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu"
    ref = pyoracle.run_pipeline(data, [("map", probe)])
    assert got == ref["output"]


def _sorted_rows(rows):
    return sorted(rows)


@pytest.mark.parametrize("name", ["sum_by_key_small", "sum_by_key_large",
                                  "count_by_key"])
def test_aggregate_by_key_gpu(name):
    from tests.pipelines import BYKEY_PIPELINES
    nm, data, columns, ops = [p for p in BYKEY_PIPELINES if p[0] == name][0]
    ctx = tuplex_amd.Context()
    from tests.pipelines import apply_ops as ap
    ds = ap(ctx.parallelize(data, columns=columns), ops)
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle.run_pipeline(data, ops, columns=columns)
    g, r = _sorted_rows(got), _sorted_rows(ref["output"])
    assert len(g) == len(r)
    for (gk, gv), (rk, rv) in zip(g, r):
        assert gk == rk
        assert abs(gv - rv) <= 1e-9 * max(1.0, abs(rv)), (gk, gv, rv)
    assert ds.exception_counts == ref["exception_counts"]


def test_text_source(tmp_path):
    from tests.extra_data import make_weblog_lines, logs_ops
    data = make_weblog_lines(10000, seed=13, bad_frac=0.01)
    p = _write(tmp_path, data, "w.log")
    ctx = tuplex_amd.Context()
    from tests.pipelines import apply_ops as ap
    ds = ap(ctx.text(p), logs_ops())
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle_csv.run_text_pipeline(data, logs_ops())
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]


def test_text_source_with_quotes(tmp_path):
    # text rows split on EVERY newline — embedded quotes are plain bytes
    lines = [b'say "hi" there', b'a,b,"c', b"plain"]
    data = b"\n".join(lines) + b"\n"
    p = _write(tmp_path, data, "q.log")
    ctx = tuplex_amd.Context()

    def upperit(x):
        return x.upper()

    ds = ctx.text(p).map(upperit)
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle_csv.run_text_pipeline(data, [("map", upperit)])
    assert got == ref["output"]


def test_unique_gpu():
    import random
    rng = random.Random(2)
    data = [rng.randint(0, 500) for _ in range(20000)]
    ctx = tuplex_amd.Context()
    ds = ctx.parallelize(data).unique()
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle.run_pipeline(data, [("unique",)])
    assert sorted(got) == sorted(ref["output"])


def test_multi_partition_mem():
    """partitionSize splitting -> multi-partition C-ABI batch (ContextOptions
    partitionSize, LocalBackend PartitionGroup analog)."""
    from tests.zillow_data import make_zillow_rows, ZILLOW_COLS
    from tests.test_codegen_compile import zillow_ops
    rows = make_zillow_rows(3000, seed=21, dirty_frac=0.02)
    ctx = tuplex_amd.Context({"partitionSize": "64KB"})
    from tests.pipelines import apply_ops as ap
    ds = ap(ctx.parallelize(rows, columns=ZILLOW_COLS), zillow_ops())
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle.run_pipeline(rows, zillow_ops(), columns=ZILLOW_COLS)
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]


def test_chunked_csv_collect_and_tocsv(tmp_path):
    """inputSplitSize chunking with exact quote-parity boundaries: quoted
    newlines must never split a row (CSVUtils.cc:1494 findLineStart purpose)."""
    import random
    from tests.zillow_data import make_zillow_csv_bytes
    from tests.test_codegen_compile import zillow_ops
    data, _ = make_zillow_csv_bytes(4000, seed=31, dirty_frac=0.02)
    # sprinkle rows with quoted embedded newlines into a copy of the stream
    rng = random.Random(4)
    lines = data.split(b"\n")
    header, body = lines[0], lines[1:-1]
    for i in range(0, len(body), 97):
        parts = body[i].split(b",")
        parts[1] = b'"line1\nline2"'  # address cell with embedded newline
        body[i] = b",".join(parts)
    data = header + b"\n" + b"\n".join(body) + b"\n"
    p = _write(tmp_path, data, "chunked.csv")

    from tests.pipelines import apply_ops as ap
    ctx = tuplex_amd.Context({"inputSplitSize": "128KB"})
    ds = ap(ctx.csv(p), zillow_ops())
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    assert ctx.metrics.data.get("chunks", 1) > 2
    ref = pyoracle_csv.run_csv_pipeline(data, zillow_ops())
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]

    outp = os.path.join(str(tmp_path), "out.csv")
    ds2 = ap(ctx.csv(p), zillow_ops())
    ds2.tocsv(outp)
    ref2 = pyoracle_csv.run_csv_pipeline(data, zillow_ops(), sink="csv")
    assert open(outp, "rb").read() == ref2["csv_text"]


def test_csv_resolver(tmp_path):
    """resolve() on a CSV-source pipeline: badparse/UDF errors replay through the
    resolver chain (ResolveTask.cc:389 order)."""
    from tests.zillow_data import make_zillow_csv_bytes
    from tests.extra_data import q6_ops  # noqa: F401
    data, _ = make_zillow_csv_bytes(2500, seed=77, dirty_frac=0.04)
    p = _write(tmp_path, data, "r.csv")

    def bd(x):
        val = x["facts and features"]
        max_idx = val.find(" bd")
        if max_idx < 0:
            max_idx = len(val)
        s = val[:max_idx]
        split_idx = s.rfind(",")
        split_idx = 0 if split_idx < 0 else split_idx + 2
        return int(s[split_idx:])

    def bd_fallback(x):
        return -1

    ops = [("withColumn", "bedrooms", bd),
           ("resolve", ValueError, bd_fallback),
           ("selectColumns", ["city", "bedrooms"])]
    ctx = tuplex_amd.Context()
    from tests.pipelines import apply_ops as ap
    ds = ap(ctx.csv(p), ops)
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle_csv.run_csv_pipeline(data, ops)
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]
    assert any(v == -1 for _, v in got)  # resolver actually fired


def test_type_hints_and_custom_nulls(tmp_path):
    """type_hints override sniffing; custom null_values make numeric columns
    Option-typed with None semantics (context.py:288/:322)."""
    from typing import Optional
    lines = [b"a,b,c"] + [b"%d,NULL,x%d" % (i, i) if i % 5 == 0
                          else b"%d,%d,x%d" % (i, i * 2, i)
                          for i in range(2000)]
    data = b"\n".join(lines) + b"\n"
    p = _write(tmp_path, data, "h.csv")

    def use(x):
        return (x["a"], str(x["b"]), x["c"])

    ctx = tuplex_amd.Context()
    ds = ctx.csv(p, null_values=["NULL"],
                 type_hints={"b": Optional[int]}).map(use)
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle_csv.run_csv_pipeline(data, [("map", use)],
                                        null_values=["NULL"],
                                        type_hints={"b": "opt_i64"})
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]
    assert sum(1 for _, b, _c in got if b == "None") == 400


def test_chunk_pipelined_large_input(tmp_path):
    """>=1M rows triggers the 4-chunk multi-stream pipeline in run_core
    (tpx_abi.cpp): per-chunk scans/writes overlap later chunks' main kernel.
    Input = 20 tiles of one 64k-row dirty base, so the expected output is the
    single-tile oracle result repeated 20x in order (merge order, exception
    replay and chunk-boundary stitching all checked)."""
    from tests.test_codegen_compile import zillow_ops
    from tests.zillow_data import make_zillow_csv_bytes

    base, _ = make_zillow_csv_bytes(64000, seed=13, dirty_frac=0.01,
                                    header=True)
    header, body = base.split(b"\n", 1)
    data = header + b"\n" + body * 20  # 1.28M rows
    p = _write(tmp_path, data, "big.csv")

    # one engine call for the whole file (default inputSplitSize would split
    # into <1M-row chunks and never reach the C=4 path)
    ctx = tuplex_amd.Context({"tuplex.inputSplitSize": "1GB"})
    ds = apply_ops(ctx.csv(p), zillow_ops())
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason

    ref = pyoracle_csv.run_csv_pipeline(base, zillow_ops())
    assert len(got) == 20 * len(ref["output"])
    assert got == ref["output"] * 20
    for k, v in ref["exception_counts"].items():
        assert ds.exception_counts.get(k, 0) == v * 20, (k, v)


def test_peephole_edge_parity(tmp_path):
    """Edge rows through the rewritten fast paths: case-insensitive scans
    (contains(lower(s), lit)), int(s.replace(',','')) stack/fallback parse,
    capitalize peephole IndexError on empty city, >31-char numeric cells."""
    from tests.test_codegen_compile import zillow_ops

    cols = "title,address,city,state,postal_code,price,facts and features,real estate provider,url,sales_date"
    rows = [
        # uppercase keywords: lower() must still match (CI scan)
        b'House For SALE,1 A St,boston,MA,2125,"$1,250,000","3 bds , 2 ba , 1,500 sqft",X,u1,2020',
        b'CONDO FOR RENT,2 B St,ny,NY,10001,"$2,000/mo","2 bds , 1 ba , 900 sqft",X,u2,2020',
        # empty city -> x[0] IndexError -> interpreter replay
        b'house for sale,3 C St,,MA,2125,"$500,000","1 bd , 1 ba , 700 sqft",X,u3,2020',
        # >31-char digit run with commas (int_drop falls back to the
        # replace+parse path; leading zeros keep the value inside i64)
        b'house for sale,4 D St,salem,MA,1970,"$1,000,000","2 bds , 2 ba , 0,000,000,000,000,000,000,001,234,567 sqft",X,u4,2020',
        # no offer keyword at all -> extractOffer returns lowered title
        b'Mystery Listing,5 E St,lynn,MA,1901,"$750,000","4 bds , 3 ba , 2,000 sqft",X,u5,2020',
        # mixed-case condo (extractType lowers via CI scan)
        b'Nice CoNdO for Sale,6 F St,dover,NH,3820,"$600,000","3 bds , 2 ba , 1,600 sqft",X,u6,2020',
    ] * 400  # enough rows to exercise real wave shapes
    data = cols.encode() + b"\n" + b"\n".join(rows) + b"\n"
    p = _write(tmp_path, data, "edge.csv")

    ctx = tuplex_amd.Context()
    ds = apply_ops(ctx.csv(p), zillow_ops())
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle_csv.run_csv_pipeline(data, zillow_ops())
    assert got == ref["output"]
    assert ds.exception_counts == ref["exception_counts"]
    assert len(got) > 0


def test_projection_pushdown(tmp_path):
    """Unused columns are cell-walked but not value-parsed: garbage in an
    unparsed numeric column diverts NO row (LogicalOptimizer selectionPushdown;
    CSVParseRowGenerator willBeSerialized). The same pipeline without pushdown
    (oracle used_cols=None) WOULD divert those rows."""
    from tests.pipelines import csv_used_cols

    lines = [b"a,b,junk,c"]
    for i in range(5000):
        junk = b"NOT_A_NUMBER" if i % 97 == 0 else b"%d" % i  # rare: keeps the sniffed type i64
        lines.append(b"%d,%d,%s,x%d" % (i, i * 2, junk, i))
    data = b"\n".join(lines) + b"\n"
    p = _write(tmp_path, data, "pd.csv")

    def use(x):
        return (x["a"] + x["b"], x["c"])

    ops = [("map", use)]
    ctx = tuplex_amd.Context()
    ds = apply_ops(ctx.csv(p), ops)
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    assert ds.exception_counts == {}          # junk column never parsed
    assert len(got) == 5000
    used = csv_used_cols(data, ops)
    assert sorted(used) == [0, 1, 3]
    ref = pyoracle_csv.run_csv_pipeline(data, ops, used_cols=used)
    assert got == ref["output"]
    assert ref["exception_counts"] == {}
    # counter-check: without pushdown the junk rows DO divert in the oracle
    ref_full = pyoracle_csv.run_csv_pipeline(data, ops, used_cols=None)
    assert sum(ref_full["exception_counts"].values()) > 0


def test_crlf_line_endings(tmp_path):
    """Windows CRLF files: \r stripped from unquoted row ends; \r\n inside a
    quoted cell is DATA (CSVUtils row-boundary semantics)."""
    def use(x):
        return (x["a"] + 1, x["b"])

    data = b"a,b\r\n" + b"".join(b"%d,x%d\r\n" % (i, i) for i in range(5000))
    p = _write(tmp_path, data, "crlf.csv")
    ctx = tuplex_amd.Context()
    ds = apply_ops(ctx.csv(p), [("map", use)])
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle_csv.run_csv_pipeline(data, [("map", use)])
    assert got == ref["output"]
    assert got[0] == (1, "x0")

    dataq = b"a,b\r\n" + b"".join(
        (b'%d,"v\r\n%d"\r\n' if i % 7 == 0 else b"%d,x%d\r\n") % (i, i)
        for i in range(3000))
    p2 = _write(tmp_path, dataq, "crlfq.csv")
    ds2 = apply_ops(ctx.csv(p2), [("map", use)])
    got2 = ds2.collect()
    assert ds2._last_outcome.mode == "gpu", ds2._last_outcome.fallback_reason
    ref2 = pyoracle_csv.run_csv_pipeline(dataq, [("map", use)])
    assert got2 == ref2["output"]
    assert got2[0][1] == "v\r\n0"


def agg_comb2(a, b):
    return a + b


def agg_by_code(a, x):
    return a + x["n"]


def test_aggregate_by_string_key_gpu():
    """STRING-key aggregateByKey on device (string hash table; slots claim
    (ptr,len), host merges duplicate slots by key bytes)."""
    import random
    rng = random.Random(17)
    codes = ["aa", "bb", "cc", "dd", "", "longer-key-name"]
    rows = [(rng.choice(codes), rng.randint(0, 100)) for _ in range(50000)]
    ctx = tuplex_amd.Context()
    ds = ctx.parallelize(rows, columns=["code", "n"]).aggregateByKey(
        agg_comb2, agg_by_code, 0, ["code"])
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle.run_pipeline(
        rows, [("aggregateByKey", agg_comb2, agg_by_code, 0, ["code"])],
        columns=["code", "n"])
    assert sorted(got) == sorted(ref["output"])
    assert len(got) == len(codes)


def test_unique_strings_gpu():
    import random
    rng = random.Random(19)
    vals = ["s%d" % rng.randint(0, 300) for _ in range(40000)] + ["", "x"]
    ctx = tuplex_amd.Context()
    ds = ctx.parallelize(vals).unique()
    got = ds.collect()
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle.run_pipeline(vals, [("unique",)])
    assert sorted(got) == sorted(ref["output"])


def fl_tocsv_use(x):
    return (x["a"], x["b"] / (x["a"] + 1.0), x["c"])


def test_f64_csv_sink_gpu(tmp_path):
    """f64 columns in the csv sink: device %f formatting, bit-identical to
    the host/oracle "%f" (PipelineBuilder.cc:1413); out-of-range values
    divert and come back through the host formatter."""
    import os
    import random
    rng = random.Random(23)
    lines = [b"a,b,c"]
    for i in range(20000):
        if i % 997 == 0:
            b = repr(rng.uniform(1e13, 1e300)).encode()  # diverts (too big)
        elif i % 499 == 0:
            b = b"0.0000005"
        else:
            b = repr(rng.uniform(-1e6, 1e6)).encode()
        lines.append(b"%d,%s,x%d" % (i, b, i))
    data = b"\n".join(lines) + b"\n"
    p = os.path.join(str(tmp_path), "f.csv")
    with open(p, "wb") as f:
        f.write(data)
    outp = os.path.join(str(tmp_path), "out.csv")
    ctx = tuplex_amd.Context()
    ds = apply_ops(ctx.csv(p), [("map", fl_tocsv_use)])
    ds.tocsv(outp)
    assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    ref = pyoracle_csv.run_csv_pipeline(data, [("map", fl_tocsv_use)],
                                        sink="csv")
    with open(outp, "rb") as f:
        got = f.read()
    assert got == ref["csv_text"]
    assert b"." in got.split(b"\n")[1]


@pytest.mark.gpu
def test_parallel_resolver_pool_engaged(tmp_path):
    """>=MIN_POOL_ROWS diverted rows engage the persistent process pool
    (presolve); output must be identical to the serial resolver's."""
    import os
    from tuplex_amd import presolve
    rows = []
    # ~5% dirty keeps the i64 sniff (>= normalcase threshold) while the
    # divert count stays above MIN_POOL_ROWS
    n = presolve.MIN_POOL_ROWS * 25
    for i in range(n):
        if i % 20 == 0:
            rows.append("notanint,%d\n" % i)   # diverts; replay raises
        else:
            rows.append("%d,x%d\n" % (i, i))   # clean
    p = os.path.join(str(tmp_path), "d.csv")
    with open(p, "w") as f:
        f.write("a,b\n" + "".join(rows))

    def du(x):
        return (x["a"] * 2, x["b"])

    outs = {}
    for procs in ("1", "2"):
        ctx = tuplex_amd.Context({"tuplex.gpu.resolveProcesses": procs})
        ds = apply_ops(ctx.csv(p), [("map", du)])
        outs[procs] = (ds.collect(), dict(ds._last_outcome.exception_counts))
        assert ds._last_outcome.mode == "gpu", ds._last_outcome.fallback_reason
    assert outs["1"] == outs["2"]
    assert outs["1"][1].get("ValueError", 0) > presolve.MIN_POOL_ROWS // 2
