"""CPU checks for the Q6/flights/logs configs: oracle consistency + plan/codegen."""
from oracle import pyoracle, pyoracle_csv
from tests import extra_data as X
from tuplex_amd import plan, csvio
from tuplex_amd import ttypes as T


def test_q6_agg_pattern_detected():
    sp = plan.build_stage([T.I64, T.F64, T.F64, T.I64], X.LINEITEM_COLS,
                          X.q6_ops())
    assert sp.compilable, sp.why_not_compilable
    assert sp.agg_expr is not None
    assert sp.agg_type == T.F64


def test_agg_pattern_sum_count():
    from tests.pipelines import agg_combine, agg_sum
    sp = plan.build_stage([T.I64], None,
                          [("aggregate", agg_combine, agg_sum, 0)])
    assert sp.agg_expr is not None and sp.agg_type == T.I64


def test_q6_oracle_value():
    data = X.make_lineitem_csv(20000, seed=42)
    ref = pyoracle_csv.run_csv_pipeline(data, X.q6_ops(),
                                        columns=X.LINEITEM_COLS, header=False,
                                        delimiter="|")
    assert len(ref["output"]) == 1
    assert ref["output"][0] > 0
    assert ref["exception_counts"] == {}


def test_flights_oracle_has_exceptions():
    data = X.make_flights_csv(2000, seed=7, bad_frac=0.02)
    ref = pyoracle_csv.run_csv_pipeline(data, X.flights_ops())
    assert sum(ref["exception_counts"].values()) > 0
    assert len(ref["output"]) > 50


def test_logs_oracle():
    data = X.make_weblog_lines(5000, seed=3, bad_frac=0.02)
    ref = pyoracle_csv.run_csv_pipeline(data, X.logs_ops(), header=False)
    assert ref["exception_counts"].get("IndexError", 0) > 0
    assert len(ref["output"]) > 1000
    ip, url, code, size = ref["output"][0]
    assert code == 200 and "." in ip


def test_delimiter_sniff():
    assert csvio.sniff_delimiter(b"a|b|c\n1|2|3\n") == b"|"
    assert csvio.sniff_delimiter(b"a,b,c\n") == b","
    assert pyoracle_csv.sniff_delimiter(b"x\ty\tz\n") == b"\t"


def test_parallel_resolver_matches_inline(tmp_path):
    """presolve.ResolverPool replays diverted rows identically to the inline
    loop (same shrunk results, same order)."""
    from tuplex_amd import presolve
    from tuplex_amd import csvio
    from tuplex_amd import ttypes as T

    def du(x):
        return (x["a"] * 2, x["b"])

    ops = [("map", du), ("filter", lambda x: x[0] % 3 != 0)]
    col_types = [T.I64, T.STR]
    names = ["a", "b"]
    payloads = []
    for i in range(300):
        if i % 7 == 0:
            payloads.append(b"notint,s%d\n" % i)
        else:
            payloads.append(b"%d,s%d\n" % (i, i))
    inline = [presolve._shrink(
        csvio.replay_csv_row(p, col_types, [""], ops, names, ","))
        for p in payloads]
    pool = presolve.ResolverPool(col_types, [""], ops, names, ",", None,
                                 False, processes=2)
    try:
        got = pool.resolve(payloads)
    finally:
        pool.close()
    assert got == inline
    assert any(r[0] == "excname" for r in got)
    assert any(r[0] == "row" for r in got)
