"""Shared pipeline definitions for GPU parity tests and kernel pre-compilation.

Each entry: (name, data, columns, ops) where ops is the logical op list used both
by tuplex_amd (DataSet chaining) and the oracle. UDFs live at module level so
inspect.getsource works both here and on the GPU box, and so __graft_entry__.build()
can pre-compile every stage's hsaco into the in-tree cache (which travels with the
repo snapshot — the GPU box then loads instead of re-compiling).
"""
import random
import string

from tests.zillow_data import ZILLOW_COLS, make_zillow_rows
from tests.test_codegen_compile import zillow_ops, zillow_input_types  # noqa: F401


# ---- UDFs (module level for source capture) --------------------------------------

def sq_map(x):
    return (x, x * x)


def div_map(x):
    return 1 // x


def div_resolver(x):
    return -1


def div_filter(x):
    return (1 // x) < 5


def str_of(x):
    return str(x)


def mapcol_div(x):
    return 1 // x


def withcol_str(x, y, z):
    return str(1 // x) + y


def withcol_resolver(x, y, z):
    return "NULL"


def swapcase_udf(x):
    return x.swapcase()


def startswith_udf(s, p):
    return s.startswith(p)


def agg_combine(a, b):
    return a + b


def agg_sum(a, x):
    return a + x


def aggby_comb(a, b):
    return a + b


def aggby_vol(a, x):
    return a + x["volume"]


def aggby_cnt(a, x):
    return a + 1


def lower_udf(x):
    return x.lower()


def find_def_udf(x):
    return x.find("def")


def floor_udf(x):
    return (x // 2, x % 3, -x)


def divmod_udf(x):
    return (x, x // 7, x % 13)


def strops_udf(x):
    return (x.find(","), x.upper(), len(x))


def int_udf(x):
    return int(x)


def gt1000(x):
    return x > 1000


def ge0(x):
    return x >= 0


def ign_filter(x):
    return (1 // x) < 5


def renamed_use(x):
    return x["n2"] * 2


def sq_then(x):
    return x * x


def gt10(x):
    return x > 10


def or235(x):
    return x == 2 or x == 3 or x == 5


def cube(x):
    return x * x * x


def chain24(x):
    return 2 < x <= 4


def addf(x):
    return x + 10.7


def idivneg(x):
    return x // -6


def strdup(a, b):
    return a * b


def center3(x):
    return x.center(3)


def center4f(x, y):
    return x.center(4, y)


def _rand_ints(n=10000, seed=42):
    rng = random.Random(seed)
    return [rng.randint(-2**40, 2**40) for _ in range(n)]


def _rand_strs(n=5000, seed=7):
    rng = random.Random(seed)
    alphabet = string.ascii_letters + string.digits + " ,$/"
    return ["".join(rng.choice(alphabet) for _ in range(rng.randint(0, 60)))
            for _ in range(n)]


PIPELINES = [
    ("config1_plumbing", [1, 2, None, 4], None, [("map", sq_map)]),
    ("map_div_exc", [1, 0, 0, 2], None, [("map", div_map)]),
    ("map_div_resolved", [1, 0, 0, 2], None,
     [("map", div_map), ("resolve", ZeroDivisionError, div_resolver)]),
    ("filter_exc", [1, 0, 0, 2], None, [("filter", div_filter)]),
    ("option_str", [1, 2, None, 3, None, None], None, [("map", str_of)]),
    ("mapcol_resolve", [(1, "a"), (0, "b"), (3, "c")], ["int", "str"],
     [("mapColumn", "int", mapcol_div),
      ("resolve", ZeroDivisionError, div_resolver)]),
    ("withcol_replace", [(1, "a", True), (0, "b", False), (3, "c", True)],
     ["num", "str", "bool"], [("withColumn", "str", withcol_str)]),
    ("swapcase", ["all lower", "ALL UPPER", "Upper && Lower", "Number *45.67++",
                  "", "pQ" * 16], None, [("map", swapcase_udf)]),
    ("startswith", [("hello", "h"), ("hello", "he"), ("Hello", "hello"),
                    ("abcde", "abcde")], None, [("map", startswith_udf)]),
    ("agg_sum", [1, 2, 3, 4, 5, 6], None,
     [("aggregate", agg_combine, agg_sum, 0)]),
    ("unicode_lower", ["", "abc", "héllo wörld", "x", "ÅBC"], None,
     [("map", lower_udf)]),
    ("unicode_find", ["abc def", "héllo wörld def", "no match here"],
     None, [("map", find_def_udf)]),
    ("floor_semantics", [-7, -1, 1, 7, -8, 8], None, [("map", floor_udf)]),
    ("rand_ints", _rand_ints(), None, [("map", divmod_udf)]),
    ("rand_strs", _rand_strs(), None, [("map", strops_udf)]),
    ("int_parse", ["42", " 17 ", "-5", "bogus", "", "123456789012", "007"], None,
     [("map", int_udf)]),
    ("filter_all", list(range(100)), None, [("filter", gt1000)]),
    ("filter_none", list(range(100)), None, [("filter", ge0)]),
    ("ignore_zerodiv", [1, 0, 2, 0, 3], None,
     [("filter", ign_filter), ("ignore", ZeroDivisionError)]),
    ("rename_column", [(1, 5), (2, 6)], ["n1", "n2"],
     [("renameColumn", "n2", "n2x"), ("renameColumn", "n2x", "n2"),
      ("map", renamed_use)]),
    ("zillow_mem", make_zillow_rows(2000, seed=42), ZILLOW_COLS, zillow_ops()),
    ("agg_sum_large", list(range(1, 100001)), None,
     [("aggregate", agg_combine, agg_sum, 0)]),
    # shapes pinned by tests/golden (reference test_filter/test_arithmetic)
    ("filter_squares", [1, 2, 3, 4, 5], None,
     [("map", sq_then), ("filter", gt10)]),
    ("filter_bool_or_cubes", [1, 2, 3, 4, 5], None,
     [("filter", or235), ("map", cube)]),
    ("filter_chained_compare", [1, 2, 3, 4, 5], None,
     [("filter", chain24)]),
    ("arith_add_float", [1, 2, 4], None, [("map", addf)]),
    ("arith_idiv_negdiv", [-10, -9, -8, -7, -6, -5], None,
     [("map", idivneg)]),
    # test_strings.py:34 test_duplication (str*int and int*str repetition)
    ("strings_dup_str_int", [("negative", -2), ("zero", 0), ("hello", 1),
                             ("goodbye", 5)], None, [("map", strdup)]),
    ("strings_dup_int_str", [(-2, "negative"), (0, "zero"), (1, "hello"),
                             (6, "foo")], None, [("map", strdup)]),
    # test_strings.py:207 test_strcenter (CPython left-bias quirk)
    ("strings_center", ["", "a", "ab", "abc", "abcd"], None,
     [("map", center3)]),
    ("strings_center_fill", [("", "|"), ("a", ","), ("ab", "+"),
                             ("abc", "2"), ("abcd", "%"), ("abcde", "*")],
     None, [("map", center4f)]),
]


def _long_rows():
    from tests.extra_data import make_long_rows
    return make_long_rows(300, seed=9)


def long_probe(x):
    return (x.find("zzz"), len(x), x[:16])


PIPELINES.append(("long_rows", _long_rows(), None, [("map", long_probe)]))


def uniq_data():
    rng = random.Random(2)
    return [rng.randint(0, 500) for _ in range(20000)]


def _by_key_rows(n=50000, seed=5):
    rng = random.Random(seed)
    return [(rng.randint(-3, 1000), float(rng.randint(-50, 50)) / 4.0)
            for _ in range(n)]


# by-key pipelines: output order is parity-unpinned -> dedicated multiset-compare
# test (tests/test_gpu_extra.py), not the generic ordered parity test
BYKEY_PIPELINES = [
    ("sum_by_key_small", [(0, 10.0), (1, 20.0), (0, -4.5)], ["id", "volume"],
     [("aggregateByKey", aggby_comb, aggby_vol, 0.0, ["id"])]),
    ("sum_by_key_large", _by_key_rows(), ["id", "volume"],
     [("aggregateByKey", aggby_comb, aggby_vol, 0.0, ["id"])]),
    ("count_by_key", _by_key_rows(8000, seed=11), ["id", "volume"],
     [("aggregateByKey", aggby_comb, aggby_cnt, 0, ["id"])]),
    ("unique_i64", uniq_data(), None, [("unique",)]),
]


def apply_ops(ds, ops):
    for op in ops:
        kind = op[0]
        if kind == "map":
            ds = ds.map(op[1])
        elif kind == "filter":
            ds = ds.filter(op[1])
        elif kind == "withColumn":
            ds = ds.withColumn(op[1], op[2])
        elif kind == "mapColumn":
            ds = ds.mapColumn(op[1], op[2])
        elif kind == "selectColumns":
            ds = ds.selectColumns(op[1])
        elif kind == "renameColumn":
            ds = ds.renameColumn(op[1], op[2])
        elif kind == "resolve":
            ds = ds.resolve(op[1], op[2])
        elif kind == "ignore":
            ds = ds.ignore(op[1])
        elif kind == "aggregate":
            ds = ds.aggregate(op[1], op[2], op[3])
        elif kind == "join":
            _, rrows, rcols, lk, rk, how, lp, ls, rp, rs = op
            rds = ds._context.parallelize(
                [r if len(r) > 1 else r[0] for r in rrows], columns=rcols)
            pre = (lp, rp) if (lp or rp) else None
            suf = (ls, rs) if (ls or rs) else None
            if how == "left":
                ds = ds.leftJoin(rds, lk, rk, prefixes=pre, suffixes=suf)
            else:
                ds = ds.join(rds, lk, rk, prefixes=pre, suffixes=suf)
        elif kind == "aggregateByKey":
            ds = ds.aggregateByKey(op[1], op[2], op[3], op[4])
        else:
            raise ValueError(kind)
    return ds


# When set (a list), stage sources are collected instead of compiled so the
# caller can fan the hipRTC compiles over a thread pool (hipRTC is
# thread-safe; the hsaco cache publish is atomic rename) — build() uses this.
_COLLECTOR = None


def _stage_compile(glib, src, desc):
    if _COLLECTOR is not None:
        _COLLECTOR.append((src, desc))
        return True
    return glib.compile_stage(src, desc, compile_only=True)


def _compile_one_job(job):
    from tuplex_amd import engine
    glib = engine.GpuLib.get()
    glib.compile_stage(job[0], job[1], compile_only=True)
    return True


def compile_collected(jobs, workers=None, verbose=False):
    """hipRTC-compile collected (src, desc) stage jobs in parallel across
    PROCESSES (hipRTC/comgr serializes compiles behind an internal lock, so
    threads don't scale; each worker process gets its own hipRTC)."""
    import os as _os
    from concurrent.futures import ProcessPoolExecutor
    import multiprocessing as _mp
    nw = workers or min(8, _os.cpu_count() or 8)
    ctx = _mp.get_context("spawn")
    with ProcessPoolExecutor(max_workers=nw, mp_context=ctx) as ex:
        list(ex.map(_compile_one_job, jobs))
    return len(jobs)


def precompile_extra(verbose=False):
    """Compile-only coverage for the stage shapes outside PIPELINES: joins
    (unique + duplicate keys, i64 + str), columnar (ORC) source, string-key
    aggregateByKey/unique, f64 csv sink — catches codegen regressions in the
    no-GPU build check and warms the travelling hsaco cache."""
    from tuplex_amd import codegen, engine, plan
    from tuplex_amd import ttypes as T

    glib = engine.GpuLib.get()
    RIGHT = [(1, "one", 1.5), (2, "two", 2.5), (3, "three", None)]
    DUP = [(1, "a", 10), (1, "b", None), (2, "c", 30), (7, "z", 5),
           (1, "d", 40)]
    SDIM = [("aa", "A1", 1), ("aa", "A2", 2), ("bb", "B1", 3)]
    specs = [
        ("join_i64", [T.I64, T.STR], ["key", "val"],
         [("join", RIGHT, ["k", "label", "w"], "key", "k", "inner",
           "", "", "", "")], "mem", "mem"),
        ("join_left", [T.I64, T.STR], ["key", "val"],
         [("join", RIGHT, ["k", "label", "w"], "key", "k", "left",
           "", "", "", "")], "mem", "mem"),
        ("join_dup_mem", [T.I64, T.STR], ["key", "val"],
         [("join", DUP, ["k", "x", "w"], "key", "k", "inner",
           "", "", "", "")], "mem", "mem"),
        ("join_dup_csv", [T.STR, T.I64], ["code", "n"],
         [("join", SDIM, ["k", "label", "r"], "code", "k", "inner",
           "", "", "", "")], "csv", "csv"),
        ("join_dup_optleft", [T.STR, ("opt", T.I64)], ["code", "m"],
         [("join", SDIM, ["k", "label", "r"], "code", "k", "left",
           "", "", "", "")], "mem", "mem"),
        ("aggby_str", [T.STR, T.I64], ["code", "n"],
         [("aggregateByKey", agg_combine, aggby_str_fn, 0, ["code"])],
         "mem", "mem"),
        ("unique_str", [T.STR], ["s"], [("unique",)], "mem", "mem"),
        ("f64_csv", [T.I64, T.F64, T.STR], ["a", "b", "c"],
         [("map", f64csv_map)], "csv", "csv"),
        ("orc_col", [T.I64, ("opt", T.I64), T.STR, T.F64, T.BOOL],
         ["a", "b", "name", "price", "active"],
         [("map", orc_map), ("filter", orc_keep)], "col", "mem"),
    ]
    n = 0
    for name, in_types, cols, ops, source, sink in specs:
        sp = plan.build_stage(in_types, cols, ops)
        assert sp.compilable, (name, sp.why_not_compilable)
        kw = {"csv_info": {"null_values": [""]}} if source == "csv" else {}
        src, desc = codegen.generate_stage(sp, source=source, sink=sink, **kw)
        _stage_compile(glib, src, desc)
        n += 1
        if verbose:
            print("  extra %s ok" % name)
    return n


def aggby_str_fn(a, x):
    return a + x["n"]


def f64csv_map(x):
    return (x["a"], x["b"] / (x["a"] + 1.0), x["c"])


def orc_map(x):
    return (x["a"], x["name"].upper(), x["b"], x["price"] * 2.0)


def orc_keep(x):
    return x[0] % 2 == 0


def precompile_csv(verbose=False):
    """Pre-compile the CSV-source stages (zillow/q6/flights/logs) so their hsacos
    are in the travelling cache."""
    from tests import extra_data as X
    from tests.zillow_data import make_zillow_csv_bytes
    from tuplex_amd import codegen, csvio, engine, plan

    glib = engine.GpuLib.get()
    jobs = []
    zdata, _ = make_zillow_csv_bytes(2000, seed=42, dirty_frac=0.02)
    jobs.append(("zillow_csv", zdata, None, None, None, zillow_ops(),
                 ["mem", "csv"]))
    jobs.append(("q6", X.make_lineitem_csv(2000), X.LINEITEM_COLS, False, "|",
                 X.q6_ops(), ["mem"]))
    jobs.append(("flights", X.make_flights_csv(500), None, None, None,
                 X.flights_ops(), ["mem"]))
    jobs.append(("logs", X.make_weblog_lines(2000), None, False, None,
                 X.logs_ops(), ["mem"]))
    n = 0
    for name, data, columns, header, delimiter, ops, sinks in jobs:
        sample = data[:1 << 20]
        delim = delimiter.encode() if delimiter else csvio.sniff_delimiter(sample)
        _h, names, col_types = csvio.sniff(sample, [""], 0.9, header, columns,
                                           delim)
        sp = plan.build_stage(col_types, names, ops)
        if not sp.compilable:
            if verbose:
                print("skip:", name, sp.why_not_compilable)
            continue
        for sink in sinks:
            src, desc = codegen.generate_stage(
                sp, source="csv", sink=sink,
                csv_info={"null_values": [""], "delimiter": delim.decode()})
            _stage_compile(glib, src, desc)
            n += 1
            if verbose:
                print("precompiled csv:", name, sink)
    # text-mode stage (quote-parity off)
    sp_t = plan.build_stage(["str"], None, X.logs_ops())
    if sp_t.compilable:
        src_t, desc_t = codegen.generate_stage(
            sp_t, source="csv", sink="mem",
            csv_info={"null_values": [], "text_mode": True})
        _stage_compile(glib, src_t, desc_t)
        n += 1
    return n


def precompile_tests(verbose=False):
    """Exact stage sources of the newer GPU tests/probes (dist engine, cache
    compiled path, mid-pipeline dup join stage 2) so fresh boxes never
    hipRTC-compile during the suite."""
    from tuplex_amd import codegen, engine, plan
    from tuplex_amd import ttypes as T
    glib = engine.GpuLib.get()
    n = 0
    # test_gpu_dist stages (csv source, 3 cols)
    from tests.test_gpu_dist import _du_map, _du_keep, _du_ab, _comb, _sumb
    for ops, sinks in ((([("map", _du_map), ("filter", _du_keep)]),
                        ("mem", "csv")),
                       (([("map", _du_ab),
                          ("aggregate", _comb, _sumb, 0)]), ("mem",))):
        sp = plan.build_stage([T.I64, T.I64, T.STR], ["a", "b", "name"], ops)
        if not sp.compilable:
            continue
        for sink in sinks:
            src, desc = codegen.generate_stage(
                sp, source="csv", sink=sink,
                csv_info={"null_values": [""], "delimiter": ","})
            _stage_compile(glib, src, desc)
            n += 1
    # test_cache_cpu gpu stages (mem source)
    from tests.test_cache_cpu import _gdiv, _gres
    for in_types, cols, ops in (
            ([T.I64, T.I64], None, [("map", _gdiv)]),
            ([T.I64, T.I64], None,
             [("map", _gdiv), ("resolve", ZeroDivisionError, _gres)]),
    ):
        sp = plan.build_stage(in_types, cols, ops)
        if sp.compilable:
            src, desc = codegen.generate_stage(sp, source="mem", sink="mem")
            _stage_compile(glib, src, desc)
            n += 1
    # flights + 2 joins (the 110-col stage: ~13 min of cold hipRTC)
    from tests import extra_data as X
    from tests.test_join import FL_J1, FL_J2, _post_join_map, _post_join_keep
    from tuplex_amd import csvio
    data = X.make_flights_csv(500)
    sample = data[:1 << 18]
    _h, names, col_types = csvio.sniff(sample, [""], 0.9, None, None, b",")
    sp = plan.build_stage(col_types, names,
                          X.flights_ops() + [FL_J1, FL_J2])
    if sp.compilable:
        src, desc = codegen.generate_stage(
            sp, source="csv", sink="mem",
            csv_info={"null_values": [""], "delimiter": ","})
        _stage_compile(glib, src, desc)
        n += 1
    # mid-pipeline dup-join stage 2 (map+filter over the joined rows)
    sp = plan.build_stage([T.STR, T.I64, T.STR, ("opt", T.I64)],
                          ["val", "key", "x", "w"],
                          [("map", _post_join_map),
                           ("filter", _post_join_keep)])
    if sp.compilable:
        src, desc = codegen.generate_stage(sp, source="mem", sink="mem")
        _stage_compile(glib, src, desc)
        n += 1
    return n


def precompile_all(verbose=False):
    """Generate + hipRTC-compile (compile-only) every pipeline's stage so the
    hsaco cache is warm. Works with no GPU."""
    from tuplex_amd import codegen, engine, plan
    from tuplex_amd import ttypes as T
    from tuplex_amd.options import Options

    glib = engine.GpuLib.get()
    opts = Options()
    n = 0
    for name, data, columns, ops in PIPELINES + BYKEY_PIPELINES:
        maj = T.infer_majority_type(data, optional_threshold=opts.optional_threshold)
        row_maj = T.row_type_of(maj)
        sp = plan.build_stage(list(T.tuple_params(row_maj)), columns, ops)
        if not sp.compilable:
            if verbose:
                print("skip (fallback):", name, sp.why_not_compilable)
            continue
        src, desc = codegen.generate_stage(sp, source="mem", sink="mem")
        _stage_compile(glib, src, desc)
        n += 1
        if verbose:
            print("precompiled:", name)
    n += precompile_csv(verbose=verbose)
    n += precompile_tests(verbose=verbose)
    return n


def csv_used_cols(data, ops, columns=None, header=None, delimiter=b",",
                  null_values=None):
    """The stage's projection-pushdown column set (plan._used_source_columns)
    for feeding the oracle's used_cols — the pushdown decision is optimizer
    metadata (LogicalOptimizer selectionPushdown), an INPUT to the semantics,
    like the delimiter."""
    from tuplex_amd import csvio, plan
    sample = data[:1 << 20]
    nl = sample.rfind(b"\n")
    if nl >= 0:
        sample = sample[:nl + 1]
    _h, names, col_types = csvio.sniff(sample, null_values or [""], 0.9,
                                       header, columns, delimiter)
    sp = plan.build_stage(col_types, names, ops)
    return sp.used_source_cols
