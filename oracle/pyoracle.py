"""ORACLE (TEST INFRASTRUCTURE ONLY) — CPU restatement of Tuplex's TransformStage
semantics, used to check the GPU engine's results. See oracle/__init__.py for the
usage restriction.

What it restates (reference file:line):
 - parallelize normal-case type inference + fallback rows:
   tuplex/python/src/PythonContext.cc:1023 inferType, :964 buildRowTypeFromSamples,
   fallback-row detection :50-123 (non-conforming rows become PYTHON_PARALLELIZE
   exception rows replayed on the interpreter).
 - per-row fast-path pipeline semantics: core/src/physical/PipelineBuilder.cc
   operator chain (map PipelineBuilder.h:310, filter :320, mapColumn :331,
   withColumn :347) with the compiled-path int()/float() parse semantics of
   runtime/src/Runtime.cc:319-383 (fast_atoi64 / fast_atod wrappers with
   python-whitespace trim) over utils/src/StringUtils.cc:22 fast_atoi64 /
   :71 fast_atod (restated verbatim below, including their quirks: "-" parses to 0,
   digit-by-digit double accumulation, exponent clamp at 308).
 - exception rows + resolve semantics: LocalBackend.cc:963-1085,
   ResolveTask.cc:389 processExceptionRow (1. true exceptions stay, 2. resolver
   functor, 3. interpreter replay) and :878 executeInOrder (merge in original row
   order). Resolvers/ignores attach to the operator immediately preceding them.
 - aggregate: thread-local fold + combine at stage end (TransformTask.cc:210-260,
   LocalBackend.cc:1180-1207).

Pinned by golden vectors transcribed from the reference's own tests in
tests/golden/*.json (each citing its reference test file:line).
"""
import math
import types
from typing import Any, Callable, List, Optional, Tuple

PY_WHITESPACE = " \t\n\r\x0b\x0c"


# ---------------------------------------------------------------------------------
# fast_atoi64 / fast_atod restatements (StringUtils.cc:22 / :71 + Runtime.cc:319
# whitespace-trimming wrappers). Return (ok, value).
# ---------------------------------------------------------------------------------

def fast_atoi64(s: str) -> Tuple[bool, int]:
    if len(s) == 0:
        return False, 0  # NULLERROR
    i, n = 0, len(s)
    neg = False
    if s[0] == "-":
        neg = True
        i += 1
    x = 0
    while i < n and "0" <= s[i] <= "9":
        x = x * 10 + (ord(s[i]) - 48)
        i += 1
    if i != n:
        return False, 0  # I64PARSE_ERROR
    if neg:
        x = -x
    # compiled path wraps to int64 (LLVM i64 arithmetic)
    x &= (1 << 64) - 1
    if x >= 1 << 63:
        x -= 1 << 64
    return True, x


def fast_atod(s: str) -> Tuple[bool, float]:
    if len(s) == 0:
        return False, 0.0
    p, n = 0, len(s)
    sign = 1.0
    if s[p] == "-":
        sign = -1.0
        p += 1
    elif s[p] == "+":
        p += 1
    value = 0.0
    while p < n and "0" <= s[p] <= "9":
        value = 10.0 * value + (ord(s[p]) - 48)
        p += 1
    if p < n and s[p] == ".":
        pow10 = 10.0
        p += 1
        while p < n and "0" <= s[p] <= "9":
            value += (ord(s[p]) - 48) / pow10
            pow10 *= 10.0
            p += 1
    frac = False
    scale = 1.0
    if p < n and s[p] in "eE":
        p += 1
        if p < n and s[p] == "-":
            frac = True
            p += 1
        elif p < n and s[p] == "+":
            p += 1
        exponent = 0
        while p < n and "0" <= s[p] <= "9":
            exponent = exponent * 10 + (ord(s[p]) - 48)
            p += 1
        if exponent > 308:
            exponent = 308
        while exponent >= 50:
            scale *= 1e50
            exponent -= 50
        while exponent >= 8:
            scale *= 1e8
            exponent -= 8
        while exponent > 0:
            scale *= 10.0
            exponent -= 1
    nanmatch = infmatch = 0
    if p == 0:
        t = "nan"
        while nanmatch < 3 and p < n and s[p] in (t[nanmatch], t[nanmatch].upper()):
            p += 1
            nanmatch += 1
    if p == 0:
        t = "infinity"
        while infmatch < 8 and p < n and s[p] in (t[infmatch], t[infmatch].upper()):
            p += 1
            infmatch += 1
    if p != n:
        return False, 0.0
    if nanmatch == 3:
        return True, math.nan
    if infmatch in (3, 8):
        return True, sign * math.inf
    return True, sign * (value / scale if frac else value * scale)


def _trim_ws(s: str) -> str:
    return s.strip(PY_WHITESPACE)


def ref_int(x):
    """Compiled-path int() — Runtime.cc:319 wrapper (trim python whitespace, then
    fast_atoi64). Non-str falls back to CPython int()."""
    if isinstance(x, str):
        ok, v = fast_atoi64(_trim_ws(x))
        if not ok:
            raise ValueError("invalid literal for int(): %r" % x)
        return v
    return int(x)


def ref_float(x):
    if isinstance(x, str):
        ok, v = fast_atod(_trim_ws(x))
        if not ok:
            raise ValueError("could not convert string to float: %r" % x)
        return v
    return float(x)


def with_ref_builtins(fn: Callable) -> Callable:
    """Return a copy of fn whose globals shadow int/float with the compiled-path
    parse semantics (what the reference's generated fast path uses)."""
    g = dict(fn.__globals__)
    g["int"] = ref_int
    g["float"] = ref_float
    new = types.FunctionType(fn.__code__, g, fn.__name__, fn.__defaults__,
                             fn.__closure__)
    return new


# ---------------------------------------------------------------------------------
# type inference (PythonContext.cc:964/:1023 restatement, value-level)
# ---------------------------------------------------------------------------------

def _vtype(v):
    if v is None:
        return "null"
    if isinstance(v, bool):
        return "bool"
    if isinstance(v, int):
        return "i64"
    if isinstance(v, float):
        return "f64"
    if isinstance(v, str):
        return "str"
    if isinstance(v, tuple):
        return ("tuple", tuple(_vtype(x) for x in v))
    return ("pyobject", type(v).__name__)


def _unify(a, b):
    if a == b:
        return a
    if a == "null":
        return b if isinstance(b, tuple) and b[0] == "opt" else ("opt", b) if b != "null" else "null"
    if b == "null":
        return a if isinstance(a, tuple) and a[0] == "opt" else ("opt", a)
    ao = isinstance(a, tuple) and a[0] == "opt"
    bo = isinstance(b, tuple) and b[0] == "opt"
    if ao or bo:
        u = _unify(a[1] if ao else a, b[1] if bo else b)
        if u is None:
            return None
        return u if isinstance(u, tuple) and u[0] == "opt" else ("opt", u)
    if isinstance(a, tuple) and a[0] == "tuple" and isinstance(b, tuple) and b[0] == "tuple":
        if len(a[1]) != len(b[1]):
            return None
        out = []
        for x, y in zip(a[1], b[1]):
            u = _unify(x, y)
            if u is None:
                return None
            out.append(u)
        return ("tuple", tuple(out))
    return None


def infer_majority_type(values, optional_threshold=0.7, sample_limit=10000):
    n = min(len(values), sample_limit)
    if n == 0:
        return ("tuple", ("f64",))
    counts = {}
    for i in range(n):
        t = _vtype(values[i])
        if t in counts:
            counts[t] += 1
        else:
            merged = False
            for k in list(counts):
                u = _unify(k, t)
                if u is not None:
                    c = counts.pop(k)
                    counts[u] = counts.get(u, 0) + c + 1
                    merged = True
                    break
            if not merged:
                counts[t] = 1
    maj, maj_c = None, -1
    maj_tup, maj_tup_c = None, -1
    for t, c in counts.items():
        if c > maj_c:
            maj, maj_c = t, c
        if isinstance(t, tuple) and t[0] == "tuple" and c > maj_tup_c:
            maj_tup, maj_tup_c = t, c
    if maj_tup is not None:
        sup = maj_tup
        num = 0
        for t, c in counts.items():
            u = _unify(t, sup)
            if u is not None:
                sup = u
                num += c
        frac = (num - counts.get(maj_tup, 0)) / float(n)
        if num > maj_c and (1 - optional_threshold) < frac < optional_threshold:
            maj = sup
    if maj not in (None, "null") and "null" in counts:
        nf = counts["null"] / float(n)
        if (1 - optional_threshold) < nf < optional_threshold:
            maj = maj if isinstance(maj, tuple) and maj[0] == "opt" else ("opt", maj)
    return maj


def conforms(v, t) -> bool:
    if isinstance(t, tuple) and t[0] == "opt":
        return v is None or conforms(v, t[1])
    if t == "null":
        return v is None
    if t == "bool":
        return isinstance(v, bool)
    if t == "i64":
        return isinstance(v, int) and not isinstance(v, bool) and -(2**63) <= v < 2**63
    if t == "f64":
        return isinstance(v, float)
    if t == "str":
        return isinstance(v, str)
    if isinstance(t, tuple) and t[0] == "tuple":
        return (isinstance(v, tuple) and len(v) == len(t[1])
                and all(conforms(x, p) for x, p in zip(v, t[1])))
    return False


# ---------------------------------------------------------------------------------
# pipeline execution (dual-mode: fast path w/ ref builtins; interpreter replay)
# ---------------------------------------------------------------------------------

class _RowExc(Exception):
    def __init__(self, exc, op_index):
        self.exc = exc
        self.op_index = op_index


def _n_params(fn):
    return fn.__code__.co_argcount


def _call_udf(fn, row, columns, fast: bool):
    """Call UDF the way the generated pipeline does: multi-param UDFs get the row
    unpacked; single-param UDFs get the bare value (1-col rows) or the tuple/dict."""
    f = with_ref_builtins(fn) if fast else fn
    if isinstance(row, tuple):
        if _n_params(fn) == len(row) and _n_params(fn) > 1:
            return f(*row)
        if columns:
            return f(dict(zip(columns, row)))
        return f(row)
    else:
        if columns and _n_params(fn) == 1:
            # single scalar col with a name: pass dict if fn subscripts? The
            # reference passes the bare value for single-col rows.
            return f(row)
        return f(row)


def _as_row(v):
    return v if isinstance(v, tuple) else (v,)


def run_pipeline(data: List[Any], ops: List[tuple], columns: Optional[List[str]] = None,
                 merge_rows_in_order: bool = True):
    """Execute a pipeline over `data` with the reference's dual-mode semantics.

    ops entries:
      ("map", fn) ("filter", fn) ("withColumn", col, fn) ("mapColumn", col, fn)
      ("selectColumns", [cols]) ("renameColumn", old, new)
      ("resolve", ExcClass, fn) ("ignore", ExcClass)
      ("aggregate", combine_fn, agg_fn, initial)
    Returns dict: output (list), exception_counts ({name: count}).
    """
    maj = infer_majority_type(data)
    # split normal vs fallback rows (PythonContext fallback detection)
    normal, fallback = [], []  # (orig_idx, value)
    for i, v in enumerate(data):
        (normal if conforms(v, maj) else fallback).append((i, v))

    # aggregate stages handled as a fold at the end
    agg = None
    aggby = None
    uniq = False
    row_ops = []
    for op in ops:
        if op[0] == "aggregate":
            agg = op
        elif op[0] == "aggregateByKey":
            aggby = op
        elif op[0] == "unique":
            uniq = True
        else:
            row_ops.append(op)

    results = []  # (orig_idx, row) for merge-in-order
    exc_counts = {}

    def record_exc(e):
        name = type(e).__name__
        exc_counts[name] = exc_counts.get(name, 0) + 1

    stream = sorted(normal + fallback) if merge_rows_in_order else normal + fallback
    fb_idx = {i for i, _ in fallback}
    for idx, v in stream:
        fast = idx not in fb_idx
        r = process_row(v, row_ops, columns, fast)
        if r[0] == "row":
            results.append((idx, r[1]))
        elif r[0] == "exc":
            record_exc(r[1])
        elif r[0] == "rows":
            for v2 in r[1]:
                results.append((idx, v2))
            for e in r[2]:
                record_exc(e)

    if merge_rows_in_order:
        results.sort(key=lambda t: t[0])
    out = [v for _, v in results]

    if agg is not None:
        _, combine_fn, agg_fn, initial = agg
        a = initial
        for v in out:
            a = agg_fn(a, _agg_row(v, columns))
        out = [a]
    elif aggby is not None:
        out = aggregate_by_key(out, aggby, columns)
    elif uniq:
        out = list(dict.fromkeys(out))  # hashmap-sink dedup (order unpinned)

    return {"output": out, "exception_counts": exc_counts}


def aggregate_by_key(rows, aggby, columns):
    """hashmap-sink fold (hashmap.cc / createFinalHashmap LocalBackend.cc:2219).
    Output order is parity-unpinned (multiset compare, test_aggregates.py:53)."""
    _, combine_fn, agg_fn, initial, key_cols = aggby
    key_idx = [columns.index(c) for c in key_cols]
    table = {}
    for v in rows:
        row = v if isinstance(v, tuple) else (v,)
        k = tuple(row[i] for i in key_idx)
        acc = table.get(k, initial)
        table[k] = agg_fn(acc, _agg_row(v, columns))
    return [k + ((val,) if not isinstance(val, tuple) else val)
            for k, val in table.items()]


def _agg_row(v, columns):
    """agg fn's row arg follows the same dict convention as other UDFs."""
    if columns and isinstance(v, tuple):
        return dict(zip(columns, v))
    return v


def process_row(value, row_ops, columns, fast):
    """Run one row through row_ops with the reference's dual-mode semantics.
    Returns ("row", value) | ("drop",) | ("exc", exception) | ("rows",
    [v...], [exc...]) (a join with duplicate build keys expands 1:N; each
    joined row continues the remaining ops independently).
    fast=True: compiled-path int()/float() semantics; False: interpreter."""
    return _process_from(value, list(columns) if columns else None, row_ops,
                         0, fast)


def _process_from(cur, cols, row_ops, k, fast):
    while k < len(row_ops):
        op = row_ops[k]
        kind = op[0]
        if kind in ("resolve", "ignore"):
            k += 1
            continue
        if kind == "join":
            matches = _join_matches(op, cur, cols)
            if len(matches) > 1:
                rows_out, excs = [], []
                for m in matches:
                    row2, cols2 = _join_build_row(op, cur, cols, m)
                    r = _process_from(row2, cols2, row_ops, k + 1, fast)
                    if r[0] == "row":
                        rows_out.append(r[1])
                    elif r[0] == "exc":
                        excs.append(r[1])
                    elif r[0] == "rows":
                        rows_out.extend(r[1])
                        excs.extend(r[2])
                return ("rows", rows_out, excs)
        try:
            cur, cols, dropped = _apply_op(op, cur, cols, fast)
            if dropped:
                return ("drop",)
        except Exception as e:  # noqa: BLE001 — row-level exception machinery
            j = k + 1  # scan following resolve/ignore ops (ResolveTask.cc:389)
            while j < len(row_ops) and row_ops[j][0] in ("resolve", "ignore"):
                rkind = row_ops[j][0]
                rcls = row_ops[j][1]
                if isinstance(e, rcls):
                    if rkind == "ignore":
                        return ("drop",)
                    try:
                        cur, cols, dropped = _apply_resolver(op, row_ops[j][2],
                                                             cur, cols, fast)
                    except Exception as e2:  # resolver itself raised
                        return ("exc", e2)
                    if dropped:
                        return ("drop",)
                    break
                j += 1
            else:
                return ("exc", e)
            k = j  # continue after the matched resolver
        k += 1
    return ("row", cur)


def _apply_op(op, cur, cols, fast):
    kind = op[0]
    if kind == "map":
        v = _call_udf(op[1], cur, cols, fast)
        return v, None, False  # map drops column names
    if kind == "filter":
        keep = _call_udf(op[1], cur, cols, fast)
        return cur, cols, not keep
    if kind == "withColumn":
        col, fn = op[1], op[2]
        row = _as_row(cur)
        cols2 = list(cols) if cols else ["column%d" % i for i in range(len(row))]
        v = _call_udf(fn, row, cols2, fast)
        if col in cols2:
            i = cols2.index(col)
            row = row[:i] + (v,) + row[i + 1:]
        else:
            cols2 = cols2 + [col]
            row = row + (v,)
        return row, cols2, False
    if kind == "mapColumn":
        col, fn = op[1], op[2]
        row = _as_row(cur)
        i = cols.index(col)
        f = with_ref_builtins(fn) if fast else fn
        v = f(row[i])
        return row[:i] + (v,) + row[i + 1:], cols, False
    if kind == "selectColumns":
        sel = op[1]
        row = _as_row(cur)
        idxs = [cols.index(c) if isinstance(c, str) else c for c in sel]
        names = [cols[i] for i in idxs]
        out = tuple(row[i] for i in idxs)
        return (out if len(out) > 1 else out[0]), names, False
    if kind == "join":
        return _apply_join(op, cur, cols)
    if kind == "renameColumn":
        old, new = op[1], op[2]
        cols2 = [new if c == old else c for c in cols]
        return cur, cols2, False
    raise ValueError("unknown op %r" % (kind,))


def _apply_resolver(op, resolver, cur, cols, fast):
    """Resolver output replaces the failing operator's UDF output
    (ResolveTask.cc:436 compiled resolve_f semantics)."""
    kind = op[0]
    if kind == "map":
        v = _call_udf(resolver, cur, cols, fast)
        return v, None, False
    if kind == "filter":
        keep = _call_udf(resolver, cur, cols, fast)
        return cur, cols, not keep
    if kind == "withColumn":
        col = op[1]
        return _apply_op(("withColumn", col, resolver), cur, cols, fast)
    if kind == "mapColumn":
        col = op[1]
        return _apply_op(("mapColumn", col, resolver), cur, cols, fast)
    raise ValueError("resolver after %r unsupported" % (kind,))


_JOIN_TABLES = {}


def _join_table(op):
    # keyed by id() but the op itself is pinned in the entry: a freed tuple's
    # address can be reused, which would silently serve the wrong build table
    ent = _JOIN_TABLES.get(id(op))
    if ent is None or ent[0] is not op:
        rrows, rcols, rk = op[1], op[2], op[4]
        rki = rcols.index(rk)
        t = {}
        for r in rrows:  # bucket lists in build order
            t.setdefault(r[rki], []).append(r)
        _JOIN_TABLES[id(op)] = (op, t)
        return t
    return ent[1]


def _join_matches(op, cur, cols):
    _, rrows, rcols, lk, rk, how, lp, ls, rp, rs = op
    row = _as_row(cur)
    cols2 = list(cols) if cols else None
    if not cols2 or lk not in cols2:
        raise ValueError("join: unknown left key column %r" % lk)
    key = row[cols2.index(lk)]
    return [] if key is None else _join_table(op).get(key, [])


def _join_build_row(op, cur, cols, m):
    _, rrows, rcols, lk, rk, how, lp, ls, rp, rs = op
    row = _as_row(cur)
    cols2 = list(cols) if cols else None
    lki = cols2.index(lk)
    rki = rcols.index(rk)
    right_vals = tuple((None if m is None else m[j])
                      for j in range(len(rcols)) if j != rki)
    out = tuple(v for i, v in enumerate(row) if i != lki) + (row[lki],) + \
        right_vals
    out_cols = ([lp + c + ls for c in cols2 if c != lk] + [lp + lk + ls] +
                [rp + c + rs for c in rcols if c != rk])
    return out, out_cols


def _apply_join(op, cur, cols):
    """0/1-match step (multi-match forks handled in _process_from);
    logical/JoinOperator.cc:164 layout."""
    how = op[5]
    matches = _join_matches(op, cur, cols)
    if not matches and how == "inner":
        return cur, cols, True
    out, out_cols = _join_build_row(op, cur, cols,
                                    matches[0] if matches else None)
    return out, out_cols, False
