"""Host interpreter replay of exception/fallback rows — the product's slow path.

The reference keeps exception handling on the host: ResolveTask.cc:389
processExceptionRow (try compiled resolver, else CPython interpreter via the
generated pure-Python pipeline, PythonPipelineBuilder.cc) and :878 executeInOrder
(in-order merge). Here every diverted row (fallback row, normal-case violation,
user exception, heap overflow) is replayed through the full operator chain with
CPython semantics, with resolver/ignore handlers applied at the operator that
raised — per north_star this path stays CPU-only and semantically identical.
"""
from typing import Any, List, Optional


def _n_params(fn):
    return fn.__code__.co_argcount


class RowView(dict):
    """Row argument for 1-param UDFs over named rows: a dict keyed by column
    name that ALSO answers positional integer (and slice) subscripts, because
    the compiled path maps x[0] on a named multi-column row to column 0
    (udf/compile.py subscript rule) — host replay must accept the same access
    patterns or diverted rows become spurious exceptions."""

    __slots__ = ("_values",)

    def __init__(self, values, columns):
        super().__init__(zip(columns, values))
        self._values = tuple(values)

    def __getitem__(self, k):
        if isinstance(k, bool) or not isinstance(k, (int, slice)):
            return dict.__getitem__(self, k)
        return self._values[k]


def _call_udf(fn, row, columns):
    if isinstance(row, tuple):
        if _n_params(fn) == len(row) and _n_params(fn) > 1:
            return fn(*row)
        if columns:
            return fn(RowView(row, columns))
        return fn(row)
    return fn(row)


def _as_row(v):
    return v if isinstance(v, tuple) else (v,)


def replay_row(value: Any, logical_ops: List[tuple], columns: Optional[List[str]],
               scalar_input: bool):
    """Replay one row through the op chain. Returns ("row", v) | ("drop",) |
    ("exc", exception) | ("rows", [v...], [exc...]) — the last when a join
    with duplicate build keys expands the row 1:N (each joined row continues
    the remaining ops independently, like the reference's per-row pipeline
    over hashmap bucket entries)."""
    row_ops = [op for op in logical_ops
               if op[0] not in ("aggregate", "aggregateByKey", "unique")]
    return _replay_from(value, list(columns) if columns else None, row_ops, 0)


def _unwrap1(v):
    """Arity-1 rows collect as scalars (the GPU merges unwrap 1-tuples the
    same way; PythonDataSet.cc fast conversion does too)."""
    return v[0] if isinstance(v, tuple) and len(v) == 1 else v


def _replay_from(cur, cols, row_ops, k):
    r = _replay_from_raw(cur, cols, row_ops, k)
    if r[0] == "row":
        return ("row", _unwrap1(r[1]))
    if r[0] == "rows":
        return ("rows", [_unwrap1(v) for v in r[1]], r[2])
    return r


def _replay_from_raw(cur, cols, row_ops, k):
    while k < len(row_ops):
        op = row_ops[k]
        kind = op[0]
        if kind in ("resolve", "ignore"):
            k += 1
            continue
        if kind == "join":
            matches = _join_matches(op, cur, cols)
            if len(matches) > 1:  # 1:N fork — remaining ops run per joined row
                rows_out, excs = [], []
                for m in matches:
                    row2, cols2 = _join_build_row(op, cur, cols, m)
                    r = _replay_from(row2, cols2, row_ops, k + 1)
                    if r[0] == "row":
                        rows_out.append(r[1])
                    elif r[0] == "exc":
                        excs.append(r[1])
                    elif r[0] == "rows":
                        rows_out.extend(r[1])
                        excs.extend(r[2])
                return ("rows", rows_out, excs)
            # 0/1 match: the plain single-row path below handles it
        try:
            cur, cols, dropped = _apply_op(op, cur, cols)
            if dropped:
                return ("drop",)
        except Exception as e:  # noqa: BLE001 — row-level dual-mode machinery
            j = k + 1
            handled = False
            while j < len(row_ops) and row_ops[j][0] in ("resolve", "ignore"):
                rkind, rcls = row_ops[j][0], row_ops[j][1]
                if isinstance(e, rcls):
                    if rkind == "ignore":
                        return ("drop",)
                    try:
                        cur, cols, dropped = _apply_resolver(op, row_ops[j][2],
                                                             cur, cols)
                    except Exception as e2:  # resolver raised
                        return ("exc", e2)
                    if dropped:
                        return ("drop",)
                    handled = True
                    break
                j += 1
            if not handled:
                return ("exc", e)
            k = j
        k += 1
    return ("row", cur)


def _apply_op(op, cur, cols):
    kind = op[0]
    if kind == "map":
        return _call_udf(op[1], cur, cols), None, False
    if kind == "filter":
        return cur, cols, not _call_udf(op[1], cur, cols)
    if kind == "withColumn":
        col, fn = op[1], op[2]
        row = _as_row(cur)
        cols2 = list(cols) if cols else ["column%d" % i for i in range(len(row))]
        v = _call_udf(fn, row, cols2)
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
        return row[:i] + (fn(row[i]),) + row[i + 1:], cols, False
    if kind == "selectColumns":
        sel = op[1]
        row = _as_row(cur)
        idxs = [cols.index(c) if isinstance(c, str) else c for c in sel]
        names = [c if isinstance(c, str) else (cols[c] if cols else "column%d" % c)
                 for c in sel]
        out = tuple(row[i] for i in idxs)
        return (out if len(out) > 1 else out[0]), names, False
    if kind == "join":
        return _apply_join(op, cur, cols)
    if kind == "renameColumn":
        old, new = op[1], op[2]
        return cur, [new if c == old else c for c in cols], False
    raise ValueError("unknown op %r" % (kind,))


def _apply_resolver(op, resolver, cur, cols):
    kind = op[0]
    if kind == "map":
        return _call_udf(resolver, cur, cols), None, False
    if kind == "filter":
        return cur, cols, not _call_udf(resolver, cur, cols)
    if kind in ("withColumn", "mapColumn"):
        return _apply_op((kind, op[1], resolver), cur, cols)
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
        for r in rrows:  # bucket LISTS in build order (hashmap.cc chains)
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
    """One joined output row (m = matched right row or None for a left-join
    miss) — logical/JoinOperator.cc:164 layout: | left cols except key | key |
    right cols except key |; the key keeps the left name, prefixes/suffixes
    rename; left join nulls the right columns."""
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
    """0/1-match join step used by the plain single-row path (multi-match
    forks are handled in _replay_from before reaching here)."""
    how = op[5]
    matches = _join_matches(op, cur, cols)
    if not matches and how == "inner":
        return cur, cols, True
    out, out_cols = _join_build_row(op, cur, cols,
                                    matches[0] if matches else None)
    return out, out_cols, False
