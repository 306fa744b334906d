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


def _call_udf(fn, row, columns):
    if isinstance(row, tuple):
        if _n_params(fn) == len(row) and _n_params(fn) > 1:
            return fn(*row)
        if columns:
            return fn(dict(zip(columns, row)))
        return fn(row)
    return fn(row)


def _as_row(v):
    return v if isinstance(v, tuple) else (v,)


def replay_row(value: Any, logical_ops: List[tuple], columns: Optional[List[str]],
               scalar_input: bool):
    """Replay one row through the op chain. Returns ("row", v) | ("drop",) |
    ("exc", exception)."""
    cur = value
    cols = list(columns) if columns else None
    row_ops = [op for op in logical_ops
               if op[0] not in ("aggregate", "aggregateByKey", "unique")]
    k = 0
    while k < len(row_ops):
        op = row_ops[k]
        kind = op[0]
        if kind in ("resolve", "ignore"):
            k += 1
            continue
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
        t = {r[rki]: r for r in rrows}
        _JOIN_TABLES[id(op)] = (op, t)
        return t
    return ent[1]


def _apply_join(op, cur, cols):
    """Inner/left hash join against a materialized build side (single-match:
    build keys are unique — logical/JoinOperator.cc:164 output layout
    | left cols except key | key | right cols except key |; the key keeps the
    left name and never nulls; left join nulls the right columns)."""
    _, rrows, rcols, lk, rk, how, lp, ls, rp, rs = op
    row = _as_row(cur)
    cols2 = list(cols) if cols else None
    if not cols2 or lk not in cols2:
        raise ValueError("join: unknown left key column %r" % lk)
    lki = cols2.index(lk)
    rki = rcols.index(rk)
    key = row[lki]
    m = None if key is None else _join_table(op).get(key)
    if m is None and how == "inner":
        return cur, cols2, True
    right_vals = tuple((None if m is None else m[j])
                      for j in range(len(rcols)) if j != rki)
    out = tuple(v for i, v in enumerate(row) if i != lki) + (key,) + right_vals
    out_cols = ([lp + c + ls for c in cols2 if c != lk] + [lp + lk + ls] +
                [rp + c + rs for c in rcols if c != rk])
    return out, out_cols, False
