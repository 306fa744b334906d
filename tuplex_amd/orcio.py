"""ORC source — columnar device ingest.

The reference reads ORC through Apache ORC C++ into its row format
(io/src/OrcTypes.cc, physical OrcReader). MI355X-native design: the Arrow
columnar buffers (int64/float64 arrays, large_string offsets+data, null masks)
are uploaded to HBM AS-IS and the generated stage kernel loads typed values
directly — no CSV parse, no row deserialize, coalesced 64-wide loads
(codegen._load_inputs_col; C-ABI tpx_stage_execute_col). Exception rows carry
no byte payload; they are replayed host-side from the original table by row
index (same ResolveTask.cc:878 merge as every other source).
"""
import ctypes
import glob as _glob
from typing import List, Optional

import numpy as np

from . import plan, resolve
from . import ttypes as T
from .engine import CollectOutcome, GpuLib, execute_stage_col, _agg_row, \
    _unwrap_row

try:
    import pyarrow as pa
    import pyarrow.compute as pc
    import pyarrow.orc as paorc
except ImportError:  # pragma: no cover
    pa = None


def _arrow_base(t):
    if pa.types.is_integer(t):
        return T.I64
    if pa.types.is_floating(t):
        return T.F64
    if pa.types.is_boolean(t):
        return T.BOOL
    if pa.types.is_string(t) or pa.types.is_large_string(t):
        return T.STR
    return None


def _target_type(base):
    return {T.I64: pa.int64(), T.F64: pa.float64(), T.BOOL: pa.bool_(),
            T.STR: pa.large_string()}[base]


def _table_rows(tab, names, lo=0, hi=None):
    """Rows of the table as python tuples (schema order); scalar for 1 col."""
    part = tab.slice(lo, (hi - lo) if hi is not None else None)
    cols = [part.column(n).to_pylist() for n in names]
    if len(names) == 1:
        return cols[0]
    return list(zip(*cols))


def run_orc(ctx, src, logical_ops, sink=None, keep_exceptions=False):
    out = CollectOutcome()
    if pa is None:
        raise RuntimeError("pyarrow is required for the ORC source")
    paths = sorted(_glob.glob(src.pattern)) or [src.pattern]
    tabs = [paorc.read_table(p) for p in paths]
    tab = pa.concat_tables(tabs) if len(tabs) > 1 else tabs[0]
    names = src.columns or list(tab.schema.names)
    if src.columns:
        tab = tab.select(src.columns)

    # schema map (data-driven Option-ness, like the majority-type sniffer)
    col_types: List = []
    bases = []
    for f in tab.schema:
        b = _arrow_base(f.type)
        bases.append(b)
    if any(b is None for b in bases):
        return _fallback_all(out, tab, names, logical_ops,
                             "unsupported ORC column type")
    tab = tab.cast(pa.schema([pa.field(n, _target_type(b))
                              for n, b in zip(names, bases)]))
    for k, b in enumerate(bases):
        col_types.append(("opt", b) if tab.column(k).null_count > 0 else b)

    sp = plan.build_stage(col_types, names, logical_ops)
    if not sp.compilable:
        return _fallback_all(out, tab, names, logical_ops,
                             sp.why_not_compilable,
                             out_cols=sp.output_columns)

    glib = GpuLib.get()
    if glib.device_count() == 0:
        raise RuntimeError("no HIP device visible — the normal-case path runs "
                           "only on GPU (no CPU fallback by design)")

    used = sp.used_source_cols
    n_rows = tab.num_rows
    dev_ptrs = []
    total_bytes = 0

    def up(byts):
        nonlocal total_bytes
        p = glib.lib.tpx_dev_alloc(max(len(byts), 1))
        if not p:
            raise RuntimeError("device alloc failed: " + glib.err())
        buf = (ctypes.c_uint8 * max(len(byts), 1)).from_buffer_copy(
            byts or b"\0")
        rc = glib.lib.tpx_dev_upload(p, buf, max(len(byts), 1))
        if rc != 0:
            raise RuntimeError(glib.err())
        dev_ptrs.append(p)
        total_bytes += len(byts)
        return p

    slots = []
    try:
        for k, t in enumerate(col_types):
            if used is not None and k not in used:
                slots += [0, 0, 0]
                continue
            base = T.deopt(t)
            arr = tab.column(k).combine_chunks()
            null_ptr = 0
            if T.is_opt(t):
                mask = pc.is_null(arr).to_numpy(zero_copy_only=False)
                null_ptr = up(np.ascontiguousarray(mask, dtype=np.uint8)
                              .tobytes())
            if base == T.STR:
                arr = arr.fill_null("")
                assert arr.offset == 0, "combine_chunks must yield offset 0"
                bufs = arr.buffers()  # [validity, offsets(i64), data]
                offs = np.frombuffer(bufs[1], dtype=np.int64,
                                     count=len(arr) + 1)
                data = bufs[2].to_pybytes() if bufs[2] is not None else b""
                slots += [up(offs.tobytes()), up(data), null_ptr]
            elif base == T.I64:
                v = arr.fill_null(0).to_numpy(zero_copy_only=False)
                slots += [up(np.ascontiguousarray(v, dtype=np.int64)
                             .tobytes()), 0, null_ptr]
            elif base == T.F64:
                v = arr.fill_null(0.0).to_numpy(zero_copy_only=False)
                slots += [up(np.ascontiguousarray(v, dtype=np.float64)
                             .tobytes()), 0, null_ptr]
            else:  # bool
                v = arr.fill_null(False).to_numpy(zero_copy_only=False)
                slots += [up(np.ascontiguousarray(v, dtype=np.uint8)
                             .tobytes()), 0, null_ptr]

        er = execute_stage_col(sp, slots, n_rows, total_bytes)
    finally:
        for p in dev_ptrs:
            glib.lib.tpx_dev_free(p)

    out.mode = "gpu"
    out.metrics = er.metrics
    scalar_input = len(col_types) == 1
    agg_cols = sp.output_columns

    def replay_iter(idxs):
        for i in sorted(idxs):
            row = _table_rows(tab, names, i, i + 1)[0]
            r = resolve.replay_row(row, logical_ops, names, scalar_input)
            yield i, r

    if sp.agg_expr is not None:
        exc_idx = [r for (r, _, _) in er.exceptions]

        def replay_rows():
            for _i, r in replay_iter(exc_idx):
                if r[0] == "row":
                    yield r[1]
                elif r[0] == "exc":
                    nm = type(r[1]).__name__
                    out.exception_counts[nm] = \
                        out.exception_counts.get(nm, 0) + 1
                elif r[0] == "rows":
                    for v2 in r[1]:
                        yield v2
                    for e in r[2]:
                        nm = type(e).__name__
                        out.exception_counts[nm] = \
                            out.exception_counts.get(nm, 0) + 1

        if sp.agg_unique and sp.agg_key_idx is not None:
            keys = {row[0] for row in er.rows}
            for row in replay_rows():
                keys.add(row if not isinstance(row, tuple) else row[0])
            out.rows = list(keys)
        elif sp.agg_key_idx is not None:
            opid_a, combine_fn, agg_fn, initial, key_cols = sp.aggregate
            table = {row[0]: initial + row[1] for row in er.rows}
            ki = agg_cols.index(key_cols[0])
            for row in replay_rows():
                rt = row if isinstance(row, tuple) else (row,)
                table[rt[ki]] = agg_fn(table.get(rt[ki], initial),
                                       _agg_row(row, agg_cols))
            out.rows = [(k, v) for k, v in table.items()]
        else:
            opid_a, combine_fn, agg_fn, initial = sp.aggregate
            acc = initial + er.rows[0][0]
            for row in replay_rows():
                acc = agg_fn(acc, _agg_row(row, agg_cols))
            out.rows = [acc]
        return out

    results = {}
    for row, local in zip(er.rows, er.row_indices):
        results.setdefault(local, []).append(_unwrap_row(row))
    for i, r in replay_iter([r for (r, _, _) in er.exceptions]):
        if r[0] == "row":
            results[i] = [r[1]]
        elif r[0] == "exc":
            if keep_exceptions:
                out.pending.append((i, _table_rows(tab, names, i, i + 1)[0]))
                continue
            nm = type(r[1]).__name__
            out.exception_counts[nm] = out.exception_counts.get(nm, 0) + 1
        elif r[0] == "rows":
            results[i] = r[1]
            for e in r[2]:
                nm = type(e).__name__
                out.exception_counts[nm] = out.exception_counts.get(nm, 0) + 1
    merged = [v for i in sorted(results) for v in results[i]]
    from .engine import finalize_merged
    out.rows = finalize_merged(merged, logical_ops, sp.output_columns)
    if keep_exceptions:
        out.row_keys = [i for i in sorted(results) for _ in results[i]]
        if len(out.row_keys) != len(out.rows):
            out.row_keys = None

        def _orc_replayer(payload, ops, _names=names,
                          _scalar=scalar_input):
            return resolve.replay_row(payload, ops, _names, _scalar)
        out.pending_replayer = _orc_replayer
    return out


def _fallback_all(out, tab, names, logical_ops, why, out_cols=None):
    out.mode = "fallback"
    out.fallback_reason = why
    scalar_input = len(names) == 1
    rows = _table_rows(tab, names)
    results = {}
    for i, v in enumerate(rows):
        r = resolve.replay_row(v, logical_ops, names, scalar_input)
        if r[0] == "row":
            results[i] = [r[1]]
        elif r[0] == "exc":
            nm = type(r[1]).__name__
            out.exception_counts[nm] = out.exception_counts.get(nm, 0) + 1
        elif r[0] == "rows":
            results[i] = r[1]
            for e in r[2]:
                nm = type(e).__name__
                out.exception_counts[nm] = out.exception_counts.get(nm, 0) + 1
    from .engine import finalize_merged, output_columns_of
    if out_cols is None:
        out_cols = output_columns_of(names, logical_ops)
    out.rows = finalize_merged([v for i in sorted(results)
                                for v in results[i]], logical_ops, out_cols)
    return out
