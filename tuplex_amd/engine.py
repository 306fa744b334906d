"""Host execution engine: drives the C-ABI GPU executor and the host resolve path.

Replaces the reference's LocalBackend orchestration (LocalBackend.cc:815
executeTransformStage: compile, task fan-out, perform, exception resolution, sink)
for the single-node hot path. The normal case runs ONLY on the GPU (fails loudly if
the HIP library or a device is missing); exception/fallback rows are replayed on the
CPython interpreter (resolve.py) and merged in order — the reference's dual-mode
contract (ResolveTask.cc:702/:878).
"""
import ctypes
import os
import time
from typing import Any, List, Optional

from . import codegen, plan, rowfmt, resolve
from . import ttypes as T
from . import ec as EC

_PKG = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_PKG, "libtpx_gpu.so")
_CACHE_DIR = os.path.join(_PKG, ".kernel_cache")


class TpxResult(ctypes.Structure):
    _fields_ = [
        ("out_data", ctypes.POINTER(ctypes.c_uint8)),
        ("out_size", ctypes.c_int64),
        ("out_num_rows", ctypes.c_int64),
        ("out_row_offsets", ctypes.POINTER(ctypes.c_int64)),
        ("out_row_indices", ctypes.POINTER(ctypes.c_int64)),
        ("exc_data", ctypes.POINTER(ctypes.c_uint8)),
        ("exc_size", ctypes.c_int64),
        ("exc_num_rows", ctypes.c_int64),
        ("t_h2d_ms", ctypes.c_double),
        ("t_kernel_ms", ctypes.c_double),
        ("t_d2h_ms", ctypes.c_double),
        ("bytes_in", ctypes.c_int64),
        ("bytes_out", ctypes.c_int64),
        ("t_boundary_ms", ctypes.c_double),
        ("t_main_ms", ctypes.c_double),
        ("t_compact_ms", ctypes.c_double),
        ("t_write_ms", ctypes.c_double),
        ("in_num_rows", ctypes.c_int64),
    ]


class TpxPartition(ctypes.Structure):
    _fields_ = [
        ("data", ctypes.POINTER(ctypes.c_uint8)),
        ("size", ctypes.c_int64),
        ("num_rows", ctypes.c_int64),
        ("row_offsets", ctypes.POINTER(ctypes.c_int64)),
    ]


class GpuLib:
    _instance = None

    def __init__(self):
        if not os.path.exists(_LIB_PATH):
            raise RuntimeError(
                "libtpx_gpu.so not built — run python tuplex_amd/csrc/build.py "
                "(the GPU normal-case path has no CPU fallback by design)")
        lib = ctypes.CDLL(_LIB_PATH)
        lib.tpx_version.restype = ctypes.c_int64
        lib.tpx_device_count.restype = ctypes.c_int64
        lib.tpx_set_device.argtypes = [ctypes.c_int64]
        lib.tpx_set_device.restype = ctypes.c_int64
        lib.tpx_last_error.restype = ctypes.c_char_p
        lib.tpx_stage_compile.restype = ctypes.c_void_p
        lib.tpx_stage_compile.argtypes = [ctypes.c_char_p, ctypes.c_char_p,
                                          ctypes.c_char_p, ctypes.c_int64]
        lib.tpx_stage_free.argtypes = [ctypes.c_void_p]
        lib.tpx_stage_source.argtypes = [ctypes.c_void_p]
        lib.tpx_stage_source.restype = ctypes.c_char_p
        lib.tpx_stage_execute.restype = ctypes.c_int64
        lib.tpx_stage_execute.argtypes = [ctypes.c_void_p,
                                          ctypes.POINTER(TpxPartition),
                                          ctypes.c_int64,
                                          ctypes.POINTER(TpxResult)]
        lib.tpx_stage_execute_csv.restype = ctypes.c_int64
        lib.tpx_stage_execute_csv.argtypes = [ctypes.c_void_p,
                                              ctypes.POINTER(ctypes.c_uint8),
                                              ctypes.c_int64, ctypes.c_int64,
                                              ctypes.POINTER(TpxResult)]
        lib.tpx_result_free.argtypes = [ctypes.POINTER(TpxResult)]
        lib.tpx_dev_alloc.restype = ctypes.c_uint64
        lib.tpx_dev_alloc.argtypes = [ctypes.c_int64]
        lib.tpx_dev_upload.restype = ctypes.c_int64
        lib.tpx_dev_upload.argtypes = [ctypes.c_uint64, ctypes.c_void_p,
                                       ctypes.c_int64]
        lib.tpx_dev_free.argtypes = [ctypes.c_uint64]
        lib.tpx_pinned_alloc.restype = ctypes.c_uint64
        lib.tpx_pinned_alloc.argtypes = [ctypes.c_int64]
        lib.tpx_pinned_free.argtypes = [ctypes.c_uint64]
        lib.tpx_stage_execute_col.restype = ctypes.c_int64
        lib.tpx_stage_execute_col.argtypes = [ctypes.c_void_p,
                                              ctypes.POINTER(ctypes.c_void_p),
                                              ctypes.c_int64, ctypes.c_int64,
                                              ctypes.c_int64, ctypes.c_int64,
                                              ctypes.c_int64,
                                              ctypes.POINTER(TpxResult)]
        lib.tpx_stage_execute_csv_dev.restype = ctypes.c_int64
        lib.tpx_stage_execute_csv_dev.argtypes = [ctypes.c_void_p,
                                                  ctypes.c_uint64,
                                                  ctypes.c_int64, ctypes.c_int64,
                                                  ctypes.c_int64,
                                                  ctypes.POINTER(TpxResult)]
        self.lib = lib
        os.makedirs(_CACHE_DIR, exist_ok=True)
        self._stage_cache = {}

    @classmethod
    def get(cls):
        if cls._instance is None:
            cls._instance = GpuLib()
        return cls._instance

    def device_count(self) -> int:
        return int(self.lib.tpx_device_count())

    def err(self) -> str:
        return self.lib.tpx_last_error().decode("utf-8", "replace")

    def compile_stage(self, src: str, desc: str, compile_only=False):
        key = (src, compile_only)
        h = self._stage_cache.get(key)
        if h:
            return h
        h = self.lib.tpx_stage_compile(src.encode(), desc.encode(),
                                       _CACHE_DIR.encode(),
                                       1 if compile_only else 0)
        if not h:
            raise RuntimeError("stage compile failed: " + self.err())
        self._stage_cache[key] = h
        return h


class ExecResult:
    def __init__(self):
        self.rows: List[tuple] = []         # normal-case output rows (value tuples)
        self.row_indices: List[int] = []    # local input row index per output row
        self.exceptions = []                # [(global_row, ec, opid)]
        self.metrics = {}


def execute_stage_mem(sp: plan.StageProgram, norm_rows: List[tuple],
                      device: int = 0, partition_size: int = 32 << 20) -> ExecResult:
    """Run the fused stage on the GPU over serialized normal-case rows, split
    into partitionSize partitions (ContextOptions.cc:202; one tpx_partition per
    arena, LocalBackend.cc:491 PartitionGroup analog)."""
    glib = GpuLib.get()
    if glib.device_count() == 0:
        raise RuntimeError("no HIP device visible — the normal-case path runs only "
                           "on GPU (no CPU fallback by design)")
    glib.lib.tpx_set_device(device)
    in_row_type = T.tup(sp.input_types)
    t0 = time.perf_counter()
    buf, offs = rowfmt.serialize_partition(norm_rows, in_row_type)
    t_ser = time.perf_counter() - t0

    src, desc = codegen.generate_stage(sp, source="mem", sink="mem")
    stage = glib.compile_stage(src, desc)

    # split the serialized rows at partition_size boundaries (row-aligned)
    cbuf = (ctypes.c_uint8 * len(buf)).from_buffer_copy(buf)
    parts_meta = []  # (byte_start, row_start, row_end)
    row_start = 0
    byte_start = 8
    for r in range(len(norm_rows) + 1):
        if r == len(norm_rows) or (offs[r] - byte_start >= partition_size
                                   and r > row_start):
            if r > row_start:
                parts_meta.append((byte_start, row_start, r))
            if r < len(norm_rows):
                byte_start = offs[r]
                row_start = r
    keep_alive = []
    parts = (TpxPartition * max(len(parts_meta), 1))()
    for pi, (bs, rs, re_) in enumerate(parts_meta):
        n_rows = re_ - rs
        # per-partition offsets relative to the partition's data pointer
        rel = (ctypes.c_int64 * (n_rows + 1))(
            *[offs[r] - bs for r in range(rs, re_)], offs[re_] - bs)
        keep_alive.append(rel)
        parts[pi] = TpxPartition(
            data=ctypes.cast(ctypes.byref(cbuf, bs),
                             ctypes.POINTER(ctypes.c_uint8)),
            size=offs[re_] - bs, num_rows=n_rows,
            row_offsets=ctypes.cast(rel, ctypes.POINTER(ctypes.c_int64)))
    res = TpxResult()
    rc = glib.lib.tpx_stage_execute(stage, parts, max(len(parts_meta), 1),
                                    ctypes.byref(res))
    if rc != 0:
        raise RuntimeError("stage execute failed: " + glib.err())
    try:
        out = ExecResult()
        out_bytes = ctypes.string_at(res.out_data, res.out_size)
        out_row_type = T.tup(sp.gpu_output_types)
        _offs = ([res.out_row_offsets[i] for i in range(res.out_num_rows)]
                 if res.out_num_rows else None)
        out.rows = rowfmt.deserialize_partition(out_bytes, out_row_type,
                                                offsets=_offs)
        out.row_indices = [res.out_row_indices[i] for i in range(res.out_num_rows)] \
            if res.out_num_rows else []
        if res.exc_num_rows:
            import struct as _s
            eb = ctypes.string_at(res.exc_data, res.exc_size)
            pos = 0
            for _ in range(res.exc_num_rows):
                row, ecode, opid, size = _s.unpack_from("<4q", eb, pos)
                pos += 32 + size
                out.exceptions.append((row, ecode, opid))
        out.metrics = {
            "t_serialize_s": t_ser,
            "t_h2d_ms": res.t_h2d_ms, "t_kernel_ms": res.t_kernel_ms,
            "t_d2h_ms": res.t_d2h_ms,
            "bytes_in": res.bytes_in, "bytes_out": res.bytes_out,
        }
        return out
    finally:
        glib.lib.tpx_result_free(ctypes.byref(res))


def execute_stage_col(sp: plan.StageProgram, dev_slots, n_rows: int,
                      in_bytes: int, device: int = 0) -> ExecResult:
    """Run the fused stage over a DEVICE-resident columnar (Arrow-layout)
    table — the ORC ingest path. dev_slots: 3 device
    pointers per input column ([values-or-offsets, string-data, null-mask];
    unused columns None)."""
    glib = GpuLib.get()
    if glib.device_count() == 0:
        raise RuntimeError("no HIP device visible — the normal-case path runs only "
                           "on GPU (no CPU fallback by design)")
    glib.lib.tpx_set_device(device)
    src, desc = codegen.generate_stage(sp, source="col", sink="mem")
    stage = glib.compile_stage(src, desc)
    arr = (ctypes.c_void_p * len(dev_slots))(
        *[ctypes.c_void_p(p or 0) for p in dev_slots])
    res = TpxResult()
    rc = glib.lib.tpx_stage_execute_col(stage, arr, len(dev_slots), n_rows,
                                        in_bytes, 0, 0, ctypes.byref(res))
    if rc != 0:
        raise RuntimeError("columnar stage execute failed: " + glib.err())
    try:
        out = ExecResult()
        out_bytes = ctypes.string_at(res.out_data, res.out_size)
        out_row_type = T.tup(sp.gpu_output_types)
        _offs = ([res.out_row_offsets[i] for i in range(res.out_num_rows)]
                 if res.out_num_rows else None)
        out.rows = rowfmt.deserialize_partition(out_bytes, out_row_type,
                                                offsets=_offs)
        out.row_indices = [res.out_row_indices[i] for i in range(res.out_num_rows)] \
            if res.out_num_rows else []
        if res.exc_num_rows:
            import struct as _s
            eb = ctypes.string_at(res.exc_data, res.exc_size)
            pos = 0
            for _ in range(res.exc_num_rows):
                row, ecode, opid, size = _s.unpack_from("<4q", eb, pos)
                pos += 32 + size
                out.exceptions.append((row, ecode, opid))
        out.metrics = {
            "t_h2d_ms": res.t_h2d_ms, "t_kernel_ms": res.t_kernel_ms,
            "t_d2h_ms": res.t_d2h_ms,
            "bytes_in": res.bytes_in, "bytes_out": res.bytes_out,
        }
        return out
    finally:
        glib.lib.tpx_result_free(ctypes.byref(res))


def run_collect(data: List[Any], logical_ops: List[tuple],
                columns: Optional[List[str]], options,
                keep_exceptions=False, keep_keys=False) -> "CollectOutcome":
    """Full dual-mode collect: majority-type inference, GPU normal case,
    interpreter replay of fallback + exception rows, in-order merge
    (mirrors LocalBackend.cc:963-1085 + ResolveTask.cc:878).

    keep_exceptions: rows that stay exceptions are STORED on the outcome
    (out.pending + out.pending_replayer) instead of counted — the cache()
    path (CacheOperator stores exceptions for later resolution). Implies
    keep_keys. keep_keys: out.row_keys aligns each output row with its
    original input index (ordered-merge key)."""
    out = CollectOutcome()
    keep_keys = keep_keys or keep_exceptions
    if not logical_ops:
        # pure materialization (collect of an empty pipeline / cached rows):
        # nothing to compute — pass rows through (the reference round-trips
        # partitions; observable rows are identical)
        out.mode = "identity"
        out.rows = list(data)
        if keep_keys:
            out.row_keys = list(range(len(data)))
        if keep_exceptions:
            out.pending_replayer = \
                lambda payload, ops, used=None: resolve.replay_row(
                    payload, ops, columns,
                    not (data and isinstance(data[0], tuple)))
        return out
    maj = T.infer_majority_type(data, optional_threshold=options.optional_threshold)
    row_maj = T.row_type_of(maj)
    scalar_input = not T.is_tuple(maj)

    norm_idx, norm_rows, fallback = [], [], []
    for i, v in enumerate(data):
        if T.value_conforms(v, maj):
            norm_idx.append(i)
            norm_rows.append((v,) if scalar_input else v)
        else:
            fallback.append((i, v))

    input_types = list(T.tuple_params(row_maj))
    sp = plan.build_stage(input_types, columns, logical_ops)

    # GPU-reducible aggregate: the stage returns partials; replayed rows are
    # folded on top with the user's agg fn (LocalBackend.cc:1180-1207 combine;
    # by-key: createFinalHashmap LocalBackend.cc:2219)
    if (sp.compilable and norm_rows and sp.agg_expr is not None):
        er = execute_stage_mem(sp, norm_rows,
                               partition_size=options.partition_size)
        out.mode = "gpu"
        out.metrics = er.metrics
        replay = [(norm_idx[r], data[norm_idx[r]]) for (r, _, _) in er.exceptions]
        replay += fallback
        agg_cols = sp.output_columns

        def replay_rows():
            for i, v in sorted(replay):
                r = resolve.replay_row(v, logical_ops, columns, scalar_input)
                if r[0] == "row":
                    yield r[1]
                elif r[0] == "exc":
                    name = type(r[1]).__name__
                    out.exception_counts[name] = \
                        out.exception_counts.get(name, 0) + 1
                elif r[0] == "rows":  # 1:N join expansion
                    for v2 in r[1]:
                        yield v2
                    for e in r[2]:
                        name = type(e).__name__
                        out.exception_counts[name] = \
                            out.exception_counts.get(name, 0) + 1

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
                k = rt[ki]
                table[k] = agg_fn(table.get(k, initial), _agg_row(row, agg_cols))
            out.rows = [(k, v) for k, v in table.items()]
        else:
            opid_a, combine_fn, agg_fn, initial = sp.aggregate
            acc = initial + er.rows[0][0]
            for row in replay_rows():
                acc = agg_fn(acc, _agg_row(row, agg_cols))
            out.rows = [acc]
        return out

    results = {}  # original index -> row value
    if sp.compilable and norm_rows:
        er = execute_stage_mem(sp, norm_rows,
                               partition_size=options.partition_size)
        out.mode = "gpu"
        out.metrics = er.metrics
        # outputs carry their exact local row index (out_row_indices) -> map back
        # to original data indices for the in-order merge (ResolveTask.cc:878)
        for row, local in zip(er.rows, er.row_indices):
            results.setdefault(norm_idx[local], []).append(_unwrap_row(row))
        replay = [(norm_idx[r], data[norm_idx[r]]) for (r, _, _) in er.exceptions]
    elif not sp.compilable:
        out.mode = "fallback"
        out.fallback_reason = sp.why_not_compilable
        replay = [(i, data[i]) for i in norm_idx]
    else:
        out.mode = "gpu"
        replay = []

    replay += fallback
    for i, v in sorted(replay):
        r = resolve.replay_row(v, logical_ops, columns, scalar_input)
        if r[0] == "row":
            results[i] = [r[1]]
        elif r[0] == "exc":
            if keep_exceptions:
                out.pending.append((i, v))
                continue
            name = type(r[1]).__name__
            out.exception_counts[name] = out.exception_counts.get(name, 0) + 1
        elif r[0] == "rows":  # 1:N join expansion
            results[i] = r[1]
            for e in r[2]:
                name = type(e).__name__
                out.exception_counts[name] = \
                    out.exception_counts.get(name, 0) + 1

    merged = [v for i in sorted(results) for v in results[i]]
    if keep_keys:
        out.row_keys = [i for i in sorted(results) for _ in results[i]]
    if keep_exceptions:
        def _mem_replayer(value, ops, used=None):
            return resolve.replay_row(value, ops, columns, scalar_input)
        out.pending_replayer = _mem_replayer
    out.rows = finalize_merged(merged, logical_ops, sp.output_columns)
    if out.row_keys is not None and len(out.rows) != len(out.row_keys):
        out.row_keys = None  # a trailing agg/unique consumed row identity
    return out


_AGG_KINDS = ("aggregate", "aggregateByKey", "unique")


def run_cached(context, src, post_ops, keep_exceptions=False):
    """Execute downstream ops over a cached dataset (CacheOperator.cc
    semantics): the materialized normal-case rows run through the engine,
    and the STORED exception rows (src.pending) are replayed through the
    full pre+post op chain — so resolvers attached after cache() see
    pre-cache exceptions, exactly like the reference."""
    row_ops = [op for op in post_ops if op[0] not in _AGG_KINDS]
    # leading resolve/ignore handlers target PRE-cache exceptions (replayed
    # below through the full chain); the materialized normal rows have no
    # exception to resolve, so strip them for the row run
    lead = 0
    while lead < len(row_ops) and row_ops[lead][0] in ("resolve", "ignore"):
        lead += 1
    row_run_ops = row_ops[lead:]
    if row_run_ops:
        out2 = run_collect(src.rows, row_run_ops, src.columns,
                           context.options_obj, keep_keys=True)
    else:  # identity over cached rows — no engine pass needed
        out2 = CollectOutcome()
        out2.mode = "gpu"
        out2.rows = list(src.rows)
        out2.row_keys = list(range(len(src.rows)))
    out = CollectOutcome()
    out.mode = out2.mode
    out.fallback_reason = out2.fallback_reason
    out.metrics = out2.metrics
    out.exception_counts = dict(out2.exception_counts)

    pairs = []
    if out2.row_keys is not None:
        for k2, row in zip(out2.row_keys, out2.rows):
            pairs.append((src.row_keys[k2], row))
    else:  # defensive: row identity lost (should not happen: no aggs here)
        for j, row in enumerate(out2.rows):
            pairs.append((j, row))

    full_ops = list(src.pre_ops) + list(post_ops)
    for key, payload in src.pending:
        r = src.replayer(payload, full_ops)
        if r[0] == "row":
            pairs.append((key, r[1]))
        elif r[0] == "rows":
            for v in r[1]:
                pairs.append((key, v))
            for e in r[2]:
                nm = type(e).__name__
                out.exception_counts[nm] = out.exception_counts.get(nm, 0) + 1
        elif r[0] == "exc":
            if keep_exceptions:
                out.pending.append((key, payload))
            else:
                nm = type(r[1]).__name__
                out.exception_counts[nm] = \
                    out.exception_counts.get(nm, 0) + 1
    if keep_exceptions:
        # cache-of-cache: a later consumer replays with its own (pre2+post2)
        # chain, which by construction is our post_ops+...; prepend OUR
        # pre-cache ops so the payload replays through the full lineage
        pre1 = list(src.pre_ops)
        rep = src.replayer
        out.pending_replayer = \
            lambda payload, ops, used=None: rep(payload, pre1 + list(ops))
    pairs.sort(key=lambda t: t[0])
    merged = [_unwrap_row(v) if isinstance(v, tuple) else v for _, v in pairs]
    out_cols = output_columns_of(src.columns, post_ops)
    out.rows = finalize_merged(merged, post_ops, out_cols)
    if keep_exceptions:
        out.row_keys = [k for k, _ in pairs] \
            if len(out.rows) == len(pairs) else None
    return out


def finalize_merged(merged, logical_ops, output_columns):
    """Host-side finalization of trailing unique/aggregate/aggregateByKey over
    interpreter-merged rows (fallback paths and non-GPU-reducible aggregates).
    Mirrors the GPU merge the engine does for reducible aggregates
    (LocalBackend.cc:1180-1207 combine; :2219 createFinalHashmap for by-key).
    `output_columns` are the PIPELINE's final columns (the agg UDF sees the
    post-rename/withColumn names, not the source names)."""
    for op in logical_ops:
        kind = op[0]
        if kind == "unique":
            merged = list(dict.fromkeys(merged))
        elif kind == "aggregate":
            _, combine_fn, agg_fn, initial = op
            a = initial
            for v in merged:
                a = agg_fn(a, _agg_row(v, output_columns))
            merged = [a]
        elif kind == "aggregateByKey":
            _, combine_fn, agg_fn, initial, key_cols = op
            kis = [output_columns.index(c) for c in key_cols]
            table, order = {}, []
            for v in merged:
                rt = v if isinstance(v, tuple) else (v,)
                k = rt[kis[0]] if len(kis) == 1 else tuple(rt[i] for i in kis)
                if k not in table:
                    order.append(k)
                table[k] = agg_fn(table.get(k, initial),
                                  _agg_row(v, output_columns))
            merged = [(k, table[k]) if len(kis) == 1 else k + (table[k],)
                      for k in order]
    return merged


def output_columns_of(columns, logical_ops):
    """Final column names of a pipeline without compiling its UDFs (for
    fallback paths that never built a StageProgram): threads names through
    build_stage with an all-str schema (column threading is type-independent)."""
    from . import ttypes as _T
    try:
        n = len(columns) if columns else 1
        return plan.build_stage([_T.STR] * n, columns, logical_ops).output_columns
    except Exception:  # noqa: BLE001 — name threading is best-effort here
        return list(columns) if columns else None


def _agg_row(v, columns):
    """agg fn row arg follows the same dict convention as other UDFs."""
    if columns and isinstance(v, tuple):
        return resolve.RowView(v, columns)
    return v


def _unwrap_row(row: tuple):
    return row[0] if len(row) == 1 else row


class CollectOutcome:
    def __init__(self):
        self.rows = []
        self.exception_counts = {}
        self.mode = None
        self.fallback_reason = None
        self.metrics = {}
        # cache()-path extras (CacheOperator semantics: exceptions stored,
        # not resolved): pending = [(order_key, payload)], replayer replays a
        # payload through an op chain; row_keys aligns rows with order keys
        self.pending = []
        self.pending_replayer = None
        self.row_keys = None
