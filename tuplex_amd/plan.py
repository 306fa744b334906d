"""Logical plan -> TransformStage program.

The reference builds a DAG of LogicalOperators (core/src/logical/), splits it into
TransformStages at pipeline breakers (PhysicalPlan.cc:423 splitIntoAndPlanStages) and
fuses each stage's operators into one function (StageBuilder.cc:602). The hot path
here has a single stage: source -> row ops (map/filter/withColumn/mapColumn/select/
rename) -> optional aggregate -> sink. This module turns the DataSet op chain into a
StageProgram: per-op TIR (compiled UDFs), schema threading, opids for exception
records, and the resolver table used by the host resolve path.
"""
from typing import List, Optional

from . import ttypes as T
from . import ec as EC
from .udf import compile_udf, UDFCompileError


class StageOp:
    def __init__(self, kind, opid, **kw):
        self.kind = kind      # map|filter|withColumn|mapColumn|selectColumns|rename
        self.opid = opid
        self.fn = kw.get("fn")            # original python callable
        self.tir = kw.get("tir")          # compiled TIR (None -> not compilable)
        self.col = kw.get("col")          # withColumn/mapColumn target
        self.cols = kw.get("cols")        # selectColumns list
        self.rename = kw.get("rename")    # (old, new)
        self.resolvers = []               # [(ec_code, exc_class, fn)]
        self.ignores = []                 # [(ec_code, exc_class)]
        # schemas threaded at build time
        self.in_types = None
        self.in_columns = None
        self.out_types = None
        self.out_columns = None


class StageProgram:
    """One fused TransformStage (normal-case path)."""

    def __init__(self, input_types: List, input_columns: Optional[List[str]]):
        self.input_types = list(input_types)
        self.input_columns = list(input_columns) if input_columns else None
        self.ops: List[StageOp] = []
        self.aggregate = None  # (opid, combine_fn, agg_fn, initial)
        # GPU-reducible aggregate (agg fn of shape `lambda a, x: a + expr(x)`,
        # the Q6/count/sum pattern — AggregateFunctions.cc fold): per-row expr TIR
        # + accumulator type. The device computes a deterministic fixed-order
        # partial sum; host folds replayed rows and the initial value on top.
        self.agg_expr = None
        self.agg_type = None
        # by-key variant (hashmap sink, hashmap.cc/int_hashmap.cc analog):
        # single i64 key column -> device open-addressing hash-reduce
        self.agg_key_idx = None
        self.agg_key_type = None
        self.agg_unique = False   # unique(): by-key count, keys-only output
        self.used_source_cols = None  # projection pushdown (None = parse all)
        self.compilable = True
        self.why_not_compilable = None

    @property
    def gpu_output_types(self):
        """Types of the partition the GPU returns (agg stages: 1 column;
        by-key: key + value)."""
        if self.agg_expr is not None:
            if self.agg_key_idx is not None:
                return [self.agg_key_type, self.agg_type]
            return [self.agg_type]
        return self.output_types

    @property
    def output_types(self):
        if self.ops:
            return self.ops[-1].out_types
        return self.input_types

    @property
    def output_columns(self):
        if self.ops:
            return self.ops[-1].out_columns
        return self.input_columns

    def signature(self) -> str:
        """Stable identity for the kernel cache (the analog of the reference's
        per-stage symbol names Stage_N)."""
        import hashlib
        import json

        def node_key(n):
            if n is None:
                return None
            out = {"op": n["op"], "t": repr(n["t"])}
            for k in ("v", "i", "w"):
                if k in n:
                    out[k] = repr(n[k])
            out["a"] = [node_key(c) for c in n["args"]]
            return out

        desc = {
            "in": repr(self.input_types), "cols": self.input_columns,
            "ops": [{"k": o.kind, "col": o.col, "cols": o.cols,
                     "r": o.rename, "tir": node_key(o.tir)} for o in self.ops],
        }
        return hashlib.sha256(json.dumps(desc, sort_keys=True).encode()).hexdigest()[:16]


def build_stage(input_types, input_columns, logical_ops) -> StageProgram:
    """logical_ops: list of tuples as produced by DataSet:
    ("map", fn) ("filter", fn) ("withColumn", col, fn) ("mapColumn", col, fn)
    ("selectColumns", [cols]) ("renameColumn", old, new)
    ("resolve", exc_class, fn) ("ignore", exc_class)
    ("aggregate", combine_fn, agg_fn, initial)
    """
    sp = StageProgram(input_types, input_columns)
    cur_types = list(input_types)
    cur_cols = list(input_columns) if input_columns else None
    opid = 0
    for entry in logical_ops:
        kind = entry[0]
        opid += 1
        if kind in ("resolve", "ignore"):
            if not sp.ops:
                raise ValueError("%s with no preceding operator" % kind)
            target = sp.ops[-1]
            cls = entry[1]
            code = EC.code_for_class(cls)
            if kind == "resolve":
                target.resolvers.append((code, cls, entry[2]))
            else:
                target.ignores.append((code, cls))
            continue
        if kind == "aggregate":
            sp.aggregate = (opid, entry[1], entry[2], entry[3])
            _compile_aggregate(sp, opid, entry[2], entry[3], cur_types, cur_cols)
            continue
        if kind == "unique":
            # single i64 column -> device hash-reduce (count, keys-only output);
            # anything else -> interpreter fallback dedup
            sp.aggregate = (opid, None, None, 0, None)
            sp.agg_unique = True
            if len(cur_types) == 1 and cur_types[0] in (T.I64, T.STR):
                from .udf import tir
                sp.agg_expr = tir.const(1)
                sp.agg_type = T.I64
                sp.agg_key_idx = 0
                sp.agg_key_type = cur_types[0]
                sp.agg_opid = opid
            else:
                _fallback(sp, "unique() on %r (single i64/str column on GPU "
                          "this round)" % (cur_types,))
            continue
        if kind == "aggregateByKey":
            key_cols = entry[4]
            sp.aggregate = (opid, entry[1], entry[2], entry[3], key_cols)
            _compile_aggregate(sp, opid, entry[2], entry[3], cur_types, cur_cols)
            if (sp.agg_expr is not None and len(key_cols) == 1 and cur_cols
                    and key_cols[0] in cur_cols):
                ki = cur_cols.index(key_cols[0])
                kt = cur_types[ki]
                if kt in (T.I64, T.STR):
                    sp.agg_key_idx = ki
                    sp.agg_key_type = kt
                else:
                    sp.agg_expr = None  # opt/f64 key: interpreter fallback
                    _fallback(sp, "aggregateByKey key type %r (i64/str on GPU "
                              "this round)" % (kt,))
            else:
                sp.agg_expr = None
                _fallback(sp, "aggregateByKey outside the GPU pattern")
            continue

        op = StageOp(kind, opid)
        op.in_types = list(cur_types)
        op.in_columns = list(cur_cols) if cur_cols else None
        if kind == "map":
            op.fn = entry[1]
            _compile_into(sp, op, cur_types, cur_cols)
            if op.tir is not None:
                rt = T.row_type_of(op.tir["t"])
                cur_types = list(T.tuple_params(rt))
            cur_cols = None  # map drops column names (reference DataSet::map)
        elif kind == "filter":
            op.fn = entry[1]
            _compile_into(sp, op, cur_types, cur_cols)
            if op.tir is not None and op.tir["t"] != T.BOOL:
                _fallback(sp, "filter UDF returns %r, not bool" % (op.tir["t"],))
                op.tir = None
        elif kind == "withColumn":
            op.col, op.fn = entry[1], entry[2]
            cols_for_udf = cur_cols or ["column%d" % i for i in range(len(cur_types))]
            _compile_into(sp, op, cur_types, cols_for_udf)
            vt = op.tir["t"] if op.tir is not None else T.STR
            if op.col in cols_for_udf:
                i = cols_for_udf.index(op.col)
                cur_types = cur_types[:i] + [vt] + cur_types[i + 1:]
            else:
                cols_for_udf = cols_for_udf + [op.col]
                cur_types = cur_types + [vt]
            cur_cols = cols_for_udf
        elif kind == "mapColumn":
            op.col, op.fn = entry[1], entry[2]
            if not cur_cols or op.col not in cur_cols:
                raise ValueError("mapColumn: unknown column %r" % op.col)
            i = cur_cols.index(op.col)
            try:
                op.tir = compile_udf(op.fn, [cur_types[i]], None)
            except UDFCompileError as e:
                _fallback(sp, str(e))
                op.tir = None
            if op.tir is not None:
                cur_types = cur_types[:i] + [op.tir["t"]] + cur_types[i + 1:]
        elif kind == "selectColumns":
            op.cols = entry[1]
            if cur_cols is None:
                idxs = [c for c in op.cols]
                if not all(isinstance(c, int) for c in idxs):
                    raise ValueError("selectColumns by name needs named columns")
            else:
                idxs = [cur_cols.index(c) if isinstance(c, str) else c
                        for c in op.cols]
            op.sel_idxs = idxs
            cur_types = [cur_types[i] for i in idxs]
            new_cols = []
            for c, i in zip(op.cols, idxs):
                if isinstance(c, str):
                    new_cols.append(c)
                elif cur_cols:
                    new_cols.append(cur_cols[i])
                else:
                    new_cols.append("column%d" % i)
            cur_cols = new_cols
        elif kind == "renameColumn":
            op.rename = (entry[1], entry[2])
            if cur_cols is None or entry[1] not in cur_cols:
                raise ValueError("renameColumn: unknown column %r" % entry[1])
            cur_cols = [entry[2] if c == entry[1] else c for c in cur_cols]
        elif kind == "join":
            # ("join", rrows, rcols, lk, rk, how, lp, ls, rp, rs) — inner/left
            # hash join, right = build side (JoinOperator.cc:164 layout:
            # left-except-key | key | right-except-key; left join nulls the
            # right columns, the key never nulls)
            _, rrows, rcols, lk, rk, how, lp, ls, rp, rs = entry
            if cur_cols is None or lk not in cur_cols:
                raise ValueError("join: unknown left key column %r" % lk)
            if rk not in rcols:
                raise ValueError("join: unknown right key column %r" % rk)
            lki = cur_cols.index(lk)
            rki = rcols.index(rk)
            keys = [r[rki] for r in rrows]
            if any(k is None for k in keys):
                raise ValueError("join: null keys in the build side are "
                                 "unsupported this round")
            op._join_dup = len(set(keys)) != len(keys)
            # per-column right types over the materialized rows
            rtypes = []
            for j in range(len(rcols)):
                vals = [r[j] for r in rrows]
                t = T.infer_majority_type([(v,) for v in vals],
                                          optional_threshold=1.0)
                rtypes.append(T.tuple_params(T.row_type_of(t))[0])
            lkt = T.deopt(cur_types[lki])
            if T.deopt(rtypes[rki]) != lkt or lkt not in (T.I64, T.STR):
                _fallback(sp, "join key type %r/%r (i64/str only on GPU)"
                          % (cur_types[lki], rtypes[rki]))
            if any(T.deopt(t) not in (T.I64, T.F64, T.BOOL, T.STR)
                   for t in rtypes):
                _fallback(sp, "join: unsupported right column type")
            if len(rrows) > 65536 or sum(
                    len(str(r[j]).encode()) for r in rrows
                    for j in range(len(rcols))
                    if isinstance(r[j], str)) > (4 << 20):
                _fallback(sp, "join build side too large for the embedded "
                          "table this round")
            op.join = (rrows, rcols, lki, rki, how)
            out_r = []
            for j, t in enumerate(rtypes):
                if j == rki:
                    continue
                out_r.append(T.opt(t) if how == "left" else t)
            cur_types = ([t for i2, t in enumerate(cur_types) if i2 != lki]
                         + [cur_types[lki]] + out_r)
            cur_cols = ([lp + c + ls for c in cur_cols if c != lk]
                        + [lp + lk + ls]
                        + [rp + c + rs for c in rcols if c != rk])
        else:
            raise ValueError("unknown op %r" % (kind,))
        op.out_types = list(cur_types)
        op.out_columns = list(cur_cols) if cur_cols else None
        sp.ops.append(op)
    # duplicate-key joins: the GPU engine supports the 1:N expansion only
    # when the join is the TERMINAL operator (its write kernel loops the
    # bucket; mid-pipeline expansion would break the one-slot-per-input-row
    # columnar contract) — otherwise the interpreter path forks the rows
    # (resolve._replay_from)
    for i2, op in enumerate(sp.ops):
        if op.kind == "join" and getattr(op, "_join_dup", False):
            if (i2 != len(sp.ops) - 1 or sp.aggregate is not None
                    or op.resolvers or op.ignores):
                _fallback(sp, "duplicate build-side join keys (1:N) "
                          "mid-pipeline: interpreter path this round")
            else:
                op.join_dup = True
    sp.used_source_cols = _used_source_columns(sp) if sp.compilable else None
    return sp


def _tir_input_idxs(node, acc):
    if node is None:
        return
    if node["op"] == "input":
        acc.add(node["i"])
    for a in node["args"]:
        _tir_input_idxs(a, acc)


def _used_source_columns(sp):
    """Backward column lineage: the set of SOURCE columns whose typed values the
    stage consumes (selection/projection pushdown — reference:
    core/src/logical/LogicalOptimizer.cc selectionPushdown via
    UDF::getAccessedColumns; CSVParseRowGenerator.cc parses a cell's value only
    when willBeSerialized). The CSV loader still walks every cell (structure +
    column-count errors keep reference semantics) but skips the typed parse of
    unused columns — a malformed value there diverts no row, exactly like the
    reference with pushdown on."""
    if sp.agg_unique:
        return frozenset(range(len(sp.input_types)))
    if sp.agg_expr is not None:
        used = set()
        _tir_input_idxs(sp.agg_expr, used)
        if sp.agg_key_idx is not None:
            used.add(sp.agg_key_idx)
    else:
        final = sp.ops[-1].out_types if sp.ops else sp.input_types
        used = set(range(len(final)))
    for op in reversed(sp.ops):
        n_in = len(op.in_types)
        if op.kind == "map":
            if op.tir is None:
                return frozenset(range(len(sp.input_types)))
            used = set()
            _tir_input_idxs(op.tir, used)
        elif op.kind == "filter":
            if op.tir is None:
                return frozenset(range(len(sp.input_types)))
            used = set(used)
            _tir_input_idxs(op.tir, used)
        elif op.kind == "mapColumn":
            pass  # positions map identity; the UDF consumes only its column
        elif op.kind == "withColumn":
            cols = op.in_columns or ["column%d" % k for k in range(n_in)]
            i = cols.index(op.col) if op.col in cols else n_in
            fed = i in used
            used = {p for p in used if p != i}
            if fed:
                if op.tir is None:
                    return frozenset(range(len(sp.input_types)))
                _tir_input_idxs(op.tir, used)
        elif op.kind == "selectColumns":
            used = {op.sel_idxs[p] for p in used}
        elif op.kind == "renameColumn":
            pass
        elif op.kind == "join":
            # out: [left except key | key | right...]; right cols have no
            # source dependency; the probe always reads the key
            lki = op.join[2]
            left_map = [i for i in range(n_in) if i != lki]
            u = {lki}
            for p in used:
                if p < n_in - 1:
                    u.add(left_map[p])
                elif p == n_in - 1:
                    u.add(lki)
            used = u
        else:
            return frozenset(range(len(sp.input_types)))
    return frozenset(used)


def _compile_aggregate(sp, opid, agg_fn, initial, cur_types, cur_cols):
    """Detect the linear-fold shape `lambda a, x: a + expr(x)` and compile expr.
    The reference emits agg_init/agg_combine functors (StageBuilder.cc:886-911);
    on MI355X the per-row expr feeds a deterministic device reduction and the
    final combine happens host-side (and over RCCL across GPUs)."""
    if not isinstance(initial, (int, float)) or isinstance(initial, bool):
        return
    acc_t = T.F64 if isinstance(initial, float) else T.I64
    try:
        from .udf.compile import compile_agg_udf
        node = compile_agg_udf(agg_fn, acc_t, list(cur_types),
                               list(cur_cols) if cur_cols else None)
    except UDFCompileError:
        return
    if node["op"] != "add":
        return
    lhs, rhs = node["args"]
    if not (lhs["op"] == "input" and lhs["i"] == 0):
        return

    # rhs must not reference the accumulator; shift input indices down by one
    def shift(n):
        if n["op"] == "input":
            if n["i"] == 0:
                raise UDFCompileError("acc used beyond a + expr")
            return dict(n, i=n["i"] - 1, args=[])
        out = dict(n)
        out["args"] = [shift(a) for a in n["args"]]
        return out

    try:
        expr = shift(rhs)
    except UDFCompileError:
        return
    if node["t"] not in (T.I64, T.F64) or expr["t"] not in (T.I64, T.F64, T.BOOL):
        return
    sp.agg_expr = expr if expr["t"] == node["t"] else \
        {"op": "float_num" if node["t"] == T.F64 else "int_i64",
         "t": node["t"], "args": [expr]}
    sp.agg_type = node["t"]
    sp.agg_opid = opid


def _compile_into(sp, op, cur_types, cur_cols):
    try:
        op.tir = compile_udf(op.fn, cur_types, cur_cols)
    except UDFCompileError as e:
        _fallback(sp, "%s (op %d)" % (e, op.opid))
        op.tir = None


def _fallback(sp, why):
    sp.compilable = False
    if sp.why_not_compilable is None:
        sp.why_not_compilable = why


def find_dup_join_split(logical_ops):
    """Index of the first duplicate-build-key join that is NOT eligible for
    the terminal-join GPU expansion (it is mid-pipeline), or None. The engine
    splits the pipeline there — the PhysicalPlan stage-split analog — so both
    halves run compiled on the GPU: stage 1 ends in the terminal 1:N join
    (bucket-looping write kernel), stage 2 continues from the materialized
    rows via the CachedSource machinery (exceptions stay deferred)."""
    for i, entry in enumerate(logical_ops):
        if entry[0] != "join" or i == len(logical_ops) - 1:
            continue
        _, rrows, rcols, lk, rk = entry[:5]
        try:
            rki = rcols.index(rk)
            keys = [r[rki] for r in rrows]
        except Exception:  # malformed join op -> let build_stage report it
            continue
        if len(set(keys)) != len(keys):
            return i
    return None
