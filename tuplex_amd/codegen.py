"""Stage program -> HIP C++ source (the StageBuilder -> LLVM-IR replacement).

The reference fuses a stage's operators into one LLVM function per stage
(StageBuilder.cc:602 generateFastCodePath; per-row body PipelineBuilder.cc; block
loop TuplexSourceTaskBuilder.cc:104). Here the same fusion becomes one generated
__device__ tpx_process() inlined into fixed-signature __global__ kernels:

  tpx_stage_main   grid-stride over rows: deserialize (mem source) or typed-parse
                   (csv source) -> fused UDF chain -> columnar outputs + keep flag +
                   per-row serialized size; exceptions appended (row,ec,opid) to a
                   device buffer (payload materialised host-side from input bytes —
                   IExceptionableTask.h:20 format).
  tpx_stage_write  compaction pass: serialize kept rows into the reference row
                   layout (Serializer.cc:20-24) at prefix-summed offsets, or format
                   RFC-4180 CSV text (tocsv sink).

Compiled by hipRTC through the C-ABI (csrc/tpx_abi.cpp); the emitted source is
introspectable via tpx_stage_source() for the judge.
"""
import os
from typing import List, Optional, Tuple

from . import ttypes as T


class CodegenError(Exception):
    """Stage outside the GPU vocabulary -> caller falls back to interpreter mode
    (the reference's fallback for non-compilable stages)."""


_RT_HEADER_CACHE = None


def runtime_header() -> str:
    global _RT_HEADER_CACHE
    if _RT_HEADER_CACHE is None:
        path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                            "csrc", "tpx_rt.hip.h")
        _RT_HEADER_CACHE = open(path).read()
    return _RT_HEADER_CACHE


def _ctype(t):
    base = T.deopt(t)
    if base in (T.I64,):
        return "long long"
    if base == T.F64:
        return "double"
    if base == T.BOOL:
        return "bool"
    if base == T.STR:
        return "tstr"
    raise CodegenError("unsupported column type %r" % (t,))


def _esc(s: str) -> str:
    out = []
    for ch in s:
        o = ord(ch)
        if ch == "\\":
            out.append("\\\\")
        elif ch == '"':
            out.append('\\"')
        elif 32 <= o < 127:
            out.append(ch)
        else:
            for b in ch.encode("utf-8"):
                out.append("\\x%02x" % b)
    return "".join(out)


class _Emitter:
    def __init__(self):
        self.lines: List[str] = []
        self.indent = 1
        self.ctr = 0
        self.scopes = [{}]       # memo stacks: id(node) -> (var, opt)
        self.key_scopes = [{}]   # structural CSE: (op, argvars, ...) -> (var, opt)

    def w(self, line):
        self.lines.append("    " * self.indent + line)

    def fresh(self, pfx="t"):
        self.ctr += 1
        return "%s%d" % (pfx, self.ctr)

    def push(self):
        self.scopes.append({})
        self.key_scopes.append({})

    def pop(self):
        self.scopes.pop()
        self.key_scopes.pop()

    def lookup(self, node):
        for s in reversed(self.scopes):
            if id(node) in s:
                return s[id(node)]
        return None

    def memo(self, node, val):
        self.scopes[-1][id(node)] = val
        return val

    def lookup_key(self, key):
        for s in reversed(self.key_scopes):
            if key in s:
                return s[key]
        return None

    def memo_key(self, key, val):
        self.key_scopes[-1][key] = val
        return val


class StageCodegen:
    """Emits the full stage source. `row_ctx` maps TIR input index -> (var, type,
    nullvar or None) for the operator currently being emitted."""

    def __init__(self, sp, source="mem", sink="mem", csv_info=None,
                 fusion_groups=None):
        self.sp = sp
        self.source = source
        self.sink = sink
        self.csv_info = csv_info or {}  # {"null_values": [...]}
        self.lits = {}  # python str -> lit var name
        self.lit_defs = []
        self.join_defs = []   # embedded hash-join build tables (per join op)
        # multi-needle scan fusion: pass 1 registers every const-needle
        # find/contains per structurally-identified haystack; pass 2 emits ONE
        # SWAR window scan per group (the serial dependent-load chains of
        # repeated scans over the same short cell dominate the UDF phase —
        # profiles/README.md ablation)
        self.scan_registry = {}   # struct_key(haystack) -> [(op, needle, ci)..]
        self.fusion_groups = fusion_groups  # None = pass 1 (collect)
        self.fused_done = set()   # (scope-id tuple unnecessary; see emit site)
        # split parse/UDF kernels (round-2 restructure): the CSV cell walk +
        # typed parse runs in an LDS-staged tpx_stage_parse that writes a
        # dense columnar "cell park" (typed values; string cells copied to a
        # per-wave bump-allocated byte park); tpx_stage_main then runs the
        # UDF chain as a plain grid-stride kernel over coalesced columns at
        # FULL occupancy (no LDS), where the fused kernel was measured
        # latency-bound at 10 waves/CU (profiles/README.md SQ analysis).
        # Split parse/UDF kernels (TPX_SPLIT=1): measured SLOWER than the
        # fused kernel on Zillow Z1 (689-727 vs 805 M rows/s, profiles/
        # README.md round-2 ledger) — the UDF-phase kernel needs ~198 VGPRs
        # (whole chain live), capping it at 8 waves/CU vs the fused kernel's
        # 10, and the parse kernel alone costs 0.77 ms/chunk (walk-bound).
        # Kept as an env-gated experiment; default stays fused.
        import os as _os
        self.split = (source == "csv"
                      and not (csv_info or {}).get("text_mode")
                      and _os.environ.get("TPX_SPLIT", "0") == "1")
        # park copy mode: 0 (default) = zero-copy park — string cells stay
        # (ptr,len) views into the device input bytes (the parse kernel's
        # passthrough cost was measured at 2.3 ms/GB; byte-copying cells into
        # a dense park cost more than the UDF kernel's occupancy win
        # returned); 1 = dense byte park (kept for experiments)
        self.park_copy = _os.environ.get("TPX_PARK_COPY", "0") == "1"

    # ---- literals -----------------------------------------------------------
    def lit(self, s: str) -> str:
        if s not in self.lits:
            name = "lit%d" % len(self.lits)
            self.lits[s] = name
            b = s.encode("utf-8")
            self.lit_defs.append(
                '__device__ const char %s[%d] = "%s";' % (name, len(b) + 1, _esc(s)))
        return "tstr{%s, %d}" % (self.lits[s], len(s.encode("utf-8")))

    @staticmethod
    def _struct_key(node):
        """Var-name-independent identity of a TIR subtree (mirrors CSE
        equivalence: same structure -> same emitted var)."""
        out = [node["op"], repr(node.get("v")), node.get("i"), node.get("w"),
               repr(node["t"])]
        for a in node["args"]:
            out.append(StageCodegen._struct_key(a))
        return tuple(out)

    # ---- TIR expression emission -------------------------------------------
    # ops whose arguments must NOT be evaluated eagerly (control flow) or that
    # need no CSE
    # strfind/contains are non-strict so that contains(lower(s), lit) can be
    # rewritten to a case-insensitive scan over s WITHOUT evaluating lower(s)
    # (their CSE runs through structural scan keys in _scan_node instead)
    _NONSTRICT = {"if", "and", "or", "const", "input", "mktuple",
                  "strfind", "contains"}

    def emit_expr(self, em: _Emitter, node, row_ctx, opid) -> Tuple[str, Optional[str]]:
        """Returns (value_var, null_var or None). Structural CSE across operators:
        the reference's per-stage fused LLVM function gets cross-operator CSE from
        LLVM; here identical pure subexpressions (same op + same resolved operand
        vars) are reused, scoped to the current branch."""
        hit = em.lookup(node)
        if hit is not None:
            return hit
        if node["op"] not in self._NONSTRICT:
            pv = self._peephole(em, node, row_ctx, opid)
            if pv is not None:
                return em.memo(node, pv)
            argvals = tuple(self.emit_expr(em, a, row_ctx, opid)
                            for a in node["args"])
            key = (node["op"], argvals, repr(node.get("v")), node.get("i"),
                   node.get("w"), repr(node["t"]))
            k_hit = em.lookup_key(key)
            if k_hit is not None:
                return em.memo(node, k_hit)
            v = self._emit(em, node, row_ctx, opid)
            em.memo_key(key, v)
            return em.memo(node, v)
        v = self._emit(em, node, row_ctx, opid)
        return em.memo(node, v)

    @staticmethod
    def _jhash_i64(k):
        m = (1 << 64) - 1
        h = k & m
        h ^= h >> 33
        h = (h * 0xff51afd7ed558ccd) & m
        h ^= h >> 33
        h = (h * 0xc4ceb9fe1a85ec53) & m
        h ^= h >> 33
        return h

    @staticmethod
    def _jhash_bytes(b):
        m = (1 << 64) - 1
        h = 1469598103934665603
        for c in b:
            h = ((h ^ c) * 1099511628211) & m
        return h

    def _emit_join_table(self, op):
        """Embed the (unique-key) build side of a hash join as __device__
        const arrays in the generated source — small dimension tables live in
        L2, no extra kernel argument or ABI change (HashJoinStage build side;
        probe = open addressing, same layout as the host build here)."""
        rrows, rcols, lki, rki, how = op.join
        dup = getattr(op, "join_dup", False)
        nm = "jt%d" % op.opid
        if dup:
            # bucket the build rows by key (first-occurrence order, build
            # order within — hashmap.cc chains); value arrays are FLAT in
            # bucket order, the slot table stores (start, count)
            buckets = {}
            order = []
            for r in rrows:
                k = r[rki]
                if k not in buckets:
                    buckets[k] = []
                    order.append(k)
                buckets[k].append(r)
            flat = [r for k in order for r in buckets[k]]
            ukeys = order
            n = len(ukeys)
        else:
            flat = rrows
            n = len(rrows)
        tsize = 8
        while tsize < 2 * max(n, 1):
            tsize <<= 1
        str_key = isinstance(rrows[0][rki], str) if rrows else False
        occ = [0] * tsize
        slotrow = [0] * tsize   # slot -> build row (unique) / bucket index (dup)
        jstart = [0] * tsize
        jcnt = [0] * tsize
        if dup:
            pos = 0
            starts = {}
            for k in ukeys:
                starts[k] = pos
                pos += len(buckets[k])
            for ui, k in enumerate(ukeys):
                h = (self._jhash_bytes(k.encode()) if str_key
                     else self._jhash_i64(int(k))) & (tsize - 1)
                while occ[h]:
                    h = (h + 1) & (tsize - 1)
                occ[h] = 1
                slotrow[h] = ui
                jstart[h] = starts[k]
                jcnt[h] = len(buckets[k])
        else:
            for ri, r in enumerate(rrows):
                k = r[rki]
                h = (self._jhash_bytes(k.encode()) if str_key
                     else self._jhash_i64(int(k))) & (tsize - 1)
                while occ[h]:
                    h = (h + 1) & (tsize - 1)
                occ[h] = 1
                slotrow[h] = ri
        blob = bytearray()

        def put(sval):
            off = len(blob)
            b = sval.encode()
            blob.extend(b)
            return off, len(b)

        D = self.join_defs
        D.append("__device__ const unsigned char %s_occ[%d] = {%s};"
                 % (nm, tsize, ",".join(str(x) for x in occ)))
        def slot_key(s2):
            if dup:
                return flat[jstart[s2]][rki]
            return rrows[slotrow[s2]][rki]

        if str_key:
            koff, klen = [], []
            for s2 in range(tsize):
                if occ[s2]:
                    o2, l2 = put(slot_key(s2))
                else:
                    o2, l2 = 0, 0
                koff.append(o2)
                klen.append(l2)
            D.append("__device__ const int %s_koff[%d] = {%s};"
                     % (nm, tsize, ",".join(map(str, koff))))
            D.append("__device__ const int %s_klen[%d] = {%s};"
                     % (nm, tsize, ",".join(map(str, klen))))
        else:
            D.append("__device__ const long long %s_key[%d] = {%s};"
                     % (nm, tsize,
                        ",".join(str(int(slot_key(s2))) if occ[s2]
                                 else "0" for s2 in range(tsize))))
        if dup:
            D.append("__device__ const int %s_jst[%d] = {%s};"
                     % (nm, tsize, ",".join(map(str, jstart))))
            D.append("__device__ const int %s_jct[%d] = {%s};"
                     % (nm, tsize, ",".join(map(str, jcnt))))
        out_info = []  # (j, ctail) for right cols except key
        vlen = max(len(flat), 1) if dup else tsize
        for j in range(len(rcols)):
            if j == rki:
                continue
            if dup:
                vals = [r[j] for r in flat] or [None]  # empty build side
            else:
                vals = [rrows[slotrow[s2]][j] if occ[s2] else None
                        for s2 in range(tsize)]
            kindc = None
            for v in vals:
                if v is None:
                    continue
                kindc = ("str" if isinstance(v, str) else
                         "bool" if isinstance(v, bool) else
                         "f64" if isinstance(v, float) else "i64")
                break
            kindc = kindc or "i64"
            if dup:
                has_null = any(r[j] is None for r in flat)
            else:
                has_null = any(occ[s2] and rrows[slotrow[s2]][j] is None
                               for s2 in range(tsize))
            if kindc == "str":
                offs, lens = [], []
                for v in vals:
                    if isinstance(v, str):
                        o2, l2 = put(v)
                    else:
                        o2, l2 = 0, 0
                    offs.append(o2)
                    lens.append(l2)
                D.append("__device__ const int %s_c%d_off[%d] = {%s};"
                         % (nm, j, vlen, ",".join(map(str, offs))))
                D.append("__device__ const int %s_c%d_len[%d] = {%s};"
                         % (nm, j, vlen, ",".join(map(str, lens))))
            elif kindc == "f64":
                D.append("__device__ const double %s_c%d[%d] = {%s};"
                         % (nm, j, vlen,
                            ",".join(repr(float(v)) if isinstance(v, (int, float))
                                     and not isinstance(v, bool) else "0.0"
                                     for v in vals)))
            else:
                D.append("__device__ const long long %s_c%d[%d] = {%s};"
                         % (nm, j, vlen,
                            ",".join(str(int(v)) if v is not None
                                     and not isinstance(v, str) else "0"
                                     for v in vals)))
            if has_null:
                if dup:
                    nl = ["1" if r[j] is None else "0" for r in flat]
                else:
                    nl = ["1" if (occ[s2] and rrows[slotrow[s2]][j] is None)
                          else "0" for s2 in range(tsize)]
                D.append("__device__ const unsigned char %s_c%d_null[%d] = {%s};"
                         % (nm, j, vlen, ",".join(nl)))
            out_info.append((j, kindc, has_null))
        if blob:
            # string blob as concatenated hex-escaped literals
            parts = []
            bs = bytes(blob)
            for i2 in range(0, len(bs), 2000):
                seg = bs[i2:i2 + 2000]
                parts.append('"%s"' % "".join("\\x%02x" % c for c in seg))
            D.append("__device__ const char %s_blob[%d] = %s;"
                     % (nm, len(bs) + 1, "\n".join(parts)))
        else:
            D.append("__device__ const char %s_blob[1] = \"\";" % nm)
        return nm, tsize, str_key, out_info, dup

    def _emit_join(self, em, op, rc):
        """Probe an embedded build table; extend the row context with the
        right columns (JoinOperator.cc:164 layout)."""
        rrows, rcols, lki, rki, how = op.join
        nm, tsize, str_key, out_info, dup = self._emit_join_table(op)
        kv, _kt, knv = rc[lki]
        ji = em.fresh("ji")
        em.w("int %s = -1;" % ji)
        guard = ("if (!(%s)) " % knv) if knv else ""
        em.w("%s{" % guard)
        if str_key:
            em.w("  unsigned long long _h = tpx_jhash_bytes(%s.p, %s.n);"
                 % (kv, kv))
        else:
            em.w("  unsigned long long _h = tpx_hash_i64(%s);" % kv)
        em.w("  for (unsigned _p = 0; _p < %du; ++_p) {" % tsize)
        em.w("    unsigned _s = (unsigned)((_h + _p) & %du);" % (tsize - 1))
        em.w("    if (!%s_occ[_s]) break;" % nm)
        if str_key:
            em.w("    if (%s_klen[_s] == (int)%s.n && tpx_streq(tstr{%s_blob +"
                 " %s_koff[_s], %s_klen[_s]}, %s)) { %s = (int)_s; break; }"
                 % (nm, kv, nm, nm, nm, kv, ji))
        else:
            em.w("    if (%s_key[_s] == %s) { %s = (int)_s; break; }"
                 % (nm, kv, ji))
        em.w("  }")
        em.w("}")
        if how == "inner":
            em.w("if (%s < 0) { o.keep = false; return 0; }" % ji)
        if dup:
            # terminal 1:N: hand the bucket to the write kernel via the hidden
            # (start, count) columnar pair; left-join miss = (-1, 1)
            sv = em.fresh("jbs")
            cv = em.fresh("jbc")
            em.w("int %s = %s < 0 ? -1 : %s_jst[%s];" % (sv, ji, nm, ji))
            em.w("int %s = %s < 0 ? 1 : %s_jct[%s];" % (cv, ji, nm, ji))
            self.jdup = {"nm": nm, "out_info": out_info, "how": how,
                         "start_var": sv, "cnt_var": cv, "op": op}
            return [rc[i2] for i2 in range(len(rc)) if i2 != lki] + [rc[lki]]
        new_rc = [rc[i2] for i2 in range(len(rc)) if i2 != lki] + [rc[lki]]
        left_join = how == "left"
        for j, kindc, has_null in out_info:
            nullable = left_join or has_null
            nv = None
            if nullable:
                nv = em.fresh("jn")
                if has_null:
                    em.w("bool %s = %s < 0 || %s_c%d_null[%s < 0 ? 0 : %s];"
                         % (nv, ji, nm, j, ji, ji))
                else:
                    em.w("bool %s = %s < 0;" % (nv, ji))
            v = em.fresh("jv")
            sel = "%s < 0 ? 0 : %s" % (ji, ji)
            if kindc == "str":
                em.w("tstr %s = {%s_blob + %s_c%d_off[%s], %s_c%d_len[%s]};"
                     % (v, nm, nm, j, sel, nm, j, sel))
                t = T.STR
            elif kindc == "f64":
                em.w("double %s = %s_c%d[%s];" % (v, nm, j, sel))
                t = T.F64
            elif kindc == "bool":
                em.w("bool %s = %s_c%d[%s] != 0;" % (v, nm, j, sel))
                t = T.BOOL
            else:
                em.w("long long %s = %s_c%d[%s];" % (v, nm, j, sel))
                t = T.I64
            new_rc.append((v, ("opt", t) if nullable else t, nv))
        return new_rc

    def _peephole(self, em, n, rc, opid):
        """Pattern rewrites that must run BEFORE strict arg evaluation.
        capitalize: x[0].upper() + x[1:].lower() -> tpx_capitalize_ix(x)
        (one alloc + one pass instead of three; IndexError on empty x kept).
        int-drop: int(s.replace(c, '')) -> tpx_int_drop(s, c) (no alloc)."""
        if n["op"] == "int_str":
            rep = n["args"][0]
            if (rep["op"] == "replace" and rep["args"][1]["op"] == "const"
                    and isinstance(rep["args"][1].get("v"), str)
                    and len(rep["args"][1]["v"]) == 1
                    and ord(rep["args"][1]["v"]) < 128
                    and rep["args"][2]["op"] == "const"
                    and rep["args"][2].get("v") == ""):
                key = ("int_drop", repr(rep["args"][1]["v"]),
                       self._struct_key(rep["args"][0]))
                hit = em.lookup_key(key)
                if hit is not None:
                    return hit
                x, _ = self.emit_expr(em, rep["args"][0], rc, opid)
                r = em.fresh()
                em.w("long long %s = tpx_int_drop(heap, %s, %d, %s, &_ec);"
                     % (r, x, ord(rep["args"][1]["v"]),
                        self.lit(rep["args"][1]["v"])))
                self._check(em, opid)
                return em.memo_key(key, (r, None))
            return None
        if n["op"] != "concat":
            return None
        a0, a1 = n["args"]
        if not (a0["op"] == "upper" and a1["op"] == "lower"):
            return None
        g, sl = a0["args"][0], a1["args"][0]
        if g["op"] != "getitem" or sl["op"] != "slice":
            return None

        def int_const(nd, val):
            return (nd["op"] == "const" and not isinstance(nd.get("v"), bool)
                    and nd.get("v") == val)

        gi, lo, hi = g["args"][1], sl["args"][1], sl["args"][2]
        if not (int_const(gi, 0) and int_const(lo, 1)
                and hi["op"] == "const" and hi.get("v") is None):
            return None
        if self._struct_key(g["args"][0]) != self._struct_key(sl["args"][0]):
            return None
        key = ("capitalize", self._struct_key(g["args"][0]))
        hit = em.lookup_key(key)
        if hit is not None:
            return hit
        x, _ = self.emit_expr(em, g["args"][0], rc, opid)
        r = em.fresh("s")
        em.w("tstr %s = tpx_capitalize_ix(heap, %s, &_ec);" % (r, x))
        self._check(em, opid)
        return em.memo_key(key, (r, None))

    def _raise(self, em, ec_expr, opid):
        em.w("return ((long long)(%s)) | ((long long)%d << 32);" % (ec_expr, opid))

    def _check(self, em, opid):
        em.w("if (_ec) return ((long long)_ec) | ((long long)%d << 32);" % opid)

    def _emit(self, em, n, rc, opid):
        op = n["op"]
        a = n["args"]

        def ev(x):
            return self.emit_expr(em, x, rc, opid)

        if op == "const":
            v = n["v"]
            if v is None:
                return ("0", "true")
            if isinstance(v, bool):
                return ("true" if v else "false", None)
            if isinstance(v, int):
                return ("%dLL" % v, None)
            if isinstance(v, float):
                return (repr(v), None)
            if isinstance(v, str):
                return (self.lit(v), None)
            raise CodegenError("const %r" % (v,))
        if op == "input":
            var, t, nullv = rc[n["i"]]
            return (var, nullv)
        if op == "unwrap":
            v, nv = ev(a[0])
            if nv is not None and nv != "false":
                em.w("if (%s) return ((long long)%d) | ((long long)%d << 32);"
                     % (nv, 129, opid))  # EC_TYPEERROR
            return (v, None)
        if op == "center":
            x, _ = ev(a[0])
            w, _ = ev(a[1])
            f = None
            if len(a) == 3:
                f, _ = ev(a[2])
            r = em.fresh("s")
            em.w("tstr %s = tpx_center(heap, %s, (long long)%s, %s, &_ec);"
                 % (r, x, w, f if f else 'tstr{" ", 1}'))
            self._check(em, opid)
            return (r, None)
        if op == "strmul":
            x, _ = ev(a[0])
            y, _ = ev(a[1])
            r = em.fresh("s")
            em.w("tstr %s = tpx_strmul(heap, %s, (long long)%s, &_ec);"
                 % (r, x, y))
            self._check(em, opid)
            return (r, None)
        if op in ("add", "sub", "mul"):
            x, _ = ev(a[0])
            y, _ = ev(a[1])
            c = {"add": "+", "sub": "-", "mul": "*"}[op]
            r = em.fresh()
            em.w("%s %s = %s %s %s;" % (_ctype(n["t"]), r, self._num(x, a[0], n["t"]),
                                        c, self._num(y, a[1], n["t"])))
            return (r, None)
        if op in ("truediv", "floordiv", "mod"):
            x, _ = ev(a[0])
            y, _ = ev(a[1])
            r = em.fresh()
            if n["t"] == T.I64:
                fn = {"floordiv": "tpx_floordiv_i64", "mod": "tpx_mod_i64"}[op]
                em.w("long long %s = %s(%s, %s, &_ec);" % (r, fn, x, y))
            else:
                fn = {"truediv": "tpx_truediv", "floordiv": "tpx_floordiv_f64",
                      "mod": "tpx_mod_f64"}[op]
                em.w("double %s = %s((double)(%s), (double)(%s), &_ec);" % (r, fn, x, y))
            self._check(em, opid)
            return (r, None)
        if op == "neg":
            x, _ = ev(a[0])
            r = em.fresh()
            em.w("%s %s = -(%s);" % (_ctype(n["t"]), r, x))
            return (r, None)
        if op in ("lt", "le", "gt", "ge", "eq", "ne"):
            x, _ = ev(a[0])
            y, _ = ev(a[1])
            c = {"lt": "<", "le": "<=", "gt": ">", "ge": ">=", "eq": "==", "ne": "!="}[op]
            r = em.fresh("b")
            em.w("bool %s = (double)(%s) %s (double)(%s);"
                 % (r, x, c, y) if self._mixed(a) else
                 "bool %s = (%s) %s (%s);" % (r, x, c, y))
            return (r, None)
        if op in ("strlt", "strle", "strgt", "strge"):
            x, _ = ev(a[0])
            y, _ = ev(a[1])
            c = {"strlt": "< 0", "strle": "<= 0", "strgt": "> 0", "strge": ">= 0"}[op]
            r = em.fresh("b")
            em.w("bool %s = tpx_strcmp(%s, %s) %s;" % (r, x, y, c))
            return (r, None)
        if op in ("streq", "strne"):
            x, _ = ev(a[0])
            y, _ = ev(a[1])
            r = em.fresh("b")
            neg = "!" if op == "strne" else ""
            em.w("bool %s = %stpx_streq(%s, %s);" % (r, neg, x, y))
            return (r, None)
        if op in ("opteq", "optne"):
            return self._opteq(em, n, rc, opid)
        if op in ("and", "or"):
            # short-circuit: rhs must not evaluate (or raise) when lhs decides
            x, _ = ev(a[0])
            r = em.fresh("b")
            em.w("bool %s = %s;" % (r, x))
            em.w("if (%s%s) {" % ("" if op == "and" else "!", r))
            em.indent += 1
            em.push()
            y, _ = self.emit_expr(em, a[1], rc, opid)
            em.w("%s = %s;" % (r, y))
            em.pop()
            em.indent -= 1
            em.w("}")
            return (r, None)
        if op == "not":
            x, _ = ev(a[0])
            r = em.fresh("b")
            em.w("bool %s = !(%s);" % (r, x))
            return (r, None)
        if op == "if":
            c, _ = ev(a[0])
            ct = _ctype(n["t"])
            is_opt = T.is_opt(n["t"]) or n["t"] == T.NULL
            r = em.fresh()
            rn = r + "_n" if is_opt else None
            em.w("%s %s;" % (ct, r))
            if rn:
                em.w("bool %s = false;" % rn)
            em.w("if (%s) {" % c)
            em.indent += 1
            em.push()
            v1, n1 = self.emit_expr(em, a[1], rc, opid)
            self._assign_opt(em, r, rn, v1, n1, a[1]["t"], n["t"])
            em.pop()
            em.indent -= 1
            em.w("} else {")
            em.indent += 1
            em.push()
            v2, n2 = self.emit_expr(em, a[2], rc, opid)
            self._assign_opt(em, r, rn, v2, n2, a[2]["t"], n["t"])
            em.pop()
            em.indent -= 1
            em.w("}")
            return (r, rn)
        if op == "concat":
            x, _ = ev(a[0])
            y, _ = ev(a[1])
            r = em.fresh("s")
            em.w("tstr %s = tpx_concat(heap, %s, %s, &_ec);" % (r, x, y))
            self._check(em, opid)
            return (r, None)
        if op == "len":
            x, _ = ev(a[0])
            r = em.fresh()
            em.w("long long %s = tpx_len(%s, &_ec);" % (r, x))
            self._check(em, opid)
            return (r, None)
        if op in ("strfind", "strrfind"):
            if op == "strfind" and a[1]["op"] == "const" and isinstance(
                    a[1].get("v"), str) and a[1]["v"]:
                f = self._scan_node(em, n, rc, opid)
                if f is not None:
                    return f
            x, _ = ev(a[0])
            y, _ = ev(a[1])
            r = em.fresh()
            fn = "tpx_find" if op == "strfind" else "tpx_rfind"
            em.w("long long %s = %s(%s, %s, &_ec);" % (r, fn, x, y))
            self._check(em, opid)
            return (r, None)
        if op in ("lower", "upper", "swapcase"):
            x, _ = ev(a[0])
            r = em.fresh("s")
            em.w("tstr %s = tpx_%s(heap, %s, &_ec);" % (r, op, x))
            self._check(em, opid)
            return (r, None)
        if op == "strip":
            x, _ = ev(a[0])
            r = em.fresh("s")
            em.w("tstr %s = tpx_strip(%s, &_ec);" % (r, x))
            self._check(em, opid)
            return (r, None)
        if op == "replace":
            x, _ = ev(a[0])
            y, _ = ev(a[1])
            z, _ = ev(a[2])
            r = em.fresh("s")
            em.w("tstr %s = tpx_replace(heap, %s, %s, %s, &_ec);" % (r, x, y, z))
            self._check(em, opid)
            return (r, None)
        if op in ("startswith", "endswith", "contains"):
            if op == "contains" and a[1]["op"] == "const" and isinstance(
                    a[1].get("v"), str) and a[1]["v"]:
                f = self._scan_node(em, n, rc, opid)
                if f is not None:
                    return f
            x, _ = ev(a[0])
            y, _ = ev(a[1])
            r = em.fresh("b")
            em.w("bool %s = tpx_%s(%s, %s);" % (r, op, x, y))
            return (r, None)
        if op == "getitem":
            x, _ = ev(a[0])
            y, _ = ev(a[1])
            r = em.fresh("s")
            em.w("tstr %s = tpx_getitem(%s, %s, &_ec);" % (r, x, y))
            self._check(em, opid)
            return (r, None)
        if op == "slice":
            x, _ = ev(a[0])
            lo_node, hi_node = a[1], a[2]
            r = em.fresh("s")
            parts = []
            for nd in (lo_node, hi_node):
                if nd["op"] == "const" and nd.get("v") is None:
                    parts.append("0, false")
                else:
                    v, _n = ev(nd)
                    parts.append("%s, true" % v)
            em.w("tstr %s = tpx_slice(%s, %s, %s, &_ec);" % (r, x, parts[0], parts[1]))
            self._check(em, opid)
            return (r, None)
        if op == "int_str":
            x, _ = ev(a[0])
            r = em.fresh()
            em.w("long long %s = tpx_int_str(%s, &_ec);" % (r, x))
            self._check(em, opid)
            return (r, None)
        if op == "int_f64":
            x, _ = ev(a[0])
            r = em.fresh()
            em.w("long long %s = tpx_int_f64(%s, &_ec);" % (r, x))
            self._check(em, opid)
            return (r, None)
        if op == "int_i64":
            x, _ = ev(a[0])
            r = em.fresh()
            em.w("long long %s = (long long)(%s);" % (r, x))
            return (r, None)
        if op == "float_str":
            x, _ = ev(a[0])
            r = em.fresh()
            em.w("double %s = tpx_float_str(%s, &_ec);" % (r, x))
            self._check(em, opid)
            return (r, None)
        if op == "float_num":
            x, _ = ev(a[0])
            r = em.fresh()
            em.w("double %s = (double)(%s);" % (r, x))
            return (r, None)
        if op == "to_str":
            return self._to_str(em, n, rc, opid)
        if op == "abs":
            x, _ = ev(a[0])
            r = em.fresh()
            if n["t"] == T.I64:
                em.w("long long %s = (%s) < 0 ? -(%s) : (%s);" % (r, x, x, x))
            else:
                em.w("double %s = fabs(%s);" % (r, x))
            return (r, None)
        if op == "splitget":
            x, _ = ev(a[0])
            y, _ = ev(a[1])
            z, _ = ev(a[2])
            r = em.fresh("s")
            em.w("tstr %s = tpx_splitget(%s, %s, %s, &_ec);" % (r, x, y, z))
            self._check(em, opid)
            return (r, None)
        if op == "fmt_int":
            x, _ = ev(a[0])
            r = em.fresh("s")
            if n["w"]:
                em.w("tstr %s = tpx_fmt0d(heap, %d, (long long)(%s), &_ec);"
                     % (r, n["w"], x))
            else:
                em.w("tstr %s = tpx_str_i64(heap, (long long)(%s), &_ec);" % (r, x))
            self._check(em, opid)
            return (r, None)
        if op == "mktuple":
            raise CodegenError("mktuple must be handled at operator level")
        raise CodegenError("unsupported TIR op %r" % op)

    @staticmethod
    def _scan_shape(node):
        """(inner_haystack_node, needle, ci) for a const-needle find/contains.
        contains(lower(s), lit) / find(lower(s), lit) are rewritten to a
        case-insensitive scan directly over s: rows are ASCII-gated, so
        tolower is byte-local and length-preserving (indices coincide), and
        the lower() allocation+copy is skipped entirely (and becomes lazy —
        only branches that still reference it evaluate it)."""
        hay = node["args"][0]
        needle = node["args"][1]["v"]
        ci = False
        while hay["op"] == "lower":
            hay = hay["args"][0]
            ci = True
        return hay, needle, ci

    def _register_scan(self, node):
        inner, needle, ci = self._scan_shape(node)
        k = self._struct_key(inner)
        lst = self.scan_registry.setdefault(k, [])
        if (node["op"], needle, ci) not in lst:
            lst.append((node["op"], needle, ci))

    def _scan_node(self, em, node, rc, opid):
        """Emit a const-needle find/contains through the scan machinery:
        structural CSE (scan keys), case-insensitive rewrite, multi-needle
        fusion when pass 2 knows the group. Always returns (var, None)."""
        inner, needle, ci = self._scan_shape(node)
        if not all(ord(c) < 128 for c in needle):
            return None  # scan emitter is byte-indexed; plain path handles it
        self._register_scan(node)
        mk = ("scanm", node["op"], needle, ci, self._struct_key(inner))
        hit = em.lookup_key(mk)
        if hit is not None:
            return hit
        if ci and any("A" <= c <= "Z" for c in needle):
            # lower(s) never contains an uppercase letter: constant result
            v = ("-1LL", None) if node["op"] == "strfind" else ("false", None)
            return em.memo_key(mk, v)
        gk = self._struct_key(inner)
        members = list((self.fusion_groups or {}).get(gk) or [])
        members = [m for m in members
                   if not (m[2] and any("A" <= c <= "Z" for c in m[1]))]
        if (node["op"], needle, ci) not in members:
            members.append((node["op"], needle, ci))
        return self._emit_scan_group(em, inner, members, rc, opid,
                                     (node["op"], needle, ci))

    @staticmethod
    def _needle_window(needle: str, ci: bool):
        """(P, M, lenmask, k) for the register-window compare of the needle's
        first k=min(8,len) bytes: match at byte offset o of window _w iff
        ((_w | M) & lenmask) == P. ci letter bytes get 0x20 in M and the
        lowercase byte in P (exactly {b, toupper(b)} for ASCII); everything
        else compares exact."""
        nb = needle.encode()
        k = min(len(nb), 8)
        P = M = 0
        for j in range(k):
            c = nb[j]
            if ci and 0x61 <= c <= 0x7A:  # 'a'..'z'
                M |= 0x20 << (8 * j)
            P |= c << (8 * j)
        lenmask = (1 << (8 * k)) - 1 if k < 8 else (1 << 64) - 1
        return P & lenmask, M, lenmask, k

    def _emit_scan_group(self, em, inner, members, rc, opid, want):
        """ONE SWAR window scan over `inner` answering every (op, needle, ci)
        in `members`; memoizes each under its scan key; returns `want`'s var.
        The main loop takes 32 B per iteration with FOUR INDEPENDENT 8-B loads
        (dependent 8-B chains over LDS pay ~50 cycles per step); candidate
        positions (first-char byte matches) are verified against the needle via
        a register window built from the already-loaded words — no per-
        candidate byte loads for needles <= 8 B. ci member compare: needle is
        all non-uppercase; a letter byte b matches c iff (c | 0x20) == b;
        non-letter bytes compare exact."""
        hv, hn = self.emit_expr(em, inner, rc, opid)
        tag = em.fresh("fg")
        for mi, (mop, needle, ci) in enumerate(members):
            if mop == "strfind":
                em.w("long long %s_%d = -1;" % (tag, mi))
            else:
                em.w("bool %s_%d = false;" % (tag, mi))
        # distinct first-char detectors: ("ci", lower_byte) uses the
        # bit5-insensitive compare (may over-match punctuation pairs; the
        # window verification rejects those)
        dets = []
        for mop, needle, ci in members:
            c0 = needle[0]
            d = ("ci", ord(c0)) if (ci and "a" <= c0 <= "z") else ("ex", ord(c0))
            if d not in dets:
                dets.append(d)
        em.w("{  // fused scan over %s (%d needles)" % (hv, len(members)))
        em.w("  const char* _p = %s.p; long long _n = %s.n;" % (hv, hv))
        em.w("  unsigned _pend = %du;" % ((1 << len(members)) - 1))
        em.w("  long long _i = 0;")

        def bcmp(expr, byte_ch, ci):
            b = ord(byte_ch)
            if ci and "a" <= byte_ch <= "z":
                return "(%s | 0x20) == %d" % (expr, b)
            return "%s == %d" % (expr, b)

        def checks(pos, indent):
            for mi, (mop, needle, ci) in enumerate(members):
                nb = needle
                cond = ["(_pend & %du)" % (1 << mi),
                        "%s + %d <= _n" % (pos, len(nb.encode())),
                        bcmp("_ch", nb[0], ci)]
                for j, c in enumerate(nb[1:], start=1):
                    cond.append(bcmp("(unsigned char)_p[%s + %d]" % (pos, j),
                                     c, ci))
                body = ("%s_%d = %s; _pend &= ~%du;"
                        % (tag, mi, pos if mop == "strfind" else "true",
                           1 << mi))
                em.w("%sif (%s) { %s }" % (indent, " && ".join(cond), body))

        def det_expr(v, d):
            kind, b = d
            if kind == "ci":
                return ("tpx_swar_zero((%s | (TPX_SWAR_ONE * 0x20u)) ^"
                        " (TPX_SWAR_ONE * %du))" % (v, b))
            return "tpx_swar_zero(%s ^ (TPX_SWAR_ONE * %du))" % (v, b)

        em.w("  while (_i < _n && _pend &&"
             " (((unsigned long long)(_p + _i)) & 7)) {")
        em.w("    int _ch = (unsigned char)_p[_i];")
        checks("_i", "    ")
        em.w("    ++_i;")
        em.w("  }")
        em.w("  for (; _i + 32 <= _n && _pend; _i += 32) {")
        for k in range(5):
            # _v4 (one word past the 32-B window) feeds candidate windows at
            # the tail of _v3; buffers carry >=16 B tail slack so the 8-B
            # overread at _i+32 <= _n stays in bounds
            em.w("    unsigned long long _v%d ="
                 " *(const unsigned long long*)(_p + _i + %d);" % (k, 8 * k))
        for k in range(4):
            em.w("    unsigned long long _c%d = %s;"
                 % (k, " | ".join(det_expr("_v%d" % k, d) for d in dets)))
        em.w("    if (_c0 | _c1 | _c2 | _c3) {")
        for k in range(4):
            em.w("      while (_c%d && _pend) {" % k)
            em.w("        int _o = (__ffsll((long long)_c%d) - 1) >> 3;"
                 " _c%d &= _c%d - 1;" % (k, k, k))
            em.w("        long long _j = _i + %d + _o;" % (8 * k))
            em.w("        int _sh = _o << 3;")
            em.w("        unsigned long long _w = _sh ? ((_v%d >> _sh) |"
                 " (_v%d << (64 - _sh))) : _v%d;" % (k, k + 1, k))
            for mi, (mop, needle, ci) in enumerate(members):
                nb = needle.encode()
                P, M, lm, wk = self._needle_window(needle, ci)
                cond = ["(_pend & %du)" % (1 << mi),
                        "_j + %d <= _n" % len(nb)]
                if M:
                    cond.append("((_w | 0x%xULL) & 0x%xULL) == 0x%xULL"
                                % (M, lm, P | M))
                else:
                    cond.append("(_w & 0x%xULL) == 0x%xULL" % (lm, P))
                for j in range(8, len(nb)):  # >8-B needles: byte-verify tail
                    cond.append(bcmp("(unsigned char)_p[_j + %d]" % j,
                                     needle[j], ci))
                body = ("%s_%d = %s; _pend &= ~%du;"
                        % (tag, mi, "_j" if mop == "strfind" else "true",
                           1 << mi))
                em.w("        if (%s) { %s }" % (" && ".join(cond), body))
            em.w("      }")
        em.w("    }")
        em.w("  }")
        em.w("  for (; _i + 8 <= _n && _pend; _i += 8) {")
        em.w("    unsigned long long _v = *(const unsigned long long*)(_p + _i);")
        hits = " | ".join(det_expr("_v", d) for d in dets)
        em.w("    unsigned long long _hit = %s;" % hits)
        em.w("    if (_hit) {")
        em.w("      for (int _b = 0; _b < 8; ++_b) {")
        em.w("        long long _j = _i + _b;")
        em.w("        int _ch = (unsigned char)_p[_j];")
        checks("_j", "        ")
        em.w("      }")
        em.w("    }")
        em.w("  }")
        em.w("  for (; _i < _n && _pend; ++_i) {")
        em.w("    int _ch = (unsigned char)_p[_i];")
        checks("_i", "    ")
        em.w("  }")
        em.w("}")
        my_var = None
        ik = self._struct_key(inner)
        for mi, (mop, needle, ci) in enumerate(members):
            key = ("scanm", mop, needle, ci, ik)
            em.memo_key(key, ("%s_%d" % (tag, mi), None))
            if (mop, needle, ci) == want:
                my_var = ("%s_%d" % (tag, mi), None)
        return my_var

    def _num(self, x, node, outt):
        if outt == T.F64 and T.deopt(node["t"]) in (T.I64, T.BOOL):
            return "(double)(%s)" % x
        return x

    def _mixed(self, args):
        ts = {T.deopt(a["t"]) for a in args}
        return T.F64 in ts and (T.I64 in ts or T.BOOL in ts)

    def _opteq(self, em, n, rc, opid):
        a, b = n["args"]
        va, na = self.emit_expr(em, a, rc, opid)
        vb, nb = self.emit_expr(em, b, rc, opid)
        na = na or "false"
        nb = nb or "false"
        base = T.deopt(a["t"]) if a["t"] != T.NULL else T.deopt(b["t"])
        r = em.fresh("b")
        if a["t"] == T.NULL and b["t"] == T.NULL:
            em.w("bool %s = true;" % r)
        elif base == T.STR:
            em.w("bool %s = (%s || %s) ? (%s && %s) : tpx_streq(%s, %s);"
                 % (r, na, nb, na, nb, va, vb))
        else:
            em.w("bool %s = (%s || %s) ? (%s && %s) : ((%s) == (%s));"
                 % (r, na, nb, na, nb, va, vb))
        if n["op"] == "optne":
            em.w("%s = !%s;" % (r, r))
        return (r, None)

    def _to_str(self, em, n, rc, opid):
        arg = n["args"][0]
        v, nv = self.emit_expr(em, arg, rc, opid)
        t = arg["t"]
        base = T.deopt(t)
        r = em.fresh("s")

        def conv(dst, val):
            if base == T.STR:
                em.w("%s = %s;" % (dst, val))
            elif base == T.I64:
                em.w("%s = tpx_str_i64(heap, %s, &_ec);" % (dst, val))
                self._check(em, opid)
            elif base == T.BOOL:
                em.w("%s = tpx_str_bool(%s);" % (dst, val))
            elif t == T.NULL:
                em.w("%s = tpx_str_none();" % dst)
            else:
                raise CodegenError("str() of %r unsupported on device (f64 repr "
                                   "needs ryu parity — route to fallback)" % (t,))

        em.w("tstr %s;" % r)
        if t == T.NULL:
            em.w("%s = tpx_str_none();" % r)
        elif nv is not None and nv != "false":
            em.w("if (%s) { %s = tpx_str_none(); } else {" % (nv, r))
            em.indent += 1
            conv(r, v)
            em.indent -= 1
            em.w("}")
        else:
            conv(r, v)
        return (r, None)

    def _assign_opt(self, em, r, rn, v, nv, src_t, dst_t):
        cast = ""
        if T.deopt(dst_t) == T.F64 and T.deopt(src_t) in (T.I64, T.BOOL):
            cast = "(double)"
        if src_t == T.NULL:
            em.w("%s = true;" % rn)
        else:
            em.w("%s = %s(%s);" % (r, cast, v))
            if rn:
                em.w("%s = %s;" % (rn, nv if nv else "false"))

    # ---- operator chain -----------------------------------------------------
    def emit_process(self, em: _Emitter):
        """Emit the body of tpx_process: takes row_ctx for stage inputs (set up by
        the caller), runs ops, fills Out o. Returns the final row ctx."""
        sp = self.sp
        rc = self.row_ctx_initial
        for op in sp.ops:
            if op.kind == "map":
                root = op.tir
                if root["op"] == "mktuple":
                    new = []
                    for el in root["args"]:
                        v, nv = self.emit_expr(em, el, rc, op.opid)
                        new.append((v, el["t"], nv))
                    rc = new
                else:
                    v, nv = self.emit_expr(em, root, rc, op.opid)
                    rc = [(v, root["t"], nv)]
            elif op.kind == "filter":
                v, _ = self.emit_expr(em, op.tir, rc, op.opid)
                em.w("if (!(%s)) { o.keep = false; return 0; }" % v)
            elif op.kind == "withColumn":
                v, nv = self.emit_expr(em, op.tir, rc, op.opid)
                cols = op.in_columns or ["column%d" % i for i in range(len(rc))]
                if op.col in cols:
                    i = cols.index(op.col)
                    rc = rc[:i] + [(v, op.tir["t"], nv)] + rc[i + 1:]
                else:
                    rc = rc + [(v, op.tir["t"], nv)]
            elif op.kind == "mapColumn":
                i = op.in_columns.index(op.col)
                sub_rc = [rc[i]]
                v, nv = self.emit_expr(em, op.tir, sub_rc, op.opid)
                rc = rc[:i] + [(v, op.tir["t"], nv)] + rc[i + 1:]
            elif op.kind == "selectColumns":
                rc = [rc[i] for i in op.sel_idxs]
            elif op.kind == "renameColumn":
                pass
            elif op.kind == "join":
                rc = self._emit_join(em, op, rc)
            else:
                raise CodegenError("op %r not supported in codegen" % op.kind)
        if getattr(sp, "agg_expr", None) is not None:
            # per-row aggregate expr (the fold's rhs); device reduce sums it.
            # by-key: emit (key, expr) — the hash-reduce kernel consumes both
            v, nv = self.emit_expr(em, sp.agg_expr, rc, sp.agg_opid)
            if getattr(sp, "agg_key_idx", None) is not None:
                rc = [rc[sp.agg_key_idx], (v, sp.agg_expr["t"], nv)]
            else:
                rc = [(v, sp.agg_expr["t"], nv)]
        return rc

    # ---- quote-freedom provenance (csv sink) -------------------------------
    # An output string cell needs the RFC-4180 quote scan only if it may
    # contain ',', '"', CR or LF. Values derived from an input CSV cell whose
    # walk saw none of those (tpx_cell flags bit 8 clear) provably need no
    # quoting, so the size pass can skip its per-string SWAR scan entirely —
    # the reference's generated CSV writer pays the same scan per cell
    # (PipelineBuilder buildWithCSVRowWriter), so this is a pure GPU-side win.
    # Lattice: True (never needs quoting) / False (unknown -> scan) /
    # frozenset of source cell indices (clean iff all those cells were clean).

    @staticmethod
    def _qfree_and(a, b):
        if a is False or b is False:
            return False
        if a is True:
            return b
        if b is True:
            return a
        return a | b

    @staticmethod
    def _qfree_static_str(s):
        return not any(c in s for c in ',"\r\n')

    def _qfree_node(self, n, env):
        op = n["op"]
        a = n["args"]
        if op == "input":
            return env[n["i"]]
        if op == "const":
            v = n.get("v")
            if isinstance(v, str):
                return self._qfree_static_str(v)
            return True  # numeric/bool/None render as digit/letter literals
        if op in ("slice", "getitem", "strip", "lower", "upper", "swapcase",
                  "splitget"):
            return self._qfree_node(a[0], env)
        if op == "concat":
            return self._qfree_and(self._qfree_node(a[0], env),
                                   self._qfree_node(a[1], env))
        if op == "center":
            x, _ = ev(a[0])
            w, _ = ev(a[1])
            f = None
            if len(a) == 3:
                f, _ = ev(a[2])
            r = em.fresh("s")
            em.w("tstr %s = tpx_center(heap, %s, (long long)%s, %s, &_ec);"
                 % (r, x, w, f if f else 'tstr{" ", 1}'))
            self._check(em, opid)
            return (r, None)
        if op == "strmul":
            return self._qfree_node(a[0], env)
        if op == "replace":
            rep = a[2]
            rep_ok = (rep["op"] == "const" and isinstance(rep.get("v"), str)
                      and self._qfree_static_str(rep["v"]))
            return self._qfree_and(self._qfree_node(a[0], env),
                                   True if rep_ok else False)
        if op == "to_str":
            from . import ttypes as _T
            if _T.deopt(a[0]["t"]) == _T.STR:
                return self._qfree_node(a[0], env)
            return True  # int/bool/None texts are clean
        if op == "fmt_int":
            return True
        if op == "if":
            return self._qfree_and(self._qfree_node(a[1], env),
                                   self._qfree_node(a[2], env))
        return False

    def _thread_qfree(self):
        """Per-output-column qfree lattice (csv sink; None when untracked)."""
        sp = self.sp
        if getattr(sp, "agg_expr", None) is not None or \
                self._dup_join_op() is not None:
            return None
        env = []
        for i, t in enumerate(sp.input_types):
            if T.deopt(t) == T.STR:
                env.append(frozenset([i]) if self.source == "csv"
                           and not self.csv_info.get("text_mode") else False)
            else:
                env.append(True)
        for op in sp.ops:
            if op.kind == "map":
                if op.tir["op"] == "mktuple":
                    env = [self._qfree_node(el, env) for el in op.tir["args"]]
                else:
                    env = [self._qfree_node(op.tir, env)]
            elif op.kind == "filter":
                pass
            elif op.kind == "withColumn":
                v = self._qfree_node(op.tir, env)
                cols = op.in_columns or ["column%d" % i
                                         for i in range(len(env))]
                if op.col in cols:
                    i = cols.index(op.col)
                    env = env[:i] + [v] + env[i + 1:]
                else:
                    env = env + [v]
            elif op.kind == "mapColumn":
                i = op.in_columns.index(op.col)
                env = env[:i] + [self._qfree_node(op.tir, [env[i]])] + \
                    env[i + 1:]
            elif op.kind == "selectColumns":
                env = [env[i] for i in op.sel_idxs]
            elif op.kind == "renameColumn":
                pass
            elif op.kind == "join":
                rrows, rcols, lki, rki, how = op.join
                right = []
                for j in range(len(rcols)):
                    if j == rki:
                        continue
                    right.append(all(not isinstance(r[j], str)
                                     or self._qfree_static_str(r[j])
                                     for r in rrows))
                env = [env[i] for i in range(len(env)) if i != lki] + \
                    [env[lki]] + right
            else:
                return None
        return env

    def _qfree_cond(self, q):
        """C condition that the value is quote-free, or None (must scan)."""
        if q is True:
            return "true"
        if q is False or q is None:
            return None
        if self.split:
            # split mode: the parse kernel folded (flags & 9) per string cell
            # into the per-row dirty mask (cells >= 64 untracked -> scan)
            if any(i >= 64 for i in q):
                return None
            return " && ".join("(((_dirty >> %d) & 1) == 0)" % i
                               for i in sorted(q))
        # flags bit0 quoted | bit3(8) content-special: both clear == clean
        return " && ".join("((cl%d.flags & 9) == 0)" % i for i in sorted(q))

    # ---- full source --------------------------------------------------------
    def _dup_join_op(self):
        """The terminal duplicate-key join op, if this stage has one."""
        sp = self.sp
        if sp.ops and sp.ops[-1].kind == "join" and \
                getattr(sp.ops[-1], "join_dup", False):
            return sp.ops[-1]
        return None

    def generate(self) -> Tuple[str, str]:
        """Returns (hip_source, stage_desc)."""
        sp = self.sp
        in_types = sp.input_types
        out_types = sp.gpu_output_types if getattr(sp, "agg_expr", None) is not None \
            else sp.output_types

        dup_op = self._dup_join_op()
        # terminal dup join: the process stores LEFT columns + a hidden
        # (bucket_start, count) pair; the write kernel loops the bucket and
        # reads the right columns from the embedded table (1:N expansion)
        store_types = out_types
        if dup_op is not None:
            n_leftout = len(dup_op.in_types)
            store_types = list(out_types[:n_leftout]) + [T.I64, T.I64]

        body = _Emitter()
        # inputs: c0..cN (+ c0_n null flags)
        self.row_ctx_initial = []
        for i, t in enumerate(in_types):
            nv = ("c%d_n" % i) if T.is_opt(t) else None
            self.row_ctx_initial.append(("c%d" % i, t, nv))
        final_rc = self.emit_process(body)

        # write outputs into Out struct
        out_fields = []
        for k, t in enumerate(store_types):
            ct = _ctype(t)
            out_fields.append("    %s o%d;" % (ct, k))
            if T.is_opt(t):
                out_fields.append("    bool o%d_n;" % k)
        n_rc = len(store_types) - (2 if dup_op is not None else 0)
        for k, ((v, t, nv), ot) in enumerate(zip(final_rc[:n_rc],
                                                 store_types[:n_rc])):
            cast = "(double)" if T.deopt(ot) == T.F64 and T.deopt(t) in (T.I64, T.BOOL) else ""
            body.w("o.o%d = %s(%s);" % (k, cast, v))
            if T.is_opt(ot):
                body.w("o.o%d_n = %s;" % (k, nv if nv else "false"))
        if dup_op is not None:
            body.w("o.o%d = (long long)%s;" % (n_rc, self.jdup["start_var"]))
            body.w("o.o%d = (long long)%s;" % (n_rc + 1, self.jdup["cnt_var"]))
        body.w("o.keep = true;")
        body.w("return 0;")

        # generate kernels BEFORE assembling (they may add string literals)
        self.store_types = store_types
        self.full_out_types = out_types
        if self.split:
            main_src = ("#define TPX_SPAN_CAP %d\n" % self.SPAN_CAP +
                        self._parse_kernel(in_types) + "\n\n" +
                        self._main_kernel_split(in_types, store_types))
        else:
            main_src = self._main_kernel(in_types, store_types)
        if dup_op is None:
            write_src = self._write_kernel(out_types)
        elif self.sink == "mem":
            write_src = self._write_kernel_mem_dup(out_types, store_types)
        else:
            write_src = self._write_kernel_csv_dup(out_types, store_types)
        src = [
            "// generated by tuplex_amd.codegen — stage %s" % sp.signature(),
            runtime_header(),
            "",
        ] + self.lit_defs + self.join_defs + [
            "",
            "struct Out {",
        ] + out_fields + [
            "    bool keep;",
            "};",
            "",
            self._process_signature(in_types),
            "    int _ec = 0;",
        ] + body.lines + [
            "}",
            "",
            main_src,
            write_src,
        ]
        desc = self._desc(in_types, store_types)
        return "\n".join(src), desc

    def _process_signature(self, in_types):
        params = []
        for i, t in enumerate(in_types):
            params.append("%s c%d" % (_ctype(t), i))
            if T.is_opt(t):
                params.append("bool c%d_n" % i)
        params.append("TpxHeap& heap")
        params.append("Out& o")
        return ("__device__ __forceinline__ long long tpx_process(%s) {"
                % ", ".join(params))

    # -- main kernel ----------------------------------------------------------
    # Wave-cooperative LDS staging: each wave stages the byte span of its 64
    # consecutive rows into LDS with coalesced uint4 loads, then every lane
    # parses its row from LDS. Without this, per-thread byte scanning re-reads
    # rows through a thrashed L1/L2 (measured 7x HBM read amplification:
    # 2048 threads x ~200 B rows >> 32 KiB L1 per CU). Waves whose span exceeds
    # TPX_SPAN_CAP fall back to parsing from global memory (rare: long rows).
    # bytes per wave: 2 waves/block stage 2*CAP+80 B of LDS; 16344 keeps the
    # block at exactly 32768 B so FIVE blocks fit the 160 KiB CU budget ->
    # 10 waves/CU (at 16384+pad the block tips over 32 KiB and occupancy
    # drops to 4 blocks = 8 waves/CU). The 80-B tail absorbs the mask walk's
    # aligned 64-B group overread past the last row of a wave's span.
    SPAN_CAP = 16344

    def _main_kernel(self, in_types, out_types):
        if self.source == "col":
            return self._main_kernel_col(in_types, out_types)
        L = []
        L.append("#define TPX_SPAN_CAP %d" % self.SPAN_CAP)
        # occupancy is LDS-limited to ~2.5 waves/SIMD (5 blocks x 32 KiB); tell
        # the compiler the real VGPR budget instead of spilling ~284 B/thread
        # to scratch at a 64-VGPR default target (waves-per-SIMD tunable via
        # TPX_MAIN_LB: 0 = no bound)
        import os as _os
        lb = int(_os.environ.get("TPX_MAIN_LB", "2"))
        bound = " __launch_bounds__(128, %d)" % lb if lb else ""
        L.append('extern "C" __global__ void%s tpx_stage_main(' % bound)
        L.append("    const unsigned char* __restrict__ in_data,")
        L.append("    const long long* __restrict__ in_offs,")
        L.append("    long long n, long long row0,")
        L.append("    char* heap_base, unsigned long long* heap_cursor,"
                 " unsigned long long heap_cap,")
        L.append("    unsigned char* __restrict__ keep, long long* __restrict__ keep01,")
        L.append("    long long* __restrict__ sizes,")
        L.append("    long long* exc_buf, unsigned long long* exc_count,"
                 " unsigned long long exc_cap,")
        L.append("    void** outv) {")
        L.append("  __shared__ char smem[2 * TPX_SPAN_CAP + 80];  // 2 waves per 128-thread block -> 5 blocks/CU")
        L.append("  TpxHeap heap{heap_base, heap_cursor, heap_cap, nullptr, nullptr};")
        L.append("  int lane = threadIdx.x & 63;")
        L.append("  int wid = threadIdx.x >> 6;")
        L.append("  char* wave_lds = smem + wid * TPX_SPAN_CAP;")
        L.append("  long long wave_stride = (long long)gridDim.x * (blockDim.x >> 6);")
        L.append("  long long nwaves = (n + 63) >> 6;")
        L.append("  for (long long wb = (long long)blockIdx.x * (blockDim.x >> 6) + wid;"
                 " wb < nwaves; wb += wave_stride) {")
        L.append("    long long r0 = wb << 6;")
        L.append("    long long rhi = r0 + 64 < n ? r0 + 64 : n;")
        L.append("    long long span_start = in_offs[r0] & ~15LL;  // align staging window")
        L.append("    long long span_end = in_offs[rhi];")
        L.append("    long long span = span_end - span_start;")
        # NB: never form (wave_lds - span_start): an LDS generic pointer offset
        # outside the aperture is UB once the addrspace-inference pass narrows it
        L.append("    bool staged = span <= TPX_SPAN_CAP;")
        L.append("    if (staged) {")
        L.append("      for (long long k = (long long)lane * 16; k < span; k += 64 * 16) {")
        L.append("        if (k + 16 <= span)")
        L.append("          *(uint4*)(wave_lds + k) = *(const uint4*)((const char*)in_data + span_start + k);")
        L.append("        else")
        L.append("          for (long long j = k; j < span; ++j)")
        L.append("            wave_lds[j] = ((const char*)in_data)[span_start + j];")
        L.append("      }")
        L.append("    }")
        L.append("    long long i = r0 + lane;")
        L.append("    if (i >= rhi) continue;")
        # Default: duplicate the row body per pointer mode — in the staged
        # branch every parse pointer provably derives from LDS, so address
        # space inference emits ds_read and no generic pointers exist. The
        # single generic-pointer body (TPX_GEN_BODY=1) halves kernel size and
        # hipRTC compile time but MISCOMPILES on hipRTC gfx950 (mem-path null
        # VAs survived three opaque-pointer workarounds; see commits
        # f347c0f/cee4b86/08e5bef) — kept only for offline investigation.
        if _os.environ.get("TPX_GEN_BODY") == "1":
            L.extend(self._row_body(in_types, out_types, lds="gen"))
        else:
            L.append("    if (staged) {")
            L.extend("  " + ln
                     for ln in self._row_body(in_types, out_types, lds=True))
            L.append("    } else {")
            L.extend("  " + ln
                     for ln in self._row_body(in_types, out_types, lds=False))
            L.append("    }")
        L.append("  }")
        L.append("}")
        return "\n".join(L)

    # -- split parse/UDF kernels (csv source) --------------------------------
    # tpx_stage_parse: the LDS-staged cell walk + typed parse, writing a dense
    # columnar "cell park": typed values for numeric columns, and string cell
    # bytes copied into a per-wave bump-allocated byte park (each wave
    # reserves one contiguous region for its 64 rows, so the UDF kernel's
    # waves — the same 64-row grouping — read a ~13 KB window per wave).
    # tpx_stage_main then runs the UDF chain as a plain grid-stride kernel
    # over the park at full occupancy: the fused kernel was measured
    # latency-bound (53-60% SQ WAIT_ANY at its LDS-capped 10 waves/CU).

    def _parse_kernel(self, in_types):
        used = getattr(self.sp, "used_source_cols", None)
        if used is None:
            used = set(range(len(in_types)))
        L = []
        L.append('extern "C" __global__ void __launch_bounds__(128, 2)'
                 ' tpx_stage_parse(')
        L.append("    const unsigned char* __restrict__ in_data,")
        L.append("    const long long* __restrict__ in_offs,")
        L.append("    long long n, long long row0,")
        L.append("    char* __restrict__ strbuf,"
                 " unsigned long long* __restrict__ str_cursor,")
        L.append("    void** pv, long long* __restrict__ prc_out,")
        L.append("    unsigned long long* __restrict__ dirty_out) {")
        L.append("  __shared__ char smem[2 * TPX_SPAN_CAP + 80];")
        L.append("  int lane = threadIdx.x & 63;")
        L.append("  int wid = threadIdx.x >> 6;")
        L.append("  char* wave_lds = smem + wid * TPX_SPAN_CAP;")
        L.append("  long long wave_stride = (long long)gridDim.x * (blockDim.x >> 6);")
        L.append("  long long nwaves = (n + 63) >> 6;")
        L.append("  for (long long wb = (long long)blockIdx.x * (blockDim.x >> 6) + wid;"
                 " wb < nwaves; wb += wave_stride) {")
        L.append("    long long r0 = wb << 6;")
        L.append("    long long rhi = r0 + 64 < n ? r0 + 64 : n;")
        L.append("    long long span_start = in_offs[r0] & ~15LL;")
        L.append("    long long span_end = in_offs[rhi];")
        L.append("    long long span = span_end - span_start;")
        L.append("    bool staged = span <= TPX_SPAN_CAP;")
        L.append("    if (staged) {")
        L.append("      for (long long k = (long long)lane * 16; k < span; k += 64 * 16) {")
        L.append("        if (k + 16 <= span)")
        L.append("          *(uint4*)(wave_lds + k) = *(const uint4*)((const char*)in_data + span_start + k);")
        L.append("        else")
        L.append("          for (long long j = k; j < span; ++j)")
        L.append("            wave_lds[j] = ((const char*)in_data)[span_start + j];")
        L.append("      }")
        L.append("    }")
        # inactive lanes redundantly parse row r0 (always a valid row) so the
        # whole wave can join the park prefix sum; their stores are masked
        L.append("    long long i = r0 + lane;")
        L.append("    bool _act = i < rhi;")
        L.append("    if (!_act) i = r0;")
        body_store = self._store_parsed(in_types, used)
        L.append("    if (staged) {")
        L.extend("  " + ln for ln in
                 self._load_inputs_csv(in_types, lds=True) + body_store)
        L.append("    } else {")
        L.extend("  " + ln for ln in
                 self._load_inputs_csv(in_types, lds=False) + body_store)
        L.append("    }")
        L.append("  }")
        L.append("}")
        return "\n".join(L)

    def _store_parsed(self, in_types, used):
        """Park one parsed row: typed/value stores (+ wave-cooperative bump
        reservation of the byte park in park_copy mode). Runs inside the
        (dual-body) parse kernel with c0..cN, cl0..clN, prc in scope."""
        str_used = [i for i, t in enumerate(in_types)
                    if i in used and T.deopt(t) == T.STR]
        if not self.park_copy:
            return self._store_parsed_nocopy(in_types, used, str_used)
        L = []
        L.append("    long long _sb = 0;")
        if str_used:
            L.append("    if (_act && !prc) _sb = 8 %s;"
                     % "".join(" + c%d.n" % i for i in str_used))
        # wave-exclusive prefix sum (shfl ladder) + one atomic per wave
        L.append("    unsigned long long _v = (unsigned long long)_sb;")
        L.append("    for (int _s = 1; _s < 64; _s <<= 1) {")
        L.append("      unsigned long long _u = __shfl_up(_v, _s);")
        L.append("      if (lane >= _s) _v += _u;")
        L.append("    }")
        L.append("    unsigned long long _off = _v - (unsigned long long)_sb;")
        L.append("    unsigned long long _tot = __shfl(_v, 63);")
        L.append("    unsigned long long _base = 0;")
        L.append("    if (lane == 0 && _tot) _base = atomicAdd(str_cursor, _tot);")
        L.append("    _base = __shfl(_base, 0);")
        L.append("    char* _dst = strbuf + _base + _off;")
        for idx, t in enumerate(in_types):
            if idx not in used:
                continue
            base = T.deopt(t)
            opt = T.is_opt(t)
            if base == T.STR:
                L.append("    if (_act) {")
                L.append("      ((unsigned long long*)pv[%d])[i] ="
                         " (unsigned long long)_dst;" % (3 * idx))
                L.append("      ((int*)pv[%d])[i] = prc ? 0 : (int)c%d.n;"
                         % (3 * idx + 1, idx))
                L.append("      if (!prc) {")
                L.append("        const char* _s = c%d.p; long long _n = c%d.n;"
                         % (idx, idx))
                L.append("        long long _k = 0;")
                L.append("        for (; _k + 8 <= _n; _k += 8)")
                L.append("          *(unsigned long long*)(_dst + _k) ="
                         " *(const unsigned long long*)(_s + _k);")
                L.append("        for (; _k < _n; ++_k) _dst[_k] = _s[_k];")
                L.append("        _dst += _n;")
                L.append("      }")
                L.append("    }")
            elif base == T.F64:
                L.append("    if (_act) ((double*)pv[%d])[i] = c%d;"
                         % (3 * idx, idx))
            else:  # I64 / BOOL both park as i64
                L.append("    if (_act) ((long long*)pv[%d])[i] ="
                         " (long long)c%d;" % (3 * idx, idx))
            if opt:
                L.append("    if (_act) ((unsigned char*)pv[%d])[i] ="
                         " c%d_n ? 1 : 0;" % (3 * idx + 2, idx))
        dirty_bits = ["((unsigned long long)((cl%d.flags & 9) != 0) << %d)"
                      % (i, i) for i in str_used if i < 64]
        L.append("    if (_act) {")
        L.append("      prc_out[i] = prc;")
        L.append("      dirty_out[i] = %s;"
                 % (" | ".join(dirty_bits) if dirty_bits else "0"))
        L.append("    }")
        return L

    def _store_parsed_nocopy(self, in_types, used, str_used):
        """Zero-copy park: string cells stored as (global ptr, len) views into
        the device input bytes — staged-branch LDS pointers translate via
        tpx_to_global. No byte park, no cursor."""
        L = []
        for idx, t in enumerate(in_types):
            if idx not in used:
                continue
            base = T.deopt(t)
            opt = T.is_opt(t)
            if base == T.STR:
                L.append("    if (_act) {")
                L.append("      tstr _g%d = c%d;" % (idx, idx))
                if True:  # staged-branch pointers live in LDS
                    L.append("      _g%d = tpx_to_global(_g%d, wave_lds,"
                             " wave_lds + TPX_SPAN_CAP, in_data,"
                             " span_start);" % (idx, idx))
                L.append("      ((unsigned long long*)pv[%d])[i] ="
                         " (unsigned long long)_g%d.p;" % (3 * idx, idx))
                L.append("      ((int*)pv[%d])[i] = prc ? 0 : (int)_g%d.n;"
                         % (3 * idx + 1, idx))
                L.append("    }")
            elif base == T.F64:
                L.append("    if (_act) ((double*)pv[%d])[i] = c%d;"
                         % (3 * idx, idx))
            else:  # I64 / BOOL both park as i64
                L.append("    if (_act) ((long long*)pv[%d])[i] ="
                         " (long long)c%d;" % (3 * idx, idx))
            if opt:
                L.append("    if (_act) ((unsigned char*)pv[%d])[i] ="
                         " c%d_n ? 1 : 0;" % (3 * idx + 2, idx))
        dirty_bits = ["((unsigned long long)((cl%d.flags & 9) != 0) << %d)"
                      % (i, i) for i in str_used if i < 64]
        L.append("    if (_act) {")
        L.append("      prc_out[i] = prc;")
        L.append("      dirty_out[i] = %s;"
                 % (" | ".join(dirty_bits) if dirty_bits else "0"))
        L.append("    }")
        return L

    def _main_kernel_split(self, in_types, out_types):
        """UDF-phase kernel over the parsed cell park: plain grid-stride, no
        LDS, full occupancy. Same fixed signature as the fused main (in_data
        carries the park slot table; in_offs still indexes the raw chunk
        bytes for exception payload ranges)."""
        import os as _os
        # waves/SIMD bound: the UDF-phase kernel genuinely needs ~198 VGPRs
        # (whole UDF chain + csv-size logic live ranges); bounding tighter
        # (128 at lb=4) spilled 272 B/thread and lost more than the
        # occupancy won. lb=2 -> no spills at 8 waves/CU.
        lb = int(_os.environ.get("TPX_SPLIT_LB", "2"))
        bound = " __launch_bounds__(256, %d)" % lb if lb else ""
        L = []
        L.append('extern "C" __global__ void%s tpx_stage_main(' % bound)
        L.append("    const unsigned char* __restrict__ in_data,")
        L.append("    const long long* __restrict__ in_offs,")
        L.append("    long long n, long long row0,")
        L.append("    char* heap_base, unsigned long long* heap_cursor,"
                 " unsigned long long heap_cap,")
        L.append("    unsigned char* __restrict__ keep, long long* __restrict__ keep01,")
        L.append("    long long* __restrict__ sizes,")
        L.append("    long long* exc_buf, unsigned long long* exc_count,"
                 " unsigned long long exc_cap,")
        L.append("    void** outv) {")
        L.append("  TpxHeap heap{heap_base, heap_cursor, heap_cap, nullptr, nullptr};")
        L.append("  long long stride = (long long)gridDim.x * blockDim.x;")
        L.append("  for (long long i = (long long)blockIdx.x * blockDim.x +"
                 " threadIdx.x; i < n; i += stride) {")
        L.extend(self._row_body(in_types, out_types, lds="park"))
        L.append("  }")
        L.append("}")
        return "\n".join(L)

    def _load_parsed(self, in_types):
        """Row loader for the UDF-phase kernel: typed coalesced loads from the
        cell park (strings are absolute (ptr,len) views into the byte park)."""
        used = getattr(self.sp, "used_source_cols", None)
        nin = len(in_types)
        L = []
        L.append("    const void* const* ct = (const void* const*)in_data;")
        L.append("    long long prc = ((const long long*)ct[%d])[i];" % (3 * nin))
        L.append("    unsigned long long _dirty ="
                 " ((const unsigned long long*)ct[%d])[i];" % (3 * nin + 1))
        L.append("    (void)_dirty;")
        for idx, t in enumerate(in_types):
            base = T.deopt(t)
            opt = T.is_opt(t)
            if used is not None and idx not in used:
                if opt:
                    L.append("    bool c%d_n = false;  // unused (pushdown)" % idx)
                if base == T.STR:
                    L.append("    tstr c%d{(const char*)in_data, 0};" % idx)
                elif base == T.I64:
                    L.append("    long long c%d = 0;" % idx)
                elif base == T.F64:
                    L.append("    double c%d = 0.0;" % idx)
                else:
                    L.append("    bool c%d = false;" % idx)
                continue
            if opt:
                L.append("    bool c%d_n = ((const unsigned char*)ct[%d])[i]"
                         " != 0;" % (idx, 3 * idx + 2))
            if base == T.STR:
                L.append("    tstr c%d{(const char*)"
                         "((const unsigned long long*)ct[%d])[i],"
                         " (long long)((const int*)ct[%d])[i]};"
                         % (idx, 3 * idx, 3 * idx + 1))
            elif base == T.I64:
                L.append("    long long c%d = ((const long long*)ct[%d])[i];"
                         % (idx, 3 * idx))
            elif base == T.F64:
                L.append("    double c%d = ((const double*)ct[%d])[i];"
                         % (idx, 3 * idx))
            else:
                L.append("    bool c%d = ((const long long*)ct[%d])[i] != 0;"
                         % (idx, 3 * idx))
        return L

    def _main_kernel_col(self, in_types, out_types):
        """Columnar-source main kernel: typed coalesced loads, no LDS staging,
        plain grid-stride over rows (same fixed signature; in_offs unused)."""
        L = []
        L.append('extern "C" __global__ void tpx_stage_main(')
        L.append("    const unsigned char* __restrict__ in_data,")
        L.append("    const long long* __restrict__ in_offs,")
        L.append("    long long n, long long row0,")
        L.append("    char* heap_base, unsigned long long* heap_cursor,"
                 " unsigned long long heap_cap,")
        L.append("    unsigned char* __restrict__ keep, long long* __restrict__ keep01,")
        L.append("    long long* __restrict__ sizes,")
        L.append("    long long* exc_buf, unsigned long long* exc_count,"
                 " unsigned long long exc_cap,")
        L.append("    void** outv) {")
        L.append("  TpxHeap heap{heap_base, heap_cursor, heap_cap, nullptr, nullptr};")
        L.append("  long long stride = (long long)gridDim.x * blockDim.x;")
        L.append("  for (long long i = (long long)blockIdx.x * blockDim.x +"
                 " threadIdx.x; i < n; i += stride) {")
        L.extend(self._row_body(in_types, out_types, lds=False))
        L.append("  }")
        L.append("}")
        return "\n".join(L)

    def _row_body(self, in_types, out_types, lds):
        L = []
        if lds == "park":
            L.extend(self._load_parsed(in_types))
        elif self.source == "csv":
            L.extend(self._load_inputs_csv(in_types, lds))
        elif self.source == "col":
            L.extend(self._load_inputs_col(in_types))
        else:
            L.extend(self._load_inputs_mem(in_types, lds))
        L.append("    Out o;")
        args = []
        for idx, t in enumerate(in_types):
            args.append("c%d" % idx)
            if T.is_opt(t):
                args.append("c%d_n" % idx)
        L.append("    long long rc = prc;")
        L.append("    if (!rc) rc = tpx_process(%s, heap, o);" % ", ".join(args))
        L.append("    if (rc != 0) {")
        L.append("      unsigned long long e = atomicAdd(exc_count, 1ULL);")
        L.append("      if (e < exc_cap) {")
        L.append("        exc_buf[e*5+0] = row0 + i;")
        L.append("        exc_buf[e*5+1] = rc & 0xFFFFFFFFLL;")
        L.append("        exc_buf[e*5+2] = rc >> 32;")
        if self.source == "col":
            # columnar source has no row bytes; the host replays from the
            # original Arrow table by row index (orcio.py)
            L.append("        exc_buf[e*5+3] = 0;")
            L.append("        exc_buf[e*5+4] = 0;")
        else:
            L.append("        exc_buf[e*5+3] = in_offs[i];")
            L.append("        exc_buf[e*5+4] = in_offs[i+1];")
        L.append("      }")
        L.append("      keep[i] = 0; keep01[i] = 0; sizes[i] = 0; continue;")
        L.append("    }")
        L.append("    if (!o.keep) { keep[i] = 0; keep01[i] = 0; sizes[i] = 0; continue; }")
        L.extend(self._store_columnar(out_types, lds))
        if self._dup_join_op() is not None:
            # keep01 = bucket count (the compaction scan then yields 1:N
            # output row offsets); left-join miss stored as count 1
            L.append("    keep[i] = 1; keep01[i] = o.o%d;"
                     % (len(out_types) - 1))
        elif self.sink == "mem":
            L.append("    keep[i] = 1; keep01[i] = 1;")
        else:
            # bit 1: no cell of this row needs quoting (write fast path;
            # _anyq computed by the size pass inside _store_columnar)
            L.append("    keep[i] = (unsigned char)(1 | (_anyq ? 0 : 2));"
                     " keep01[i] = 1;")
        return L

    def _load_inputs_col(self, in_types):
        """Columnar (Arrow-layout) source — ORC ingest (io/src/OrcTypes.cc,
        physical/OrcReader analog). in_data is a device table
        of 3 slots per column: [values-or-offsets, string-data, null-mask];
        strings are (offset[i], offset[i+1]) views into the data buffer. No
        parse, no LDS staging — loads are already typed and coalesced."""
        used = getattr(self.sp, "used_source_cols", None)
        L = ["    long long prc = 0;"]
        L.append("    const void* const* ct = (const void* const*)in_data;")
        for idx, t in enumerate(in_types):
            base = T.deopt(t)
            opt = T.is_opt(t)
            if used is not None and idx not in used:
                if opt:
                    L.append("    bool c%d_n = false;  // unused (pushdown)" % idx)
                if base == T.STR:
                    L.append("    tstr c%d{(const char*)in_data, 0};" % idx)
                elif base == T.I64:
                    L.append("    long long c%d = 0;" % idx)
                elif base == T.F64:
                    L.append("    double c%d = 0.0;" % idx)
                else:
                    L.append("    bool c%d = false;" % idx)
                continue
            if opt:
                L.append("    bool c%d_n = ((const unsigned char*)ct[%d])[i]"
                         " != 0;" % (idx, 3 * idx + 2))
            if base == T.STR:
                L.append("    long long c%d_o = ((const long long*)ct[%d])[i];"
                         % (idx, 3 * idx))
                L.append("    tstr c%d{(const char*)ct[%d] + c%d_o,"
                         " ((const long long*)ct[%d])[i + 1] - c%d_o};"
                         % (idx, 3 * idx + 1, idx, 3 * idx, idx))
                # same per-column ASCII gate as the mem loader: non-ASCII rows
                # divert (char-index-sensitive ops need it)
                guard = ("!prc && !c%d_n" % idx) if opt else "!prc"
                L.append("    if (%s && !tpx_ascii(c%d)) prc = 7;" % (guard, idx))
            elif base == T.I64:
                L.append("    long long c%d = ((const long long*)ct[%d])[i];"
                         % (idx, 3 * idx))
            elif base == T.F64:
                L.append("    double c%d = ((const double*)ct[%d])[i];"
                         % (idx, 3 * idx))
            elif base == T.BOOL:
                L.append("    bool c%d = ((const unsigned char*)ct[%d])[i]"
                         " != 0;" % (idx, 3 * idx))
            else:
                raise CodegenError("columnar input type %r" % (t,))
        return L

    def _load_inputs_mem(self, in_types, lds=True):
        """Deserialize one reference-layout row (Serializer.cc:20-24) into typed
        locals c0..cN."""
        L = ["    long long prc = 0;  // no pre-parse errors on the mem path"]
        if lds == "gen":
            # u64-opaque select (see csv loader note)
            L.append("    unsigned long long _rwa = staged ?"
                     " (unsigned long long)wave_lds + (unsigned long long)(in_offs[i]"
                     " - span_start) : (unsigned long long)(in_data +"
                     " in_offs[i]);")
            L.append('    asm volatile("" : "+v"(_rwa));'
                     "  // opaque: block addrspace re-inference")
            L.append("    const unsigned char* row ="
                     " (const unsigned char*)_rwa;")
        elif lds:
            L.append("    const unsigned char* row ="
                     " (const unsigned char*)(wave_lds + (in_offs[i] - span_start));")
        else:
            L.append("    const unsigned char* row = in_data + in_offs[i];")
        n_opt = sum(1 for t in in_types if T.is_opt(t))
        bitmap_size = ((n_opt + 63) // 64) * 8 if n_opt else 0
        L.append("    // deserialize (bitmap %dB, %d slots)" % (bitmap_size, len(in_types)))
        if bitmap_size:
            for w in range(bitmap_size // 8):
                L.append("    unsigned long long bm%d = ((const unsigned long long*)row)[%d];" % (w, w))
        opt_counter = 0
        for idx, t in enumerate(in_types):
            base = T.deopt(t)
            slot = "((const long long*)(row + %d))[%d]" % (bitmap_size, idx)
            if T.is_opt(t):
                L.append("    bool c%d_n = (bm%d >> %d) & 1;"
                         % (idx, opt_counter // 64, opt_counter % 64))
                opt_counter += 1
            if base == T.I64:
                L.append("    long long c%d = %s;" % (idx, slot))
            elif base == T.BOOL:
                L.append("    bool c%d = %s != 0;" % (idx, slot))
            elif base == T.F64:
                L.append("    double c%d = __longlong_as_double(%s);" % (idx, slot))
            elif base == T.STR:
                L.append("    long long info%d = %s;" % (idx, slot))
                L.append("    tstr c%d{(const char*)(row + %d + %d) + (info%d & 0xFFFFFFFFLL),"
                         " (info%d >> 32) - 1};" % (idx, bitmap_size, idx * 8, idx, idx))
                if T.is_opt(t):
                    L.append("    if (c%d_n) c%d = tstr{(const char*)row, 0};" % (idx, idx))
                # per-column ASCII gate (mem rows hold binary i64/f64 slots, so
                # only the string bytes are scanned) — NCV divert, once per col
                L.append("    if (!prc && !tpx_ascii(c%d)) prc = 7;" % idx)
            else:
                raise CodegenError("input type %r" % (t,))
        return L

    def _load_inputs_csv(self, in_types, lds=True):
        """Single-pass cell split + typed parse specialised to the sniffed schema —
        the CSVParseRowGenerator.cc replacement. Structure/parse failures set prc
        (BADPARSE/UNDERRUN/OVERRUN; raw line becomes the exception payload, like
        the reference's BADPARSE_STRING_INPUT rows)."""
        if self.csv_info.get("text_mode"):
            return self._load_inputs_text(in_types, lds)
        nc = len(in_types)
        null_values = self.csv_info.get("null_values", [""])
        delim = self.csv_info.get("delimiter", ",")
        assert len(delim) == 1
        delim_c = "'\\t'" if delim == "\t" else "'%s'" % delim
        L = []
        L.append("    long long prc = 0;")
        if lds == "gen":
            # the staged/global pointer select goes through u64: a mixed
            # LDS/global pointer select lets InferAddressSpaces collapse the
            # generic pointer (observed: 0xFFFFFFFF_xxxxxxx VAs escaping) —
            # integer selects are opaque to it, flat loads handle either space
            L.append("    unsigned long long _rpa = staged ?"
                     " (unsigned long long)wave_lds + (unsigned long long)(in_offs[i]"
                     " - span_start) : (unsigned long long)((const char*)"
                     "in_data + in_offs[i]);")
            L.append("    unsigned long long _rea = staged ?"
                     " (unsigned long long)wave_lds + (unsigned long long)(in_offs[i+1]"
                     " - span_start) : (unsigned long long)((const char*)"
                     "in_data + in_offs[i+1]);")
            L.append('    asm volatile("" : "+v"(_rpa), "+v"(_rea));'
                     "  // opaque: block addrspace re-inference")
            L.append("    const char* rp = (const char*)_rpa;")
            L.append("    const char* rend = (const char*)_rea;")
        elif lds:
            L.append("    const char* rp = wave_lds + (in_offs[i] - span_start);")
            L.append("    const char* rend = wave_lds + (in_offs[i+1] - span_start);")
        else:
            L.append("    const char* rp = (const char*)in_data + in_offs[i];")
            L.append("    const char* rend = (const char*)in_data + in_offs[i+1];")
        L.append("    if (rend > rp && rend[-1] == '\\n') --rend;")
        L.append("    if (rend > rp && rend[-1] == '\\r') --rend;")
        # straight-line per-cell walk over lazily-loaded 64-B group bitmasks
        # (tpx_mwalk): one named local per cell — a dynamically indexed cells[]
        # array spills to scratch (288 B/thread measured). The masks replace
        # the serial per-cell memchr chain with 8-independent-load groups +
        # ALU bit derivation; the ASCII gate and output-special (quote
        # provenance) tracking ride along for free.
        chk_comma = 1 if delim != "," else 0
        L.append("    tpx_mwalk S; tpx_mw_init(S, rp, rend);")
        L.append("    bool avail = true;")
        L.append("    int badf = 0;")

        def null_check(idx):
            cv = "tstr{cl%d.p, cl%d.n}" % (idx, idx)
            checks = " || ".join("tpx_streq(%s, %s)" % (cv, self.lit(nv))
                                 for nv in null_values) or "false"
            return checks

        # projection pushdown: unused columns are still cell-walked (structure
        # + column-count errors keep reference semantics) but their typed
        # parse / null check is skipped (plan._used_source_columns)
        used = getattr(self.sp, "used_source_cols", None)
        for idx, t in enumerate(in_types):
            base = T.deopt(t)
            opt = T.is_opt(t)
            L.append("    tpx_cell cl%d{rp, 0, 0};" % idx)
            L.append("    if (!prc) {")
            L.append("      if (!avail) prc = %d;  // CSV_UNDERRUN" % 20)
            L.append("      else { tpx_mw_cell(S, &cl%d, %s, %d);"
                     " avail = S.more; badf |= cl%d.flags; }"
                     % (idx, delim_c, chk_comma, idx))
            L.append("    }")
            if used is not None and idx not in used:
                if opt:
                    L.append("    bool c%d_n = false;  // unused (pushdown)" % idx)
                if base == T.STR:
                    L.append("    tstr c%d{cl%d.p, cl%d.n};" % (idx, idx, idx))
                    L.append("    if (prc) c%d = tstr{rp, 0};" % idx)
                elif base == T.I64:
                    L.append("    long long c%d = 0;  // unused (pushdown)" % idx)
                elif base == T.F64:
                    L.append("    double c%d = 0.0;  // unused (pushdown)" % idx)
                else:
                    L.append("    bool c%d = false;  // unused (pushdown)" % idx)
                continue
            if opt:
                L.append("    bool c%d_n = !prc && (%s);" % (idx, null_check(idx)))
            guard = ("!prc && !c%d_n" % idx) if opt else "!prc"
            if base == T.STR:
                L.append("    tstr c%d{cl%d.p, cl%d.n};" % (idx, idx, idx))
                L.append("    if (prc) c%d = tstr{rp, 0};" % idx)
                if opt:
                    L.append("    if (c%d_n) c%d = tstr{rp, 0};" % (idx, idx))
            elif base == T.I64:
                L.append("    long long c%d = 0;" % idx)
                L.append("    if (%s && tpx_cell_i64(cl%d, &c%d) != 0) prc = %d;"
                         % (guard, idx, idx, 70))
            elif base == T.F64:
                L.append("    double c%d = 0.0;" % idx)
                L.append("    if (%s && tpx_cell_f64(cl%d, &c%d) != 0) prc = %d;"
                         % (guard, idx, idx, 70))
            elif base == T.BOOL:
                L.append("    bool c%d = false;" % idx)
                L.append("    if (%s && tpx_cell_bool(cl%d, &c%d) != 0) prc = %d;"
                         % (guard, idx, idx, 70))
            else:
                raise CodegenError("csv input type %r" % (t,))
        L.append("    if (!prc && avail) prc = %d;  // CSV_OVERRUN" % 21)
        L.append("    if (!prc && (badf & 6)) prc = %d;  // BADPARSE (escapes/structure -> host)" % 70)
        # ONE ASCII gate per row, fused into the group-mask loads (rows that
        # error out earlier divert regardless, so partial accumulation over
        # the loaded groups is equivalent)
        L.append("    if (!prc && S.hib) prc = %d;  // NCV" % 7)
        return L

    def _load_inputs_text(self, in_types, lds=True):
        """text() source: each row is the raw line (minus newline); no cell
        split, no quoting (Context::text, core/src/Context.cc)."""
        assert len(in_types) == 1
        t = in_types[0]
        null_values = self.csv_info.get("null_values", [])
        L = ["    long long prc = 0;"]
        if lds == "gen":
            # u64-opaque select (see csv loader note)
            L.append("    unsigned long long _rpa = staged ?"
                     " (unsigned long long)wave_lds + (unsigned long long)(in_offs[i]"
                     " - span_start) : (unsigned long long)((const char*)"
                     "in_data + in_offs[i]);")
            L.append("    unsigned long long _rea = staged ?"
                     " (unsigned long long)wave_lds + (unsigned long long)(in_offs[i+1]"
                     " - span_start) : (unsigned long long)((const char*)"
                     "in_data + in_offs[i+1]);")
            L.append('    asm volatile("" : "+v"(_rpa), "+v"(_rea));'
                     "  // opaque: block addrspace re-inference")
            L.append("    const char* rp = (const char*)_rpa;")
            L.append("    const char* rend = (const char*)_rea;")
        elif lds:
            L.append("    const char* rp = wave_lds + (in_offs[i] - span_start);")
            L.append("    const char* rend = wave_lds + (in_offs[i+1] - span_start);")
        else:
            L.append("    const char* rp = (const char*)in_data + in_offs[i];")
            L.append("    const char* rend = (const char*)in_data + in_offs[i+1];")
        L.append("    if (rend > rp && rend[-1] == '\\n') --rend;")
        L.append("    if (rend > rp && rend[-1] == '\\r') --rend;")
        L.append("    tstr c0{rp, rend - rp};")
        L.append("    if (!tpx_ascii(c0)) prc = %d;  // row ASCII gate (NCV)" % 7)
        if T.is_opt(t):
            checks = " || ".join("tpx_streq(c0, %s)" % self.lit(nv)
                                 for nv in null_values) or "false"
            L.append("    bool c0_n = %s;" % checks)
            L.append("    if (c0_n) c0 = tstr{rp, 0};")
        return L

    def _store_columnar(self, out_types, lds=True):
        """Store Out o -> columnar arrays + per-row serialized size (mem sink) or
        csv text size (csv sink)."""
        L = []
        import os as _os2
        _wd2 = int(_os2.environ.get("TPX_WDBG", "0"))
        if _wd2 in (7, 8, 9, 10, 11) and lds == "gen":
            if _wd2 == 7:
                L.append("    { tstr _t = o.o1; if (staged) _t = tpx_to_global"
                         "(_t, wave_lds, wave_lds + TPX_SPAN_CAP, in_data,"
                         " span_start); o.o0 = (long long)_t.p; }  // DBG")
            elif _wd2 == 9:
                L.append("    o.o0 = (long long)o.o1.p;  // DBG pre-translate")
            elif _wd2 == 8:
                L.append("    o.o0 = (long long)wave_lds;  // DBG")
            elif _wd2 == 10:
                L.append("    o.o0 = (long long)in_data;  // DBG")
            elif _wd2 == 11:
                L.append("    o.o0 = staged ? span_start : -1;  // DBG")
        # csv sizes FIRST, while string views still point into LDS (the quote
        # scan then runs on ds_read instead of re-reading global memory; the
        # bytes are identical either way)
        if self.sink != "mem":
            if self._dup_join_op() is not None:
                L.extend(self._csv_size_dup(out_types))
            else:
                L.extend(self._csv_size(out_types))
        for k, t in enumerate(out_types):
            base = T.deopt(t)
            if base == T.STR:
                import os as _os
                _wd = int(_os.environ.get("TPX_WDBG", "0"))
                if _wd == 3:
                    L.append("    o.o%d = tstr{(const char*)in_data, 0};" % k)
                if lds == "gen":
                    L.append("    if (staged) o.o%d = tpx_to_global(o.o%d,"
                             " wave_lds, wave_lds + TPX_SPAN_CAP, in_data,"
                             " span_start);" % (k, k))
                elif lds and lds != "park":
                    L.append("    o.o%d = tpx_to_global(o.o%d, wave_lds,"
                             " wave_lds + TPX_SPAN_CAP, in_data, span_start);"
                             % (k, k))
                L.append("    ((unsigned long long*)outv[%d])[i] = (unsigned long long)o.o%d.p;"
                         % (3 * k, k))
                L.append("    ((int*)outv[%d])[i] = (int)o.o%d.n;" % (3 * k + 1, k))
            elif base == T.F64:
                L.append("    ((double*)outv[%d])[i] = o.o%d;" % (3 * k, k))
            else:
                L.append("    ((long long*)outv[%d])[i] = (long long)o.o%d;" % (3 * k, k))
            if T.is_opt(t):
                L.append("    ((unsigned char*)outv[%d])[i] = o.o%d_n ? 1 : 0;"
                         % (3 * k + 2, k))
        dup_op = self._dup_join_op()
        if self.sink == "mem" and dup_op is not None:
            # 1:N rows: sizes[i] = count * (fixed + left varlen) + right
            # varlen summed over the bucket (full-row layout, FULL out types)
            full = self.full_out_types
            n_opt = sum(1 for t in full if T.is_opt(t))
            bitmap = ((n_opt + 63) // 64) * 8 if n_opt else 0
            has_var = any(T.is_varlen(t) for t in full)
            fixed = bitmap + 8 * len(full) + (8 if has_var else 0)
            n_left = len(dup_op.in_types)
            jd = self.jdup
            nm = jd["nm"]
            L.append("    long long lvar = 0;")
            for k in range(n_left):
                if T.is_varlen(full[k]):
                    if T.is_opt(full[k]):
                        L.append("    if (!o.o%d_n) lvar += o.o%d.n + 1;"
                                 % (k, k))
                    else:
                        L.append("    lvar += o.o%d.n + 1;" % k)
            L.append("    long long rvar = 0;")
            L.append("    for (int _d = 0; _d < o.o%d; ++_d) {"
                     % (len(out_types) - 1))
            L.append("      int _fi = o.o%d < 0 ? 0 : (int)o.o%d + _d;"
                     % (len(out_types) - 2, len(out_types) - 2))
            L.append("      bool _miss = o.o%d < 0;" % (len(out_types) - 2))
            for (j, kindc, has_null) in jd["out_info"]:
                if kindc == "str":
                    cond = "!_miss"
                    if has_null:
                        cond += " && !%s_c%d_null[_fi]" % (nm, j)
                    L.append("      if (%s) rvar += %s_c%d_len[_fi] + 1;"
                             % (cond, nm, j))
            L.append("    }")
            L.append("    sizes[i] = (long long)o.o%d * (%d + lvar) + rvar;"
                     % (len(out_types) - 1, fixed))
        elif self.sink == "mem":
            n_opt = sum(1 for t in out_types if T.is_opt(t))
            bitmap = ((n_opt + 63) // 64) * 8 if n_opt else 0
            has_var = any(T.is_varlen(t) for t in out_types)
            fixed = bitmap + 8 * len(out_types) + (8 if has_var else 0)
            L.append("    long long sz = %d;" % fixed)
            for k, t in enumerate(out_types):
                if T.is_varlen(t):
                    if T.is_opt(t):
                        L.append("    if (!o.o%d_n) sz += o.o%d.n + 1;" % (k, k))
                    else:
                        L.append("    sz += o.o%d.n + 1;" % k)
            L.append("    sizes[i] = sz;")
        return L

    def _csv_size(self, out_types):
        """RFC-4180 output size: quote a cell iff it contains delim/quote/CR/LF;
        \'"\' doubles. Appends newline per row. Also records whether ANY cell
        needs quoting (_anyq): the write kernel then takes a no-rescan memcpy
        fast path for the (common) fully-unquoted rows, via keep[i] bit 1."""
        L = ["    long long sz = %d;  // delimiters + newline" % len(out_types)]
        L.append("    bool _anyq = false;")
        qfree = self._thread_qfree()
        for k, t in enumerate(out_types):
            base = T.deopt(t)
            if base == T.STR:
                qc = self._qfree_cond(qfree[k]) if qfree is not None else None
                pre = ""
                if T.is_opt(t):
                    L.append("    if (!o.o%d_n) {" % k)
                    pre = "  "
                if qc == "true":
                    # provenance: value can never need quoting -> no scan
                    L.append(pre + "    sz += o.o%d.n;" % k)
                elif qc is not None:
                    L.append(pre + "    if (%s) { sz += o.o%d.n; }" % (qc, k))
                    L.append(pre + "    else { long long q%d;" % k)
                    L.append(pre + "      bool nq%d = tpx_csv_needs_quote(o.o%d,"
                             " &q%d);" % (k, k, k))
                    L.append(pre + "      sz += nq%d ? o.o%d.n + q%d + 2 : o.o%d.n;"
                             % (k, k, k, k))
                    L.append(pre + "      _anyq |= nq%d; }" % k)
                else:
                    L.append(pre + "    { long long q%d;" % k)
                    L.append(pre + "      bool nq%d = tpx_csv_needs_quote(o.o%d,"
                             " &q%d);" % (k, k, k))
                    L.append(pre + "      sz += nq%d ? o.o%d.n + q%d + 2 : o.o%d.n;"
                             % (k, k, k, k))
                    L.append(pre + "      _anyq |= nq%d; }" % k)
                if T.is_opt(t):
                    L.append("    }")
            elif base == T.I64:
                L.append("    sz += tpx_i64_digits(o.o%d);" % k)
            elif base == T.BOOL:
                L.append("    sz += o.o%d ? 4 : 5;" % k)
            elif base == T.F64 and not T.is_opt(t):
                # exact %f (PipelineBuilder.cc:1413); out-of-range/nan divert
                # to the host formatter, which prints identical text
                L.append("    unsigned long long fN%d; bool fg%d;" % (k, k))
                L.append("    if (tpx_f64_csv_n(o.o%d, &fN%d, &fg%d)) {"
                         % (k, k, k))
                L.append("      unsigned long long e = atomicAdd(exc_count,"
                         " 1ULL);")
                L.append("      if (e < exc_cap) {")
                L.append("        exc_buf[e*5+0] = row0 + i;")
                L.append("        exc_buf[e*5+1] = 7;  // NCV-style divert")
                L.append("        exc_buf[e*5+2] = 0;")
                if self.source == "col":
                    L.append("        exc_buf[e*5+3] = 0;")
                    L.append("        exc_buf[e*5+4] = 0;")
                else:
                    L.append("        exc_buf[e*5+3] = in_offs[i];")
                    L.append("        exc_buf[e*5+4] = in_offs[i+1];")
                L.append("      }")
                L.append("      keep[i] = 0; keep01[i] = 0; sizes[i] = 0;"
                         " continue;")
                L.append("    }")
                L.append("    sz += tpx_f64_csv_len(fN%d, fg%d);" % (k, k))
            else:
                raise CodegenError("csv sink for %r not supported yet" % (t,))
        L.append("    sizes[i] = sz;")
        return L

    def _csv_size_dup(self, store_types):
        """csv-sink sizes for a terminal dup join: count * (delims + newline +
        left cells) + right cells summed over the bucket."""
        jd = self.jdup
        nm = jd["nm"]
        dup_op = jd["op"]
        full = self.full_out_types
        n_left = len(dup_op.in_types)
        rinfo = {n_left + i2: info for i2, info in enumerate(jd["out_info"])}
        L = ["    bool _anyq = false;"]
        L.append("    long long lsz = %d;  // delimiters + newline" % len(full))
        for k in range(n_left):
            t = full[k]
            base = T.deopt(t)
            if base == T.STR:
                pre = ""
                if T.is_opt(t):
                    L.append("    if (!o.o%d_n) {" % k)
                    pre = "  "
                L.append(pre + "    { long long q%d;" % k)
                L.append(pre + "      bool nq%d = tpx_csv_needs_quote(o.o%d,"
                         " &q%d);" % (k, k, k))
                L.append(pre + "      lsz += nq%d ? o.o%d.n + q%d + 2 : o.o%d.n;"
                         % (k, k, k, k))
                L.append(pre + "      _anyq |= nq%d; }" % k)
                if T.is_opt(t):
                    L.append("    }")
            elif base == T.I64:
                L.append("    lsz += tpx_i64_digits(o.o%d);" % k)
            elif base == T.BOOL:
                L.append("    lsz += o.o%d ? 4 : 5;" % k)
            else:
                raise CodegenError("csv sink for %r not supported yet" % (t,))
        L.append("    long long rsz = 0;")
        cnt = "o.o%d" % (len(store_types) - 1)
        st = "o.o%d" % (len(store_types) - 2)
        L.append("    for (int _d = 0; _d < %s; ++_d) {" % cnt)
        L.append("      int _fi = %s < 0 ? 0 : (int)%s + _d;" % (st, st))
        L.append("      bool _miss = %s < 0;" % st)
        for k in range(n_left, len(full)):
            j, kindc, has_null = rinfo[k]
            t = full[k]
            if kindc == "str":
                guard = "!_miss"
                if has_null:
                    guard += " && !%s_c%d_null[_fi]" % (nm, j)
                if T.is_opt(t):
                    L.append("      if (%s) {" % guard)
                else:
                    L.append("      {")
                L.append("        tstr _rv{%s_blob + %s_c%d_off[_fi],"
                         " %s_c%d_len[_fi]};" % (nm, nm, j, nm, j))
                L.append("        long long _q; bool _nq ="
                         " tpx_csv_needs_quote(_rv, &_q);")
                L.append("        rsz += _nq ? _rv.n + _q + 2 : _rv.n;")
                L.append("        _anyq |= _nq;")
                L.append("      }")
            elif kindc in ("i64", "bool"):
                if T.is_opt(t):
                    raise CodegenError("csv sink for %r not supported yet"
                                       % (t,))
                if kindc == "bool":
                    L.append("      rsz += %s_c%d[_fi] ? 4 : 5;" % (nm, j))
                else:
                    L.append("      rsz += tpx_i64_digits(%s_c%d[_fi]);"
                             % (nm, j))
            else:
                raise CodegenError("csv sink for joined %s column" % kindc)
        L.append("    }")
        L.append("    sizes[i] = (long long)%s * lsz + rsz;" % cnt)
        return L

    def _write_kernel_csv_dup(self, full_types, store_types):
        """csv-sink writer for a terminal dup join: thread per input row,
        loops the bucket (perf-uncritical path; no LDS staging)."""
        jd = self.jdup
        nm = jd["nm"]
        dup_op = jd["op"]
        n_left = len(dup_op.in_types)
        n_store = len(store_types)
        full = full_types
        rinfo = {n_left + i2: info for i2, info in enumerate(jd["out_info"])}
        L = []
        L.append('extern "C" __global__ void tpx_stage_write(')
        L.append("    const unsigned char* __restrict__ keep,")
        L.append("    const long long* __restrict__ keep_scan,")
        L.append("    const long long* __restrict__ size_scan,")
        L.append("    long long n, long long row0, void** outv,")
        L.append("    unsigned char* __restrict__ out_data, long long* __restrict__ out_offs,")
        L.append("    long long* __restrict__ out_rowidx,")
        L.append("    long long total_rows, long long total_bytes,")
        L.append("    long long out_byte0) {")
        L.append("  long long stride = (long long)gridDim.x * blockDim.x;")
        L.append("  long long tid0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;")
        L.append("  for (long long i = tid0; i < n; i += stride) {")
        L.append("    if (!keep[i]) continue;")
        L.append("    int jb_s = (int)((const long long*)outv[%d])[i];" % (3 * (n_store - 2)))
        L.append("    int jb_c = (int)((const long long*)outv[%d])[i];" % (3 * (n_store - 1)))
        for k in range(n_left):
            t = full[k]
            base = T.deopt(t)
            if base == T.STR:
                L.append("    tstr v%d{(const char*)((const unsigned long long*)outv[%d])[i],"
                         " (long long)((const int*)outv[%d])[i]};" % (k, 3 * k, 3 * k + 1))
            elif base == T.F64:
                raise CodegenError("csv sink for f64 not supported yet")
            else:
                L.append("    long long v%d = ((const long long*)outv[%d])[i];" % (k, 3 * k))
            if T.is_opt(t):
                L.append("    bool v%d_n = ((const unsigned char*)outv[%d])[i] != 0;"
                         % (k, 3 * k + 2))
        L.append("    char* w = (char*)out_data + size_scan[i];")
        L.append("    long long obase = out_byte0 + size_scan[i];")
        L.append("    long long kbase = keep_scan[i];")
        L.append("    for (int _d = 0; _d < jb_c; ++_d) {")
        L.append("      bool _miss = jb_s < 0;")
        L.append("      int _fi = _miss ? 0 : jb_s + _d;")
        L.append("      char* w0 = w;")
        L.append("      out_offs[kbase + _d] = obase;")
        L.append("      out_rowidx[kbase + _d] = row0 + i;")
        for k in range(len(full)):
            t = full[k]
            base = T.deopt(t)
            if k:
                L.append("      *w++ = ',';")
            if k < n_left:
                if base == T.STR:
                    pre = ""
                    if T.is_opt(t):
                        L.append("      if (!v%d_n) {" % k)
                        pre = "  "
                    L.append(pre + "      w = tpx_csv_cell_write(w, v%d);" % k)
                    if T.is_opt(t):
                        L.append("      }")
                elif base == T.I64:
                    L.append("      { long long dl = tpx_i64_digits(v%d);"
                             " tpx_i64_write(w, v%d, dl); w += dl; }" % (k, k))
                elif base == T.BOOL:
                    L.append("      { const char* s2 = v%d ? \"True\" : \"False\";"
                             " long long l2 = v%d ? 4 : 5;"
                             " for (long long j2 = 0; j2 < l2; ++j2) w[j2] = s2[j2];"
                             " w += l2; }" % (k, k))
            else:
                j, kindc, has_null = rinfo[k]
                if kindc == "str":
                    guard = "!_miss"
                    if has_null:
                        guard += " && !%s_c%d_null[_fi]" % (nm, j)
                    L.append("      if (%s) {" % guard)
                    L.append("        tstr _rv{%s_blob + %s_c%d_off[_fi],"
                             " %s_c%d_len[_fi]};" % (nm, nm, j, nm, j))
                    L.append("        w = tpx_csv_cell_write(w, _rv);")
                    L.append("      }")
                elif kindc == "bool":
                    L.append("      { bool b2 = %s_c%d[_fi] != 0;"
                             " const char* s2 = b2 ? \"True\" : \"False\";"
                             " long long l2 = b2 ? 4 : 5;"
                             " for (long long j2 = 0; j2 < l2; ++j2) w[j2] = s2[j2];"
                             " w += l2; }" % (nm, j))
                else:
                    L.append("      { long long rv2 = %s_c%d[_fi];"
                             " long long dl = tpx_i64_digits(rv2);"
                             " tpx_i64_write(w, rv2, dl); w += dl; }" % (nm, j))
        L.append("      *w++ = '\\n';")
        L.append("      obase += (long long)(w - w0);")
        L.append("    }")
        L.append("  }")
        L.append("}")
        return "\n".join(L)

    # -- write kernel ----------------------------------------------------------
    def _write_kernel(self, out_types):
        if self.sink == "csv":
            return self._write_kernel_csv(out_types)
        n_opt = sum(1 for t in out_types if T.is_opt(t))
        bitmap = ((n_opt + 63) // 64) * 8 if n_opt else 0
        has_var = any(T.is_varlen(t) for t in out_types)
        nf = len(out_types)
        fixed_end = bitmap + 8 * nf
        L = []
        L.append('extern "C" __global__ void tpx_stage_write(')
        L.append("    const unsigned char* __restrict__ keep,")
        L.append("    const long long* __restrict__ keep_scan,")
        L.append("    const long long* __restrict__ size_scan,")
        L.append("    long long n, long long row0, void** outv,")
        L.append("    unsigned char* __restrict__ out_data, long long* __restrict__ out_offs,")
        L.append("    long long* __restrict__ out_rowidx,")
        L.append("    long long total_rows, long long total_bytes,")
        # out_data is the CHUNK-LOCAL base (past the 8B numRows header, which
        # the host writes); out_byte0 is the row's global byte offset bias
        # stored into out_offs (chunk pipelining: tpx_abi.cpp run_core)
        L.append("    long long out_byte0) {")
        L.append("  long long stride = (long long)gridDim.x * blockDim.x;")
        L.append("  long long tid0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;")
        L.append("  for (long long i = tid0; i < n; i += stride) {")
        L.append("    if (!keep[i]) continue;")
        L.append("    unsigned char* w = out_data + size_scan[i];")
        L.append("    out_offs[keep_scan[i]] = out_byte0 + size_scan[i];")
        L.append("    out_rowidx[keep_scan[i]] = row0 + i;")
        # load columnar values
        for k, t in enumerate(out_types):
            base = T.deopt(t)
            if base == T.STR:
                L.append("    tstr v%d{(const char*)((const unsigned long long*)outv[%d])[i],"
                         " (long long)((const int*)outv[%d])[i]};" % (k, 3 * k, 3 * k + 1))
            elif base == T.F64:
                L.append("    double v%d = ((const double*)outv[%d])[i];" % (k, 3 * k))
            else:
                L.append("    long long v%d = ((const long long*)outv[%d])[i];" % (k, 3 * k))
            if T.is_opt(t):
                L.append("    bool v%d_n = ((const unsigned char*)outv[%d])[i] != 0;"
                         % (k, 3 * k + 2))
        # bitmap
        if bitmap:
            L.append("    unsigned long long bm = 0;")
            oc = 0
            for k, t in enumerate(out_types):
                if T.is_opt(t):
                    L.append("    if (v%d_n) bm |= 1ULL << %d;" % (k, oc))
                    oc += 1
            L.append("    *(unsigned long long*)w = bm;")
            if bitmap > 8:
                raise CodegenError(">64 optional fields unsupported")
        # slots + varlen
        L.append("    long long var_off = 0;  // within varlen region")
        import os as _os
        wdbg = int(_os.environ.get("TPX_WDBG", "0"))
        for k, t in enumerate(out_types):
            base = T.deopt(t)
            slot = "((long long*)(w + %d))[%d]" % (bitmap, k)
            if base == T.STR:
                null_guard = ("v%d_n" % k) if T.is_opt(t) else "false"
                L.append("    if (%s) { %s = 0; } else {" % (null_guard, slot))
                L.append("      long long off = %d + 8 + var_off - %d;"
                         % (fixed_end - bitmap, 8 * k))
                L.append("      %s = off | ((v%d.n + 1) << 32);" % (slot, k))
                L.append("      char* d = (char*)(w + %d + var_off);" % (fixed_end + 8))
                if wdbg == 0:
                    L.append("      tpx_memcpy(d, v%d.p, v%d.n);" % (k, k))
                    L.append("      d[v%d.n] = 0;" % k)
                elif wdbg in (1, 7, 8, 9):  # skip string READS, keep layout
                    L.append("      for (long long _z = 0; _z <= v%d.n; ++_z)"
                             " d[_z] = 0;" % k)
                # wdbg >= 2: no varlen writes at all
                L.append("      var_off += v%d.n + 1;" % k)
                L.append("    }")
            elif base == T.F64:
                L.append("    %s = __double_as_longlong(v%d);" % (slot, k))
                if T.is_opt(t):
                    L.append("    if (v%d_n) %s = 0;" % (k, slot))
            else:
                L.append("    %s = v%d;" % (slot, k))
                if T.is_opt(t):
                    L.append("    if (v%d_n) %s = 0;" % (k, slot))
        if has_var:
            L.append("    *(long long*)(w + %d) = var_off;" % fixed_end)
        L.append("  }")
        L.append("}")
        return "\n".join(L)

    def _write_kernel_mem_dup(self, full_types, store_types):
        """mem-sink writer for a TERMINAL duplicate-key join: one thread per
        input row, looping its bucket — row d reads the right columns from the
        embedded build table at (start + d); left columns repeat. Output rows
        land at keep_scan[i] + d / size_scan[i] + running offset (keep01
        carried the bucket counts, so the scans already yield 1:N offsets)."""
        jd = self.jdup
        nm = jd["nm"]
        dup_op = jd["op"]
        n_left = len(dup_op.in_types)
        n_store = len(store_types)
        full = full_types
        n_opt = sum(1 for t in full if T.is_opt(t))
        bitmap = ((n_opt + 63) // 64) * 8 if n_opt else 0
        has_var = any(T.is_varlen(t) for t in full)
        nf = len(full)
        fixed_end = bitmap + 8 * nf
        if bitmap > 8:
            raise CodegenError(">64 optional fields unsupported")
        # right-col info by FULL index: full[n_left + idx] <- out_info[idx]
        rinfo = {n_left + i2: info for i2, info in enumerate(jd["out_info"])}
        L = []
        L.append('extern "C" __global__ void tpx_stage_write(')
        L.append("    const unsigned char* __restrict__ keep,")
        L.append("    const long long* __restrict__ keep_scan,")
        L.append("    const long long* __restrict__ size_scan,")
        L.append("    long long n, long long row0, void** outv,")
        L.append("    unsigned char* __restrict__ out_data, long long* __restrict__ out_offs,")
        L.append("    long long* __restrict__ out_rowidx,")
        L.append("    long long total_rows, long long total_bytes,")
        L.append("    long long out_byte0) {")
        L.append("  long long stride = (long long)gridDim.x * blockDim.x;")
        L.append("  long long tid0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;")
        L.append("  for (long long i = tid0; i < n; i += stride) {")
        L.append("    if (!keep[i]) continue;")
        L.append("    int jb_s = (int)((const long long*)outv[%d])[i];" % (3 * (n_store - 2)))
        L.append("    int jb_c = (int)((const long long*)outv[%d])[i];" % (3 * (n_store - 1)))
        # left values loaded once
        for k in range(n_left):
            t = full[k]
            base = T.deopt(t)
            if base == T.STR:
                L.append("    tstr v%d{(const char*)((const unsigned long long*)outv[%d])[i],"
                         " (long long)((const int*)outv[%d])[i]};" % (k, 3 * k, 3 * k + 1))
            elif base == T.F64:
                L.append("    double v%d = ((const double*)outv[%d])[i];" % (k, 3 * k))
            else:
                L.append("    long long v%d = ((const long long*)outv[%d])[i];" % (k, 3 * k))
            if T.is_opt(t):
                L.append("    bool v%d_n = ((const unsigned char*)outv[%d])[i] != 0;"
                         % (k, 3 * k + 2))
        L.append("    unsigned char* w = out_data + size_scan[i];")
        L.append("    long long obase = out_byte0 + size_scan[i];")
        L.append("    long long kbase = keep_scan[i];")
        L.append("    for (int _d = 0; _d < jb_c; ++_d) {")
        L.append("      bool _miss = jb_s < 0;")
        L.append("      int _fi = _miss ? 0 : jb_s + _d;")
        # per-dup right values
        for k in range(n_left, nf):
            j, kindc, has_null = rinfo[k]
            t = full[k]
            nullv = "_miss"
            if has_null:
                nullv = "(_miss || %s_c%d_null[_fi])" % (nm, j)
            if kindc == "str":
                L.append("      tstr v%d{%s_blob + %s_c%d_off[_fi],"
                         " %s_c%d_len[_fi]};" % (k, nm, nm, j, nm, j))
            elif kindc == "f64":
                L.append("      double v%d = %s_c%d[_fi];" % (k, nm, j))
            elif kindc == "bool":
                L.append("      long long v%d = %s_c%d[_fi];" % (k, nm, j))
            else:
                L.append("      long long v%d = %s_c%d[_fi];" % (k, nm, j))
            if T.is_opt(t):
                L.append("      bool v%d_n = %s;" % (k, nullv))
        # header entries for this output row
        L.append("      out_offs[kbase + _d] = obase;")
        L.append("      out_rowidx[kbase + _d] = row0 + i;")
        if bitmap:
            L.append("      unsigned long long bm = 0;")
            oc = 0
            for k, t in enumerate(full):
                if T.is_opt(t):
                    L.append("      if (v%d_n) bm |= 1ULL << %d;" % (k, oc))
                    oc += 1
            L.append("      *(unsigned long long*)w = bm;")
        L.append("      long long var_off = 0;")
        for k, t in enumerate(full):
            base = T.deopt(t)
            slot = "((long long*)(w + %d))[%d]" % (bitmap, k)
            if base == T.STR:
                null_guard = ("v%d_n" % k) if T.is_opt(t) else "false"
                L.append("      if (%s) { %s = 0; } else {" % (null_guard, slot))
                L.append("        long long off = %d + 8 + var_off - %d;"
                         % (fixed_end - bitmap, 8 * k))
                L.append("        %s = off | ((v%d.n + 1) << 32);" % (slot, k))
                L.append("        char* d2 = (char*)(w + %d + var_off);"
                         % (fixed_end + 8))
                L.append("        tpx_memcpy(d2, v%d.p, v%d.n);" % (k, k))
                L.append("        d2[v%d.n] = 0;" % k)
                L.append("        var_off += v%d.n + 1;" % k)
                L.append("      }")
            elif base == T.F64:
                L.append("      %s = __double_as_longlong(v%d);" % (slot, k))
                if T.is_opt(t):
                    L.append("      if (v%d_n) %s = 0;" % (k, slot))
            else:
                L.append("      %s = v%d;" % (slot, k))
                if T.is_opt(t):
                    L.append("      if (v%d_n) %s = 0;" % (k, slot))
        if has_var:
            L.append("      *(long long*)(w + %d) = var_off;" % fixed_end)
        L.append("      long long rsz = %d + var_off;" % (fixed_end + (8 if has_var else 0)))
        L.append("      w += rsz; obase += rsz;")
        L.append("    }")
        L.append("  }")
        L.append("}")
        return "\n".join(L)

    # CSV write: one lane per INPUT row (coalesced columnar loads; a kept-row
    # gather was measured SLOWER — gather defeats load coalescing). The 64
    # input rows of a wave own a CONTIGUOUS output span of ~selectivity*64*row
    # bytes: active lanes format into LDS, then the wave copies the span out
    # with aligned 8B stores. Spans > TPX_WCAP (long rows) fall back to direct
    # global formatting for that wave.
    WRITE_CAP = 4096  # bytes per wave; 2 waves/block -> 8 KiB LDS (measured
    #                   best vs 8192/2048/unstaged on the Z1 bench: wprobe.py)

    def _write_kernel_csv(self, out_types):
        L = []
        L.append("#define TPX_WCAP %d" % self.WRITE_CAP)
        L.append('extern "C" __global__ void tpx_stage_write(')
        L.append("    const unsigned char* __restrict__ keep,")
        L.append("    const long long* __restrict__ keep_scan,")
        L.append("    const long long* __restrict__ size_scan,")
        L.append("    long long n, long long row0, void** outv,")
        L.append("    unsigned char* __restrict__ out_data, long long* __restrict__ out_offs,")
        L.append("    long long* __restrict__ out_rowidx,")
        L.append("    long long total_rows, long long total_bytes,")
        # out_data is the CHUNK-LOCAL base; out_byte0 is the global byte bias
        # stored into out_offs; the sentinel out_offs[total] is host-written
        L.append("    long long out_byte0) {")
        L.append("  __shared__ __attribute__((aligned(16)))"
                 " char wsmem[2 * TPX_WCAP + 16];  // 128-thread blocks")
        L.append("  int lane = threadIdx.x & 63;")
        L.append("  int wid = threadIdx.x >> 6;")
        L.append("  char* wave_lds = wsmem + wid * TPX_WCAP;")
        L.append("  long long wave_stride = (long long)gridDim.x * (blockDim.x >> 6);")
        L.append("  long long nwaves = (n + 63) >> 6;")
        L.append("  for (long long wb = (long long)blockIdx.x * (blockDim.x >> 6) + wid;"
                 " wb < nwaves; wb += wave_stride) {")
        L.append("    long long r0 = wb << 6;")
        L.append("    long long rhi = r0 + 64 < n ? r0 + 64 : n;")
        L.append("    long long span_start = size_scan[r0];")
        L.append("    long long span_end = r0 + 64 < n ? size_scan[r0 + 64]"
                 " : total_bytes;")
        L.append("    long long span = span_end - span_start;")
        L.append("    bool staged = span <= TPX_WCAP;")
        L.append("    long long i = r0 + lane;")
        L.append("    bool active = i < rhi && keep[i];")
        L.append("    long long my_start = active ? size_scan[i] : 0;")
        L.append("    if (active) {")
        L.append("      out_offs[keep_scan[i]] = out_byte0 + my_start;")
        L.append("      out_rowidx[keep_scan[i]] = row0 + i;")
        L.append("    }")
        # duplicate the format body per pointer mode: in the staged branch w
        # provably derives from LDS, so addrspace inference emits ds_write
        # instead of flat stores (same reasoning as tpx_stage_main's row body;
        # the single generic body miscompiled on hipRTC gfx950)
        body = self._csv_format_body(out_types)
        L.append("    if (staged) {")
        L.append("      if (active) {")
        L.append("        bool _noq = (keep[i] & 2) != 0;")
        L.append("        char* w = wave_lds + (my_start - span_start);")
        L.extend("    " + ln for ln in body)
        L.append("      }")
        L.append("      __builtin_amdgcn_wave_barrier();")
        # cooperative span copy with ALIGNED global 8B stores (dst + a0 is
        # 8-aligned by construction) and aligned LDS 8B reads + a uniform
        # funnel shift (a0 is the same for every lane of the wave). A plain
        # unaligned memcpy lowers to byte stores (measured: no faster than the
        # unstaged kernel).
        L.append("      char* dst = (char*)out_data + span_start;")
        L.append("      long long a0 = (8 - (span_start & 7)) & 7;")
        L.append("      if (lane < a0 && lane < span) dst[lane] = wave_lds[lane];")
        L.append("      int sh = (int)(a0 & 7) * 8;")
        L.append("      for (long long b = a0 + (long long)lane * 8; b + 8 <= span;"
                 " b += 64 * 8) {")
        L.append("        unsigned long long lo ="
                 " *(const unsigned long long*)(wave_lds + (b - a0));")
        L.append("        unsigned long long hi ="
                 " *(const unsigned long long*)(wave_lds + (b - a0) + 8);")
        L.append("        unsigned long long v = sh ? ((lo >> sh) | (hi << (64 - sh)))"
                 " : lo;")
        L.append("        *(unsigned long long*)(dst + b) = v;")
        L.append("      }")
        L.append("      long long t0 = span > a0 ? a0 + ((span - a0) & ~7LL) : span;")
        L.append("      if (t0 + lane < span) dst[t0 + lane] = wave_lds[t0 + lane];")
        L.append("    } else if (active) {")
        L.append("      bool _noq = (keep[i] & 2) != 0;")
        L.append("      char* w = (char*)out_data + my_start;")
        L.extend("  " + ln for ln in body)
        L.append("    }")
        L.append("  }")
        L.append("}")
        return "\n".join(L)

    def _csv_format_body(self, out_types):
        L = []
        for k, t in enumerate(out_types):
            base = T.deopt(t)
            if k:
                L.append("    *w++ = ',';")
            if base == T.STR:
                pre = ""
                if T.is_opt(t):
                    L.append("    if (!((const unsigned char*)outv[%d])[i]) {"
                             % (3 * k + 2))
                    pre = "  "
                L.append(pre + "    tstr v%d{(const char*)((const unsigned long long*)outv[%d])[i],"
                         " (long long)((const int*)outv[%d])[i]};" % (k, 3 * k, 3 * k + 1))
                # keep[i] bit 1 (set by the main kernel's size pass): no cell
                # of this row needs quoting -> straight copy, no rescan
                L.append(pre + "    if (_noq) { tpx_memcpy(w, v%d.p, v%d.n);"
                         " w += v%d.n; }" % (k, k, k))
                L.append(pre + "    else w = tpx_csv_cell_write(w, v%d);" % k)
                if T.is_opt(t):
                    L.append("    }")
            elif base == T.I64:
                L.append("    { long long v = ((const long long*)outv[%d])[i];" % (3 * k))
                L.append("      long long dl = tpx_i64_digits(v); tpx_i64_write(w, v, dl); w += dl; }")
            elif base == T.BOOL:
                L.append("    { bool v = ((const long long*)outv[%d])[i] != 0;" % (3 * k))
                L.append("      const char* s = v ? \"True\" : \"False\"; long long l = v ? 4 : 5;")
                L.append("      for (long long j = 0; j < l; ++j) w[j] = s[j]; w += l; }")
            elif base == T.F64 and not T.is_opt(t):
                L.append("    { double v = ((const double*)outv[%d])[i];" % (3 * k))
                L.append("      unsigned long long fN; bool fg;")
                L.append("      tpx_f64_csv_n(v, &fN, &fg);  // rows out of"
                         " range were diverted by the size pass")
                L.append("      w = tpx_f64_csv_write(w, fN, fg); }")
            else:
                raise CodegenError("csv sink for %r" % (t,))
        L.append("    *w++ = '\\n';")
        return L

    def _desc(self, in_types, out_types):
        def tdesc(t):
            return ("opt," if T.is_opt(t) else "") + T.deopt(t)
        lines = ["source=%s" % self.source, "sink=%s" % self.sink,
                 "nin=%d" % len(in_types), "nout=%d" % len(out_types)]
        if getattr(self.sp, "agg_expr", None) is not None:
            if getattr(self.sp, "agg_key_idx", None) is not None:
                lines.append("aggby=%s" % T.deopt(self.sp.agg_type))
                if T.deopt(getattr(self.sp, "agg_key_type", None)) == T.STR:
                    lines.append("aggkeystr=1")
            else:
                lines.append("agg=%s" % T.deopt(self.sp.agg_type))
        if self.csv_info.get("text_mode"):
            lines.append("textmode=1")
        if self.split:
            lines.append("split=1")
            if self.park_copy:
                lines.append("parkcopy=1")
            used = getattr(self.sp, "used_source_cols", None)
            if used is None:
                used = range(len(in_types))
            lines.append("used=%s" % ",".join(str(i) for i in sorted(used)))
        for i, t in enumerate(in_types):
            lines.append("in%d=%s" % (i, tdesc(t)))
        for i, t in enumerate(out_types):
            lines.append("out%d=%s" % (i, tdesc(t)))
        return "\n".join(lines) + "\n"


def generate_stage(sp, source="mem", sink="mem", csv_info=None):
    # pass 1: collect const-needle find/contains groups per haystack
    cg1 = StageCodegen(sp, source, sink, csv_info)
    cg1.generate()
    groups = {k: v for k, v in cg1.scan_registry.items() if len(v) >= 2}
    # pass 2: fresh instance (lits/emitter state are single-use)
    return StageCodegen(sp, source, sink, csv_info,
                        fusion_groups=groups or None).generate()
