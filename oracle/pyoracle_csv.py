"""ORACLE (TEST INFRASTRUCTURE ONLY) — CPU restatement of the reference's CSV
TransformStage semantics. See oracle/__init__.py for the usage restriction.

Restates (reference file:line):
 - row-boundary detection: CSVReader.cc:390 + utils/src/CSVUtils.cc:1494
   findLineStart — expressed here as the RFC-4180 quote-parity rule ('\\n' is a
   boundary iff preceded by an even number of '"');
 - cell split + quote/escape handling: CSVParseRowGenerator.cc state machine;
   cells containing '""' escapes or structural errors divert to the interpreter
   path (BADPARSE_STRING_INPUT with the raw line, ExceptionCodes.h:118);
 - type sniffing: CSVStatistic.cc behavior at rule level (normal-case threshold
   ContextOptions.cc:216; per-column i64 -> f64 -> bool -> str with Option
   wrapping for numeric columns when null-value strings were sampled; str columns
   stay plain str under nullValueOptimization=false — the Zillow Z1 config);
 - typed cell parse: trim + fast_atoi64/fast_atod/fast_atob
   (Runtime.cc:319-383 over StringUtils.cc:22/:71/:186, restated in pyoracle.py);
 - dual-mode execution + in-order merge: LocalBackend.cc:963-1085,
   ResolveTask.cc:389/:878 (interpreter replay parses the raw line with full
   RFC-4180 unescaping and CPython conversions);
 - CSV output: quote iff the cell contains delim/quote/CR/LF, '"' doubled
   (codegen'd writer, PipelineBuilder.h:238 buildWithCSVRowWriter).
"""
import csv as _pycsv
import io
import math
from typing import List, Optional

from . import pyoracle

PY_WHITESPACE = pyoracle.PY_WHITESPACE


# ---- row/cell split (identical rules to the GPU scan + split) --------------------

def split_rows(data: bytes) -> List[bytes]:
    rows = []
    start = 0
    parity = 0
    for i, b in enumerate(data):
        if b == 0x22:
            parity ^= 1
        elif b == 0x0A and parity == 0:
            rows.append(data[start:i + 1])
            start = i + 1
    if start < len(data):
        rows.append(data[start:])
    return rows


def split_cells(line: bytes, delim: bytes = b","):
    end = len(line)
    while end > 0 and line[end - 1:end] in (b"\n", b"\r"):
        end -= 1
    cells, flags = [], 0
    p = 0
    more = True
    while more:
        more = False
        if p < end and line[p:p + 1] == b'"':
            s = p + 1
            q = s
            esc = False
            while q < end:
                if line[q:q + 1] == b'"':
                    if q + 1 < end and line[q + 1:q + 2] == b'"':
                        esc = True
                        q += 2
                        continue
                    break
                q += 1
            if q >= end:
                cells.append(line[p:end])
                flags |= 4
                return cells, flags
            cells.append(line[s:q])
            if esc:
                flags |= 2
            q += 1
            if q < end and line[q:q + 1] != delim:
                flags |= 4
            while q < end and line[q:q + 1] != delim:
                q += 1
            if q < end:
                more = True
                q += 1
            p = q
        else:
            q = p
            while q < end and line[q:q + 1] != delim:
                q += 1
            cells.append(line[p:q])
            if q < end:
                more = True
                q += 1
            p = q
    return cells, flags


# ---- sniffing --------------------------------------------------------------------

def _try_i64(s: str):
    t = s.strip(PY_WHITESPACE)
    if not t:
        return None
    ok, v = pyoracle.fast_atoi64(t)
    return v if ok else None


def _accept_f64(s: str) -> bool:
    t = s.strip(PY_WHITESPACE)
    if not t:
        return False
    ok, _ = pyoracle.fast_atod(t)
    return ok


_BOOL_TRUE = ("true", "t", "yes", "y", "1")
_BOOL_FALSE = ("false", "f", "no", "n", "0")


def _try_bool(s: str):
    t = s.strip(PY_WHITESPACE).lower()
    if t in _BOOL_TRUE:
        return True
    if t in _BOOL_FALSE:
        return False
    return None


def sniff_delimiter(sample: bytes) -> bytes:
    first = split_rows(sample)[0]
    best, bestn = b",", -1
    for cand in (b",", b";", b"|", b"\t"):
        n = len(split_cells(first, cand)[0])
        if n > bestn:
            best, bestn = cand, n
    return best


def sniff(sample: bytes, null_values, threshold, header, columns,
          delim: bytes = b","):
    rows = [split_cells(r, delim)[0] for r in split_rows(sample)]
    rows = [r for r in rows if r]
    txt = [[c.decode("utf-8", "replace") for c in r] for r in rows]

    def numericish(c):
        return _try_i64(c) is not None or _accept_f64(c) or _try_bool(c) is not None

    if header is None:
        r0 = txt[0]
        rest = txt[1:1001]
        has_header = (len(txt) > 1 and not any(numericish(c) for c in r0 if c)
                      and any(numericish(c) for r in rest for c in r))
    else:
        has_header = header
    names = txt[0] if has_header else None
    data_rows = txt[1:] if has_header else txt
    if columns:
        names = list(columns)
    ncols = max(len(r) for r in data_rows[:1000]) if data_rows else len(names or [])
    if names is None:
        names = ["column%d" % i for i in range(ncols)]
    ncols = len(names)

    nulls = set(null_values)
    types = []
    for k in range(ncols):
        n = n_i = n_f = n_b = n_nul = 0
        for r in data_rows[:10000]:
            if k >= len(r):
                continue
            c = r[k]
            n += 1
            if c in nulls:
                n_nul += 1
            elif _try_i64(c) is not None:
                n_i += 1
                n_f += 1
            elif _accept_f64(c):
                n_f += 1
            elif _try_bool(c) is not None:
                n_b += 1
        if n == 0:
            types.append("str")
            continue
        if (n_i + n_nul) / n >= threshold and n_i > 0:
            t = "i64"
        elif (n_f + n_nul) / n >= threshold and n_f > 0:
            t = "f64"
        elif (n_b + n_nul) / n >= threshold and n_b > 0:
            t = "bool"
        else:
            types.append("str")
            continue
        types.append(("opt", t) if n_nul > 0 else t)
    return has_header, names, types


# ---- execution -------------------------------------------------------------------

def _deopt(t):
    return t[1] if isinstance(t, tuple) and t[0] == "opt" else t


def _is_opt(t):
    return isinstance(t, tuple) and t[0] == "opt"


def _fast_parse_row(cells, flags, col_types, null_values, used=None):
    """Normal-case typed parse. Returns ("row", tuple) or ("bad", ec_name).
    used: projection-pushdown column set (LogicalOptimizer selectionPushdown /
    CSVParseRowGenerator willBeSerialized): unused columns keep their raw cell
    text and never fail a typed parse."""
    if flags & 6:
        return ("bad", "escape/structure")
    if len(cells) != len(col_types):
        return ("bad", "cellcount")
    vals = []
    for ci, (c, t) in enumerate(zip(cells, col_types)):
        s = c.decode("utf-8", "replace")
        base = _deopt(t)
        if used is not None and ci not in used:
            vals.append(s)
            continue
        if _is_opt(t) and s in null_values:
            vals.append(None)
            continue
        if base == "str":
            vals.append(s)
            continue
        st = s.strip(PY_WHITESPACE)
        if base == "i64":
            ok, v = pyoracle.fast_atoi64(st)
            if not ok:
                return ("bad", "i64")
            vals.append(v)
        elif base == "f64":
            ok, v = pyoracle.fast_atod(st)
            if not ok:
                return ("bad", "f64")
            vals.append(v)
        elif base == "bool":
            b = _try_bool(st)
            if b is None:
                return ("bad", "bool")
            vals.append(b)
        else:
            raise ValueError(t)
    return ("row", tuple(vals))


class _BadParse(Exception):
    pass


_BadParse.__name__ = "BadParseStringInput"


def _replay_row(raw_line: bytes, col_types, null_values, row_ops, columns,
                delim: str = ",", used=None):
    """Interpreter replay: full RFC-4180 parse + CPython conversions (matches the
    product's csvio.replay_csv_row rules)."""
    text = raw_line.decode("utf-8", "replace").rstrip("\n").rstrip("\r")
    try:
        cells = next(_pycsv.reader(io.StringIO(text), delimiter=delim))
    except (StopIteration, _pycsv.Error):
        return ("exc", _BadParse("unparseable"))
    if len(cells) != len(col_types):
        return ("exc", _BadParse("cellcount"))
    vals = []  # noqa: replay path
    for ci, (c, t) in enumerate(zip(cells, col_types)):
        base = _deopt(t)
        if used is not None and ci not in used:
            vals.append(c)
            continue
        if _is_opt(t) and c in null_values:
            vals.append(None)
            continue
        try:
            if base == "i64":
                vals.append(int(c.strip(PY_WHITESPACE)))
            elif base == "f64":
                vals.append(float(c.strip(PY_WHITESPACE)))
            elif base == "bool":
                b = _try_bool(c)
                if b is None:
                    raise ValueError(c)
                vals.append(b)
            else:
                vals.append(c)
        except ValueError as e:
            return ("exc", e)
    val = vals[0] if len(col_types) == 1 else tuple(vals)
    return pyoracle.process_row(val, row_ops, columns, fast=False)


def run_csv_pipeline(csv_bytes: bytes, ops, columns=None, header=None,
                     null_values=None, threshold=0.9, sink="collect",
                     delimiter=None, type_hints=None, used_cols=None):
    """Full oracle CSV pipeline. sink='collect' -> values; 'csv' -> output text
    (bytes, incl. header line)."""
    if null_values is None:
        null_values = [""]
    sample = csv_bytes[:1 << 20]
    nl = sample.rfind(b"\n")
    if nl >= 0:
        sample = sample[:nl + 1]
    delim = delimiter.encode() if delimiter else sniff_delimiter(sample)
    has_header, names, col_types = sniff(sample, null_values, threshold, header,
                                         columns, delim)
    if type_hints:  # string lattice forms: "i64" / "opt_i64" / ...
        for k, h in type_hints.items():
            i = names.index(k) if isinstance(k, str) else int(k)
            col_types[i] = ("opt", h[4:]) if h.startswith("opt_") else h
    data = csv_bytes
    if has_header:
        p = data.find(b"\n")
        data = data[p + 1:] if p >= 0 else b""

    agg = None
    aggby = None
    row_ops = []
    for op in ops:
        if op[0] == "aggregate":
            agg = op
        elif op[0] == "aggregateByKey":
            aggby = op
        else:
            row_ops.append(op)

    out_rows = {}
    exc_counts = {}

    def record(e):
        nm = type(e).__name__
        exc_counts[nm] = exc_counts.get(nm, 0) + 1

    for i, line in enumerate(split_rows(data)):
        cells, flags = split_cells(line, delim)
        pr = _fast_parse_row(cells, flags, col_types, null_values,
                             used=used_cols)
        if pr[0] == "row":
            val = pr[1][0] if len(col_types) == 1 else pr[1]  # 1-col = scalar row
            r = pyoracle.process_row(val, row_ops, names, fast=True)
            if sink == "csv" and r[0] == "row" and not _f64_row_fits(r[1]):
                # device %f covers |v| < ~9.2e12; larger/nan floats divert
                # and REPARSE through CPython on replay (fast_atod's value
                # may differ there — documented divergence)
                r = _replay_row(line, col_types, null_values, row_ops, names,
                                delim.decode(), used=used_cols)
        else:
            r = _replay_row(line, col_types, null_values, row_ops, names,
                            delim.decode(), used=used_cols)
        if r[0] == "row":
            out_rows[i] = [r[1]]
        elif r[0] == "exc":
            record(r[1])
        elif r[0] == "rows":  # 1:N join expansion
            out_rows[i] = r[1]
            for e in r[2]:
                record(e)

    rows = []
    for i in sorted(out_rows):
        for v in out_rows[i]:
            if isinstance(v, tuple) and len(v) == 1:
                v = v[0]
            rows.append(v)
    result = {"exception_counts": exc_counts, "columns": names,
              "col_types": col_types, "has_header": has_header}
    if agg is not None:
        _, combine_fn, agg_fn, initial = agg
        a = initial
        for v in rows:
            a = agg_fn(a, pyoracle._agg_row(v, names))
        rows = [a]
    elif aggby is not None:
        rows = pyoracle.aggregate_by_key(rows, aggby, names)
    if sink == "collect":
        result["output"] = rows
        return result

    # csv sink
    out_cols = _output_columns(row_ops, names)
    if not out_cols and rows:
        w = len(rows[0]) if isinstance(rows[0], tuple) else 1
        out_cols = ["column%d" % i for i in range(w)]  # product default
    segs = [format_csv_row(out_cols)]
    for v in rows:
        row = v if isinstance(v, tuple) else (v,)
        segs.append(format_csv_row(list(row)))
    result["csv_text"] = b"".join(segs)
    result["output"] = rows
    return result


def split_lines(data: bytes):
    """text(): rows on every newline (no quote parity)."""
    out = data.split(b"\n")
    if out and out[-1] == b"":
        out.pop()
    return out


def run_text_pipeline(text_bytes: bytes, ops, null_values=None):
    """Oracle for Context.text(): one str column per line; null_values map
    matching lines to None (core Context::text semantics)."""
    null_values = null_values or []
    agg = None
    row_ops = []
    for op in ops:
        if op[0] == "aggregate":
            agg = op
        else:
            row_ops.append(op)
    out_rows = {}
    exc_counts = {}
    for i, line in enumerate(split_lines(text_bytes)):
        v = line.decode("utf-8", "replace")
        if v.endswith("\r"):
            v = v[:-1]
        if null_values and v in null_values:
            v2 = None
        else:
            v2 = v
        r = pyoracle.process_row(v2, row_ops, None, fast=True)
        if r[0] == "row":
            vv = r[1]
            if isinstance(vv, tuple) and len(vv) == 1:
                vv = vv[0]
            out_rows[i] = vv
        elif r[0] == "exc":
            nm = type(r[1]).__name__
            exc_counts[nm] = exc_counts.get(nm, 0) + 1
    rows = [out_rows[i] for i in sorted(out_rows)]
    if agg is not None:
        _, combine_fn, agg_fn, initial = agg
        a = initial
        for v in rows:
            a = agg_fn(a, v)
        rows = [a]
    return {"output": rows, "exception_counts": exc_counts}


def _output_columns(row_ops, names):
    cols = list(names) if names else None
    for op in row_ops:
        if op[0] == "map":
            cols = None
        elif op[0] == "withColumn":
            if cols is not None and op[1] not in cols:
                cols = cols + [op[1]]
        elif op[0] == "selectColumns":
            cols = [c if isinstance(c, str) else (cols[c] if cols else
                                                  "column%d" % c)
                    for c in op[1]]
        elif op[0] == "renameColumn":
            if cols:
                cols = [op[2] if c == op[1] else c for c in cols]
        elif op[0] == "join":
            _, _rr, rcols, lk, rk, _how, lp, ls, rp, rs = op
            if cols:
                cols = ([lp + c + ls for c in cols if c != lk] +
                        [lp + lk + ls] +
                        [rp + c + rs for c in rcols if c != rk])
    return cols or []


def _f64_fits_device(v):
    """Mirror of tpx_f64_csv_n's range: True iff the device %f path formats
    this double (N = round(v*10^6) fits 63 bits; finite)."""
    import math
    import struct
    if not isinstance(v, float):
        return True
    if math.isnan(v) or math.isinf(v):
        return False
    bits = struct.unpack("<Q", struct.pack("<d", v))[0]
    exp = (bits >> 52) & 0x7FF
    man = bits & ((1 << 52) - 1)
    if exp == 0:
        m, e = man, -1074
    else:
        m, e = man | (1 << 52), exp - 1075
    M = m * 15625
    k = e + 6
    if k >= 0:
        return k < 62 and (M << k) < (1 << 63)
    k = -k
    if k >= 69:
        return True
    N = M >> k
    rem = M & ((1 << k) - 1)
    half = 1 << (k - 1)
    if rem > half or (rem == half and (N & 1)):
        N += 1
    return N < (1 << 63)


def _f64_row_fits(v):
    row = v if isinstance(v, tuple) else (v,)
    return all(_f64_fits_device(c) for c in row)


def format_cell(v) -> bytes:
    if v is None:
        s = ""
    elif isinstance(v, bool):
        s = "True" if v else "False"
    elif isinstance(v, float):
        s = "%f" % v  # the reference formats csv doubles with "%f"
                      # (PipelineBuilder.cc:1413)
    else:
        s = str(v)
    if any(c in s for c in ',"\n\r'):
        s = '"' + s.replace('"', '""') + '"'
    return s.encode("utf-8")


def format_csv_row(vals) -> bytes:
    return b",".join(format_cell(v) for v in vals) + b"\n"
