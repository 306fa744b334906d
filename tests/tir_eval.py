"""TEST INFRASTRUCTURE: reference evaluator for TIR nodes.

Evaluates a compiled UDF's TIR DAG on Python values with the same semantics the HIP
codegen must implement (including the compiled-path int()/float() parse semantics of
Runtime.cc:319-383). Used to check tuplex_amd.udf.compile output against plain
CPython execution of the same UDF; the GPU parity tests then check the generated HIP
kernel against the same expectations. Not part of the product path."""
import math

from oracle import pyoracle


class TirExc(Exception):
    def __init__(self, code_name):
        self.code_name = code_name
        super().__init__(code_name)


def ev(node, row):
    memo = {}

    def rec(n):
        key = id(n)
        if key in memo:
            return memo[key]
        v = _ev(n, row, rec)
        memo[key] = v
        return v

    return rec(node)


def _pyslice(s, lo, hi):
    return s[slice(lo, hi)]


def _ev(n, row, rec):
    op = n["op"]
    a = n["args"]
    if op == "const":
        return n["v"]
    if op == "input":
        return row[n["i"]]
    if op == "unwrap":
        v = rec(a[0])
        if v is None:
            raise TypeError("unsupported operand type: 'NoneType'")
        return v
    if op in ("add", "sub", "mul", "truediv", "floordiv", "mod"):
        x, y = rec(a[0]), rec(a[1])
        if op == "add":
            return x + y
        if op == "sub":
            return x - y
        if op == "mul":
            return x * y
        if op == "truediv":
            return x / y
        if op == "floordiv":
            return x // y
        return x % y
    if op == "concat":
        return rec(a[0]) + rec(a[1])
    if op == "strmul":
        return rec(a[0]) * int(rec(a[1]))
    if op == "center":
        f = rec(a[2]) if len(a) == 3 else " "
        return rec(a[0]).center(int(rec(a[1])), f)
    if op in ("lt", "le", "gt", "ge", "eq", "ne", "strlt", "strle", "strgt",
              "strge", "streq", "strne", "opteq", "optne"):
        x, y = rec(a[0]), rec(a[1])
        base = op[3:] if op.startswith(("str", "opt")) else op
        return {"lt": x < y, "le": x <= y, "gt": x > y, "ge": x >= y,
                "eq": x == y, "ne": x != y}[base]
    if op == "and":
        return rec(a[0]) and rec(a[1])
    if op == "or":
        return rec(a[0]) or rec(a[1])
    if op == "not":
        return not rec(a[0])
    if op == "neg":
        return -rec(a[0])
    if op == "if":
        return rec(a[1]) if rec(a[0]) else rec(a[2])
    if op == "len":
        return len(rec(a[0]))
    if op == "strfind":
        return rec(a[0]).find(rec(a[1]))
    if op == "strrfind":
        return rec(a[0]).rfind(rec(a[1]))
    if op in ("lower", "upper", "strip", "swapcase"):
        return getattr(rec(a[0]), op)()
    if op == "replace":
        return rec(a[0]).replace(rec(a[1]), rec(a[2]))
    if op == "startswith":
        return rec(a[0]).startswith(rec(a[1]))
    if op == "endswith":
        return rec(a[0]).endswith(rec(a[1]))
    if op == "contains":
        return rec(a[1]) in rec(a[0])
    if op == "getitem":
        return rec(a[0])[rec(a[1])]
    if op == "slice":
        lo = rec(a[1])
        hi = rec(a[2])
        return _pyslice(rec(a[0]), lo, hi)
    if op == "int_str":
        return pyoracle.ref_int(rec(a[0]))
    if op == "int_f64":
        return int(rec(a[0]))
    if op == "int_i64":
        return int(rec(a[0]))
    if op == "float_str":
        return pyoracle.ref_float(rec(a[0]))
    if op == "float_num":
        return float(rec(a[0]))
    if op == "to_str":
        v = rec(a[0])
        if isinstance(v, float):
            return repr(v)
        return str(v)
    if op == "abs":
        return abs(rec(a[0]))
    if op == "splitget":
        return rec(a[0]).split(rec(a[1]))[rec(a[2])]
    if op == "fmt_int":
        v = rec(a[0])
        return ("%0" + str(n["w"]) + "d") % v if n["w"] else "%d" % v
    if op == "mktuple":
        return tuple(rec(x) for x in a)
    raise ValueError("unknown op %r" % op)
