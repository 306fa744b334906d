"""Typed expression IR (TIR) — the replacement for the reference's typed UDF AST.

The reference parses UDF source with ANTLR, type-annotates from a sample trace
(tuplex/codegen/src/TypeAnnotatorVisitor.cc) and emits LLVM IR per operation
(BlockGeneratorVisitor.cc, FunctionRegistry.cc). Here a UDF becomes a DAG of typed
nodes; codegen.py turns the DAG into HIP C++ statements.

Node = dict {op, t (result type), args:[nodes], plus op-specific fields v/i/w}.
Ops that can raise carry implicit exception semantics; the raising ops and their
codes (ExceptionCodes.h:26):
  floordiv/mod/truediv by zero -> ZERODIVISIONERROR(136)
  getitem out of range         -> INDEXERROR(111)
  int_str parse failure        -> VALUEERROR(135)   (Runtime.cc:319 fast_atoi64 wrapper)
  float_str parse failure      -> VALUEERROR(135)
  null where value required    -> TYPEERROR(129)    (CPython: unsupported operand NoneType)
"""
from .. import ttypes as T


class TirError(Exception):
    pass


def mk(op, t, args=(), **kw):
    n = {"op": op, "t": t, "args": list(args)}
    n.update(kw)
    return n


def const(v):
    return mk("const", T.type_of_value(v), v=v)


def inp(i, t):
    return mk("input", t, i=i)


_NUM = (T.I64, T.F64, T.BOOL)


def _numt(a, b):
    """Numeric promotion (Python semantics: bool<int<float)."""
    ta, tb = a["t"], b["t"]
    if ta not in _NUM or tb not in _NUM:
        return None
    if T.F64 in (ta, tb):
        return T.F64
    return T.I64


def _deopt_node(n):
    """Use of a possibly-null value where a value is required: wrap in a null guard
    (raises TYPEERROR at runtime, like CPython None arithmetic)."""
    if T.is_opt(n["t"]):
        return mk("unwrap", T.deopt(n["t"]), [n])
    return n


def binop(op, a, b):
    if op in ("add", "sub", "mul", "truediv", "floordiv", "mod"):
        a, b = _deopt_node(a), _deopt_node(b)
        if op == "add" and a["t"] == T.STR and b["t"] == T.STR:
            return mk("concat", T.STR, [a, b])
        if op == "mod" and a["t"] == T.STR:
            raise TirError("%-format must go through fmt()")
        if op == "mul":
            # str * int / int * str repetition (python sequence semantics)
            if a["t"] == T.STR and b["t"] in (T.I64, T.BOOL):
                return mk("strmul", T.STR, [a, b])
            if b["t"] == T.STR and a["t"] in (T.I64, T.BOOL):
                return mk("strmul", T.STR, [b, a])
        nt = _numt(a, b)
        if nt is None:
            raise TirError("unsupported operand types for %s: %r %r" % (op, a["t"], b["t"]))
        if op == "truediv":
            nt = T.F64
        return mk(op, nt, [a, b])
    if op in ("lt", "le", "gt", "ge"):
        a, b = _deopt_node(a), _deopt_node(b)
        if a["t"] == T.STR and b["t"] == T.STR:
            return mk("str" + op, T.BOOL, [a, b])
        if _numt(a, b) is None:
            raise TirError("unsupported comparison %r %r" % (a["t"], b["t"]))
        return mk(op, T.BOOL, [a, b])
    if op in ("eq", "ne"):
        # None comparisons are valid Python (x == None)
        if a["t"] == T.NULL or b["t"] == T.NULL or T.is_opt(a["t"]) or T.is_opt(b["t"]):
            return mk("opteq" if op == "eq" else "optne", T.BOOL, [a, b])
        if a["t"] == T.STR and b["t"] == T.STR:
            return mk("streq" if op == "eq" else "strne", T.BOOL, [a, b])
        if _numt(a, b) is None:
            raise TirError("unsupported equality %r %r" % (a["t"], b["t"]))
        return mk(op, T.BOOL, [a, b])
    raise TirError("unknown binop %s" % op)


def boolop(op, a, b):
    if a["t"] != T.BOOL or b["t"] != T.BOOL:
        raise TirError("and/or operands must be bool-typed (truthiness subset)")
    return mk(op, T.BOOL, [a, b])


def notop(a):
    if a["t"] != T.BOOL:
        raise TirError("not operand must be bool")
    return mk("not", T.BOOL, [a])


def neg(a):
    a = _deopt_node(a)
    if a["t"] not in _NUM:
        raise TirError("neg needs numeric")
    return mk("neg", T.I64 if a["t"] in (T.I64, T.BOOL) else T.F64, [a])


def ifexpr(c, a, b):
    if c["t"] != T.BOOL:
        raise TirError("if condition must be bool")
    u = T.unify(a["t"], b["t"], auto_upcast=True)
    if u is None:
        raise TirError("if branches don't unify: %r %r" % (a["t"], b["t"]))
    return mk("if", u, [c, a, b])


def call(name, args):
    """String methods / builtins. Mirrors the reference's FunctionRegistry surface
    (codegen/src/FunctionRegistry.cc) for the benchmark UDF vocabulary."""
    a = [(_deopt_node(x) if name != "to_str" else x) for x in args]
    if name == "len":
        if a[0]["t"] != T.STR:
            raise TirError("len on non-str")
        return mk("len", T.I64, a)
    if name in ("find", "rfind"):
        _need(a, (T.STR, T.STR))
        return mk("str" + name, T.I64, a)
    if name in ("lower", "upper", "strip", "swapcase"):
        _need(a, (T.STR,))
        return mk(name, T.STR, a)
    if name == "replace":
        _need(a, (T.STR, T.STR, T.STR))
        return mk("replace", T.STR, a)
    if name == "center":
        # s.center(width[, fill]) — FunctionRegistry strCenter
        # (StringFunctions.cc:224); bool widths coerce like CPython
        if len(a) not in (2, 3) or a[0]["t"] != T.STR:
            raise TirError("center args")
        if a[1]["t"] not in (T.I64, T.BOOL):
            raise TirError("center width must be int")
        if len(a) == 3 and a[2]["t"] != T.STR:
            raise TirError("center fill must be str")
        return mk("center", T.STR, a)
    if name in ("startswith", "endswith"):
        _need(a, (T.STR, T.STR))
        return mk(name, T.BOOL, a)
    if name == "contains":  # `sub in s` -> contains(s, sub)
        _need(a, (T.STR, T.STR))
        return mk("contains", T.BOOL, a)
    if name == "int":
        t = a[0]["t"]
        if t == T.STR:
            return mk("int_str", T.I64, a)
        if t == T.F64:
            return mk("int_f64", T.I64, a)
        if t in (T.I64, T.BOOL):
            return mk("int_i64", T.I64, a)
        raise TirError("int() on %r" % (t,))
    if name == "float":
        t = a[0]["t"]
        if t == T.STR:
            return mk("float_str", T.F64, a)
        if t in (T.I64, T.BOOL, T.F64):
            return mk("float_num", T.F64, a)
        raise TirError("float() on %r" % (t,))
    if name == "to_str":
        return mk("to_str", T.STR, args)  # null-aware: str(None) == 'None'
    if name == "abs":
        if a[0]["t"] not in _NUM:
            raise TirError("abs on %r" % (a[0]["t"],))
        return mk("abs", T.F64 if a[0]["t"] == T.F64 else T.I64, a)
    raise TirError("unsupported function %s" % name)


def _need(args, types):
    if len(args) != len(types):
        raise TirError("arity")
    for x, t in zip(args, types):
        if x["t"] != t:
            raise TirError("arg type %r != %r" % (x["t"], t))


def getitem(s, i):
    s = _deopt_node(s)
    if s["t"] != T.STR or i["t"] != T.I64:
        raise TirError("getitem needs (str, i64)")
    return mk("getitem", T.STR, [s, i])


def strslice(s, lo, hi):
    """s[lo:hi]; lo/hi None or i64 nodes; Python clamp semantics, no step."""
    s = _deopt_node(s)
    if s["t"] != T.STR:
        raise TirError("slice on non-str")
    args = [s, lo if lo is not None else const(None), hi if hi is not None else const(None)]
    return mk("slice", T.STR, args)


def splitget(s, sep, idx):
    """s.split(sep)[i] — fused (the logs-pipeline idiom). Part indexing is
    byte-exact for UTF-8 (parts are substrings, not char indices)."""
    s = _deopt_node(s)
    if s["t"] != T.STR or sep["t"] != T.STR or idx["t"] != T.I64:
        raise TirError("split needs (str, str)[int]")
    return mk("splitget", T.STR, [s, sep, idx])


def fmt(spec, arg):
    """'%05d' % x style formatting; spec like (width, zero_pad) for %d."""
    arg = _deopt_node(arg)
    if arg["t"] not in (T.I64, T.BOOL):
        raise TirError("fmt %d needs int")
    return mk("fmt_int", T.STR, [arg], w=spec)


def walk(node, seen=None):
    """Iterate DAG nodes once each (post-order)."""
    if seen is None:
        seen = set()
    if id(node) in seen:
        return
    seen.add(id(node))
    for c in node["args"]:
        yield from walk(c, seen)
    yield node
