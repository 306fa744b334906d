"""UDF Python-AST -> TIR compiler.

The reference extracts UDF source via reflection (tuplex/python/tuplex/utils/
reflection.py:184 get_source), parses it with ANTLR and type-annotates from a traced
sample (codegen/src/TypeAnnotatorVisitor.cc, TraceVisitor.cc). Here: inspect/ast
source capture; direct symbolic execution of the (loop-free) statement list into a
typed expression DAG. A UDF outside the supported vocabulary raises UDFCompileError
and the pipeline falls back to the interpreter path — the reference's fallback mode
for non-compilable UDFs.

Supported statements: Assign, AugAssign, If/elif/else, Return, Expr (ignored).
Supported expressions: constants, names, bin/bool/unary ops, chained comparisons,
`in`, IfExp, str method calls (find/rfind/lower/upper/replace/strip/startswith/
endswith/swapcase), len/int/float/str/abs, subscript (column access, str index,
str slice), '%0Nd' % x format, tuple literals (return position only).
"""
import ast
import inspect
import textwrap
from typing import Callable, Dict, List, Optional

from .. import ttypes as T
from . import tir


class UDFCompileError(Exception):
    pass


def get_source(fn: Callable) -> str:
    """Reference: reflection.py:184 get_source — robust lambda/function source
    extraction."""
    try:
        src = inspect.getsource(fn)
    except (OSError, TypeError) as e:
        raise UDFCompileError("cannot get source: %s" % e)
    return textwrap.dedent(src)


def _extract_fn_ast(fn: Callable):
    src = get_source(fn)
    # source may be an expression statement containing the lambda inside a call;
    # parse and find the first Lambda or FunctionDef node
    try:
        tree = ast.parse(src)
    except SyntaxError:
        # e.g. source line is `.filter(lambda x: ...)` — strip leading dots/chars
        for strip_try in range(1, 5):
            try:
                tree = ast.parse(src.strip().lstrip("."))
                break
            except SyntaxError:
                src2 = "(" + src.strip().rstrip("\\").rstrip() + ")"
                try:
                    tree = ast.parse(src2)
                    break
                except SyntaxError:
                    continue
        else:
            raise UDFCompileError("cannot parse UDF source")
    name = getattr(fn, "__name__", "<lambda>")
    found = []
    for node in ast.walk(tree):
        if isinstance(node, ast.Lambda):
            found.append(node)
        elif isinstance(node, ast.FunctionDef) and (name == "<lambda>" or node.name == name):
            found.append(node)
    if not found:
        raise UDFCompileError("no function found in source")
    if name == "<lambda>":
        lams = [n for n in found if isinstance(n, ast.Lambda)]
        if len(lams) != 1:
            raise UDFCompileError("ambiguous lambda source")
        return lams[0]
    return found[0]


class _Ret:
    def __init__(self, node):
        self.node = node


class _Compiler:
    def __init__(self, fn, input_types: List, columns: Optional[List[str]],
                 acc_mode: bool = False):
        self.fn = fn
        self.input_types = input_types
        self.columns = columns
        self.acc_mode = acc_mode   # aggregate fn: arg0 = accumulator (input 0),
        self.row_base = 0          # arg1 = row over inputs row_base..
        self.globals = dict(getattr(fn, "__globals__", {}))
        closure = getattr(fn, "__closure__", None)
        if closure:
            for name, cell in zip(fn.__code__.co_freevars, closure):
                try:
                    self.globals[name] = cell.cell_contents
                except ValueError:
                    pass
        self.row_param = None  # param name bound to whole row (dict/tuple access)

    def compile(self):
        node = _extract_fn_ast(self.fn)
        args = node.args.args
        if node.args.vararg or node.args.kwonlyargs or node.args.kwarg:
            raise UDFCompileError("varargs not supported")
        env: Dict[str, dict] = {}
        if self.acc_mode:
            if len(args) != 2:
                raise UDFCompileError("aggregate UDF must take (acc, row)")
            env[args[0].arg] = tir.inp(0, self.input_types[0])
            self.row_param = args[1].arg
            self.row_base = 1
            if len(self.input_types) == 2:
                env[args[1].arg] = tir.inp(1, self.input_types[1])
        elif len(args) == 1:
            p = args[0].arg
            self.row_param = p  # x['col'] / x[i] row access always allowed
            if len(self.input_types) == 1:
                env[p] = tir.inp(0, self.input_types[0])  # bare-scalar use
        elif len(args) == len(self.input_types):
            for i, a in enumerate(args):
                env[a.arg] = tir.inp(i, self.input_types[i])
        else:
            raise UDFCompileError("UDF arity %d vs row arity %d"
                                  % (len(args), len(self.input_types)))
        if isinstance(node, ast.Lambda):
            return self.expr(node.body, env)
        res = self.exec_stmts(node.body, 0, env)
        if isinstance(res, _Ret):
            return res.node
        raise UDFCompileError("UDF may fall off the end (implicit None return)")

    # ---- statements ----------------------------------------------------------
    def exec_stmts(self, stmts, i, env):
        while i < len(stmts):
            s = stmts[i]
            if isinstance(s, ast.Return):
                if s.value is None:
                    raise UDFCompileError("bare return")
                return _Ret(self.expr(s.value, env))
            if isinstance(s, ast.Assign):
                if len(s.targets) != 1 or not isinstance(s.targets[0], ast.Name):
                    raise UDFCompileError("only simple assignment supported")
                env[s.targets[0].id] = self.expr(s.value, env)
                i += 1
                continue
            if isinstance(s, ast.AugAssign):
                if not isinstance(s.target, ast.Name):
                    raise UDFCompileError("only simple augassign")
                cur = env.get(s.target.id)
                if cur is None:
                    raise UDFCompileError("augassign of unbound %s" % s.target.id)
                env[s.target.id] = self._binop_ast(s.op, cur, self.expr(s.value, env))
                i += 1
                continue
            if isinstance(s, ast.If):
                cond = self.expr(s.test, env)
                rest = stmts[i + 1:]
                r1 = self.exec_stmts(list(s.body) + rest, 0, dict(env))
                r2 = self.exec_stmts(list(s.orelse) + rest, 0, dict(env))
                if isinstance(r1, _Ret) and isinstance(r2, _Ret):
                    return _Ret(self._merge_ret(cond, r1.node, r2.node))
                raise UDFCompileError("UDF path falls off the end")
            if isinstance(s, ast.Expr):
                i += 1  # docstring / no-op
                continue
            if isinstance(s, ast.Pass):
                i += 1
                continue
            raise UDFCompileError("unsupported statement %s" % type(s).__name__)
        return env  # fell off the end

    def _merge_ret(self, cond, a, b):
        # tuples merge elementwise
        if a.get("op") == "mktuple" and b.get("op") == "mktuple":
            if len(a["args"]) != len(b["args"]):
                raise UDFCompileError("branch tuple arity mismatch")
            elems = [tir.ifexpr(cond, x, y) for x, y in zip(a["args"], b["args"])]
            return tir.mk("mktuple", T.tup(e["t"] for e in elems), elems)
        try:
            return tir.ifexpr(cond, a, b)
        except tir.TirError as e:
            raise UDFCompileError(str(e))

    # ---- expressions ---------------------------------------------------------
    def expr(self, e, env):
        try:
            return self._expr(e, env)
        except tir.TirError as ex:
            raise UDFCompileError(str(ex))

    def _expr(self, e, env):
        if isinstance(e, ast.Constant):
            if e.value is None or isinstance(e.value, (bool, int, float, str)):
                return tir.const(e.value)
            raise UDFCompileError("unsupported constant %r" % (e.value,))
        if isinstance(e, ast.Name):
            if e.id in env:
                return env[e.id]
            if e.id in self.globals:
                v = self.globals[e.id]
                if v is None or isinstance(v, (bool, int, float, str)):
                    return tir.const(v)
            raise UDFCompileError("unbound name %s" % e.id)
        if isinstance(e, ast.BinOp):
            if isinstance(e.op, ast.Mod) and self._is_str_expr(e.left):
                return self._fmt(e, env)
            return self._binop_ast(e.op, self._expr(e.left, env), self._expr(e.right, env))
        if isinstance(e, ast.UnaryOp):
            v = self._expr(e.operand, env)
            if isinstance(e.op, ast.USub):
                if v["op"] == "const" and isinstance(v.get("v"), (int, float)):
                    return tir.const(-v["v"])
                return tir.neg(v)
            if isinstance(e.op, ast.Not):
                return tir.notop(v)
            raise UDFCompileError("unsupported unary op")
        if isinstance(e, ast.BoolOp):
            op = "and" if isinstance(e.op, ast.And) else "or"
            node = self._expr(e.values[0], env)
            for v in e.values[1:]:
                node = tir.boolop(op, node, self._expr(v, env))
            return node
        if isinstance(e, ast.Compare):
            left = self._expr(e.left, env)
            result = None
            for op, comp in zip(e.ops, e.comparators):
                right = self._expr(comp, env)
                c = self._cmp_ast(op, left, right)
                result = c if result is None else tir.boolop("and", result, c)
                left = right
            return result
        if isinstance(e, ast.IfExp):
            return tir.ifexpr(self._expr(e.test, env), self._expr(e.body, env),
                              self._expr(e.orelse, env))
        if isinstance(e, ast.Call):
            return self._call(e, env)
        if isinstance(e, ast.Subscript):
            return self._subscript(e, env)
        if isinstance(e, ast.Tuple):
            elems = [self._expr(x, env) for x in e.elts]
            return tir.mk("mktuple", T.tup(x["t"] for x in elems), elems)
        raise UDFCompileError("unsupported expression %s" % type(e).__name__)

    def _is_str_expr(self, e):
        return isinstance(e, ast.Constant) and isinstance(e.value, str)

    def _fmt(self, e, env):
        spec = e.left.value
        arg = self._expr(e.right, env)
        # subset: single %d / %0Nd / %s directive
        import re
        m = re.fullmatch(r"%0(\d+)d", spec)
        if m:  # '%05d' -> zero-padded width N
            return tir.fmt(int(m.group(1)), arg)
        if spec == "%d":
            return tir.fmt(0, arg)
        if spec == "%s":
            return tir.call("to_str", [arg])
        raise UDFCompileError("unsupported format spec %r" % spec)

    def _binop_ast(self, op, a, b):
        m = {ast.Add: "add", ast.Sub: "sub", ast.Mult: "mul", ast.Div: "truediv",
             ast.FloorDiv: "floordiv", ast.Mod: "mod"}
        k = m.get(type(op))
        if k is None:
            raise UDFCompileError("unsupported operator %s" % type(op).__name__)
        return tir.binop(k, a, b)

    def _cmp_ast(self, op, a, b):
        m = {ast.Lt: "lt", ast.LtE: "le", ast.Gt: "gt", ast.GtE: "ge",
             ast.Eq: "eq", ast.NotEq: "ne"}
        if isinstance(op, ast.In):
            return tir.call("contains", [b, a])  # contains(haystack, needle)
        if isinstance(op, ast.NotIn):
            return tir.notop(tir.call("contains", [b, a]))
        if isinstance(op, (ast.Is, ast.IsNot)):
            # `x is None` style
            k = "eq" if isinstance(op, ast.Is) else "ne"
            return tir.binop(k, a, b)
        k = m.get(type(op))
        if k is None:
            raise UDFCompileError("unsupported comparison")
        return tir.binop(k, a, b)

    def _call(self, e, env):
        if isinstance(e.func, ast.Attribute):
            obj = self._expr(e.func.value, env)
            meth = e.func.attr
            args = [self._expr(a, env) for a in e.args]
            if meth in ("find", "rfind", "lower", "upper", "strip", "replace",
                        "startswith", "endswith", "swapcase", "center"):
                return tir.call(meth if meth not in ("find", "rfind") else meth,
                                [obj] + args)
            raise UDFCompileError("unsupported method .%s" % meth)
        if isinstance(e.func, ast.Name):
            name = e.func.id
            args = [self._expr(a, env) for a in e.args]
            if name == "str":
                return tir.call("to_str", args)
            if name in ("int", "float", "len", "abs"):
                return tir.call(name, args)
            raise UDFCompileError("unsupported call %s()" % name)
        raise UDFCompileError("unsupported call form")

    def _subscript(self, e, env):
        # row access: x['col'] / x[i] on the row param. Like the reference's traced
        # typing, the meaning of x[...] depends on what x IS: when x is bound to a
        # single str value, x[0] / x[1:] are string ops (CPython semantics); a str
        # key is always column access (str[str] is invalid Python).
        if isinstance(e.value, ast.Name) and e.value.id == self.row_param:
            scalar = env.get(self.row_param)
            scalar_is_str = scalar is not None and T.deopt(scalar["t"]) == T.STR
            if (isinstance(e.slice, ast.Constant)
                    and isinstance(e.slice.value, str)):
                key = e.slice.value
                if not self.columns or key not in self.columns:
                    raise UDFCompileError("unknown column %r" % key)
                i = self.row_base + self.columns.index(key)
                return tir.inp(i, self.input_types[i])
            if not scalar_is_str:
                if (isinstance(e.slice, ast.Constant)
                        and isinstance(e.slice.value, int)):
                    key = self.row_base + e.slice.value
                    if not (0 <= key < len(self.input_types)):
                        raise UDFCompileError("column index out of range")
                    return tir.inp(key, self.input_types[key])
                raise UDFCompileError("row subscript must be a constant")
            # else: fall through to string getitem/slice on the scalar binding
        # s.split(sep)[i] -> fused splitget (logs pipelines)
        if (isinstance(e.value, ast.Call) and isinstance(e.value.func, ast.Attribute)
                and e.value.func.attr == "split" and len(e.value.args) == 1
                and not isinstance(e.slice, ast.Slice)):
            s = self._expr(e.value.func.value, env)
            sep = self._expr(e.value.args[0], env)
            idx = self._expr(e.slice, env)
            return tir.splitget(s, sep, idx)
        obj = self._expr(e.value, env)
        if isinstance(e.slice, ast.Slice):
            if e.slice.step is not None:
                raise UDFCompileError("slice step not supported")
            lo = self._expr(e.slice.lower, env) if e.slice.lower else None
            hi = self._expr(e.slice.upper, env) if e.slice.upper else None
            return tir.strslice(obj, lo, hi)
        idx = self._expr(e.slice, env)
        return tir.getitem(obj, idx)


def compile_udf(fn: Callable, input_types: List, columns: Optional[List[str]] = None):
    """Compile a Python UDF to a TIR node given the input row's column types.
    Returns the root node (output type in node['t']). Raises UDFCompileError for
    anything outside the vocabulary -> caller falls back to interpreter mode."""
    return _Compiler(fn, input_types, columns).compile()


def compile_agg_udf(fn: Callable, acc_type, row_types: List,
                    columns: Optional[List[str]] = None):
    """Compile an aggregate UDF `lambda a, x: ...` — input 0 is the accumulator,
    inputs 1.. are the row columns (AggregateFunctions.cc functor shape)."""
    return _Compiler(fn, [acc_type] + list(row_types), columns,
                     acc_mode=True).compile()
