"""Type lattice subset of the reference's python::Type.

Reference: tuplex/utils/src/TypeSystem.cc (type lattice),
tuplex/python/src/PythonContext.cc:964 buildRowTypeFromSamples / :1023 inferType
(majority-type + optionize rules used by parallelize).

Types are represented as plain strings / tuples:
  'i64' 'f64' 'bool' 'str' 'null' 'emptytuple'
  ('opt', T)            Option[T]
  ('tuple', (T1,...Tn)) tuple / row type
"""
from typing import Any, Optional, Tuple, List

I64 = "i64"
F64 = "f64"
BOOL = "bool"
STR = "str"
NULL = "null"
EMPTYTUPLE = "emptytuple"


def opt(t):
    if is_opt(t) or t == NULL:
        return t
    return ("opt", t)


def is_opt(t) -> bool:
    return isinstance(t, tuple) and t[0] == "opt"


def deopt(t):
    return t[1] if is_opt(t) else t


def tup(ts) -> tuple:
    return ("tuple", tuple(ts))


def is_tuple(t) -> bool:
    return isinstance(t, tuple) and t[0] == "tuple"


def tuple_params(t):
    return t[1]


def is_varlen(t) -> bool:
    """Does a field of this type occupy the varlen section?
    (Serializer.cc: strings are var fields; null-valued option<str> still makes the
    SCHEMA a varlen schema.)"""
    t = deopt(t)
    return t == STR


def is_optional(t) -> bool:
    return is_opt(t)


def type_of_value(v: Any):
    """python value -> tuplex type (reference: python::mapPythonClassToTuplexType)."""
    if v is None:
        return NULL
    if isinstance(v, bool):
        return BOOL
    if isinstance(v, int):
        return I64
    if isinstance(v, float):
        return F64
    if isinstance(v, str):
        return STR
    if isinstance(v, tuple):
        if len(v) == 0:
            return EMPTYTUPLE
        return tup(type_of_value(x) for x in v)
    raise TypeError("unsupported python value type: %r" % (type(v),))


def unify(a, b, auto_upcast: bool = False):
    """Unify two types or return None (reference: unifyTypes, PythonContext.cc /
    TypeSystem). null+T -> Option[T]; T+Option[T] -> Option[T]; tuples elementwise;
    numeric upcasts only when auto_upcast."""
    if a == b:
        return a
    if a == NULL:
        return opt(b) if b != NULL else NULL
    if b == NULL:
        return opt(a)
    if is_opt(a) or is_opt(b):
        u = unify(deopt(a), deopt(b), auto_upcast)
        return opt(u) if u is not None else None
    if auto_upcast:
        num = {BOOL: 0, I64: 1, F64: 2}
        if a in num and b in num:
            return a if num[a] >= num[b] else b
    if is_tuple(a) and is_tuple(b):
        pa, pb = tuple_params(a), tuple_params(b)
        if len(pa) != len(pb):
            return None
        out = []
        for x, y in zip(pa, pb):
            u = unify(x, y, auto_upcast)
            if u is None:
                return None
            out.append(u)
        return tup(out)
    return None


def infer_majority_type(values: List[Any], optional_threshold: float = 0.7,
                        auto_upcast: bool = False, sample_limit: int = 10000):
    """Majority ("normal case") type of a parallelize list.

    Restates tuplex/python/src/PythonContext.cc:1023 inferType +
    :964 buildRowTypeFromSamples: count types over the sample (merging
    unifiable keys), pick the most frequent; if a tuple majority exists and the
    optionized super-tuple would cover a fraction in (1-thr, thr) beyond it, use the
    super-tuple; if NULLVALUE rows form a fraction in (1-thr, thr), wrap in Option.
    """
    n = min(len(values), sample_limit)
    if n == 0:
        return tup([F64])  # reference defaults empty parallelize to f64 schema
    counts = {}  # type -> count
    for i in range(n):
        t = type_of_value(values[i])
        if t in counts:
            counts[t] += 1
        else:
            merged = False
            for k in list(counts.keys()):
                u = unify(k, t, auto_upcast)
                if u is not None:
                    c = counts.pop(k)
                    counts[u] = counts.get(u, 0) + c + 1
                    merged = True
                    break
            if not merged:
                counts[t] = 1

    # majority, preferring "bigger" types on tie order (reference sorts by subclass)
    maj_type, maj_count = None, -1
    maj_tuple, maj_tuple_count = None, -1
    for t, c in counts.items():
        if c > maj_count:
            maj_type, maj_count = t, c
        if is_tuple(t) and c > maj_tuple_count:
            maj_tuple, maj_tuple_count = t, c

    if maj_tuple is not None:
        # optionized super-tuple check (buildRowTypeFromSamples, PythonContext.cc:999)
        super_t = maj_tuple
        num = 0
        for t, c in counts.items():
            u = unify(t, super_t, auto_upcast)
            if u is not None:
                super_t = u
                num += c
        frac = (num - counts.get(maj_tuple, 0)) / float(n)
        if num > maj_count and (1 - optional_threshold) < frac < optional_threshold:
            maj_type = super_t

    if maj_type not in (None, NULL) and NULL in counts:
        none_frac = counts[NULL] / float(n)
        if (1 - optional_threshold) < none_frac < optional_threshold:
            maj_type = opt(maj_type)
    return maj_type


def value_conforms(v: Any, t) -> bool:
    """Does value v fit type t exactly (normal-case check, no upcast)?"""
    if is_opt(t):
        return v is None or value_conforms(v, deopt(t))
    if t == NULL:
        return v is None
    if t == BOOL:
        return isinstance(v, bool)
    if t == I64:
        return isinstance(v, int) and not isinstance(v, bool) and -(2**63) <= v < 2**63
    if t == F64:
        return isinstance(v, float)
    if t == STR:
        return isinstance(v, str)
    if t == EMPTYTUPLE:
        return isinstance(v, tuple) and len(v) == 0
    if is_tuple(t):
        ps = tuple_params(t)
        return (isinstance(v, tuple) and len(v) == len(ps)
                and all(value_conforms(x, p) for x, p in zip(v, ps)))
    return False


def row_type_of(t):
    """Normalize a scalar/row type to a row ('tuple') type: scalars become 1-tuples
    (the reference's row always is a tuple type)."""
    if is_tuple(t):
        return t
    return tup([t])
