"""UDF compiler tests: compile the benchmark UDF vocabulary (Zillow pipeline of
benchmarks/zillow/Z1/runtuplex.py:12-110, the test_exceptions lambdas) to TIR and
check TIR evaluation == CPython execution row by row."""
import pytest

from tuplex_amd import ttypes as T
from tuplex_amd.udf import compile_udf, UDFCompileError
from tests import tir_eval


# --- Zillow UDFs (same shapes as benchmarks/zillow/Z1/runtuplex.py) ---------------

def extractBd(x):
    val = x["facts and features"]
    max_idx = val.find(" bd")
    if max_idx < 0:
        max_idx = len(val)
    s = val[:max_idx]
    split_idx = s.rfind(",")
    if split_idx < 0:
        split_idx = 0
    else:
        split_idx += 2
    r = s[split_idx:]
    return int(r)


def extractType(x):
    t = x["title"].lower()
    type = "unknown"
    if "condo" in t or "apartment" in t:
        type = "condo"
    if "house" in t:
        type = "house"
    return type


def extractOffer(x):
    offer = x["title"].lower()
    if "sale" in offer:
        return "sale"
    if "rent" in offer:
        return "rent"
    if "sold" in offer:
        return "sold"
    if "foreclose" in offer.lower():
        return "foreclosed"
    return offer


def extractPrice(x):
    price = x["price"]
    p = 0
    if x["offer"] == "sold":
        val = x["facts and features"]
        s = val[val.find("Price/sqft:") + len("Price/sqft:") + 1:]
        r = s[s.find("$") + 1:s.find(", ") - 1]
        price_per_sqft = int(r)
        p = price_per_sqft * x["sqft"]
    elif x["offer"] == "rent":
        max_idx = price.rfind("/")
        p = int(price[1:max_idx].replace(",", ""))
    else:
        p = int(price[1:].replace(",", ""))
    return p


ZCOLS = ["title", "price", "facts and features", "offer", "sqft", "postal_code"]
ZTYPES = [T.STR, T.STR, T.STR, T.STR, T.I64, T.STR]

ZROWS = [
    ("House For Sale", "$489,000", "3 bds , 1 ba , 1,560 sqft", "sale", 1560, "1801"),
    ("Condo For Rent", "$3,700/mo", "2 bds , 1 ba , 920 sqft", "rent", 920, "2215"),
    ("House Sold", "$250,000", "Price/sqft: $161, 3 bds", "sold", 1550, "60614"),
    ("Apartment for sale", "$1,250,000", "4 bds , 3 ba , 2,480 sqft", "sale", 2480, "94107"),
]


def _check(fn, rows=ZROWS, cols=ZCOLS, typs=ZTYPES):
    node = compile_udf(fn, typs, cols)
    for row in rows:
        expect = fn(dict(zip(cols, row)))
        got = tir_eval.ev(node, row)
        assert got == expect, (fn.__name__, row, got, expect)
    return node


def test_extract_bd():
    n = _check(extractBd)
    assert n["t"] == T.I64


def test_extract_type():
    n = _check(extractType)
    assert n["t"] == T.STR


def test_extract_offer():
    _check(extractOffer)


def test_extract_price():
    n = _check(extractPrice)
    assert n["t"] == T.I64


def test_zipcode_format():
    fn = lambda x: "%05d" % int(x["postal_code"])  # noqa: E731
    n = _check(fn)
    assert n["t"] == T.STR


def test_city_mapcolumn_udf():
    fn = lambda x: x[0].upper() + x[1:].lower()  # noqa: E731
    node = compile_udf(fn, [T.STR], None)
    for s in ["wOBURN", "boston", "X", "ab"]:
        assert tir_eval.ev(node, (s,)) == fn(s)


def test_filter_chain_compare():
    fn = lambda x: 100000 < x["price"] < 2e7  # noqa: E731
    node = compile_udf(fn, [T.I64], ["price"])
    assert node["t"] == T.BOOL
    for v in [99999, 100000, 100001, 2 * 10**7, 5 * 10**6]:
        assert tir_eval.ev(node, (v,)) == fn({"price": v})


def test_simple_lambda_division():
    node = compile_udf(lambda x: 1 // x, [T.I64], None)
    assert tir_eval.ev(node, (2,)) == 0
    with pytest.raises(ZeroDivisionError):
        tir_eval.ev(node, (0,))


def test_tuple_output():
    node = compile_udf(lambda x: (x, x * x), [T.I64], None)
    assert node["t"] == T.tup([T.I64, T.I64])
    assert tir_eval.ev(node, (3,)) == (3, 9)


def test_multi_param():
    node = compile_udf(lambda s, p: s.startswith(p), [T.STR, T.STR], None)
    assert tir_eval.ev(node, ("hello", "he")) is True
    assert tir_eval.ev(node, ("Hello", "hello")) is False


def test_option_str_of():
    node = compile_udf(lambda x: str(x), [T.opt(T.I64)], None)
    assert node["t"] == T.STR
    assert tir_eval.ev(node, (None,)) == "None"
    assert tir_eval.ev(node, (5,)) == "5"


def test_uncompilable_falls_out():
    with pytest.raises(UDFCompileError):
        compile_udf(lambda x: [i for i in range(x)], [T.I64], None)
    with pytest.raises(UDFCompileError):
        compile_udf(lambda x: x.unknown_method(), [T.STR], None)


def test_getitem_index_error():
    node = compile_udf(lambda x: x[0], [T.STR], None)
    with pytest.raises(IndexError):
        tir_eval.ev(node, ("",))


def test_int_parse_valueerror():
    node = compile_udf(lambda x: int(x), [T.STR], None)
    assert tir_eval.ev(node, ("42",)) == 42
    with pytest.raises(ValueError):
        tir_eval.ev(node, ("4x",))
