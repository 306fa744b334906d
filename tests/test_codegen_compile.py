"""Codegen + hipRTC compile checks (CPU container: compile-only, no module load).

These catch generated-source syntax/type errors before any GPU time is spent, and
verify the C-ABI library loads and exports every symbol include/tpx_abi.h declares.
"""
import ctypes
import os

import pytest

from tuplex_amd import codegen, plan
from tuplex_amd import ttypes as T

HERE = os.path.dirname(os.path.abspath(__file__))
LIB = os.path.join(os.path.dirname(HERE), "tuplex_amd", "libtpx_gpu.so")


def _lib():
    if not os.path.exists(LIB):
        pytest.skip("libtpx_gpu.so not built")
    lib = ctypes.CDLL(LIB)
    lib.tpx_stage_compile.restype = ctypes.c_void_p
    lib.tpx_stage_compile.argtypes = [ctypes.c_char_p] * 3 + [ctypes.c_int64]
    lib.tpx_last_error.restype = ctypes.c_char_p
    lib.tpx_stage_free.argtypes = [ctypes.c_void_p]
    return lib


def test_abi_symbols():
    lib = _lib()
    for sym in ["tpx_version", "tpx_device_count", "tpx_set_device",
                "tpx_last_error", "tpx_stage_compile", "tpx_stage_free",
                "tpx_stage_execute", "tpx_stage_execute_csv", "tpx_result_free",
                "tpx_stage_source"]:
        assert getattr(lib, sym) is not None


def _compile_only(sp, source="mem", sink="mem"):
    src, desc = codegen.generate_stage(sp, source=source, sink=sink)
    lib = _lib()
    cache = os.path.join(os.path.dirname(HERE), "tuplex_amd", ".kernel_cache")
    os.makedirs(cache, exist_ok=True)
    h = lib.tpx_stage_compile(src.encode(), desc.encode(), cache.encode(), 1)
    if not h:
        err = lib.tpx_last_error().decode()
        raise AssertionError("hipRTC compile failed:\n%s\n--- source ---\n%s"
                             % (err[:4000], _numbered(src)))
    lib.tpx_stage_free(h)
    return src


def _numbered(src):
    return "\n".join("%4d %s" % (i + 1, l)
                     for i, l in enumerate(src.splitlines()))


def square_map(x):
    return (x, x * x)


def test_compile_square_map():
    sp = plan.build_stage([T.I64], None, [("map", square_map)])
    assert sp.compilable, sp.why_not_compilable
    src = _compile_only(sp)
    assert "tpx_process" in src and "tpx_stage_main" in src


def div_map(x):
    return 1 // x


def test_compile_div_map():
    sp = plan.build_stage([T.I64], None, [("map", div_map)])
    assert sp.compilable
    _compile_only(sp)


def str_of(x):
    return str(x)


def test_compile_option_str():
    sp = plan.build_stage([T.opt(T.I64)], None, [("map", str_of)])
    assert sp.compilable, sp.why_not_compilable
    _compile_only(sp)


def pred(x):
    return (1 // x) < 5


def test_compile_filter():
    sp = plan.build_stage([T.I64], None, [("filter", pred)])
    assert sp.compilable
    _compile_only(sp)


# ---- Zillow stage (full pipeline of benchmarks/zillow/Z1/runtuplex.py:192) -------

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


def extractBa(x):
    val = x["facts and features"]
    max_idx = val.find(" ba")
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


def extractSqft(x):
    val = x["facts and features"]
    max_idx = val.find(" sqft")
    if max_idx < 0:
        max_idx = len(val)
    s = val[:max_idx]
    split_idx = s.rfind("ba ,")
    if split_idx < 0:
        split_idx = 0
    else:
        split_idx += 5
    r = s[split_idx:]
    r = r.replace(",", "")
    return int(r)


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


def extractType(x):
    t = x["title"].lower()
    type = "unknown"
    if "condo" in t or "apartment" in t:
        type = "condo"
    if "house" in t:
        type = "house"
    return type


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


ZILLOW_COLS = ["title", "address", "city", "state", "postal_code", "price",
               "facts and features", "real estate provider", "url", "sales_date"]


def zillow_ops():
    return [
        ("withColumn", "bedrooms", extractBd),
        ("filter", lambda x: x["bedrooms"] < 10),
        ("withColumn", "type", extractType),
        ("filter", lambda x: x["type"] == "house"),
        ("withColumn", "zipcode", lambda x: "%05d" % int(x["postal_code"])),
        ("mapColumn", "city", lambda x: x[0].upper() + x[1:].lower()),
        ("withColumn", "bathrooms", extractBa),
        ("withColumn", "sqft", extractSqft),
        ("withColumn", "offer", extractOffer),
        ("withColumn", "price", extractPrice),
        ("filter", lambda x: 100000 < x["price"] < 2e7),
        ("selectColumns", ["url", "zipcode", "address", "city", "state",
                           "bedrooms", "bathrooms", "sqft", "offer", "type",
                           "price"]),
    ]


def zillow_input_types():
    # sniffed zillow schema: all str except postal_code (f64-looking "1801.0")
    return [T.STR, T.STR, T.STR, T.STR, T.F64, T.STR, T.STR, T.STR, T.STR, T.STR]


def test_compile_zillow_stage_mem():
    sp = plan.build_stage(zillow_input_types(), ZILLOW_COLS, zillow_ops())
    assert sp.compilable, sp.why_not_compilable
    assert [T.deopt(t) for t in sp.output_types] == \
        [T.STR, T.STR, T.STR, T.STR, T.STR, T.I64, T.I64, T.I64, T.STR, T.STR, T.I64]
    _compile_only(sp, source="mem", sink="mem")


def test_compile_zillow_stage_csv_sink():
    sp = plan.build_stage(zillow_input_types(), ZILLOW_COLS, zillow_ops())
    assert sp.compilable, sp.why_not_compilable
    _compile_only(sp, source="mem", sink="csv")


def test_needle_window_mask_math():
    """CPU check of _emit_scan_group's register-window compare: for random
    8-byte windows, ((w | M) & lenmask) == P must hold iff every needle byte
    matches under the ci rule (letter byte b matches c iff (c|0x20)==b)."""
    import random
    from tuplex_amd.codegen import StageCodegen
    rng = random.Random(5)
    pool = "abzAZB Q,:/q3"
    for _ in range(3000):
        nlen = rng.randint(1, 8)
        needle = "".join(rng.choice("abz :,q") for _ in range(nlen))
        ci = rng.random() < 0.5
        P, M, lm, k = StageCodegen._needle_window(needle, ci)
        win = [rng.choice(pool) for _ in range(8)]
        if rng.random() < 0.5:  # force a (possibly case-flipped) match
            for j, c in enumerate(needle[:8]):
                win[j] = c.upper() if (ci and rng.random() < 0.5 and
                                       "a" <= c <= "z") else c
        w = 0
        for j, c in enumerate(win):
            w |= ord(c) << (8 * j)
        got = ((w | M) & lm) == P
        exp = True
        for j, c in enumerate(needle[:k]):
            b = ord(win[j])
            if ci and "a" <= c <= "z":
                if (b | 0x20) != ord(c):
                    exp = False
            elif b != ord(c):
                exp = False
        assert got == exp, (needle, ci, win, hex(P), hex(M))
