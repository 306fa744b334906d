"""CPU regression net for the device %f algorithm: the same exact
round-half-even N = v*10^6 computation (128-bit fixed point) must agree with
CPython's correctly-rounded "%f" wherever the device range check accepts the
value (tpx_f64_csv_n / oracle _f64_fits_device)."""
import random
import struct

from oracle.pyoracle_csv import _f64_fits_device


def fmt_exact(v):
    bits = struct.unpack("<Q", struct.pack("<d", v))[0]
    neg = bits >> 63
    exp = (bits >> 52) & 0x7FF
    man = bits & ((1 << 52) - 1)
    if exp == 0x7FF:
        return None
    if exp == 0:
        m, e = man, -1074
    else:
        m, e = man | (1 << 52), exp - 1075
    M = m * 15625
    k = e + 6
    if k >= 0:
        if k >= 62 or (M << k) >> 63:
            return None
        N = M << k
    else:
        k = -k
        if k >= 69:
            N = 0
        else:
            N = M >> k
            rem = M & ((1 << k) - 1)
            half = 1 << (k - 1)
            if rem > half or (rem == half and (N & 1)):
                N += 1
            if N >> 63:
                return None
    s = str(N).rjust(7, "0")
    return ("-" if neg else "") + s[:-6] + "." + s[-6:]


def test_fmt_exact_matches_cpython():
    rng = random.Random(3)
    checked = 0
    for _ in range(20000):
        c = rng.random()
        if c < 0.3:
            v = rng.uniform(-1e13, 1e13)
        elif c < 0.6:
            v = rng.uniform(-1000, 1000)
        elif c < 0.8:
            v = rng.uniform(-1e-5, 1e-5)
        else:
            v = struct.unpack("<d",
                              struct.pack("<Q", rng.getrandbits(64)))[0]
        got = fmt_exact(v)
        assert (got is not None) == _f64_fits_device(v), repr(v)
        if got is None:
            continue
        assert got == "%f" % v, repr(v)
        checked += 1
    assert checked > 10000


def test_fmt_exact_edges():
    for v in [0.0, -0.0, 5e-7, 1.5e-6, 5e-324, -5e-324, 9.2e12, 1e-7,
              123456789.123456789]:
        assert fmt_exact(v) == "%f" % v
    for v in [float("nan"), float("inf"), 1e300, 1.9e13]:
        assert not _f64_fits_device(v)
