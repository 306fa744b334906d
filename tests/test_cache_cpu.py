"""cache() semantics (CacheOperator.cc analog): exceptions are STORED at the
cache point and resolved by downstream resolvers — a cached pipeline must be
indistinguishable from the same pipeline without cache(). CPU-runnable
(interpreter paths)."""
import os

import pytest

import tuplex_amd


def _div(x):
    # sorted() keeps the stage on the interpreter (CPU-runnable test)
    return (sorted([x[0], x[0]])[0], 100 // x[1])


def _res(x):
    return (x[0], -1)


def _tri(x):
    return sorted([x * 3, x * 3])[0]


def _evenkeep(x):
    return sorted([x, x])[0] % 2 == 0


def _oddkeep(x):
    return sorted([x[1], x[1]])[0] != 3


def _sortkey(x):
    # sorted() -> outside compiled vocabulary -> interpreter path on CPU
    return (sorted([x["a"], x["a"]])[0], x["b"])


DATA = [(1, 2), (2, 0), (3, 5), (4, 0), (5, 1)]


def _ident(x):
    return sorted([x[0], x[0]])[0], x[1]


def test_cache_defers_exceptions_mem():
    ctx = tuplex_amd.Context()
    plain = ctx.parallelize(DATA).map(_div).resolve(ZeroDivisionError, _res)
    want = plain.collect()

    cached = (ctx.parallelize(DATA).map(_div).cache()
              .resolve(ZeroDivisionError, _res))
    got = cached.collect()
    assert got == want
    assert cached._last_outcome.exception_counts == \
        plain._last_outcome.exception_counts


def test_cache_unresolved_exceptions_still_counted_mem():
    ctx = tuplex_amd.Context()
    plain = ctx.parallelize(DATA).map(_div)
    want = plain.collect()
    cached = ctx.parallelize(DATA).map(_div).cache()
    got_ds = cached.map(_ident)
    got = got_ds.collect()
    assert got == want
    assert got_ds._last_outcome.exception_counts == \
        plain._last_outcome.exception_counts == {"ZeroDivisionError": 2}
    # the cache point itself reported nothing resolved/raised yet
    assert cached._last_outcome.exception_counts == {}


def test_cache_post_filter_and_map():
    ctx = tuplex_amd.Context()
    plain = (ctx.parallelize(list(range(20))).map(_tri)
             .filter(_evenkeep))
    want = plain.collect()
    cached = (ctx.parallelize(list(range(20))).map(_tri).cache()
              .filter(_evenkeep))
    assert cached.collect() == want


def test_cache_csv_fallback_defers_exceptions(tmp_path):
    p = os.path.join(str(tmp_path), "in.csv")
    with open(p, "w") as f:
        f.write("a,b\n")
        for i in range(50):
            f.write("%d,%d\n" % (i, i % 7))
        f.write("notanint,5\n")
    ctx = tuplex_amd.Context()
    plain = ctx.csv(p).map(_sortkey).filter(_oddkeep)
    want = plain.collect()
    wantc = plain._last_outcome.exception_counts

    cached = ctx.csv(p).map(_sortkey).cache().filter(_oddkeep)
    got = cached.collect()
    assert got == want
    assert cached._last_outcome.exception_counts == wantc


def test_cache_of_cache():
    ctx = tuplex_amd.Context()
    plain = ctx.parallelize(DATA).map(_div).resolve(ZeroDivisionError, _res)
    want = plain.collect()
    c2 = (ctx.parallelize(DATA).map(_div).cache().cache()
          .resolve(ZeroDivisionError, _res))
    assert c2.collect() == want


def _trim1(x):
    return sorted([x, x])[0]


def _sagg(a, x):
    return a + sorted([x, x])[0]


def test_cache_trailing_aggregate_materializes():
    ctx = tuplex_amd.Context()
    ds = (ctx.parallelize([1, 2, 3, 4]).map(_trim1)
          .aggregate(lambda a, b: a + b, _sagg, 0).cache())
    assert ds.collect() == [10]


def _gdiv(x):
    return (x[0], 100 // x[1])


def _gres(x):
    return (x[0], -1)


@pytest.mark.gpu
def test_cache_defers_exceptions_gpu():
    """Compiled-path cache: GPU stage before AND after the cache point;
    stored exceptions resolved by a post-cache resolver."""
    ctx = tuplex_amd.Context()
    data = [(i, i % 5) for i in range(2000)]
    plain = ctx.parallelize(data).map(_gdiv).resolve(ZeroDivisionError, _gres)
    want = plain.collect()
    cached = (ctx.parallelize(data).map(_gdiv).cache()
              .resolve(ZeroDivisionError, _gres))
    got = cached.collect()
    assert got == want
    assert cached._last_outcome.exception_counts == \
        plain._last_outcome.exception_counts


@pytest.mark.gpu
def test_cache_gpu_post_stage():
    ctx = tuplex_amd.Context()
    data = [(i, "s%d" % i) for i in range(5000)]
    plain = (ctx.parallelize(data).map(lambda x: (x[0] * 2, x[1]))
             .filter(lambda x: x[0] % 3 == 0))
    want = plain.collect()
    cached = (ctx.parallelize(data).map(lambda x: (x[0] * 2, x[1])).cache()
              .filter(lambda x: x[0] % 3 == 0))
    got = cached.collect()
    assert got == want


def test_cache_tocsv_matches_uncached(tmp_path):
    import os
    ctx = tuplex_amd.Context()
    p1 = os.path.join(str(tmp_path), "a.csv")
    p2 = os.path.join(str(tmp_path), "b.csv")
    plain = ctx.parallelize(DATA).map(_div).resolve(ZeroDivisionError, _res)
    plain.tocsv(p1)
    (ctx.parallelize(DATA).map(_div).cache()
     .resolve(ZeroDivisionError, _res).tocsv(p2))
    assert open(p1, "rb").read() == open(p2, "rb").read()
