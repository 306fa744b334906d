"""Property round trip for the reference row format (Serializer.cc layout):
serialize_partition -> deserialize_partition is the identity for arbitrary
typed rows (bitmaps, 8B slots, varlen offsets, NUL-terminated strings)."""
import hypothesis.strategies as st
from hypothesis import given, settings

from tuplex_amd import rowfmt
from tuplex_amd import ttypes as T

ascii_str = st.text(alphabet=st.characters(min_codepoint=32,
                                           max_codepoint=126), max_size=30)


def value_for(t):
    base = T.deopt(t)
    if base == T.I64:
        v = st.integers(min_value=-2**63, max_value=2**63 - 1)
    elif base == T.F64:
        v = st.floats(allow_nan=False, width=64)
    elif base == T.BOOL:
        v = st.booleans()
    else:
        v = ascii_str
    if T.is_opt(t):
        return st.one_of(st.none(), v)
    return v


types_strategy = st.lists(
    st.sampled_from([T.I64, T.F64, T.BOOL, T.STR, ("opt", T.I64),
                     ("opt", T.STR), ("opt", T.F64), ("opt", T.BOOL)]),
    min_size=1, max_size=6)


@settings(max_examples=200, deadline=None)
@given(st.data())
def test_rowfmt_roundtrip(data):
    types = data.draw(types_strategy)
    rows = data.draw(st.lists(
        st.tuples(*[value_for(t) for t in types]), min_size=0, max_size=25))
    rt = T.tup(types)
    buf, offs = rowfmt.serialize_partition(rows, rt)
    back = rowfmt.deserialize_partition(bytes(buf), rt)
    assert back == rows
    assert len(offs) == len(rows) + 1


def test_fast_deserializer_matches_scalar_random():
    """_deserialize_fast must agree with the scalar row loop on random
    schemas/rows (incl. optionals, empty/unicode strings, extreme ints) both
    with and without explicit row offsets."""
    import random
    from tuplex_amd import rowfmt
    from tuplex_amd import ttypes as T
    rng = random.Random(11)
    bases = [T.I64, T.F64, T.BOOL, T.STR]
    for trial in range(12):
        nf = rng.randint(1, 6)
        ps = []
        for _ in range(nf):
            b = rng.choice(bases)
            ps.append(T.opt(b) if rng.random() < 0.4 else b)
        rt = T.tup(ps)
        rows = []
        for i in range(300):
            row = []
            for p in ps:
                if T.is_opt(p) and rng.random() < 0.25:
                    row.append(None)
                    continue
                b = T.deopt(p)
                if b == T.I64:
                    row.append(rng.randint(-2**63, 2**63 - 1))
                elif b == T.F64:
                    row.append(rng.choice([0.0, -0.0, 1e300, -1.5,
                                           float("inf"), 3.14159]))
                elif b == T.BOOL:
                    row.append(bool(rng.getrandbits(1)))
                else:
                    row.append(rng.choice(["", "a", "héllo", "x" * 50,
                                           "q,\"\n"]))
            rows.append(tuple(row))
        buf, offs = rowfmt.serialize_partition(rows, rt)
        scalar = [rowfmt.deserialize_row(buf, o, rt)[0] for o in offs[:-1]]
        fast1 = rowfmt._deserialize_fast(buf, rt, len(rows), offs)
        fast2 = rowfmt._deserialize_fast(buf, rt, len(rows), None)
        assert fast1 is not None and fast2 is not None
        assert list(map(tuple, fast1)) == scalar, trial
        assert list(map(tuple, fast2)) == scalar, trial
