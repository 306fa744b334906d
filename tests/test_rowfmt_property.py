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
