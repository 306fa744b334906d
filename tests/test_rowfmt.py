"""Row-format tests: product serializer (tuplex_amd/rowfmt.py) against the byte
layouts pinned from the reference's SerializerTest (tests/golden citations)."""
import json
import os
import struct

from tuplex_amd import rowfmt
from tuplex_amd import ttypes as T

HERE = os.path.dirname(os.path.abspath(__file__))
GOLDEN = json.load(open(os.path.join(HERE, "golden", "reference_goldens.json")))["cases"]


def _parse_type(x):
    if isinstance(x, str):
        return x
    if x[0] == "opt":
        return ("opt", _parse_type(x[1]))
    raise ValueError(x)


def test_tuple_with_options_layout():
    """SerializerTest.cc:80 TupleWithOptions: bitmap word == 0x02, word 2 == 42."""
    c = GOLDEN["serializer_tuple_with_options"]
    rt = T.tup([_parse_type(t) for t in c["row_type"]])
    buf = rowfmt.serialize_row(c["row"], rt)
    words = struct.unpack_from("<3Q", buf, 0)
    assert words[0] == c["expect_word0"]
    assert words[2] == c["expect_word2"]
    # round trip
    row, n = rowfmt.deserialize_row(buf, 0, rt)
    assert row == ("test", 42, None)
    assert n == len(buf)


def test_invariance_round_trip():
    c = GOLDEN["serializer_invariance_I"]
    rt = T.tup([_parse_type(t) for t in c["row_type"]])
    row = tuple(c["row"])
    buf = rowfmt.serialize_row(row, rt)
    got, n = rowfmt.deserialize_row(buf, 0, rt)
    assert got == row
    assert n == len(buf)


def test_string_nul_terminated_and_info_word():
    """Serializer.cc:265: strings stored with trailing NUL, size includes it;
    :1097: info word = offset|size<<32, offset from the slot's own address."""
    rt = T.tup([T.STR])
    buf = rowfmt.serialize_row(("hi",), rt)
    # layout: [slot][varlen_total][bytes 'h''i''\0']
    slot = struct.unpack_from("<Q", buf, 0)[0]
    offset, size = slot & 0xFFFFFFFF, slot >> 32
    assert size == 3
    assert offset == 16  # slot at 0, data at 8(slot)+8(varlen word) = 16
    assert struct.unpack_from("<q", buf, 8)[0] == 3  # varlen total
    assert buf[16:19] == b"hi\x00"


def test_varlen_word_present_when_all_null():
    """Serializer.cc:1061 note: option<str> all-NULL rows still carry the varlen
    total word (schema-varlen, not value-varlen)."""
    rt = T.tup([("opt", T.STR)])
    buf = rowfmt.serialize_row((None,), rt)
    # [bitmap][slot][varlen_total=0]
    assert len(buf) == 24
    assert struct.unpack_from("<Q", buf, 0)[0] == 1  # null bit set
    assert struct.unpack_from("<q", buf, 16)[0] == 0


def test_empty_string():
    """SerializerTest.cc:34 EmptyString round trip."""
    rt = T.tup([T.STR])
    buf = rowfmt.serialize_row(("",), rt)
    row, _ = rowfmt.deserialize_row(buf, 0, rt)
    assert row == ("",)


def test_partition_round_trip():
    rt = T.tup([T.I64, ("opt", T.STR), T.F64, T.BOOL])
    rows = [(1, "a", 1.5, True), (2, None, -2.25, False), (3, "xyz", 0.0, True)]
    buf, offs = rowfmt.serialize_partition(rows, rt)
    assert struct.unpack_from("<q", buf, 0)[0] == 3
    assert offs[0] == 8 and offs[-1] == len(buf)
    assert rowfmt.deserialize_partition(buf, rt) == rows


def test_negative_and_extreme_ints():
    rt = T.tup([T.I64])
    for v in (0, -1, 2**63 - 1, -(2**63), 42):
        buf = rowfmt.serialize_row((v,), rt)
        assert rowfmt.deserialize_row(buf, 0, rt)[0] == (v,)
