"""Tungsten-style row (de)serialization — the reference's partition byte layout.

Reference: tuplex/utils/src/Serializer.cc:20-24 (layout note), :29 calcBitmapSize
(bitmap is a multiple of 64 bits covering OPTIONAL fields only, in field order),
:265 appendWithoutInference(string) (strings stored with trailing NUL; size includes
it), :1016 serialize (bitmap | fixed slots | varlen-total | varlen bytes; the
varlen-total word is present iff the SCHEMA has varlen-typed fields, even when every
value is null), :1097 (varlen slot word = offset | size<<32, offset measured from the
slot's own address). Partition layout: [int64 numRows][rows...] (Partition.h:38).

Round-1 schema subset: flat rows of i64/f64/bool/str and Options thereof (nested
tuples and lists go down the fallback path).
"""
import struct
from typing import Any, List, Sequence, Tuple

from . import ttypes as T


def _field_info(row_type):
    ps = T.tuple_params(row_type)
    optional = [T.is_opt(p) for p in ps]
    varlen = [T.is_varlen(p) for p in ps]
    n_opt = sum(optional)
    bitmap_size = ((n_opt + 63) // 64) * 8 if n_opt else 0
    has_var = any(varlen)
    return ps, optional, varlen, bitmap_size, has_var


def serialized_length(row: Sequence[Any], row_type) -> int:
    ps, optional, varlen, bitmap_size, has_var = _field_info(row_type)
    size = bitmap_size + 8 * len(ps) + (8 if has_var else 0)
    for v, p in zip(row, ps):
        if T.is_varlen(p) and v is not None:
            size += len(v.encode("utf-8")) + 1
    return size


def serialize_row(row: Sequence[Any], row_type) -> bytes:
    ps, optional, varlen, bitmap_size, has_var = _field_info(row_type)
    assert len(row) == len(ps), "row arity mismatch"
    bitmap = 0
    opt_counter = 0
    slots = []
    varbuf = bytearray()
    var_positions = []  # (slot_index, var_offset_in_buf, size)
    for i, (v, p) in enumerate(zip(row, ps)):
        if optional[i]:
            if v is None:
                bitmap |= 1 << opt_counter
            opt_counter += 1
        base = T.deopt(p)
        if v is None:
            slots.append(0)
        elif base == T.I64:
            slots.append(int(v) & 0xFFFFFFFFFFFFFFFF)
        elif base == T.BOOL:
            slots.append(1 if v else 0)
        elif base == T.F64:
            slots.append(struct.unpack("<Q", struct.pack("<d", float(v)))[0])
        elif base == T.STR:
            b = v.encode("utf-8") + b"\x00"
            var_positions.append((i, len(varbuf), len(b)))
            varbuf += b
            slots.append(0)  # patched below
        else:
            raise TypeError("unsupported field type %r" % (p,))
    # patch varlen info words: offset from the slot's own address to the data
    # (Serializer.cc:1080-1105: data begins at fixed_end+8; slot i sits at i*8 within
    # the post-bitmap region)
    fixed_len = 8 * len(ps)
    for slot_i, var_off, sz in var_positions:
        offset = (fixed_len + 8 + var_off) - 8 * slot_i
        slots[slot_i] = (offset | (sz << 32)) & 0xFFFFFFFFFFFFFFFF
    out = bytearray()
    if bitmap_size:
        out += bitmap.to_bytes(bitmap_size, "little")
    out += struct.pack("<%dQ" % len(slots), *slots)
    if has_var:
        out += struct.pack("<q", len(varbuf))
        out += varbuf
    return bytes(out)


def deserialize_row(buf: bytes, pos: int, row_type) -> Tuple[tuple, int]:
    """Returns (row values, bytes consumed)."""
    ps, optional, varlen, bitmap_size, has_var = _field_info(row_type)
    start = pos
    bitmap = int.from_bytes(buf[pos:pos + bitmap_size], "little") if bitmap_size else 0
    pos += bitmap_size
    vals = []
    opt_counter = 0
    var_total = 0
    fixed_base = pos
    for i, p in enumerate(ps):
        slot = struct.unpack_from("<Q", buf, pos)[0]
        is_null = False
        if optional[i]:
            is_null = bool((bitmap >> opt_counter) & 1)
            opt_counter += 1
        base = T.deopt(p)
        if is_null:
            vals.append(None)
        elif base == T.I64:
            v = slot if slot < 2**63 else slot - 2**64
            vals.append(v)
        elif base == T.BOOL:
            vals.append(bool(slot))
        elif base == T.F64:
            vals.append(struct.unpack("<d", struct.pack("<Q", slot))[0])
        elif base == T.STR:
            offset = slot & 0xFFFFFFFF
            sz = slot >> 32
            sptr = pos + offset
            vals.append(buf[sptr:sptr + sz - 1].decode("utf-8"))
        else:
            raise TypeError("unsupported field type %r" % (p,))
        pos += 8
    if has_var:
        var_total = struct.unpack_from("<q", buf, pos)[0]
        pos += 8 + var_total
    return tuple(vals), pos - start


def serialize_partition(rows: List[Sequence[Any]], row_type) -> Tuple[bytes, List[int]]:
    """[int64 numRows][rows...]; returns (bytes, row_offsets[n+1] relative to start)."""
    out = bytearray(struct.pack("<q", len(rows)))
    offsets = [len(out)]
    for r in rows:
        out += serialize_row(r, row_type)
        offsets.append(len(out))
    return bytes(out), offsets


def deserialize_partition(buf: bytes, row_type, offsets=None) -> List[tuple]:
    """`offsets`: optional per-row byte offsets into buf (n(+1) entries,
    absolute, first == 8) — the GPU write kernel returns them
    (tpx_result.out_row_offsets), enabling the vectorized decoder."""
    n = struct.unpack_from("<q", buf, 0)[0]
    if n >= 256:
        rows = _deserialize_fast(buf, row_type, n, offsets)
        if rows is not None:
            return rows
    pos = 8
    rows = []
    for _ in range(n):
        row, consumed = deserialize_row(buf, pos, row_type)
        rows.append(row)
        pos += consumed
    return rows


def _deserialize_fast(buf, row_type, n, offsets):
    """numpy-vectorized partition decode (the pure-python row loop costs
    ~10 us/row and dominates collect() whole-job time). Falls back (None)
    for layouts outside the flat-row subset."""
    import numpy as np
    ps, optional, varlen, bitmap_size, has_var = _field_info(row_type)
    if bitmap_size > 8:
        return None  # >64 optional fields: scalar path
    arr = np.frombuffer(buf, dtype=np.uint8)
    if offsets is not None:
        starts = np.asarray(offsets[:n], dtype=np.int64)
    else:
        # row stride varies only via var_total: one cheap scan
        fixed = bitmap_size + 8 * len(ps) + (8 if has_var else 0)
        starts = np.empty(n, dtype=np.int64)
        pos = 8
        if has_var:
            vt_off = bitmap_size + 8 * len(ps)
            u = struct.unpack_from
            for k in range(n):
                starts[k] = pos
                pos += fixed + u("<q", buf, pos + vt_off)[0]
        else:
            starts = 8 + fixed * np.arange(n, dtype=np.int64)

    def gather_u64(idx):
        b = arr[idx[:, None] + np.arange(8)]
        return np.ascontiguousarray(b).view(np.uint64).ravel()

    bm = gather_u64(starts) & ((1 << (8 * bitmap_size)) - 1) \
        if bitmap_size else None
    slotbase = starts + bitmap_size
    cols = []
    opt_counter = 0
    for i, p in enumerate(ps):
        idx = slotbase + 8 * i
        slots = gather_u64(idx)
        base = T.deopt(p)
        if base == T.I64:
            col = slots.view(np.int64).tolist()
        elif base == T.BOOL:
            col = (slots != 0).tolist()
        elif base == T.F64:
            col = slots.view(np.float64).tolist()
        elif base == T.STR:
            off = (slots & 0xFFFFFFFF).astype(np.int64)
            sz = (slots >> np.uint64(32)).astype(np.int64)
            sp = (idx + off).tolist()
            ep = (idx + off + sz - 1).tolist()
            col = [buf[s:e].decode("utf-8") for s, e in zip(sp, ep)]
        else:
            return None
        if optional[i]:
            nulls = (bm >> np.uint64(opt_counter)) & np.uint64(1)
            opt_counter += 1
            if nulls.any():
                nl = nulls.tolist()
                col = [None if z else v for z, v in zip(nl, col)]
        cols.append(col)
    if not cols:
        return [()] * n
    return list(zip(*cols))
