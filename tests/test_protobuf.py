"""Protobuf decode tests with a hand-written Python encoder as oracle."""
import random

import pytest

from spark_rapids_jni_amd.columnar import Column, DType

random.seed(37)


def _varint(v):
    out = b""
    v &= (1 << 64) - 1
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out += bytes([b | 0x80])
        else:
            out += bytes([b])
            return out


def _tag(fnum, wire):
    return _varint((fnum << 3) | wire)


def encode(fields):
    """fields: list of (fnum, kind, value)"""
    import struct as st
    out = b""
    for fnum, kind, v in fields:
        if kind in ("int64", "int32", "bool"):
            out += _tag(fnum, 0) + _varint(int(v))
        elif kind == "sint64":
            zz = (v << 1) ^ (v >> 63)
            out += _tag(fnum, 0) + _varint(zz)
        elif kind == "double":
            out += _tag(fnum, 1) + st.pack("<d", v)
        elif kind == "float":
            out += _tag(fnum, 5) + st.pack("<f", v)
        elif kind in ("string", "bytes"):
            b = v.encode() if isinstance(v, str) else v
            out += _tag(fnum, 2) + _varint(len(b)) + b
    return out


@pytest.mark.gpu
def test_protobuf_decode():
    from spark_rapids_jni_amd.ops import protobuf
    msgs = [
        encode([(1, "int64", 42), (2, "string", "hello"), (3, "double", 1.5)]),
        encode([(2, "string", ""), (1, "int64", -1)]),
        encode([(3, "double", -2.25), (9, "int64", 99)]),  # 9: unknown field
        b"",
        None,
        encode([(1, "int64", 7), (1, "int64", 8)]),  # last-one-wins
        encode([(4, "sint64", -5), (5, "bool", 1), (6, "float", 0.5)]),
    ]
    col = Column.from_pylist(msgs, DType.STRING, "cuda")
    t = protobuf.decode(col, [(1, "int64"), (2, "string"), (3, "double"),
                              (4, "sint64"), (5, "bool"), (6, "float")])
    assert t.columns[0].to_pylist() == [42, -1, None, None, None, 8, None]
    assert t.columns[1].to_pylist() == ["hello", "", None, None, None, None,
                                        None]
    assert t.columns[2].to_pylist() == [1.5, None, -2.25, None, None, None,
                                        None]
    assert t.columns[3].to_pylist() == [None, None, None, None, None, None, -5]
    assert t.columns[4].to_pylist() == [None] * 6 + [True]
    assert t.columns[5].to_pylist() == [None] * 6 + [0.5]


@pytest.mark.gpu
def test_protobuf_malformed():
    from spark_rapids_jni_amd.ops import protobuf
    msgs = [encode([(1, "int64", 5)]), b"\xff\xff\xff", b"\x0a\xff"]
    col = Column.from_pylist(msgs, DType.STRING, "cuda")
    t = protobuf.decode(col, [(1, "int64")])
    assert t.columns[0].to_pylist() == [5, None, None]


@pytest.mark.gpu
def test_protobuf_nested_messages():
    """Nested message fields decode recursively into STRUCT columns."""
    def inner_msg(a, s):
        return encode([(1, "int64", a), (2, "string", s)])

    rows = []
    exp = []
    for i in range(500):
        if i % 11 == 3:
            rows.append(encode([(3, "int64", i)]))  # nested field absent
            exp.append((i, None))
        else:
            sub = inner_msg(i * 7, f"n{i}")
            rows.append(encode([(3, "int64", i)]) +
                        _tag(4, 2) + _varint(len(sub)) + sub)
            exp.append((i, (i * 7, f"n{i}")))
    col = Column.from_pylist([bytes(r) for r in rows], DType.STRING, "cuda")
    from spark_rapids_jni_amd.ops.protobuf import decode
    tbl = decode(col, [(3, "int64"),
                       (4, ("message", [(1, "int64"), (2, "string")]))])
    top = tbl.columns[0].to_pylist()
    nested = tbl.columns[1]
    assert nested.dtype == DType.STRUCT
    got_a = nested.children[0].to_pylist()
    got_s = nested.children[1].to_pylist()
    for i, (t, sub) in enumerate(exp):
        assert top[i] == t
        if sub is None:
            assert nested.is_valid_host(i) is False or \
                (got_a[i] is None and got_s[i] is None)
        else:
            assert (got_a[i], got_s[i]) == sub


@pytest.mark.gpu
def test_protobuf_repeated_int64():
    """repeated int64 fields (both unpacked tags and packed blobs) decode
    to LIST<INT64>."""
    rows = []
    exp = []
    for i in range(400):
        vals = list(range(i % 6))
        body = b""
        if i % 3 == 0:
            # packed: one length-delimited blob of varints
            blob = b"".join(_varint(v * 3) for v in vals)
            body += _tag(5, 2) + _varint(len(blob)) + blob
        else:
            for v in vals:
                body += _tag(5, 0) + _varint(v * 3)
        body += _tag(1, 0) + _varint(i)
        rows.append(body)
        exp.append([v * 3 for v in vals] if (vals or i % 3 == 0 and i % 6 == 0)
                   else ([v * 3 for v in vals] if vals else None))
    col = Column.from_pylist([bytes(r) for r in rows], DType.STRING, "cuda")
    from spark_rapids_jni_amd.ops.protobuf import decode
    tbl = decode(col, [(1, "int64"), (5, "repeated_int64")])
    assert tbl.columns[0].to_pylist() == list(range(400))
    lst = tbl.columns[1]
    got = lst.to_pylist()
    for i in range(400):
        vals = [v * 3 for v in range(i % 6)]
        if i % 3 == 0:
            # packed blob always present (possibly empty)
            assert got[i] == vals, i
        elif vals:
            assert got[i] == vals, i
        else:
            assert got[i] is None, i


@pytest.mark.gpu
def test_protobuf_repeated_all_types():
    """Reference protobuf_kernels.cuh:150-361 batched variants: repeated
    int32/bool/sint64/double/float decode to LISTs (packed and unpacked),
    repeated string to LIST<STRING>."""
    import struct as st
    from spark_rapids_jni_amd.ops import protobuf as pb

    def enc_row(i):
        out = b""
        # f1 repeated int32: unpacked for even rows, packed for odd
        vals32 = [i, i * 7 + 1, -i][: (i % 4)]
        if i % 2 == 0:
            for v in vals32:
                out += _tag(1, 0) + _varint(v & (2**64 - 1))
        else:
            blob = b"".join(_varint(v & (2**64 - 1)) for v in vals32)
            if blob:
                out += _tag(1, 2) + _varint(len(blob)) + blob
        # f2 repeated bool (packed)
        bools = [bool((i >> k) & 1) for k in range(i % 3)]
        blob = b"".join(_varint(int(b)) for b in bools)
        if blob:
            out += _tag(2, 2) + _varint(len(blob)) + blob
        # f3 repeated sint64 (unpacked, zigzag)
        sints = [-(i * 3), i * 5][: (i % 3)]
        for v in sints:
            out += _tag(3, 0) + _varint(((v << 1) ^ (v >> 63)) & (2**64 - 1))
        # f4 repeated double: packed even rows, unpacked odd
        dbls = [i * 1.5, -i / 3.0][: (i % 3)]
        if i % 2 == 0:
            blob = b"".join(st.pack("<d", d) for d in dbls)
            if blob:
                out += _tag(4, 2) + _varint(len(blob)) + blob
        else:
            for d in dbls:
                out += _tag(4, 1) + st.pack("<d", d)
        # f5 repeated float (packed)
        flts = [float(i), float(i) * 0.5, 9.25][: (i % 4)]
        blob = b"".join(st.pack("<f", f) for f in flts)
        if blob:
            out += _tag(5, 2) + _varint(len(blob)) + blob
        # f6 repeated string
        strs = [f"s{i}", "", f"xx{i*3}"][: (i % 4)]
        for s in strs:
            b = s.encode()
            out += _tag(6, 2) + _varint(len(b)) + b
        return out, (vals32, bools, sints, dbls, flts, strs)

    n = 500
    rows = [enc_row(i) for i in range(n)]
    col = Column.from_pylist([r[0] for r in rows], DType.STRING, "cuda")
    tbl = pb.decode(col, [(1, "repeated_int32"), (2, "repeated_bool"),
                          (3, "repeated_sint64"), (4, "repeated_double"),
                          (5, "repeated_float"), (6, "repeated_string")])
    got = [c.to_pylist() for c in tbl.columns]
    for i in range(n):
        vals32, bools, sints, dbls, flts, strs = rows[i][1]
        assert (got[0][i] or []) == (vals32 if vals32 else []) or \
            (got[0][i] is None and not vals32), (i, got[0][i], vals32)
        if vals32:
            assert got[0][i] == vals32, i
        if bools:
            assert got[1][i] == bools, i
        if sints:
            assert got[2][i] == sints, i
        if dbls:
            assert len(got[3][i]) == len(dbls) and all(
                abs(a - b) < 1e-12 for a, b in zip(got[3][i], dbls)), i
        if flts:
            assert len(got[4][i]) == len(flts), i
        if strs:
            assert got[5][i] == strs, i


def test_duplicate_field_numbers_rejected():
    """Reference ProtobufSchemaDescriptorTest: duplicate field numbers under
    ONE parent are invalid; the same number under different parents is
    fine (validated per message level, no GPU needed to reject)."""
    import pytest as _pt
    from spark_rapids_jni_amd.ops import protobuf as pb
    from spark_rapids_jni_amd.columnar import Column, DType
    col = Column.from_pylist([b""], DType.STRING)
    with _pt.raises(ValueError, match="duplicate"):
        pb.decode(col, [(1, "int64"), (1, "string")])
    with _pt.raises(ValueError, match="duplicate"):
        pb.decode(col, [(1, "int64"),
                        (2, ("message", [(3, "int64"), (3, "bool")]))])
