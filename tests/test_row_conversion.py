"""JCUDF row conversion tests: layout oracle + GPU roundtrip."""
import random
import struct

import pytest
import torch

from spark_rapids_jni_amd.columnar import Column, DType, Table
from spark_rapids_jni_amd.ops.row_conversion import row_layout

random.seed(31)


def test_row_layout_example():
    # example from RowConversion.java:77-85: BOOL8, INT16, INT32
    offs, validity_off, row_size = row_layout(
        [DType.BOOL8, DType.INT16, DType.INT32])
    assert offs == [0, 2, 4]
    assert validity_off == 8
    assert row_size == 16
    # ordered large-to-small: 4,2,1 -> no padding inside
    offs2, voff2, rs2 = row_layout([DType.INT32, DType.INT16, DType.BOOL8])
    assert offs2 == [0, 4, 6]
    assert voff2 == 7
    assert rs2 == 8


def _expected_rows(colvals, dtypes):
    offs, voff, rs = row_layout(dtypes)
    n = len(colvals[0])
    out = bytearray(n * rs)
    fmt = {DType.BOOL8: "<b", DType.INT8: "<b", DType.INT16: "<h",
           DType.INT32: "<i", DType.INT64: "<q", DType.FLOAT32: "<f",
           DType.FLOAT64: "<d", DType.DATE32: "<i", DType.TIMESTAMP_US: "<q"}
    for r in range(n):
        base = r * rs
        for c, (vals, dt) in enumerate(zip(colvals, dtypes)):
            v = vals[r]
            if v is not None:
                x = int(v) if dt == DType.BOOL8 else v
                struct.pack_into(fmt[dt], out, base + offs[c], x)
                out[base + voff + c // 8] |= 1 << (c % 8)
    return bytes(out)


@pytest.mark.gpu
def test_to_rows_matches_oracle():
    from spark_rapids_jni_amd.ops.row_conversion import convert_to_rows
    n = 100
    dtypes = [DType.BOOL8, DType.INT16, DType.INT32, DType.INT64,
              DType.FLOAT64, DType.INT8]
    colvals = []
    for dt in dtypes:
        if dt == DType.BOOL8:
            colvals.append([None if i % 13 == 5 else bool(i % 2)
                            for i in range(n)])
        elif dt in (DType.INT8,):
            colvals.append([None if i % 7 == 1 else (i % 200) - 100
                            for i in range(n)])
        elif dt == DType.INT16:
            colvals.append([random.randint(-30000, 30000) for _ in range(n)])
        elif dt == DType.FLOAT64:
            colvals.append([None if i % 5 == 2 else random.random()
                            for i in range(n)])
        else:
            colvals.append([None if i % 11 == 3 else random.randint(-10**6, 10**6)
                            for i in range(n)])
    cols = [Column.from_pylist(v, dt, "cuda") for v, dt in zip(colvals, dtypes)]
    batches = convert_to_rows(Table(cols))
    assert len(batches) == 1
    got = batches[0][0].cpu().numpy().tobytes()
    assert got == _expected_rows(colvals, dtypes)


@pytest.mark.gpu
def test_row_roundtrip():
    from spark_rapids_jni_amd.ops.row_conversion import (convert_from_rows,
                                                         convert_to_rows)
    n = 1000
    dtypes = [DType.INT64, DType.INT32, DType.FLOAT32, DType.BOOL8]
    colvals = [
        [None if i % 17 == 0 else random.randint(-2**62, 2**62) for i in range(n)],
        [random.randint(-2**31 + 1, 2**31 - 1) for _ in range(n)],
        [None if i % 3 == 1 else float(i) * 0.5 for i in range(n)],
        [bool(i % 2) for i in range(n)],
    ]
    cols = [Column.from_pylist(v, dt, "cuda") for v, dt in zip(colvals, dtypes)]
    batches = convert_to_rows(Table(cols))
    back = convert_from_rows(batches, dtypes)
    for orig, got, vals in zip(cols, back.columns, colvals):
        assert got.to_pylist() == vals


def test_var_row_layout():
    from spark_rapids_jni_amd.ops.row_conversion import var_row_layout
    # INT64, STRING, INT8: string slot is 8B aligned to 4
    offs, voff, fixed = var_row_layout([DType.INT64, DType.STRING, DType.INT8])
    assert offs == [0, 8, 16]
    assert voff == 17
    assert fixed == 24


@pytest.mark.gpu
def test_var_row_roundtrip():
    from spark_rapids_jni_amd.ops.row_conversion import (
        convert_from_rows_varwidth, convert_to_rows_varwidth)
    n = 777
    svals = [None if i % 13 == 4 else ("s%d" % i) * (i % 9) for i in range(n)]
    ivals = [None if i % 7 == 2 else i * 3 for i in range(n)]
    s2vals = ["x" * (i % 4) for i in range(n)]
    dtypes = [DType.STRING, DType.INT64, DType.STRING]
    cols = [Column.from_pylist(svals, DType.STRING, "cuda"),
            Column.from_pylist(ivals, DType.INT64, "cuda"),
            Column.from_pylist(s2vals, DType.STRING, "cuda")]
    buf, row_offs = convert_to_rows_varwidth(Table(cols))
    assert row_offs.numel() == n + 1
    # every row 8-byte aligned
    ro = row_offs.cpu().tolist()
    assert all(o % 8 == 0 for o in ro)
    back = convert_from_rows_varwidth(buf, row_offs, dtypes)
    assert back.columns[0].to_pylist() == svals
    assert back.columns[1].to_pylist() == ivals
    assert back.columns[2].to_pylist() == s2vals


@pytest.mark.gpu
def test_var_row_layout_bytes():
    """Check the on-wire fixed-section layout: (offset,len) pair + chars
    after validity."""
    from spark_rapids_jni_amd.ops.row_conversion import (
        convert_to_rows_varwidth, var_row_layout)
    dtypes = [DType.INT32, DType.STRING]
    cols = [Column.from_pylist([7, 8], DType.INT32, "cuda"),
            Column.from_pylist(["abc", ""], DType.STRING, "cuda")]
    buf, row_offs = convert_to_rows_varwidth(Table(cols))
    offs, voff, fixed = var_row_layout(dtypes)
    raw = buf.cpu().numpy().tobytes()
    ro = row_offs.cpu().tolist()
    r0 = raw[ro[0]:ro[1]]
    i32, spos, slen = struct.unpack_from("<i", r0, offs[0])[0], \
        struct.unpack_from("<ii", r0, offs[1])[0], \
        struct.unpack_from("<ii", r0, offs[1])[1]
    assert i32 == 7 and spos == fixed and slen == 3
    assert r0[spos:spos + 3] == b"abc"
    assert r0[voff] == 0b11
