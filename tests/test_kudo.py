"""Kudo serializer round-trip tests (CPU; reference KudoSerializerTest +
KudoConcatValidityTest sliced-validity cases)."""
import io
import random
import struct

import pytest

from spark_rapids_jni_amd import kudo
from spark_rapids_jni_amd.columnar import Column, DType

random.seed(99)


def _roundtrip(schema_cols, slices):
    """slices: list of (row_offset, num_rows); returns merged pylists."""
    bufs = []
    for off, n in slices:
        out = io.BytesIO()
        kudo.write_partition(schema_cols, off, n, out)
        bufs.append(out.getvalue())
    merged = kudo.merge_on_host(bufs, schema_cols)
    return [c.to_pylist() for c in merged]


def _expected(cols, slices):
    out = []
    for c in cols:
        vals = c.to_pylist()
        out.append([v for off, n in slices for v in vals[off:off + n]])
    return out


def test_header_roundtrip():
    h = kudo.KudoTableHeader(3, 100, 16, 8, 64, 5, b"\x1f")
    out = io.BytesIO()
    h.write(out)
    out.seek(0)
    h2 = kudo.KudoTableHeader.read(out)
    assert h2 == h
    # header is big-endian on the wire
    raw = out.getvalue()
    assert raw[:4] == b"KUD0"
    assert struct.unpack(">i", raw[4:8])[0] == 3


def test_fixed_width_no_nulls():
    c = Column.from_pylist(list(range(100)), DType.INT64)
    got = _roundtrip([c], [(0, 50), (50, 50)])
    assert got[0] == list(range(100))


def test_fixed_width_unaligned_slices_with_nulls():
    vals = [None if i % 5 == 0 else i for i in range(100)]
    c = Column.from_pylist(vals, DType.INT32)
    slices = [(3, 6), (9, 13), (22, 1), (23, 40), (63, 37)]
    got = _roundtrip([c], slices)
    assert got[0] == _expected([c], slices)[0]


def test_multi_dtype_table():
    n = 64
    cols = [
        Column.from_pylist([random.randint(-100, 100) for _ in range(n)],
                           DType.INT32),
        Column.from_pylist([None if i % 7 == 2 else random.random()
                            for i in range(n)], DType.FLOAT64),
        Column.from_pylist([bool(i % 3) for i in range(n)], DType.BOOL8),
        Column.from_pylist([None if i % 11 == 0 else i * 100 - 3000
                            for i in range(n)], DType.INT16),
    ]
    slices = [(0, 10), (10, 17), (27, 37)]
    got = _roundtrip(cols, slices)
    exp = _expected(cols, slices)
    for g, e in zip(got, exp):
        assert g == e


def test_strings_with_nulls():
    vals = [None if i % 4 == 1 else f"s{i}" * (i % 3) for i in range(50)]
    c = Column.from_pylist(vals, DType.STRING)
    slices = [(5, 11), (16, 3), (19, 31)]
    got = _roundtrip([c], slices)
    assert got[0] == _expected([c], slices)[0]


def test_empty_slice():
    c = Column.from_pylist([1, 2, 3], DType.INT64)
    got = _roundtrip([c], [(0, 3), (1, 0), (1, 2)])
    assert got[0] == [1, 2, 3, 2, 3]


def test_struct_column():
    n = 40
    a = Column.from_pylist([None if i % 6 == 0 else i for i in range(n)],
                           DType.INT32)
    b = Column.from_pylist([f"v{i}" if i % 3 else None for i in range(n)],
                           DType.STRING)
    s = Column(DType.STRUCT, n, None, None, None, [a, b])
    slices = [(2, 9), (11, 29)]
    bufs = []
    for off, cnt in slices:
        out = io.BytesIO()
        kudo.write_partition([s], off, cnt, out)
        bufs.append(out.getvalue())
    merged = kudo.merge_on_host(bufs, [s])[0]
    exp_a = [v for off, cnt in slices for v in a.to_pylist()[off:off + cnt]]
    exp_b = [v for off, cnt in slices for v in b.to_pylist()[off:off + cnt]]
    got = merged.to_pylist()
    assert [g[0] for g in got] == exp_a
    assert [g[1] for g in got] == exp_b


def test_list_column():
    n = 30
    child_vals = []
    offs = [0]
    for i in range(n):
        ln = random.randint(0, 4)
        child_vals.extend(random.randint(0, 99) for _ in range(ln))
        offs.append(offs[-1] + ln)
    import torch
    child = Column.from_pylist(child_vals, DType.INT64)
    lst = Column(DType.LIST, n, None, None,
                 torch.tensor(offs, dtype=torch.int32), [child])
    expected = lst.to_pylist()
    slices = [(1, 9), (13, 17)]
    bufs = []
    for off, cnt in slices:
        out = io.BytesIO()
        kudo.write_partition([lst], off, cnt, out)
        bufs.append(out.getvalue())
    merged = kudo.merge_on_host(bufs, [lst])[0]
    exp = [v for off, cnt in slices for v in expected[off:off + cnt]]
    assert merged.to_pylist() == exp


def test_body_alignment():
    # validity part must be padded to 4B counting the header; total body 4B
    c = Column.from_pylist([None, 1, 2], DType.INT8)
    out = io.BytesIO()
    n = kudo.write_partition([c], 0, 3, out)
    raw = out.getvalue()
    assert len(raw) == n
    t = kudo.read_one(io.BytesIO(raw))
    assert (t.header.header_len() + t.header.validity_len) % 4 == 0
    # whole record (header + body) ends 4-byte aligned
    assert (t.header.header_len() + t.header.total_len) % 4 == 0


def test_arrow_interop_roundtrip():
    pa = pytest.importorskip("pyarrow")
    from spark_rapids_jni_amd.columnar import from_arrow, to_arrow
    cases = [
        pa.array([1, None, 3, -7], type=pa.int64()),
        pa.array([1.5, None, -2.25], type=pa.float64()),
        pa.array(["ab", None, "", "xyz"], type=pa.string()),
        pa.array(list(range(300)), type=pa.int32()),
        pa.array([None] * 5, type=pa.int64()),
    ]
    for arr in cases:
        col = from_arrow(arr)
        assert col.to_pylist() == arr.to_pylist()
        back = to_arrow(col)
        assert back.to_pylist() == arr.to_pylist()
        assert back.type == arr.type


def test_kudo_fuzz_roundtrip():
    """Random flat/nested schemas, random slices -> write_partition ->
    merge_on_host must reproduce the sliced rows exactly."""
    import random as rnd
    import torch
    r = rnd.Random(907)

    def rand_col(n, depth=0):
        t = r.choice(["i64", "i32", "str", "f64"] +
                     (["struct", "list"] if depth < 2 else []))
        if t == "i64":
            vals = [None if r.random() < 0.15 else r.randint(-10**9, 10**9)
                    for _ in range(n)]
            return Column.from_pylist(vals, DType.INT64), vals
        if t == "i32":
            vals = [None if r.random() < 0.1 else r.randint(-10**5, 10**5)
                    for _ in range(n)]
            return Column.from_pylist(vals, DType.INT32), vals
        if t == "f64":
            vals = [None if r.random() < 0.1 else r.random() * 100
                    for _ in range(n)]
            return Column.from_pylist(vals, DType.FLOAT64), vals
        if t == "str":
            vals = [None if r.random() < 0.2 else
                    "s" * r.randrange(4) + str(r.randrange(100))
                    for _ in range(n)]
            return Column.from_pylist(vals, DType.STRING), vals
        if t == "struct":
            a, av = rand_col(n, depth + 1)
            b, bv = rand_col(n, depth + 1)
            return (Column(DType.STRUCT, n, None, None, None, [a, b]),
                    list(zip(av, bv)))
        # list of int64
        offs = [0]
        child_vals = []
        for _ in range(n):
            ln = r.randrange(4)
            child_vals.extend(r.randrange(100) for _ in range(ln))
            offs.append(offs[-1] + ln)
        child = Column.from_pylist(child_vals, DType.INT64)
        col = Column(DType.LIST, n, None, None,
                     torch.tensor(offs, dtype=torch.int32), [child])
        return col, col.to_pylist()

    for trial in range(12):
        n = r.randrange(1, 120)
        cols, expected = zip(*[rand_col(n) for _ in range(r.randrange(1, 4))])
        nparts = r.randrange(1, 5)
        bounds = sorted(r.randrange(n + 1) for _ in range(nparts - 1))
        bounds = [0] + bounds + [n]
        bufs = []
        for p in range(nparts):
            out = io.BytesIO()
            kudo.write_partition(list(cols), bounds[p],
                                 bounds[p + 1] - bounds[p], out)
            bufs.append(out.getvalue())
        merged = kudo.merge_on_host(bufs, list(cols))
        for mc, exp, col in zip(merged, expected, cols):
            got = mc.to_pylist()
            want = [exp[i] for p in range(nparts)
                    for i in range(bounds[p], bounds[p + 1])]
            if col.dtype == DType.STRUCT:
                got = [tuple(g) if g is not None else None for g in got]
            assert got == want, f"trial {trial}"


def test_zero_row_slice_at_validity_boundary():
    # ADVICE fix: a 0-row slice whose start byte lies past the end of the
    # validity buffer (64-row nullable column sliced at row 64) must still
    # emit the 1 advertised validity byte, or the reader desyncs.
    vals = [None if i % 3 == 0 else i for i in range(64)]
    c = Column.from_pylist(vals, DType.INT64)
    slices = [(0, 64), (64, 0)]
    got = _roundtrip([c], slices)
    assert got[0] == _expected([c], slices)[0]


def test_zero_row_nullable_column():
    # A 0-row nullable column: validity_nbytes(0)==0 but the wire spec says
    # at least 1 byte travels.
    vals = [None, 1, 2, None]
    c = Column.from_pylist(vals, DType.INT32)
    got = _roundtrip([c], [(2, 0), (0, 4), (4, 0)])
    assert got[0] == _expected([c], [(2, 0), (0, 4), (4, 0)])[0]


def test_list_of_struct_deep_roundtrip():
    """LIST<STRUCT<int, LIST<string>>> host kudo round-trip — the nesting
    shape the parquet reader now produces (nulls at every level)."""
    import torch
    n = 24
    rng = random.Random(17)
    inner_vals, inner_offs = [], [0]
    n_entries = 40
    for i in range(n_entries):
        ln = rng.randint(0, 3)
        inner_vals.extend(f"s{i}_{k}" if (i + k) % 5 else None
                          for k in range(ln))
        inner_offs.append(inner_offs[-1] + ln)
    inner_child = Column.from_pylist(inner_vals, DType.STRING)
    inner = Column(DType.LIST, n_entries, None, None,
                   torch.tensor(inner_offs, dtype=torch.int32),
                   [inner_child])
    ints = Column.from_pylist([None if i % 7 == 3 else i * 11
                               for i in range(n_entries)], DType.INT32)
    entries = Column(DType.STRUCT, n_entries, None, None, None,
                     [ints, inner])
    outer_offs = [0]
    for i in range(n):
        outer_offs.append(min(outer_offs[-1] + rng.randint(0, 3), n_entries))
    col = Column(DType.LIST, n, None, None,
                 torch.tensor(outer_offs, dtype=torch.int32), [entries])
    expected = col.to_pylist()
    slices = [(0, 7), (9, 12)]
    bufs = []
    for off, cnt in slices:
        out = io.BytesIO()
        kudo.write_partition([col], off, cnt, out)
        bufs.append(out.getvalue())
    merged = kudo.merge_on_host(bufs, [col])[0]
    exp = [v for off, cnt in slices for v in expected[off:off + cnt]]
    assert merged.to_pylist() == exp


def test_write_simple_reference_byte_counts():
    """Reference KudoSerializerTest.testWriteSimple: the EXACT byte counts
    and header fields for the 7-column (a,b,c,c1,d,d1,d2) 4-row table —
    172 bytes total, validity 7, offsets 40, body 143, validity bitset
    10111110."""
    import torch
    from spark_rapids_jni_amd.columnar import validity_from_bools
    vb = validity_from_bools([True, True, False, True], "cpu")
    a = Column.from_pylist([1, 2, 3, 4], DType.INT32)
    b = Column.from_pylist(["1", "12", None, "45"], DType.STRING)
    c1 = Column.from_pylist([1, None, 3, 4, 5, 6, 7, 8, 9], DType.INT32)
    c = Column(DType.LIST, 4, None, vb,
               torch.tensor([0, 3, 6, 6, 9], dtype=torch.int32), [c1])
    d1 = Column.from_pylist([1, 2, None, 3], DType.INT8)
    d2 = Column.from_pylist([11, None, None, 33], DType.INT64)
    d = Column(DType.STRUCT, 4, None, vb.clone(), None, [d1, d2])
    out = io.BytesIO()
    kudo.write_partition([a, b, c, d], 0, 4, out)
    buf = out.getvalue()
    assert len(buf) == 172
    hdr = kudo.KudoTableHeader.read(io.BytesIO(buf))
    assert hdr.num_columns == 7
    assert hdr.offset == 0
    assert hdr.num_rows == 4
    assert hdr.validity_len == 7
    assert hdr.offset_len == 40
    assert hdr.total_len == 143
    assert not hdr.has_validity_buffer(0)
    for i in range(1, 7):
        assert hdr.has_validity_buffer(i), i


def test_row_count_only_reference_bytes():
    """Reference testRowCountOnly: a 28-byte header-only batch."""
    out = io.BytesIO()
    n = kudo.write_row_count(out, 5)
    assert n == 28 and len(out.getvalue()) == 28
    hdr = kudo.KudoTableHeader.read(io.BytesIO(out.getvalue()))
    assert (hdr.num_columns, hdr.offset, hdr.num_rows, hdr.validity_len,
            hdr.offset_len, hdr.total_len) == (0, 0, 5, 0, 0, 0)


def test_serialize_validity_unaligned_tail_slice():
    """Reference testSerializeValidity: a 3-row slice at offset 509 of a
    512-row nullable column — validity bits cross byte boundaries at an
    unaligned offset and must merge back exactly."""
    col = Column.from_pylist([None, None] + list(range(2, 512)), DType.INT32)
    out = io.BytesIO()
    kudo.write_partition([col], 509, 3, out)
    merged = kudo.merge_on_host([out.getvalue()], [col])[0]
    assert merged.to_pylist() == [509, 510, 511]


def test_merge_fuzz_random_slices():
    """Randomized merge fuzz (covers the KudoConcatValidityTest ground):
    arbitrary unaligned slices of nullable int/string/list columns must
    reassemble exactly, across many random slicings."""
    import torch
    rng = random.Random(7863832)
    n = 200
    a = Column.from_pylist([None if rng.random() < 0.3 else i
                            for i in range(n)], DType.INT32)
    b = Column.from_pylist([None if rng.random() < 0.2 else f"s{i % 17}"
                            for i in range(n)], DType.STRING)
    offs = [0]
    child_vals = []
    for i in range(n):
        ln = rng.randint(0, 3)
        child_vals.extend(rng.randint(0, 9) for _ in range(ln))
        offs.append(offs[-1] + ln)
    c = Column(DType.LIST, n, None, None,
               torch.tensor(offs, dtype=torch.int32),
               [Column.from_pylist(child_vals, DType.INT64)])
    cols = [a, b, c]
    expected = [col.to_pylist() for col in cols]
    for trial in range(25):
        cuts = sorted(rng.sample(range(1, n), rng.randint(1, 6)))
        bounds = [0] + cuts + [n]
        bufs = []
        for s0, s1 in zip(bounds, bounds[1:]):
            out = io.BytesIO()
            kudo.write_partition(cols, s0, s1 - s0, out)
            bufs.append(out.getvalue())
        merged = kudo.merge_on_host(bufs, cols)
        for col, exp in zip(merged, expected):
            assert col.to_pylist() == exp, trial
