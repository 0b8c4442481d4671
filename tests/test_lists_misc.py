"""list/map/iceberg/round/AST tests + HostTable/fileio/version/arms (CPU)."""
import random

import pytest
import torch

from spark_rapids_jni_amd.columnar import Column, DType, Table

random.seed(91)


def _mk_list(vals_per_row, dtype=DType.INT64, device="cuda"):
    child_vals = []
    offs = [0]
    valid = []
    for row in vals_per_row:
        valid.append(row is not None)
        if row:
            child_vals.extend(row)
        offs.append(len(child_vals))
    from spark_rapids_jni_amd.columnar import validity_from_bools
    child = Column.from_pylist(child_vals, dtype, device)
    v = validity_from_bools(valid, device) if not all(valid) else None
    return Column(DType.LIST, len(vals_per_row), None, v,
                  torch.tensor(offs, dtype=torch.int32, device=device), [child],
                  null_count=None)


@pytest.mark.gpu
def test_list_slice():
    from spark_rapids_jni_amd.ops.lists import list_slice
    rows = [[1, 2, 3, 4, 5], [], [9], None, [10, 20, 30]]
    col = _mk_list(rows)
    got = list_slice(col, 2, 2).to_pylist()
    assert got == [[2, 3], [], [], None, [20, 30]]
    got = list_slice(col, -2, 5).to_pylist()
    assert got == [[4, 5], [], [], None, [20, 30]]
    starts = Column.from_pylist([1, 1, 1, 1, 3], DType.INT32, "cuda")
    got = list_slice(col, starts, 1).to_pylist()
    assert got == [[1], [], [9], None, [30]]


@pytest.mark.gpu
def test_map_ops():
    from spark_rapids_jni_amd.ops.lists import (is_valid_map, map_from_entries,
                                                sort_map_column)
    def mk_map(rows):
        keys, vals, offs = [], [], [0]
        for r in rows:
            for k, v in r:
                keys.append(k)
                vals.append(v)
            offs.append(len(keys))
        kc = Column.from_pylist(keys, DType.STRING, "cuda")
        vc = Column.from_pylist(vals, DType.INT64, "cuda")
        entry = Column(DType.STRUCT, len(keys), None, None, None, [kc, vc])
        return Column(DType.LIST, len(rows), None, None,
                      torch.tensor(offs, dtype=torch.int32, device="cuda"),
                      [entry])

    good = mk_map([[("b", 2), ("a", 1)], [], [("x", 9), ("y", 8), ("c", 7)]])
    assert is_valid_map(good).to_pylist() == [True, True, True]
    dup = mk_map([[("a", 1), ("a", 2)]])
    assert is_valid_map(dup).to_pylist() == [False]
    with pytest.raises(ValueError):
        map_from_entries(dup)
    s = sort_map_column(good).to_pylist()
    assert s[0] == [("a", 1), ("b", 2)]
    assert s[2] == [("c", 7), ("x", 9), ("y", 8)]


@pytest.mark.gpu
def test_map_zip():
    from spark_rapids_jni_amd.ops.lists import map_zip, sort_map_column
    def mk(rows):
        keys, vals, offs = [], [], [0]
        for r in rows:
            for k, v in r:
                keys.append(k)
                vals.append(v)
            offs.append(len(keys))
        kc = Column.from_pylist(keys, DType.INT64, "cuda")
        vc = Column.from_pylist(vals, DType.INT64, "cuda")
        entry = Column(DType.STRUCT, len(keys), None, None, None, [kc, vc])
        return Column(DType.LIST, len(rows), None, None,
                      torch.tensor(offs, dtype=torch.int32, device="cuda"),
                      [entry])

    m1 = mk([[(1, 10), (3, 30)], [(5, 50)]])
    m2 = mk([[(2, 200), (3, 300)], []])
    z = map_zip(m1, m2)
    out = z.to_pylist()
    assert out[0] == [(1, (10, None)), (2, (None, 200)), (3, (30, 300))]
    assert out[1] == [(5, (50, None))]


@pytest.mark.gpu
def test_iceberg_transforms():
    from spark_rapids_jni_amd.ops.lists import (iceberg_bucket,
                                                iceberg_datetime_transform,
                                                iceberg_truncate)
    # iceberg spec test vectors: bucket hash of 34 (int/long) = 2017239379
    from spark_rapids_jni_amd.utils import sparkref
    import struct as st
    h = sparkref.murmur3_bytes(st.pack("<q", 34), 0)
    assert sparkref._to_signed32(h) == 2017239379
    col = Column.from_pylist([34, None, -1], DType.INT64, "cuda")
    b = iceberg_bucket(col, 16).to_pylist()
    assert b[0] == (2017239379 & 0x7FFFFFFF) % 16
    assert b[1] is None
    # spec Appendix B: date 2017-11-16 (17486 days) hashes to -653330422,
    # timestamp 2017-11-16T22:31:08 (micros) to -2047944441 — both via the
    # int64 path (dates/timestamps bucket as longs)
    dcol2 = Column.from_pylist([17486], DType.DATE32, "cuda")
    assert iceberg_bucket(dcol2, 128).to_pylist() == \
        [(-653330422 & 0x7FFFFFFF) % 128]
    tcol = Column.from_pylist([1510871468000000], DType.TIMESTAMP_US, "cuda")
    assert iceberg_bucket(tcol, 128).to_pylist() == \
        [(-2047944441 & 0x7FFFFFFF) % 128]
    # truncate: W=10: 11->10, -11->-20 (floored)
    t = iceberg_truncate(Column.from_pylist([11, -11, 0], DType.INT64, "cuda"),
                         10).to_pylist()
    assert t == [10, -20, 0]
    # datetime: 2021-06-15 -> year 51 (2021-1970), month 617, day=days
    import datetime
    days = (datetime.date(2021, 6, 15) - datetime.date(1970, 1, 1)).days
    dcol = Column.from_pylist([days], DType.DATE32, "cuda")
    assert iceberg_datetime_transform(dcol, "year").to_pylist() == [51]
    assert iceberg_datetime_transform(dcol, "month").to_pylist() == [51 * 12 + 5]
    assert iceberg_datetime_transform(dcol, "day").to_pylist() == [days]


@pytest.mark.gpu
def test_round_double():
    from spark_rapids_jni_amd.ops.lists import round_double
    vals = [2.5, 3.5, -2.5, 1.25, 1.35, 0.005, 123.456, None, 2.675]
    col = Column.from_pylist(vals, DType.FLOAT64, "cuda")
    up = round_double(col, 0).to_pylist()
    assert up[:3] == [3.0, 4.0, -3.0]  # HALF_UP away from zero
    assert up[6] == 123.0
    assert up[7] is None
    even = round_double(col, 1, half_even=True).to_pylist()
    assert even[3] == 1.2  # 1.25 -> 1.2 (even)
    assert even[4] == 1.4  # 1.35 is actually 1.35000000000000008882 -> 1.4
    two = round_double(col, 2).to_pylist()
    # Spark rounds via the SHORTEST decimal form (BigDecimal.valueOf /
    # Double.toString): 2.675 -> "2.675" -> HALF_UP -> 2.68
    assert two[8] == 2.68


@pytest.mark.gpu
def test_ast_filter_and_matched_rows():
    from spark_rapids_jni_amd.ops.join import HashJoinTable
    from spark_rapids_jni_amd.ops.lists import (Ast, filter_gather_maps_by_ast,
                                                get_matched_rows)
    b_keys = Column.from_pylist([1, 2, 3, 4], DType.INT64, "cuda")
    b_val = Column.from_pylist([10, 20, 30, 40], DType.INT64, "cuda")
    p_keys = Column.from_pylist([1, 2, 3, 4, 2], DType.INT64, "cuda")
    p_val = Column.from_pylist([5, 25, 50, 35, 19], DType.INT64, "cuda")
    bi, pi = HashJoinTable.build(b_keys).inner_join(p_keys)
    # mixed join: keep pairs where b_val < p_val  (cols = [b_val, p_val])
    ast = Ast().left_col(0).right_col(1).op("<")
    fl, fr = filter_gather_maps_by_ast([b_val, p_val], ast, bi, pi)
    pairs = set(zip(fl.cpu().tolist(), fr.cpu().tolist()))
    # b=2(20) vs p=25 keep; b=3(30) vs 50 keep; b=4(40) vs 35 drop;
    # b=1(10) vs 5 drop; b=2(20) vs 19 drop
    assert pairs == {(1, 1), (2, 2)}
    m = get_matched_rows(fl, 4).to_pylist()
    assert m == [False, True, True, False]


@pytest.mark.gpu
def test_host_table_roundtrip():
    from spark_rapids_jni_amd.hosttable import HostTable
    cols = [Column.from_pylist([1, None, 3], DType.INT64, "cuda"),
            Column.from_pylist(["a", "bb", None], DType.STRING, "cuda")]
    t = Table(cols)
    ht = HostTable.from_table_async(t)
    torch.cuda.synchronize()
    back = ht.to_table_async()
    torch.cuda.synchronize()
    assert back.columns[0].to_pylist() == [1, None, 3]
    assert back.columns[1].to_pylist() == ["a", "bb", None]


def test_fileio_local(tmp_path):
    from spark_rapids_jni_amd.fileio import LocalFileIO
    fio = LocalFileIO()
    p = str(tmp_path / "f.bin")
    with fio.new_output_file(p).create() as f:
        f.write(b"0123456789")
    inf = fio.new_input_file(p)
    assert inf.length() == 10
    s = inf.open()
    assert s.read_fully(3, 4) == b"3456"
    # reference RapidsInputFileTest: tail reads + clean failures
    assert inf.read_tail(4) == b"6789"
    assert inf.read_tail(10) == b"0123456789"
    with pytest.raises(EOFError):
        inf.read_tail(11)
    with pytest.raises(EOFError):
        s.read_fully(8, 4)
    with pytest.raises(ValueError):
        s.read_fully(-1, 2)
    with pytest.raises(ValueError):
        s.read_fully(0, -2)


def test_version_gates():
    from spark_rapids_jni_amd.version import SparkPlatformType, Version
    v = Version(SparkPlatformType.VANILLA_SPARK, 3, 2)
    assert v.is_vanilla_320()
    assert not v.is_vanilla_330_or_later()
    db = Version(SparkPlatformType.DATABRICKS, 14, 3)
    assert db.is_databricks_14_3_or_later()


def test_arms():
    from spark_rapids_jni_amd.utils.arms import (Pair, check_argument,
                                                 close_quietly, closing_all)
    p = Pair(1, "x")
    assert p.left == 1 and p.right == "x"
    with pytest.raises(ValueError):
        check_argument(False, "nope")

    class C:
        closed = False

        def close(self):
            self.closed = True

    c1, c2 = C(), C()
    with closing_all(c1, c2):
        pass
    assert c1.closed and c2.closed
    close_quietly(None, 5)  # no-op, no raise


@pytest.mark.gpu
def test_q3_pipeline_vs_torch_oracle():
    from spark_rapids_jni_amd import exec as ex
    n = 200000
    ss = ex.gen_store_sales(n, n_items=500, device="cuda")
    dd = ex.gen_date_dim(device="cuda")
    it = ex.gen_item(500, device="cuda")
    keys, res = ex.q3_like(ss, dd, it, year=2000, manufact_id=50)
    got = dict(zip(keys.columns[0].to_pylist(), res[0].to_pylist()))
    # torch oracle
    sold, item_sk = ss.columns[0].data, ss.columns[1].data
    qty, price = ss.columns[2].data, ss.columns[3].data
    d_sk, d_year = dd.columns[0].data, dd.columns[1].data
    i_sk, i_brand, i_man = (it.columns[0].data, it.columns[1].data,
                            it.columns[2].data)
    good_dates = set(d_sk[d_year == 2000].cpu().tolist())
    man_items = {int(s): int(b) for s, b, m in
                 zip(i_sk.cpu().tolist(), i_brand.cpu().tolist(),
                     i_man.cpu().tolist()) if m == 50}
    exp = {}
    for s, isk, p in zip(sold.cpu().tolist(), item_sk.cpu().tolist(),
                         price.cpu().tolist()):
        if s in good_dates and isk in man_items:
            b = man_items[isk]
            exp[b] = exp.get(b, 0.0) + p
    assert set(got) == set(exp)
    for b in exp:
        assert abs(got[b] - exp[b]) < 1e-6 * max(1.0, abs(exp[b]))


@pytest.mark.gpu
def test_iceberg_bucket_decimal_and_binary():
    """Iceberg spec Appendix B vectors: decimal buckets hash the minimal
    big-endian two's-complement unscaled bytes; binary hashes raw bytes."""
    from spark_rapids_jni_amd.ops.lists import iceberg_bucket
    from spark_rapids_jni_amd.utils import sparkref as ref

    # iceberg spec Appendix B vectors (STANDARD murmur3, not Spark's
    # tail-as-int variant): decimal 14.20 -> bytes 0x05 0x8C ->
    # -500754589; "iceberg" -> 1210000089
    assert ref.murmur3_bytes_std(b"\x05\x8c", 0) & 0xFFFFFFFF == \
        (-500754589) & 0xFFFFFFFF
    assert ref.murmur3_bytes_std(b"iceberg", 0) & 0xFFFFFFFF == \
        1210000089
    for dt in (DType.DECIMAL32, DType.DECIMAL64, DType.DECIMAL128):
        col = Column.from_pylist([1420, None, -7, 0, 2**30], dt,
                                 device="cuda", scale=2)
        got = iceberg_bucket(col, 16).to_pylist()
        import struct as _s
        for i, v in enumerate([1420, None, -7, 0, 2**30]):
            if v is None:
                assert got[i] is None
                continue
            exp_bytes = ref.java_bigint_bytes(v)
            h = ref.murmur3_bytes_std(exp_bytes, 0)
            assert got[i] == (h & 0x7FFFFFFF) % 16, (dt, i)

    # strings: spec vector "iceberg" -> 1210000089 through the kernel
    scol = Column.from_pylist(["iceberg", "abc", None], DType.STRING,
                              device="cuda")
    sgot = iceberg_bucket(scol, 64).to_pylist()
    assert sgot[0] == (1210000089 & 0x7FFFFFFF) % 64
    assert sgot[1] == (ref.murmur3_bytes_std(b"abc", 0) & 0x7FFFFFFF) % 64
    assert sgot[2] is None

    # binary payloads go through the string/bytes path
    col = Column.from_pylist([b"\x00\x01\x02\x03", b"", None], DType.STRING,
                             device="cuda")
    got = iceberg_bucket(col, 128).to_pylist()
    exp = (ref.murmur3_bytes_std(b"\x00\x01\x02\x03", 0) & 0x7FFFFFFF) % 128
    assert got[0] == exp
    # spec: binary 0x00 01 02 03 hashes to -188683207
    assert ref.murmur3_bytes_std(b"\x00\x01\x02\x03", 0) & 0xFFFFFFFF == \
        (-188683207) & 0xFFFFFFFF
