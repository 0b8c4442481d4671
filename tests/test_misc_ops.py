"""Misc op tests vs Python oracles (bloom, zorder/hilbert, strings, agg64,
multiply, datetime rebase/trunc, case_when)."""
import datetime
import random
import struct

import pytest
import torch

from spark_rapids_jni_amd.columnar import Column, DType

random.seed(63)


@pytest.mark.gpu
def test_case_when_select_first_true():
    from spark_rapids_jni_amd.ops.misc import select_first_true_index
    c1 = Column.from_pylist([True, False, None, False], DType.BOOL8, "cuda")
    c2 = Column.from_pylist([False, True, False, None], DType.BOOL8, "cuda")
    got = select_first_true_index([c1, c2]).to_pylist()
    assert got == [0, 1, 2, 2]


@pytest.mark.gpu
@pytest.mark.parametrize("version", [1, 2])
def test_bloom_filter_roundtrip(version):
    from spark_rapids_jni_amd.ops.misc import BloomFilter
    bf = BloomFilter(version, num_hashes=3, num_longs=64, seed=42)
    keys = [random.randint(-10**9, 10**9) for _ in range(500)]
    bf.put(Column.from_pylist(keys, DType.INT64, "cuda"))
    probe = keys[:100] + [10**12 + i for i in range(200)] + [None]
    got = bf.might_contain(
        Column.from_pylist(probe, DType.INT64, "cuda")).to_pylist()
    for i in range(100):
        assert got[i] is True  # no false negatives
    fp = sum(1 for i in range(100, 300) if got[i])
    assert fp < 50  # loose false-positive sanity bound
    assert got[300] is None


@pytest.mark.gpu
def test_bloom_filter_merge():
    from spark_rapids_jni_amd.ops.misc import BloomFilter
    a = BloomFilter(2, 3, 32, seed=7)
    b = BloomFilter(2, 3, 32, seed=7)
    a.put(Column.from_pylist([1, 2, 3], DType.INT64, "cuda"))
    b.put(Column.from_pylist([100, 200], DType.INT64, "cuda"))
    a.merge(b)
    got = a.might_contain(
        Column.from_pylist([1, 2, 3, 100, 200], DType.INT64, "cuda")).to_pylist()
    assert all(got)


def _hilbert_oracle(coords, nbits):
    X = list(coords)
    n = len(X)
    M = 1 << (nbits - 1)
    Q = M
    while Q > 1:
        P = Q - 1
        for i in range(n):
            if X[i] & Q:
                X[0] ^= P
            else:
                t = (X[0] ^ X[i]) & P
                X[0] ^= t
                X[i] ^= t
        Q >>= 1
    for i in range(1, n):
        X[i] ^= X[i - 1]
    t = 0
    Q = M
    while Q > 1:
        if X[n - 1] & Q:
            t ^= Q - 1
        Q >>= 1
    for i in range(n):
        X[i] ^= t
    idx = 0
    for b in range(nbits - 1, -1, -1):
        for j in range(n):
            idx = (idx << 1) | ((X[j] >> b) & 1)
    return idx


@pytest.mark.gpu
def test_hilbert_index():
    from spark_rapids_jni_amd.ops.misc import hilbert_index
    nbits = 10
    n = 300
    xs = [random.randrange(1 << nbits) for _ in range(n)]
    ys = [random.randrange(1 << nbits) for _ in range(n)]
    cols = [Column.from_pylist(xs, DType.INT32, "cuda"),
            Column.from_pylist(ys, DType.INT32, "cuda")]
    got = hilbert_index(nbits, cols).to_pylist()
    for i in range(n):
        assert got[i] == _hilbert_oracle([xs[i], ys[i]], nbits), i
    # locality sanity: consecutive curve points are grid neighbors
    inv = {}
    for x in range(8):
        for y in range(8):
            inv[_hilbert_oracle([x, y], 3)] = (x, y)
    pts = [inv[i] for i in range(64)]
    for a, b in zip(pts, pts[1:]):
        assert abs(a[0] - b[0]) + abs(a[1] - b[1]) == 1


@pytest.mark.gpu
def test_interleave_bits():
    from spark_rapids_jni_amd.ops.misc import interleave_bits
    a = Column.from_pylist([0xFFFFFFFF - 2**31, 0], DType.INT32, "cuda")
    b = Column.from_pylist([0, -2**31], DType.INT32, "cuda")
    # value semantics: a row0 = 0x7fffffff, b row0 = 0
    out = interleave_bits([a, b])
    raw = out.data.cpu().numpy().tobytes()
    r0 = raw[:8]
    # a=0111..., b=0000... interleaved (a first) -> 00 01 01 01 ... = 0x15? per
    # bit j: col j%2, bit j/2: j0=a.bit0(0) j1=b.bit0(0) j2=a.bit1(1) j3=b.bit1(0)
    bits = []
    av, bv = 0x7FFFFFFF, 0
    for j in range(64):
        c = j % 2
        k = j // 2
        v = av if c == 0 else bv
        bits.append((v >> (31 - k)) & 1)
    exp0 = bytes(sum(bits[i * 8 + t] << (7 - t) for t in range(8))
                 for i in range(8))
    assert r0 == exp0


@pytest.mark.gpu
def test_bytes_to_hex_and_uuid():
    from spark_rapids_jni_amd.ops.misc import bytes_to_hex, random_uuids
    vals = ["abc", "", None, "\x00\xff"]
    col = Column.from_pylist(vals, DType.STRING, "cuda")
    got = bytes_to_hex(col).to_pylist()
    assert got[0] == "616263"
    assert got[1] == ""
    assert got[2] is None
    assert got[3] == "00C3BF"  # utf-8 bytes of "\x00\xff"
    u = random_uuids(100, seed=5).to_pylist()
    assert len(set(u)) == 100
    for s in u:
        assert len(s) == 36 and s[8] == s[13] == s[18] == s[23] == "-"
        assert s[14] == "4"
        assert s[19] in "89ab"


@pytest.mark.gpu
def test_substring_index():
    from spark_rapids_jni_amd.ops.misc import substring_index
    vals = ["www.apache.org", "a.b", "nodot", "", None, "a.b.c.d"]
    col = Column.from_pylist(vals, DType.STRING, "cuda")
    assert substring_index(col, ".", 2).to_pylist() == \
        ["www.apache", "a.b", "nodot", "", None, "a.b"]
    assert substring_index(col, ".", -2).to_pylist() == \
        ["apache.org", "a.b", "nodot", "", None, "c.d"]
    assert substring_index(col, ".", 0).to_pylist() == \
        ["", "", "", "", None, ""]
    # reference GpuSubstringIndexUtilsTest vectors: count beyond the number
    # of delimiters returns the whole string; multi-char and multi-byte
    # UTF-8 delimiters count as units
    assert substring_index(col, ".", 3).to_pylist()[0] == "www.apache.org"
    assert substring_index(col, ".", -3).to_pylist()[0] == "www.apache.org"
    cn = Column.from_pylist(["大千世界大千世界"], DType.STRING, "cuda")
    assert substring_index(cn, "千", 2).to_pylist() == ["大千世界大"]
    bars = Column.from_pylist(["www||apache||org"], DType.STRING, "cuda")
    assert substring_index(bars, "||", 2).to_pylist() == ["www||apache"]


@pytest.mark.gpu
def test_literal_range_pattern():
    from spark_rapids_jni_amd.ops.misc import literal_range_pattern
    vals = ["abc123x", "abc12", "xxabc999", "abc", None, "zabc00z"]
    col = Column.from_pylist(vals, DType.STRING, "cuda")
    got = literal_range_pattern(col, "abc", 3, "0", "9").to_pylist()
    assert got == [True, False, True, False, None, False]


@pytest.mark.gpu
def test_aggregation64_chunks():
    from spark_rapids_jni_amd.ops.misc import Aggregation64Utils as A
    vals = [0, 1, -1, 2**40, -(2**40), 2**62, -(2**62), None]
    col = Column.from_pylist(vals, DType.INT64, "cuda")
    lo = A.extract_int32_chunk(col, 0)
    hi = A.extract_int32_chunk(col, 1)
    lov, hiv = lo.data.cpu().tolist(), hi.data.cpu().tolist()
    for i, v in enumerate(vals):
        if v is None:
            continue
        assert (lov[i] + (hiv[i] << 32)) == v, f"row {i}"
    s, ov = A.combine_int64_sum_chunks(lo, hi)
    sv = s.data.cpu().tolist()
    for i, v in enumerate(vals):
        if v is not None:
            assert sv[i] == v
    assert not any(ov.data.cpu().tolist()[:7])


@pytest.mark.gpu
def test_multiply_ansi():
    from spark_rapids_jni_amd.ops.misc import OverflowError64, multiply_int64
    a = Column.from_pylist([2, 3, 2**62, None], DType.INT64, "cuda")
    b = Column.from_pylist([5, -7, 4, 9], DType.INT64, "cuda")
    got = multiply_int64(a, b).to_pylist()
    assert got == [10, -21, None, None]  # overflow -> null (non-ANSI)
    with pytest.raises(OverflowError64) as ei:
        multiply_int64(a, b, ansi=True)
    assert ei.value.row_with_error == 2


def _days(y, m, d):
    return (datetime.date(y, m, d) - datetime.date(1970, 1, 1)).days


@pytest.mark.gpu
def test_rebase_days_roundtrip():
    from spark_rapids_jni_amd.ops.misc import (rebase_gregorian_to_julian,
                                               rebase_julian_to_gregorian)
    days = [_days(2020, 1, 1), _days(1582, 10, 15), _days(1582, 10, 14) - 10,
            _days(1200, 6, 1), _days(100, 1, 1), 0, None]
    col = Column.from_pylist(days, DType.DATE32, "cuda")
    jul = rebase_gregorian_to_julian(col)
    back = rebase_julian_to_gregorian(jul).to_pylist()
    assert back == days
    julv = jul.to_pylist()
    # modern dates unchanged; 1200-06-01 proleptic gregorian = julian +7 days
    assert julv[0] == days[0]
    assert julv[1] == days[1]
    assert julv[3] != days[3]


@pytest.mark.gpu
def test_truncate_timestamp():
    from spark_rapids_jni_amd.ops.misc import truncate_timestamp
    base = datetime.datetime(2021, 7 - 0, 15, 13, 45, 59, 123456,
                             tzinfo=datetime.timezone.utc)
    epoch = datetime.datetime(1970, 1, 1, tzinfo=datetime.timezone.utc)
    us = int((base - epoch).total_seconds() * 1e6) + 456  # keep micros exact
    us = int((base - epoch) // datetime.timedelta(microseconds=1))
    col = Column.from_pylist([us, None], DType.TIMESTAMP_US, "cuda")

    def expect(**kw):
        t = base.replace(**kw)
        return int((t - epoch) // datetime.timedelta(microseconds=1))

    assert truncate_timestamp(col, "YEAR").to_pylist()[0] == expect(
        month=1, day=1, hour=0, minute=0, second=0, microsecond=0)
    assert truncate_timestamp(col, "QUARTER").to_pylist()[0] == expect(
        month=7, day=1, hour=0, minute=0, second=0, microsecond=0)
    assert truncate_timestamp(col, "MONTH").to_pylist()[0] == expect(
        day=1, hour=0, minute=0, second=0, microsecond=0)
    assert truncate_timestamp(col, "DAY").to_pylist()[0] == expect(
        hour=0, minute=0, second=0, microsecond=0)
    assert truncate_timestamp(col, "HOUR").to_pylist()[0] == expect(
        minute=0, second=0, microsecond=0)
    assert truncate_timestamp(col, "SECOND").to_pylist()[0] == expect(
        microsecond=0)
    assert truncate_timestamp(col, "MILLISECOND").to_pylist()[0] == expect(
        microsecond=123000)
    # WEEK: 2021-07-15 is Thursday -> Monday 2021-07-12
    wk = truncate_timestamp(col, "WEEK").to_pylist()[0]
    exp_wk = int((datetime.datetime(2021, 7, 12, tzinfo=datetime.timezone.utc)
                  - epoch) // datetime.timedelta(microseconds=1))
    assert wk == exp_wk
    assert truncate_timestamp(col, "DAY").to_pylist()[1] is None


@pytest.mark.gpu
def test_radix_sort_i64():
    from spark_rapids_jni_amd.ops.sort import sort_pairs_i64
    for n in [0, 1, 63, 2048, 5000, 100000]:
        t = torch.randint(-2**62, 2**62, (n,), dtype=torch.int64, device="cuda")
        if n > 10:
            t[:5] = torch.tensor([0, -1, 1, -2**62, 2**62 - 1], device="cuda")
        sk, order = sort_pairs_i64(t)
        ref, ref_idx = torch.sort(t, stable=True)
        assert torch.equal(sk, ref), f"n={n}"
        assert torch.equal(t[order], sk), f"n={n} payload"


@pytest.mark.gpu
def test_radix_sort_stability():
    from spark_rapids_jni_amd.ops.sort import sort_pairs_i64
    t = torch.tensor([5, 3, 5, 3, 5, 3] * 700, dtype=torch.int64,
                     device="cuda")
    sk, order = sort_pairs_i64(t)
    oh = order.cpu().tolist()
    threes = [i for i in oh[:2100]]
    assert threes == sorted(threes)  # stable: original order preserved
    fives = [i for i in oh[2100:]]
    assert fives == sorted(fives)


@pytest.mark.gpu
def test_sort_merge_join_matches_hash_join():
    from spark_rapids_jni_amd.ops.join import HashJoinTable
    from spark_rapids_jni_amd.ops.sort import sort_merge_inner_join
    import random as rnd
    rnd.seed(3)
    bvals = [rnd.randint(0, 500) if rnd.random() > 0.02 else None
             for _ in range(1500)]
    pvals = [rnd.randint(0, 600) if rnd.random() > 0.02 else None
             for _ in range(4000)]
    b = Column.from_pylist(bvals, DType.INT64, "cuda")
    p = Column.from_pylist(pvals, DType.INT64, "cuda")
    sb, sp = sort_merge_inner_join(b, p)
    hb, hp = HashJoinTable.build(b).inner_join(p)
    assert set(zip(sb.cpu().tolist(), sp.cpu().tolist())) == \
        set(zip(hb.cpu().tolist(), hp.cpu().tolist()))


@pytest.mark.gpu
def test_literal_range_pattern_codepoints():
    """Reference RegexRewriteUtilsTest vectors: digit class and a CJK
    codepoint class (19968..40869) — ranges are CODEPOINTS, not bytes."""
    from spark_rapids_jni_amd.ops.misc import literal_range_pattern
    col = Column.from_pylist(["abc123", "aabc123", "aabc12", "abc1232",
                              "aabc1232"], DType.STRING, "cuda")
    got = literal_range_pattern(col, "abc", 3, "0", "9").to_pylist()
    assert got == [True, True, False, True, True]
    cn = Column.from_pylist(["数据砖块", "火花-急流英伟达", "英伟达Nvidia",
                             "火花-急流"], DType.STRING, "cuda")
    got = literal_range_pattern(cn, "英", 2, chr(19968),
                                chr(40869)).to_pylist()
    assert got == [False, True, True, False]
