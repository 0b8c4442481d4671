"""Join / aggregate / gather / partition tests against Python oracles."""
import random
from collections import defaultdict

import pytest
import torch

from spark_rapids_jni_amd.columnar import Column, DType, Table

random.seed(7)


def _pairs(build_idx, probe_idx):
    return set(zip(build_idx.cpu().tolist(), probe_idx.cpu().tolist()))


@pytest.mark.gpu
def test_inner_join_int64():
    from spark_rapids_jni_amd.ops.join import HashJoinTable
    nb, np_ = 2000, 5000
    bvals = [random.randint(0, 700) for _ in range(nb)]
    pvals = [random.randint(0, 1000) for _ in range(np_)]
    b = Column.from_pylist(bvals, DType.INT64, "cuda")
    p = Column.from_pylist(pvals, DType.INT64, "cuda")
    tbl = HashJoinTable.build(b)
    bi, pi = tbl.inner_join(p)
    expect = set()
    index = defaultdict(list)
    for i, v in enumerate(bvals):
        index[v].append(i)
    for j, v in enumerate(pvals):
        for i in index.get(v, ()):
            expect.add((i, j))
    assert _pairs(bi, pi) == expect


@pytest.mark.gpu
def test_inner_join_nulls_never_match():
    from spark_rapids_jni_amd.ops.join import HashJoinTable
    bvals = [1, None, 2, 3, None]
    pvals = [None, 1, 3, None, 9]
    b = Column.from_pylist(bvals, DType.INT64, "cuda")
    p = Column.from_pylist(pvals, DType.INT64, "cuda")
    bi, pi = HashJoinTable.build(b).inner_join(p)
    assert _pairs(bi, pi) == {(0, 1), (3, 2)}


@pytest.mark.gpu
def test_inner_join_multicol_string():
    from spark_rapids_jni_amd.ops.join import HashJoinTable
    nb, np_ = 300, 800
    bk1 = [random.randint(0, 20) for _ in range(nb)]
    bk2 = [random.choice(["a", "bb", "ccc", None, "dd"]) for _ in range(nb)]
    pk1 = [random.randint(0, 25) for _ in range(np_)]
    pk2 = [random.choice(["a", "bb", "ccc", None, "x"]) for _ in range(np_)]
    b = [Column.from_pylist(bk1, DType.INT32, "cuda"),
         Column.from_pylist(bk2, DType.STRING, "cuda")]
    p = [Column.from_pylist(pk1, DType.INT32, "cuda"),
         Column.from_pylist(pk2, DType.STRING, "cuda")]
    bi, pi = HashJoinTable.build(b).inner_join(p)
    expect = set()
    for i in range(nb):
        if bk2[i] is None:
            continue
        for j in range(np_):
            if pk2[j] is None:
                continue
            if bk1[i] == pk1[j] and bk2[i] == pk2[j]:
                expect.add((i, j))
    assert _pairs(bi, pi) == expect


@pytest.mark.gpu
def test_semi_anti_join():
    from spark_rapids_jni_amd.ops.join import HashJoinTable
    bvals = [1, 2, 3]
    pvals = [0, 1, 2, 5, 2, None]
    tbl = HashJoinTable.build(Column.from_pylist(bvals, DType.INT64, "cuda"))
    p = Column.from_pylist(pvals, DType.INT64, "cuda")
    semi = sorted(tbl.semi_join(p).cpu().tolist())
    anti = sorted(tbl.semi_join(p, anti=True).cpu().tolist())
    assert semi == [1, 2, 4]
    assert anti == [0, 3, 5]  # null key: no match -> anti emits it


@pytest.mark.gpu
def test_left_outer_full_outer():
    from spark_rapids_jni_amd.ops.join import HashJoinTable, make_full_outer, make_left_outer
    bvals = [10, 20, 30]
    pvals = [20, 99, 10]
    tbl = HashJoinTable.build(Column.from_pylist(bvals, DType.INT64, "cuda"))
    p = Column.from_pylist(pvals, DType.INT64, "cuda")
    bi, pi, matched = tbl.inner_join(p, track_build_matches=True)
    lb, lp = make_left_outer(3, bi, pi)
    pairs = set(zip(lb.cpu().tolist(), lp.cpu().tolist()))
    assert pairs == {(1, 0), (0, 2), (-1, 1)}
    fb, fp = make_full_outer(matched, bi, pi, 3)
    pairs = set(zip(fb.cpu().tolist(), fp.cpu().tolist()))
    assert pairs == {(1, 0), (0, 2), (-1, 1), (2, -1)}


@pytest.mark.gpu
def test_gather_with_nulls():
    from spark_rapids_jni_amd.ops.copying import gather_column
    vals = [1.5, None, 3.5, 4.5]
    col = Column.from_pylist(vals, DType.FLOAT64, "cuda")
    gmap = torch.tensor([3, -1, 1, 0, 2], dtype=torch.int64, device="cuda")
    out = gather_column(col, gmap)
    assert out.to_pylist() == [4.5, None, None, 1.5, 3.5]


@pytest.mark.gpu
def test_gather_strings():
    from spark_rapids_jni_amd.ops.copying import gather_column
    vals = ["hello", None, "", "world", "xyzzy"]
    col = Column.from_pylist(vals, DType.STRING, "cuda")
    gmap = torch.tensor([4, 0, -1, 2, 1, 3], dtype=torch.int64, device="cuda")
    out = gather_column(col, gmap)
    assert out.to_pylist() == ["xyzzy", "hello", None, "", None, "world"]


@pytest.mark.gpu
def test_groupby_sums_counts():
    from spark_rapids_jni_amd.ops.aggregate import Agg, groupby
    n = 5000
    keys = [random.randint(0, 50) if random.random() > 0.05 else None
            for _ in range(n)]
    vals = [random.randint(-100, 100) if random.random() > 0.1 else None
            for _ in range(n)]
    fvals = [random.random() * 10 if v is not None else None for v in vals]
    kc = Column.from_pylist(keys, DType.INT64, "cuda")
    vc = Column.from_pylist(vals, DType.INT64, "cuda")
    fc = Column.from_pylist(fvals, DType.FLOAT64, "cuda")
    kt, res = groupby(kc, [(Agg.COUNT_ALL, None), (Agg.COUNT_VALID, vc),
                           (Agg.SUM, vc), (Agg.MIN, vc), (Agg.MAX, vc),
                           (Agg.SUM, fc)])
    out_keys = kt.columns[0].to_pylist()
    got = {k: (c, cv, s, mn, mx, fs) for k, c, cv, s, mn, mx, fs in zip(
        out_keys, res[0].to_pylist(), res[1].to_pylist(), res[2].to_pylist(),
        res[3].to_pylist(), res[4].to_pylist(), res[5].to_pylist())}
    exp = defaultdict(lambda: [0, 0, None, None, None, None])
    for k, v, f in zip(keys, vals, fvals):
        e = exp[k]
        e[0] += 1
        if v is not None:
            e[1] += 1
            e[2] = v if e[2] is None else e[2] + v
            e[3] = v if e[3] is None else min(e[3], v)
            e[4] = v if e[4] is None else max(e[4], v)
            e[5] = f if e[5] is None else e[5] + f
    assert set(got.keys()) == set(exp.keys())
    for k, e in exp.items():
        c, cv, s, mn, mx, fs = got[k]
        assert (c, cv, s, mn, mx) == tuple(e[:5]), f"key {k}"
        if e[5] is None:
            assert fs is None
        else:
            assert abs(fs - e[5]) < 1e-6


@pytest.mark.gpu
def test_groupby_multicol_keys():
    from spark_rapids_jni_amd.ops.aggregate import Agg, groupby
    n = 2000
    k1 = [random.randint(0, 5) for _ in range(n)]
    k2 = [random.choice(["x", "y", None]) for _ in range(n)]
    c1 = Column.from_pylist(k1, DType.INT32, "cuda")
    c2 = Column.from_pylist(k2, DType.STRING, "cuda")
    kt, res = groupby([c1, c2], [(Agg.COUNT_ALL, None)])
    gk1 = kt.columns[0].to_pylist()
    gk2 = kt.columns[1].to_pylist()
    got = {(a, b): c for a, b, c in zip(gk1, gk2, res[0].to_pylist())}
    exp = defaultdict(int)
    for a, b in zip(k1, k2):
        exp[(a, b)] += 1
    assert got == dict(exp)


@pytest.mark.gpu
def test_partition_roundtrip():
    from spark_rapids_jni_amd.ops import hashing
    from spark_rapids_jni_amd.ops.copying import partition_map, spark_partition_ids
    n, nparts = 10000, 8
    vals = [random.randint(-10**9, 10**9) for _ in range(n)]
    col = Column.from_pylist(vals, DType.INT64, "cuda")
    h = hashing.murmur3([col])
    pids = spark_partition_ids(h, nparts)
    offsets, perm = partition_map(pids, nparts)
    offs = offsets.cpu().tolist()
    permh = perm.cpu().tolist()
    pidsh = pids.cpu().tolist()
    assert offs[0] == 0 and offs[-1] == n
    assert sorted(permh) == list(range(n))
    for p in range(nparts):
        for d in range(offs[p], offs[p + 1]):
            assert pidsh[permh[d]] == p


@pytest.mark.gpu
def test_i64_fast_path_matches_generic():
    from spark_rapids_jni_amd.ops.join import HashJoinTable
    nb, np_ = 3000, 9000
    bvals = [random.randint(0, 1500) if random.random() > 0.02 else None
             for _ in range(nb)]
    pvals = [random.randint(0, 2000) if random.random() > 0.02 else None
             for _ in range(np_)]
    b = Column.from_pylist(bvals, DType.INT64, "cuda")
    p = Column.from_pylist(pvals, DType.INT64, "cuda")
    fast = HashJoinTable.build(b)
    slow = HashJoinTable.build(b, force_generic=True)
    assert fast.i64_fast and not slow.i64_fast
    fb, fp = fast.inner_join(p)
    sb, sp = slow.inner_join(p)
    assert _pairs(fb, fp) == _pairs(sb, sp)


@pytest.mark.gpu
def test_groupby_i64_fast_path():
    """Single non-null int64 key takes the specialized Slot64 kernel; must
    agree with a host oracle, including the INT64_MIN sentinel key."""
    from spark_rapids_jni_amd.ops.aggregate import Agg, groupby
    n = 20000
    keys = [random.choice([-2**63, -2**63 + 1, 0, 7, 2**62, -5])
            if random.random() < 0.3 else random.randint(-10**9, 10**9)
            for _ in range(n)]
    vals = [random.randint(-1000, 1000) for _ in range(n)]
    kc = Column.from_pylist(keys, DType.INT64, "cuda")
    vc = Column.from_pylist(vals, DType.INT64, "cuda")
    assert kc.validity is None  # must select the fast path
    kt, res = groupby(kc, [(Agg.COUNT_ALL, None), (Agg.SUM, vc),
                           (Agg.MIN, vc), (Agg.MAX, vc)])
    got = {k: (c, s, mn, mx) for k, c, s, mn, mx in zip(
        kt.columns[0].to_pylist(), res[0].to_pylist(), res[1].to_pylist(),
        res[2].to_pylist(), res[3].to_pylist())}
    exp = {}
    for k, v in zip(keys, vals):
        if k not in exp:
            exp[k] = [0, 0, v, v]
        e = exp[k]
        e[0] += 1
        e[1] += v
        e[2] = min(e[2], v)
        e[3] = max(e[3], v)
    assert len(got) == len(exp)
    for k, e in exp.items():
        assert got[k] == tuple(e), f"key {k}"


@pytest.mark.gpu
def test_groupby_i64_fast_path_large_sparse():
    """Mostly-unique keys stress the insert/CAS path of the Slot64 table."""
    from spark_rapids_jni_amd.ops.aggregate import Agg, groupby
    n = 100_000
    keys = torch.randint(-2**62, 2**62, (n,), dtype=torch.int64,
                         device="cuda")
    kc = Column(DType.INT64, n, keys)
    kt, res = groupby(kc, [(Agg.COUNT_ALL, None)])
    total = int(res[0].data.sum().item())
    assert total == n
    import collections
    cnt = collections.Counter(keys.cpu().tolist())
    got = dict(zip(kt.columns[0].to_pylist(), res[0].to_pylist()))
    assert got == dict(cnt)


@pytest.mark.gpu
def test_partitioned_probe_matches_direct():
    """Force the radix-partitioned probe path (large capacity via a big build)
    on a small validation: results must equal the direct probe as multisets."""
    from spark_rapids_jni_amd.ops.join import HashJoinTable
    from spark_rapids_jni_amd import _native
    g = _native.gpu()
    n = 200_000
    keys = torch.randint(0, n // 2, (n,), dtype=torch.int64, device="cuda")
    bcol = Column(DType.INT64, n, keys)
    tbl = HashJoinTable.build(bcol)
    probe = torch.randint(0, n, (n,), dtype=torch.int64, device="cuda")
    pcol = Column(DType.INT64, n, probe)
    bi, pi = tbl.inner_join(pcol)
    # partition manually and probe with idxmap — must match the direct result
    pbits = 4
    np_ = 1 << pbits
    stream = _native.current_stream()
    hist = torch.zeros(np_, dtype=torch.int64, device="cuda")
    g.part_hist(probe.data_ptr(), n, pbits, hist.data_ptr(), stream)
    assert int(hist.sum().item()) == n
    cursors = torch.zeros(np_, dtype=torch.int64, device="cuda")
    torch.cumsum(hist[:np_ - 1], 0, out=cursors[1:])
    part_keys = torch.empty(n, dtype=torch.int64, device="cuda")
    part_idx = torch.empty(n, dtype=torch.int32, device="cuda")
    g.part_scatter(probe.data_ptr(), n, pbits, cursors.data_ptr(),
                   part_keys.data_ptr(), part_idx.data_ptr(), stream)
    # partitioned keys are a permutation of the input
    assert torch.equal(part_keys.sort().values, probe.sort().values)
    assert torch.equal(probe[part_idx.long()], part_keys)
    counter = torch.zeros(1, dtype=torch.int64, device="cuda")
    out_b = torch.empty(bi.numel(), dtype=torch.int32, device="cuda")
    out_p = torch.empty(bi.numel(), dtype=torch.int64, device="cuda")
    g.join_probe_i64(part_keys.data_ptr(), 0, n, tbl.slots.data_ptr(),
                     tbl.capacity, counter.data_ptr(), out_b.data_ptr(),
                     out_p.data_ptr(), bi.numel(), 0, 1, part_idx.data_ptr(),
                     stream)
    assert int(counter.item()) == bi.numel()
    direct = sorted(zip(pi.cpu().tolist(), bi.cpu().tolist()))
    parted = sorted(zip(out_p.cpu().tolist(), out_b.cpu().tolist()))
    assert direct == parted


@pytest.mark.gpu
def test_int32_keys_take_fast_path():
    """int32 (NDS surrogate-key dtype) joins route through the specialized
    table via an int64 upcast and agree with the generic path."""
    from spark_rapids_jni_amd.ops.join import HashJoinTable
    n = 40_000
    bk = torch.randint(-2**31, 2**31 - 1, (n,), dtype=torch.int32,
                       device="cuda")
    pk = torch.cat([bk[:n // 2], torch.randint(-2**31, 2**31 - 1, (n,),
                                               dtype=torch.int32,
                                               device="cuda")])
    bcol = Column.from_torch(bk)
    pcol = Column.from_torch(pk)
    tbl = HashJoinTable.build(bcol)
    assert tbl.i64_fast
    bi, pi = tbl.inner_join(pcol)
    gtbl = HashJoinTable.build(bcol, force_generic=True)
    gbi, gpi = gtbl.inner_join(pcol)
    assert sorted(zip(pi.cpu().tolist(), bi.cpu().tolist())) == \
        sorted(zip(gpi.cpu().tolist(), gbi.cpu().tolist()))


@pytest.mark.gpu
def test_groupby_int32_key_fast_path():
    from spark_rapids_jni_amd.ops.aggregate import Agg, groupby
    n = 30_000
    keys = torch.randint(-1000, 1000, (n,), dtype=torch.int32, device="cuda")
    vals = torch.randint(0, 100, (n,), dtype=torch.int64, device="cuda")
    kt, res = groupby(Column.from_torch(keys),
                      [(Agg.COUNT_ALL, None),
                       (Agg.SUM, Column.from_torch(vals))])
    import collections
    exp_c = collections.Counter(keys.cpu().tolist())
    exp_s = collections.defaultdict(int)
    for k, v in zip(keys.cpu().tolist(), vals.cpu().tolist()):
        exp_s[k] += v
    got = {k: (c, s) for k, c, s in zip(kt.columns[0].to_pylist(),
                                        res[0].to_pylist(),
                                        res[1].to_pylist())}
    assert got == {k: (exp_c[k], exp_s[k]) for k in exp_c}


@pytest.mark.gpu
def test_groupby_lds_low_cardinality():
    """hinted low-cardinality group-by takes the LDS pre-aggregation kernel;
    results must match the host oracle exactly (incl. min/max/nullable sum
    and hint-exceeding cardinality falling through to the global path)."""
    from spark_rapids_jni_amd.ops.aggregate import Agg, groupby
    n = 200_000
    for true_groups, hint in ((50, 64), (900, 1000), (5000, 64)):
        keys = torch.randint(0, true_groups, (n,), dtype=torch.int64,
                             device="cuda")
        vals = [None if i % 13 == 5 else (i % 2001) - 1000 for i in range(n)]
        vc = Column.from_pylist(vals, DType.INT64, "cuda")
        kt, res = groupby(Column.from_torch(keys),
                          [(Agg.COUNT_ALL, None), (Agg.SUM, vc),
                           (Agg.MIN, vc)], num_groups_hint=hint)
        import collections
        exp = collections.defaultdict(lambda: [0, None, None])
        for k, v in zip(keys.cpu().tolist(), vals):
            e = exp[k]
            e[0] += 1
            if v is not None:
                e[1] = v if e[1] is None else e[1] + v
                e[2] = v if e[2] is None else min(e[2], v)
        got = {k: (c, s, mn) for k, c, s, mn in zip(
            kt.columns[0].to_pylist(), res[0].to_pylist(),
            res[1].to_pylist(), res[2].to_pylist())}
        assert len(got) == len(exp), (true_groups, hint)
        for k, e in exp.items():
            assert got[k] == tuple(e), (true_groups, hint, k)


@pytest.mark.gpu
def test_groupby_multicol_packed_keys():
    """Narrow multi-int keys pack into one int64 (null-safe) and take the
    specialized path; results must equal the generic-path oracle exactly."""
    from spark_rapids_jni_amd.ops.aggregate import Agg, groupby
    n = 50_000
    k1 = [None if i % 19 == 7 else i % 7 for i in range(n)]
    k2 = [(i * 31) % 1000 - 500 for i in range(n)]
    vals = [i % 100 for i in range(n)]
    c1 = Column.from_pylist(k1, DType.INT32, "cuda")
    c2 = Column.from_pylist(k2, DType.INT64, "cuda")
    vc = Column.from_pylist(vals, DType.INT64, "cuda")
    kt, res = groupby(Table([c1, c2]), [(Agg.COUNT_ALL, None), (Agg.SUM, vc)])
    import collections
    exp = collections.defaultdict(lambda: [0, 0])
    for a, b, v in zip(k1, k2, vals):
        exp[(a, b)][0] += 1
        exp[(a, b)][1] += v
    got = {}
    g1 = kt.columns[0].to_pylist()
    g2 = kt.columns[1].to_pylist()
    for i, (c, sm) in enumerate(zip(res[0].to_pylist(), res[1].to_pylist())):
        got[(g1[i], g2[i])] = [c, sm]
    assert got == dict(exp)


@pytest.mark.gpu
def test_multikey_packed_join():
    """Multi-int join keys pack into one int64; results must equal the
    generic path, including null keys (never match) and probe values
    outside the build-side range."""
    from spark_rapids_jni_amd.ops.join import HashJoinTable
    n = 30_000
    b1 = [i % 500 for i in range(n)]
    b2 = [None if i % 23 == 9 else (i * 7) % 40 - 20 for i in range(n)]
    # first half mirrors build rows (guaranteed matches), second half random
    # with values outside the build range
    p1 = b1[:n // 2] + [(i * 3) % 700 for i in range(n // 2)]
    p2 = b2[:n // 2] + [None if i % 17 == 3 else (i * 5) % 44 - 22
                        for i in range(n // 2)]
    bt = Table([Column.from_pylist(b1, DType.INT32, "cuda"),
                Column.from_pylist(b2, DType.INT64, "cuda")])
    pt = Table([Column.from_pylist(p1, DType.INT32, "cuda"),
                Column.from_pylist(p2, DType.INT64, "cuda")])
    tbl = HashJoinTable.build(bt)
    assert tbl.i64_fast and getattr(tbl, "_pack", None) is not None
    bi, pi = tbl.inner_join(pt)
    gt = HashJoinTable.build(bt, force_generic=True)
    gbi, gpi = gt.inner_join(pt)
    assert sorted(zip(pi.cpu().tolist(), bi.cpu().tolist())) == \
        sorted(zip(gpi.cpu().tolist(), gbi.cpu().tolist()))
    assert bi.numel() > 0


@pytest.mark.gpu
def test_groupby_bad_hint_recovers():
    """A cardinality hint far below the true group count must not hang or
    miscount — both specialized and generic paths flag overflow and re-run."""
    from spark_rapids_jni_amd.ops.aggregate import Agg, groupby
    n = 300_000
    # int64 path
    keys = torch.randint(0, 50_000, (n,), dtype=torch.int64, device="cuda")
    kt, res = groupby(Column.from_torch(keys), [(Agg.COUNT_ALL, None)],
                      num_groups_hint=16)
    assert int(res[0].data.sum().item()) == n
    # generic (string-keyed) path
    svals = [f"k{i % 20000}" for i in range(n)]
    sc = Column.from_pylist(svals, DType.STRING, "cuda")
    kt2, res2 = groupby(sc, [(Agg.COUNT_ALL, None)], num_groups_hint=16)
    assert int(res2[0].data.sum().item()) == n
    assert kt2.num_rows == 20000


@pytest.mark.gpu
def test_groupby_multicol_wide_range_generic():
    """Key ranges whose product overflows the int64 packing domain must stay
    on the generic representative-row path and still be exact."""
    from spark_rapids_jni_amd.ops.aggregate import Agg, groupby
    n = 20_000
    k1 = torch.randint(-2**60, 2**60, (n,), dtype=torch.int64,
                       device="cuda") % 97
    k2 = torch.randint(-2**60, 2**60, (n,), dtype=torch.int64, device="cuda")
    # k2 spans ~2^61 values -> range product overflows -> generic path
    kt, res = groupby(Table([Column.from_torch(k1), Column.from_torch(k2)]),
                      [(Agg.COUNT_ALL, None)])
    assert int(res[0].data.sum().item()) == n
    import collections
    exp = collections.Counter(zip(k1.cpu().tolist(), k2.cpu().tolist()))
    got = collections.Counter()
    g1 = kt.columns[0].to_pylist()
    g2 = kt.columns[1].to_pylist()
    for i, c in enumerate(res[0].to_pylist()):
        got[(g1[i], g2[i])] = c
    assert got == exp


@pytest.mark.gpu
def test_empty_inputs_all_paths():
    """0-row build/probe/group-by inputs on every dispatch path."""
    from spark_rapids_jni_amd.ops.aggregate import Agg, groupby
    from spark_rapids_jni_amd.ops.join import HashJoinTable
    e64 = Column.from_torch(torch.empty(0, dtype=torch.int64, device="cuda"))
    e32 = Column.from_torch(torch.empty(0, dtype=torch.int32, device="cuda"))
    tbl = HashJoinTable.build(e64)
    bi, pi = tbl.inner_join(Column.from_torch(
        torch.arange(10, dtype=torch.int64, device="cuda")))
    assert bi.numel() == 0
    tbl2 = HashJoinTable.build(Table([e32, e64]))
    bi2, pi2 = tbl2.inner_join(Table([e32, e64]))
    assert bi2.numel() == 0
    kt, res = groupby(Table([e32, e64]), [(Agg.COUNT_ALL, None)])
    assert kt.num_rows == 0
    kt2, res2 = groupby(e64, [(Agg.COUNT_ALL, None)], num_groups_hint=8)
    assert kt2.num_rows == 0


@pytest.mark.gpu
def test_multikey_semi_anti_join():
    # r2 regression: the generic-table rebuild inside semi_join hashed only
    # the first key column, silently missing every multi-key match
    import random as _r
    from spark_rapids_jni_amd.ops.join import HashJoinTable
    _r.seed(7)
    n_b, n_p = 500, 2000
    bk1 = [_r.randint(0, 50) for _ in range(n_b)]
    bk2 = [_r.randint(0, 50) for _ in range(n_b)]
    pk1 = [_r.randint(0, 60) for _ in range(n_p)]
    pk2 = [_r.randint(0, 60) for _ in range(n_p)]
    bset = set(zip(bk1, bk2))
    exp_semi = sorted(i for i in range(n_p) if (pk1[i], pk2[i]) in bset)
    b = [Column.from_pylist(bk1, DType.INT64, device="cuda"),
         Column.from_pylist(bk2, DType.INT64, device="cuda")]
    p = [Column.from_pylist(pk1, DType.INT64, device="cuda"),
         Column.from_pylist(pk2, DType.INT64, device="cuda")]
    tbl = HashJoinTable.build(b)
    semi = sorted(tbl.semi_join(p).cpu().tolist())
    anti = sorted(tbl.semi_join(p, anti=True).cpu().tolist())
    assert semi == exp_semi
    assert anti == sorted(set(range(n_p)) - set(exp_semi))


@pytest.mark.gpu
def test_mark_matches_small_build_huge_probe():
    # right-semi building block: matched flags must be set even with a
    # capacity-0 (mark-only) probe
    from spark_rapids_jni_amd.ops.join import HashJoinTable
    import random as _r
    _r.seed(11)
    build_vals = list(range(0, 200, 2))  # evens < 200
    probe_vals = [_r.randint(0, 400) for _ in range(500_000)]
    b = Column.from_pylist(build_vals, DType.INT64, device="cuda")
    p = Column.from_pylist(probe_vals, DType.INT64, device="cuda")
    tbl = HashJoinTable.build(b)
    m = tbl.mark_matches(p).cpu().tolist()
    pset = set(probe_vals)
    exp = [1 if v in pset else 0 for v in build_vals]
    assert m == exp


@pytest.mark.gpu
def test_global_agg_chunked_lds():
    # the NDS engine computes global aggregates via 1-group LDS passes;
    # verify sum/avg/min/max/count against a plain torch fp64 reference
    import torch as _t
    from spark_rapids_jni_amd.nds.plan import Engine, Frame, Val
    from spark_rapids_jni_amd.nds.plan import Agg as NAgg
    from spark_rapids_jni_amd.nds.expr import col
    n = 1_000_000
    g = _t.Generator(device="cuda").manual_seed(5)
    vals = _t.rand(n, dtype=_t.float64, device="cuda", generator=g) * 100
    valid = _t.rand(n, device="cuda", generator=g) > 0.1
    iv = _t.randint(0, 1000, (n,), dtype=_t.int64, device="cuda",
                    generator=g)
    f = Frame({"x": Val(vals, valid), "i": Val(iv)}, n)
    e = Engine({}, device="cuda")
    e.register("t", f)
    from spark_rapids_jni_amd.nds.plan import Scan
    out = e.run(NAgg(Scan("t"), [], [
        ("s", "sum", col("x")), ("a", "avg", col("x")),
        ("mn", "min", col("x")), ("mx", "max", col("x")),
        ("c", "count", col("x")), ("si", "sum", col("i")),
        ("call", "count", None)]))
    row = out.to_rows()[0]
    xs = vals[valid]
    exp = (float(xs.sum()), float(xs.mean()), float(xs.min()),
           float(xs.max()), int(valid.sum()), int(iv.sum()), n)
    for got, want in zip(row, exp):
        if isinstance(want, float):
            assert abs(got - want) / max(abs(want), 1) < 1e-9, (row, exp)
        else:
            assert got == want, (row, exp)
