"""NDS query-shape pipelines (q1/q9/q95-like) vs host oracles."""
import pytest
import torch

from spark_rapids_jni_amd.columnar import Column, DType, Table

pytestmark = pytest.mark.gpu


def test_q9_like_buckets():
    from spark_rapids_jni_amd.exec import gen_store_sales, q9_like
    ss = gen_store_sales(200_000, device="cuda")
    out = q9_like(ss, [20, 40, 60, 80, 100])
    qty = ss.columns[2].data.cpu()
    price = ss.columns[3].data.cpu()
    lo = 0
    for (cnt, avg), hi in zip(out, [20, 40, 60, 80, 100]):
        m = (qty > lo) & (qty <= hi)
        assert cnt == int(m.sum())
        if cnt:
            assert abs(avg - float(price[m].mean())) < 1e-9
        lo = hi
    assert sum(c for c, _ in out) == 200_000


def test_q95_like_semi_anti():
    from spark_rapids_jni_amd.exec import q95_like
    n = 50_000
    g = torch.Generator(device="cuda")
    g.manual_seed(11)
    orders = torch.randint(0, 20_000, (n,), dtype=torch.int64, device="cuda",
                           generator=g)
    ship = torch.randint(0, 100, (n,), dtype=torch.int64, device="cuda",
                         generator=g)
    profit = torch.rand(n, dtype=torch.float64, device="cuda", generator=g)
    ws = Table([Column.from_torch(orders), Column.from_torch(ship),
                Column(DType.FLOAT64, n, profit)])
    ret = Column.from_torch(torch.arange(0, 5000, dtype=torch.int64,
                                         device="cuda"))
    window = Column.from_torch(torch.arange(10, 30, dtype=torch.int64,
                                            device="cuda"))
    semi_n, anti_n, semi_sum = q95_like(ws, ret, window)
    oc = orders.cpu()
    sc = ship.cpu()
    pc = profit.cpu()
    in_win = (sc >= 10) & (sc < 30)
    in_ret = oc < 5000
    assert semi_n == int((in_win & in_ret).sum())
    assert anti_n == int((in_win & ~in_ret).sum())
    assert abs(semi_sum - float(pc[in_win & in_ret].sum())) < 1e-6


def test_q1_like_having():
    from spark_rapids_jni_amd.exec import q1_like
    n = 30_000
    g = torch.Generator(device="cuda")
    g.manual_seed(5)
    cust = torch.randint(0, 3000, (n,), dtype=torch.int64, device="cuda",
                         generator=g)
    store = torch.randint(0, 10, (n,), dtype=torch.int64, device="cuda",
                          generator=g)
    amt = torch.randint(1, 100, (n,), dtype=torch.int64, device="cuda",
                        generator=g)
    sr = Table([Column.from_torch(cust), Column.from_torch(store),
                Column.from_torch(amt)])
    got = sorted(q1_like(sr).to_pylist())
    import collections
    totals = collections.defaultdict(int)
    for c, s, a in zip(cust.cpu().tolist(), store.cpu().tolist(),
                       amt.cpu().tolist()):
        totals[(c, s)] += a
    by_store = collections.defaultdict(list)
    for (c, s), t in totals.items():
        by_store[s].append(t)
    exp = sorted(c for (c, s), t in totals.items()
                 if t > 1.2 * (sum(by_store[s]) / len(by_store[s])))
    assert got == exp


def test_q14_distinct_like():
    from spark_rapids_jni_amd.exec import gen_store_sales, q14_distinct_like
    ss = gen_store_sales(500_000, n_items=18000, device="cuda")
    approx, exact = q14_distinct_like(ss)
    assert exact <= 18000
    assert abs(approx - exact) / exact < 0.05
