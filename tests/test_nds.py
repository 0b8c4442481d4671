"""NDS harness tests.

CPU tier: every one of the 99 queries executes on the oracle backend and a
few have hand-checked result invariants; the distributed (world=2, gloo)
run of representative queries matches the world=1 run exactly.

GPU tier (@pytest.mark.gpu): all 99 queries cross-checked GPU kernels vs
CPU oracle at SF 0.05 (VERDICT r01 item 1 acceptance criterion).
"""
import os
import subprocess
import sys

import pytest
import torch

from spark_rapids_jni_amd.nds import gen_catalog
from spark_rapids_jni_amd.nds.plan import CpuBackend, Engine
from spark_rapids_jni_amd.nds.queries import QUERIES
from spark_rapids_jni_amd.nds.runner import (compare_frames,
                                             frame_sorted_rows, power_run,
                                             verify)

SF = 0.01


@pytest.fixture(scope="module")
def catalog():
    return gen_catalog(sf=SF)


def test_all_99_queries_defined():
    assert sorted(QUERIES) == list(range(1, 100))


@pytest.mark.parametrize("n", sorted(QUERIES))
def test_query_runs_cpu(catalog, n):
    e = Engine(catalog, device="cpu")
    f = QUERIES[n](e)
    assert f.nrows >= 0
    # materialization must succeed (decodes dictionaries, applies nulls)
    rows = f.to_rows()
    assert len(rows) == f.nrows


def test_q3_invariants(catalog):
    e = Engine(catalog, device="cpu")
    f = QUERIES[3](e)
    rows = f.to_rows()
    assert f.names() == ["d_year", "i_brand", "i_brand_id", "sum_agg"]
    # all brands must come from manufacturer 1 items sold in November
    sums = [r[3] for r in rows]
    assert all(s is None or s > 0 for s in sums)
    # sorted by year then sum desc
    years = [r[0] for r in rows]
    assert years == sorted(years)


def test_q96_count_matches_manual(catalog):
    e = Engine(catalog, device="cpu")
    f = QUERIES[96](e)
    (cnt,) = f.to_rows()[0]
    # manual recompute with numpy
    import numpy as np
    ss = catalog["store_sales"]
    hd = catalog["household_demographics"]
    st = catalog["store"]
    tsk = ss.columns["ss_sold_time_sk"]
    tv = ss.valid["ss_sold_time_sk"]
    hour = tsk // 3600
    minute = (tsk % 3600) // 60
    time_ok = (hour == 20) & (minute >= 30)
    if tv is not None:
        time_ok &= tv
    dep7 = set(np.nonzero(hd.columns["hd_dep_count"] == 7)[0] + 1)
    names = st.dicts["s_store_name"]
    stores = {i + 1 for i, c in enumerate(st.columns["s_store_name"])
              if names[c] == "store_a"}
    hds = ss.columns["ss_hdemo_sk"]
    hdv = ss.valid["ss_hdemo_sk"]
    sts = ss.columns["ss_store_sk"]
    stv = ss.valid["ss_store_sk"]
    manual = 0
    for i in np.nonzero(time_ok)[0]:
        if hdv is not None and not hdv[i]:
            continue
        if stv is not None and not stv[i]:
            continue
        if int(hds[i]) in dep7 and int(sts[i]) in stores:
            manual += 1
    assert cnt == manual


def test_sort_limit_deterministic(catalog):
    # two engines over the same catalog produce identical Limit cuts
    a = QUERIES[42](Engine(catalog, device="cpu"))
    b = QUERIES[42](Engine(catalog, device="cpu"))
    assert frame_sorted_rows(a) == frame_sorted_rows(b)


def test_power_run_subset(catalog):
    e = Engine(catalog, device="cpu")
    out = power_run(e, queries=[3, 7, 42], quiet=True)
    assert out["n_queries"] == 3
    assert all(q["seconds"] >= 0 for q in out["queries"].values())


# --- distributed (world=2, gloo) -------------------------------------------

_DIST_QUERIES = [2, 3, 6, 15, 16, 28, 38, 51, 59, 67, 78, 95, 97]

_WORKER = r"""
import os, sys, json
import torch
import torch.distributed as dist
sys.path.insert(0, {repo!r})
from spark_rapids_jni_amd.nds import gen_catalog
from spark_rapids_jni_amd.nds.plan import Engine
from spark_rapids_jni_amd.nds.queries import QUERIES
from spark_rapids_jni_amd.nds.runner import frame_sorted_rows

rank = int(os.environ["RANK"])
dist.init_process_group("gloo")
world = dist.get_world_size()
cat = gen_catalog(sf={sf}, world=world, rank=rank)
out = {{}}
for n in {queries}:
    e = Engine(cat, device="cpu", world=world, rank=rank)
    f = QUERIES[n](e)
    out[n] = frame_sorted_rows(f)
if rank == 0:
    with open({out!r}, "w") as fh:
        json.dump({{str(k): [[repr(x) for x in row] for row in v]
                   for k, v in out.items()}}, fh)
dist.barrier()
dist.destroy_process_group()
"""


def test_distributed_matches_single(tmp_path, catalog):
    """world=2 gloo run (sharded facts + exchange/partial-agg paths) must
    reproduce the world=1 result for representative queries."""
    import json
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = str(tmp_path / "dist.json")
    script = tmp_path / "worker.py"
    script.write_text(_WORKER.format(repo=repo, sf=SF,
                                     queries=_DIST_QUERIES, out=out))
    env = dict(os.environ)
    env.update({"MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29517",
                "WORLD_SIZE": "2"})
    procs = []
    for r in range(2):
        env_r = dict(env)
        env_r["RANK"] = str(r)
        procs.append(subprocess.Popen(
            [sys.executable, str(script)], env=env_r,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE))
    for p in procs:
        so, se = p.communicate(timeout=600)
        assert p.returncode == 0, se.decode()[-3000:]
    with open(out) as fh:
        dist_rows = json.load(fh)
    for n in _DIST_QUERIES:
        e = Engine(catalog, device="cpu")
        f = QUERIES[n](e)
        single = [[repr(x) for x in row] for row in frame_sorted_rows(f)]
        got = dist_rows[str(n)]
        assert got == single, \
            f"q{n}: world=2 differs from world=1 " \
            f"({len(got)} vs {len(single)} rows)"


# --- GPU tier ---------------------------------------------------------------

@pytest.mark.gpu
def test_gpu_matches_cpu_oracle_all_queries():
    failures = verify(sf=0.05)
    assert not failures, f"GPU/CPU mismatches: {failures}"


def test_rollup_cascade_matches_bruteforce(catalog):
    """The cascade re-aggregation in Engine._rollup (level L-1 from level
    L's result) must equal a brute-force per-level aggregation computed in
    plain Python — sums, counts, min/max, null keys included."""
    from collections import defaultdict
    from spark_rapids_jni_amd.nds.expr import col
    from spark_rapids_jni_amd.nds.queries import A, S

    e = Engine(catalog, device="cpu")
    plan = A(S("item", "i_category", "i_class", "i_current_price",
               "i_brand_id"),
             ["i_category", "i_class"],
             [("s", "sum", col("i_current_price")),
              ("mx", "max", col("i_brand_id")),
              ("c", "count", None)], rollup=True)
    f = e.run(plan)
    got = {}
    for cat_, cls, s, mx, c, lvl in f.to_rows():
        got[(lvl, cat_, cls)] = (None if s is None else round(s, 4), mx, c)

    base = e.run(S("item", "i_category", "i_class", "i_current_price",
                   "i_brand_id")).to_rows()
    exp = {}
    for lvl in (2, 1, 0):
        acc = defaultdict(lambda: [0.0, False, None, 0])
        for cat_, cls, price, brand in base:
            key = (lvl, cat_ if lvl >= 1 else None,
                   cls if lvl >= 2 else None)
            a = acc[key]
            if price is not None:
                a[0] += price
                a[1] = True
            if brand is not None:
                a[2] = brand if a[2] is None else max(a[2], brand)
            a[3] += 1
        for key, (s, any_s, mx, c) in acc.items():
            exp[key] = (round(s, 4) if any_s else None, mx, c)
    assert got == exp


def test_day_pivot_decomposition_equivalent(catalog):
    """day_pivot_sums (two-stage q2/q59 rewrite) must produce EXACTLY the
    single-stage sum(case_when(d_day_name=X)) aggregation it replaced."""
    from spark_rapids_jni_amd.nds.expr import case_when, col, lit
    from spark_rapids_jni_amd.nds.queries import A, J, S, day_pivot_sums

    days = ["Sunday", "Monday", "Tuesday", "Wednesday", "Thursday",
            "Friday", "Saturday"]
    child = J(S("store_sales", "ss_sold_date_sk", "ss_store_sk",
                "ss_sales_price"),
              S("date_dim", "d_date_sk", "d_week_seq", "d_day_name"),
              [("ss_sold_date_sk", "d_date_sk")])
    keys = ["d_week_seq", "ss_store_sk"]
    e1 = Engine(catalog, device="cpu")
    direct = e1.run(A(child, keys,
                      [(dy.lower()[:3] + "_sales", "sum",
                        case_when((col("d_day_name") == dy,
                                   col("ss_sales_price")),
                                  otherwise=lit(0.0))) for dy in days]))
    e2 = Engine(catalog, device="cpu")
    pivot = e2.run(day_pivot_sums(child, keys, col("ss_sales_price"), days))
    assert frame_sorted_rows(direct) == frame_sorted_rows(pivot)
    assert direct.nrows > 0


def test_q78_filter_pushdown_equivalent(catalog):
    """q78's d_year=2000 pushdown below the aggregate must not change the
    result: filtering AFTER the (year-keyed) aggregation gives the same
    rows as the pushed-down plan used by the query."""
    from spark_rapids_jni_amd.nds.expr import col
    from spark_rapids_jni_amd.nds.queries import A, F, J, S

    child = J(J(S("store_sales", "ss_sold_date_sk", "ss_item_sk",
                  "ss_customer_sk", "ss_quantity"),
                S("date_dim", "d_date_sk", "d_year"),
                [("ss_sold_date_sk", "d_date_sk")]),
              S("item", "i_item_sk"), [("ss_item_sk", "i_item_sk")])
    keys = ["d_year", "ss_item_sk", "ss_customer_sk"]
    aggs = [("q", "sum", col("ss_quantity"))]
    e1 = Engine(catalog, device="cpu")
    post = e1.run(F(A(child, keys, aggs), col("d_year") == 2000))
    e2 = Engine(catalog, device="cpu")
    pushed = e2.run(A(F(child, col("d_year") == 2000), keys, aggs))
    assert frame_sorted_rows(post) == frame_sorted_rows(pushed)
    assert post.nrows > 0


def test_q16_runtime_semifilter_equivalent(catalog):
    """q16's runtime semi-filter before the countd aggregate must not
    change which orders qualify as multi-warehouse."""
    from spark_rapids_jni_amd.nds.expr import col
    from spark_rapids_jni_amd.nds.queries import A, F, J, P, S, dsk

    lo, hi = dsk("2002-02-01"), dsk("2002-04-02")
    cs1 = F(S("catalog_sales", "cs_ship_date_sk", "cs_order_number"),
            col("cs_ship_date_sk").between(lo, hi))
    base = S("catalog_sales", "cs_order_number", "cs_warehouse_sk")
    agg = [("nwh", "countd", col("cs_warehouse_sk"))]

    def qualifying(plan_child, eng):
        f = eng.run(P(F(A(plan_child, ["cs_order_number"], agg),
                        col("nwh") > 1), ("ono", col("cs_order_number"))))
        return sorted(r[0] for r in f.to_rows())

    e1 = Engine(catalog, device="cpu")
    full = qualifying(base, e1)
    e2 = Engine(catalog, device="cpu")
    filtered = qualifying(
        J(base, P(cs1, ("fono", col("cs_order_number"))),
          [("cs_order_number", "fono")], how="semi"), e2)
    e3 = Engine(catalog, device="cpu")
    cs1_orders = {r[0] for r in
                  e3.run(P(cs1, ("o", col("cs_order_number")))).to_rows()}
    assert filtered == [o for o in full if o in cs1_orders]


def test_all_true_memo_and_drop():
    """_precheck_all_true resolves memos in one batch; _val_to_column drops
    all-true masks (keeps real ones) and caches the decision per Val."""
    from spark_rapids_jni_amd.nds.expr import Val
    from spark_rapids_jni_amd.nds.plan import (_precheck_all_true,
                                               _val_to_column)
    a = Val(torch.arange(10), torch.ones(10, dtype=torch.bool))
    b = Val(torch.arange(10), torch.tensor([True] * 9 + [False]))
    c = Val(torch.arange(10))  # no mask
    _precheck_all_true([a, b, c, None])
    assert a._all_true is True and b._all_true is False
    ca = _val_to_column(a)
    cb = _val_to_column(b)
    cc = _val_to_column(c)
    assert ca.validity is None          # all-true mask dropped
    assert cb.validity is not None      # real nulls preserved
    assert not cb.is_valid_host(9) and cb.is_valid_host(0)
    assert cc.validity is None
