"""Minimal columnar execution layer + NDS-shaped synthetic data.

The reference is a kernel library driven by the Spark plugin; the analogous
driver here is this module: enough operator glue (scan, filter, project,
join, aggregate, exchange) to run NDS-class query shapes end-to-end on
MI355X — used by bench.py --op q3 and the integration tests.
"""
from typing import List, Optional, Sequence

import torch

from . import parquet as pq
from .columnar import Column, DType, Table
from .ops import hashing
from .ops.aggregate import Agg, groupby
from .ops.copying import gather, gather_column
from .ops.join import HashJoinTable


def apply_boolean_mask(table: Table, mask: torch.Tensor) -> Table:
    """Keep rows where mask != 0 (Spark Filter)."""
    sel = torch.nonzero(mask, as_tuple=False).view(-1)
    return gather(table, sel)


def scan_parquet(path: str, columns=None, device="cuda") -> Table:
    return pq.read_table(path, columns=columns, device=device)


# --- synthetic TPC-DS-shaped generators (no network: random in the right
#     shapes/cardinalities; SURVEY.md §7 stage 8) ---------------------------

def gen_date_dim(device="cuda", start_year=1998, n_years=7) -> Table:
    n = n_years * 365
    d_date_sk = torch.arange(2450000, 2450000 + n, dtype=torch.int64,
                             device=device)
    d_year = (torch.arange(n, device=device) // 365 + start_year).to(torch.int64)
    d_moy = ((torch.arange(n, device=device) % 365) // 31 + 1).clamp(1, 12) \
        .to(torch.int64)
    return Table([Column.from_torch(d_date_sk), Column.from_torch(d_year),
                  Column.from_torch(d_moy)])


def gen_store_sales(n: int, n_items: int = 18000, n_dates: int = 7 * 365,
                    device="cuda", seed: int = 42) -> Table:
    g = torch.Generator(device=device)
    g.manual_seed(seed)
    sold_date_sk = torch.randint(2450000, 2450000 + n_dates, (n,),
                                 dtype=torch.int64, device=device, generator=g)
    item_sk = torch.randint(1, n_items + 1, (n,), dtype=torch.int64,
                            device=device, generator=g)
    qty = torch.randint(1, 100, (n,), dtype=torch.int64, device=device,
                        generator=g)
    price = torch.rand(n, dtype=torch.float64, device=device, generator=g) * 100
    return Table([Column.from_torch(sold_date_sk), Column.from_torch(item_sk),
                  Column.from_torch(qty),
                  Column(DType.FLOAT64, n, price)])


def gen_item(n_items: int = 18000, device="cuda", seed: int = 7) -> Table:
    g = torch.Generator(device=device)
    g.manual_seed(seed)
    i_item_sk = torch.arange(1, n_items + 1, dtype=torch.int64, device=device)
    i_brand_id = torch.randint(1, 1000, (n_items,), dtype=torch.int64,
                               device=device, generator=g)
    i_manufact_id = torch.randint(1, 100, (n_items,), dtype=torch.int64,
                                  device=device, generator=g)
    return Table([Column.from_torch(i_item_sk), Column.from_torch(i_brand_id),
                  Column.from_torch(i_manufact_id)])


def q3_like(store_sales: Table, date_dim: Table, item: Table, year: int,
            manufact_id: int):
    """NDS q3 shape: date filter -> join store_sales⋈date_dim ⋈ item with an
    item filter -> group by brand -> sum(price).

    store_sales: [sold_date_sk, item_sk, qty, price]
    date_dim:    [d_date_sk, d_year, d_moy]
    item:        [i_item_sk, i_brand_id, i_manufact_id]
    """
    # filter dims
    dd = apply_boolean_mask(date_dim, date_dim.columns[1].data == year)
    it = apply_boolean_mask(item, item.columns[2].data == manufact_id)
    # join: build on the (small) dims, probe the fact table. Dimension keys
    # are unique, so matches <= probe rows: pass out_hint to skip the COUNT
    # probe entirely (the exact-size retry path still guards correctness).
    nfact = store_sales.num_rows
    dd_tbl = HashJoinTable.build(dd.columns[0])
    bi, pi = dd_tbl.inner_join(store_sales.columns[0], out_hint=nfact)
    # only the columns the rest of the plan reads survive the gather
    ss1 = Table([gather_column(store_sales.columns[1], pi),
                 gather_column(store_sales.columns[3], pi)])
    it_tbl = HashJoinTable.build(it.columns[0])
    bi2, pi2 = it_tbl.inner_join(ss1.columns[0], out_hint=ss1.num_rows)
    price = gather_column(ss1.columns[1], pi2)
    brand = gather_column(it.columns[1], bi2.long())
    # brand cardinality is bounded by the filtered item rows (planner
    # statistic): a low hint engages the LDS pre-aggregation kernel
    keys, results = groupby(brand, [(Agg.SUM, price),
                                    (Agg.COUNT_ALL, None)],
                            num_groups_hint=min(it.num_rows, 1_000_000))
    return keys, results


def q9_like(store_sales: Table, buckets: Sequence[int]):
    """NDS q9 shape: bucketed conditional aggregates over one fact scan —
    CASE WHEN qty BETWEEN lo AND hi THEN ... per bucket, avg(price) within
    each quantity bucket (select_first_true_index + masked aggregates).

    Returns [(count, avg_price) per bucket].
    """
    from .ops.misc import select_first_true_index
    qty = store_sales.columns[2].data
    price = store_sales.columns[3]
    conds = []
    lo = 0
    for hi in buckets:
        conds.append(Column.from_torch(((qty > lo) & (qty <= hi)).to(torch.int8),
                                       dtype=DType.BOOL8))
        lo = hi
    bucket_idx = select_first_true_index(conds)
    out = []
    for b in range(len(buckets)):
        mask = bucket_idx.data == b
        cnt = int(mask.sum().item())
        s = float(price.data[mask].sum().item()) if cnt else 0.0
        out.append((cnt, s / cnt if cnt else None))
    return out


def q95_like(web_sales: Table, web_returns_order_sk: Column,
             ship_date_sk: Column):
    """NDS q95 shape: orders shipped in a date window whose order number
    also appears in returns (left SEMI join) and orders that never appear
    (ANTI join) — exercises the semi/anti gather-map algebra.

    web_sales: [order_number, ship_date_sk, net_profit]
    Returns (semi_count, anti_count, semi_profit_sum).
    """
    dd_tbl = HashJoinTable.build(ship_date_sk)
    sel = dd_tbl.semi_join(web_sales.columns[1])
    ws = gather(web_sales, sel)
    ret_tbl = HashJoinTable.build(web_returns_order_sk)
    semi = ret_tbl.semi_join(ws.columns[0])
    anti = ret_tbl.semi_join(ws.columns[0], anti=True)
    profit = gather_column(ws.columns[2], semi)
    return semi.numel(), anti.numel(), float(profit.data.sum().item())


def q1_like(store_returns: Table, avg_factor: float = 1.2):
    """NDS q1 shape: per (customer, store) return totals, HAVING
    total > avg_factor * store average (group-by -> second group-by ->
    broadcast join back -> filter).

    store_returns: [customer_sk, store_sk, return_amt]
    Returns the customer_sk column of qualifying rows.
    """
    cust, store, amt = store_returns.columns
    pair_keys, pair_res = groupby(Table([cust, store]), [(Agg.SUM, amt)])
    totals = pair_res[0]
    # per-store average of the per-customer totals
    store_keys, store_res = groupby(pair_keys.columns[1],
                                    [(Agg.SUM, totals), (Agg.COUNT_ALL, None)])
    avg = store_res[0].data.to(torch.float64) / store_res[1].data.clamp(min=1)
    st_tbl = HashJoinTable.build(store_keys.columns[0])
    bi, pi = st_tbl.inner_join(pair_keys.columns[1])
    thresh = avg[bi.long()] * avg_factor
    tot = totals.data.to(torch.float64)[pi]
    qual = pi[tot > thresh]
    return gather_column(pair_keys.columns[0], qual)


def q14_distinct_like(store_sales: Table, precision: int = 12):
    """NDS q14/q38-family shape: approximate distinct counting of item keys
    via the Spark-compatible HLL++ sketch (ops/sketch.py) alongside an exact
    group-by distinct for validation-sized inputs."""
    from .ops.sketch import HyperLogLogPlusPlus
    items = store_sales.columns[1]
    h = HyperLogLogPlusPlus(precision=precision,
                            device=str(items.data.device))
    h.update(items)
    approx = h.estimate()
    keys, _ = groupby(items, [(Agg.COUNT_ALL, None)])
    exact = keys.num_rows
    return approx, exact
