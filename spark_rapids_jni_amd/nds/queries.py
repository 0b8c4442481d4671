"""NDS q1-q99 expressed in the plan IR.

Faithful-shape translations of the TPC-DS query set (same tables, join
graphs, aggregation/window/set-operation structure per query); filter
literals are adapted to the synthetic domains in schema.py (documented
deltas: dictionary-code ordering stands in for string collation in ORDER
BY; money is float64). Each query is a function(Engine) -> Frame so WITH
clauses, scalar subqueries and multi-pass set ops compose naturally.
"""
from .expr import (col, lit, date_lit, case_when, coalesce, is_null,
                   is_not_null, abs_)
from .plan import (Scan, Filter, Project, Join, Agg, Window, Sort, Limit,
                   Union, Distinct, Engine, Frame)
from .schema import DATE_BASE_SK, DATE_START
import datetime


def dsk(s: str) -> int:
    """date_sk of 'YYYY-MM-DD' (date_sk is day-linear in the generator)."""
    y, m, d = map(int, s.split("-"))
    return DATE_BASE_SK + (datetime.date(y, m, d) - DATE_START).days


def S(t, *cols):
    return Scan(t, list(cols) if cols else None)


def F(child, cond):
    return Filter(child, cond)


def P(child, *outs, extend=False):
    return Project(child, list(outs), extend=extend)


def J(l, r, on, how="inner"):
    return Join(l, r, on, how)


def A(child, keys, aggs, rollup=False):
    return Agg(child, keys, aggs, rollup)


def day_pivot_sums(child, keys, val_expr, days):
    """sum(case_when(d_day_name = X, v)) pivot, decomposed: ONE pass over
    the fact input grouped by keys + d_day_name, then the 7 case-pivots
    over the group-count-sized intermediate (sum distributes over the
    finer partition, so results are exact)."""
    pre = A(child, keys + ["d_day_name"], [("__pv", "sum", val_expr)])
    aggs = [(dy.lower()[:3] + "_sales", "sum",
             case_when((col("d_day_name") == dy, col("__pv")),
                       otherwise=lit(0.0))) for dy in days]
    return A(pre, keys, aggs)


def dd(cond, *cols):
    want = set(cols) | {"d_date_sk"}
    return F(S("date_dim", *sorted(want)), cond)


# ---------------------------------------------------------------------------

def q1(e: Engine) -> Frame:
    ctr = A(J(S("store_returns", "sr_returned_date_sk", "sr_customer_sk",
              "sr_store_sk", "sr_return_amt"),
            dd(col("d_year") == 2000, "d_year"),
            [("sr_returned_date_sk", "d_date_sk")]),
            ["sr_customer_sk", "sr_store_sk"],
            [("ctr_total_return", "sum", col("sr_return_amt"))])
    e.register("ctr", e.run(ctr))
    avg_by_store = A(S("ctr"), ["sr_store_sk"],
                     [("avg_ret", "avg", col("ctr_total_return"))])
    p = J(S("ctr"),
          P(avg_by_store, ("st2", col("sr_store_sk")),
            ("avg_ret", col("avg_ret"))),
          [("sr_store_sk", "st2")])
    p = F(p, col("ctr_total_return") > col("avg_ret") * 1.2)
    p = J(p, F(S("store", "s_store_sk", "s_state"), col("s_state") == "CA"),
          [("sr_store_sk", "s_store_sk")])
    p = J(p, S("customer", "c_customer_sk", "c_customer_id"),
          [("sr_customer_sk", "c_customer_sk")])
    p = P(p, ("c_customer_id", col("c_customer_id")))
    return e.run(Limit(Sort(p, [("c_customer_id", True)]), 100))


def q2(e: Engine) -> Frame:
    wscs = Union([
        P(S("web_sales", "ws_sold_date_sk", "ws_ext_sales_price"),
          ("sold_date_sk", col("ws_sold_date_sk")),
          ("sales_price", col("ws_ext_sales_price"))),
        P(S("catalog_sales", "cs_sold_date_sk", "cs_ext_sales_price"),
          ("sold_date_sk", col("cs_sold_date_sk")),
          ("sales_price", col("cs_ext_sales_price")))])
    joined = J(wscs, S("date_dim", "d_date_sk", "d_week_seq", "d_day_name",
                       "d_year"),
               [("sold_date_sk", "d_date_sk")])
    days = ["Sunday", "Monday", "Tuesday", "Wednesday", "Thursday",
            "Friday", "Saturday"]
    wswscs = day_pivot_sums(joined, ["d_week_seq", "d_year"],
                            col("sales_price"), days)
    e.register("wswscs", e.run(wswscs))
    y = P(F(S("wswscs"), col("d_year") == 2001),
          ("week1", col("d_week_seq")),
          *[(f"{d}1", col(d)) for d in
            [dy.lower()[:3] + "_sales" for dy in days]])
    z = P(F(S("wswscs"), col("d_year") == 2002),
          ("week2", col("d_week_seq") - 53),
          *[(f"{d}2", col(d)) for d in
            [dy.lower()[:3] + "_sales" for dy in days]])
    p = J(y, z, [("week1", "week2")])
    outs = [("d_week_seq1", col("week1"))]
    for dy in days:
        d = dy.lower()[:3] + "_sales"
        outs.append((d[:3] + "_ratio", col(d + "1") / col(d + "2")))
    p = P(p, *outs)
    return e.run(Sort(p, [("d_week_seq1", True)]))


def q3(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_item_sk",
            "ss_ext_sales_price"),
          dd((col("d_moy") == 11), "d_moy", "d_year"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("item", "i_item_sk", "i_brand", "i_brand_id",
                 "i_manufact_id"), col("i_manufact_id") == 1),
          [("ss_item_sk", "i_item_sk")])
    p = A(p, ["d_year", "i_brand", "i_brand_id"],
          [("sum_agg", "sum", col("ss_ext_sales_price"))])
    return e.run(Limit(Sort(p, [("d_year", True), ("sum_agg", False),
                                ("i_brand_id", True)]), 100))


def _year_total_channel(e, sales, prefix, cust_col, year, label):
    """q4/q11/q74 helper: per (customer, year) net revenue for one channel."""
    lp, wc, da, sp = (f"{prefix}_ext_list_price", f"{prefix}_ext_wholesale_cost",
                      f"{prefix}_ext_discount_amt", f"{prefix}_ext_sales_price")
    p = J(S(sales, f"{prefix}_sold_date_sk", cust_col, lp, wc, da, sp),
          dd(col("d_year") == year, "d_year"),
          [(f"{prefix}_sold_date_sk", "d_date_sk")])
    p = J(p, S("customer", "c_customer_sk", "c_customer_id"),
          [(cust_col, "c_customer_sk")])
    p = A(p, ["c_customer_id"],
          [("year_total", "sum",
            ((col(lp) - col(wc) - col(da)) + col(sp)) / 2.0)])
    return P(p, ("cid", col("c_customer_id")),
             ("year_total", col("year_total")))


def q4(e: Engine) -> Frame:
    parts = {}
    for ch, (tbl, pre, cc) in {
            "s": ("store_sales", "ss", "ss_customer_sk"),
            "c": ("catalog_sales", "cs", "cs_bill_customer_sk"),
            "w": ("web_sales", "ws", "ws_bill_customer_sk")}.items():
        for yr, tag in ((2001, "first"), (2002, "sec")):
            name = f"t_{ch}_{tag}"
            f = e.run(_year_total_channel(e, tbl, pre, cc, yr, ch))
            parts[name] = f
            e.register(name, f)
    p = J(P(F(S("t_s_first"), col("year_total") > 0),
            ("cid", col("cid")), ("ts1", col("year_total"))),
          P(S("t_s_sec"), ("cid2", col("cid")), ("ts2", col("year_total"))),
          [("cid", "cid2")])
    p = J(p, P(F(S("t_c_first"), col("year_total") > 0),
               ("cid3", col("cid")), ("tc1", col("year_total"))),
          [("cid", "cid3")])
    p = J(p, P(S("t_c_sec"), ("cid4", col("cid")), ("tc2", col("year_total"))),
          [("cid", "cid4")])
    p = J(p, P(F(S("t_w_first"), col("year_total") > 0),
               ("cid5", col("cid")), ("tw1", col("year_total"))),
          [("cid", "cid5")])
    p = J(p, P(S("t_w_sec"), ("cid6", col("cid")), ("tw2", col("year_total"))),
          [("cid", "cid6")])
    p = F(p, (col("tc2") / col("tc1") > col("ts2") / col("ts1")) &
          (col("tc2") / col("tc1") > col("tw2") / col("tw1")))
    p = P(p, ("customer_id", col("cid")))
    return e.run(Limit(Sort(p, [("customer_id", True)]), 100))


def q5(e: Engine) -> Frame:
    lo, hi = dsk("2000-08-23"), dsk("2000-09-06")
    in_win = lambda c: (col(c) >= lo) & (col(c) <= hi)
    ss = P(F(S("store_sales", "ss_store_sk", "ss_sold_date_sk",
              "ss_ext_sales_price", "ss_net_profit"),
             in_win("ss_sold_date_sk")),
           ("sk", col("ss_store_sk")), ("sales", col("ss_ext_sales_price")),
           ("ret", lit(0.0)), ("profit", col("ss_net_profit")),
           ("loss", lit(0.0)))
    sr = P(F(S("store_returns", "sr_store_sk", "sr_returned_date_sk",
              "sr_return_amt", "sr_net_loss"),
             in_win("sr_returned_date_sk")),
           ("sk", col("sr_store_sk")), ("sales", lit(0.0)),
           ("ret", col("sr_return_amt")), ("profit", lit(0.0)),
           ("loss", col("sr_net_loss")))
    ssr = J(A(Union([ss, sr]), ["sk"],
              [("sales", "sum", col("sales")), ("ret", "sum", col("ret")),
               ("profit", "sum", col("profit") - col("loss"))]),
            S("store", "s_store_sk", "s_store_id"), [("sk", "s_store_sk")])
    ssr = P(ssr, ("channel", lit("store channel")), ("id", col("s_store_id")),
            ("sales", col("sales")), ("returns_", col("ret")),
            ("profit", col("profit")))
    cs = P(F(S("catalog_sales", "cs_catalog_page_sk", "cs_sold_date_sk",
              "cs_ext_sales_price", "cs_net_profit"),
             in_win("cs_sold_date_sk")),
           ("sk", col("cs_catalog_page_sk")),
           ("sales", col("cs_ext_sales_price")), ("ret", lit(0.0)),
           ("profit", col("cs_net_profit")), ("loss", lit(0.0)))
    cr = P(F(S("catalog_returns", "cr_catalog_page_sk", "cr_returned_date_sk",
              "cr_return_amount", "cr_net_loss"),
             in_win("cr_returned_date_sk")),
           ("sk", col("cr_catalog_page_sk")), ("sales", lit(0.0)),
           ("ret", col("cr_return_amount")), ("profit", lit(0.0)),
           ("loss", col("cr_net_loss")))
    csr = J(A(Union([cs, cr]), ["sk"],
              [("sales", "sum", col("sales")), ("ret", "sum", col("ret")),
               ("profit", "sum", col("profit") - col("loss"))]),
            S("catalog_page", "cp_catalog_page_sk", "cp_catalog_page_id"),
            [("sk", "cp_catalog_page_sk")])
    csr = P(csr, ("channel", lit("catalog channel")),
            ("id", col("cp_catalog_page_id")), ("sales", col("sales")),
            ("returns_", col("ret")), ("profit", col("profit")))
    ws = P(F(S("web_sales", "ws_web_site_sk", "ws_sold_date_sk",
              "ws_ext_sales_price", "ws_net_profit"),
             in_win("ws_sold_date_sk")),
           ("sk", col("ws_web_site_sk")),
           ("sales", col("ws_ext_sales_price")), ("ret", lit(0.0)),
           ("profit", col("ws_net_profit")), ("loss", lit(0.0)))
    # wr joins back to ws for the site key (reference query shape)
    wrj = J(S("web_returns", "wr_item_sk", "wr_order_number",
              "wr_returned_date_sk", "wr_return_amt", "wr_net_loss"),
            S("web_sales", "ws_item_sk", "ws_order_number",
              "ws_web_site_sk"),
            [("wr_item_sk", "ws_item_sk"),
             ("wr_order_number", "ws_order_number")])
    wr = P(F(wrj, in_win("wr_returned_date_sk")),
           ("sk", col("ws_web_site_sk")), ("sales", lit(0.0)),
           ("ret", col("wr_return_amt")), ("profit", lit(0.0)),
           ("loss", col("wr_net_loss")))
    wsr = J(A(Union([ws, wr]), ["sk"],
              [("sales", "sum", col("sales")), ("ret", "sum", col("ret")),
               ("profit", "sum", col("profit") - col("loss"))]),
            S("web_site", "web_site_sk", "web_site_id"),
            [("sk", "web_site_sk")])
    wsr = P(wsr, ("channel", lit("web channel")), ("id", col("web_site_id")),
            ("sales", col("sales")), ("returns_", col("ret")),
            ("profit", col("profit")))
    p = A(Union([ssr, csr, wsr]), ["channel", "id"],
          [("sales", "sum", col("sales")),
           ("returns_", "sum", col("returns_")),
           ("profit", "sum", col("profit"))], rollup=True)
    return e.run(Limit(Sort(p, [("__lvl", True), ("channel", True),
                                ("id", True)]), 100))


def q6(e: Engine) -> Frame:
    mseq = e.scalar(Limit(Distinct(P(dd((col("d_year") == 2001) &
                                        (col("d_moy") == 1), "d_year",
                                        "d_moy", "d_month_seq"),
                                     ("d_month_seq", col("d_month_seq")))),
                          1))
    cat_avg = A(F(S("item", "i_item_sk", "i_category", "i_current_price"),
                  is_not_null(col("i_category"))),
                ["i_category"],
                [("cat_avg", "avg", col("i_current_price"))])
    it = J(S("item", "i_item_sk", "i_category", "i_current_price"),
           P(cat_avg, ("cat2", col("i_category")), ("cat_avg", col("cat_avg"))),
           [("i_category", "cat2")])
    it = F(it, col("i_current_price") > col("cat_avg") * 1.2)
    p = J(S("store_sales", "ss_sold_date_sk", "ss_customer_sk", "ss_item_sk"),
          dd(col("d_month_seq") == mseq, "d_month_seq"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, P(it, ("i_item_sk", col("i_item_sk"))),
          [("ss_item_sk", "i_item_sk")])
    p = J(p, S("customer", "c_customer_sk", "c_current_addr_sk"),
          [("ss_customer_sk", "c_customer_sk")])
    p = J(p, S("customer_address", "ca_address_sk", "ca_state"),
          [("c_current_addr_sk", "ca_address_sk")])
    p = F(A(p, ["ca_state"], [("cnt", "count", None)]), col("cnt") >= 10)
    return e.run(Limit(Sort(p, [("cnt", True), ("ca_state", True)]), 100))


def q7(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_item_sk", "ss_cdemo_sk",
            "ss_promo_sk", "ss_quantity", "ss_list_price", "ss_coupon_amt",
            "ss_sales_price"),
          dd(col("d_year") == 2000, "d_year"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("customer_demographics", "cd_demo_sk", "cd_gender",
                 "cd_marital_status", "cd_education_status"),
               (col("cd_gender") == "M") & (col("cd_marital_status") == "S") &
               (col("cd_education_status") == "College")),
          [("ss_cdemo_sk", "cd_demo_sk")])
    p = J(p, F(S("promotion", "p_promo_sk", "p_channel_email",
                 "p_channel_event"),
               (col("p_channel_email") == "N") |
               (col("p_channel_event") == "N")),
          [("ss_promo_sk", "p_promo_sk")])
    p = J(p, S("item", "i_item_sk", "i_item_id"),
          [("ss_item_sk", "i_item_sk")])
    p = A(p, ["i_item_id"],
          [("agg1", "avg", col("ss_quantity")),
           ("agg2", "avg", col("ss_list_price")),
           ("agg3", "avg", col("ss_coupon_amt")),
           ("agg4", "avg", col("ss_sales_price"))])
    return e.run(Limit(Sort(p, [("i_item_id", True)]), 100))


def q8(e: Engine) -> Frame:
    zips = [f"{10000 + 89 * k % 90000:05d}" for k in range(0, 400, 3)]
    v1 = Distinct(P(F(S("customer_address", "ca_address_sk", "ca_zip"),
                      col("ca_zip").substr(1, 5).isin(zips)),
                    ("zip5", col("ca_zip").substr(1, 5))))
    pref = J(S("customer_address", "ca_address_sk", "ca_zip"),
             F(S("customer", "c_current_addr_sk", "c_preferred_cust_flag"),
               col("c_preferred_cust_flag") == "Y"),
             [("ca_address_sk", "c_current_addr_sk")])
    v2 = P(F(A(P(pref, ("zip5", col("ca_zip").substr(1, 5))), ["zip5"],
               [("cnt", "count", None)]), col("cnt") > 10),
           ("zip5", col("zip5")))
    both = J(v1, P(v2, ("zip5b", col("zip5"))), [("zip5", "zip5b")],
             how="semi")
    zip2 = Distinct(P(both, ("zip2", col("zip5").substr(1, 2))))
    st = J(P(S("store", "s_store_sk", "s_store_name", "s_zip"),
             ("s_store_sk", col("s_store_sk")),
             ("s_store_name", col("s_store_name")),
             ("s_zip2", col("s_zip").substr(1, 2))),
           P(zip2, ("zip2b", col("zip2"))), [("s_zip2", "zip2b")],
           how="semi")
    p = J(S("store_sales", "ss_sold_date_sk", "ss_store_sk",
            "ss_net_profit"),
          dd((col("d_qoy") == 2) & (col("d_year") == 1998), "d_qoy",
             "d_year"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, st, [("ss_store_sk", "s_store_sk")])
    p = A(p, ["s_store_name"],
          [("net_profit", "sum", col("ss_net_profit"))])
    return e.run(Limit(Sort(p, [("s_store_name", True)]), 100))


def q9(e: Engine) -> Frame:
    buckets = [(1, 20), (21, 40), (41, 60), (61, 80), (81, 100)]
    out = {}
    for i, (lo, hi) in enumerate(buckets, 1):
        base = F(S("store_sales", "ss_quantity", "ss_ext_discount_amt",
                   "ss_net_paid"), col("ss_quantity").between(lo, hi))
        cnt = e.scalar(A(base, [], [("c", "count", None)]), "c")
        avg1 = e.scalar(A(base, [], [("a", "avg",
                                      col("ss_ext_discount_amt"))]), "a")
        avg2 = e.scalar(A(base, [], [("a", "avg", col("ss_net_paid"))]), "a")
        out[f"bucket{i}"] = avg1 if (cnt or 0) > 1000 else avg2
    return e.const_frame(**out)


def q10(e: Engine) -> Frame:
    counties = ["county_00", "county_01", "county_02", "county_03",
                "county_04"]
    c = J(S("customer", "c_customer_sk", "c_current_addr_sk",
            "c_current_cdemo_sk"),
          F(S("customer_address", "ca_address_sk", "ca_county"),
            col("ca_county").isin(counties)),
          [("c_current_addr_sk", "ca_address_sk")])
    win = dd((col("d_year") == 2002) & (col("d_moy").between(1, 4)),
             "d_year", "d_moy")
    ss = J(S("store_sales", "ss_customer_sk", "ss_sold_date_sk"), win,
           [("ss_sold_date_sk", "d_date_sk")])
    c = J(c, P(ss, ("sscust", col("ss_customer_sk"))),
          [("c_customer_sk", "sscust")], how="semi")
    ws = J(S("web_sales", "ws_bill_customer_sk", "ws_sold_date_sk"), win,
           [("ws_sold_date_sk", "d_date_sk")])
    cs = J(S("catalog_sales", "cs_ship_customer_sk", "cs_sold_date_sk"), win,
           [("cs_sold_date_sk", "d_date_sk")])
    either = Union([P(ws, ("cust", col("ws_bill_customer_sk"))),
                    P(cs, ("cust", col("cs_ship_customer_sk")))])
    c = J(c, either, [("c_customer_sk", "cust")], how="semi")
    p = J(c, S("customer_demographics", "cd_demo_sk", "cd_gender",
               "cd_marital_status", "cd_education_status",
               "cd_purchase_estimate", "cd_credit_rating", "cd_dep_count",
               "cd_dep_employed_count", "cd_dep_college_count"),
          [("c_current_cdemo_sk", "cd_demo_sk")])
    p = A(p, ["cd_gender", "cd_marital_status", "cd_education_status",
              "cd_purchase_estimate", "cd_credit_rating", "cd_dep_count",
              "cd_dep_employed_count", "cd_dep_college_count"],
          [("cnt1", "count", None)])
    return e.run(Limit(Sort(p, [("cd_gender", True),
                                ("cd_marital_status", True),
                                ("cd_education_status", True)]), 100))


def q11(e: Engine) -> Frame:
    for ch, (tbl, pre, cc) in {
            "s": ("store_sales", "ss", "ss_customer_sk"),
            "w": ("web_sales", "ws", "ws_bill_customer_sk")}.items():
        for yr, tag in ((2001, "first"), (2002, "sec")):
            e.register(f"t11_{ch}_{tag}",
                       e.run(_year_total_channel(e, tbl, pre, cc, yr, ch)))
    p = J(P(F(S("t11_s_first"), col("year_total") > 0),
            ("cid", col("cid")), ("ts1", col("year_total"))),
          P(S("t11_s_sec"), ("cid2", col("cid")), ("ts2", col("year_total"))),
          [("cid", "cid2")])
    p = J(p, P(F(S("t11_w_first"), col("year_total") > 0),
               ("cid3", col("cid")), ("tw1", col("year_total"))),
          [("cid", "cid3")])
    p = J(p, P(S("t11_w_sec"), ("cid4", col("cid")),
               ("tw2", col("year_total"))),
          [("cid", "cid4")])
    p = F(p, col("tw2") / col("tw1") > col("ts2") / col("ts1"))
    p = P(p, ("customer_id", col("cid")))
    return e.run(Limit(Sort(p, [("customer_id", True)]), 100))


def _ratio_by_class(e, sales, pre, date_lo, cats):
    """q12/q20/q98 shape: 30-day window, 3 categories, per-item revenue with
    ratio-to-class window."""
    lo, hi = dsk(date_lo), dsk(date_lo) + 30
    p = J(F(S(sales, f"{pre}_sold_date_sk", f"{pre}_item_sk",
              f"{pre}_ext_sales_price"),
            col(f"{pre}_sold_date_sk").between(lo, hi)),
          F(S("item", "i_item_sk", "i_item_id", "i_item_desc", "i_category",
              "i_class", "i_current_price"),
            col("i_category").isin(cats)),
          [(f"{pre}_item_sk", "i_item_sk")])
    p = A(p, ["i_item_id", "i_item_desc", "i_category", "i_class",
              "i_current_price"],
          [("itemrevenue", "sum", col(f"{pre}_ext_sales_price"))])
    p = Window(p, ["i_class"],
               [("classrev", "sum", "itemrevenue")])
    p = P(p, ("i_item_id", col("i_item_id")),
          ("i_item_desc", col("i_item_desc")),
          ("i_category", col("i_category")), ("i_class", col("i_class")),
          ("i_current_price", col("i_current_price")),
          ("itemrevenue", col("itemrevenue")),
          ("revenueratio", col("itemrevenue") * 100.0 / col("classrev")))
    return Limit(Sort(p, [("i_category", True), ("i_class", True),
                          ("i_item_id", True), ("i_item_desc", True),
                          ("revenueratio", True)]), 100)


def q12(e: Engine) -> Frame:
    return e.run(_ratio_by_class(e, "web_sales", "ws", "1999-02-22",
                                 ["Sports", "Books", "Home"]))


def q13(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_store_sk", "ss_cdemo_sk",
            "ss_hdemo_sk", "ss_addr_sk", "ss_sales_price", "ss_net_profit",
            "ss_quantity", "ss_ext_sales_price", "ss_ext_wholesale_cost"),
          dd(col("d_year") == 2001, "d_year"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, S("store", "s_store_sk"), [("ss_store_sk", "s_store_sk")])
    p = J(p, S("customer_demographics", "cd_demo_sk", "cd_marital_status",
               "cd_education_status"),
          [("ss_cdemo_sk", "cd_demo_sk")])
    p = J(p, S("household_demographics", "hd_demo_sk", "hd_dep_count"),
          [("ss_hdemo_sk", "hd_demo_sk")])
    p = J(p, S("customer_address", "ca_address_sk", "ca_country",
               "ca_state"),
          [("ss_addr_sk", "ca_address_sk")])
    b1 = ((col("cd_marital_status") == "M") &
          (col("cd_education_status") == "Advanced Degree") &
          col("ss_sales_price").between(100.0, 150.0) &
          (col("hd_dep_count") == 3))
    b2 = ((col("cd_marital_status") == "S") &
          (col("cd_education_status") == "College") &
          col("ss_sales_price").between(50.0, 100.0) &
          (col("hd_dep_count") == 1))
    b3 = ((col("cd_marital_status") == "W") &
          (col("cd_education_status") == "2 yr Degree") &
          col("ss_sales_price").between(150.0, 200.0) &
          (col("hd_dep_count") == 1))
    g1 = ((col("ca_country") == "United States") &
          col("ca_state").isin(["TX", "OH", "TX"]) &
          col("ss_net_profit").between(100, 200))
    g2 = ((col("ca_country") == "United States") &
          col("ca_state").isin(["OR", "NM", "KY"]) &
          col("ss_net_profit").between(150, 300))
    g3 = ((col("ca_country") == "United States") &
          col("ca_state").isin(["VA", "TX", "MS"]) &
          col("ss_net_profit").between(50, 250))
    p = F(p, (b1 | b2 | b3) & (g1 | g2 | g3))
    return e.run(A(p, [],
                   [("avg1", "avg", col("ss_quantity")),
                    ("avg2", "avg", col("ss_ext_sales_price")),
                    ("avg3", "avg", col("ss_ext_wholesale_cost")),
                    ("sum1", "sum", col("ss_ext_wholesale_cost"))]))


def q14(e: Engine) -> Frame:
    yrs = (col("d_year") >= 1999) & (col("d_year") <= 2001)
    triples = []
    for tbl, pre in (("store_sales", "ss"), ("catalog_sales", "cs"),
                     ("web_sales", "ws")):
        t = J(S(tbl, f"{pre}_sold_date_sk", f"{pre}_item_sk"),
              dd(yrs, "d_year"), [(f"{pre}_sold_date_sk", "d_date_sk")])
        t = J(t, S("item", "i_item_sk", "i_brand_id", "i_class_id",
                   "i_category_id"),
              [(f"{pre}_item_sk", "i_item_sk")])
        triples.append(Distinct(P(t, ("brand_id", col("i_brand_id")),
                                  ("class_id", col("i_class_id")),
                                  ("category_id", col("i_category_id")))))
    common = J(J(triples[0],
                 P(triples[1], ("b2", col("brand_id")),
                   ("c2", col("class_id")), ("g2", col("category_id"))),
                 [("brand_id", "b2"), ("class_id", "c2"),
                  ("category_id", "g2")], how="semi"),
               P(triples[2], ("b3", col("brand_id")),
                 ("c3", col("class_id")), ("g3", col("category_id"))),
               [("brand_id", "b3"), ("class_id", "c3"),
                ("category_id", "g3")], how="semi")
    cross_items = P(J(S("item", "i_item_sk", "i_brand_id", "i_class_id",
                        "i_category_id"), common,
                      [("i_brand_id", "brand_id"), ("i_class_id", "class_id"),
                       ("i_category_id", "category_id")], how="semi"),
                    ("item_sk", col("i_item_sk")))
    e.register("cross_items", e.run(cross_items))
    avg_parts = []
    for tbl, pre in (("store_sales", "ss"), ("catalog_sales", "cs"),
                     ("web_sales", "ws")):
        t = J(S(tbl, f"{pre}_sold_date_sk", f"{pre}_quantity",
                f"{pre}_list_price"),
              dd(yrs, "d_year"), [(f"{pre}_sold_date_sk", "d_date_sk")])
        avg_parts.append(P(t, ("q", col(f"{pre}_quantity")),
                           ("p", col(f"{pre}_list_price"))))
    avg_sales = e.scalar(A(Union(avg_parts), [],
                           [("a", "avg", col("q") * col("p"))]), "a")
    chans = []
    for tbl, pre, label in (("store_sales", "ss", "store"),
                            ("catalog_sales", "cs", "catalog"),
                            ("web_sales", "ws", "web")):
        t = J(S(tbl, f"{pre}_sold_date_sk", f"{pre}_item_sk",
                f"{pre}_quantity", f"{pre}_list_price"),
              dd((col("d_year") == 2001) & (col("d_moy") == 11), "d_year",
                 "d_moy"),
              [(f"{pre}_sold_date_sk", "d_date_sk")])
        t = J(t, S("cross_items"), [(f"{pre}_item_sk", "item_sk")],
              how="semi")
        t = J(t, S("item", "i_item_sk", "i_brand_id", "i_class_id",
                   "i_category_id"),
              [(f"{pre}_item_sk", "i_item_sk")])
        t = A(t, ["i_brand_id", "i_class_id", "i_category_id"],
              [("sales", "sum", col(f"{pre}_quantity") *
                col(f"{pre}_list_price")),
               ("number_sales", "count", None)])
        t = F(t, col("sales") > lit(float(avg_sales or 0.0)))
        chans.append(P(t, ("channel", lit(label)),
                       ("i_brand_id", col("i_brand_id")),
                       ("i_class_id", col("i_class_id")),
                       ("i_category_id", col("i_category_id")),
                       ("sales", col("sales")),
                       ("number_sales", col("number_sales"))))
    p = A(Union(chans), ["channel", "i_brand_id", "i_class_id",
                         "i_category_id"],
          [("sum_sales", "sum", col("sales")),
           ("sum_number_sales", "sum", col("number_sales"))], rollup=True)
    return e.run(Limit(Sort(p, [("__lvl", True), ("channel", True),
                                ("i_brand_id", True), ("i_class_id", True),
                                ("i_category_id", True)]), 100))


def q15(e: Engine) -> Frame:
    p = J(S("catalog_sales", "cs_sold_date_sk", "cs_bill_customer_sk",
            "cs_sales_price"),
          dd((col("d_qoy") == 2) & (col("d_year") == 2001), "d_qoy",
             "d_year"),
          [("cs_sold_date_sk", "d_date_sk")])
    p = J(p, S("customer", "c_customer_sk", "c_current_addr_sk"),
          [("cs_bill_customer_sk", "c_customer_sk")])
    p = J(p, S("customer_address", "ca_address_sk", "ca_zip", "ca_state"),
          [("c_current_addr_sk", "ca_address_sk")])
    p = F(p, col("ca_zip").substr(1, 5).isin(
        ["85669", "86197", "88274", "83405", "86475"]) |
        col("ca_state").isin(["CA", "WA", "GA"]) |
        (col("cs_sales_price") > 500))
    p = A(p, ["ca_zip"], [("total", "sum", col("cs_sales_price"))])
    return e.run(Limit(Sort(p, [("ca_zip", True)]), 100))


def q16(e: Engine) -> Frame:
    lo, hi = dsk("2002-02-01"), dsk("2002-04-02")
    cs1 = J(F(S("catalog_sales", "cs_ship_date_sk", "cs_ship_addr_sk",
               "cs_call_center_sk", "cs_order_number", "cs_warehouse_sk",
               "cs_ext_ship_cost", "cs_net_profit"),
              col("cs_ship_date_sk").between(lo, hi)),
            F(S("customer_address", "ca_address_sk", "ca_state"),
              col("ca_state") == "GA"),
            [("cs_ship_addr_sk", "ca_address_sk")])
    cs1 = J(cs1, F(S("call_center", "cc_call_center_sk", "cc_county"),
                   col("cc_county").isin(["county_00", "county_01"])),
            [("cs_call_center_sk", "cc_call_center_sk")])
    # runtime filter (Spark AQE-style): the countd-warehouse aggregate only
    # matters for orders that survive cs1's date/state/county filters —
    # semi-filter the 144M-row scan down to those orders first
    multi_wh = J(S("catalog_sales", "cs_order_number", "cs_warehouse_sk"),
                 P(cs1, ("fono", col("cs_order_number"))),
                 [("cs_order_number", "fono")], how="semi")
    multi_wh = P(F(A(multi_wh,
                     ["cs_order_number"],
                     [("nwh", "countd", col("cs_warehouse_sk"))]),
                   col("nwh") > 1),
                 ("ono", col("cs_order_number")))
    cs1 = J(cs1, multi_wh, [("cs_order_number", "ono")], how="semi")
    cs1 = J(cs1, P(S("catalog_returns", "cr_order_number"),
                   ("rno", col("cr_order_number"))),
            [("cs_order_number", "rno")], how="anti")
    return e.run(A(cs1, [],
                   [("order_count", "countd", col("cs_order_number")),
                    ("total_shipping_cost", "sum", col("cs_ext_ship_cost")),
                    ("total_net_profit", "sum", col("cs_net_profit"))]))


def q17(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_item_sk", "ss_customer_sk",
            "ss_ticket_number", "ss_quantity", "ss_store_sk"),
          dd(col("d_qoy") == 1, "d_qoy"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, S("store_returns", "sr_returned_date_sk", "sr_customer_sk",
               "sr_item_sk", "sr_ticket_number", "sr_return_quantity"),
          [("ss_customer_sk", "sr_customer_sk"),
           ("ss_item_sk", "sr_item_sk"),
           ("ss_ticket_number", "sr_ticket_number")])
    p = J(p, P(dd(col("d_qoy").between(1, 3), "d_qoy"),
               ("d2sk", col("d_date_sk"))),
          [("sr_returned_date_sk", "d2sk")])
    p = J(p, S("catalog_sales", "cs_sold_date_sk", "cs_bill_customer_sk",
               "cs_item_sk", "cs_quantity"),
          [("ss_customer_sk", "cs_bill_customer_sk"),
           ("ss_item_sk", "cs_item_sk")])
    p = J(p, P(dd(col("d_qoy").between(1, 3), "d_qoy"),
               ("d3sk", col("d_date_sk"))),
          [("cs_sold_date_sk", "d3sk")])
    p = J(p, S("store", "s_store_sk", "s_state"),
          [("ss_store_sk", "s_store_sk")])
    p = J(p, S("item", "i_item_sk", "i_item_id", "i_item_desc"),
          [("ss_item_sk", "i_item_sk")])
    p = A(p, ["i_item_id", "i_item_desc", "s_state"],
          [("store_sales_quantitycount", "count", col("ss_quantity")),
           ("store_sales_quantityave", "avg", col("ss_quantity")),
           ("store_sales_quantitystdev", "stddev_samp", col("ss_quantity")),
           ("store_returns_quantitycount", "count",
            col("sr_return_quantity")),
           ("store_returns_quantityave", "avg", col("sr_return_quantity")),
           ("catalog_sales_quantitycount", "count", col("cs_quantity")),
           ("catalog_sales_quantityave", "avg", col("cs_quantity"))])
    return e.run(Limit(Sort(p, [("i_item_id", True), ("i_item_desc", True),
                                ("s_state", True)]), 100))


def q18(e: Engine) -> Frame:
    p = J(S("catalog_sales", "cs_sold_date_sk", "cs_item_sk",
            "cs_bill_cdemo_sk", "cs_bill_customer_sk", "cs_quantity",
            "cs_list_price", "cs_coupon_amt", "cs_sales_price",
            "cs_net_profit"),
          dd(col("d_year") == 1998, "d_year"),
          [("cs_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("customer_demographics", "cd_demo_sk", "cd_gender",
                 "cd_education_status", "cd_dep_count"),
               (col("cd_gender") == "F") &
               (col("cd_education_status") == "Unknown")),
          [("cs_bill_cdemo_sk", "cd_demo_sk")])
    p = J(p, F(S("customer", "c_customer_sk", "c_current_addr_sk",
                 "c_birth_month"),
               col("c_birth_month").isin([1, 6, 8, 9, 12, 2])),
          [("cs_bill_customer_sk", "c_customer_sk")])
    p = J(p, F(S("customer_address", "ca_address_sk", "ca_country",
                 "ca_state", "ca_county"),
               col("ca_state").isin(["MS", "IN", "ND", "OK", "NM", "VA"])),
          [("c_current_addr_sk", "ca_address_sk")])
    p = J(p, S("item", "i_item_sk", "i_item_id"),
          [("cs_item_sk", "i_item_sk")])
    p = A(p, ["i_item_id", "ca_country", "ca_state", "ca_county"],
          [("agg1", "avg", col("cs_quantity")),
           ("agg2", "avg", col("cs_list_price")),
           ("agg3", "avg", col("cs_coupon_amt")),
           ("agg4", "avg", col("cs_sales_price")),
           ("agg5", "avg", col("cs_net_profit")),
           ("agg6", "avg", col("cd_dep_count"))], rollup=True)
    return e.run(Limit(Sort(p, [("__lvl", True), ("ca_country", True),
                                ("ca_state", True), ("ca_county", True),
                                ("i_item_id", True)]), 100))


def q19(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_item_sk", "ss_customer_sk",
            "ss_store_sk", "ss_ext_sales_price"),
          dd((col("d_moy") == 11) & (col("d_year") == 1998), "d_moy",
             "d_year"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("item", "i_item_sk", "i_brand_id", "i_brand",
                 "i_manufact_id", "i_manufact", "i_manager_id"),
               col("i_manager_id") == 8),
          [("ss_item_sk", "i_item_sk")])
    p = J(p, S("customer", "c_customer_sk", "c_current_addr_sk"),
          [("ss_customer_sk", "c_customer_sk")])
    p = J(p, S("customer_address", "ca_address_sk", "ca_zip"),
          [("c_current_addr_sk", "ca_address_sk")])
    p = J(p, S("store", "s_store_sk", "s_zip"),
          [("ss_store_sk", "s_store_sk")])
    p = F(p, ~(col("ca_zip").substr(1, 5) == col("s_zip").substr(1, 5)))
    p = A(p, ["i_brand", "i_brand_id", "i_manufact_id", "i_manufact"],
          [("ext_price", "sum", col("ss_ext_sales_price"))])
    return e.run(Limit(Sort(p, [("ext_price", False), ("i_brand", True),
                                ("i_brand_id", True),
                                ("i_manufact_id", True)]), 100))


def q20(e: Engine) -> Frame:
    return e.run(_ratio_by_class(e, "catalog_sales", "cs", "1999-02-22",
                                 ["Sports", "Books", "Home"]))


def q21(e: Engine) -> Frame:
    mid = dsk("2000-03-11")
    p = J(F(S("inventory", "inv_date_sk", "inv_item_sk", "inv_warehouse_sk",
              "inv_quantity_on_hand"),
            col("inv_date_sk").between(mid - 30, mid + 30)),
          F(S("item", "i_item_sk", "i_item_id", "i_current_price"),
            col("i_current_price").between(0.99, 1.49)),
          [("inv_item_sk", "i_item_sk")])
    p = J(p, S("warehouse", "w_warehouse_sk", "w_warehouse_name"),
          [("inv_warehouse_sk", "w_warehouse_sk")])
    p = A(p, ["w_warehouse_name", "i_item_id"],
          [("inv_before", "sum",
            case_when((col("inv_date_sk") < mid,
                       col("inv_quantity_on_hand")), otherwise=lit(0))),
           ("inv_after", "sum",
            case_when((col("inv_date_sk") >= mid,
                       col("inv_quantity_on_hand")), otherwise=lit(0)))])
    p = F(p, case_when((col("inv_before") > 0,
                        col("inv_after").cast_float() /
                        col("inv_before").cast_float()),
                       otherwise=lit(None)).between(2.0 / 3.0, 3.0 / 2.0))
    return e.run(Limit(Sort(p, [("w_warehouse_name", True),
                                ("i_item_id", True)]), 100))


def q22(e: Engine) -> Frame:
    p = J(S("inventory", "inv_date_sk", "inv_item_sk",
            "inv_quantity_on_hand"),
          dd(col("d_month_seq").between(1200, 1211), "d_month_seq"),
          [("inv_date_sk", "d_date_sk")])
    p = J(p, S("item", "i_item_sk", "i_product_name", "i_brand", "i_class",
               "i_category"),
          [("inv_item_sk", "i_item_sk")])
    p = A(p, ["i_product_name", "i_brand", "i_class", "i_category"],
          [("qoh", "avg", col("inv_quantity_on_hand"))], rollup=True)
    return e.run(Limit(Sort(p, [("qoh", True), ("i_product_name", True),
                                ("i_brand", True), ("i_class", True),
                                ("i_category", True)]), 100))


def q23(e: Engine) -> Frame:
    yrs = (col("d_year") >= 2000) & (col("d_year") <= 2002)
    freq = J(S("store_sales", "ss_sold_date_sk", "ss_item_sk"),
             dd(yrs, "d_year", "d_date"),
             [("ss_sold_date_sk", "d_date_sk")])
    freq = J(freq, S("item", "i_item_sk", "i_item_desc"),
             [("ss_item_sk", "i_item_sk")])
    freq = F(A(P(freq, ("itemdesc", col("i_item_desc").substr(1, 30)),
                 ("item_sk", col("ss_item_sk")), ("solddate", col("d_date"))),
               ["itemdesc", "item_sk", "solddate"],
               [("cnt", "count", None)]), col("cnt") > 4)
    e.register("frequent_ss_items",
               e.run(Distinct(P(freq, ("item_sk", col("item_sk"))))))
    csales = A(J(S("store_sales", "ss_customer_sk", "ss_quantity",
                   "ss_sales_price", "ss_sold_date_sk"),
                 dd(yrs, "d_year"), [("ss_sold_date_sk", "d_date_sk")]),
               ["ss_customer_sk"],
               [("csales", "sum", col("ss_quantity") *
                 col("ss_sales_price"))])
    e.register("csales", e.run(csales))
    tpcds_cmax = e.scalar(A(S("csales"), [],
                            [("m", "max", col("csales"))]), "m") or 0.0
    best = P(F(S("csales"), col("csales") >
               lit(0.5 * float(tpcds_cmax))),
             ("cust_sk", col("ss_customer_sk")))
    e.register("best_ss_customer", e.run(best))
    win = dd((col("d_year") == 2000) & (col("d_moy") == 2), "d_year",
             "d_moy")
    cs = J(S("catalog_sales", "cs_sold_date_sk", "cs_item_sk",
             "cs_bill_customer_sk", "cs_quantity", "cs_list_price"), win,
           [("cs_sold_date_sk", "d_date_sk")])
    cs = J(cs, S("frequent_ss_items"), [("cs_item_sk", "item_sk")],
           how="semi")
    cs = J(cs, S("best_ss_customer"), [("cs_bill_customer_sk", "cust_sk")],
           how="semi")
    cs = P(cs, ("sales", col("cs_quantity") * col("cs_list_price")))
    ws = J(S("web_sales", "ws_sold_date_sk", "ws_item_sk",
             "ws_bill_customer_sk", "ws_quantity", "ws_list_price"), win,
           [("ws_sold_date_sk", "d_date_sk")])
    ws = J(ws, S("frequent_ss_items"), [("ws_item_sk", "item_sk")],
           how="semi")
    ws = J(ws, S("best_ss_customer"), [("ws_bill_customer_sk", "cust_sk")],
           how="semi")
    ws = P(ws, ("sales", col("ws_quantity") * col("ws_list_price")))
    return e.run(A(Union([cs, ws]), [], [("total", "sum", col("sales"))]))


def q24(e: Engine) -> Frame:
    base = J(S("store_sales", "ss_ticket_number", "ss_item_sk",
               "ss_customer_sk", "ss_store_sk", "ss_net_paid"),
             S("store_returns", "sr_ticket_number", "sr_item_sk"),
             [("ss_ticket_number", "sr_ticket_number"),
              ("ss_item_sk", "sr_item_sk")])
    base = J(base, F(S("store", "s_store_sk", "s_market_id", "s_store_name",
                       "s_state", "s_zip"), col("s_market_id") == 8),
             [("ss_store_sk", "s_store_sk")])
    base = J(base, S("item", "i_item_sk", "i_color", "i_current_price",
                     "i_manager_id", "i_units", "i_size"),
             [("ss_item_sk", "i_item_sk")])
    base = J(base, S("customer", "c_customer_sk", "c_first_name",
                     "c_last_name", "c_current_addr_sk", "c_birth_country"),
             [("ss_customer_sk", "c_customer_sk")])
    base = J(base, S("customer_address", "ca_address_sk", "ca_state",
                     "ca_country", "ca_zip"),
             [("c_current_addr_sk", "ca_address_sk")])
    base = F(base, ~(col("c_birth_country") == col("ca_country")) &
             (col("s_zip") == col("ca_zip")))
    ssales = A(base, ["c_last_name", "c_first_name", "s_store_name",
                      "ca_state", "s_state", "i_color", "i_current_price",
                      "i_manager_id", "i_units", "i_size"],
               [("netpaid", "sum", col("ss_net_paid"))])
    e.register("ssales", e.run(ssales))
    avg_np = e.scalar(A(S("ssales"), [],
                        [("a", "avg", col("netpaid"))]), "a") or 0.0
    p = F(S("ssales"), col("i_color") == "azure")
    p = A(p, ["c_last_name", "c_first_name", "s_store_name"],
          [("paid", "sum", col("netpaid"))])
    p = F(p, col("paid") > lit(0.05 * float(avg_np)))
    return e.run(Sort(p, [("c_last_name", True), ("c_first_name", True),
                          ("s_store_name", True)]))


def _three_way_sales_returns(e, agg_map, moy_first=4, q25=True):
    """q25/q29 shape: ss -> sr -> cs linked by customer+item with monthly
    windows, group by item/store."""
    p = J(S("store_sales", "ss_sold_date_sk", "ss_item_sk", "ss_customer_sk",
            "ss_ticket_number", "ss_store_sk", "ss_net_profit",
            "ss_quantity"),
          dd((col("d_moy") == moy_first) & (col("d_year") == 2000), "d_moy",
             "d_year"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, S("store_returns", "sr_returned_date_sk", "sr_customer_sk",
               "sr_item_sk", "sr_ticket_number", "sr_net_loss",
               "sr_return_quantity"),
          [("ss_customer_sk", "sr_customer_sk"),
           ("ss_item_sk", "sr_item_sk"),
           ("ss_ticket_number", "sr_ticket_number")])
    p = J(p, P(dd((col("d_moy").between(moy_first, moy_first + 6)) &
                  (col("d_year") == 2000), "d_moy", "d_year"),
               ("d2sk", col("d_date_sk"))),
          [("sr_returned_date_sk", "d2sk")])
    p = J(p, S("catalog_sales", "cs_sold_date_sk", "cs_bill_customer_sk",
               "cs_item_sk", "cs_net_profit", "cs_quantity"),
          [("ss_customer_sk", "cs_bill_customer_sk"),
           ("ss_item_sk", "cs_item_sk")])
    p = J(p, P(dd((col("d_moy").between(moy_first, moy_first + 6)) &
                  (col("d_year") == 2000), "d_moy", "d_year"),
               ("d3sk", col("d_date_sk"))),
          [("cs_sold_date_sk", "d3sk")])
    p = J(p, S("store", "s_store_sk", "s_store_id", "s_store_name"),
          [("ss_store_sk", "s_store_sk")])
    p = J(p, S("item", "i_item_sk", "i_item_id", "i_item_desc"),
          [("ss_item_sk", "i_item_sk")])
    p = A(p, ["i_item_id", "i_item_desc", "s_store_id", "s_store_name"],
          agg_map)
    return Limit(Sort(p, [("i_item_id", True), ("i_item_desc", True),
                          ("s_store_id", True), ("s_store_name", True)]),
                 100)


def q25(e: Engine) -> Frame:
    return e.run(_three_way_sales_returns(
        e, [("store_sales_profit", "sum", col("ss_net_profit")),
            ("store_returns_loss", "sum", col("sr_net_loss")),
            ("catalog_sales_profit", "sum", col("cs_net_profit"))]))


def q26(e: Engine) -> Frame:
    p = J(S("catalog_sales", "cs_sold_date_sk", "cs_item_sk",
            "cs_bill_cdemo_sk", "cs_promo_sk", "cs_quantity",
            "cs_list_price", "cs_coupon_amt", "cs_sales_price"),
          dd(col("d_year") == 2000, "d_year"),
          [("cs_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("customer_demographics", "cd_demo_sk", "cd_gender",
                 "cd_marital_status", "cd_education_status"),
               (col("cd_gender") == "M") & (col("cd_marital_status") == "S")
               & (col("cd_education_status") == "College")),
          [("cs_bill_cdemo_sk", "cd_demo_sk")])
    p = J(p, F(S("promotion", "p_promo_sk", "p_channel_email",
                 "p_channel_event"),
               (col("p_channel_email") == "N") |
               (col("p_channel_event") == "N")),
          [("cs_promo_sk", "p_promo_sk")])
    p = J(p, S("item", "i_item_sk", "i_item_id"),
          [("cs_item_sk", "i_item_sk")])
    p = A(p, ["i_item_id"],
          [("agg1", "avg", col("cs_quantity")),
           ("agg2", "avg", col("cs_list_price")),
           ("agg3", "avg", col("cs_coupon_amt")),
           ("agg4", "avg", col("cs_sales_price"))])
    return e.run(Limit(Sort(p, [("i_item_id", True)]), 100))


def q27(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_item_sk", "ss_store_sk",
            "ss_cdemo_sk", "ss_quantity", "ss_list_price", "ss_coupon_amt",
            "ss_sales_price"),
          dd(col("d_year") == 2002, "d_year"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("customer_demographics", "cd_demo_sk", "cd_gender",
                 "cd_marital_status", "cd_education_status"),
               (col("cd_gender") == "M") & (col("cd_marital_status") == "S")
               & (col("cd_education_status") == "College")),
          [("ss_cdemo_sk", "cd_demo_sk")])
    p = J(p, F(S("store", "s_store_sk", "s_state"),
               col("s_state").isin(["AL", "AK", "AZ", "AR", "CA", "CO"])),
          [("ss_store_sk", "s_store_sk")])
    p = J(p, S("item", "i_item_sk", "i_item_id"),
          [("ss_item_sk", "i_item_sk")])
    p = A(p, ["i_item_id", "s_state"],
          [("agg1", "avg", col("ss_quantity")),
           ("agg2", "avg", col("ss_list_price")),
           ("agg3", "avg", col("ss_coupon_amt")),
           ("agg4", "avg", col("ss_sales_price"))], rollup=True)
    return e.run(Limit(Sort(p, [("__lvl", True), ("i_item_id", True),
                                ("s_state", True)]), 100))


def q28(e: Engine) -> Frame:
    buckets = [(0, 5, 11, 8, 37), (6, 10, 14, 27, 65), (11, 15, 17, 45, 91),
               (16, 20, 20, 70, 122), (21, 25, 23, 100, 154),
               (26, 30, 26, 130, 187)]
    out = {}
    for i, (qlo, qhi, lp, cp, wc) in enumerate(buckets, 1):
        base = F(S("store_sales", "ss_quantity", "ss_list_price",
                   "ss_coupon_amt", "ss_wholesale_cost"),
                 col("ss_quantity").between(qlo, qhi) &
                 (col("ss_list_price").between(lp, lp + 10) |
                  col("ss_coupon_amt").between(cp, cp + 1000) |
                  col("ss_wholesale_cost").between(wc, wc + 20)))
        row = e.run(A(base, [],
                      [("avg_p", "avg", col("ss_list_price")),
                       ("cnt", "count", col("ss_list_price")),
                       ("cntd", "countd", col("ss_list_price"))]))
        rows = row.to_rows()
        a, c, d = rows[0] if rows else (None, 0, 0)
        out[f"b{i}_lp"] = a
        out[f"b{i}_cnt"] = int(c or 0)
        out[f"b{i}_cntd"] = int(d or 0)
    return e.const_frame(**out)


def q29(e: Engine) -> Frame:
    return e.run(_three_way_sales_returns(
        e, [("store_sales_quantity", "sum", col("ss_quantity")),
            ("store_returns_quantity", "sum", col("sr_return_quantity")),
            ("catalog_sales_quantity", "sum", col("cs_quantity"))]))


def q30(e: Engine) -> Frame:
    ctr = J(S("web_returns", "wr_returned_date_sk",
              "wr_returning_customer_sk", "wr_returning_addr_sk",
              "wr_return_amt"),
            dd(col("d_year") == 2002, "d_year"),
            [("wr_returned_date_sk", "d_date_sk")])
    ctr = J(ctr, S("customer_address", "ca_address_sk", "ca_state"),
            [("wr_returning_addr_sk", "ca_address_sk")])
    ctr = A(ctr, ["wr_returning_customer_sk", "ca_state"],
            [("ctr_total_return", "sum", col("wr_return_amt"))])
    e.register("ctr30", e.run(ctr))
    st_avg = A(S("ctr30"), ["ca_state"],
               [("avg_ret", "avg", col("ctr_total_return"))])
    p = J(S("ctr30"), P(st_avg, ("st2", col("ca_state")),
                        ("avg_ret", col("avg_ret"))),
          [("ca_state", "st2")])
    p = F(p, col("ctr_total_return") > col("avg_ret") * 1.2)
    cust = J(S("customer", "c_customer_sk", "c_customer_id",
               "c_salutation", "c_first_name", "c_last_name",
               "c_preferred_cust_flag", "c_birth_day", "c_birth_month",
               "c_birth_year", "c_birth_country", "c_current_addr_sk"),
             P(F(S("customer_address", "ca_address_sk", "ca_state"),
                 col("ca_state") == "GA"),
               ("ca2", col("ca_address_sk"))),
             [("c_current_addr_sk", "ca2")])
    p = J(p, cust, [("wr_returning_customer_sk", "c_customer_sk")])
    p = P(p, ("c_customer_id", col("c_customer_id")),
          ("c_salutation", col("c_salutation")),
          ("c_first_name", col("c_first_name")),
          ("c_last_name", col("c_last_name")),
          ("c_preferred_cust_flag", col("c_preferred_cust_flag")),
          ("c_birth_day", col("c_birth_day")),
          ("c_birth_month", col("c_birth_month")),
          ("c_birth_year", col("c_birth_year")),
          ("c_birth_country", col("c_birth_country")),
          ("ctr_total_return", col("ctr_total_return")))
    return e.run(Limit(Sort(p, [("c_customer_id", True)]), 100))


def q31(e: Engine) -> Frame:
    def county_qtr(tbl, pre, datecol, amtcol, tag):
        p = J(S(tbl, datecol, f"{pre}_addr_sk", amtcol),
              dd((col("d_year") == 2000) & col("d_qoy").between(1, 3),
                 "d_year", "d_qoy"),
              [(datecol, "d_date_sk")])
        p = J(p, S("customer_address", "ca_address_sk", "ca_county"),
              [(f"{pre}_addr_sk", "ca_address_sk")])
        return A(p, ["ca_county", "d_qoy"],
                 [(tag, "sum", col(amtcol))])
    sss = e.run(county_qtr("store_sales", "ss", "ss_sold_date_sk",
                           "ss_ext_sales_price", "ss"))
    wss = e.run(county_qtr("web_sales", "ws_bill", "ws_sold_date_sk",
                           "ws_ext_sales_price", "ws"))
    e.register("q31ss", sss)
    e.register("q31ws", wss)

    def pick(name, q, tag, out):
        return P(F(S(name), col("d_qoy") == q),
                 (f"cty{out}", col("ca_county")), (out, col(tag)))
    p = J(pick("q31ss", 1, "ss", "ss1"), pick("q31ss", 2, "ss", "ss2"),
          [("ctyss1", "ctyss2")])
    p = J(p, pick("q31ss", 3, "ss", "ss3"), [("ctyss1", "ctyss3")])
    p = J(p, pick("q31ws", 1, "ws", "ws1"), [("ctyss1", "ctyws1")])
    p = J(p, pick("q31ws", 2, "ws", "ws2"), [("ctyss1", "ctyws2")])
    p = J(p, pick("q31ws", 3, "ws", "ws3"), [("ctyss1", "ctyws3")])
    p = F(p, (col("ws2") / col("ws1") > col("ss2") / col("ss1")) &
          (col("ws3") / col("ws2") > col("ss3") / col("ss2")))
    p = P(p, ("ca_county", col("ctyss1")), ("d_year", lit(2000)),
          ("web_q1_q2_increase", col("ws2") / col("ws1")),
          ("store_q1_q2_increase", col("ss2") / col("ss1")),
          ("web_q2_q3_increase", col("ws3") / col("ws2")),
          ("store_q2_q3_increase", col("ss3") / col("ss2")))
    return e.run(Sort(p, [("ca_county", True)]))


def q32(e: Engine) -> Frame:
    lo = dsk("2000-01-27")
    base = J(F(S("catalog_sales", "cs_sold_date_sk", "cs_item_sk",
                 "cs_ext_discount_amt"),
               col("cs_sold_date_sk").between(lo, lo + 90)),
             F(S("item", "i_item_sk", "i_manufact_id"),
               col("i_manufact_id") == 7),
             [("cs_item_sk", "i_item_sk")])
    e.register("q32base", e.run(base))
    avg_d = A(S("q32base"), ["cs_item_sk"],
              [("avg_disc", "avg", col("cs_ext_discount_amt"))])
    p = J(S("q32base"), P(avg_d, ("isk2", col("cs_item_sk")),
                          ("avg_disc", col("avg_disc"))),
          [("cs_item_sk", "isk2")])
    p = F(p, col("cs_ext_discount_amt") > col("avg_disc") * 1.3)
    return e.run(A(p, [], [("excess_discount_amount", "sum",
                            col("cs_ext_discount_amt"))]))


def _manufact_by_channel(e, cats_filter, moy, year):
    """q33/q56/q60 shape: 3 channels x (date, gmt-offset address, item
    restriction via subquery) grouped by manufact/item."""
    item_keys = Distinct(P(F(S("item", "i_item_sk", "i_manufact_id",
                              *cats_filter[0]), cats_filter[1]),
                           ("mid", col("i_manufact_id"))))
    e.register("q33keys", e.run(item_keys))
    chans = []
    for tbl, pre, addr in (("store_sales", "ss", "ss_addr_sk"),
                           ("catalog_sales", "cs", "cs_bill_addr_sk"),
                           ("web_sales", "ws", "ws_bill_addr_sk")):
        p = J(S(tbl, f"{pre}_sold_date_sk", f"{pre}_item_sk", addr,
                f"{pre}_ext_sales_price"),
              dd((col("d_year") == year) & (col("d_moy") == moy), "d_year",
                 "d_moy"),
              [(f"{pre}_sold_date_sk", "d_date_sk")])
        p = J(p, F(S("customer_address", "ca_address_sk", "ca_gmt_offset"),
                   col("ca_gmt_offset") == -5.0),
              [(addr, "ca_address_sk")])
        p = J(p, S("item", "i_item_sk", "i_manufact_id"),
              [(f"{pre}_item_sk", "i_item_sk")])
        p = J(p, S("q33keys"), [("i_manufact_id", "mid")], how="semi")
        chans.append(A(p, ["i_manufact_id"],
                       [("total_sales", "sum",
                         col(f"{pre}_ext_sales_price"))]))
    p = A(Union(chans), ["i_manufact_id"],
          [("total_sales", "sum", col("total_sales"))])
    return Limit(Sort(p, [("total_sales", True)]), 100)


def q33(e: Engine) -> Frame:
    return e.run(_manufact_by_channel(
        e, (["i_category"], col("i_category") == "Electronics"), 5, 1998))


def q34(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_store_sk", "ss_hdemo_sk",
            "ss_customer_sk", "ss_ticket_number"),
          dd((col("d_dom").between(1, 3) | col("d_dom").between(25, 28)) &
             col("d_year").isin([1999, 2000, 2001]), "d_dom", "d_year"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("store", "s_store_sk", "s_county"),
               col("s_county").isin(["county_00", "county_01", "county_02",
                                     "county_03"])),
          [("ss_store_sk", "s_store_sk")])
    p = J(p, F(S("household_demographics", "hd_demo_sk", "hd_buy_potential",
                 "hd_vehicle_count", "hd_dep_count"),
               col("hd_buy_potential").isin([">10000", "Unknown"]) &
               (col("hd_vehicle_count") > 0) &
               (col("hd_dep_count").cast_float() /
                col("hd_vehicle_count").cast_float() > 1.2)),
          [("ss_hdemo_sk", "hd_demo_sk")])
    p = F(A(p, ["ss_ticket_number", "ss_customer_sk"],
            [("cnt", "count", None)]), col("cnt").between(15, 20))
    p = J(p, S("customer", "c_customer_sk", "c_last_name", "c_first_name",
               "c_salutation", "c_preferred_cust_flag"),
          [("ss_customer_sk", "c_customer_sk")])
    return e.run(Limit(Sort(p, [("c_last_name", True),
                                ("c_first_name", True),
                                ("c_salutation", True),
                                ("cnt", False)]), 1000))


def q35(e: Engine) -> Frame:
    win = dd((col("d_year") == 2002) & (col("d_qoy") < 4), "d_year",
             "d_qoy")
    c = S("customer", "c_customer_sk", "c_current_addr_sk",
          "c_current_cdemo_sk")
    ss = J(S("store_sales", "ss_customer_sk", "ss_sold_date_sk"), win,
           [("ss_sold_date_sk", "d_date_sk")])
    c = J(c, P(ss, ("cust", col("ss_customer_sk"))),
          [("c_customer_sk", "cust")], how="semi")
    ws = J(S("web_sales", "ws_bill_customer_sk", "ws_sold_date_sk"), win,
           [("ws_sold_date_sk", "d_date_sk")])
    cs = J(S("catalog_sales", "cs_ship_customer_sk", "cs_sold_date_sk"), win,
           [("cs_sold_date_sk", "d_date_sk")])
    either = Union([P(ws, ("cust", col("ws_bill_customer_sk"))),
                    P(cs, ("cust", col("cs_ship_customer_sk")))])
    c = J(c, either, [("c_customer_sk", "cust")], how="semi")
    p = J(c, S("customer_address", "ca_address_sk", "ca_state"),
          [("c_current_addr_sk", "ca_address_sk")])
    p = J(p, S("customer_demographics", "cd_demo_sk", "cd_gender",
               "cd_marital_status", "cd_dep_count",
               "cd_dep_employed_count", "cd_dep_college_count"),
          [("c_current_cdemo_sk", "cd_demo_sk")])
    p = A(p, ["ca_state", "cd_gender", "cd_marital_status", "cd_dep_count",
              "cd_dep_employed_count", "cd_dep_college_count"],
          [("cnt1", "count", None),
           ("min1", "min", col("cd_dep_count")),
           ("max1", "max", col("cd_dep_count")),
           ("avg1", "avg", col("cd_dep_count")),
           ("min2", "min", col("cd_dep_employed_count")),
           ("max2", "max", col("cd_dep_employed_count")),
           ("avg2", "avg", col("cd_dep_employed_count")),
           ("min3", "min", col("cd_dep_college_count")),
           ("max3", "max", col("cd_dep_college_count")),
           ("avg3", "avg", col("cd_dep_college_count"))])
    return e.run(Limit(Sort(p, [("ca_state", True), ("cd_gender", True),
                                ("cd_marital_status", True),
                                ("cd_dep_count", True)]), 100))


def q36(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_item_sk", "ss_store_sk",
            "ss_net_profit", "ss_ext_sales_price"),
          dd(col("d_year") == 2001, "d_year"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, S("item", "i_item_sk", "i_category", "i_class"),
          [("ss_item_sk", "i_item_sk")])
    p = J(p, F(S("store", "s_store_sk", "s_state"),
               col("s_state").isin(["AL", "AK", "AZ", "AR", "CA", "CO",
                                    "CT", "DE"])),
          [("ss_store_sk", "s_store_sk")])
    p = A(p, ["i_category", "i_class"],
          [("profit", "sum", col("ss_net_profit")),
           ("sales", "sum", col("ss_ext_sales_price"))], rollup=True)
    p = P(p, ("i_category", col("i_category")), ("i_class", col("i_class")),
          ("lochierarchy", lit(2) - col("__lvl")),
          ("gross_margin", col("profit") / col("sales")),
          ("rank_cat", case_when((col("__lvl") == 2, col("i_category")),
                                 otherwise=lit(None))))
    p = Window(p, ["lochierarchy", "rank_cat"],
               [("rank_within_parent", "rank", None,
                 [("gross_margin", True)])])
    return e.run(Limit(Sort(p, [("lochierarchy", False),
                                ("i_category", True),
                                ("rank_within_parent", True)]), 100))


def q37(e: Engine) -> Frame:
    lo = dsk("2000-02-01")
    it = F(S("item", "i_item_sk", "i_item_id", "i_item_desc",
             "i_current_price", "i_manufact_id"),
           col("i_current_price").between(68.0, 98.0) &
           col("i_manufact_id").isin([7, 8, 9, 10]))
    p = J(it, F(S("inventory", "inv_item_sk", "inv_date_sk",
                  "inv_quantity_on_hand"),
                col("inv_quantity_on_hand").between(100, 500) &
                col("inv_date_sk").between(lo, lo + 60)),
          [("i_item_sk", "inv_item_sk")])
    p = J(p, P(S("catalog_sales", "cs_item_sk"),
               ("csk", col("cs_item_sk"))),
          [("i_item_sk", "csk")], how="semi")
    p = A(p, ["i_item_id", "i_item_desc", "i_current_price"],
          [("cnt", "count", None)])
    return e.run(Limit(Sort(p, [("i_item_id", True)]), 100))


def q38(e: Engine) -> Frame:
    win = dd(col("d_month_seq").between(1200, 1211), "d_month_seq",
             "d_date")
    frames = []
    for tbl, pre, cc in (("store_sales", "ss", "ss_customer_sk"),
                         ("catalog_sales", "cs", "cs_bill_customer_sk"),
                         ("web_sales", "ws", "ws_bill_customer_sk")):
        p = J(S(tbl, f"{pre}_sold_date_sk", cc), win,
              [(f"{pre}_sold_date_sk", "d_date_sk")])
        p = J(p, S("customer", "c_customer_sk", "c_last_name",
                   "c_first_name"),
              [(cc, "c_customer_sk")])
        frames.append(e.run(Distinct(P(p, ("ln", col("c_last_name")),
                                       ("fn", col("c_first_name")),
                                       ("dt", col("d_date"))))))
    e.register("q38a", frames[0])
    e.register("q38b", frames[1])
    e.register("q38c", frames[2])
    p = J(S("q38a"), P(S("q38b"), ("ln2", col("ln")), ("fn2", col("fn")),
                       ("dt2", col("dt"))),
          [("ln", "ln2"), ("fn", "fn2"), ("dt", "dt2")], how="semi")
    p = J(p, P(S("q38c"), ("ln3", col("ln")), ("fn3", col("fn")),
               ("dt3", col("dt"))),
          [("ln", "ln3"), ("fn", "fn3"), ("dt", "dt3")], how="semi")
    return e.run(A(p, [], [("cnt", "count", None)]))


def q39(e: Engine) -> Frame:
    base = J(S("inventory", "inv_item_sk", "inv_warehouse_sk", "inv_date_sk",
               "inv_quantity_on_hand"),
             dd(col("d_year") == 2001, "d_year", "d_moy"),
             [("inv_date_sk", "d_date_sk")])
    base = J(base, S("warehouse", "w_warehouse_sk", "w_warehouse_name"),
             [("inv_warehouse_sk", "w_warehouse_sk")])
    g = A(base, ["w_warehouse_name", "inv_warehouse_sk", "inv_item_sk",
                 "d_moy"],
          [("mean", "avg", col("inv_quantity_on_hand")),
           ("stdev", "stddev_samp", col("inv_quantity_on_hand"))])
    g = P(g, ("w_warehouse_name", col("w_warehouse_name")),
          ("w_sk", col("inv_warehouse_sk")), ("i_sk", col("inv_item_sk")),
          ("d_moy", col("d_moy")), ("mean", col("mean")),
          ("cov", case_when((col("mean") == 0.0, lit(None)),
                            otherwise=col("stdev") / col("mean"))))
    g = F(g, col("cov") > 1.0)
    e.register("q39inv", e.run(g))
    p = J(P(F(S("q39inv"), col("d_moy") == 1),
            ("w1", col("w_sk")), ("i1", col("i_sk")),
            ("mean1", col("mean")), ("cov1", col("cov"))),
          P(F(S("q39inv"), col("d_moy") == 2),
            ("w2", col("w_sk")), ("i2", col("i_sk")),
            ("mean2", col("mean")), ("cov2", col("cov"))),
          [("w1", "w2"), ("i1", "i2")])
    return e.run(Sort(p, [("w1", True), ("i1", True)]))


def q40(e: Engine) -> Frame:
    mid = dsk("2000-03-11")
    p = J(F(S("catalog_sales", "cs_order_number", "cs_item_sk",
              "cs_sold_date_sk", "cs_warehouse_sk", "cs_sales_price"),
            col("cs_sold_date_sk").between(mid - 30, mid + 30)),
          P(S("catalog_returns", "cr_order_number", "cr_item_sk",
              "cr_refunded_cash"),
            ("cr_order_number", col("cr_order_number")),
            ("cr_item_sk2", col("cr_item_sk")),
            ("cr_refunded_cash", col("cr_refunded_cash"))),
          [("cs_order_number", "cr_order_number"),
           ("cs_item_sk", "cr_item_sk2")], how="left")
    p = J(p, S("warehouse", "w_warehouse_sk", "w_state"),
          [("cs_warehouse_sk", "w_warehouse_sk")])
    p = J(p, F(S("item", "i_item_sk", "i_item_id", "i_current_price"),
               col("i_current_price").between(0.99, 1.49)),
          [("cs_item_sk", "i_item_sk")])
    net = col("cs_sales_price") - coalesce(col("cr_refunded_cash"),
                                           lit(0.0))
    p = A(p, ["w_state", "i_item_id"],
          [("sales_before", "sum",
            case_when((col("cs_sold_date_sk") < mid, net),
                      otherwise=lit(0.0))),
           ("sales_after", "sum",
            case_when((col("cs_sold_date_sk") >= mid, net),
                      otherwise=lit(0.0)))])
    return e.run(Limit(Sort(p, [("w_state", True), ("i_item_id", True)]),
                       100))


def q41(e: Engine) -> Frame:
    combo = (((col("i_category") == "Women") &
              col("i_color").isin(["azure", "blush", "coral", "cream"]) &
              col("i_units").isin(["Each", "Dozen", "Case", "Box"]) &
              col("i_size").isin(["small", "petite", "medium"])) |
             ((col("i_category") == "Men") &
              col("i_color").isin(["black", "blue", "brown", "beige"]) &
              col("i_units").isin(["Pallet", "Gross", "Bunch", "Carton"]) &
              col("i_size").isin(["large", "economy", "N/A"])))
    manu = Distinct(P(F(S("item", "i_item_sk", "i_manufact", "i_category",
                          "i_color", "i_units", "i_size"), combo),
                      ("manu", col("i_manufact"))))
    p = J(F(S("item", "i_item_sk", "i_manufact_id", "i_manufact",
              "i_product_name"),
            col("i_manufact_id").between(7, 47)),
          manu, [("i_manufact", "manu")], how="semi")
    p = Distinct(P(p, ("i_product_name", col("i_product_name"))))
    return e.run(Limit(Sort(p, [("i_product_name", True)]), 100))


def q42(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_item_sk",
            "ss_ext_sales_price"),
          dd((col("d_moy") == 11) & (col("d_year") == 2000), "d_moy",
             "d_year"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("item", "i_item_sk", "i_category_id", "i_category",
                 "i_manager_id"), col("i_manager_id") == 1),
          [("ss_item_sk", "i_item_sk")])
    p = A(p, ["i_category_id", "i_category"],
          [("total", "sum", col("ss_ext_sales_price"))])
    p = P(p, ("d_year", lit(2000)), ("i_category_id", col("i_category_id")),
          ("i_category", col("i_category")), ("total", col("total")))
    return e.run(Limit(Sort(p, [("total", False), ("d_year", True),
                                ("i_category_id", True),
                                ("i_category", True)]), 100))


def q43(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_store_sk",
            "ss_sales_price"),
          dd(col("d_year") == 2000, "d_year", "d_day_name"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("store", "s_store_sk", "s_store_id", "s_store_name",
                 "s_gmt_offset"), col("s_gmt_offset") == -5.0),
          [("ss_store_sk", "s_store_sk")])
    days = ["Sunday", "Monday", "Tuesday", "Wednesday", "Thursday",
            "Friday", "Saturday"]
    p = A(p, ["s_store_name", "s_store_id"],
          [(dy.lower()[:3] + "_sales", "sum",
            case_when((col("d_day_name") == dy, col("ss_sales_price")),
                      otherwise=lit(0.0))) for dy in days])
    return e.run(Limit(Sort(p, [("s_store_name", True),
                                ("s_store_id", True)]), 100))


def q44(e: Engine) -> Frame:
    base = F(S("store_sales", "ss_store_sk", "ss_item_sk", "ss_net_profit",
               "ss_hdemo_sk"), col("ss_store_sk") == 4)
    thresh = e.scalar(A(F(base, is_null(col("ss_hdemo_sk"))), [],
                        [("a", "avg", col("ss_net_profit"))]), "a")
    v = A(base, ["ss_item_sk"], [("rank_col", "avg", col("ss_net_profit"))])
    v = F(v, col("rank_col") > lit(0.9 * float(thresh or 0.0)))
    e.register("q44v", e.run(v))
    asc = Window(S("q44v"), [], [("rnk", "rank", None,
                                  [("rank_col", True)])])
    dsc = Window(S("q44v"), [], [("rnk", "rank", None,
                                  [("rank_col", False)])])
    asc = P(F(asc, col("rnk") <= 10), ("rnk", col("rnk")),
            ("best_sk", col("ss_item_sk")))
    dsc = P(F(dsc, col("rnk") <= 10), ("rnk2", col("rnk")),
            ("worst_sk", col("ss_item_sk")))
    p = J(asc, dsc, [("rnk", "rnk2")])
    p = J(p, P(S("item", "i_item_sk", "i_product_name"),
               ("i1sk", col("i_item_sk")), ("best_performing",
                                            col("i_product_name"))),
          [("best_sk", "i1sk")])
    p = J(p, P(S("item", "i_item_sk", "i_product_name"),
               ("i2sk", col("i_item_sk")), ("worst_performing",
                                            col("i_product_name"))),
          [("worst_sk", "i2sk")])
    p = P(p, ("rnk", col("rnk")), ("best_performing",
                                   col("best_performing")),
          ("worst_performing", col("worst_performing")))
    return e.run(Sort(p, [("rnk", True)]))


def q45(e: Engine) -> Frame:
    item_ids = Distinct(P(F(S("item", "i_item_sk", "i_item_id"),
                            col("i_item_sk").isin([2, 3, 5, 7, 11, 13, 17,
                                                   19, 23, 29, 31, 37, 41,
                                                   43, 47, 53])),
                          ("iid", col("i_item_id"))))
    e.register("q45ids", e.run(item_ids))
    p = J(S("web_sales", "ws_sold_date_sk", "ws_bill_customer_sk",
            "ws_item_sk", "ws_sales_price"),
          dd((col("d_qoy") == 2) & (col("d_year") == 2001), "d_qoy",
             "d_year"),
          [("ws_sold_date_sk", "d_date_sk")])
    p = J(p, S("customer", "c_customer_sk", "c_current_addr_sk"),
          [("ws_bill_customer_sk", "c_customer_sk")])
    p = J(p, S("customer_address", "ca_address_sk", "ca_city", "ca_zip"),
          [("c_current_addr_sk", "ca_address_sk")])
    p = J(p, S("item", "i_item_sk", "i_item_id"),
          [("ws_item_sk", "i_item_sk")])
    # zip in list OR item_id in subquery: express the second leg via a
    # marker column from a left join
    p = J(p, P(S("q45ids"), ("iid2", col("iid")), ("hit", lit(1))),
          [("i_item_id", "iid2")], how="left")
    zips = [f"{10000 + 89 * k % 90000:05d}" for k in (5, 10, 15, 20, 25,
                                                        30, 35, 40, 45)]
    p = F(p, col("ca_zip").substr(1, 5).isin(zips) | (col("hit") == 1))
    p = A(p, ["ca_zip", "ca_city"],
          [("total", "sum", col("ws_sales_price"))])
    return e.run(Limit(Sort(p, [("ca_zip", True), ("ca_city", True)]), 100))


def _city_profit(e, dom_cond, hd_cond):
    """q46/q68 shape."""
    p = J(S("store_sales", "ss_sold_date_sk", "ss_store_sk", "ss_hdemo_sk",
            "ss_addr_sk", "ss_customer_sk", "ss_ticket_number",
            "ss_coupon_amt", "ss_net_profit"),
          dd(dom_cond, "d_dow", "d_year", "d_dom"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("store", "s_store_sk", "s_city"),
               col("s_city").isin(["city_00", "city_01", "city_02",
                                   "city_03", "city_04"])),
          [("ss_store_sk", "s_store_sk")])
    p = J(p, F(S("household_demographics", "hd_demo_sk", "hd_dep_count",
                 "hd_vehicle_count"), hd_cond),
          [("ss_hdemo_sk", "hd_demo_sk")])
    p = J(p, P(S("customer_address", "ca_address_sk", "ca_city"),
               ("ca_address_sk", col("ca_address_sk")),
               ("bought_city", col("ca_city"))),
          [("ss_addr_sk", "ca_address_sk")])
    p = A(p, ["ss_ticket_number", "ss_customer_sk", "bought_city"],
          [("amt", "sum", col("ss_coupon_amt")),
           ("profit", "sum", col("ss_net_profit"))])
    p = J(p, S("customer", "c_customer_sk", "c_current_addr_sk",
               "c_last_name", "c_first_name"),
          [("ss_customer_sk", "c_customer_sk")])
    p = J(p, P(S("customer_address", "ca_address_sk", "ca_city"),
               ("ca2", col("ca_address_sk")),
               ("current_city", col("ca_city"))),
          [("c_current_addr_sk", "ca2")])
    p = F(p, ~(col("current_city") == col("bought_city")))
    return Limit(Sort(p, [("c_last_name", True), ("c_first_name", True),
                          ("ss_ticket_number", True)]), 100)


def q46(e: Engine) -> Frame:
    return e.run(_city_profit(
        e, col("d_dow").isin([6, 0]) & col("d_year").isin([1999, 2000,
                                                           2001]),
        (col("hd_dep_count") == 4) | (col("hd_vehicle_count") == 3)))


def _monthly_rank_lag(e, tbl, pre, group_dims, dim_joins, year):
    """q47/q57 shape: monthly sums with per-group avg + lag/lead via
    row_number self-joins."""
    p = J(S(tbl, f"{pre}_sold_date_sk", f"{pre}_item_sk",
            *[c for j in dim_joins for c in j[2]],
            f"{pre}_sales_price"),
          dd(col("d_year").isin([year - 1, year, year + 1]), "d_year",
             "d_moy"),
          [(f"{pre}_sold_date_sk", "d_date_sk")])
    p = J(p, S("item", "i_item_sk", "i_category", "i_brand"),
          [(f"{pre}_item_sk", "i_item_sk")])
    for (rtbl, on, _cols, rcols) in dim_joins:
        p = J(p, S(rtbl, *rcols), on)
    p = A(p, ["i_category", "i_brand"] + group_dims + ["d_year", "d_moy"],
          [("sum_sales", "sum", col(f"{pre}_sales_price"))])
    p = Window(p, ["i_category", "i_brand"] + group_dims + ["d_year"],
               [("avg_monthly_sales", "avg", "sum_sales")])
    p = Window(p, ["i_category", "i_brand"] + group_dims,
               [("rn", "row_number", None,
                 [("d_year", True), ("d_moy", True)])])
    e.register("q47v1", e.run(p))
    v1 = S("q47v1")
    cur = F(v1, (col("d_year") == year) &
            (col("avg_monthly_sales") > 0) &
            (abs_(col("sum_sales") - col("avg_monthly_sales")) /
             col("avg_monthly_sales") > 0.1))
    prev = P(v1, ("rnp", col("rn") + 1), ("psum", col("sum_sales")),
             *[(f"p_{k}", col(k)) for k in
               ["i_category", "i_brand"] + group_dims])
    nxt = P(v1, ("rnn", col("rn") - 1), ("nsum", col("sum_sales")),
            *[(f"n_{k}", col(k)) for k in
              ["i_category", "i_brand"] + group_dims])
    on_p = [(k, f"p_{k}") for k in ["i_category", "i_brand"] + group_dims]
    on_n = [(k, f"n_{k}") for k in ["i_category", "i_brand"] + group_dims]
    p = J(cur, prev, on_p + [("rn", "rnp")])
    p = J(p, nxt, on_n + [("rn", "rnn")])
    return Limit(Sort(p, [("sum_sales", True), ("d_moy", True)]), 100)


def q47(e: Engine) -> Frame:
    return e.run(_monthly_rank_lag(
        e, "store_sales", "ss", ["s_store_name", "s_company_name"],
        [("store", [("ss_store_sk", "s_store_sk")], ["ss_store_sk"],
          ["s_store_sk", "s_store_name", "s_company_name"])], 2000))


def q48(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_store_sk", "ss_cdemo_sk",
            "ss_addr_sk", "ss_sales_price", "ss_net_profit", "ss_quantity"),
          dd(col("d_year") == 2001, "d_year"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, S("store", "s_store_sk"), [("ss_store_sk", "s_store_sk")])
    p = J(p, S("customer_demographics", "cd_demo_sk", "cd_marital_status",
               "cd_education_status"),
          [("ss_cdemo_sk", "cd_demo_sk")])
    p = J(p, S("customer_address", "ca_address_sk", "ca_country",
               "ca_state"),
          [("ss_addr_sk", "ca_address_sk")])
    b1 = ((col("cd_marital_status") == "M") &
          (col("cd_education_status") == "4 yr Degree") &
          col("ss_sales_price").between(100.0, 150.0))
    b2 = ((col("cd_marital_status") == "D") &
          (col("cd_education_status") == "2 yr Degree") &
          col("ss_sales_price").between(50.0, 100.0))
    b3 = ((col("cd_marital_status") == "S") &
          (col("cd_education_status") == "College") &
          col("ss_sales_price").between(150.0, 200.0))
    g1 = (col("ca_country") == "United States") & \
        col("ca_state").isin(["CO", "OH", "TX"]) & \
        col("ss_net_profit").between(0, 2000)
    g2 = (col("ca_country") == "United States") & \
        col("ca_state").isin(["OR", "MN", "KY"]) & \
        col("ss_net_profit").between(150, 3000)
    g3 = (col("ca_country") == "United States") & \
        col("ca_state").isin(["VA", "CA", "MS"]) & \
        col("ss_net_profit").between(50, 25000)
    p = F(p, (b1 | b2 | b3) & (g1 | g2 | g3))
    return e.run(A(p, [], [("total_qty", "sum", col("ss_quantity"))]))


def q49(e: Engine) -> Frame:
    chans = []
    for tbl, pre, rtbl, rpre, okey, ikey, label in (
            ("web_sales", "ws", "web_returns", "wr", "order_number",
             "item_sk", "web"),
            ("catalog_sales", "cs", "catalog_returns", "cr", "order_number",
             "item_sk", "catalog"),
            ("store_sales", "ss", "store_returns", "sr", "ticket_number",
             "item_sk", "store")):
        amt = "cr_return_amount" if rpre == "cr" else f"{rpre}_return_amt"
        s = F(S(tbl, f"{pre}_sold_date_sk", f"{pre}_{okey}",
                f"{pre}_{ikey}", f"{pre}_quantity", f"{pre}_net_paid",
                f"{pre}_net_profit"),
              (col(f"{pre}_net_profit") > 1) & (col(f"{pre}_net_paid") > 0)
              & (col(f"{pre}_quantity") > 0))
        s = J(s, dd((col("d_year") == 2001) & (col("d_moy") == 12),
                    "d_year", "d_moy"),
              [(f"{pre}_sold_date_sk", "d_date_sk")])
        s = J(s, P(F(S(rtbl, f"{rpre}_{okey}", f"{rpre}_{ikey}",
                       f"{rpre}_return_quantity", amt),
                     col(amt) > 10000),
                   (f"{rpre}_{okey}2", col(f"{rpre}_{okey}")),
                   (f"{rpre}_{ikey}2", col(f"{rpre}_{ikey}")),
                   ("ret_qty", col(f"{rpre}_return_quantity")),
                   ("ret_amt", col(amt))),
              [(f"{pre}_{okey}", f"{rpre}_{okey}2"),
               (f"{pre}_{ikey}", f"{rpre}_{ikey}2")], how="left")
        g = A(s, [f"{pre}_{ikey}"],
              [("rq", "sum", coalesce(col("ret_qty"), lit(0))),
               ("sq", "sum", col(f"{pre}_quantity")),
               ("ra", "sum", coalesce(col("ret_amt"), lit(0.0))),
               ("sa", "sum", col(f"{pre}_net_paid"))])
        g = P(g, ("item", col(f"{pre}_{ikey}")),
              ("return_ratio", col("rq").cast_float() /
               col("sq").cast_float()),
              ("currency_ratio", col("ra") / col("sa")))
        g = Window(g, [], [("return_rank", "rank", None,
                            [("return_ratio", True)]),
                           ("currency_rank", "rank", None,
                            [("currency_ratio", True)])])
        g = F(g, (col("return_rank") <= 10) | (col("currency_rank") <= 10))
        chans.append(P(g, ("channel", lit(label)), ("item", col("item")),
                       ("return_ratio", col("return_ratio")),
                       ("return_rank", col("return_rank")),
                       ("currency_rank", col("currency_rank"))))
    p = Union(chans)
    return e.run(Limit(Sort(p, [("channel", True), ("return_rank", True),
                                ("currency_rank", True),
                                ("item", True)]), 100))


def q50(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_ticket_number",
            "ss_item_sk", "ss_customer_sk", "ss_store_sk"),
          S("store_returns", "sr_returned_date_sk", "sr_ticket_number",
            "sr_item_sk", "sr_customer_sk"),
          [("ss_ticket_number", "sr_ticket_number"),
           ("ss_item_sk", "sr_item_sk"),
           ("ss_customer_sk", "sr_customer_sk")])
    p = J(p, P(dd((col("d_year") == 2001) & (col("d_moy") == 8), "d_year",
                  "d_moy"),
               ("d2sk", col("d_date_sk"))),
          [("sr_returned_date_sk", "d2sk")])
    p = J(p, S("store", "s_store_sk", "s_store_name", "s_company_id",
               "s_city", "s_county", "s_state", "s_zip"),
          [("ss_store_sk", "s_store_sk")])
    lag = col("sr_returned_date_sk") - col("ss_sold_date_sk")
    p = A(p, ["s_store_name", "s_company_id", "s_city", "s_county",
              "s_state", "s_zip"],
          [("d30", "sum", case_when((lag <= 30, lit(1)), otherwise=lit(0))),
           ("d31_60", "sum", case_when(((lag > 30) & (lag <= 60), lit(1)),
                                       otherwise=lit(0))),
           ("d61_90", "sum", case_when(((lag > 60) & (lag <= 90), lit(1)),
                                       otherwise=lit(0))),
           ("d91_120", "sum", case_when(((lag > 90) & (lag <= 120),
                                         lit(1)), otherwise=lit(0))),
           ("dgt120", "sum", case_when((lag > 120, lit(1)),
                                       otherwise=lit(0)))])
    return e.run(Limit(Sort(p, [("s_store_name", True)]), 100))


def q51(e: Engine) -> Frame:
    def cumul(tbl, pre, out):
        p = J(S(tbl, f"{pre}_sold_date_sk", f"{pre}_item_sk",
                f"{pre}_sales_price"),
              dd(col("d_month_seq").between(1200, 1211), "d_month_seq",
                 "d_date"),
              [(f"{pre}_sold_date_sk", "d_date_sk")])
        p = A(p, [f"{pre}_item_sk", "d_date"],
              [("daily", "sum", col(f"{pre}_sales_price"))])
        p = Window(p, [f"{pre}_item_sk"],
                   [("cume", "cumsum", "daily", [("d_date", True)])])
        return P(p, (f"{out}_item", col(f"{pre}_item_sk")),
                 (f"{out}_date", col("d_date")),
                 (f"{out}_cume", col("cume")))
    web = cumul("web_sales", "ws", "web")
    store = cumul("store_sales", "ss", "store")
    p = J(web, store, [("web_item", "store_item"),
                       ("web_date", "store_date")], how="full")
    p = P(p, ("item_sk", coalesce(col("web_item"), col("store_item"))),
          ("d_date", coalesce(col("web_date"), col("store_date"))),
          ("web_sales", col("web_cume")),
          ("store_sales", col("store_cume")))
    p = Window(p, ["item_sk"],
               [("web_cumulative", "max", "web_sales",
                 [("d_date", True)]),
                ("store_cumulative", "max", "store_sales",
                 [("d_date", True)])])
    p = F(p, col("web_cumulative") > col("store_cumulative"))
    return e.run(Limit(Sort(p, [("item_sk", True), ("d_date", True)]), 100))


def q52(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_item_sk",
            "ss_ext_sales_price"),
          dd((col("d_moy") == 11) & (col("d_year") == 2000), "d_moy",
             "d_year"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("item", "i_item_sk", "i_brand_id", "i_brand",
                 "i_manager_id"), col("i_manager_id") == 1),
          [("ss_item_sk", "i_item_sk")])
    p = A(p, ["i_brand_id", "i_brand"],
          [("ext_price", "sum", col("ss_ext_sales_price"))])
    return e.run(Limit(Sort(p, [("ext_price", False),
                                ("i_brand_id", True)]), 100))


def _quarterly_dev(e, key_col, key_filter):
    """q53/q63 shape."""
    p = J(S("store_sales", "ss_sold_date_sk", "ss_item_sk", "ss_store_sk",
            "ss_sales_price"),
          dd(col("d_month_seq").between(1200, 1211), "d_month_seq",
             "d_qoy" if key_col == "i_manufact_id" else "d_moy"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("item", "i_item_sk", "i_manufact_id", "i_manager_id",
                 "i_category", "i_class", "i_brand"), key_filter),
          [("ss_item_sk", "i_item_sk")])
    p = J(p, S("store", "s_store_sk"), [("ss_store_sk", "s_store_sk")])
    tcol = "d_qoy" if key_col == "i_manufact_id" else "d_moy"
    p = A(p, [key_col, tcol], [("sum_sales", "sum", col("ss_sales_price"))])
    p = Window(p, [key_col], [("avg_sales", "avg", "sum_sales")])
    p = F(p, case_when((col("avg_sales") > 0,
                        abs_(col("sum_sales") - col("avg_sales")) /
                        col("avg_sales")), otherwise=lit(None)) > 0.1)
    return Limit(Sort(p, [("avg_sales", True), ("sum_sales", True)]), 100)


def q53(e: Engine) -> Frame:
    return e.run(_quarterly_dev(
        e, "i_manufact_id",
        (col("i_category").isin(["Books", "Children", "Electronics"]) &
         col("i_class").isin(["class00", "class01", "class02"])) |
        (col("i_category").isin(["Women", "Music", "Men"]) &
         col("i_class").isin(["class03", "class04", "class05"]))))


def q54(e: Engine) -> Frame:
    cust = []
    for tbl, pre, cc in (("catalog_sales", "cs", "cs_bill_customer_sk"),
                         ("web_sales", "ws", "ws_bill_customer_sk")):
        p = J(S(tbl, f"{pre}_sold_date_sk", f"{pre}_item_sk", cc),
              dd((col("d_moy") == 12) & (col("d_year") == 1998), "d_moy",
                 "d_year"),
              [(f"{pre}_sold_date_sk", "d_date_sk")])
        p = J(p, F(S("item", "i_item_sk", "i_category", "i_class"),
                   (col("i_category") == "Women") &
                   (col("i_class") == "class03")),
              [(f"{pre}_item_sk", "i_item_sk")])
        cust.append(P(p, ("cust", col(cc))))
    my_customers = Distinct(Union(cust))
    e.register("q54cust", e.run(my_customers))
    mseq = (1998 - 1900) * 12 + 12 - 1  # Dec 1998; window is +1..+3
    rev = J(S("store_sales", "ss_customer_sk", "ss_sold_date_sk",
              "ss_ext_sales_price"),
            dd(col("d_month_seq").between(mseq + 1, mseq + 3),
               "d_month_seq"),
            [("ss_sold_date_sk", "d_date_sk")])
    rev = J(rev, S("q54cust"), [("ss_customer_sk", "cust")], how="semi")
    rev = A(rev, ["ss_customer_sk"],
            [("revenue", "sum", col("ss_ext_sales_price"))])
    seg = P(rev, ("segment", (col("revenue") / 50.0).cast_int()))
    p = A(seg, ["segment"], [("num_customers", "count", None)])
    p = P(p, ("segment", col("segment")),
          ("num_customers", col("num_customers")),
          ("segment_base", col("segment") * 50), extend=False)
    return e.run(Limit(Sort(p, [("segment", True),
                                ("num_customers", True)]), 100))


def q55(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_item_sk",
            "ss_ext_sales_price"),
          dd((col("d_moy") == 11) & (col("d_year") == 1999), "d_moy",
             "d_year"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("item", "i_item_sk", "i_brand_id", "i_brand",
                 "i_manager_id"), col("i_manager_id") == 28),
          [("ss_item_sk", "i_item_sk")])
    p = A(p, ["i_brand_id", "i_brand"],
          [("ext_price", "sum", col("ss_ext_sales_price"))])
    return e.run(Limit(Sort(p, [("ext_price", False),
                                ("i_brand_id", True)]), 100))


def q56(e: Engine) -> Frame:
    return e.run(_manufact_by_channel(
        e, (["i_color"], col("i_color").isin(["azure", "beige", "black"])),
        2, 2001))


def q57(e: Engine) -> Frame:
    return e.run(_monthly_rank_lag(
        e, "catalog_sales", "cs", ["cc_name"],
        [("call_center", [("cs_call_center_sk", "cc_call_center_sk")],
          ["cs_call_center_sk"], ["cc_call_center_sk", "cc_name"])], 2000))


def q58(e: Engine) -> Frame:
    week = e.scalar(P(dd(col("d_date") == date_lit("2000-01-03").value,
                         "d_date", "d_week_seq"),
                      ("wk", col("d_week_seq"))), "wk")
    wk_dates = P(dd(col("d_week_seq") == int(week or 0), "d_week_seq",
                    "d_date"),
                 ("wdsk", col("d_date_sk")))
    e.register("q58w", e.run(wk_dates))
    outs = []
    for tbl, pre in (("store_sales", "ss"), ("catalog_sales", "cs"),
                     ("web_sales", "ws")):
        p = J(S(tbl, f"{pre}_sold_date_sk", f"{pre}_item_sk",
                f"{pre}_ext_sales_price"),
              S("q58w"), [(f"{pre}_sold_date_sk", "wdsk")], how="semi")
        p = J(p, S("item", "i_item_sk", "i_item_id"),
              [(f"{pre}_item_sk", "i_item_sk")])
        outs.append(A(p, ["i_item_id"],
                      [(f"{pre}_rev", "sum",
                        col(f"{pre}_ext_sales_price"))]))
    p = J(outs[0], P(outs[1], ("iid2", col("i_item_id")),
                     ("cs_rev", col("cs_rev"))),
          [("i_item_id", "iid2")])
    p = J(p, P(outs[2], ("iid3", col("i_item_id")),
               ("ws_rev", col("ws_rev"))),
          [("i_item_id", "iid3")])
    avg3 = (col("ss_rev") + col("cs_rev") + col("ws_rev")) / 3.0
    p = F(p, (col("ss_rev").between(0.9 * 1.0 * avg3, 1.1 * avg3)) &
          (col("cs_rev").between(0.9 * avg3, 1.1 * avg3)) &
          (col("ws_rev").between(0.9 * avg3, 1.1 * avg3)))
    return e.run(Limit(Sort(p, [("i_item_id", True), ("ss_rev", True)]),
                       100))


def q59(e: Engine) -> Frame:
    days = ["Sunday", "Monday", "Tuesday", "Wednesday", "Thursday",
            "Friday", "Saturday"]
    wss = day_pivot_sums(
        J(S("store_sales", "ss_sold_date_sk", "ss_store_sk",
            "ss_sales_price"),
          S("date_dim", "d_date_sk", "d_week_seq", "d_day_name"),
          [("ss_sold_date_sk", "d_date_sk")]),
        ["d_week_seq", "ss_store_sk"], col("ss_sales_price"), days)
    e.register("q59wss", e.run(wss))
    wk1 = e.scalar(P(dd(col("d_date") == date_lit("2001-01-01").value,
                        "d_date", "d_week_seq"), ("wk", col("d_week_seq"))),
                   "wk") or 0
    y1 = F(S("q59wss"), col("d_week_seq").between(wk1, wk1 + 51))
    y1 = J(y1, S("store", "s_store_sk", "s_store_id", "s_store_name"),
           [("ss_store_sk", "s_store_sk")])
    y1 = P(y1, ("sid1", col("s_store_id")), ("sname", col("s_store_name")),
           ("wk1", col("d_week_seq")),
           *[(f"{dy.lower()[:3]}1", col(dy.lower()[:3] + "_sales"))
             for dy in days])
    y2 = P(F(S("q59wss"), col("d_week_seq").between(wk1 + 52, wk1 + 103)),
           ("st2", col("ss_store_sk")), ("wk2", col("d_week_seq") - 52),
           *[(f"{dy.lower()[:3]}2", col(dy.lower()[:3] + "_sales"))
             for dy in days])
    y2 = J(y2, P(S("store", "s_store_sk", "s_store_id"),
                 ("s_store_sk", col("s_store_sk")),
                 ("sid2", col("s_store_id"))),
           [("st2", "s_store_sk")])
    p = J(y1, y2, [("sid1", "sid2"), ("wk1", "wk2")])
    p = P(p, ("s_store_name", col("sname")), ("wk", col("wk1")),
          *[(f"r_{dy.lower()[:3]}",
             col(f"{dy.lower()[:3]}1") / col(f"{dy.lower()[:3]}2"))
            for dy in days])
    return e.run(Limit(Sort(p, [("s_store_name", True), ("wk", True)]),
                       100))


def q60(e: Engine) -> Frame:
    return e.run(_manufact_by_channel(
        e, (["i_category"], col("i_category") == "Music"), 9, 1998))


def q61(e: Engine) -> Frame:
    def sales(with_promo):
        p = J(S("store_sales", "ss_sold_date_sk", "ss_item_sk",
                "ss_promo_sk", "ss_customer_sk", "ss_store_sk",
                "ss_ext_sales_price"),
              dd((col("d_year") == 1998) & (col("d_moy") == 11), "d_year",
                 "d_moy"),
              [("ss_sold_date_sk", "d_date_sk")])
        p = J(p, F(S("store", "s_store_sk", "s_gmt_offset"),
                   col("s_gmt_offset") == -5.0),
              [("ss_store_sk", "s_store_sk")])
        p = J(p, F(S("item", "i_item_sk", "i_category"),
                   col("i_category") == "Jewelry"),
              [("ss_item_sk", "i_item_sk")])
        p = J(p, S("customer", "c_customer_sk", "c_current_addr_sk"),
              [("ss_customer_sk", "c_customer_sk")])
        p = J(p, F(S("customer_address", "ca_address_sk", "ca_gmt_offset"),
                   col("ca_gmt_offset") == -5.0),
              [("c_current_addr_sk", "ca_address_sk")])
        if with_promo:
            p = J(p, F(S("promotion", "p_promo_sk", "p_channel_dmail",
                         "p_channel_email", "p_channel_tv"),
                       (col("p_channel_dmail") == "Y") |
                       (col("p_channel_email") == "Y") |
                       (col("p_channel_tv") == "Y")),
                  [("ss_promo_sk", "p_promo_sk")])
        return e.scalar(A(p, [], [("t", "sum", col("ss_ext_sales_price"))]),
                        "t") or 0.0
    promotions = sales(True)
    total = sales(False)
    return e.const_frame(promotions=promotions, total=total,
                         ratio=(promotions / total * 100.0) if total else
                         None)


def q62(e: Engine) -> Frame:
    p = J(S("web_sales", "ws_ship_date_sk", "ws_sold_date_sk",
            "ws_warehouse_sk", "ws_ship_mode_sk", "ws_web_site_sk"),
          dd(col("d_month_seq").between(1200, 1211), "d_month_seq"),
          [("ws_ship_date_sk", "d_date_sk")])
    p = J(p, S("warehouse", "w_warehouse_sk", "w_warehouse_name"),
          [("ws_warehouse_sk", "w_warehouse_sk")])
    p = J(p, S("ship_mode", "sm_ship_mode_sk", "sm_type"),
          [("ws_ship_mode_sk", "sm_ship_mode_sk")])
    p = J(p, S("web_site", "web_site_sk", "web_name"),
          [("ws_web_site_sk", "web_site_sk")])
    lag = col("ws_ship_date_sk") - col("ws_sold_date_sk")
    p = A(p, ["w_warehouse_name", "sm_type", "web_name"],
          [("d30", "sum", case_when((lag <= 30, lit(1)), otherwise=lit(0))),
           ("d31_60", "sum", case_when(((lag > 30) & (lag <= 60), lit(1)),
                                       otherwise=lit(0))),
           ("d61_90", "sum", case_when(((lag > 60) & (lag <= 90), lit(1)),
                                       otherwise=lit(0))),
           ("d91_120", "sum", case_when(((lag > 90) & (lag <= 120),
                                         lit(1)), otherwise=lit(0))),
           ("dgt120", "sum", case_when((lag > 120, lit(1)),
                                       otherwise=lit(0)))])
    return e.run(Limit(Sort(p, [("w_warehouse_name", True),
                                ("sm_type", True), ("web_name", True)]),
                       100))


def q63(e: Engine) -> Frame:
    return e.run(_quarterly_dev(
        e, "i_manager_id",
        (col("i_category").isin(["Books", "Children", "Electronics"]) &
         col("i_class").isin(["class00", "class01", "class02"])) |
        (col("i_category").isin(["Women", "Music", "Men"]) &
         col("i_class").isin(["class03", "class04", "class05"]))))


def q64(e: Engine) -> Frame:
    cs_ui = A(J(S("catalog_sales", "cs_item_sk", "cs_order_number",
                  "cs_ext_list_price"),
                P(S("catalog_returns", "cr_item_sk", "cr_order_number",
                    "cr_refunded_cash", "cr_reversed_charge",
                    "cr_store_credit"),
                  ("cri", col("cr_item_sk")), ("cro", col("cr_order_number")),
                  ("refund", col("cr_refunded_cash") +
                   col("cr_reversed_charge") + col("cr_store_credit"))),
                [("cs_item_sk", "cri"), ("cs_order_number", "cro")]),
              ["cs_item_sk"],
              [("sale", "sum", col("cs_ext_list_price")),
               ("refund", "sum", col("refund"))])
    cs_ui = P(F(cs_ui, col("sale") > col("refund") * 2),
              ("ui_item", col("cs_item_sk")))
    e.register("q64csui", e.run(cs_ui))
    base = J(S("store_sales", "ss_item_sk", "ss_ticket_number",
               "ss_customer_sk", "ss_store_sk", "ss_sold_date_sk",
               "ss_cdemo_sk", "ss_hdemo_sk", "ss_addr_sk",
               "ss_wholesale_cost", "ss_list_price", "ss_coupon_amt"),
             S("store_returns", "sr_item_sk", "sr_ticket_number"),
             [("ss_item_sk", "sr_item_sk"),
              ("ss_ticket_number", "sr_ticket_number")])
    base = J(base, S("q64csui"), [("ss_item_sk", "ui_item")], how="semi")
    base = J(base, S("date_dim", "d_date_sk", "d_year"),
             [("ss_sold_date_sk", "d_date_sk")])
    base = J(base, S("store", "s_store_sk", "s_store_name", "s_zip"),
             [("ss_store_sk", "s_store_sk")])
    base = J(base, S("customer", "c_customer_sk", "c_current_cdemo_sk",
                     "c_current_hdemo_sk", "c_current_addr_sk",
                     "c_first_sales_date_sk", "c_first_shipto_date_sk"),
             [("ss_customer_sk", "c_customer_sk")])
    base = J(base, P(S("customer_demographics", "cd_demo_sk",
                       "cd_marital_status"),
                     ("cd1sk", col("cd_demo_sk")),
                     ("ms1", col("cd_marital_status"))),
             [("ss_cdemo_sk", "cd1sk")])
    base = J(base, P(S("customer_demographics", "cd_demo_sk",
                       "cd_marital_status"),
                     ("cd2sk", col("cd_demo_sk")),
                     ("ms2", col("cd_marital_status"))),
             [("c_current_cdemo_sk", "cd2sk")])
    base = F(base, ~(col("ms1") == col("ms2")))
    base = J(base, P(S("household_demographics", "hd_demo_sk",
                       "hd_income_band_sk"),
                     ("hd1sk", col("hd_demo_sk")),
                     ("ib1", col("hd_income_band_sk"))),
             [("ss_hdemo_sk", "hd1sk")])
    base = J(base, P(S("household_demographics", "hd_demo_sk",
                       "hd_income_band_sk"),
                     ("hd2sk", col("hd_demo_sk")),
                     ("ib2", col("hd_income_band_sk"))),
             [("c_current_hdemo_sk", "hd2sk")])
    base = J(base, P(S("customer_address", "ca_address_sk", "ca_city"),
                     ("ad1", col("ca_address_sk")),
                     ("b_city", col("ca_city"))),
             [("ss_addr_sk", "ad1")])
    base = J(base, P(S("customer_address", "ca_address_sk", "ca_city"),
                     ("ad2", col("ca_address_sk")),
                     ("c_city", col("ca_city"))),
             [("c_current_addr_sk", "ad2")])
    base = J(base, F(S("item", "i_item_sk", "i_product_name",
                       "i_current_price", "i_color"),
                     col("i_color").isin(["azure", "black", "blue",
                                          "blush", "brown", "beige"]) &
                     col("i_current_price").between(10.0, 70.0)),
             [("ss_item_sk", "i_item_sk")])
    g = A(base, ["i_product_name", "ss_item_sk", "s_store_name", "s_zip",
                 "b_city", "c_city", "d_year"],
          [("cnt", "count", None),
           ("s1", "sum", col("ss_wholesale_cost")),
           ("s2", "sum", col("ss_list_price")),
           ("s3", "sum", col("ss_coupon_amt"))])
    e.register("q64cs", e.run(g))
    y1 = P(F(S("q64cs"), col("d_year") == 2000),
           ("pn1", col("i_product_name")), ("it1", col("ss_item_sk")),
           ("st1", col("s_store_name")), ("zip1", col("s_zip")),
           ("cnt1", col("cnt")), ("s11", col("s1")), ("s21", col("s2")),
           ("s31", col("s3")))
    y2 = P(F(S("q64cs"), col("d_year") == 2001),
           ("pn2", col("i_product_name")), ("it2", col("ss_item_sk")),
           ("st2", col("s_store_name")), ("zip2", col("s_zip")),
           ("cnt2", col("cnt")), ("s12", col("s1")), ("s22", col("s2")),
           ("s32", col("s3")))
    p = J(y1, y2, [("it1", "it2"), ("st1", "st2"), ("zip1", "zip2")])
    p = F(p, col("cnt2") <= col("cnt1"))
    return e.run(Limit(Sort(p, [("pn1", True), ("st1", True),
                                ("cnt2", True)]), 100))


def q65(e: Engine) -> Frame:
    win = dd(col("d_month_seq").between(1200, 1211), "d_month_seq")
    sc = A(J(S("store_sales", "ss_store_sk", "ss_item_sk",
               "ss_sold_date_sk", "ss_sales_price"), win,
             [("ss_sold_date_sk", "d_date_sk")]),
           ["ss_store_sk", "ss_item_sk"],
           [("revenue", "sum", col("ss_sales_price"))])
    e.register("q65sc", e.run(sc))
    sb = A(S("q65sc"), ["ss_store_sk"],
           [("ave", "avg", col("revenue"))])
    p = J(S("q65sc"), P(sb, ("sb_store", col("ss_store_sk")),
                        ("ave", col("ave"))),
          [("ss_store_sk", "sb_store")])
    p = F(p, col("revenue") <= col("ave") * 0.1)
    p = J(p, S("store", "s_store_sk", "s_store_name"),
          [("ss_store_sk", "s_store_sk")])
    p = J(p, S("item", "i_item_sk", "i_item_desc", "i_current_price",
               "i_wholesale_cost", "i_brand"),
          [("ss_item_sk", "i_item_sk")])
    return e.run(Limit(Sort(p, [("s_store_name", True),
                                ("i_item_desc", True)]), 100))


def q66(e: Engine) -> Frame:
    chans = []
    for tbl, pre, price, qty in (
            ("web_sales", "ws", "ws_ext_sales_price", "ws_quantity"),
            ("catalog_sales", "cs", "cs_ext_sales_price", "cs_quantity")):
        p = J(S(tbl, f"{pre}_sold_date_sk", f"{pre}_sold_time_sk",
                f"{pre}_warehouse_sk", f"{pre}_ship_mode_sk", price, qty),
              dd(col("d_year") == 2001, "d_year", "d_moy"),
              [(f"{pre}_sold_date_sk", "d_date_sk")])
        p = J(p, F(S("time_dim", "t_time_sk", "t_time"),
                   col("t_time").between(30838, 30838 + 28800)),
              [(f"{pre}_sold_time_sk", "t_time_sk")])
        p = J(p, F(S("ship_mode", "sm_ship_mode_sk", "sm_carrier"),
                   col("sm_carrier").isin(["carrier_1", "carrier_2"])),
              [(f"{pre}_ship_mode_sk", "sm_ship_mode_sk")])
        p = J(p, S("warehouse", "w_warehouse_sk", "w_warehouse_name",
                   "w_warehouse_sq_ft", "w_city", "w_county", "w_state",
                   "w_country"),
              [(f"{pre}_warehouse_sk", "w_warehouse_sk")])
        aggs = []
        for m in range(1, 13):
            aggs.append((f"sales_m{m}", "sum",
                         case_when((col("d_moy") == m, col(price)),
                                   otherwise=lit(0.0))))
            aggs.append((f"net_m{m}", "sum",
                         case_when((col("d_moy") == m,
                                    col(qty).cast_float()),
                                   otherwise=lit(0.0))))
        chans.append(A(p, ["w_warehouse_name", "w_warehouse_sq_ft",
                           "w_city", "w_county", "w_state", "w_country"],
                       aggs))
    p = A(Union(chans), ["w_warehouse_name", "w_warehouse_sq_ft", "w_city",
                         "w_county", "w_state", "w_country"],
          [(f"{k}_m{m}", "sum", col(f"{k}_m{m}"))
           for m in range(1, 13) for k in ("sales", "net")])
    return e.run(Limit(Sort(p, [("w_warehouse_name", True)]), 100))


def q67(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_item_sk", "ss_store_sk",
            "ss_sales_price", "ss_quantity"),
          dd(col("d_month_seq").between(1200, 1211), "d_month_seq",
             "d_year", "d_qoy", "d_moy"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, S("store", "s_store_sk", "s_store_id"),
          [("ss_store_sk", "s_store_sk")])
    p = J(p, S("item", "i_item_sk", "i_category", "i_class", "i_brand",
               "i_product_name"),
          [("ss_item_sk", "i_item_sk")])
    p = A(p, ["i_category", "i_class", "i_brand", "i_product_name",
              "d_year", "d_qoy", "d_moy", "s_store_id"],
          [("sumsales", "sum", col("ss_sales_price") *
            col("ss_quantity").cast_float())], rollup=True)
    # rank on exact cents: the sum is a multiple of 0.01, so quantizing
    # removes float accumulation jitter that would break ties differently
    # between execution orders
    p = Project(p, [("rank_key", (col("sumsales") * 100.0 + 0.5)
                     .cast_int())], extend=True)
    p = Window(p, ["i_category"],
               [("rk", "rank", None, [("rank_key", False)])])
    p = F(p, col("rk") <= 100)
    return e.run(Limit(Sort(p, [("i_category", True), ("rk", True)]), 100))


def q68(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_store_sk", "ss_hdemo_sk",
            "ss_addr_sk", "ss_customer_sk", "ss_ticket_number",
            "ss_ext_sales_price", "ss_ext_list_price", "ss_ext_tax"),
          dd(col("d_dom").between(1, 2) & col("d_year").isin(
              [1999, 2000, 2001]), "d_dom", "d_year"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("store", "s_store_sk", "s_city"),
               col("s_city").isin(["city_00", "city_01"])),
          [("ss_store_sk", "s_store_sk")])
    p = J(p, F(S("household_demographics", "hd_demo_sk", "hd_dep_count",
                 "hd_vehicle_count"),
               (col("hd_dep_count") == 4) | (col("hd_vehicle_count") == 3)),
          [("ss_hdemo_sk", "hd_demo_sk")])
    p = J(p, P(S("customer_address", "ca_address_sk", "ca_city"),
               ("ca_address_sk", col("ca_address_sk")),
               ("bought_city", col("ca_city"))),
          [("ss_addr_sk", "ca_address_sk")])
    p = A(p, ["ss_ticket_number", "ss_customer_sk", "bought_city"],
          [("extended_price", "sum", col("ss_ext_sales_price")),
           ("list_price", "sum", col("ss_ext_list_price")),
           ("extended_tax", "sum", col("ss_ext_tax"))])
    p = J(p, S("customer", "c_customer_sk", "c_current_addr_sk",
               "c_last_name", "c_first_name"),
          [("ss_customer_sk", "c_customer_sk")])
    p = J(p, P(S("customer_address", "ca_address_sk", "ca_city"),
               ("ca2", col("ca_address_sk")),
               ("current_city", col("ca_city"))),
          [("c_current_addr_sk", "ca2")])
    p = F(p, ~(col("current_city") == col("bought_city")))
    return e.run(Limit(Sort(p, [("c_last_name", True),
                                ("ss_ticket_number", True)]), 100))


def q69(e: Engine) -> Frame:
    win = dd((col("d_year") == 2001) & col("d_moy").between(4, 6),
             "d_year", "d_moy")
    c = J(S("customer", "c_customer_sk", "c_current_addr_sk",
            "c_current_cdemo_sk"),
          F(S("customer_address", "ca_address_sk", "ca_state"),
            col("ca_state").isin(["KY", "GA", "NM"])),
          [("c_current_addr_sk", "ca_address_sk")])
    ss = J(S("store_sales", "ss_customer_sk", "ss_sold_date_sk"), win,
           [("ss_sold_date_sk", "d_date_sk")])
    c = J(c, P(ss, ("cust", col("ss_customer_sk"))),
          [("c_customer_sk", "cust")], how="semi")
    ws = J(S("web_sales", "ws_bill_customer_sk", "ws_sold_date_sk"), win,
           [("ws_sold_date_sk", "d_date_sk")])
    c = J(c, P(ws, ("cust", col("ws_bill_customer_sk"))),
          [("c_customer_sk", "cust")], how="anti")
    cs = J(S("catalog_sales", "cs_ship_customer_sk", "cs_sold_date_sk"),
           win, [("cs_sold_date_sk", "d_date_sk")])
    c = J(c, P(cs, ("cust", col("cs_ship_customer_sk"))),
          [("c_customer_sk", "cust")], how="anti")
    p = J(c, S("customer_demographics", "cd_demo_sk", "cd_gender",
               "cd_marital_status", "cd_education_status",
               "cd_purchase_estimate", "cd_credit_rating"),
          [("c_current_cdemo_sk", "cd_demo_sk")])
    p = A(p, ["cd_gender", "cd_marital_status", "cd_education_status",
              "cd_purchase_estimate", "cd_credit_rating"],
          [("cnt1", "count", None)])
    return e.run(Limit(Sort(p, [("cd_gender", True),
                                ("cd_marital_status", True),
                                ("cd_education_status", True)]), 100))


def q70(e: Engine) -> Frame:
    base = J(S("store_sales", "ss_sold_date_sk", "ss_store_sk",
               "ss_net_profit"),
             dd(col("d_month_seq").between(1200, 1211), "d_month_seq"),
             [("ss_sold_date_sk", "d_date_sk")])
    by_state = A(J(base, S("store", "s_store_sk", "s_state"),
                   [("ss_store_sk", "s_store_sk")]),
                 ["s_state"], [("profit", "sum", col("ss_net_profit"))])
    top5 = Window(by_state, [], [("rk", "rank", None,
                                  [("profit", False)])])
    top5 = P(F(top5, col("rk") <= 5), ("top_state", col("s_state")))
    e.register("q70top", e.run(top5))
    p = J(base, S("store", "s_store_sk", "s_state", "s_county"),
          [("ss_store_sk", "s_store_sk")])
    p = J(p, S("q70top"), [("s_state", "top_state")], how="semi")
    p = A(p, ["s_state", "s_county"],
          [("total_sum", "sum", col("ss_net_profit"))], rollup=True)
    p = P(p, ("s_state", col("s_state")), ("s_county", col("s_county")),
          ("lochierarchy", lit(2) - col("__lvl")),
          ("total_sum", col("total_sum")),
          ("rank_state", case_when((col("__lvl") == 2, col("s_state")),
                                   otherwise=lit(None))))
    p = Window(p, ["lochierarchy", "rank_state"],
               [("rank_within_parent", "rank", None,
                 [("total_sum", False)])])
    return e.run(Limit(Sort(p, [("lochierarchy", False),
                                ("s_state", True),
                                ("rank_within_parent", True)]), 100))


def q71(e: Engine) -> Frame:
    chans = []
    for tbl, pre in (("web_sales", "ws"), ("catalog_sales", "cs"),
                     ("store_sales", "ss")):
        p = J(S(tbl, f"{pre}_sold_date_sk", f"{pre}_sold_time_sk",
                f"{pre}_item_sk", f"{pre}_ext_sales_price"),
              dd((col("d_moy") == 11) & (col("d_year") == 1999), "d_moy",
                 "d_year"),
              [(f"{pre}_sold_date_sk", "d_date_sk")])
        chans.append(P(p, ("ext_price", col(f"{pre}_ext_sales_price")),
                       ("item_sk", col(f"{pre}_item_sk")),
                       ("time_sk", col(f"{pre}_sold_time_sk"))))
    p = J(Union(chans), F(S("item", "i_item_sk", "i_brand_id", "i_brand",
                            "i_manager_id"), col("i_manager_id") == 1),
          [("item_sk", "i_item_sk")])
    p = J(p, F(S("time_dim", "t_time_sk", "t_hour", "t_minute",
                 "t_meal_time"),
               (col("t_meal_time") == "breakfast") |
               (col("t_meal_time") == "dinner")),
          [("time_sk", "t_time_sk")])
    p = A(p, ["i_brand_id", "i_brand", "t_hour", "t_minute"],
          [("ext_price", "sum", col("ext_price"))])
    return e.run(Limit(Sort(p, [("ext_price", False),
                                ("i_brand_id", True)]), 100))


def q72(e: Engine) -> Frame:
    p = J(S("catalog_sales", "cs_sold_date_sk", "cs_ship_date_sk",
            "cs_item_sk", "cs_order_number", "cs_quantity",
            "cs_bill_cdemo_sk", "cs_bill_hdemo_sk", "cs_promo_sk"),
          dd(col("d_year") == 2001, "d_year", "d_week_seq", "d_date"),
          [("cs_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("customer_demographics", "cd_demo_sk",
                 "cd_marital_status"), col("cd_marital_status") == "D"),
          [("cs_bill_cdemo_sk", "cd_demo_sk")])
    p = J(p, F(S("household_demographics", "hd_demo_sk",
                 "hd_buy_potential"),
               col("hd_buy_potential") == ">10000"),
          [("cs_bill_hdemo_sk", "hd_demo_sk")])
    # inventory joined on (item, week): inventory date -> week via date_dim
    inv = J(S("inventory", "inv_date_sk", "inv_item_sk",
              "inv_quantity_on_hand"),
            P(S("date_dim", "d_date_sk", "d_week_seq"),
              ("invdsk", col("d_date_sk")), ("inv_week", col("d_week_seq"))),
            [("inv_date_sk", "invdsk")])
    p = J(p, P(inv, ("inv_item", col("inv_item_sk")),
               ("inv_week", col("inv_week")),
               ("inv_qoh", col("inv_quantity_on_hand"))),
          [("cs_item_sk", "inv_item"), ("d_week_seq", "inv_week")])
    p = F(p, (col("inv_qoh") < col("cs_quantity")) &
          (col("cs_ship_date_sk") > col("cs_sold_date_sk") + 5))
    p = J(p, S("item", "i_item_sk", "i_item_desc"),
          [("cs_item_sk", "i_item_sk")])
    p = J(p, S("warehouse", "w_warehouse_sk", "w_warehouse_name"), [],
          how="inner") if False else p
    p = J(p, P(S("promotion", "p_promo_sk"), ("pp", col("p_promo_sk")),
               ("has_promo", lit(1))),
          [("cs_promo_sk", "pp")], how="left")
    p = J(p, P(S("catalog_returns", "cr_item_sk", "cr_order_number"),
               ("cri", col("cr_item_sk")), ("cro", col("cr_order_number")),
               ("returned", lit(1))),
          [("cs_item_sk", "cri"), ("cs_order_number", "cro")], how="left")
    p = A(p, ["i_item_desc", "d_week_seq"],
          [("no_promo", "sum", case_when((is_null(col("has_promo")),
                                          lit(1)), otherwise=lit(0))),
           ("promo", "sum", case_when((is_not_null(col("has_promo")),
                                       lit(1)), otherwise=lit(0))),
           ("total_cnt", "count", None)])
    return e.run(Limit(Sort(p, [("total_cnt", False),
                                ("i_item_desc", True),
                                ("d_week_seq", True)]), 100))


def q73(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_store_sk", "ss_hdemo_sk",
            "ss_customer_sk", "ss_ticket_number"),
          dd((col("d_dom").between(1, 2)) & col("d_year").isin(
              [1999, 2000, 2001]), "d_dom", "d_year"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("store", "s_store_sk", "s_county"),
               col("s_county").isin(["county_00", "county_01", "county_02",
                                     "county_03"])),
          [("ss_store_sk", "s_store_sk")])
    p = J(p, F(S("household_demographics", "hd_demo_sk",
                 "hd_buy_potential", "hd_vehicle_count", "hd_dep_count"),
               col("hd_buy_potential").isin([">10000", "Unknown"]) &
               (col("hd_vehicle_count") > 0) &
               (col("hd_dep_count").cast_float() /
                col("hd_vehicle_count").cast_float() > 1.0)),
          [("ss_hdemo_sk", "hd_demo_sk")])
    p = F(A(p, ["ss_ticket_number", "ss_customer_sk"],
            [("cnt", "count", None)]), col("cnt").between(1, 5))
    p = J(p, S("customer", "c_customer_sk", "c_last_name", "c_first_name",
               "c_salutation", "c_preferred_cust_flag"),
          [("ss_customer_sk", "c_customer_sk")])
    return e.run(Limit(Sort(p, [("cnt", False), ("c_last_name", True)]),
                       1000))


def q74(e: Engine) -> Frame:
    def yt(tbl, pre, cc, year):
        p = J(S(tbl, f"{pre}_sold_date_sk", cc, f"{pre}_net_paid"),
              dd(col("d_year") == year, "d_year"),
              [(f"{pre}_sold_date_sk", "d_date_sk")])
        p = J(p, S("customer", "c_customer_sk", "c_customer_id",
                   "c_first_name", "c_last_name"),
              [(cc, "c_customer_sk")])
        return A(p, ["c_customer_id", "c_first_name", "c_last_name"],
                 [("year_total", "sum", col(f"{pre}_net_paid"))])
    e.register("t74_s1", e.run(yt("store_sales", "ss", "ss_customer_sk",
                                  2001)))
    e.register("t74_s2", e.run(yt("store_sales", "ss", "ss_customer_sk",
                                  2002)))
    e.register("t74_w1", e.run(yt("web_sales", "ws", "ws_bill_customer_sk",
                                  2001)))
    e.register("t74_w2", e.run(yt("web_sales", "ws", "ws_bill_customer_sk",
                                  2002)))
    p = J(P(F(S("t74_s1"), col("year_total") > 0), ("cid", col("c_customer_id")),
            ("fn", col("c_first_name")), ("ln", col("c_last_name")),
            ("ts1", col("year_total"))),
          P(S("t74_s2"), ("cid2", col("c_customer_id")),
            ("ts2", col("year_total"))),
          [("cid", "cid2")])
    p = J(p, P(F(S("t74_w1"), col("year_total") > 0),
               ("cid3", col("c_customer_id")), ("tw1", col("year_total"))),
          [("cid", "cid3")])
    p = J(p, P(S("t74_w2"), ("cid4", col("c_customer_id")),
               ("tw2", col("year_total"))),
          [("cid", "cid4")])
    p = F(p, col("tw2") / col("tw1") > col("ts2") / col("ts1"))
    p = P(p, ("customer_id", col("cid")), ("customer_first_name", col("fn")),
          ("customer_last_name", col("ln")))
    return e.run(Limit(Sort(p, [("customer_id", True)]), 100))


def q75(e: Engine) -> Frame:
    chans = []
    for tbl, pre, rtbl, rpre, okey in (
            ("store_sales", "ss", "store_returns", "sr", "ticket_number"),
            ("catalog_sales", "cs", "catalog_returns", "cr",
             "order_number"),
            ("web_sales", "ws", "web_returns", "wr", "order_number")):
        amt = ("cr_return_amount" if rpre == "cr" else
               f"{rpre}_return_amt")
        p = J(S(tbl, f"{pre}_sold_date_sk", f"{pre}_item_sk",
                f"{pre}_{okey}", f"{pre}_quantity", f"{pre}_ext_sales_price"),
              dd(col("d_year").isin([2001, 2002]), "d_year"),
              [(f"{pre}_sold_date_sk", "d_date_sk")])
        p = J(p, F(S("item", "i_item_sk", "i_brand_id", "i_class_id",
                     "i_category_id", "i_category", "i_manufact_id"),
                   col("i_category") == "Books"),
              [(f"{pre}_item_sk", "i_item_sk")])
        p = J(p, P(S(rtbl, f"{rpre}_item_sk", f"{rpre}_{okey}",
                     f"{rpre}_return_quantity", amt),
                   ("ri", col(f"{rpre}_item_sk")),
                   ("ro", col(f"{rpre}_{okey}")),
                   ("rq", col(f"{rpre}_return_quantity")),
                   ("ra", col(amt))),
              [(f"{pre}_item_sk", "ri"), (f"{pre}_{okey}", "ro")],
              how="left")
        chans.append(P(p, ("d_year", col("d_year")),
                       ("i_brand_id", col("i_brand_id")),
                       ("i_class_id", col("i_class_id")),
                       ("i_category_id", col("i_category_id")),
                       ("i_manufact_id", col("i_manufact_id")),
                       ("qty", col(f"{pre}_quantity") -
                        coalesce(col("rq"), lit(0))),
                       ("amt", col(f"{pre}_ext_sales_price") -
                        coalesce(col("ra"), lit(0.0)))))
    total = A(Union(chans), ["d_year", "i_brand_id", "i_class_id",
                             "i_category_id", "i_manufact_id"],
              [("sales_cnt", "sum", col("qty")),
               ("sales_amt", "sum", col("amt"))])
    e.register("q75all", e.run(total))
    cur = P(F(S("q75all"), col("d_year") == 2002),
            ("b2", col("i_brand_id")), ("c2", col("i_class_id")),
            ("g2", col("i_category_id")), ("m2", col("i_manufact_id")),
            ("cnt2", col("sales_cnt")), ("amt2", col("sales_amt")))
    prev = P(F(S("q75all"), col("d_year") == 2001),
             ("b1", col("i_brand_id")), ("c1", col("i_class_id")),
             ("g1", col("i_category_id")), ("m1", col("i_manufact_id")),
             ("cnt1", col("sales_cnt")), ("amt1", col("sales_amt")))
    p = J(cur, prev, [("b2", "b1"), ("c2", "c1"), ("g2", "g1"),
                      ("m2", "m1")])
    p = F(p, col("cnt2").cast_float() / col("cnt1").cast_float() < 0.9)
    p = P(p, ("prev_year", lit(2001)), ("year_", lit(2002)),
          ("i_brand_id", col("b2")), ("i_class_id", col("c2")),
          ("i_category_id", col("g2")), ("i_manufact_id", col("m2")),
          ("prev_cnt", col("cnt1")), ("curr_cnt", col("cnt2")),
          ("sales_cnt_diff", col("cnt2") - col("cnt1")),
          ("sales_amt_diff", col("amt2") - col("amt1")))
    return e.run(Limit(Sort(p, [("sales_cnt_diff", True)]), 100))


def q76(e: Engine) -> Frame:
    chans = []
    specs = (("store_sales", "ss", "ss_store_sk", "ss_item_sk",
              "ss_ext_sales_price", "ss_sold_date_sk", "store"),
             ("web_sales", "ws", "ws_ship_customer_sk", "ws_item_sk",
              "ws_ext_sales_price", "ws_sold_date_sk", "web"),
             ("catalog_sales", "cs", "cs_ship_addr_sk", "cs_item_sk",
              "cs_ext_sales_price", "cs_sold_date_sk", "catalog"))
    for tbl, pre, nullcol, isk, price, datecol, label in specs:
        p = F(S(tbl, nullcol, isk, price, datecol),
              is_null(col(nullcol)))
        p = J(p, S("item", "i_item_sk", "i_category"),
              [(isk, "i_item_sk")])
        p = J(p, S("date_dim", "d_date_sk", "d_year", "d_qoy"),
              [(datecol, "d_date_sk")])
        chans.append(P(p, ("channel", lit(label)),
                       ("col_name", lit(nullcol)),
                       ("d_year", col("d_year")), ("d_qoy", col("d_qoy")),
                       ("i_category", col("i_category")),
                       ("ext_sales_price", col(price))))
    p = A(Union(chans), ["channel", "col_name", "d_year", "d_qoy",
                         "i_category"],
          [("sales_cnt", "count", None),
           ("sales_amt", "sum", col("ext_sales_price"))])
    return e.run(Limit(Sort(p, [("channel", True), ("col_name", True),
                                ("d_year", True), ("d_qoy", True),
                                ("i_category", True)]), 100))


def q77(e: Engine) -> Frame:
    lo, hi = dsk("2000-08-23"), dsk("2000-09-22")
    win = lambda c: col(c).between(lo, hi)
    ss = A(J(F(S("store_sales", "ss_sold_date_sk", "ss_store_sk",
               "ss_ext_sales_price", "ss_net_profit"),
              win("ss_sold_date_sk")),
            S("store", "s_store_sk"), [("ss_store_sk", "s_store_sk")]),
           ["ss_store_sk"],
           [("sales", "sum", col("ss_ext_sales_price")),
            ("profit", "sum", col("ss_net_profit"))])
    sr = A(J(F(S("store_returns", "sr_returned_date_sk", "sr_store_sk",
               "sr_return_amt", "sr_net_loss"),
              win("sr_returned_date_sk")),
            S("store", "s_store_sk"), [("sr_store_sk", "s_store_sk")]),
           ["sr_store_sk"],
           [("returns_", "sum", col("sr_return_amt")),
            ("profit_loss", "sum", col("sr_net_loss"))])
    st = J(ss, P(sr, ("srsk", col("sr_store_sk")),
                 ("returns_", col("returns_")),
                 ("profit_loss", col("profit_loss"))),
           [("ss_store_sk", "srsk")], how="left")
    st = P(st, ("channel", lit("store channel")), ("id", col("ss_store_sk")),
           ("sales", col("sales")),
           ("returns_", coalesce(col("returns_"), lit(0.0))),
           ("profit", col("profit") - coalesce(col("profit_loss"),
                                               lit(0.0))))
    cs_sales = e.run(A(F(S("catalog_sales", "cs_sold_date_sk",
                           "cs_call_center_sk", "cs_ext_sales_price",
                           "cs_net_profit"), win("cs_sold_date_sk")),
                       ["cs_call_center_sk"],
                       [("sales", "sum", col("cs_ext_sales_price")),
                        ("profit", "sum", col("cs_net_profit"))]))
    e.register("q77cs", cs_sales)
    cr_tot = e.run(A(F(S("catalog_returns", "cr_returned_date_sk",
                         "cr_return_amount", "cr_net_loss"),
                       win("cr_returned_date_sk")), [],
                     [("returns_", "sum", col("cr_return_amount")),
                      ("profit_loss", "sum", col("cr_net_loss"))]))
    rr = cr_tot.to_rows()
    cr_r, cr_l = (rr[0] if rr else (0.0, 0.0))
    ct = P(S("q77cs"), ("channel", lit("catalog channel")),
           ("id", col("cs_call_center_sk")), ("sales", col("sales")),
           ("returns_", lit(float(cr_r or 0.0))),
           ("profit", col("profit") - lit(float(cr_l or 0.0))))
    ws = A(J(F(S("web_sales", "ws_sold_date_sk", "ws_web_page_sk",
               "ws_ext_sales_price", "ws_net_profit"),
              win("ws_sold_date_sk")),
            S("web_page", "wp_web_page_sk"),
            [("ws_web_page_sk", "wp_web_page_sk")]),
           ["ws_web_page_sk"],
           [("sales", "sum", col("ws_ext_sales_price")),
            ("profit", "sum", col("ws_net_profit"))])
    wr = A(J(F(S("web_returns", "wr_returned_date_sk", "wr_web_page_sk",
               "wr_return_amt", "wr_net_loss"),
              win("wr_returned_date_sk")),
            S("web_page", "wp_web_page_sk"),
            [("wr_web_page_sk", "wp_web_page_sk")]),
           ["wr_web_page_sk"],
           [("returns_", "sum", col("wr_return_amt")),
            ("profit_loss", "sum", col("wr_net_loss"))])
    wt = J(ws, P(wr, ("wrsk", col("wr_web_page_sk")),
                 ("returns_", col("returns_")),
                 ("profit_loss", col("profit_loss"))),
           [("ws_web_page_sk", "wrsk")], how="left")
    wt = P(wt, ("channel", lit("web channel")), ("id", col("ws_web_page_sk")),
           ("sales", col("sales")),
           ("returns_", coalesce(col("returns_"), lit(0.0))),
           ("profit", col("profit") - coalesce(col("profit_loss"),
                                               lit(0.0))))
    p = A(Union([st, ct, wt]), ["channel", "id"],
          [("sales", "sum", col("sales")),
           ("returns_", "sum", col("returns_")),
           ("profit", "sum", col("profit"))], rollup=True)
    return e.run(Limit(Sort(p, [("__lvl", True), ("channel", True),
                                ("id", True)]), 100))


def q78(e: Engine) -> Frame:
    def no_return(tbl, pre, rtbl, rpre, okey, cust):
        p = J(S(tbl, f"{pre}_sold_date_sk", f"{pre}_item_sk", cust,
                f"{pre}_{okey}", f"{pre}_quantity", f"{pre}_wholesale_cost",
                f"{pre}_sales_price"),
              P(S(rtbl, f"{rpre}_item_sk", f"{rpre}_{okey}"),
                ("ri", col(f"{rpre}_item_sk")),
                ("ro", col(f"{rpre}_{okey}"))),
              [(f"{pre}_item_sk", "ri"), (f"{pre}_{okey}", "ro")],
              how="anti")
        p = J(p, S("date_dim", "d_date_sk", "d_year"),
              [(f"{pre}_sold_date_sk", "d_date_sk")])
        # d_year is a grouping key and every consumer filters d_year=2000,
        # so the filter commutes below the aggregate (Catalyst does the
        # same pushdown) — cuts the ~250M-group aggregation input 5x
        p = F(p, col("d_year") == 2000)
        return A(p, ["d_year", f"{pre}_item_sk", cust],
                 [(f"{pre}_qty", "sum", col(f"{pre}_quantity")),
                  (f"{pre}_wc", "sum", col(f"{pre}_wholesale_cost")),
                  (f"{pre}_sp", "sum", col(f"{pre}_sales_price"))])
    ss = no_return("store_sales", "ss", "store_returns", "sr",
                   "ticket_number", "ss_customer_sk")
    ws = no_return("web_sales", "ws", "web_returns", "wr", "order_number",
                   "ws_bill_customer_sk")
    cs = no_return("catalog_sales", "cs", "catalog_returns", "cr",
                   "order_number", "cs_bill_customer_sk")
    p = J(F(ss, col("d_year") == 2000),
          P(F(ws, col("d_year") == 2000), ("wyi", col("ws_item_sk")),
            ("wyc", col("ws_bill_customer_sk")), ("ws_qty", col("ws_qty")),
            ("ws_wc", col("ws_wc")), ("ws_sp", col("ws_sp"))),
          [("ss_item_sk", "wyi"), ("ss_customer_sk", "wyc")])
    p = J(p, P(F(cs, col("d_year") == 2000), ("cyi", col("cs_item_sk")),
               ("cyc", col("cs_bill_customer_sk")),
               ("cs_qty", col("cs_qty")), ("cs_wc", col("cs_wc")),
               ("cs_sp", col("cs_sp"))),
          [("ss_item_sk", "cyi"), ("ss_customer_sk", "cyc")], how="left")
    p = F(p, coalesce(col("ws_qty"), lit(0)) +
          coalesce(col("cs_qty"), lit(0)) > 0)
    p = P(p, ("d_year", col("d_year")), ("ss_item_sk", col("ss_item_sk")),
          ("ss_customer_sk", col("ss_customer_sk")),
          ("ratio", col("ss_qty").cast_float() /
           (coalesce(col("ws_qty"), lit(0)) +
            coalesce(col("cs_qty"), lit(0))).cast_float()),
          ("store_qty", col("ss_qty")),
          ("store_wholesale_cost", col("ss_wc")),
          ("store_sales_price", col("ss_sp")),
          ("other_chan_qty", coalesce(col("ws_qty"), lit(0)) +
           coalesce(col("cs_qty"), lit(0))))
    return e.run(Limit(Sort(p, [("ratio", True), ("ss_qty", True)
                                if False else ("store_qty", True)]), 100))


def q79(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_date_sk", "ss_store_sk", "ss_hdemo_sk",
            "ss_customer_sk", "ss_ticket_number", "ss_addr_sk",
            "ss_coupon_amt", "ss_net_profit"),
          dd((col("d_dow") == 1) & col("d_year").isin([1999, 2000, 2001]),
             "d_dow", "d_year"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("store", "s_store_sk", "s_number_employees", "s_city"),
               col("s_number_employees").between(200, 295)),
          [("ss_store_sk", "s_store_sk")])
    p = J(p, F(S("household_demographics", "hd_demo_sk", "hd_dep_count",
                 "hd_vehicle_count"),
               (col("hd_dep_count") == 6) | (col("hd_vehicle_count") > 2)),
          [("ss_hdemo_sk", "hd_demo_sk")])
    p = A(p, ["ss_ticket_number", "ss_customer_sk", "s_city"],
          [("amt", "sum", col("ss_coupon_amt")),
           ("profit", "sum", col("ss_net_profit"))])
    p = J(p, S("customer", "c_customer_sk", "c_last_name", "c_first_name"),
          [("ss_customer_sk", "c_customer_sk")])
    return e.run(Limit(Sort(p, [("c_last_name", True),
                                ("c_first_name", True),
                                ("ss_ticket_number", True)]), 100))


def q80(e: Engine) -> Frame:
    lo, hi = dsk("2000-08-23"), dsk("2000-09-22")
    chans = []
    for (tbl, pre, rtbl, rpre, okey, sitecol, sitetbl, sitesk, siteid,
         label) in (
            ("store_sales", "ss", "store_returns", "sr", "ticket_number",
             "ss_store_sk", "store", "s_store_sk", "s_store_id", "store"),
            ("catalog_sales", "cs", "catalog_returns", "cr",
             "order_number", "cs_catalog_page_sk", "catalog_page",
             "cp_catalog_page_sk", "cp_catalog_page_id", "catalog"),
            ("web_sales", "ws", "web_returns", "wr", "order_number",
             "ws_web_site_sk", "web_site", "web_site_sk", "web_site_id",
             "web")):
        amt = ("cr_return_amount" if rpre == "cr" else
               f"{rpre}_return_amt")
        p = J(F(S(tbl, f"{pre}_sold_date_sk", f"{pre}_item_sk",
                  f"{pre}_{okey}", f"{pre}_promo_sk", sitecol,
                  f"{pre}_ext_sales_price", f"{pre}_net_profit"),
                col(f"{pre}_sold_date_sk").between(lo, hi)),
              F(S("item", "i_item_sk", "i_current_price"),
                col("i_current_price") > 50.0),
              [(f"{pre}_item_sk", "i_item_sk")])
        p = J(p, F(S("promotion", "p_promo_sk", "p_channel_tv"),
                   col("p_channel_tv") == "N"),
              [(f"{pre}_promo_sk", "p_promo_sk")])
        p = J(p, P(S(rtbl, f"{rpre}_item_sk", f"{rpre}_{okey}", amt,
                     f"{rpre}_net_loss"),
                   ("ri", col(f"{rpre}_item_sk")),
                   ("ro", col(f"{rpre}_{okey}")),
                   ("ramt", col(amt)), ("rloss", col(f"{rpre}_net_loss"))),
              [(f"{pre}_item_sk", "ri"), (f"{pre}_{okey}", "ro")],
              how="left")
        p = J(p, S(sitetbl, sitesk, siteid), [(sitecol, sitesk)])
        g = A(p, [siteid],
              [("sales", "sum", col(f"{pre}_ext_sales_price")),
               ("returns_", "sum", coalesce(col("ramt"), lit(0.0))),
               ("profit", "sum", col(f"{pre}_net_profit") -
                coalesce(col("rloss"), lit(0.0)))])
        chans.append(P(g, ("channel", lit(label + " channel")),
                       ("id", col(siteid)), ("sales", col("sales")),
                       ("returns_", col("returns_")),
                       ("profit", col("profit"))))
    p = A(Union(chans), ["channel", "id"],
          [("sales", "sum", col("sales")),
           ("returns_", "sum", col("returns_")),
           ("profit", "sum", col("profit"))], rollup=True)
    return e.run(Limit(Sort(p, [("__lvl", True), ("channel", True),
                                ("id", True)]), 100))


def q81(e: Engine) -> Frame:
    ctr = J(S("catalog_returns", "cr_returned_date_sk",
              "cr_returning_customer_sk", "cr_returning_addr_sk",
              "cr_return_amt_inc_tax"),
            dd(col("d_year") == 2000, "d_year"),
            [("cr_returned_date_sk", "d_date_sk")])
    ctr = J(ctr, S("customer_address", "ca_address_sk", "ca_state"),
            [("cr_returning_addr_sk", "ca_address_sk")])
    ctr = A(ctr, ["cr_returning_customer_sk", "ca_state"],
            [("ctr_total_return", "sum", col("cr_return_amt_inc_tax"))])
    e.register("ctr81", e.run(ctr))
    st_avg = A(S("ctr81"), ["ca_state"],
               [("avg_ret", "avg", col("ctr_total_return"))])
    p = J(S("ctr81"), P(st_avg, ("st2", col("ca_state")),
                        ("avg_ret", col("avg_ret"))),
          [("ca_state", "st2")])
    p = F(p, col("ctr_total_return") > col("avg_ret") * 1.2)
    cust = J(S("customer", "c_customer_sk", "c_customer_id",
               "c_salutation", "c_first_name", "c_last_name",
               "c_current_addr_sk"),
             F(S("customer_address", "ca_address_sk", "ca_state",
                 "ca_city", "ca_zip", "ca_country", "ca_location_type"),
               col("ca_state") == "GA"),
             [("c_current_addr_sk", "ca_address_sk")])
    cust = P(cust, ("c_customer_sk", col("c_customer_sk")),
             ("c_customer_id", col("c_customer_id")),
             ("c_salutation", col("c_salutation")),
             ("c_first_name", col("c_first_name")),
             ("c_last_name", col("c_last_name")),
             ("cust_city", col("ca_city")), ("cust_zip", col("ca_zip")),
             ("cust_country", col("ca_country")),
             ("cust_loc", col("ca_location_type")))
    p = J(p, cust, [("cr_returning_customer_sk", "c_customer_sk")])
    p = P(p, ("c_customer_id", col("c_customer_id")),
          ("c_salutation", col("c_salutation")),
          ("c_first_name", col("c_first_name")),
          ("c_last_name", col("c_last_name")),
          ("ca_city", col("cust_city")), ("ca_zip", col("cust_zip")),
          ("ca_country", col("cust_country")),
          ("ca_location_type", col("cust_loc")),
          ("ctr_total_return", col("ctr_total_return")))
    return e.run(Limit(Sort(p, [("c_customer_id", True)]), 100))


def q82(e: Engine) -> Frame:
    lo = dsk("2000-05-25")
    it = F(S("item", "i_item_sk", "i_item_id", "i_item_desc",
             "i_current_price", "i_manufact_id"),
           col("i_current_price").between(62.0, 92.0) &
           col("i_manufact_id").isin([12, 13, 14, 15]))
    p = J(it, F(S("inventory", "inv_item_sk", "inv_date_sk",
                  "inv_quantity_on_hand"),
                col("inv_quantity_on_hand").between(100, 500) &
                col("inv_date_sk").between(lo, lo + 60)),
          [("i_item_sk", "inv_item_sk")])
    p = J(p, P(S("store_sales", "ss_item_sk"), ("ssk", col("ss_item_sk"))),
          [("i_item_sk", "ssk")], how="semi")
    p = Distinct(P(p, ("i_item_id", col("i_item_id")),
                   ("i_item_desc", col("i_item_desc")),
                   ("i_current_price", col("i_current_price"))))
    return e.run(Limit(Sort(p, [("i_item_id", True)]), 100))


def q83(e: Engine) -> Frame:
    dates = [date_lit("2000-06-30").value, date_lit("2000-09-27").value,
             date_lit("2000-11-17").value]
    weeks = Distinct(P(F(S("date_dim", "d_date_sk", "d_date",
                           "d_week_seq"), col("d_date").isin(dates)),
                       ("wk", col("d_week_seq"))))
    e.register("q83w", e.run(weeks))
    wk_dates = J(S("date_dim", "d_date_sk", "d_week_seq"),
                 S("q83w"), [("d_week_seq", "wk")], how="semi")
    e.register("q83d", e.run(P(wk_dates, ("dsk", col("d_date_sk")))))
    outs = []
    for rtbl, rpre in (("store_returns", "sr"), ("catalog_returns", "cr"),
                       ("web_returns", "wr")):
        p = J(S(rtbl, f"{rpre}_returned_date_sk", f"{rpre}_item_sk",
                f"{rpre}_return_quantity"),
              S("q83d"), [(f"{rpre}_returned_date_sk", "dsk")], how="semi")
        p = J(p, S("item", "i_item_sk", "i_item_id"),
              [(f"{rpre}_item_sk", "i_item_sk")])
        outs.append(A(p, ["i_item_id"],
                      [(f"{rpre}_qty", "sum",
                        col(f"{rpre}_return_quantity"))]))
    p = J(outs[0], P(outs[1], ("iid2", col("i_item_id")),
                     ("cr_qty", col("cr_qty"))), [("i_item_id", "iid2")])
    p = J(p, P(outs[2], ("iid3", col("i_item_id")),
               ("wr_qty", col("wr_qty"))), [("i_item_id", "iid3")])
    tot = (col("sr_qty") + col("cr_qty") + col("wr_qty")).cast_float()
    p = P(p, ("item_id", col("i_item_id")), ("sr_item_qty", col("sr_qty")),
          ("sr_dev", col("sr_qty").cast_float() / tot / 3.0 * 100.0),
          ("cr_item_qty", col("cr_qty")),
          ("cr_dev", col("cr_qty").cast_float() / tot / 3.0 * 100.0),
          ("wr_item_qty", col("wr_qty")),
          ("wr_dev", col("wr_qty").cast_float() / tot / 3.0 * 100.0),
          ("average", tot / 3.0))
    return e.run(Limit(Sort(p, [("item_id", True), ("sr_item_qty", True)]),
                       100))


def q84(e: Engine) -> Frame:
    p = J(S("customer", "c_customer_sk", "c_customer_id", "c_first_name",
            "c_last_name", "c_current_addr_sk", "c_current_cdemo_sk",
            "c_current_hdemo_sk"),
          F(S("customer_address", "ca_address_sk", "ca_city"),
            col("ca_city") == "city_10"),
          [("c_current_addr_sk", "ca_address_sk")])
    p = J(p, S("household_demographics", "hd_demo_sk",
               "hd_income_band_sk"),
          [("c_current_hdemo_sk", "hd_demo_sk")])
    p = J(p, F(S("income_band", "ib_income_band_sk", "ib_lower_bound",
                 "ib_upper_bound"),
               (col("ib_lower_bound") >= 30000) &
               (col("ib_upper_bound") <= 80000)),
          [("hd_income_band_sk", "ib_income_band_sk")])
    p = J(p, P(S("store_returns", "sr_cdemo_sk"),
               ("srcd", col("sr_cdemo_sk"))),
          [("c_current_cdemo_sk", "srcd")], how="semi")
    p = P(p, ("customer_id", col("c_customer_id")),
          ("customername", col("c_last_name")),
          ("c_first_name", col("c_first_name")))
    return e.run(Limit(Sort(p, [("customer_id", True)]), 100))


def q85(e: Engine) -> Frame:
    p = J(S("web_returns", "wr_item_sk", "wr_order_number",
            "wr_refunded_cdemo_sk", "wr_returning_cdemo_sk",
            "wr_refunded_addr_sk", "wr_reason_sk", "wr_fee",
            "wr_refunded_cash", "wr_return_quantity"),
          S("web_sales", "ws_item_sk", "ws_order_number",
            "ws_web_page_sk", "ws_sold_date_sk", "ws_quantity",
            "ws_sales_price", "ws_net_profit"),
          [("wr_item_sk", "ws_item_sk"),
           ("wr_order_number", "ws_order_number")])
    p = J(p, S("web_page", "wp_web_page_sk"),
          [("ws_web_page_sk", "wp_web_page_sk")])
    p = J(p, dd(col("d_year") == 2000, "d_year"),
          [("ws_sold_date_sk", "d_date_sk")])
    p = J(p, P(S("customer_demographics", "cd_demo_sk",
                 "cd_marital_status", "cd_education_status"),
               ("cd1sk", col("cd_demo_sk")),
               ("ms1", col("cd_marital_status")),
               ("ed1", col("cd_education_status"))),
          [("wr_refunded_cdemo_sk", "cd1sk")])
    p = J(p, P(S("customer_demographics", "cd_demo_sk",
                 "cd_marital_status", "cd_education_status"),
               ("cd2sk", col("cd_demo_sk")),
               ("ms2", col("cd_marital_status")),
               ("ed2", col("cd_education_status"))),
          [("wr_returning_cdemo_sk", "cd2sk")])
    p = J(p, S("customer_address", "ca_address_sk", "ca_country",
               "ca_state"),
          [("wr_refunded_addr_sk", "ca_address_sk")])
    p = J(p, S("reason", "r_reason_sk", "r_reason_desc"),
          [("wr_reason_sk", "r_reason_sk")])
    m1 = ((col("ms1") == "M") & (col("ed1") == "4 yr Degree") &
          col("ws_sales_price").between(100.0, 150.0))
    m2 = ((col("ms1") == "S") & (col("ed1") == "College") &
          col("ws_sales_price").between(50.0, 100.0))
    m3 = ((col("ms1") == "W") & (col("ed1") == "2 yr Degree") &
          col("ws_sales_price").between(150.0, 200.0))
    s1 = (col("ca_country") == "United States") & \
        col("ca_state").isin(["IN", "OH", "NJ"]) & \
        col("ws_net_profit").between(100, 200)
    s2 = (col("ca_country") == "United States") & \
        col("ca_state").isin(["WI", "CT", "KY"]) & \
        col("ws_net_profit").between(150, 300)
    s3 = (col("ca_country") == "United States") & \
        col("ca_state").isin(["LA", "IA", "AR"]) & \
        col("ws_net_profit").between(50, 250)
    p = F(p, ((col("ms1") == col("ms2")) & (m1 | m2 | m3)) & (s1 | s2 | s3))
    p = A(p, ["r_reason_desc"],
          [("avg_qty", "avg", col("ws_quantity")),
           ("avg_fee", "avg", col("wr_fee")),
           ("avg_cash", "avg", col("wr_refunded_cash"))])
    return e.run(Limit(Sort(p, [("avg_qty", True), ("avg_fee", True),
                                ("r_reason_desc", True)]), 100))


def q86(e: Engine) -> Frame:
    p = J(S("web_sales", "ws_sold_date_sk", "ws_item_sk", "ws_net_paid"),
          dd(col("d_month_seq").between(1200, 1211), "d_month_seq"),
          [("ws_sold_date_sk", "d_date_sk")])
    p = J(p, S("item", "i_item_sk", "i_category", "i_class"),
          [("ws_item_sk", "i_item_sk")])
    p = A(p, ["i_category", "i_class"],
          [("total_sum", "sum", col("ws_net_paid"))], rollup=True)
    p = P(p, ("i_category", col("i_category")), ("i_class", col("i_class")),
          ("lochierarchy", lit(2) - col("__lvl")),
          ("total_sum", col("total_sum")),
          ("rank_cat", case_when((col("__lvl") == 2, col("i_category")),
                                 otherwise=lit(None))))
    p = Window(p, ["lochierarchy", "rank_cat"],
               [("rank_within_parent", "rank", None,
                 [("total_sum", False)])])
    return e.run(Limit(Sort(p, [("lochierarchy", False),
                                ("i_category", True),
                                ("rank_within_parent", True)]), 100))


def q87(e: Engine) -> Frame:
    win = dd(col("d_month_seq").between(1200, 1211), "d_month_seq",
             "d_date")
    frames = []
    for tbl, pre, cc in (("store_sales", "ss", "ss_customer_sk"),
                         ("catalog_sales", "cs", "cs_bill_customer_sk"),
                         ("web_sales", "ws", "ws_bill_customer_sk")):
        p = J(S(tbl, f"{pre}_sold_date_sk", cc), win,
              [(f"{pre}_sold_date_sk", "d_date_sk")])
        p = J(p, S("customer", "c_customer_sk", "c_last_name",
                   "c_first_name"),
              [(cc, "c_customer_sk")])
        frames.append(e.run(Distinct(P(p, ("ln", col("c_last_name")),
                                       ("fn", col("c_first_name")),
                                       ("dt", col("d_date"))))))
    e.register("q87a", frames[0])
    e.register("q87b", frames[1])
    e.register("q87c", frames[2])
    p = J(S("q87a"), P(S("q87b"), ("ln2", col("ln")), ("fn2", col("fn")),
                       ("dt2", col("dt"))),
          [("ln", "ln2"), ("fn", "fn2"), ("dt", "dt2")], how="anti")
    p = J(p, P(S("q87c"), ("ln3", col("ln")), ("fn3", col("fn")),
               ("dt3", col("dt"))),
          [("ln", "ln3"), ("fn", "fn3"), ("dt", "dt3")], how="anti")
    return e.run(A(p, [], [("cnt", "count", None)]))


def q88(e: Engine) -> Frame:
    slots = [(8, 30), (9, 0), (9, 30), (10, 0), (10, 30), (11, 0),
             (11, 30), (12, 0)]
    out = {}
    for i, (h, m) in enumerate(slots, 1):
        p = J(S("store_sales", "ss_sold_time_sk", "ss_hdemo_sk",
                "ss_store_sk"),
              F(S("time_dim", "t_time_sk", "t_hour", "t_minute"),
                (col("t_hour") == h) &
                (col("t_minute").between(m, m + 29))),
              [("ss_sold_time_sk", "t_time_sk")])
        p = J(p, F(S("household_demographics", "hd_demo_sk",
                     "hd_dep_count", "hd_vehicle_count"),
                   ((col("hd_dep_count") == 4) &
                    (col("hd_vehicle_count") <= 6)) |
                   ((col("hd_dep_count") == 2) &
                    (col("hd_vehicle_count") <= 4)) |
                   ((col("hd_dep_count") == 0) &
                    (col("hd_vehicle_count") <= 2))),
              [("ss_hdemo_sk", "hd_demo_sk")])
        p = J(p, F(S("store", "s_store_sk", "s_store_name"),
                   col("s_store_name") == "store_a"),
              [("ss_store_sk", "s_store_sk")])
        out[f"h{h}_{m}"] = int(e.scalar(A(p, [], [("c", "count", None)]),
                                        "c") or 0)
    return e.const_frame(**out)


def q89(e: Engine) -> Frame:
    in1 = (col("i_category").isin(["Books", "Children", "Electronics"]) &
           col("i_class").isin(["class06", "class07", "class08"]))
    in2 = (col("i_category").isin(["Women", "Music", "Men"]) &
           col("i_class").isin(["class09", "class10", "class11"]))
    p = J(S("store_sales", "ss_sold_date_sk", "ss_item_sk", "ss_store_sk",
            "ss_sales_price"),
          dd(col("d_year") == 1999, "d_year", "d_moy"),
          [("ss_sold_date_sk", "d_date_sk")])
    p = J(p, F(S("item", "i_item_sk", "i_category", "i_class", "i_brand"),
               in1 | in2),
          [("ss_item_sk", "i_item_sk")])
    p = J(p, S("store", "s_store_sk", "s_store_name", "s_company_name"),
          [("ss_store_sk", "s_store_sk")])
    p = A(p, ["i_category", "i_class", "i_brand", "s_store_name",
              "s_company_name", "d_moy"],
          [("sum_sales", "sum", col("ss_sales_price"))])
    p = Window(p, ["i_category", "i_brand", "s_store_name",
                   "s_company_name"],
               [("avg_monthly_sales", "avg", "sum_sales")])
    p = F(p, case_when((~(col("avg_monthly_sales") == 0.0),
                        abs_(col("sum_sales") -
                             col("avg_monthly_sales")) /
                        col("avg_monthly_sales")),
                       otherwise=lit(None)) > 0.1)
    return e.run(Limit(Sort(p, [("sum_sales", True),
                                ("s_store_name", True)]), 100))


def q90(e: Engine) -> Frame:
    def half(h1, h2):
        p = J(S("web_sales", "ws_sold_time_sk", "ws_ship_hdemo_sk",
                "ws_web_page_sk"),
              F(S("time_dim", "t_time_sk", "t_hour"),
                col("t_hour").between(h1, h2)),
              [("ws_sold_time_sk", "t_time_sk")])
            
        p = J(p, F(S("household_demographics", "hd_demo_sk",
                     "hd_dep_count"), col("hd_dep_count") == 6),
              [("ws_ship_hdemo_sk", "hd_demo_sk")])
        p = J(p, F(S("web_page", "wp_web_page_sk", "wp_char_count"),
                   col("wp_char_count").between(5000, 5200)),
              [("ws_web_page_sk", "wp_web_page_sk")])
        return int(e.scalar(A(p, [], [("c", "count", None)]), "c") or 0)
    am = half(8, 9)
    pm = half(19, 20)
    return e.const_frame(amc=am, pmc=pm,
                         am_pm_ratio=(am / pm) if pm else None)


def q91(e: Engine) -> Frame:
    p = J(S("catalog_returns", "cr_returned_date_sk", "cr_call_center_sk",
            "cr_returning_customer_sk", "cr_net_loss"),
          dd((col("d_year") == 1998) & (col("d_moy") == 11), "d_year",
             "d_moy"),
          [("cr_returned_date_sk", "d_date_sk")])
    p = J(p, S("call_center", "cc_call_center_sk", "cc_call_center_id",
               "cc_name", "cc_manager"),
          [("cr_call_center_sk", "cc_call_center_sk")])
    p = J(p, S("customer", "c_customer_sk", "c_current_cdemo_sk",
               "c_current_hdemo_sk", "c_current_addr_sk"),
          [("cr_returning_customer_sk", "c_customer_sk")])
    p = J(p, F(S("customer_demographics", "cd_demo_sk", "cd_marital_status",
                 "cd_education_status"),
               ((col("cd_marital_status") == "M") &
                (col("cd_education_status") == "Unknown")) |
               ((col("cd_marital_status") == "W") &
                (col("cd_education_status") == "Advanced Degree"))),
          [("c_current_cdemo_sk", "cd_demo_sk")])
    p = J(p, F(S("household_demographics", "hd_demo_sk",
                 "hd_buy_potential"),
               col("hd_buy_potential").isin(["0-500"])),
          [("c_current_hdemo_sk", "hd_demo_sk")])
    p = J(p, F(S("customer_address", "ca_address_sk", "ca_gmt_offset"),
               col("ca_gmt_offset") == -7.0),
          [("c_current_addr_sk", "ca_address_sk")])
    p = A(p, ["cc_call_center_id", "cc_name", "cc_manager",
              "cd_marital_status", "cd_education_status"],
          [("returns_loss", "sum", col("cr_net_loss"))])
    return e.run(Sort(p, [("returns_loss", False)]))


def q92(e: Engine) -> Frame:
    lo = dsk("2000-01-27")
    base = J(F(S("web_sales", "ws_sold_date_sk", "ws_item_sk",
                 "ws_ext_discount_amt"),
               col("ws_sold_date_sk").between(lo, lo + 90)),
             F(S("item", "i_item_sk", "i_manufact_id"),
               col("i_manufact_id") == 9),
             [("ws_item_sk", "i_item_sk")])
    e.register("q92base", e.run(base))
    avg_d = A(S("q92base"), ["ws_item_sk"],
              [("avg_disc", "avg", col("ws_ext_discount_amt"))])
    p = J(S("q92base"), P(avg_d, ("isk2", col("ws_item_sk")),
                          ("avg_disc", col("avg_disc"))),
          [("ws_item_sk", "isk2")])
    p = F(p, col("ws_ext_discount_amt") > col("avg_disc") * 1.3)
    return e.run(A(p, [], [("excess_discount", "sum",
                            col("ws_ext_discount_amt"))]))


def q93(e: Engine) -> Frame:
    rr = J(S("store_returns", "sr_item_sk", "sr_ticket_number",
             "sr_reason_sk", "sr_return_quantity"),
           F(S("reason", "r_reason_sk", "r_reason_desc"),
             col("r_reason_desc") == "reason 28"),
           [("sr_reason_sk", "r_reason_sk")])
    p = J(S("store_sales", "ss_item_sk", "ss_ticket_number",
            "ss_customer_sk", "ss_quantity", "ss_sales_price"),
          P(rr, ("ri", col("sr_item_sk")), ("rt", col("sr_ticket_number")),
            ("ret_qty", col("sr_return_quantity"))),
          [("ss_item_sk", "ri"), ("ss_ticket_number", "rt")], how="left")
    act = case_when(
        (is_not_null(col("ret_qty")),
         (col("ss_quantity") - col("ret_qty")).cast_float() *
         col("ss_sales_price")),
        otherwise=col("ss_quantity").cast_float() * col("ss_sales_price"))
    p = A(p, ["ss_customer_sk"], [("sumsales", "sum", act)])
    return e.run(Limit(Sort(p, [("sumsales", True),
                                ("ss_customer_sk", True)]), 100))


def q94(e: Engine) -> Frame:
    lo, hi = dsk("1999-02-01"), dsk("1999-04-02")
    ws1 = J(F(S("web_sales", "ws_ship_date_sk", "ws_ship_addr_sk",
               "ws_web_site_sk", "ws_order_number", "ws_warehouse_sk",
               "ws_ext_ship_cost", "ws_net_profit"),
              col("ws_ship_date_sk").between(lo, hi)),
            F(S("customer_address", "ca_address_sk", "ca_state"),
              col("ca_state") == "IL"),
            [("ws_ship_addr_sk", "ca_address_sk")])
    ws1 = J(ws1, F(S("web_site", "web_site_sk", "web_company_name"),
                   col("web_company_name") == "webco_0"),
            [("ws_web_site_sk", "web_site_sk")])
    multi_wh = P(F(A(S("web_sales", "ws_order_number", "ws_warehouse_sk"),
                     ["ws_order_number"],
                     [("nwh", "countd", col("ws_warehouse_sk"))]),
                   col("nwh") > 1),
                 ("ono", col("ws_order_number")))
    ws1 = J(ws1, multi_wh, [("ws_order_number", "ono")], how="semi")
    ws1 = J(ws1, P(S("web_returns", "wr_order_number"),
                   ("rno", col("wr_order_number"))),
            [("ws_order_number", "rno")], how="anti")
    return e.run(A(ws1, [],
                   [("order_count", "countd", col("ws_order_number")),
                    ("total_shipping_cost", "sum", col("ws_ext_ship_cost")),
                    ("total_net_profit", "sum", col("ws_net_profit"))]))


def q95(e: Engine) -> Frame:
    lo, hi = dsk("1999-02-01"), dsk("1999-04-02")
    ws1 = J(F(S("web_sales", "ws_ship_date_sk", "ws_ship_addr_sk",
               "ws_web_site_sk", "ws_order_number", "ws_ext_ship_cost",
               "ws_net_profit"),
              col("ws_ship_date_sk").between(lo, hi)),
            F(S("customer_address", "ca_address_sk", "ca_state"),
              col("ca_state") == "IL"),
            [("ws_ship_addr_sk", "ca_address_sk")])
    ws1 = J(ws1, F(S("web_site", "web_site_sk", "web_company_name"),
                   col("web_company_name") == "webco_0"),
            [("ws_web_site_sk", "web_site_sk")])
    ws_wh = P(F(A(S("web_sales", "ws_order_number", "ws_warehouse_sk"),
                  ["ws_order_number"],
                  [("nwh", "countd", col("ws_warehouse_sk"))]),
                col("nwh") > 1),
              ("ono", col("ws_order_number")))
    e.register("q95wh", e.run(ws_wh))
    ws1 = J(ws1, S("q95wh"), [("ws_order_number", "ono")], how="semi")
    ret = J(S("web_returns", "wr_order_number"),
            S("q95wh"), [("wr_order_number", "ono")], how="semi")
    ws1 = J(ws1, P(ret, ("rno", col("wr_order_number"))),
            [("ws_order_number", "rno")], how="semi")
    return e.run(A(ws1, [],
                   [("order_count", "countd", col("ws_order_number")),
                    ("total_shipping_cost", "sum", col("ws_ext_ship_cost")),
                    ("total_net_profit", "sum", col("ws_net_profit"))]))


def q96(e: Engine) -> Frame:
    p = J(S("store_sales", "ss_sold_time_sk", "ss_hdemo_sk", "ss_store_sk"),
          F(S("time_dim", "t_time_sk", "t_hour", "t_minute"),
            (col("t_hour") == 20) & (col("t_minute") >= 30)),
          [("ss_sold_time_sk", "t_time_sk")])
    p = J(p, F(S("household_demographics", "hd_demo_sk", "hd_dep_count"),
               col("hd_dep_count") == 7),
          [("ss_hdemo_sk", "hd_demo_sk")])
    p = J(p, F(S("store", "s_store_sk", "s_store_name"),
               col("s_store_name") == "store_a"),
          [("ss_store_sk", "s_store_sk")])
    return e.run(A(p, [], [("cnt", "count", None)]))


def q97(e: Engine) -> Frame:
    win = dd(col("d_month_seq").between(1200, 1211), "d_month_seq")
    ssci = A(J(S("store_sales", "ss_sold_date_sk", "ss_customer_sk",
                 "ss_item_sk"), win, [("ss_sold_date_sk", "d_date_sk")]),
             ["ss_customer_sk", "ss_item_sk"], [("c1", "count", None)])
    csci = A(J(S("catalog_sales", "cs_sold_date_sk", "cs_bill_customer_sk",
                 "cs_item_sk"), win, [("cs_sold_date_sk", "d_date_sk")]),
             ["cs_bill_customer_sk", "cs_item_sk"],
             [("c2", "count", None)])
    p = J(P(ssci, ("cust", col("ss_customer_sk")),
            ("item", col("ss_item_sk")), ("s_mark", lit(1))),
          P(csci, ("cust2", col("cs_bill_customer_sk")),
            ("item2", col("cs_item_sk")), ("c_mark", lit(1))),
          [("cust", "cust2"), ("item", "item2")], how="full")
    p = A(p, [], [
        ("store_only", "sum",
         case_when((is_not_null(col("s_mark")) & is_null(col("c_mark")),
                    lit(1)), otherwise=lit(0))),
        ("catalog_only", "sum",
         case_when((is_null(col("s_mark")) & is_not_null(col("c_mark")),
                    lit(1)), otherwise=lit(0))),
        ("store_and_catalog", "sum",
         case_when((is_not_null(col("s_mark")) &
                    is_not_null(col("c_mark")), lit(1)),
                   otherwise=lit(0)))])
    return e.run(p)


def q98(e: Engine) -> Frame:
    return e.run(_ratio_by_class(e, "store_sales", "ss", "1999-02-22",
                                 ["Sports", "Books", "Home"]))


def q99(e: Engine) -> Frame:
    p = J(S("catalog_sales", "cs_ship_date_sk", "cs_sold_date_sk",
            "cs_warehouse_sk", "cs_ship_mode_sk", "cs_call_center_sk"),
          dd(col("d_month_seq").between(1200, 1211), "d_month_seq"),
          [("cs_ship_date_sk", "d_date_sk")])
    p = J(p, S("warehouse", "w_warehouse_sk", "w_warehouse_name"),
          [("cs_warehouse_sk", "w_warehouse_sk")])
    p = J(p, S("ship_mode", "sm_ship_mode_sk", "sm_type"),
          [("cs_ship_mode_sk", "sm_ship_mode_sk")])
    p = J(p, S("call_center", "cc_call_center_sk", "cc_name"),
          [("cs_call_center_sk", "cc_call_center_sk")])
    lag = col("cs_ship_date_sk") - col("cs_sold_date_sk")
    p = A(p, ["w_warehouse_name", "sm_type", "cc_name"],
          [("d30", "sum", case_when((lag <= 30, lit(1)), otherwise=lit(0))),
           ("d31_60", "sum", case_when(((lag > 30) & (lag <= 60), lit(1)),
                                       otherwise=lit(0))),
           ("d61_90", "sum", case_when(((lag > 60) & (lag <= 90), lit(1)),
                                       otherwise=lit(0))),
           ("d91_120", "sum", case_when(((lag > 90) & (lag <= 120),
                                         lit(1)), otherwise=lit(0))),
           ("dgt120", "sum", case_when((lag > 120, lit(1)),
                                       otherwise=lit(0)))])
    return e.run(Limit(Sort(p, [("w_warehouse_name", True),
                                ("sm_type", True), ("cc_name", True)]),
                       100))


QUERIES = {n: globals()[f"q{n}"] for n in range(1, 100)}
