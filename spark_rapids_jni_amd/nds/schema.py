"""Synthetic TPC-DS-shaped tables for the NDS harness.

No network: data is generated in the TPC-DS shapes (same tables, columns,
key relationships and rough cardinalities/domain scaling as the spec's
dsdgen), with numpy RNG. Returns sample (ticket/order, item, customer)
triples from their sales table so sales<->returns joins behave like real
NDS data. String attributes are dictionary-encoded (codes in the column,
dictionary in metadata) — see expr.py for why this is the MI355X-first
layout.

Scaling: fact tables grow linearly with SF; dimensions follow approximate
TPC-DS domain scaling (sub-linear exponents fitted to the published SF1 /
SF1000 / SF3000 row counts). `sharded` tables are generated per-rank
(seeded by (table, rank)) for the one-process-per-GPU distributed mode.
"""
import datetime
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np

_EPOCH = datetime.date(1970, 1, 1)


@dataclass
class Dataset:
    name: str
    nrows: int
    columns: Dict[str, np.ndarray] = field(default_factory=dict)
    valid: Dict[str, Optional[np.ndarray]] = field(default_factory=dict)
    dicts: Dict[str, Optional[List[str]]] = field(default_factory=dict)
    sharded: bool = False

    def add(self, name, arr, valid=None, dic=None):
        if arr.dtype == np.int64 and arr.size:
            # every NDS integer domain (surrogate keys, ticket numbers up
            # to SF3000, codes, quantities) fits 32 bits: downcasting
            # halves HBM footprint and scan bandwidth
            mn, mx = arr.min(), arr.max()
            if -2**31 <= mn and mx < 2**31:
                arr = arr.astype(np.int32)
        self.columns[name] = arr
        self.valid[name] = valid
        self.dicts[name] = dic


# (rows at SF1, scaling exponent, sharded)
TABLES = {
    "store_sales":       (2_880_000, 1.0, True),
    "store_returns":     (288_000, 1.0, True),
    "catalog_sales":     (1_441_000, 1.0, True),
    "catalog_returns":   (144_000, 1.0, True),
    "web_sales":         (719_000, 1.0, True),
    "web_returns":       (72_000, 1.0, True),
    "inventory":         (11_745_000, 0.6, True),
    "date_dim":          (0, 0.0, False),   # fixed span, see _gen_date_dim
    "time_dim":          (86_400, 0.0, False),
    "item":              (18_000, 0.41, False),
    "customer":          (100_000, 0.69, False),
    "customer_address":  (50_000, 0.69, False),
    "customer_demographics": (1_920_800, 0.0, False),
    "household_demographics": (7_200, 0.0, False),
    "income_band":       (20, 0.0, False),
    "store":             (12, 0.64, False),
    "warehouse":         (5, 0.2, False),
    "web_site":          (30, 0.085, False),
    "web_page":          (60, 0.57, False),
    "call_center":       (6, 0.28, False),
    "catalog_page":      (11_718, 0.136, False),
    "promotion":         (300, 0.233, False),
    "reason":            (35, 0.1, False),
    "ship_mode":         (20, 0.0, False),
}

# date_dim span: 1990-01-01 .. 2009-12-31; facts sell in 1998..2002
DATE_BASE_SK = 2450000
DATE_START = datetime.date(1990, 1, 1)
DATE_END = datetime.date(2009, 12, 31)
N_DATES = (DATE_END - DATE_START).days + 1
SELL_FIRST = (datetime.date(1998, 1, 1) - DATE_START).days
SELL_LAST = (datetime.date(2002, 12, 31) - DATE_START).days

CATEGORIES = ["Books", "Children", "Electronics", "Home", "Jewelry",
              "Men", "Music", "Shoes", "Sports", "Women"]
STATES = ["AL", "AK", "AZ", "AR", "CA", "CO", "CT", "DE", "FL", "GA", "HI",
          "ID", "IL", "IN", "IA", "KS", "KY", "LA", "ME", "MD", "MA", "MI",
          "MN", "MS", "MO", "MT", "NE", "NV", "NH", "NJ", "NM", "NY", "NC",
          "ND", "OH", "OK", "OR", "PA", "RI", "SC", "SD", "TN", "TX", "UT",
          "VT", "VA", "WA", "WV", "WI", "WY"]
CITIES = [f"city_{i:02d}" for i in range(60)]
COUNTIES = [f"county_{i:02d}" for i in range(30)]
COLORS = ["almond", "aquamarine", "azure", "beige", "black", "blue",
          "blush", "brown", "burlywood", "chartreuse", "chiffon", "coral",
          "cornflower", "cream", "cyan", "dark", "deep", "dim", "dodger",
          "firebrick"]
SIZES = ["small", "medium", "large", "extra large", "economy", "N/A",
         "petite"]
UNITS = ["Unknown", "Each", "Dozen", "Case", "Pallet", "Gross", "Box",
         "Bunch", "Bundle", "Carton"]
EDU = ["Primary", "Secondary", "College", "2 yr Degree", "4 yr Degree",
       "Advanced Degree", "Unknown"]
MARITAL = ["M", "S", "D", "W", "U"]
BUY_POTENTIAL = [">10000", "5001-10000", "1001-5000", "501-1000", "0-500",
                 "Unknown"]
CREDIT = ["Low Risk", "Good", "High Risk", "Unknown"]
SHIP_TYPES = ["EXPRESS", "NEXT DAY", "OVERNIGHT", "REGULAR", "TWO DAY",
              "LIBRARY"]
DAY_NAMES = ["Sunday", "Monday", "Tuesday", "Wednesday", "Thursday",
             "Friday", "Saturday"]
MEALS = ["breakfast", "lunch", "dinner"]


def table_rows(name: str, sf: float) -> int:
    if name == "date_dim":
        return N_DATES
    base, exp, _sh = TABLES[name]
    if name == "customer_demographics" and sf < 1.0:
        # spec-fixed at 1.92M, but scaled below SF1 so tiny validation
        # catalogs stay tiny (documented delta)
        return max(int(base * sf ** 0.5), 64)
    if exp == 0.0:
        return base
    return max(int(base * sf ** exp), 8)


# Fact tables are generated in NBLOCKS deterministic blocks (block b is
# seeded by (table, b)); rank r of a world-w run owns blocks b % w == r.
# The union of all ranks' shards is therefore byte-identical to the
# world=1 table for ANY world size — distributed results must equal
# single-GPU results, and the driver's scaling curve runs the same data.
NBLOCKS = 8


def _block_rows(n: int, b: int) -> int:
    return n // NBLOCKS + (1 if b < n % NBLOCKS else 0)


class _Gen:
    def __init__(self, sf, world, rank):
        self.sf = sf
        self.world = world
        self.rank = rank
        self.sizes = {t: table_rows(t, sf) for t in TABLES}

    def rng(self, table):
        # crc32, not hash(): ranks must agree on dimension content and
        # Python string hashing is randomized per process
        import zlib
        seed = (zlib.crc32(table.encode()) & 0xFFFF) * 10007 + 7
        return np.random.default_rng(seed)

    def nrows(self, table):
        return self.sizes[table]

    # -- column helpers ----------------------------------------------------
    @staticmethod
    def money(r, n, lo, hi):
        return np.round(r.uniform(lo, hi, n), 2)

    @staticmethod
    def fk(r, n, domain, null_frac=0.0):
        arr = r.integers(1, domain + 1, n, dtype=np.int64)
        if null_frac <= 0:
            return arr, None
        valid = r.random(n) >= null_frac
        return arr, valid

    def date_fk(self, r, n, null_frac=0.0):
        day = r.integers(SELL_FIRST, SELL_LAST + 1, n, dtype=np.int64)
        arr = day + DATE_BASE_SK
        if null_frac <= 0:
            return arr, None
        valid = r.random(n) >= null_frac
        return arr, valid


class _BlockGen(_Gen):
    """View of the generator for one fact block: sharded tables size to the
    block and seed by (table, block)."""

    def __init__(self, g: _Gen, block: int):
        self.sf = g.sf
        self.world = g.world
        self.rank = g.rank
        self.sizes = g.sizes
        self.block = block

    def rng(self, table):
        import zlib
        seed = ((zlib.crc32(table.encode()) & 0xFFFF) * 10007 + 7 +
                self.block * 131071)
        return np.random.default_rng(seed)

    def nrows(self, table):
        n = self.sizes[table]
        if TABLES[table][2]:
            return _block_rows(n, self.block)
        return n


def _concat_datasets(name: str, parts: List[Dataset]) -> Dataset:
    total = sum(p.nrows for p in parts)
    out = Dataset(name, total, sharded=True)
    for cname in parts[0].columns:
        arr = np.concatenate([p.columns[cname] for p in parts])
        if any(p.valid[cname] is not None for p in parts):
            valid = np.concatenate([
                p.valid[cname] if p.valid[cname] is not None
                else np.ones(p.nrows, dtype=bool) for p in parts])
        else:
            valid = None
        out.add(cname, arr, valid, parts[0].dicts[cname])
    return out

    # -- column helpers ----------------------------------------------------
    @staticmethod
    def money(r, n, lo, hi):
        return np.round(r.uniform(lo, hi, n), 2)

    @staticmethod
    def fk(r, n, domain, null_frac=0.0):
        arr = r.integers(1, domain + 1, n, dtype=np.int64)
        if null_frac <= 0:
            return arr, None
        valid = r.random(n) >= null_frac
        return arr, valid

    def date_fk(self, r, n, null_frac=0.0):
        day = r.integers(SELL_FIRST, SELL_LAST + 1, n, dtype=np.int64)
        arr = day + DATE_BASE_SK
        if null_frac <= 0:
            return arr, None
        valid = r.random(n) >= null_frac
        return arr, valid


def _gen_date_dim(g: _Gen) -> Dataset:
    n = N_DATES
    d = Dataset("date_dim", n)
    i = np.arange(n, dtype=np.int64)
    days = np.array([(DATE_START + datetime.timedelta(days=int(x)) - _EPOCH)
                     .days for x in i], dtype=np.int64)
    dates = [DATE_START + datetime.timedelta(days=int(x)) for x in i]
    d.add("d_date_sk", i + DATE_BASE_SK)
    d.add("d_date", days)
    d.add("d_year", np.array([x.year for x in dates], dtype=np.int64))
    d.add("d_moy", np.array([x.month for x in dates], dtype=np.int64))
    d.add("d_dom", np.array([x.day for x in dates], dtype=np.int64))
    d.add("d_qoy", np.array([(x.month - 1) // 3 + 1 for x in dates],
                            dtype=np.int64))
    dow = np.array([(x.weekday() + 1) % 7 for x in dates], dtype=np.int64)
    d.add("d_dow", dow)
    d.add("d_day_name", dow, dic=DAY_NAMES)
    # sequential week/month numbers (same semantics as TPC-DS *_seq columns)
    first_dow = (DATE_START.weekday() + 1) % 7
    d.add("d_week_seq", (i + first_dow) // 7 + 5270)
    d.add("d_month_seq", np.array(
        [(x.year - 1900) * 12 + x.month - 1 for x in dates], dtype=np.int64))
    return d


def _gen_time_dim(g: _Gen) -> Dataset:
    n = 86_400
    d = Dataset("time_dim", n)
    i = np.arange(n, dtype=np.int64)
    d.add("t_time_sk", i)
    d.add("t_time", i)
    d.add("t_hour", i // 3600)
    d.add("t_minute", (i % 3600) // 60)
    meal = np.full(n, -1, dtype=np.int64)
    h = i // 3600
    meal[(h >= 6) & (h < 9)] = 0
    meal[(h >= 11) & (h < 14)] = 1
    meal[(h >= 17) & (h < 20)] = 2
    d.add("t_meal_time", np.clip(meal, 0, None), valid=meal >= 0, dic=MEALS)
    return d


def _dict_col(r, n, values, null_frac=0.0):
    codes = r.integers(0, len(values), n, dtype=np.int64)
    valid = None
    if null_frac > 0:
        valid = r.random(n) >= null_frac
    return codes, valid, list(values)


def _gen_item(g: _Gen) -> Dataset:
    r = g.rng("item")
    n = g.sizes["item"]
    d = Dataset("item", n)
    i = np.arange(1, n + 1, dtype=np.int64)
    d.add("i_item_sk", i)
    # item_id repeats roughly every 2 items (TPC-DS reuses ids across
    # item versions)
    nid = max(n // 2, 1)
    ids = [f"AAAAAAAA{k:08d}" for k in range(nid)]
    d.add("i_item_id", np.arange(n, dtype=np.int64) % nid, dic=ids)
    ndesc = max(n // 4, 1)
    d.add("i_item_desc", r.integers(0, ndesc, n),
          dic=[f"item description {k}" for k in range(ndesc)])
    d.add("i_current_price", g.money(r, n, 0.5, 100.0),
          valid=r.random(n) >= 0.01)
    d.add("i_wholesale_cost", g.money(r, n, 0.3, 80.0))
    nbrand = min(max(n // 20, 10), 1000)
    brand_id = r.integers(1, nbrand + 1, n)
    d.add("i_brand_id", brand_id * 1000 + r.integers(1, 10, n))
    d.add("i_brand", brand_id - 1, dic=[f"brand#{k+1}" for k in range(nbrand)])
    nclass = 16
    cls = r.integers(0, nclass, n)
    cat = r.integers(0, len(CATEGORIES), n)
    d.add("i_class_id", cls + 1)
    d.add("i_class", cls, dic=[f"class{k:02d}" for k in range(nclass)])
    d.add("i_category_id", cat + 1)
    d.add("i_category", cat, valid=r.random(n) >= 0.005, dic=CATEGORIES)
    nmanu = min(max(n // 20, 10), 1000)
    manu = r.integers(1, nmanu + 1, n)
    d.add("i_manufact_id", manu)
    d.add("i_manufact", manu - 1, dic=[f"manu#{k+1}" for k in range(nmanu)])
    d.add("i_size", *_dict_col(r, n, SIZES))
    d.add("i_color", *_dict_col(r, n, COLORS))
    d.add("i_units", *_dict_col(r, n, UNITS))
    d.add("i_container", r.integers(0, 2, n), dic=["Unknown", "Plastic"])
    d.add("i_manager_id", r.integers(1, 101, n))
    nprod = n
    d.add("i_product_name", np.arange(n, dtype=np.int64),
          dic=[f"product{k:07d}" for k in range(nprod)])
    return d


def _gen_customer(g: _Gen) -> Dataset:
    r = g.rng("customer")
    n = g.sizes["customer"]
    d = Dataset("customer", n)
    d.add("c_customer_sk", np.arange(1, n + 1, dtype=np.int64))
    d.add("c_customer_id", np.arange(n, dtype=np.int64),
          dic=[f"AAAAAAAA{k:08d}" for k in range(n)])
    cd, cdv = g.fk(r, n, g.sizes["customer_demographics"], 0.03)
    d.add("c_current_cdemo_sk", cd, cdv)
    hd, hdv = g.fk(r, n, g.sizes["household_demographics"], 0.03)
    d.add("c_current_hdemo_sk", hd, hdv)
    ca, cav = g.fk(r, n, g.sizes["customer_address"], 0.02)
    d.add("c_current_addr_sk", ca, cav)
    fs, fsv = g.date_fk(r, n, 0.03)
    d.add("c_first_sales_date_sk", fs, fsv)
    d.add("c_first_shipto_date_sk", fs + 30, fsv)
    nname = min(max(n // 10, 100), 6000)
    d.add("c_first_name", r.integers(0, nname, n),
          dic=[f"First{k}" for k in range(nname)])
    d.add("c_last_name", r.integers(0, nname, n),
          dic=[f"Last{k}" for k in range(nname)])
    d.add("c_salutation", *_dict_col(r, n, ["Mr.", "Mrs.", "Ms.", "Dr.",
                                            "Miss", "Sir"], 0.02))
    d.add("c_preferred_cust_flag", *_dict_col(r, n, ["N", "Y"], 0.03))
    d.add("c_birth_day", r.integers(1, 29, n))
    d.add("c_birth_month", r.integers(1, 13, n))
    d.add("c_birth_year", r.integers(1924, 1993, n),
          valid=r.random(n) >= 0.02)
    d.add("c_birth_country", *_dict_col(
        r, n, [f"COUNTRY_{k:02d}" for k in range(40)], 0.02))
    d.add("c_email_address", np.arange(n, dtype=np.int64),
          dic=[f"c{k}@example.com" for k in range(n)])
    return d


def _gen_customer_address(g: _Gen) -> Dataset:
    r = g.rng("customer_address")
    n = g.sizes["customer_address"]
    d = Dataset("customer_address", n)
    d.add("ca_address_sk", np.arange(1, n + 1, dtype=np.int64))
    d.add("ca_city", *_dict_col(r, n, CITIES, 0.01))
    d.add("ca_county", *_dict_col(r, n, COUNTIES, 0.01))
    d.add("ca_state", *_dict_col(r, n, STATES, 0.01))
    nzip = 1000
    d.add("ca_zip", r.integers(0, nzip, n),
          valid=r.random(n) >= 0.01,
          dic=[f"{10000 + 89 * k % 90000:05d}" for k in range(nzip)])
    d.add("ca_country", np.zeros(n, dtype=np.int64),
          valid=r.random(n) >= 0.005, dic=["United States"])
    # cyclic so every offset value is guaranteed present (queries filter
    # on gmt_offset == -5)
    d.add("ca_gmt_offset", (-5.0 - (np.arange(n) % 4)).astype(np.float64),
          valid=r.random(n) >= 0.01)
    d.add("ca_location_type", *_dict_col(
        r, n, ["apartment", "condo", "single family"], 0.01))
    return d


def _gen_customer_demographics(g: _Gen) -> Dataset:
    r = g.rng("customer_demographics")
    n = g.sizes["customer_demographics"]
    d = Dataset("customer_demographics", n)
    d.add("cd_demo_sk", np.arange(1, n + 1, dtype=np.int64))
    d.add("cd_gender", *_dict_col(r, n, ["M", "F"]))
    d.add("cd_marital_status", *_dict_col(r, n, MARITAL))
    d.add("cd_education_status", *_dict_col(r, n, EDU))
    d.add("cd_purchase_estimate", r.integers(1, 21, n) * 500)
    d.add("cd_credit_rating", *_dict_col(r, n, CREDIT))
    d.add("cd_dep_count", r.integers(0, 10, n))
    d.add("cd_dep_employed_count", r.integers(0, 7, n))
    d.add("cd_dep_college_count", r.integers(0, 7, n))
    return d


def _gen_household_demographics(g: _Gen) -> Dataset:
    r = g.rng("household_demographics")
    n = g.sizes["household_demographics"]
    d = Dataset("household_demographics", n)
    d.add("hd_demo_sk", np.arange(1, n + 1, dtype=np.int64))
    d.add("hd_income_band_sk", r.integers(1, 21, n))
    d.add("hd_buy_potential", *_dict_col(r, n, BUY_POTENTIAL))
    d.add("hd_dep_count", r.integers(0, 10, n))
    d.add("hd_vehicle_count", r.integers(-1, 5, n))
    return d


def _gen_income_band(g: _Gen) -> Dataset:
    n = 20
    d = Dataset("income_band", n)
    i = np.arange(n, dtype=np.int64)
    d.add("ib_income_band_sk", i + 1)
    d.add("ib_lower_bound", i * 10000)
    d.add("ib_upper_bound", (i + 1) * 10000)
    return d


def _gen_store(g: _Gen) -> Dataset:
    r = g.rng("store")
    n = g.sizes["store"]
    d = Dataset("store", n)
    d.add("s_store_sk", np.arange(1, n + 1, dtype=np.int64))
    d.add("s_store_id", np.arange(n, dtype=np.int64) // 2,
          dic=[f"AAAAAAAA{k:08d}" for k in range(max(n // 2, 1) + 1)])
    nname = min(n, 10)
    d.add("s_store_name", np.arange(n, dtype=np.int64) % nname,
          dic=[f"store_{chr(97+k)}" for k in range(nname)])
    d.add("s_number_employees", r.integers(200, 301, n))
    d.add("s_city", *_dict_col(r, n, CITIES[:20]))
    d.add("s_county", *_dict_col(r, n, COUNTIES[:15]))
    d.add("s_state", *_dict_col(r, n, STATES[:12]))
    d.add("s_zip", r.integers(0, 400, n),
          dic=[f"{10000 + 89 * k % 90000:05d}" for k in range(400)])
    ncomp = max(min(n // 2, 6), 1)
    comp = r.integers(1, ncomp + 1, n)
    d.add("s_company_id", comp)
    d.add("s_company_name", comp - 1,
          dic=[f"company_{k+1}" for k in range(ncomp)])
    d.add("s_gmt_offset", (-5.0 - (np.arange(n) % 4)).astype(np.float64))
    d.add("s_market_id", r.integers(1, 11, n))
    return d


def _gen_warehouse(g: _Gen) -> Dataset:
    r = g.rng("warehouse")
    n = g.sizes["warehouse"]
    d = Dataset("warehouse", n)
    d.add("w_warehouse_sk", np.arange(1, n + 1, dtype=np.int64))
    d.add("w_warehouse_name", np.arange(n, dtype=np.int64),
          dic=[f"warehouse_{k}" for k in range(n)])
    d.add("w_warehouse_sq_ft", r.integers(50_000, 1_000_001, n))
    d.add("w_city", *_dict_col(r, n, CITIES[:20]))
    d.add("w_county", *_dict_col(r, n, COUNTIES[:15]))
    d.add("w_state", *_dict_col(r, n, STATES[:12]))
    d.add("w_country", np.zeros(n, dtype=np.int64), dic=["United States"])
    return d


def _gen_web_site(g: _Gen) -> Dataset:
    r = g.rng("web_site")
    n = g.sizes["web_site"]
    d = Dataset("web_site", n)
    d.add("web_site_sk", np.arange(1, n + 1, dtype=np.int64))
    d.add("web_site_id", np.arange(n, dtype=np.int64),
          dic=[f"AAAAAAAA{k:08d}" for k in range(n)])
    nname = max(n // 2, 1)
    d.add("web_name", np.arange(n, dtype=np.int64) % nname,
          dic=[f"site_{k}" for k in range(nname)])
    d.add("web_company_name", *_dict_col(
        r, n, [f"webco_{k}" for k in range(6)]))
    return d


def _gen_web_page(g: _Gen) -> Dataset:
    r = g.rng("web_page")
    n = g.sizes["web_page"]
    d = Dataset("web_page", n)
    d.add("wp_web_page_sk", np.arange(1, n + 1, dtype=np.int64))
    d.add("wp_char_count", r.integers(100, 8001, n))
    return d


def _gen_call_center(g: _Gen) -> Dataset:
    r = g.rng("call_center")
    n = g.sizes["call_center"]
    d = Dataset("call_center", n)
    d.add("cc_call_center_sk", np.arange(1, n + 1, dtype=np.int64))
    d.add("cc_call_center_id", np.arange(n, dtype=np.int64),
          dic=[f"AAAAAAAA{k:08d}" for k in range(n)])
    nname = max(n // 2, 1)
    d.add("cc_name", np.arange(n, dtype=np.int64) % nname,
          dic=[f"call_center_{k}" for k in range(nname)])
    d.add("cc_county", *_dict_col(r, n, COUNTIES[:8]))
    d.add("cc_manager", *_dict_col(r, n, [f"Manager{k}" for k in range(20)]))
    return d


def _gen_catalog_page(g: _Gen) -> Dataset:
    n = g.sizes["catalog_page"]
    d = Dataset("catalog_page", n)
    d.add("cp_catalog_page_sk", np.arange(1, n + 1, dtype=np.int64))
    d.add("cp_catalog_page_id", np.arange(n, dtype=np.int64),
          dic=[f"AAAAAAAA{k:08d}" for k in range(n)])
    return d


def _gen_promotion(g: _Gen) -> Dataset:
    r = g.rng("promotion")
    n = g.sizes["promotion"]
    d = Dataset("promotion", n)
    d.add("p_promo_sk", np.arange(1, n + 1, dtype=np.int64))
    d.add("p_promo_id", np.arange(n, dtype=np.int64),
          dic=[f"AAAAAAAA{k:08d}" for k in range(n)])
    for c in ("p_channel_dmail", "p_channel_email", "p_channel_tv",
              "p_channel_event", "p_channel_catalog"):
        d.add(c, *_dict_col(r, n, ["N", "Y"], 0.01))
    return d


def _gen_reason(g: _Gen) -> Dataset:
    n = g.sizes["reason"]
    d = Dataset("reason", n)
    d.add("r_reason_sk", np.arange(1, n + 1, dtype=np.int64))
    d.add("r_reason_desc", np.arange(n, dtype=np.int64),
          dic=[f"reason {k}" for k in range(n)])
    return d


def _gen_ship_mode(g: _Gen) -> Dataset:
    r = g.rng("ship_mode")
    n = g.sizes["ship_mode"]
    d = Dataset("ship_mode", n)
    d.add("sm_ship_mode_sk", np.arange(1, n + 1, dtype=np.int64))
    d.add("sm_type", np.arange(n, dtype=np.int64) % len(SHIP_TYPES),
          dic=SHIP_TYPES)
    d.add("sm_carrier", *_dict_col(r, n, [f"carrier_{k}" for k in range(20)]))
    return d


# --- fact tables ------------------------------------------------------------

def _gen_store_sales(g: _Gen) -> Dataset:
    r = g.rng("store_sales")
    n = g.nrows("store_sales")
    d = Dataset("store_sales", n, sharded=True)
    # ticket-level attributes: all line items of one ticket share date,
    # time, customer, demographics, address and store (as dsdgen does) —
    # group-by-ticket queries (q34/q46/q68/q73/q79) depend on this
    nt = max(n // 12, 1)
    tk = r.integers(0, nt, n)

    def per_ticket(vals, valid):
        return vals[tk], (valid[tk] if valid is not None else None)

    d.add("ss_sold_date_sk", *per_ticket(*g.date_fk(r, nt, 0.02)))
    tt = r.integers(0, 86_400, nt)
    ttv = r.random(nt) >= 0.02
    d.add("ss_sold_time_sk", tt[tk], ttv[tk])
    d.add("ss_item_sk", *g.fk(r, n, g.sizes["item"]))
    d.add("ss_customer_sk",
          *per_ticket(*g.fk(r, nt, g.sizes["customer"], 0.03)))
    d.add("ss_cdemo_sk",
          *per_ticket(*g.fk(r, nt, g.sizes["customer_demographics"], 0.03)))
    d.add("ss_hdemo_sk",
          *per_ticket(*g.fk(r, nt, g.sizes["household_demographics"],
                            0.03)))
    d.add("ss_addr_sk",
          *per_ticket(*g.fk(r, nt, g.sizes["customer_address"], 0.03)))
    d.add("ss_store_sk", *per_ticket(*g.fk(r, nt, g.sizes["store"], 0.02)))
    pr, prv = g.fk(r, n, g.sizes["promotion"], 0.02)
    d.add("ss_promo_sk", pr, prv)
    d.add("ss_ticket_number", tk + 1)
    d.add("ss_quantity", r.integers(1, 101, n), valid=r.random(n) >= 0.02)
    whole = g.money(r, n, 1, 100)
    lst = np.round(whole * r.uniform(1.0, 2.0, n), 2)
    sales = np.round(lst * r.uniform(0.2, 1.0, n), 2)
    qty = d.columns["ss_quantity"]
    d.add("ss_wholesale_cost", whole, valid=r.random(n) >= 0.02)
    d.add("ss_list_price", lst, valid=r.random(n) >= 0.02)
    d.add("ss_sales_price", sales, valid=r.random(n) >= 0.02)
    d.add("ss_ext_discount_amt", np.round((lst - sales) * qty, 2))
    d.add("ss_ext_sales_price", np.round(sales * qty, 2),
          valid=r.random(n) >= 0.02)
    d.add("ss_ext_wholesale_cost", np.round(whole * qty, 2))
    d.add("ss_ext_list_price", np.round(lst * qty, 2))
    d.add("ss_ext_tax", np.round(sales * qty * 0.08, 2))
    d.add("ss_coupon_amt", np.round(
        np.where(r.random(n) < 0.2, sales * qty * 0.1, 0.0), 2))
    net_paid = np.round(sales * qty - d.columns["ss_coupon_amt"], 2)
    d.add("ss_net_paid", net_paid, valid=r.random(n) >= 0.02)
    d.add("ss_net_paid_inc_tax", np.round(net_paid * 1.08, 2))
    d.add("ss_net_profit", np.round(net_paid - whole * qty, 2),
          valid=r.random(n) >= 0.02)
    return d


def _sample_sales(r, sales: Dataset, n):
    idx = r.integers(0, sales.nrows, n)
    return idx


def _gen_store_returns(g: _Gen, ss: Dataset) -> Dataset:
    r = g.rng("store_returns")
    n = g.nrows("store_returns")
    d = Dataset("store_returns", n, sharded=True)
    idx = _sample_sales(r, ss, n)
    dk, dkv = g.date_fk(r, n, 0.02)
    d.add("sr_returned_date_sk", dk, dkv)
    d.add("sr_item_sk", ss.columns["ss_item_sk"][idx])
    ck = ss.columns["ss_customer_sk"][idx]
    ckv = (ss.valid["ss_customer_sk"][idx]
           if ss.valid["ss_customer_sk"] is not None else None)
    d.add("sr_customer_sk", ck, ckv)
    cd, cdv = g.fk(r, n, g.sizes["customer_demographics"], 0.03)
    d.add("sr_cdemo_sk", cd, cdv)
    hd, hdv = g.fk(r, n, g.sizes["household_demographics"], 0.03)
    d.add("sr_hdemo_sk", hd, hdv)
    st = ss.columns["ss_store_sk"][idx]
    stv = (ss.valid["ss_store_sk"][idx]
           if ss.valid["ss_store_sk"] is not None else None)
    d.add("sr_store_sk", st, stv)
    re_, rev = g.fk(r, n, g.sizes["reason"], 0.02)
    d.add("sr_reason_sk", re_, rev)
    d.add("sr_ticket_number", ss.columns["ss_ticket_number"][idx])
    d.add("sr_return_quantity", r.integers(1, 51, n),
          valid=r.random(n) >= 0.02)
    amt = g.money(r, n, 1, 2000)
    d.add("sr_return_amt", amt, valid=r.random(n) >= 0.02)
    d.add("sr_return_tax", np.round(amt * 0.08, 2))
    d.add("sr_return_amt_inc_tax", np.round(amt * 1.08, 2))
    d.add("sr_fee", g.money(r, n, 0.5, 100))
    d.add("sr_return_ship_cost", g.money(r, n, 0, 500))
    d.add("sr_refunded_cash", np.round(amt * r.uniform(0, 1, n), 2))
    d.add("sr_reversed_charge", g.money(r, n, 0, 300))
    d.add("sr_store_credit", g.money(r, n, 0, 300))
    d.add("sr_net_loss", g.money(r, n, 0.5, 1000),
          valid=r.random(n) >= 0.02)
    return d


def _gen_catalog_sales(g: _Gen) -> Dataset:
    r = g.rng("catalog_sales")
    n = g.nrows("catalog_sales")
    d = Dataset("catalog_sales", n, sharded=True)
    # order-level attributes (dates, customers, addresses, call center,
    # ship mode) are shared by an order's line items; warehouse and
    # catalog page stay per-item (q16/q94-style multi-warehouse orders)
    nt = max(n // 10, 1)
    tk = r.integers(0, nt, n)

    def per_order(vals, valid):
        return vals[tk], (valid[tk] if valid is not None else None)

    dk, dkv = g.date_fk(r, nt, 0.02)
    d.add("cs_sold_date_sk", *per_order(dk, dkv))
    tt = r.integers(0, 86_400, nt)
    d.add("cs_sold_time_sk", tt[tk])
    sh = dk + r.integers(1, 90, nt)
    d.add("cs_ship_date_sk", *per_order(sh, dkv))
    ck, ckv = g.fk(r, nt, g.sizes["customer"], 0.02)
    d.add("cs_bill_customer_sk", *per_order(ck, ckv))
    cd, cdv = g.fk(r, nt, g.sizes["customer_demographics"], 0.02)
    d.add("cs_bill_cdemo_sk", *per_order(cd, cdv))
    hd, hdv = g.fk(r, nt, g.sizes["household_demographics"], 0.02)
    d.add("cs_bill_hdemo_sk", *per_order(hd, hdv))
    ba, bav = g.fk(r, nt, g.sizes["customer_address"], 0.02)
    d.add("cs_bill_addr_sk", *per_order(ba, bav))
    # ship customer == bill customer for ~85% of orders
    ship = ck.copy()
    other = r.integers(1, g.sizes["customer"] + 1, nt)
    swap = r.random(nt) < 0.15
    ship[swap] = other[swap]
    d.add("cs_ship_customer_sk", *per_order(ship, ckv))
    sa, sav = g.fk(r, nt, g.sizes["customer_address"], 0.02)
    d.add("cs_ship_addr_sk", *per_order(sa, sav))
    cc, ccv = g.fk(r, nt, g.sizes["call_center"], 0.02)
    d.add("cs_call_center_sk", *per_order(cc, ccv))
    cp, cpv = g.fk(r, n, g.sizes["catalog_page"], 0.02)
    d.add("cs_catalog_page_sk", cp, cpv)
    sm, smv = g.fk(r, nt, g.sizes["ship_mode"], 0.02)
    d.add("cs_ship_mode_sk", *per_order(sm, smv))
    wh, whv = g.fk(r, n, g.sizes["warehouse"], 0.02)
    d.add("cs_warehouse_sk", wh, whv)
    d.add("cs_item_sk", *g.fk(r, n, g.sizes["item"]))
    pr, prv = g.fk(r, n, g.sizes["promotion"], 0.02)
    d.add("cs_promo_sk", pr, prv)
    d.add("cs_order_number", tk + 1)
    d.add("cs_quantity", r.integers(1, 101, n), valid=r.random(n) >= 0.02)
    whole = g.money(r, n, 1, 100)
    lst = np.round(whole * r.uniform(1.0, 2.0, n), 2)
    sales = np.round(lst * r.uniform(0.2, 1.0, n), 2)
    qty = d.columns["cs_quantity"]
    d.add("cs_wholesale_cost", whole, valid=r.random(n) >= 0.02)
    d.add("cs_list_price", lst, valid=r.random(n) >= 0.02)
    d.add("cs_sales_price", sales, valid=r.random(n) >= 0.02)
    d.add("cs_ext_discount_amt", np.round((lst - sales) * qty, 2))
    d.add("cs_ext_sales_price", np.round(sales * qty, 2))
    d.add("cs_ext_wholesale_cost", np.round(whole * qty, 2))
    d.add("cs_ext_list_price", np.round(lst * qty, 2))
    d.add("cs_ext_tax", np.round(sales * qty * 0.08, 2))
    d.add("cs_coupon_amt", np.round(
        np.where(r.random(n) < 0.2, sales * qty * 0.1, 0.0), 2))
    d.add("cs_ext_ship_cost", g.money(r, n, 0, 200))
    net_paid = np.round(sales * qty - d.columns["cs_coupon_amt"], 2)
    d.add("cs_net_paid", net_paid)
    d.add("cs_net_paid_inc_tax", np.round(net_paid * 1.08, 2))
    d.add("cs_net_paid_inc_ship",
          np.round(net_paid + d.columns["cs_ext_ship_cost"], 2))
    d.add("cs_net_paid_inc_ship_tax", np.round(
        net_paid * 1.08 + d.columns["cs_ext_ship_cost"], 2))
    d.add("cs_net_profit", np.round(net_paid - whole * qty, 2))
    return d


def _gen_catalog_returns(g: _Gen, cs: Dataset) -> Dataset:
    r = g.rng("catalog_returns")
    n = g.nrows("catalog_returns")
    d = Dataset("catalog_returns", n, sharded=True)
    idx = _sample_sales(r, cs, n)
    dk, dkv = g.date_fk(r, n, 0.02)
    d.add("cr_returned_date_sk", dk, dkv)
    d.add("cr_item_sk", cs.columns["cs_item_sk"][idx])
    ck = cs.columns["cs_bill_customer_sk"][idx]
    ckv = (cs.valid["cs_bill_customer_sk"][idx]
           if cs.valid["cs_bill_customer_sk"] is not None else None)
    d.add("cr_refunded_customer_sk", ck, ckv)
    d.add("cr_returning_customer_sk", ck, ckv)
    ra, rav = g.fk(r, n, g.sizes["customer_address"], 0.02)
    d.add("cr_returning_addr_sk", ra, rav)
    cc = cs.columns["cs_call_center_sk"][idx]
    ccv = (cs.valid["cs_call_center_sk"][idx]
           if cs.valid["cs_call_center_sk"] is not None else None)
    d.add("cr_call_center_sk", cc, ccv)
    cp, cpv = g.fk(r, n, g.sizes["catalog_page"], 0.02)
    d.add("cr_catalog_page_sk", cp, cpv)
    re_, rev = g.fk(r, n, g.sizes["reason"], 0.02)
    d.add("cr_reason_sk", re_, rev)
    wh, whv = g.fk(r, n, g.sizes["warehouse"], 0.02)
    d.add("cr_warehouse_sk", wh, whv)
    d.add("cr_order_number", cs.columns["cs_order_number"][idx])
    d.add("cr_return_quantity", r.integers(1, 51, n),
          valid=r.random(n) >= 0.02)
    amt = g.money(r, n, 1, 2000)
    d.add("cr_return_amount", amt, valid=r.random(n) >= 0.02)
    d.add("cr_return_tax", np.round(amt * 0.08, 2))
    d.add("cr_return_amt_inc_tax", np.round(amt * 1.08, 2))
    d.add("cr_fee", g.money(r, n, 0.5, 100))
    d.add("cr_return_ship_cost", g.money(r, n, 0, 500))
    d.add("cr_refunded_cash", np.round(amt * r.uniform(0, 1, n), 2))
    d.add("cr_reversed_charge", g.money(r, n, 0, 300))
    d.add("cr_store_credit", g.money(r, n, 0, 300))
    d.add("cr_net_loss", g.money(r, n, 0.5, 1000))
    return d


def _gen_web_sales(g: _Gen) -> Dataset:
    r = g.rng("web_sales")
    n = g.nrows("web_sales")
    d = Dataset("web_sales", n, sharded=True)
    nt = max(n // 8, 1)
    tk = r.integers(0, nt, n)

    def per_order(vals, valid):
        return vals[tk], (valid[tk] if valid is not None else None)

    dk, dkv = g.date_fk(r, nt, 0.02)
    d.add("ws_sold_date_sk", *per_order(dk, dkv))
    d.add("ws_sold_time_sk", r.integers(0, 86_400, nt)[tk])
    d.add("ws_ship_date_sk", *per_order(dk + r.integers(1, 90, nt), dkv))
    d.add("ws_item_sk", *g.fk(r, n, g.sizes["item"]))
    ck, ckv = g.fk(r, nt, g.sizes["customer"], 0.02)
    d.add("ws_bill_customer_sk", *per_order(ck, ckv))
    cd, cdv = g.fk(r, nt, g.sizes["customer_demographics"], 0.02)
    d.add("ws_bill_cdemo_sk", *per_order(cd, cdv))
    hd, hdv = g.fk(r, nt, g.sizes["household_demographics"], 0.02)
    d.add("ws_bill_hdemo_sk", *per_order(hd, hdv))
    shd, shdv = g.fk(r, nt, g.sizes["household_demographics"], 0.02)
    d.add("ws_ship_hdemo_sk", *per_order(shd, shdv))
    ba, bav = g.fk(r, nt, g.sizes["customer_address"], 0.02)
    d.add("ws_bill_addr_sk", *per_order(ba, bav))
    ship = ck.copy()
    other = r.integers(1, g.sizes["customer"] + 1, nt)
    swap = r.random(nt) < 0.15
    ship[swap] = other[swap]
    d.add("ws_ship_customer_sk", *per_order(ship, ckv))
    sa, sav = g.fk(r, nt, g.sizes["customer_address"], 0.02)
    d.add("ws_ship_addr_sk", *per_order(sa, sav))
    wp, wpv = g.fk(r, nt, g.sizes["web_page"], 0.02)
    d.add("ws_web_page_sk", *per_order(wp, wpv))
    wsit, wsitv = g.fk(r, nt, g.sizes["web_site"], 0.02)
    d.add("ws_web_site_sk", *per_order(wsit, wsitv))
    sm, smv = g.fk(r, nt, g.sizes["ship_mode"], 0.02)
    d.add("ws_ship_mode_sk", *per_order(sm, smv))
    wh, whv = g.fk(r, n, g.sizes["warehouse"], 0.02)
    d.add("ws_warehouse_sk", wh, whv)
    pr, prv = g.fk(r, n, g.sizes["promotion"], 0.02)
    d.add("ws_promo_sk", pr, prv)
    d.add("ws_order_number", tk + 1)
    d.add("ws_quantity", r.integers(1, 101, n), valid=r.random(n) >= 0.02)
    whole = g.money(r, n, 1, 100)
    lst = np.round(whole * r.uniform(1.0, 2.0, n), 2)
    sales = np.round(lst * r.uniform(0.2, 1.0, n), 2)
    qty = d.columns["ws_quantity"]
    d.add("ws_wholesale_cost", whole, valid=r.random(n) >= 0.02)
    d.add("ws_list_price", lst, valid=r.random(n) >= 0.02)
    d.add("ws_sales_price", sales, valid=r.random(n) >= 0.02)
    d.add("ws_ext_discount_amt", np.round((lst - sales) * qty, 2))
    d.add("ws_ext_sales_price", np.round(sales * qty, 2))
    d.add("ws_ext_wholesale_cost", np.round(whole * qty, 2))
    d.add("ws_ext_list_price", np.round(lst * qty, 2))
    d.add("ws_ext_tax", np.round(sales * qty * 0.08, 2))
    d.add("ws_coupon_amt", np.round(
        np.where(r.random(n) < 0.2, sales * qty * 0.1, 0.0), 2))
    d.add("ws_ext_ship_cost", g.money(r, n, 0, 200))
    net_paid = np.round(sales * qty - d.columns["ws_coupon_amt"], 2)
    d.add("ws_net_paid", net_paid)
    d.add("ws_net_paid_inc_tax", np.round(net_paid * 1.08, 2))
    d.add("ws_net_paid_inc_ship",
          np.round(net_paid + d.columns["ws_ext_ship_cost"], 2))
    d.add("ws_net_paid_inc_ship_tax", np.round(
        net_paid * 1.08 + d.columns["ws_ext_ship_cost"], 2))
    d.add("ws_net_profit", np.round(net_paid - whole * qty, 2))
    return d


def _gen_web_returns(g: _Gen, ws: Dataset) -> Dataset:
    r = g.rng("web_returns")
    n = g.nrows("web_returns")
    d = Dataset("web_returns", n, sharded=True)
    idx = _sample_sales(r, ws, n)
    dk, dkv = g.date_fk(r, n, 0.02)
    d.add("wr_returned_date_sk", dk, dkv)
    d.add("wr_item_sk", ws.columns["ws_item_sk"][idx])
    ck = ws.columns["ws_bill_customer_sk"][idx]
    ckv = (ws.valid["ws_bill_customer_sk"][idx]
           if ws.valid["ws_bill_customer_sk"] is not None else None)
    d.add("wr_refunded_customer_sk", ck, ckv)
    d.add("wr_returning_customer_sk", ck, ckv)
    cd, cdv = g.fk(r, n, g.sizes["customer_demographics"], 0.02)
    d.add("wr_refunded_cdemo_sk", cd, cdv)
    d.add("wr_returning_cdemo_sk", cd, cdv)
    ra, rav = g.fk(r, n, g.sizes["customer_address"], 0.02)
    d.add("wr_refunded_addr_sk", ra, rav)
    d.add("wr_returning_addr_sk", ra, rav)
    wp, wpv = g.fk(r, n, g.sizes["web_page"], 0.02)
    d.add("wr_web_page_sk", wp, wpv)
    re_, rev = g.fk(r, n, g.sizes["reason"], 0.02)
    d.add("wr_reason_sk", re_, rev)
    d.add("wr_order_number", ws.columns["ws_order_number"][idx])
    d.add("wr_return_quantity", r.integers(1, 51, n),
          valid=r.random(n) >= 0.02)
    amt = g.money(r, n, 1, 2000)
    d.add("wr_return_amt", amt, valid=r.random(n) >= 0.02)
    d.add("wr_return_tax", np.round(amt * 0.08, 2))
    d.add("wr_return_amt_inc_tax", np.round(amt * 1.08, 2))
    d.add("wr_fee", g.money(r, n, 0.5, 100))
    d.add("wr_return_ship_cost", g.money(r, n, 0, 500))
    d.add("wr_refunded_cash", np.round(amt * r.uniform(0, 1, n), 2))
    d.add("wr_reversed_charge", g.money(r, n, 0, 300))
    d.add("wr_account_credit", g.money(r, n, 0, 300))
    d.add("wr_net_loss", g.money(r, n, 0.5, 1000))
    return d


def _gen_inventory(g: _Gen) -> Dataset:
    r = g.rng("inventory")
    n = g.nrows("inventory")
    d = Dataset("inventory", n, sharded=True)
    dk, _ = g.date_fk(r, n)
    d.add("inv_date_sk", dk)
    d.add("inv_item_sk", *g.fk(r, n, g.sizes["item"]))
    d.add("inv_warehouse_sk", *g.fk(r, n, g.sizes["warehouse"]))
    d.add("inv_quantity_on_hand", r.integers(0, 1001, n),
          valid=r.random(n) >= 0.05)
    return d


_DIM_GENS = {
    "date_dim": _gen_date_dim, "time_dim": _gen_time_dim, "item": _gen_item,
    "customer": _gen_customer, "customer_address": _gen_customer_address,
    "customer_demographics": _gen_customer_demographics,
    "household_demographics": _gen_household_demographics,
    "income_band": _gen_income_band, "store": _gen_store,
    "warehouse": _gen_warehouse, "web_site": _gen_web_site,
    "web_page": _gen_web_page, "call_center": _gen_call_center,
    "catalog_page": _gen_catalog_page, "promotion": _gen_promotion,
    "reason": _gen_reason, "ship_mode": _gen_ship_mode,
    "inventory": _gen_inventory, "store_sales": _gen_store_sales,
    "catalog_sales": _gen_catalog_sales, "web_sales": _gen_web_sales,
}


# fact -> (sales generator, ticket/order column, line items per ticket,
#          paired returns table + generator)
_FACTS = {
    "store_sales": (_gen_store_sales, "ss_ticket_number", 12,
                    "store_returns", _gen_store_returns),
    "catalog_sales": (_gen_catalog_sales, "cs_order_number", 10,
                      "catalog_returns", _gen_catalog_returns),
    "web_sales": (_gen_web_sales, "ws_order_number", 8,
                  "web_returns", _gen_web_returns),
}


def gen_catalog(sf: float = 0.01, world: int = 1, rank: int = 0,
                tables: Optional[List[str]] = None) -> Dict[str, Dataset]:
    """Generate the full NDS catalog (or a subset) at scale factor `sf`.

    Sharded (fact) tables are built from this rank's deterministic blocks;
    the union over ranks equals the world=1 table exactly.
    """
    g = _Gen(sf, world, rank)
    want = set(tables) if tables else set(TABLES)
    for sales, (_f, _t, _k, rets, _rf) in _FACTS.items():
        if rets in want:
            want.add(sales)
    out: Dict[str, Dataset] = {}
    for name, fn in _DIM_GENS.items():
        if TABLES[name][2]:
            continue  # facts handled block-wise below
        if name in want:
            out[name] = fn(g)
    my_blocks = [b for b in range(NBLOCKS) if b % world == rank]
    if "inventory" in want:
        out["inventory"] = _concat_datasets(
            "inventory", [_gen_inventory(_BlockGen(g, b))
                          for b in my_blocks])
    for sales, (sfn, tcol, k, rets, rfn) in _FACTS.items():
        if sales not in want:
            continue
        # per-block order-id stride keeps ticket numbers globally unique
        stride = g.sizes[sales] // NBLOCKS // k + 2
        sparts, rparts = [], []
        for b in my_blocks:
            bg = _BlockGen(g, b)
            sb = sfn(bg)
            sb.columns[tcol] = sb.columns[tcol] + b * stride
            sparts.append(sb)
            if rets in want:
                rparts.append(rfn(bg, sb))
        out[sales] = _concat_datasets(sales, sparts)
        if rets in want:
            out[rets] = _concat_datasets(rets, rparts)
    return out
