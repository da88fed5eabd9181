"""Deterministic synthetic TPC-DS data generator (dsdgen stand-in).

There is no network access for real dsdgen data, so tables are generated
synthetically with dsdgen-like shapes: correct schemas, referential
integrity (fact FKs -> dim PKs), realistic null fractions, and the
1998-2002 sales calendar. Deterministic per (table, part): rank r of a
distributed job generates exactly its shard.

Parquet written via pyarrow (host); scans read it back through
ParquetScan (host page read -> device columns).
"""
from __future__ import annotations

import os
from typing import List, Optional

import numpy as np

from .schema import BASE_ROWS, FACT_TABLES, SCHEMAS, row_count

JULIAN_BASE = 2415022  # d_date_sk of calendar index 0 == 1900-01-01
EPOCH_IDX = 25567  # calendar index of 1970-01-01
SALES_START_IDX = 35794  # calendar index of 1998-01-01
SALES_DAYS = 1826  # 1998-01-01 .. 2002-12-31

_CATEGORIES = ["Books", "Children", "Electronics", "Home", "Jewelry",
               "Men", "Music", "Shoes", "Sports", "Women"]
_STATES = ["TN", "GA", "AL", "SC", "NC", "KY", "VA", "FL", "MS", "LA",
           "TX", "OH", "IL", "IN", "MI", "CA", "NY", "PA", "WA", "OR"]
_COUNTRIES = ["United States"] * 9 + ["Canada"]
_CITIES = ["Midway", "Fairview", "Oak Grove", "Five Points", "Centerville",
           "Liberty", "Pleasant Hill", "Mount Zion", "Salem", "Union",
           "Riverside", "Greenfield", "Oakland", "Springdale", "Shiloh",
           "Edgewood"]
# county universe = the literals the benchmark queries filter on plus
# city-derived names (q54 joins ca_county = s_county, so both tables draw
# from this same pool)
_COUNTIES = ["Williamson County", "Rush County", "Toole County",
             "Jefferson County", "Dona Ana County", "La Porte County",
             "Bronx County", "Franklin Parish", "Orange County"] + \
    [f"{c} County" for c in _CITIES]
_FIRST = ["James", "Mary", "John", "Patricia", "Robert", "Jennifer", "Michael",
          "Linda", "William", "Barbara", "David", "Susan", "Richard", "Jessica"]
_LAST = ["Smith", "Johnson", "Williams", "Brown", "Jones", "Garcia", "Miller",
         "Davis", "Rodriguez", "Martinez", "Hernandez", "Lopez", "Gonzalez"]
_BUY_POTENTIAL = [">10000", "5001-10000", "1001-5000", "501-1000", "0-500", "Unknown"]
_EDUCATION = ["Primary", "Secondary", "College", "2 yr Degree", "4 yr Degree",
              "Advanced Degree", "Unknown"]

# dsdgen item domains: per-category class lists (tpcds.dst classes) and the
# brand-name vocabulary; the named brands are the exact literals the
# benchmark queries filter on, kept dense enough to select rows at small SF
_CLASSES = [
    ["arts", "business", "computers", "cooking", "entertainments", "fiction",
     "history", "home repair", "mystery", "parenting", "reference", "romance",
     "science", "self-help", "sports", "travel"],  # Books
    ["infants", "newborn", "school-uniforms", "toddlers"],  # Children
    ["audio", "automotive", "cameras", "camcorders", "dvd/vcr players",
     "karoke", "memory", "monitors", "musical", "personal", "portable",
     "scanners", "stereo", "televisions", "wireless", "disk drives"],  # Electronics
    ["accent", "bathroom", "bedding", "blinds/shades", "curtains/drapes",
     "decor", "flatware", "furniture", "glassware", "kids", "lighting",
     "mattresses", "paint", "rugs", "tables", "wallpaper"],  # Home
    ["birdal", "costume", "diamonds", "estate", "gold", "jewelry boxes",
     "loose stones", "mens watch", "pendants", "rings", "semi-precious",
     "womens watch"],  # Jewelry
    ["accessories", "pants", "shirts", "sportswear"],  # Men
    ["classical", "country", "pop", "rock"],  # Music
    ["athletic", "kids", "mens", "womens"],  # Shoes
    ["archery", "athletic shoes", "baseball", "basketball", "camping",
     "fishing", "fitness", "football", "golf", "guns", "hockey", "optics",
     "outdoor", "pools", "sailing", "tennis"],  # Sports
    ["dresses", "fragrances", "maternity", "swimwear"],  # Women
]
_NAMED_BRANDS = [
    "amalgimporto #1", "edu packscholar #1", "exportiimporto #1",
    "exportiunivamalg #9", "importoamalg #1", "scholaramalgamalg #14",
    "scholaramalgamalg #7", "scholaramalgamalg #9",
]
_BRAND_SYL = ["amalg", "importo", "edu pack", "scholar", "brand", "corp",
              "maxi", "univ", "exporti", "nameless"]

_DAY_NAMES = ["Monday", "Tuesday", "Wednesday", "Thursday", "Friday",
              "Saturday", "Sunday"]  # 1900-01-01 was a Monday


def _seed(table: str, part: int) -> int:
    # stable across processes (builtin hash() is randomized per process,
    # which would make regenerated datasets irreproducible)
    import zlib

    return zlib.crc32(f"auron-tpcds-v1:{table}:{part}".encode()) % (2 ** 31)


def _rng(table: str, part: int) -> np.random.Generator:
    return np.random.default_rng(np.random.SeedSequence([_seed(table, part)]))


def _with_nulls(rng, arr, frac):
    if frac <= 0:
        return arr, None
    mask = rng.random(len(arr)) < frac
    return arr, ~mask  # validity


def _money(rng, n, lo=1.0, hi=100.0):
    return np.round(rng.uniform(lo, hi, n), 2)


def _id_str(prefix, arr):
    return [f"{prefix}{int(v):012d}" for v in arr]



def _row_item(j, n_item):
    """Stateless pseudo-random item for fact row j: returns tables can
    reconstruct the (line, item) pair of the sale they reference the way
    dsdgen's returns reference actual sales lines."""
    h = (j.astype(np.uint64) * np.uint64(0x9E3779B97F4A7C15)) >> np.uint64(17)
    return (h % np.uint64(max(n_item, 1))).astype(np.int64) + 1


def _calendar():
    n = BASE_ROWS["date_dim"]
    idx = np.arange(n, dtype=np.int64)
    dates = np.datetime64("1900-01-01") + idx.astype("timedelta64[D]")
    months = dates.astype("M8[M]")
    years = (dates.astype("M8[Y]").astype(np.int64) + 1970).astype(np.int32)
    moy = ((months.astype(np.int64) % 12) + 12) % 12 + 1
    dom = (dates - months.astype("M8[D]")).astype(np.int64) + 1
    return idx, dates, years, moy.astype(np.int32), dom.astype(np.int32)


def generate_table(name: str, sf: float, part: int = 0, nparts: int = 1):
    """-> pyarrow.Table for rows [part] of [nparts] shards."""
    import pyarrow as pa

    total = row_count(name, sf)
    lo = total * part // nparts
    hi = total * (part + 1) // nparts
    n = hi - lo
    rng = _rng(name, part)
    sks = np.arange(lo + 1, hi + 1, dtype=np.int64)  # 1-based surrogate keys

    n_item = row_count("item", sf)
    n_cust = row_count("customer", sf)
    n_addr = row_count("customer_address", sf)
    n_store = row_count("store", sf)
    n_cdemo = BASE_ROWS["customer_demographics"]
    n_hdemo = BASE_ROWS["household_demographics"]
    n_promo = BASE_ROWS["promotion"]
    n_wh = BASE_ROWS["warehouse"]

    def date_fk(null_frac=0.02):
        v = JULIAN_BASE + SALES_START_IDX + rng.integers(0, SALES_DAYS, n)
        return _with_nulls(rng, v, null_frac)

    def fk(card, null_frac=0.03):
        return _with_nulls(rng, rng.integers(1, card + 1, n), null_frac)

    cols = {}
    valids = {}

    def put(col, arr, validity=None):
        cols[col] = arr
        if validity is not None:
            valids[col] = validity

    if name == "date_dim":
        idx, dates, years, moy, dom = _calendar()
        put("d_date_sk", JULIAN_BASE + idx)
        put("d_date", (idx - EPOCH_IDX).astype(np.int32))
        put("d_year", years)
        put("d_moy", moy)
        put("d_dom", dom)
        put("d_qoy", ((moy - 1) // 3 + 1).astype(np.int32))
        put("d_day_name", [_DAY_NAMES[i % 7] for i in idx])
        put("d_month_seq", ((years.astype(np.int64) - 1900) * 12 + moy - 1).astype(np.int32))
        put("d_week_seq", (idx // 7).astype(np.int32) + 1)
        put("d_dow", (idx % 7).astype(np.int32))
        qoy = ((moy - 1) // 3 + 1)
        put("d_quarter_name", [f"{int(y)}Q{int(q)}" for y, q in zip(years, qoy)])
    elif name == "time_dim":
        t = np.arange(n, dtype=np.int64) + lo
        put("t_time_sk", t)
        put("t_time", t)
        hour = (t // 3600).astype(np.int32)
        put("t_hour", hour)
        put("t_minute", ((t % 3600) // 60).astype(np.int32))
        meal = np.where(hour < 9, "breakfast",
                        np.where((hour >= 11) & (hour < 14), "lunch",
                                 np.where((hour >= 17) & (hour < 21), "dinner", "")))
        put("t_meal_time", meal.tolist())
    elif name == "item":
        put("i_item_sk", sks)
        put("i_item_id", _id_str("AAAAAAAA", sks))
        cat_id = ((sks - 1) % 10 + 1).astype(np.int32)
        put("i_category", [_CATEGORIES[c - 1] for c in cat_id])
        put("i_category_id", cat_id)
        mfg = rng.integers(1, 1001, n).astype(np.int32)
        put("i_manufact_id", mfg, _with_nulls(rng, mfg, 0.01)[1])
        brand_id = (cat_id.astype(np.int64) * 1000000 + mfg * 100 + rng.integers(1, 10, n)).astype(np.int32)
        put("i_brand_id", brand_id)
        # dsdgen-style brand names: ~30% of items carry one of the named
        # brands the benchmark queries filter on; the rest get random
        # syllable-pair brands (mk_word emulation)
        named = rng.random(n) < 0.30
        bname_named = rng.integers(0, len(_NAMED_BRANDS), n)
        s1 = rng.integers(0, len(_BRAND_SYL), n)
        s2 = rng.integers(0, len(_BRAND_SYL), n)
        bnum = rng.integers(1, 17, n)
        put("i_brand", [(_NAMED_BRANDS[int(bn)] if nm else
                         f"{_BRAND_SYL[int(a)]}{_BRAND_SYL[int(b)]} #{int(k)}")
                        for nm, bn, a, b, k in zip(named, bname_named, s1, s2, bnum)])
        cls_idx = rng.integers(0, 64, n)
        put("i_class", [_CLASSES[int(c - 1)][int(v) % len(_CLASSES[int(c - 1)])]
                        for c, v in zip(cat_id, cls_idx)])
        put("i_class_id", (cls_idx % 16 + 1).astype(np.int32))
        put("i_current_price", _money(rng, n, 0.09, 99.0), _with_nulls(rng, sks, 0.01)[1])
        put("i_wholesale_cost", _money(rng, n, 0.02, 88.0), _with_nulls(rng, sks, 0.01)[1])
        put("i_manager_id", rng.integers(1, 101, n).astype(np.int32))
        put("i_product_name", [f"product{int(s)}" for s in sks])
        put("i_item_desc", [f"the quite famous item number {int(s)} description" for s in sks])
        _colors = ["red", "blue", "green", "yellow", "black", "white", "purple",
                   "orange", "pink", "brown", "gray", "cyan", "pale", "powder",
                   "khaki", "midnight", "snow", "forest", "ghost", "floral",
                   "blanched", "burlywood", "burnished", "chiffon",
                   "cornflower", "deep", "frosted", "honeydew", "indian",
                   "light", "medium", "papaya", "slate", "spring"]
        put("i_color", [_colors[int(v)] for v in rng.integers(0, len(_colors), n)])
        _units = ["Ounce", "Oz", "Bunch", "Ton", "N/A", "Dozen", "Box", "Pound",
                  "Pallet", "Gross", "Cup", "Dram", "Each", "Tbl", "Lb", "Bundle"]
        put("i_units", [_units[int(v)] for v in rng.integers(0, len(_units), n)])
        put("i_size", [["small", "medium", "large", "extra large", "petite", "N/A"][int(v)]
                       for v in rng.integers(0, 6, n)])
        put("i_manufact", [f"manufact{int(m)}" for m in mfg])
    elif name == "customer":
        put("c_customer_sk", sks)
        put("c_customer_id", _id_str("AAAAAAAA", sks))
        put("c_first_name", [_FIRST[int(v)] for v in rng.integers(0, len(_FIRST), n)])
        put("c_last_name", [_LAST[int(v)] for v in rng.integers(0, len(_LAST), n)])
        a, av = fk(n_addr, 0.02)
        put("c_current_addr_sk", a, av)
        cd, cdv = fk(n_cdemo, 0.02)
        put("c_current_cdemo_sk", cd, cdv)
        hd, hdv = fk(n_hdemo, 0.02)
        put("c_current_hdemo_sk", hd, hdv)
        # dsdgen stores birth country uppercased (q24 matches upper(ca_country))
        put("c_birth_country", [_COUNTRIES[int(v)].upper() for v in rng.integers(0, len(_COUNTRIES), n)])
        put("c_birth_year", rng.integers(1924, 1993, n).astype(np.int32))
        put("c_birth_day", rng.integers(1, 29, n).astype(np.int32))
        put("c_birth_month", rng.integers(1, 13, n).astype(np.int32))
        put("c_login", ["" for _ in range(n)])
        put("c_email_address", [f"c{int(s)}@example.com" for s in sks])
        d, dv = date_fk(0.02)
        put("c_first_sales_date_sk", d, dv)
        d2, d2v = date_fk(0.02)
        put("c_first_shipto_date_sk", d2, d2v)
        put("c_preferred_cust_flag", ["Y" if v else "N" for v in rng.random(n) < 0.5])
        d3, d3v = date_fk(0.02)
        put("c_last_review_date_sk", d3, d3v)
        put("c_salutation", [["Mr.", "Mrs.", "Ms.", "Dr."][int(v)] for v in rng.integers(0, 4, n)])
    elif name == "customer_address":
        put("ca_address_sk", sks)
        put("ca_state", [_STATES[int(v)] for v in rng.integers(0, len(_STATES), n)],
            _with_nulls(rng, sks, 0.02)[1])
        put("ca_zip", [f"{int(v):05d}" for v in rng.integers(10000, 99999, n)])
        put("ca_country", [_COUNTRIES[int(v)] for v in rng.integers(0, len(_COUNTRIES), n)])
        put("ca_city", [_CITIES[int(v)] for v in rng.integers(0, len(_CITIES), n)])
        put("ca_county", [_COUNTIES[int(v)] for v in rng.integers(0, len(_COUNTIES), n)])
        put("ca_gmt_offset", rng.choice([-5.0, -6.0, -7.0, -8.0], n))
        put("ca_street_name", [f"{_LAST[int(v)]} St" for v in rng.integers(0, len(_LAST), n)])
        put("ca_street_type", [["Street", "Avenue", "Blvd", "Court", "Lane"][int(v)] for v in rng.integers(0, 5, n)])
        put("ca_location_type", [["apartment", "condo", "single family"][int(v)] for v in rng.integers(0, 3, n)])
        put("ca_suite_number", [f"Suite {int(v)}" for v in rng.integers(0, 100, n)])
        put("ca_street_number", [str(int(v)) for v in rng.integers(1, 1000, n)])
    elif name == "customer_demographics":
        put("cd_demo_sk", sks)
        put("cd_gender", ["M" if (s - 1) % 2 == 0 else "F" for s in sks])
        put("cd_marital_status", [["M", "S", "D", "W", "U"][(int(s) - 1) // 2 % 5] for s in sks])
        put("cd_education_status", [_EDUCATION[(int(s) - 1) // 10 % 7] for s in sks])
        put("cd_purchase_estimate", (((sks - 1) // 70 % 20 + 1) * 500).astype(np.int32))
        put("cd_credit_rating", [["Low Risk", "High Risk", "Good", "Unknown"][(int(s) - 1) // 1400 % 4] for s in sks])
        put("cd_dep_count", ((sks - 1) // 5600 % 7).astype(np.int32))
        put("cd_dep_employed_count", ((sks - 1) // 39200 % 7).astype(np.int32))
        put("cd_dep_college_count", ((sks - 1) // 274400 % 7).astype(np.int32))
    elif name == "household_demographics":
        put("hd_demo_sk", sks)
        put("hd_income_band_sk", (sks - 1) % 20 + 1)
        put("hd_buy_potential", [_BUY_POTENTIAL[(int(s) - 1) // 20 % 6] for s in sks])
        put("hd_dep_count", ((sks - 1) // 120 % 10).astype(np.int32))
        put("hd_vehicle_count", ((sks - 1) // 1200 % 6 - 1).astype(np.int32))
    elif name == "income_band":
        put("ib_income_band_sk", sks)
        put("ib_lower_bound", ((sks - 1) * 10000).astype(np.int32))
        put("ib_upper_bound", (sks * 10000).astype(np.int32))
    elif name == "store":
        put("s_store_sk", sks)
        put("s_store_id", _id_str("AAAAAAAA", (sks + 1) // 2))
        put("s_store_name", [["ought", "able", "pri", "ese", "anti", "cally"][int(s) % 6] for s in sks])
        # dsdgen: SF<=1000 stores are mostly TN
        put("s_state", ["TN" if int(s) % 4 != 0 else _STATES[int(s) % len(_STATES)] for s in sks])
        # half the stores share county names with the address universe so
        # county-equi joins (q54) are non-trivial
        put("s_county", ["Williamson County" if int(s2) % 2 == 0
                         else _COUNTIES[int(s2) % len(_COUNTIES)]
                         for s2 in sks])
        put("s_zip", [f"{int(v):05d}" for v in rng.integers(30000, 40000, n)])
        put("s_city", [_CITIES[int(s) % len(_CITIES)] for s in sks])
        put("s_number_employees", rng.integers(200, 301, n).astype(np.int32))
        put("s_gmt_offset", np.full(n, -5.0))
        put("s_company_id", np.ones(n, dtype=np.int32))
        put("s_company_name", ["Unknown" for _ in range(n)])
        put("s_street_number", [str(int(v)) for v in rng.integers(1, 1000, n)])
        put("s_street_type", [["Street", "Avenue", "Blvd", "Court", "Lane"][int(s) % 5] for s in sks])
        put("s_suite_number", [f"Suite {int(s) % 100}" for s in sks])
        put("s_market_id", ((sks - 1) % 10 + 1).astype(np.int32))
        put("s_street_name", [f"{_LAST[int(s) % len(_LAST)]} Blvd" for s in sks])
    elif name == "warehouse":
        put("w_warehouse_sk", sks)
        put("w_warehouse_name", [f"Warehouse {int(s)}" for s in sks])
        put("w_warehouse_sq_ft", rng.integers(50000, 1000000, n).astype(np.int32))
        put("w_state", ["TN"] * n)
        put("w_county", ["Williamson County"] * n)
        put("w_city", [_CITIES[int(s) % len(_CITIES)] for s in sks])
        put("w_country", ["United States"] * n)
    elif name == "promotion":
        put("p_promo_sk", sks)
        put("p_promo_id", _id_str("AAAAAAAA", sks))
        for c in ("p_channel_email", "p_channel_event", "p_channel_dmail", "p_channel_tv"):
            put(c, ["N" if v < 0.85 else "Y" for v in rng.random(n)])
    elif name == "reason":
        put("r_reason_sk", sks)
        put("r_reason_desc", [f"reason {int(s)}" for s in sks])
    elif name == "ship_mode":
        put("sm_ship_mode_sk", sks)
        put("sm_type", [["EXPRESS", "AIR", "SURFACE", "SEA", "OVERNIGHT"][int(s) % 5] for s in sks])
        put("sm_carrier", [["UPS", "FEDEX", "AIRBORNE", "USPS", "DHL"][int(s) % 5] for s in sks])
    elif name == "call_center":
        put("cc_call_center_sk", sks)
        put("cc_call_center_id", _id_str("AAAAAAAA", sks))
        put("cc_county", ["Williamson County"] * n)
        put("cc_name", [f"call center {int(s)}" for s in sks])
        put("cc_manager", [_FIRST[int(s) % len(_FIRST)] + " " + _LAST[int(s) % len(_LAST)] for s in sks])
    elif name == "web_site":
        put("web_site_sk", sks)
        put("web_site_id", _id_str("AAAAAAAA", sks))
        put("web_name", [f"site_{int(s) % 10}" for s in sks])
        put("web_company_name", [["pri", "able", "ese", "anti", "cally"][int(s) % 5] for s in sks])
    elif name == "web_page":
        put("wp_web_page_sk", sks)
        put("wp_char_count", rng.integers(100, 8000, n).astype(np.int32))
    elif name == "catalog_page":
        put("cp_catalog_page_sk", sks)
        put("cp_catalog_page_id", _id_str("AAAAAAAA", sks))
    elif name == "store_sales":
        d, dv = date_fk()
        put("ss_sold_date_sk", d, dv)
        t, tv = fk(86400, 0.02)
        put("ss_sold_time_sk", t, tv)
        put("ss_item_sk", _row_item(lo + np.arange(n, dtype=np.int64), n_item))
        c, cv = fk(n_cust)
        put("ss_customer_sk", c, cv)
        cd, cdv = fk(n_cdemo)
        put("ss_cdemo_sk", cd, cdv)
        hd, hdv = fk(n_hdemo)
        put("ss_hdemo_sk", hd, hdv)
        a, av = fk(n_addr)
        put("ss_addr_sk", a, av)
        st, stv = fk(n_store, 0.02)
        put("ss_store_sk", st, stv)
        p, pv = fk(n_promo, 0.02)
        put("ss_promo_sk", p, pv)
        put("ss_ticket_number", (lo + np.arange(n, dtype=np.int64)) // 3 + 1)
        qty = rng.integers(1, 101, n).astype(np.int32)
        put("ss_quantity", qty, _with_nulls(rng, qty, 0.02)[1])
        whole = _money(rng, n, 1, 100)
        lst = np.round(whole * rng.uniform(1.0, 2.5, n), 2)
        sales = np.round(lst * rng.uniform(0.0, 1.0, n), 2)
        put("ss_wholesale_cost", whole)
        put("ss_list_price", lst, _with_nulls(rng, lst, 0.02)[1])
        put("ss_sales_price", sales, _with_nulls(rng, sales, 0.02)[1])
        ext_sales = np.round(sales * qty, 2)
        ext_whole = np.round(whole * qty, 2)
        ext_list = np.round(lst * qty, 2)
        coupon = np.where(rng.random(n) < 0.1, np.round(ext_sales * rng.uniform(0, 0.5, n), 2), 0.0)
        put("ss_ext_discount_amt", np.round(ext_list - ext_sales, 2))
        put("ss_ext_sales_price", ext_sales, _with_nulls(rng, ext_sales, 0.02)[1])
        put("ss_ext_wholesale_cost", ext_whole)
        put("ss_ext_list_price", ext_list)
        put("ss_ext_tax", np.round(ext_sales * 0.09 * rng.random(n), 2))
        put("ss_coupon_amt", coupon)
        net_paid = np.round(ext_sales - coupon, 2)
        put("ss_net_paid", net_paid)
        put("ss_net_paid_inc_tax", np.round(net_paid * 1.05, 2))
        put("ss_net_profit", np.round(net_paid - ext_whole, 2), _with_nulls(rng, net_paid, 0.02)[1])
    elif name == "store_returns":
        d, dv = date_fk()
        put("sr_returned_date_sk", d, dv)
        t, tv = fk(86400, 0.02)
        put("sr_return_time_sk", t, tv)
        src_rows = rng.integers(0, max(row_count("store_sales", sf), 1), n)
        put("sr_item_sk", _row_item(src_rows, n_item))
        c, cv = fk(n_cust)
        put("sr_customer_sk", c, cv)
        cd, cdv = fk(n_cdemo)
        put("sr_cdemo_sk", cd, cdv)
        hd, hdv = fk(n_hdemo)
        put("sr_hdemo_sk", hd, hdv)
        a, av = fk(n_addr)
        put("sr_addr_sk", a, av)
        st, stv = fk(n_store, 0.02)
        put("sr_store_sk", st, stv)
        r, rv = fk(BASE_ROWS["reason"], 0.02)
        put("sr_reason_sk", r, rv)
        put("sr_ticket_number", src_rows // 3 + 1)
        q = rng.integers(1, 50, n).astype(np.int32)
        put("sr_return_quantity", q, _with_nulls(rng, q, 0.02)[1])
        amt = _money(rng, n, 1, 2000)
        put("sr_return_amt", amt, _with_nulls(rng, amt, 0.02)[1])
        put("sr_return_tax", np.round(amt * 0.09 * rng.random(n), 2))
        put("sr_return_amt_inc_tax", np.round(amt * 1.05, 2))
        put("sr_fee", _money(rng, n, 0.5, 100))
        put("sr_return_ship_cost", _money(rng, n, 0, 500))
        put("sr_refunded_cash", np.round(amt * rng.uniform(0, 1, n), 2))
        put("sr_reversed_charge", np.round(amt * rng.uniform(0, 0.5, n), 2))
        put("sr_store_credit", np.round(amt * rng.uniform(0, 0.5, n), 2))
        put("sr_net_loss", _money(rng, n, 0.5, 1000), _with_nulls(rng, amt, 0.02)[1])
    elif name in ("catalog_sales", "web_sales"):
        pre = "cs" if name == "catalog_sales" else "ws"
        d, dv = date_fk()
        put(f"{pre}_sold_date_sk", d, dv)
        t, tv = fk(86400, 0.02)
        put(f"{pre}_sold_time_sk", t, tv)
        sd, sdv = date_fk()
        put(f"{pre}_ship_date_sk", sd, sdv)
        put(f"{pre}_item_sk", _row_item(lo + np.arange(n, dtype=np.int64), n_item))
        c, cv = fk(n_cust)
        put(f"{pre}_bill_customer_sk", c, cv)
        cd, cdv = fk(n_cdemo)
        put(f"{pre}_bill_cdemo_sk", cd, cdv)
        hd, hdv = fk(n_hdemo)
        put(f"{pre}_bill_hdemo_sk", hd, hdv)
        a, av = fk(n_addr)
        put(f"{pre}_bill_addr_sk", a, av)
        c2, c2v = fk(n_cust)
        put(f"{pre}_ship_customer_sk", c2, c2v)
        a2, a2v = fk(n_addr)
        put(f"{pre}_ship_addr_sk", a2, a2v)
        if pre == "cs":
            cc, ccv = fk(BASE_ROWS["call_center"], 0.02)
            put("cs_call_center_sk", cc, ccv)
            cp, cpv = fk(BASE_ROWS["catalog_page"], 0.02)
            put("cs_catalog_page_sk", cp, cpv)
        else:
            wp, wpv = fk(BASE_ROWS["web_page"], 0.02)
            put("ws_web_page_sk", wp, wpv)
            sh, shv = fk(n_hdemo)
            put("ws_ship_hdemo_sk", sh, shv)
            wsi, wsiv = fk(BASE_ROWS["web_site"], 0.02)
            put("ws_web_site_sk", wsi, wsiv)
        sm, smv = fk(BASE_ROWS["ship_mode"], 0.02)
        put(f"{pre}_ship_mode_sk", sm, smv)
        w, wv = fk(n_wh, 0.02)
        put(f"{pre}_warehouse_sk", w, wv)
        pp, ppv = fk(n_promo, 0.02)
        put(f"{pre}_promo_sk", pp, ppv)
        put(f"{pre}_order_number", (lo + np.arange(n, dtype=np.int64)) // 4 + 1)
        qty = rng.integers(1, 101, n).astype(np.int32)
        put(f"{pre}_quantity", qty, _with_nulls(rng, qty, 0.02)[1])
        whole = _money(rng, n, 1, 100)
        lst = np.round(whole * rng.uniform(1.0, 2.5, n), 2)
        sales = np.round(lst * rng.uniform(0.0, 1.0, n), 2)
        put(f"{pre}_wholesale_cost", whole)
        put(f"{pre}_list_price", lst)
        put(f"{pre}_sales_price", sales, _with_nulls(rng, sales, 0.02)[1])
        ext_sales = np.round(sales * qty, 2)
        ext_whole = np.round(whole * qty, 2)
        ext_list = np.round(lst * qty, 2)
        coupon = np.where(rng.random(n) < 0.1, np.round(ext_sales * rng.uniform(0, 0.5, n), 2), 0.0)
        put(f"{pre}_ext_discount_amt", np.round(ext_list - ext_sales, 2))
        put(f"{pre}_ext_sales_price", ext_sales, _with_nulls(rng, ext_sales, 0.02)[1])
        put(f"{pre}_ext_wholesale_cost", ext_whole)
        put(f"{pre}_ext_list_price", ext_list)
        put(f"{pre}_ext_tax", np.round(ext_sales * 0.09 * rng.random(n), 2))
        put(f"{pre}_coupon_amt", coupon)
        put(f"{pre}_ext_ship_cost", _money(rng, n, 0, 200))
        net_paid = np.round(ext_sales - coupon, 2)
        put(f"{pre}_net_paid", net_paid)
        put(f"{pre}_net_paid_inc_tax", np.round(net_paid * 1.05, 2))
        if pre == "cs":
            put("cs_net_paid_inc_ship", np.round(net_paid + 50, 2))
            put("cs_net_paid_inc_ship_tax", np.round(net_paid * 1.05 + 50, 2))
        put(f"{pre}_net_profit", np.round(net_paid - ext_whole, 2), _with_nulls(rng, net_paid, 0.02)[1])
    elif name in ("catalog_returns", "web_returns"):
        pre = "cr" if name == "catalog_returns" else "wr"
        d, dv = date_fk()
        put(f"{pre}_returned_date_sk", d, dv)
        src_fact = "catalog_sales" if pre == "cr" else "web_sales"
        src_rows = rng.integers(0, max(row_count(src_fact, sf), 1), n)
        put(f"{pre}_item_sk", _row_item(src_rows, n_item))
        put(f"{pre}_order_number", src_rows // 4 + 1)
        c, cv = fk(n_cust)
        put(f"{pre}_returning_customer_sk", c, cv)
        ra, rav = fk(n_addr)
        put(f"{pre}_returning_addr_sk", ra, rav)
        if pre == "wr":
            rc1, rc1v = fk(n_cdemo)
            put("wr_refunded_cdemo_sk", rc1, rc1v)
            rc2, rc2v = fk(n_cdemo)
            put("wr_returning_cdemo_sk", rc2, rc2v)
            ra2, ra2v = fk(n_addr)
            put("wr_refunded_addr_sk", ra2, ra2v)
            rr, rrv = fk(BASE_ROWS["reason"], 0.02)
            put("wr_reason_sk", rr, rrv)
            wwp, wwpv = fk(BASE_ROWS["web_page"], 0.02)
            put("wr_web_page_sk", wwp, wwpv)
        if pre == "cr":
            cp, cpv = fk(BASE_ROWS["catalog_page"], 0.02)
            put("cr_catalog_page_sk", cp, cpv)
            cc, ccv = fk(BASE_ROWS["call_center"], 0.02)
            put("cr_call_center_sk", cc, ccv)
        q = rng.integers(1, 50, n).astype(np.int32)
        put(f"{pre}_return_quantity", q, _with_nulls(rng, q, 0.02)[1])
        amt = _money(rng, n, 1, 2000)
        amt_name = "cr_return_amount" if pre == "cr" else "wr_return_amt"
        put(amt_name, amt, _with_nulls(rng, amt, 0.02)[1])
        if pre == "cr":
            put("cr_return_tax", np.round(amt * 0.09 * rng.random(n), 2))
            put("cr_return_amt_inc_tax", np.round(amt * 1.05, 2))
        else:
            put("wr_fee", _money(rng, n, 0.5, 100))
        put(f"{pre}_net_loss", _money(rng, n, 0.5, 1000))
        put(f"{pre}_refunded_cash", np.round(amt * rng.uniform(0, 1, n), 2))
        put(f"{pre}_reversed_charge", np.round(amt * rng.uniform(0, 0.5, n), 2))
        if pre == "cr":
            put("cr_store_credit", np.round(amt * rng.uniform(0, 0.5, n), 2))
        else:
            put("wr_account_credit", np.round(amt * rng.uniform(0, 0.5, n), 2))
    elif name == "inventory":
        put("inv_date_sk", JULIAN_BASE + SALES_START_IDX + ((sks - 1) % (SALES_DAYS // 7)) * 7)
        put("inv_item_sk", (sks - 1) % n_item + 1)
        put("inv_warehouse_sk", (sks - 1) % n_wh + 1)
        q = rng.integers(0, 1000, n).astype(np.int32)
        put("inv_quantity_on_hand", q, _with_nulls(rng, q, 0.05)[1])
    else:
        raise KeyError(name)

    import pyarrow as pa

    from .. import dtypes as dt

    def decimal_array(values, mask, precision, scale):
        """Exact decimal128 from float money values (2dp): unscaled int
        cents packed into 128-bit little-endian buffers."""
        v = np.asarray(values, dtype=np.float64)
        cents = np.round(v * (10 ** scale)).astype(np.int64)
        n2 = len(cents)
        packed = np.zeros(2 * n2, dtype=np.int64)
        packed[0::2] = cents
        packed[1::2] = np.where(cents < 0, -1, 0)  # sign extension
        if mask is not None:
            validity = np.packbits(~mask, bitorder="little")
            bufs = [pa.py_buffer(validity.tobytes()), pa.py_buffer(packed.tobytes())]
        else:
            bufs = [None, pa.py_buffer(packed.tobytes())]
        return pa.Array.from_buffers(pa.decimal128(precision, scale), n2, bufs)

    schema = SCHEMAS[name]
    arrays = []
    for colname, dtype in schema.items():
        arr = cols[colname]
        mask = None
        if colname in valids:
            mask = ~valids[colname]
        if dtype.code == dt.DECIMAL64:
            arrays.append(decimal_array(arr, mask, dtype.precision, dtype.scale))
            continue
        atype = {
            dt.INT32: pa.int32(), dt.INT64: pa.int64(), dt.FLOAT64: pa.float64(),
            dt.STRING: pa.string(), dt.DATE32: pa.date32(),
        }[dtype.code]
        if isinstance(arr, list):
            arrays.append(pa.array(arr, type=atype))
        else:
            if dtype.code == dt.DATE32:
                arrays.append(pa.array(np.asarray(arr, dtype=np.int32)).cast(pa.date32()))
            else:
                arrays.append(pa.array(np.asarray(arr), type=atype,
                                       mask=mask if mask is not None else None))
    return pa.table(dict(zip(schema.keys(), arrays)))


def _nparts_for(table: str, sf: float) -> int:
    rows = row_count(table, sf)
    if table in ("store_sales", "catalog_sales", "web_sales", "inventory",
                 "store_returns", "catalog_returns", "web_returns"):
        # >=8 parts so an 8-GPU node shards fact scans evenly; large parts
        # keep decode launches few and grids full (8M rows per part)
        return max(8, min(256, (rows + 8_000_000 - 1) // 8_000_000))
    if table == "customer_demographics":
        return max(1, min(16, (rows + 2_000_000 - 1) // 2_000_000))
    return 1


DATAGEN_VERSION = 15


def dataset_root(root: str, sf: float) -> str:
    tag = f"sf{sf:g}-v{DATAGEN_VERSION}"
    return os.path.join(root, tag)


def dataset_paths(root: str, sf: float, table: str) -> List[str]:
    d = os.path.join(dataset_root(root, sf), table)
    return sorted(os.path.join(d, f) for f in os.listdir(d) if f.endswith(".parquet"))


def write_dataset(root: str, sf: float, tables: Optional[List[str]] = None,
                  rank: int = 0, world: int = 1, force: bool = False) -> str:
    """Write the dataset under root/sf{sf}/{table}/part-*.parquet.

    Work is sharded across ranks by (table, part) index; call from every
    rank, then barrier before reading."""
    base = dataset_root(root, sf)
    tables = tables or list(SCHEMAS.keys())
    todo = []
    job = 0
    for t in tables:
        nparts = _nparts_for(t, sf)
        tdir = os.path.join(base, t)
        os.makedirs(tdir, exist_ok=True)
        for p in range(nparts):
            path = os.path.join(tdir, f"part-{p:04d}.parquet")
            if job % world == rank and (force or not os.path.exists(path)):
                todo.append((t, sf, p, nparts, path))
            job += 1
    # cap default workers: each in-flight (table, part) job materializes
    # a multi-GB arrow table, so 256-core boxes must not run 254 at once
    workers = int(os.environ.get("AURON_DATAGEN_WORKERS",
                                 str(max(1, min(32, (os.cpu_count() or 4) - 2)))))
    if len(todo) > 3 and workers > 1:
        # dsdgen-style parallel generation: (table, part) jobs fan out
        # across processes (numpy-only work, no device state involved)
        from concurrent.futures import ProcessPoolExecutor

        with ProcessPoolExecutor(max_workers=min(workers, len(todo))) as ex:
            list(ex.map(_write_one, todo, chunksize=1))
    else:
        for t in todo:
            _write_one(t)
    return base


def _write_one(args):
    import pyarrow.parquet as pq

    t, sf, p, nparts, path = args
    tbl = generate_table(t, sf, p, nparts)
    # uncompressed pages: the GPU parquet decoder (parquet_native.py +
    # csrc/parquet.hip) consumes page bytes directly in HBM — numerics as
    # PLAIN, strings as RLE_DICTIONARY (indices decoded + bytes gathered
    # on device); no host decode at all
    str_cols = [f.name for f in tbl.schema
                if f.type == __import__("pyarrow").string()]
    pq.write_table(tbl, path, compression="NONE",
                   use_dictionary=str_cols,
                   data_page_version="1.0",
                   store_decimal_as_integer=True,
                   dictionary_pagesize_limit=1 << 26,
                   data_page_size=64 << 10,
                   row_group_size=1 << 20)
