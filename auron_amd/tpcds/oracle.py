"""pandas oracle implementations of the TPC-DS queries (ground truth for
result comparison — the QueryResultComparator role from the reference's
dev/auron-it harness, with double tolerance)."""
from __future__ import annotations

import numpy as np
import pandas as pd

from . import datagen


def _read(root, sf, table, columns=None):
    import pyarrow.parquet as pq

    paths = datagen.dataset_paths(root, sf, table)
    import pyarrow as pa

    t = pa.concat_tables([pq.read_table(p, columns=columns) for p in paths])
    return t.to_pandas()


def _merge(l, r, lk, rk, how="inner"):
    """SQL-semantics merge: NULL keys never match."""
    lv = l.dropna(subset=[lk]) if how in ("inner",) else l
    rv = r.dropna(subset=[rk])
    if how == "inner":
        return lv.dropna(subset=[lk]).merge(rv, left_on=lk, right_on=rk, how="inner")
    raise NotImplementedError(how)


def q1(root, sf):
    sr = _read(root, sf, "store_returns", ["sr_returned_date_sk", "sr_customer_sk",
                                           "sr_store_sk", "sr_return_amt"])
    dd = _read(root, sf, "date_dim", ["d_date_sk", "d_year"])
    dd = dd[dd.d_year == 2000]
    j = _merge(sr, dd, "sr_returned_date_sk", "d_date_sk")
    ctr = (j.groupby(["sr_customer_sk", "sr_store_sk"], dropna=False)
            .sr_return_amt.sum(min_count=1).reset_index())
    ctr.columns = ["ctr_customer_sk", "ctr_store_sk", "ctr_total_return"]
    av = ctr.groupby("ctr_store_sk", dropna=False).ctr_total_return.mean().reset_index()
    av.columns = ["av_store_sk", "av"]
    j2 = _merge(ctr, av, "ctr_store_sk", "av_store_sk")
    f = j2[j2.ctr_total_return > j2.av * 1.2]
    st = _read(root, sf, "store", ["s_store_sk", "s_state"])
    st = st[st.s_state == "TN"]
    j3 = _merge(f, st, "ctr_store_sk", "s_store_sk")
    cust = _read(root, sf, "customer", ["c_customer_sk", "c_customer_id"])
    j4 = _merge(j3, cust, "ctr_customer_sk", "c_customer_sk")
    out = j4[["c_customer_id"]].sort_values("c_customer_id").head(100)
    return out.reset_index(drop=True)


def _star_q(root, sf, dd_filter, it_filter, group_cols, measure="ss_ext_sales_price",
            agg_name="sum_agg"):
    ss = _read(root, sf, "store_sales", ["ss_sold_date_sk", "ss_item_sk", measure])
    dd = _read(root, sf, "date_dim", ["d_date_sk", "d_year", "d_moy"])
    dd = dd[dd_filter(dd)]
    it = _read(root, sf, "item")
    it = it[it_filter(it)]
    j = _merge(ss, dd, "ss_sold_date_sk", "d_date_sk")
    j = _merge(j, it, "ss_item_sk", "i_item_sk")
    g = j.groupby(group_cols, dropna=False)[measure].sum(min_count=1).reset_index()
    g.columns = group_cols + [agg_name]
    return g


def q3(root, sf):
    g = _star_q(root, sf, lambda d: d.d_moy == 11,
                lambda i: i.i_manufact_id == 128,
                ["d_year", "i_brand_id", "i_brand"])
    g = g.sort_values(["d_year", "sum_agg", "i_brand_id"],
                      ascending=[True, False, True]).head(100)
    return g.reset_index(drop=True)


def q6(root, sf):
    dd_all = _read(root, sf, "date_dim", ["d_date_sk", "d_month_seq", "d_year", "d_moy"])
    ms = dd_all[(dd_all.d_year == 2001) & (dd_all.d_moy == 1)].d_month_seq.iloc[0]
    it = _read(root, sf, "item", ["i_item_sk", "i_category", "i_current_price"])
    cat_avg = it.groupby("i_category", dropna=False).i_current_price.mean().reset_index()
    cat_avg.columns = ["i_category", "cat_avg_price"]
    itj = _merge(it, cat_avg, "i_category", "i_category")
    itj = itj.rename(columns={"i_category_x": "i_category"}) if "i_category_x" in itj else itj
    it_f = itj[itj.i_current_price > itj.cat_avg_price * 1.2]
    ss = _read(root, sf, "store_sales", ["ss_sold_date_sk", "ss_customer_sk", "ss_item_sk"])
    dd = dd_all[dd_all.d_month_seq == ms]
    j = _merge(ss, dd, "ss_sold_date_sk", "d_date_sk")
    j = _merge(j, it_f, "ss_item_sk", "i_item_sk")
    cust = _read(root, sf, "customer", ["c_customer_sk", "c_current_addr_sk"])
    j = _merge(j, cust, "ss_customer_sk", "c_customer_sk")
    ca = _read(root, sf, "customer_address", ["ca_address_sk", "ca_state"])
    j = _merge(j, ca, "c_current_addr_sk", "ca_address_sk")
    g = j.groupby("ca_state", dropna=False).size().reset_index(name="cnt")
    g = g[g.cnt >= 10]
    g.columns = ["state", "cnt"]
    g = g.sort_values(["cnt", "state"], na_position="first").head(100)
    return g.reset_index(drop=True)


def q7(root, sf):
    ss = _read(root, sf, "store_sales", ["ss_sold_date_sk", "ss_item_sk", "ss_cdemo_sk",
                                         "ss_promo_sk", "ss_quantity", "ss_list_price",
                                         "ss_coupon_amt", "ss_sales_price"])
    cd = _read(root, sf, "customer_demographics")
    cd = cd[(cd.cd_gender == "M") & (cd.cd_marital_status == "S")
            & (cd.cd_education_status == "College")]
    dd = _read(root, sf, "date_dim", ["d_date_sk", "d_year"])
    dd = dd[dd.d_year == 2000]
    pr = _read(root, sf, "promotion")
    pr = pr[(pr.p_channel_email == "N") | (pr.p_channel_event == "N")]
    it = _read(root, sf, "item", ["i_item_sk", "i_item_id"])
    j = _merge(ss, cd, "ss_cdemo_sk", "cd_demo_sk")
    j = _merge(j, dd, "ss_sold_date_sk", "d_date_sk")
    j = _merge(j, pr, "ss_promo_sk", "p_promo_sk")
    j = _merge(j, it, "ss_item_sk", "i_item_sk")
    g = j.groupby("i_item_id", dropna=False).agg(
        agg1=("ss_quantity", "mean"), agg2=("ss_list_price", "mean"),
        agg3=("ss_coupon_amt", "mean"), agg4=("ss_sales_price", "mean")).reset_index()
    return g.sort_values("i_item_id").head(100).reset_index(drop=True)


def q42(root, sf):
    g = _star_q(root, sf, lambda d: (d.d_moy == 11) & (d.d_year == 2000),
                lambda i: i.i_manager_id == 1,
                ["d_year", "i_category_id", "i_category"], agg_name="total_sales")
    g = g.sort_values(["total_sales", "d_year", "i_category_id", "i_category"],
                      ascending=[False, True, True, True]).head(100)
    return g.reset_index(drop=True)


def q52(root, sf):
    g = _star_q(root, sf, lambda d: (d.d_moy == 11) & (d.d_year == 2000),
                lambda i: i.i_manager_id == 1,
                ["d_year", "i_brand_id", "i_brand"], agg_name="ext_price")
    g = g.sort_values(["d_year", "ext_price", "i_brand_id"],
                      ascending=[True, False, True]).head(100)
    return g.reset_index(drop=True)


def q55(root, sf):
    g = _star_q(root, sf, lambda d: (d.d_moy == 11) & (d.d_year == 1999),
                lambda i: i.i_manager_id == 28,
                ["i_brand_id", "i_brand"], agg_name="ext_price")
    g = g.sort_values(["ext_price", "i_brand_id"], ascending=[False, True]).head(100)
    return g.reset_index(drop=True)


def q19(root, sf):
    ss = _read(root, sf, "store_sales", ["ss_sold_date_sk", "ss_item_sk", "ss_customer_sk",
                                         "ss_store_sk", "ss_ext_sales_price"])
    dd = _read(root, sf, "date_dim", ["d_date_sk", "d_year", "d_moy"])
    dd = dd[(dd.d_moy == 11) & (dd.d_year == 1998)]
    it = _read(root, sf, "item")
    it = it[it.i_manager_id == 8]
    cust = _read(root, sf, "customer", ["c_customer_sk", "c_current_addr_sk"])
    ca = _read(root, sf, "customer_address", ["ca_address_sk", "ca_zip"])
    st = _read(root, sf, "store", ["s_store_sk", "s_zip"])
    j = _merge(ss, dd, "ss_sold_date_sk", "d_date_sk")
    j = _merge(j, it, "ss_item_sk", "i_item_sk")
    j = _merge(j, cust, "ss_customer_sk", "c_customer_sk")
    j = _merge(j, ca, "c_current_addr_sk", "ca_address_sk")
    j = _merge(j, st, "ss_store_sk", "s_store_sk")
    j = j[j.ca_zip.str[:5] != j.s_zip.str[:5]]
    g = j.groupby(["i_brand", "i_brand_id", "i_manufact_id"], dropna=False) \
         .ss_ext_sales_price.sum(min_count=1).reset_index()
    g.columns = ["i_brand", "i_brand_id", "i_manufact_id", "ext_price"]
    g = g.sort_values(["ext_price", "i_brand", "i_brand_id", "i_manufact_id"],
                      ascending=[False, True, True, True]).head(100)
    return g.reset_index(drop=True)


def q96(root, sf):
    ss = _read(root, sf, "store_sales", ["ss_sold_time_sk", "ss_hdemo_sk", "ss_store_sk"])
    hd = _read(root, sf, "household_demographics", ["hd_demo_sk", "hd_dep_count"])
    hd = hd[hd.hd_dep_count == 7]
    td = _read(root, sf, "time_dim")
    td = td[(td.t_hour == 20) & (td.t_minute >= 30)]
    st = _read(root, sf, "store", ["s_store_sk", "s_store_name"])
    st = st[st.s_store_name == "ese"]
    j = _merge(ss, hd, "ss_hdemo_sk", "hd_demo_sk")
    j = _merge(j, td, "ss_sold_time_sk", "t_time_sk")
    j = _merge(j, st, "ss_store_sk", "s_store_sk")
    return pd.DataFrame({"cnt": [len(j)]})


def q68(root, sf):
    ss = _read(root, sf, "store_sales", ["ss_sold_date_sk", "ss_store_sk", "ss_hdemo_sk",
                                         "ss_addr_sk", "ss_customer_sk", "ss_ticket_number",
                                         "ss_ext_sales_price", "ss_ext_list_price", "ss_ext_tax"])
    dd = _read(root, sf, "date_dim", ["d_date_sk", "d_dom", "d_year"])
    dd = dd[(dd.d_dom >= 1) & (dd.d_dom <= 2) & dd.d_year.isin([1999, 2000, 2001])]
    st = _read(root, sf, "store", ["s_store_sk", "s_city"])
    st = st[st.s_city.isin(["Midway", "Fairview"])]
    hd = _read(root, sf, "household_demographics")
    hd = hd[(hd.hd_dep_count == 4) | (hd.hd_vehicle_count == 3)]
    ca = _read(root, sf, "customer_address", ["ca_address_sk", "ca_city"])
    j = _merge(ss, dd, "ss_sold_date_sk", "d_date_sk")
    j = _merge(j, st, "ss_store_sk", "s_store_sk")
    j = _merge(j, hd, "ss_hdemo_sk", "hd_demo_sk")
    j = _merge(j, ca, "ss_addr_sk", "ca_address_sk")
    g = j.groupby(["ss_ticket_number", "ss_customer_sk", "ca_city"], dropna=False).agg(
        extended_price=("ss_ext_sales_price", lambda x: x.sum(min_count=1)),
        list_price=("ss_ext_list_price", lambda x: x.sum(min_count=1)),
        extended_tax=("ss_ext_tax", lambda x: x.sum(min_count=1))).reset_index()
    g = g.rename(columns={"ca_city": "bought_city"})
    cust = _read(root, sf, "customer", ["c_customer_sk", "c_current_addr_sk",
                                        "c_first_name", "c_last_name"])
    j2 = _merge(g, cust, "ss_customer_sk", "c_customer_sk")
    ca2 = ca.rename(columns={"ca_address_sk": "current_addr_sk", "ca_city": "current_city"})
    j3 = _merge(j2, ca2, "c_current_addr_sk", "current_addr_sk")
    f = j3[j3.current_city != j3.bought_city].copy()
    f = f[~(f.current_city.isna() | f.bought_city.isna())]
    out = f[["c_last_name", "c_first_name", "bought_city", "ss_ticket_number",
             "extended_price", "extended_tax", "list_price"]]
    out = out.sort_values(["c_last_name", "ss_ticket_number"]).head(100)
    return out.reset_index(drop=True)


ORACLES = {"q1": q1, "q3": q3, "q6": q6, "q7": q7, "q19": q19, "q42": q42,
           "q52": q52, "q55": q55, "q68": q68, "q96": q96}
