"""TPC-DS table schemas (column subset used by the implemented queries).

Role parity: the TPC-DS integration harness of the reference
(/root/reference/dev/auron-it, tpcds-queries/q*.sql). Money columns are
float64 in round 1 (decimal64 kernels are tracked in ARCHITECTURE.md).
"""
from __future__ import annotations

from .. import dtypes as dt

f64 = dt.float64
d72 = dt.decimal64(7, 2)
i32 = dt.int32
i64 = dt.int64
s = dt.string
d32 = dt.date32

SCHEMAS = {
    "date_dim": {
        "d_date_sk": i64, "d_date": d32, "d_year": i32, "d_moy": i32,
        "d_dom": i32, "d_qoy": i32, "d_day_name": s, "d_month_seq": i32,
        "d_week_seq": i32, "d_dow": i32, "d_quarter_name": s,
    },
    "time_dim": {"t_time_sk": i64, "t_time": i64, "t_hour": i32, "t_minute": i32, "t_meal_time": s},
    "item": {
        "i_item_sk": i64, "i_item_id": s, "i_item_desc": s, "i_category": s,
        "i_category_id": i32, "i_brand": s, "i_brand_id": i32,
        "i_manufact_id": i32, "i_class": s, "i_class_id": i32,
        "i_current_price": d72, "i_wholesale_cost": d72, "i_manager_id": i32, "i_product_name": s, "i_color": s, "i_units": s, "i_size": s, "i_manufact": s,
    },
    "customer": {
        "c_customer_sk": i64, "c_customer_id": s, "c_first_name": s,
        "c_last_name": s, "c_current_addr_sk": i64, "c_current_cdemo_sk": i64,
        "c_current_hdemo_sk": i64, "c_birth_country": s, "c_birth_year": i32, "c_birth_day": i32,
        "c_birth_month": i32, "c_login": s, "c_last_review_date_sk": i64, "c_email_address": s, "c_first_sales_date_sk": i64,
        "c_first_shipto_date_sk": i64, "c_preferred_cust_flag": s, "c_salutation": s,
    },
    "customer_address": {
        "ca_address_sk": i64, "ca_state": s, "ca_zip": s, "ca_country": s,
        "ca_city": s, "ca_county": s, "ca_gmt_offset": f64, "ca_street_name": s, "ca_street_type": s,
        "ca_location_type": s, "ca_suite_number": s, "ca_street_number": s,
    },
    "customer_demographics": {
        "cd_demo_sk": i64, "cd_gender": s, "cd_marital_status": s,
        "cd_education_status": s, "cd_purchase_estimate": i32, "cd_credit_rating": s,
        "cd_dep_count": i32, "cd_dep_employed_count": i32, "cd_dep_college_count": i32,
    },
    "household_demographics": {
        "hd_demo_sk": i64, "hd_income_band_sk": i64, "hd_buy_potential": s,
        "hd_dep_count": i32, "hd_vehicle_count": i32,
    },
    "income_band": {"ib_income_band_sk": i64, "ib_lower_bound": i32, "ib_upper_bound": i32},
    "store": {
        "s_store_sk": i64, "s_store_id": s, "s_store_name": s, "s_state": s,
        "s_county": s, "s_zip": s, "s_city": s, "s_number_employees": i32,
        "s_gmt_offset": f64, "s_company_id": i32, "s_company_name": s, "s_street_name": s,
        "s_street_number": s, "s_street_type": s, "s_suite_number": s, "s_market_id": i32,
    },
    "warehouse": {
        "w_warehouse_sk": i64, "w_warehouse_name": s, "w_warehouse_sq_ft": i32,
        "w_state": s, "w_county": s, "w_city": s, "w_country": s,
    },
    "promotion": {
        "p_promo_sk": i64, "p_promo_id": s, "p_channel_email": s,
        "p_channel_event": s, "p_channel_dmail": s, "p_channel_tv": s,
    },
    "reason": {"r_reason_sk": i64, "r_reason_desc": s},
    "ship_mode": {"sm_ship_mode_sk": i64, "sm_type": s, "sm_carrier": s},
    "call_center": {
        "cc_call_center_sk": i64, "cc_call_center_id": s, "cc_county": s,
        "cc_name": s, "cc_manager": s,
    },
    "web_site": {"web_site_sk": i64, "web_site_id": s, "web_name": s, "web_company_name": s},
    "web_page": {"wp_web_page_sk": i64, "wp_char_count": i32},
    "catalog_page": {"cp_catalog_page_sk": i64, "cp_catalog_page_id": s},
    "store_sales": {
        "ss_sold_date_sk": i64, "ss_sold_time_sk": i64, "ss_item_sk": i64,
        "ss_customer_sk": i64, "ss_cdemo_sk": i64, "ss_hdemo_sk": i64,
        "ss_addr_sk": i64, "ss_store_sk": i64, "ss_promo_sk": i64,
        "ss_ticket_number": i64, "ss_quantity": i32, "ss_wholesale_cost": d72,
        "ss_list_price": d72, "ss_sales_price": d72, "ss_ext_discount_amt": d72,
        "ss_ext_sales_price": d72, "ss_ext_wholesale_cost": d72,
        "ss_ext_list_price": d72, "ss_ext_tax": d72, "ss_coupon_amt": d72,
        "ss_net_paid": d72, "ss_net_paid_inc_tax": d72, "ss_net_profit": d72,
    },
    "store_returns": {
        "sr_returned_date_sk": i64, "sr_return_time_sk": i64, "sr_item_sk": i64,
        "sr_customer_sk": i64, "sr_cdemo_sk": i64, "sr_hdemo_sk": i64,
        "sr_addr_sk": i64, "sr_store_sk": i64, "sr_reason_sk": i64,
        "sr_ticket_number": i64, "sr_return_quantity": i32, "sr_return_amt": d72,
        "sr_return_tax": d72, "sr_return_amt_inc_tax": d72, "sr_fee": d72,
        "sr_return_ship_cost": d72, "sr_refunded_cash": d72,
        "sr_reversed_charge": d72, "sr_store_credit": d72, "sr_net_loss": d72,
    },
    "catalog_sales": {
        "cs_sold_date_sk": i64, "cs_sold_time_sk": i64, "cs_ship_date_sk": i64,
        "cs_bill_customer_sk": i64, "cs_bill_cdemo_sk": i64, "cs_bill_hdemo_sk": i64,
        "cs_bill_addr_sk": i64, "cs_ship_customer_sk": i64, "cs_ship_addr_sk": i64,
        "cs_call_center_sk": i64, "cs_catalog_page_sk": i64, "cs_ship_mode_sk": i64,
        "cs_warehouse_sk": i64, "cs_item_sk": i64, "cs_promo_sk": i64,
        "cs_order_number": i64, "cs_quantity": i32, "cs_wholesale_cost": d72,
        "cs_list_price": d72, "cs_sales_price": d72, "cs_ext_discount_amt": d72,
        "cs_ext_sales_price": d72, "cs_ext_wholesale_cost": d72,
        "cs_ext_list_price": d72, "cs_ext_tax": d72, "cs_coupon_amt": d72,
        "cs_ext_ship_cost": d72, "cs_net_paid": d72, "cs_net_paid_inc_tax": d72,
        "cs_net_paid_inc_ship": f64, "cs_net_paid_inc_ship_tax": d72, "cs_net_profit": d72,
    },
    "catalog_returns": {
        "cr_returned_date_sk": i64, "cr_item_sk": i64, "cr_order_number": i64,
        "cr_returning_customer_sk": i64, "cr_returning_addr_sk": i64, "cr_catalog_page_sk": i64,
        "cr_return_quantity": i32, "cr_return_amount": d72, "cr_return_tax": d72, "cr_return_amt_inc_tax": d72,
        "cr_net_loss": d72, "cr_refunded_cash": d72, "cr_reversed_charge": d72,
        "cr_store_credit": d72, "cr_call_center_sk": i64,
    },
    "web_sales": {
        "ws_sold_date_sk": i64, "ws_sold_time_sk": i64, "ws_ship_date_sk": i64,
        "ws_item_sk": i64, "ws_bill_customer_sk": i64, "ws_bill_cdemo_sk": i64,
        "ws_bill_hdemo_sk": i64, "ws_bill_addr_sk": i64, "ws_ship_customer_sk": i64, "ws_ship_addr_sk": i64,
        "ws_web_page_sk": i64, "ws_web_site_sk": i64, "ws_ship_mode_sk": i64, "ws_ship_hdemo_sk": i64,
        "ws_warehouse_sk": i64, "ws_promo_sk": i64, "ws_order_number": i64,
        "ws_quantity": i32, "ws_wholesale_cost": d72, "ws_list_price": d72,
        "ws_sales_price": d72, "ws_ext_discount_amt": d72, "ws_ext_sales_price": d72,
        "ws_ext_wholesale_cost": d72, "ws_ext_list_price": d72, "ws_ext_tax": d72,
        "ws_coupon_amt": d72, "ws_ext_ship_cost": d72, "ws_net_paid": d72,
        "ws_net_paid_inc_tax": d72, "ws_net_profit": d72,
    },
    "web_returns": {
        "wr_returned_date_sk": i64, "wr_item_sk": i64, "wr_order_number": i64,
        "wr_returning_customer_sk": i64, "wr_returning_addr_sk": i64, "wr_refunded_cdemo_sk": i64, "wr_returning_cdemo_sk": i64, "wr_refunded_addr_sk": i64, "wr_reason_sk": i64, "wr_web_page_sk": i64, "wr_return_quantity": i32,
        "wr_return_amt": d72, "wr_net_loss": d72, "wr_fee": d72,
        "wr_refunded_cash": d72, "wr_reversed_charge": d72, "wr_account_credit": d72,
    },
    "inventory": {
        "inv_date_sk": i64, "inv_item_sk": i64, "inv_warehouse_sk": i64,
        "inv_quantity_on_hand": i32,
    },
}


def table_schema(name: str):
    return SCHEMAS[name]


# dsdgen row counts at SF=1 (facts scale linearly with SF, dims per spec)
BASE_ROWS = {
    "date_dim": 73049, "time_dim": 86400, "item": 18000, "customer": 100000,
    "customer_address": 50000, "customer_demographics": 1920800,
    "household_demographics": 7200, "income_band": 20, "store": 12,
    "warehouse": 5, "promotion": 300, "reason": 35, "ship_mode": 20,
    "call_center": 6, "web_site": 30, "web_page": 60, "catalog_page": 11718,
    "store_sales": 2880404, "store_returns": 287514, "catalog_sales": 1441548,
    "catalog_returns": 144067, "web_sales": 719384, "web_returns": 71763,
    "inventory": 11745000,
}

FACT_TABLES = {"store_sales", "store_returns", "catalog_sales", "catalog_returns",
               "web_sales", "web_returns", "inventory"}
# dims that grow sublinearly with SF (simplified: sqrt-ish growth via lookup)
SCALED_DIMS = {"item": 2.0, "customer": 3.0, "customer_address": 3.0}


_DIM_MIN = {"item": 900, "customer": 5000, "customer_address": 2500,
            "customer_demographics": 48020}


def row_count(table: str, sf: float) -> int:
    base = BASE_ROWS[table]
    if table in FACT_TABLES:
        return max(int(base * sf), 1000)
    if sf < 1 and table in _DIM_MIN:
        return max(int(base * sf), _DIM_MIN[table])
    if table in SCALED_DIMS and sf > 1:
        return int(base * min(sf ** 0.5 * SCALED_DIMS[table] / 2, sf))
    return base
