"""Synthetic TPC-DS-shaped tables (star schema subset) for query tests.

Mirrors the column subset the test queries touch; scale parameter sizes
store_sales like tpch_data.py sizes lineitem.  (Reference role: the
TPC-DS derived benchmarks the reference runs, e.g. BodoSQL TPC-DS suite.)
"""

from __future__ import annotations

import numpy as np
import pandas as pd


def gen_tpcds(scale: float = 0.02, seed: int = 0):
    rng = np.random.default_rng(seed)
    n_item = max(40, int(2000 * scale))
    n_cust = max(50, int(5000 * scale))
    n_store = max(4, int(12 * scale) or 4)
    n_dates = 365 * 3
    n_ss = max(1000, int(600_000 * scale))

    date_dim = pd.DataFrame({
        "d_date_sk": np.arange(n_dates, dtype=np.int64),
        "d_date": pd.date_range("1999-01-01", periods=n_dates, freq="D"),
    })
    date_dim["d_year"] = date_dim["d_date"].dt.year
    date_dim["d_moy"] = date_dim["d_date"].dt.month
    date_dim["d_qoy"] = date_dim["d_date"].dt.quarter

    item = pd.DataFrame({
        "i_item_sk": np.arange(n_item, dtype=np.int64),
        "i_item_id": [f"ITEM{i:08d}" for i in range(n_item)],
        "i_brand_id": rng.integers(1, 30, n_item).astype(np.int64),
        "i_brand": rng.choice([f"brand#{b}" for b in range(1, 30)], n_item),
        "i_category": rng.choice(
            ["Books", "Electronics", "Home", "Music", "Sports"], n_item),
        "i_manager_id": rng.integers(1, 100, n_item).astype(np.int64),
        "i_current_price": np.round(rng.uniform(0.5, 300.0, n_item), 2),
    })

    customer = pd.DataFrame({
        "c_customer_sk": np.arange(n_cust, dtype=np.int64),
        "c_customer_id": [f"CUST{i:08d}" for i in range(n_cust)],
        "c_birth_country": rng.choice(
            ["UNITED STATES", "CANADA", "MEXICO", "GERMANY"], n_cust),
    })

    store = pd.DataFrame({
        "s_store_sk": np.arange(n_store, dtype=np.int64),
        "s_store_name": [f"store_{i}" for i in range(n_store)],
        "s_state": rng.choice(["TN", "CA", "TX", "WA"], n_store),
    })

    store_sales = pd.DataFrame({
        "ss_sold_date_sk": rng.integers(0, n_dates, n_ss).astype(np.int64),
        "ss_item_sk": rng.integers(0, n_item, n_ss).astype(np.int64),
        "ss_customer_sk": rng.integers(0, n_cust, n_ss).astype(np.int64),
        "ss_store_sk": rng.integers(0, n_store, n_ss).astype(np.int64),
        "ss_quantity": rng.integers(1, 100, n_ss).astype(np.int64),
        "ss_sales_price": np.round(rng.uniform(0.5, 200.0, n_ss), 2),
        "ss_ext_sales_price": np.round(rng.uniform(1.0, 2000.0, n_ss), 2),
        "ss_net_profit": np.round(rng.uniform(-500.0, 500.0, n_ss), 2),
    })

    return {"date_dim": date_dim, "item": item, "customer": customer,
            "store": store, "store_sales": store_sales}
