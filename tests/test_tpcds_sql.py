"""TPC-DS-shaped SQL queries over the engine vs pandas oracles
(reference role: the BodoSQL TPC-DS suite; star joins + CTEs + windows)."""

import os
import sys

import numpy as np
import pandas as pd
import pytest

sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "benchmarks"))

from tpcds_data import gen_tpcds  # noqa: E402

from bodo_amd.sql import BodoSQLContext  # noqa: E402


@pytest.fixture(scope="module")
def ctx():
    t = gen_tpcds(0.02)
    return BodoSQLContext(t), t


def test_tpcds_q3_shape(ctx):
    """Q3: year/brand revenue for one manufacturer-ish filter."""
    bc, t = ctx
    got = bc.sql("""
        select d.d_year, i.i_brand_id, i.i_brand,
               sum(ss.ss_ext_sales_price) as sum_agg
        from store_sales ss
        join date_dim d on ss.ss_sold_date_sk = d.d_date_sk
        join item i on ss.ss_item_sk = i.i_item_sk
        where i.i_manager_id < 20 and d.d_moy = 12
        group by d.d_year, i.i_brand_id, i.i_brand
        order by d.d_year, sum_agg desc, i.i_brand_id
        limit 100
    """).to_pandas()
    ss, dd, it = t["store_sales"], t["date_dim"], t["item"]
    j = ss.merge(dd, left_on="ss_sold_date_sk", right_on="d_date_sk") \
          .merge(it, left_on="ss_item_sk", right_on="i_item_sk")
    f = j[(j.i_manager_id < 20) & (j.d_moy == 12)]
    exp = f.groupby(["d_year", "i_brand_id", "i_brand"], as_index=False) \
        .agg(sum_agg=("ss_ext_sales_price", "sum")) \
        .sort_values(["d_year", "sum_agg", "i_brand_id"],
                     ascending=[True, False, True]).head(100)
    assert len(got) == len(exp)
    np.testing.assert_allclose(
        sorted(got["sum_agg"]), sorted(exp["sum_agg"]), rtol=1e-9)


def test_tpcds_q42_shape(ctx):
    """Q42: category revenue for a (year, month)."""
    bc, t = ctx
    got = bc.sql("""
        select d.d_year, i.i_category, sum(ss.ss_ext_sales_price) as rev
        from store_sales ss
        join date_dim d on ss.ss_sold_date_sk = d.d_date_sk
        join item i on ss.ss_item_sk = i.i_item_sk
        where d.d_year = 2000 and d.d_moy = 11
        group by d.d_year, i.i_category
        order by rev desc
    """).to_pandas()
    ss, dd, it = t["store_sales"], t["date_dim"], t["item"]
    j = ss.merge(dd, left_on="ss_sold_date_sk", right_on="d_date_sk") \
          .merge(it, left_on="ss_item_sk", right_on="i_item_sk")
    f = j[(j.d_year == 2000) & (j.d_moy == 11)]
    exp = f.groupby("i_category")["ss_ext_sales_price"].sum() \
        .sort_values(ascending=False)
    assert len(got) == len(exp)
    np.testing.assert_allclose(got["rev"].to_numpy(), exp.to_numpy(),
                               rtol=1e-9)


def test_tpcds_q55_brand_revenue(ctx):
    bc, t = ctx
    got = bc.sql("""
        select i.i_brand_id as brand_id, i.i_brand as brand,
               sum(ss.ss_ext_sales_price) as ext_price
        from store_sales ss
        join date_dim d on ss.ss_sold_date_sk = d.d_date_sk
        join item i on ss.ss_item_sk = i.i_item_sk
        where i.i_manager_id = 28 and d.d_moy = 11 and d.d_year = 1999
        group by i.i_brand_id, i.i_brand
        order by ext_price desc, brand_id
    """).to_pandas()
    ss, dd, it = t["store_sales"], t["date_dim"], t["item"]
    j = ss.merge(dd, left_on="ss_sold_date_sk", right_on="d_date_sk") \
          .merge(it, left_on="ss_item_sk", right_on="i_item_sk")
    f = j[(j.i_manager_id == 28) & (j.d_moy == 11) & (j.d_year == 1999)]
    exp = f.groupby(["i_brand_id", "i_brand"], as_index=False) \
        .agg(ext_price=("ss_ext_sales_price", "sum"))
    assert len(got) == len(exp)
    np.testing.assert_allclose(sorted(got["ext_price"]),
                               sorted(exp["ext_price"]), rtol=1e-9)


def test_tpcds_cte_store_ranking(ctx):
    """CTE + window: per-state store revenue ranking (q70 shape)."""
    bc, t = ctx
    got = bc.sql("""
        with sr as (
            select s.s_state, s.s_store_name,
                   sum(ss.ss_net_profit) as profit
            from store_sales ss
            join store s on ss.ss_store_sk = s.s_store_sk
            group by s.s_state, s.s_store_name
        )
        select s_state, s_store_name, profit
        from sr
        qualify rank() over (partition by s_state
                             order by profit desc) <= 2
        order by s_state, profit desc
    """).to_pandas()
    ss, st = t["store_sales"], t["store"]
    j = ss.merge(st, left_on="ss_store_sk", right_on="s_store_sk")
    sr = j.groupby(["s_state", "s_store_name"], as_index=False) \
        .agg(profit=("ss_net_profit", "sum"))
    sr["rk"] = sr.groupby("s_state")["profit"].rank(method="min",
                                                    ascending=False)
    exp = sr[sr.rk <= 2].sort_values(["s_state", "profit"],
                                     ascending=[True, False])
    assert len(got) == len(exp)
    np.testing.assert_allclose(got["profit"].to_numpy(),
                               exp["profit"].to_numpy(), rtol=1e-9)


def test_tpcds_q7_customer_avg(ctx):
    """Aggregate over join with multiple avg aggs (q7 shape)."""
    bc, t = ctx
    got = bc.sql("""
        select i.i_item_id, avg(ss.ss_quantity) as agg1,
               avg(ss.ss_sales_price) as agg2,
               count(*) as cnt
        from store_sales ss
        join item i on ss.ss_item_sk = i.i_item_sk
        where i.i_category = 'Books'
        group by i.i_item_id
        order by i.i_item_id
        limit 50
    """).to_pandas()
    ss, it = t["store_sales"], t["item"]
    j = ss.merge(it, left_on="ss_item_sk", right_on="i_item_sk")
    f = j[j.i_category == "Books"]
    exp = f.groupby("i_item_id", as_index=False).agg(
        agg1=("ss_quantity", "mean"), agg2=("ss_sales_price", "mean"),
        cnt=("ss_quantity", "size")).sort_values("i_item_id").head(50)
    assert len(got) == len(exp)
    np.testing.assert_allclose(got["agg1"].to_numpy(),
                               exp["agg1"].to_numpy(), rtol=1e-9)
    assert got["cnt"].tolist() == exp["cnt"].tolist()
