"""Standard TPC-H SQL texts through BodoSQLContext, differential against the
pandas-form query implementations (reference analog: BodoSQL test suites)."""

import os
import sys

import numpy as np
import pandas as pd
import pytest

sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "benchmarks"))

from bodo_amd.sql import BodoSQLContext  # noqa: E402
from tpch_data import gen_all  # noqa: E402
from tpch_queries import ALL as PANDAS_Q  # noqa: E402
from tpch_sql import Q as SQL_Q  # noqa: E402
from tests.test_tpch import _decat_df, _norm  # noqa: E402

SF = 0.02

# SQL output column names differ from the pandas-form implementations; for
# comparison we align by POSITION after normalizing + sorting all columns.
SUPPORTED = [n for n in range(1, 23) if SQL_Q.get(n)]


@pytest.fixture(scope="module")
def ctx_tables():
    t = gen_all(SF)
    bc = BodoSQLContext({k: v for k, v in t.items()})
    return bc, t


@pytest.mark.parametrize("qnum", SUPPORTED)
def test_tpch_sql(ctx_tables, qnum):
    bc, t = ctx_tables
    got = bc.sql(SQL_Q[qnum]).to_pandas()
    exp = PANDAS_Q[qnum](pd, {k: _decat_df(v) for k, v in t.items()})
    got = _norm(got)
    exp = _norm(exp)
    # align by position: same column count expected for the shared queries
    if qnum in (2,):
        # q2 SQL selects a different column order than the pandas form
        exp = exp[["S_ACCTBAL", "S_NAME", "N_NAME", "P_PARTKEY", "P_MFGR",
                   "S_ADDRESS", "S_PHONE", "S_COMMENT"]]
    if qnum == 8:
        # pandas form rounds mkt_share to 2 digits; SQL text does not
        got["mkt_share"] = got["mkt_share"].round(2)
    if qnum == 10:
        # column order differs between the spec SQL and the pandas form
        exp = exp[["C_CUSTKEY", "C_NAME", "REVENUE", "C_ACCTBAL", "N_NAME",
                   "C_ADDRESS", "C_PHONE", "C_COMMENT"]]
        got["revenue"] = got["revenue"].round(2)
    if qnum == 14:
        got["promo_revenue"] = got["promo_revenue"].round(2)
    if qnum == 18:
        exp = exp[["C_NAME", "C_CUSTKEY", "O_ORDERKEY", "O_ORDERDATE",
                   "O_TOTALPRICE", "L_QUANTITY"]]
    if qnum == 20:
        # the pandas form keeps one row per qualifying partsupp pair; the
        # SQL IN-subquery is per-supplier (spec semantics)
        exp = exp.drop_duplicates().reset_index(drop=True)
    assert got.shape[0] == exp.shape[0], (got.shape, exp.shape)
    assert got.shape[1] == exp.shape[1], (list(got.columns), list(exp.columns))
    got.columns = list(range(got.shape[1]))
    exp.columns = list(range(exp.shape[1]))
    order = list(got.columns)
    got = got.sort_values(order).reset_index(drop=True)
    exp = exp.sort_values(order).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False,
                                  atol=1e-6, rtol=1e-6)
