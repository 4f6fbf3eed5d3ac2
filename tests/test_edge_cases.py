"""Edge-case hardening: empty frames, all-null columns, single rows,
unicode, zero-group aggregations — shapes that commonly break columnar
engines (reference analog: the check_func parameter matrix over
dataframe_common.py fixtures)."""

import numpy as np
import pandas as pd
import pytest

import bodo_amd.pandas as bpd
from bodo_amd.sql import BodoSQLContext


def _decat(d):
    d = d.copy()
    for c in d.columns:
        if isinstance(d[c].dtype, pd.CategoricalDtype):
            d[c] = d[c].astype(object)
    return d


def test_empty_frame_ops():
    df = pd.DataFrame({"a": pd.Series([], dtype="int64"),
                       "b": pd.Series([], dtype="float64")})
    b = bpd.from_pandas(df)
    assert len(b[b.a > 0].to_pandas()) == 0
    g = b.groupby("a", as_index=False).agg(s=bpd.NamedAgg("b", "sum"))
    assert len(g.to_pandas()) == 0
    assert len(b.sort_values("a").to_pandas()) == 0
    assert b.b.sum() == 0.0


def test_empty_after_filter_groupby():
    df = pd.DataFrame({"a": [1, 2, 3], "b": [1.0, 2.0, 3.0]})
    b = bpd.from_pandas(df)
    out = b[b.a > 99].groupby("a", as_index=False).agg(
        s=bpd.NamedAgg("b", "sum")).to_pandas()
    assert len(out) == 0


def test_single_row():
    df = pd.DataFrame({"a": [7], "b": [3.5], "c": ["only"]})
    b = bpd.from_pandas(df)
    out = _decat(b[b.a > 0].to_pandas())
    pd.testing.assert_frame_equal(out, df, check_dtype=False)
    assert b.b.median() == 3.5
    g = b.groupby("c", as_index=False).agg(
        n=bpd.NamedAgg("a", "count")).to_pandas()
    assert g["n"].iloc[0] == 1


def test_all_null_column():
    df = pd.DataFrame({"a": [1, 2, 3], "b": [np.nan] * 3})
    b = bpd.from_pandas(df)
    assert b.b.count() == 0
    assert np.isnan(b.b.mean()) or b.b.mean() is None
    out = b.dropna(subset=["b"]).to_pandas()
    assert len(out) == 0
    g = b.groupby("a", as_index=False).agg(
        s=bpd.NamedAgg("b", "sum")).to_pandas()
    assert len(g) == 3


def test_unicode_strings():
    df = pd.DataFrame({"s": ["héllo", "wörld", "naïve", "héllo", "日本語"],
                       "v": [1.0, 2.0, 3.0, 4.0, 5.0]})
    b = bpd.from_pandas(df)
    g = b.groupby("s", as_index=False).agg(
        s2=bpd.NamedAgg("v", "sum")).to_pandas()
    g["s"] = g["s"].astype(str)
    exp = df.groupby("s", as_index=False).agg(s2=("v", "sum"))
    pd.testing.assert_frame_equal(
        g.sort_values("s").reset_index(drop=True),
        exp.sort_values("s").reset_index(drop=True), check_dtype=False)
    up = b.s.str.upper().to_pandas().astype(str)
    assert (up.to_numpy() == df.s.str.upper().to_numpy()).all()


def test_join_no_matches():
    left = pd.DataFrame({"k": [1, 2, 3], "v": [1.0, 2.0, 3.0]})
    right = pd.DataFrame({"k": [10, 20], "w": [0.1, 0.2]})
    b = bpd.from_pandas(left).merge(bpd.from_pandas(right), on="k")
    assert len(b.to_pandas()) == 0
    bl = bpd.from_pandas(left).merge(bpd.from_pandas(right), on="k",
                                     how="left").to_pandas()
    assert len(bl) == 3 and bl["w"].isna().all()


def test_duplicate_heavy_sort():
    df = pd.DataFrame({"a": [5] * 100 + [1] * 100, "b": range(200)})
    b = bpd.from_pandas(df)
    out = b.sort_values("a").to_pandas()
    assert (out.a.to_numpy() == np.sort(df.a.to_numpy())).all()


def test_sql_empty_result():
    df = pd.DataFrame({"a": [1, 2], "b": [0.5, 1.5]})
    bc = BodoSQLContext({"t": df})
    out = bc.sql("select a, sum(b) as s from t where a > 100 "
                 "group by a order by a").to_pandas()
    assert len(out) == 0
    assert list(out.columns) == ["a", "s"]


def test_sql_null_arithmetic():
    df = pd.DataFrame({"a": [1.0, np.nan, 3.0]})
    bc = BodoSQLContext({"t": df})
    out = bc.sql("select a + 1 as x, coalesce(a, 0) as c from t").to_pandas()
    assert np.isnan(out.x.iloc[1])
    assert out.c.iloc[1] == 0.0


def test_wide_frame():
    data = {f"c{i}": np.arange(10) * i for i in range(60)}
    df = pd.DataFrame(data)
    b = bpd.from_pandas(df)
    out = b[b.c1 > 2][["c0", "c30", "c59"]].to_pandas()
    exp = df[df.c1 > 2][["c0", "c30", "c59"]].reset_index(drop=True)
    pd.testing.assert_frame_equal(out, exp, check_dtype=False)


def test_empty_and_single_row_device_ops():
    """Empty/1-row shards through the new device paths (shift/fill/rolling/
    explode/list/decimal) must not fault."""
    from decimal import Decimal

    e = pd.DataFrame({"x": pd.Series([], dtype="float64")})
    be = bpd.from_pandas(e)
    assert len(be.x.shift(2).to_pandas()) == 0
    assert len(be.x.ffill().to_pandas()) == 0
    assert len(be.x.rolling(3).mean().to_pandas()) == 0
    one = pd.DataFrame({"x": [2.5],
                        "l": pd.Series([[1, 2]]),
                        "m": pd.Series([Decimal("1.50")])})
    b1 = bpd.from_pandas(one)
    assert pd.isna(b1.x.shift(1).to_pandas().iloc[0])
    assert b1.explode("l").to_pandas()["l"].tolist() == [1, 2]
    assert float(b1.m.sum()) == 1.5
    empty_list = pd.DataFrame({"l": pd.Series([], dtype=object)})
    # all-empty object column infers null type; explode is a no-op frame
    bl = bpd.from_pandas(empty_list)
    assert len(bl.to_pandas()) == 0
