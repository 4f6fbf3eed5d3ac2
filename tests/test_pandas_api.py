"""Differential tests for the extended pandas API surface (reference:
bodo/pandas frame.py/series.py method inventory)."""

import numpy as np
import pandas as pd
import pytest

import bodo_amd.pandas as bpd


def _decat(obj):
    if isinstance(obj, pd.DataFrame):
        out = obj.copy()
        for c in out.columns:
            if isinstance(out[c].dtype, pd.CategoricalDtype):
                out[c] = out[c].astype(object)
        return out
    if isinstance(obj.dtype, pd.CategoricalDtype):
        return obj.astype(object)
    return obj


@pytest.fixture()
def df():
    rng = np.random.default_rng(0)
    return pd.DataFrame({
        "a": rng.integers(0, 10, 200),
        "b": np.where(rng.random(200) < 0.1, np.nan, rng.random(200) * 100),
        "c": rng.choice(["xx", "yy", "zz", "ww"], 200),
        "t": pd.to_datetime("2020-01-01")
        + pd.to_timedelta(rng.integers(0, 3 * 365, 200), unit="D"),
    })


def test_series_reductions(df):
    b = bpd.from_pandas(df)
    assert abs(b.b.median() - df.b.median()) < 1e-9
    assert abs(b.b.quantile(0.3) - df.b.quantile(0.3)) < 1e-9
    assert abs(b.a.prod() - float(df.a.astype("float64").prod())) < 1e-6
    pd.testing.assert_series_equal(b.b.describe(), df.b.describe(),
                                   check_dtype=False)


def test_series_transforms(df):
    b = bpd.from_pandas(df)
    pd.testing.assert_series_equal(b.b.between(20, 60).to_pandas(),
                                   df.b.between(20, 60), check_names=False,
                                   check_dtype=False)
    pd.testing.assert_series_equal(b.b.clip(10, 90).to_pandas(),
                                   df.b.clip(10, 90), check_names=False,
                                   check_dtype=False)
    assert list(b.b.nlargest(5)) == list(df.b.nlargest(5))
    assert list(b.b.nsmallest(5)) == list(df.b.nsmallest(5))
    assert sorted(b.b.dropna().to_pandas()) == sorted(df.b.dropna())
    assert list(b.c.mode()) == list(df.c.mode())


def test_frame_astype_fillna(df):
    b = bpd.from_pandas(df)
    out = b.astype({"a": "float64"}).to_pandas()
    assert out["a"].dtype == np.float64
    f = b.fillna({"b": -1.0}).to_pandas()
    exp = df.fillna({"b": -1.0})
    pd.testing.assert_series_equal(f["b"], exp["b"], check_dtype=False)


def test_frame_query_iloc_describe(df):
    b = bpd.from_pandas(df)
    got = _decat(b.query("a > 3 and b < 80").to_pandas())
    exp = df.query("a > 3 and b < 80").reset_index(drop=True)
    pd.testing.assert_frame_equal(got.reset_index(drop=True), exp,
                                  check_dtype=False)
    pd.testing.assert_frame_equal(_decat(b.iloc[:7].to_pandas()), df.iloc[:7],
                                  check_dtype=False)
    pd.testing.assert_frame_equal(b.describe(), df[["a", "b"]].describe(),
                                  check_dtype=False)
    assert dict(b.nunique()) == dict(df.nunique())


def test_str_generic(df):
    b = bpd.from_pandas(df)
    cases = [
        ("replace", ("x", "Q"), {}),
        ("zfill", (5,), {}),
        ("isalpha", (), {}),
        ("find", ("y",), {}),
        ("rstrip", ("wz",), {}),
        ("repeat", (2,), {}),
    ]
    for op, args, kw in cases:
        got = _decat(getattr(b.c.str, op)(*args, **kw).to_pandas())
        exp = getattr(df.c.str, op)(*args, **kw)
        pd.testing.assert_series_equal(got, exp, check_names=False,
                                       check_dtype=False)


def test_dt_extended(df):
    b = bpd.from_pandas(df)
    for fld in ["is_month_start", "is_month_end", "is_quarter_start",
                "is_year_start", "is_year_end", "days_in_month"]:
        got = getattr(b.t.dt, fld).to_pandas().to_numpy()
        exp = getattr(df.t.dt, fld).to_numpy()
        assert (got == exp).all(), fld
    got = _decat(b.t.dt.month_name().to_pandas()).astype(str)
    assert (got.to_numpy() == df.t.dt.month_name().to_numpy()).all()
    got = _decat(b.t.dt.day_name().to_pandas()).astype(str)
    assert (got.to_numpy() == df.t.dt.day_name().to_numpy()).all()


def test_series_sort_values(df):
    b = bpd.from_pandas(df)
    got = b.a.sort_values().to_pandas().to_numpy()
    assert (got == np.sort(df.a.to_numpy())).all()
    got = b.a.sort_values(ascending=False).to_pandas().to_numpy()
    assert (got == np.sort(df.a.to_numpy())[::-1]).all()


def test_melt(df):
    b = bpd.from_pandas(df)
    got = b.melt(id_vars=["a"], value_vars=["b"]).to_pandas()
    got["variable"] = got["variable"].astype(str)
    got = got.sort_values(["a", "value"]).reset_index(drop=True)
    exp = df.melt(id_vars=["a"], value_vars=["b"]).sort_values(
        ["a", "value"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_pivot_table(df):
    b = bpd.from_pandas(df)
    got = b.pivot_table(values="b", index="a", columns="c", aggfunc="mean")
    exp = df.pivot_table(values="b", index="a", columns="c", aggfunc="mean")
    got.index = got.index.astype(exp.index.dtype)
    got.columns = [str(c) for c in got.columns]
    exp.columns = [str(c) for c in exp.columns]
    pd.testing.assert_frame_equal(got.sort_index(), exp.sort_index(),
                                  check_dtype=False, check_names=False)


def test_read_sql_roundtrip(tmp_path, df):
    dbp = str(tmp_path / "t.db")
    b = bpd.from_pandas(df[["a", "b", "c"]])
    b.to_sql("tab", dbp, if_exists="replace")
    got = bpd.read_sql("select a, b, c from tab where a > 3", dbp).to_pandas()
    got = _decat(got)
    exp = df[df.a > 3][["a", "b", "c"]].reset_index(drop=True)
    pd.testing.assert_frame_equal(
        got.sort_values(["a", "b"]).reset_index(drop=True),
        exp.sort_values(["a", "b"]).reset_index(drop=True), check_dtype=False)


def test_to_json(tmp_path, df):
    p = str(tmp_path / "out.json")
    bpd.from_pandas(df[["a", "b"]]).to_json(p)
    got = pd.read_json(p, orient="records", lines=True)
    pd.testing.assert_frame_equal(got, df[["a", "b"]].reset_index(drop=True),
                                  check_dtype=False)


def test_cumulative(df):
    b = bpd.from_pandas(df)
    for f in ["cumsum", "cumprod", "cummin", "cummax"]:
        got = getattr(b.b, f)().to_pandas()
        exp = getattr(df.b, f)().reset_index(drop=True)
        pd.testing.assert_series_equal(got, exp, check_names=False,
                                       check_dtype=False)
    got = b.a.cumsum().to_pandas()
    pd.testing.assert_series_equal(got, df.a.cumsum().reset_index(drop=True),
                                   check_names=False, check_dtype=False)


def test_shift_diff(df):
    b = bpd.from_pandas(df)
    for k in [1, 3, -2]:
        pd.testing.assert_series_equal(
            b.b.shift(k).to_pandas(), df.b.shift(k).reset_index(drop=True),
            check_names=False, check_dtype=False)
    pd.testing.assert_series_equal(
        b.b.diff().to_pandas(), df.b.diff().reset_index(drop=True),
        check_names=False, check_dtype=False)


def test_groupby_list_agg(df):
    b = bpd.from_pandas(df)
    got = b.groupby("a", as_index=False)["b"].agg(["sum", "max"]).to_pandas()
    got = got.sort_values("a").reset_index(drop=True)
    exp = df.groupby("a", as_index=False)["b"].agg(["sum", "max"]).sort_values(
        "a").reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_frame_reductions(df):
    b = bpd.from_pandas(df)
    pd.testing.assert_series_equal(b.sum(), df[["a", "b"]].sum(),
                                   check_dtype=False)
    pd.testing.assert_series_equal(b.mean(), df[["a", "b"]].mean(),
                                   check_dtype=False)
    pd.testing.assert_series_equal(b.min(), df[["a", "b"]].min(),
                                   check_dtype=False)


def test_series_rank(df):
    b = bpd.from_pandas(df)
    for m in ["average", "min", "dense"]:
        got = b.b.rank(method=m).to_pandas()
        exp = df.b.rank(method=m).reset_index(drop=True)
        pd.testing.assert_series_equal(got, exp, check_names=False,
                                       check_dtype=False)


def test_str_split_get(df):
    b = bpd.from_pandas(df)
    src = pd.DataFrame({"s": ["a-b-c", "x-y", "solo", None] * 50})
    bs = bpd.from_pandas(src)
    got = bs.s.str.split("-").str.get(1).to_pandas()
    exp = src.s.str.split("-").str.get(1)
    got = _decat(got).where(lambda x: x.notna(), np.nan)
    exp = exp.where(exp.notna(), np.nan)
    pd.testing.assert_series_equal(got, exp, check_names=False,
                                   check_dtype=False)


def test_groupby_head(df):
    b = bpd.from_pandas(df)
    got = _decat(b.groupby("a").head(3).to_pandas())
    got = got.sort_values(["a", "b"]).reset_index(drop=True)
    exp = df.groupby("a").head(3).sort_values(["a", "b"]).reset_index(
        drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_corr_cov(df):
    b = bpd.from_pandas(df)
    exp = df[["a", "b"]].astype(float)
    assert abs(b.a.corr(b.b) - exp.a.corr(exp.b)) < 1e-9
    assert abs(b.a.cov(b.b) - exp.a.cov(exp.b)) < 1e-9
    pd.testing.assert_frame_equal(b.corr(), df[["a", "b"]].corr(),
                                  check_dtype=False, atol=1e-9)


def test_series_getitem_mask(df):
    b = bpd.from_pandas(df)
    got = b.b[b.a > 5].to_pandas()
    exp = df.b[df.a > 5].reset_index(drop=True)
    pd.testing.assert_series_equal(got, exp, check_names=False,
                                   check_dtype=False)
    got2 = b.a[:7].to_pandas()
    assert list(got2) == list(df.a[:7])


def test_groupby_apply_frame_return(df):
    import warnings

    b = bpd.from_pandas(df)

    def top2(g):
        return g.nlargest(2, "b")

    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        got = _decat(b.groupby("a").apply(top2).to_pandas())
        got = got.sort_values(["a", "b"]).reset_index(drop=True)
        exp = df.groupby("a", sort=False).apply(top2).reset_index(drop=True)
    exp = exp.sort_values(["a", "b"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got[exp.columns], exp, check_dtype=False)


def test_series_pct_change_duplicated(df):
    b = bpd.from_pandas(df)
    pd.testing.assert_series_equal(
        b.b.pct_change().to_pandas(),
        df.b.pct_change(fill_method=None).reset_index(drop=True),
        check_names=False, check_dtype=False)
    pd.testing.assert_series_equal(
        b.a.duplicated().to_pandas(), df.a.duplicated().reset_index(drop=True),
        check_names=False, check_dtype=False)
    assert list(b.b.to_frame("x").to_pandas().columns) == ["x"]


def test_agg_extras(df):
    b = bpd.from_pandas(df)
    assert list((-b.a).to_pandas()) == list(-df.a)
    assert abs(b.b.agg("sum") - df.b.sum()) < 1e-9
    ser = b.b.agg(["sum", "mean"])
    assert abs(ser["sum"] - df.b.sum()) < 1e-9
    ag = b.agg({"a": "max", "b": "sum"})
    assert ag["a"] == df.a.max() and abs(ag["b"] - df.b.sum()) < 1e-9
    assert b.memory_usage()["a"] > 0


def test_string_shift_ffill():
    src = pd.DataFrame({"s": ["a", None, "b", None, None, "c"] * 10})
    b = bpd.from_pandas(src)
    got = _decat(b.s.shift(1).to_pandas()).astype(object)
    exp = src.s.shift(1).astype(object)
    assert (got.fillna("~") == exp.fillna("~")).all()
    got = _decat(b.s.ffill().to_pandas()).astype(str)
    assert (got.to_numpy() == src.s.ffill().astype(str).to_numpy()).all()


def test_get_dummies(df):
    b = bpd.from_pandas(df)
    got = bpd.get_dummies(b.c).to_pandas().astype(bool)
    exp = pd.get_dummies(df.c).astype(bool).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)
    g2 = bpd.get_dummies(b[["a", "c"]], columns=["c"]).to_pandas()
    e2 = pd.get_dummies(df[["a", "c"]], columns=["c"])
    assert list(g2.columns) == list(e2.columns)


def test_cut_qcut(df):
    b = bpd.from_pandas(df)
    bins = [0.0, 25.0, 50.0, 75.0, 100.0]
    got = bpd.cut(b.b, bins, labels=False).to_pandas()
    exp = pd.cut(df.b, bins, labels=False).reset_index(drop=True)
    pd.testing.assert_series_equal(got, exp, check_names=False,
                                   check_dtype=False)
    gq = bpd.qcut(b.b.dropna(), 4, labels=False).to_pandas()
    eq = pd.qcut(df.b.dropna(), 4, labels=False).reset_index(drop=True)
    pd.testing.assert_series_equal(gq, eq.astype(float), check_names=False,
                                   check_dtype=False)


def test_groupby_tail_nth_filter(df):
    b = bpd.from_pandas(df)
    got = _decat(b.groupby("a").tail(2).to_pandas())
    got = got.sort_values(["a", "b"]).reset_index(drop=True)
    exp = df.groupby("a").tail(2).sort_values(["a", "b"]).reset_index(
        drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)
    got = _decat(b.groupby("a").nth(0).to_pandas())
    got = got.sort_values(["a", "b"]).reset_index(drop=True)
    exp = df.groupby("a", as_index=False).nth(0).sort_values(
        ["a", "b"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)
    got = _decat(b.groupby("a").filter(
        lambda g: len(g) > 15).to_pandas())
    got = got.sort_values(["a", "b"]).reset_index(drop=True)
    exp = df.groupby("a").filter(lambda g: len(g) > 15).sort_values(
        ["a", "b"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_cross_merge():
    l = pd.DataFrame({"a": [1, 2]})
    r = pd.DataFrame({"b": ["x", "y", "z"]})
    got = _decat(bpd.from_pandas(l).merge(bpd.from_pandas(r),
                                          how="cross").to_pandas())
    exp = l.merge(r, how="cross")
    pd.testing.assert_frame_equal(
        got.sort_values(["a", "b"]).reset_index(drop=True),
        exp.sort_values(["a", "b"]).reset_index(drop=True),
        check_dtype=False)


def test_series_tail_idx(df):
    b = bpd.from_pandas(df)
    assert list(b.a.tail(4)) == list(df.a.tail(4))
    assert b.b.idxmax() == int(df.b.idxmax())
    assert b.b.idxmin() == int(df.b.idxmin())
    assert len(b.b.sample(n=9, random_state=1).to_pandas()) == 9


def test_frame_filter_get_reset(df):
    b = bpd.from_pandas(df)
    assert list(b.filter(items=["a", "b"]).columns) == ["a", "b"]
    assert list(b.filter(regex="^[ab]$").columns) == ["a", "b"]
    assert b.get("nope", 3) == 3
    assert b.reset_index(drop=True) is b


def test_groupby_any_all_skew(df):
    src = df.copy()
    src["flag"] = src.a % 2 == 0
    b = bpd.from_pandas(src)
    got = b.groupby("c", as_index=False).agg(
        a=bpd.NamedAgg("flag", "any"), l=bpd.NamedAgg("flag", "all"),
        s=bpd.NamedAgg("b", "skew")).to_pandas()
    got = _decat(got).sort_values("c").reset_index(drop=True)
    exp = src.groupby("c", as_index=False).agg(
        a=("flag", "any"), l=("flag", "all"),
        s=("b", "skew")).sort_values("c").reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False, atol=1e-6)


def test_astype_category(df):
    b = bpd.from_pandas(df)
    out = b.c.astype("category").to_pandas()
    assert out.dtype.name == "category"
    assert sorted(out.astype(str).unique()) == sorted(df.c.unique())


def test_to_parquet_partition_cols(tmp_path, df):
    p = str(tmp_path / "out")
    bpd.from_pandas(df[["a", "c", "b"]]).to_parquet(p, partition_cols=["c"])
    back = pd.read_parquet(p)
    assert len(back) == len(df)
    import os

    assert any(d.startswith("c=") for d in os.listdir(p))


def test_read_parquet_filters(tmp_path, df):
    p = str(tmp_path / "t.parquet")
    df[["a", "b"]].to_parquet(p, row_group_size=32)
    out = bpd.read_parquet(p, filters=[("a", ">", 4)]).to_pandas()
    exp = df[df.a > 4][["a", "b"]].reset_index(drop=True)
    pd.testing.assert_frame_equal(
        out.sort_values(["a", "b"]).reset_index(drop=True),
        exp.sort_values(["a", "b"]).reset_index(drop=True),
        check_dtype=False)


def test_series_replace(df):
    b = bpd.from_pandas(df)
    got = _decat(b.c.replace({"xx": "QQ"}).to_pandas()).astype(str)
    exp = df.c.replace({"xx": "QQ"})
    assert (got.to_numpy() == exp.to_numpy()).all()
    gn = b.a.replace([0, 1], -1).to_pandas()
    assert (gn.to_numpy() == df.a.replace([0, 1], -1).to_numpy()).all()


def test_frame_replace():
    df = pd.DataFrame({"a": [1, 2, 3], "b": [2.0, 5.0, 2.0]})
    got = bpd.from_pandas(df).replace(2, 99).to_pandas()
    pd.testing.assert_frame_equal(got, df.replace(2, 99), check_dtype=False)


def test_frame_round_value_counts(df):
    b = bpd.from_pandas(df)
    got = b[["b"]].round(1).to_pandas()
    pd.testing.assert_frame_equal(got, df[["b"]].round(1).reset_index(
        drop=True), check_dtype=False)
    vc = b.value_counts(subset=["a"])
    exp = df.value_counts(subset=["a"])
    assert vc.sum() == exp.sum() and vc.max() == exp.max()


def test_autocorr(df):
    b = bpd.from_pandas(df)
    exp = df.b.autocorr(1)
    got = b.b.autocorr(1)
    assert abs(got - exp) < 1e-9


def test_cross_frame_series_arithmetic():
    a = bpd.from_pandas(pd.DataFrame({"x": [1.0, 2.0, 3.0]}))
    b = bpd.from_pandas(pd.DataFrame({"y": [10.0, 20.0, 30.0]}))
    assert (a.x + b.y).to_pandas().tolist() == [11.0, 22.0, 33.0]
    assert (b.y - a.x).to_pandas().tolist() == [9.0, 18.0, 27.0]
    assert (a.x > b.y).to_pandas().tolist() == [False] * 3


def test_cross_frame_setitem():
    a = bpd.from_pandas(pd.DataFrame({"x": [1, 2, 3]}))
    b = bpd.from_pandas(pd.DataFrame({"y": [10.0, 20.0, 30.0]}))
    a["z"] = b.y
    out = a.to_pandas()
    assert out.z.tolist() == [10.0, 20.0, 30.0]


def test_set_index_reset_index_multiindex():
    rng = np.random.default_rng(17)
    df = pd.DataFrame({"a": rng.integers(0, 5, 50),
                       "b": rng.integers(0, 3, 50),
                       "v": rng.random(50)})
    b = bpd.from_pandas(df)
    one = b.set_index("a")
    got = one.to_pandas()
    assert got.index.name == "a"
    assert list(got.columns) == ["b", "v"] or "a" in got.columns
    # MultiIndex
    two = b.set_index(["a", "b"])
    got2 = two.to_pandas()
    assert list(got2.index.names) == ["a", "b"]
    assert list(got2.columns) == ["v"]
    # reset back
    r = two.reset_index().to_pandas()
    assert list(r.columns)[:2] == ["a", "b"]
    rd = two.reset_index(drop=True).to_pandas()
    assert list(rd.columns) == ["v"]
    # sort_index
    si = two.sort_index().to_pandas()
    assert list(si.index.get_level_values(0)) == sorted(df["a"].tolist())


def test_list_columns_and_explode():
    """LIST (arrow list<child>) columns: round trip, gather, concat and
    pandas-semantics explode (reference: array_item_arr_ext.py +
    _lateral.cpp FLATTEN)."""
    import pyarrow as pa

    df = pd.DataFrame({"k": [1, 2, 3, 4, 5],
                       "l": pd.Series([[1, 2], [3], [], None, [4, 5, 6]])})
    b = bpd.from_pandas(df)
    got = b.explode("l").to_pandas().reset_index(drop=True)
    exp = df.explode("l").reset_index(drop=True)
    assert got["k"].tolist() == exp["k"].tolist()
    gv = ["" if pd.isna(v) else float(v) for v in got["l"]]
    ev = ["" if pd.isna(v) else float(v) for v in exp["l"]]
    assert gv == ev
    # explode then aggregate (the lateral-join shape)
    out = b.explode("l").groupby("k", as_index=False).agg(
        s=bpd.NamedAgg("l", "sum")).sort_values("k").to_pandas()
    exp2 = df.explode("l").groupby("k", as_index=False).agg(
        s=("l", "sum")).sort_values("k").reset_index(drop=True)
    assert [float(v) for v in out["s"]] == [float(v) for v in exp2["s"]]


def test_list_string_explode():
    df = pd.DataFrame({"k": [1, 2],
                       "l": pd.Series([["a", "bb"], ["ccc"]])})
    b = bpd.from_pandas(df)
    got = b.explode("l").to_pandas().reset_index(drop=True)
    assert got["l"].astype(str).tolist() == ["a", "bb", "ccc"]


def test_str_split_list_and_explode():
    """str.split materializes as a LIST<string> series; explode after split
    is the csv-tags lateral pattern."""
    df = pd.DataFrame({"k": [1, 2, 3],
                       "s": ["a,b", "c", "d,e,f"]})
    b = bpd.from_pandas(df)
    parts = b.s.str.split(",")
    got = parts.to_pandas()
    assert [list(v) for v in got] == [["a", "b"], ["c"], ["d", "e", "f"]]
    # fused element access still works
    first = b.s.str.split(",").get(0).to_pandas()
    assert first.astype(str).tolist() == ["a", "c", "d"]
    # assign + explode
    b["parts"] = b.s.str.split(",")._series()
    out = b.explode("parts").to_pandas()
    assert out["parts"].astype(str).tolist() == ["a", "b", "c", "d", "e", "f"]
    assert out["k"].tolist() == [1, 1, 2, 3, 3, 3]


def test_rolling_device_path_variants():
    """Global rolling sum/mean/min/max/count via device prefix ops (host
    pandas only for exotic funcs)."""
    import bodo_amd.engine.executor as E

    rng = np.random.default_rng(51)
    n = 5000
    df = pd.DataFrame({"x": rng.random(n)})
    df.loc[rng.random(n) < 0.1, "x"] = np.nan
    b = bpd.from_pandas(df)
    for f in ("sum", "mean", "min", "max", "count"):
        for w, mp in ((7, None), (12, 3)):
            r = b.x.rolling(w, min_periods=mp)
            got = getattr(r, f)().to_pandas().reset_index(drop=True)
            exp = getattr(df.x.rolling(w, min_periods=mp), f)().reset_index(
                drop=True)
            pd.testing.assert_series_equal(got, exp, check_names=False,
                                           check_dtype=False, atol=1e-9)


def test_ffill_bfill_device_differential():
    rng = np.random.default_rng(61)
    n = 3000
    x = rng.random(n)
    x[rng.random(n) < 0.3] = np.nan
    x[:5] = np.nan  # leading nulls stay null under ffill
    df = pd.DataFrame({"x": x, "i": rng.integers(0, 9, n).astype("float64")})
    df.loc[rng.random(n) < 0.2, "i"] = np.nan
    b = bpd.from_pandas(df)
    for f in ("ffill", "bfill"):
        got = getattr(b.x, f)().to_pandas().reset_index(drop=True)
        exp = getattr(df.x, f)().reset_index(drop=True)
        pd.testing.assert_series_equal(got, exp, check_names=False,
                                       check_dtype=False)
        got2 = getattr(b.i, f)().to_pandas().reset_index(drop=True)
        exp2 = getattr(df.i, f)().reset_index(drop=True)
        pd.testing.assert_series_equal(got2, exp2, check_names=False,
                                       check_dtype=False)


def test_str_findall_extract_rsplit():
    df = pd.DataFrame({"s": ["a1b22c333", "x9", "nope", None]})
    b = bpd.from_pandas(df)
    got = b.s.str.findall(r"\d+").to_pandas()
    exp = df.s.str.findall(r"\d+")
    assert [None if v is None else list(v) for v in got] == \
        [None if not isinstance(v, list) else v for v in exp]
    got2 = b.s.str.extract(r"([a-z])\d", expand=False).to_pandas()
    exp2 = df.s.str.extract(r"([a-z])\d", expand=False)
    assert [None if pd.isna(v) else v for v in got2] == \
        [None if pd.isna(v) else v for v in exp2]
    got3 = b.s.str.rsplit("b").to_pandas()
    exp3 = df.s.str.rsplit("b")
    assert [None if v is None else list(v) for v in got3] == \
        [None if not isinstance(v, list) else v for v in exp3]


def test_list_accessor_len_get():
    df = pd.DataFrame({"l": pd.Series([[1, 2], [3], [], None, [4, 5, 6]])})
    b = bpd.from_pandas(df)
    ln = b.l.list.len().to_pandas()
    assert [None if pd.isna(v) else int(v) for v in ln] == [2, 1, 0, None, 3]
    g0 = b.l.list.get(0).to_pandas()
    assert [None if pd.isna(v) else int(v) for v in g0] == [1, 3, None, None, 4]
    gm1 = b.l.list[-1].to_pandas()
    assert [None if pd.isna(v) else int(v) for v in gm1] == \
        [2, 3, None, None, 6]
    # SQL surface
    from bodo_amd.sql import BodoSQLContext

    bc = BodoSQLContext({"t": df})
    out = bc.sql("select array_size(l) as n, get(l, 1) as e from t"
                 ).to_pandas()
    assert [None if pd.isna(v) else int(v) for v in out["n"]] == \
        [2, 1, 0, None, 3]
    assert [None if pd.isna(v) else int(v) for v in out["e"]] == \
        [2, None, None, None, 5]


def test_outer_join_null_keys_preserved():
    """Outer/left joins keep NULL keys as nulls — not storage fill values
    (coalesce must honor validity masks, not only NaN payloads)."""
    l = pd.DataFrame({"k": [1.0, np.nan, 2.0], "v": [10, 20, 30]})
    r = pd.DataFrame({"k": [1.0, 3.0], "w": [100, 200]})
    got = (bpd.from_pandas(l).merge(bpd.from_pandas(r), on="k", how="outer")
           .to_pandas())
    exp = l.merge(r, on="k", how="outer")
    g = got.sort_values(["k", "v"], na_position="last").reset_index(drop=True)
    e = exp.sort_values(["k", "v"], na_position="last").reset_index(drop=True)
    pd.testing.assert_frame_equal(g, e, check_dtype=False)


def test_outer_join_mixed_int_float_keys():
    """int64 left key vs float64-with-NaN right key coalesces in float64
    (casting NaN to int64 min corrupted unmatched rows)."""
    l = pd.DataFrame({"k": [1, 2], "v": [10, 20]})
    r = pd.DataFrame({"k": [2.0, np.nan, 4.0], "w": [7, 8, 9]})
    got = (bpd.from_pandas(l).merge(bpd.from_pandas(r), on="k", how="outer")
           .to_pandas())
    exp = l.merge(r, on="k", how="outer")
    g = got.sort_values(["k", "w"], na_position="last").reset_index(drop=True)
    e = exp.sort_values(["k", "w"], na_position="last").reset_index(drop=True)
    pd.testing.assert_frame_equal(g, e, check_dtype=False)


def test_outer_join_empty_left_string_key():
    """Empty left frame (degenerate null schema) outer-joined on a string
    key must yield string values, not dictionary codes."""
    l = pd.DataFrame({"k": pd.Series([], dtype=object),
                      "v": pd.Series([], dtype=float)})
    r = pd.DataFrame({"k": ["a", "b"], "w": [1, 2]})
    got = (bpd.from_pandas(l).merge(bpd.from_pandas(r), on="k", how="outer")
           .to_pandas())
    assert sorted(map(str, got["k"])) == ["a", "b"]
    assert sorted(got["w"]) == [1, 2]


def test_crosstab_and_pivot_index_only():
    rng = np.random.default_rng(0)
    df = pd.DataFrame({"g": rng.choice(["a", "b", "c"], 100),
                       "h": rng.choice(["p", "q"], 100),
                       "y": rng.random(100)})
    b = bpd.from_pandas(df)
    got = bpd.crosstab(b["g"], b["h"])
    exp = pd.crosstab(df["g"], df["h"])
    assert (got.values == exp.values).all()
    got2 = bpd.crosstab(b["g"], b["h"], values=b["y"], aggfunc="mean")
    exp2 = pd.crosstab(df["g"], df["h"], values=df["y"], aggfunc="mean")
    np.testing.assert_allclose(got2.values, exp2.values)
    gotp = b.pivot_table(values="y", index="g", aggfunc="mean")
    expp = df.pivot_table(values="y", index="g", aggfunc="mean")
    np.testing.assert_allclose(gotp.values, expp.values)


def test_groupby_extended_methods():
    """ngroup/idxmax/idxmin/cummin/cummax/pct_change/sample/expanding/
    describe on groupby (round-2 coverage)."""
    rng = np.random.default_rng(2)
    df = pd.DataFrame({"g": rng.choice(["a", "b", "c"], 120),
                       "x": rng.integers(0, 50, 120),
                       "y": np.where(rng.random(120) < 0.1, np.nan,
                                     rng.random(120))})
    b = bpd.from_pandas(df)
    assert np.asarray(b.groupby("g").ngroup().to_pandas()).tolist() == \
        df.groupby("g").ngroup().tolist()
    for m in ("cummin", "cummax"):
        got = getattr(b.groupby("g")["y"], m)().to_pandas()
        want = getattr(df.groupby("g")["y"], m)()
        np.testing.assert_allclose(got.fillna(-9e9), want.fillna(-9e9))
    got = b.groupby("g")["y"].pct_change().to_pandas()
    want = df.groupby("g")["y"].pct_change(fill_method=None)
    np.testing.assert_allclose(got.fillna(-9e9), want.fillna(-9e9))
    assert list(b.groupby("g")["y"].idxmax()) == \
        list(df.groupby("g")["y"].idxmax())
    assert list(b.groupby("g")["y"].idxmin()) == \
        list(df.groupby("g")["y"].idxmin())
    s = b.groupby("g").sample(2, random_state=0).to_pandas()
    assert len(s) == 6
    got = b.groupby("g")["y"].expanding().mean().to_pandas()
    want = df.groupby("g")["y"].transform(lambda t: t.expanding(1).mean())
    np.testing.assert_allclose(got.fillna(-9e9), want.fillna(-9e9))
    d = b.groupby("g")["y"].describe()
    d = d.to_pandas() if hasattr(d, "to_pandas") else d
    assert len(d) == 3 and "mean" in d.columns


def test_merge_asof_and_to_datetime_options():
    l = pd.DataFrame({"t": [1, 5, 10, 3, 8], "v": [1.0, 2.0, 3.0, 4.0, 5.0]})
    r = pd.DataFrame({"t": [0, 4, 9], "w": [10, 20, 30]})
    got = bpd.merge_asof(bpd.from_pandas(l), bpd.from_pandas(r),
                         on="t").to_pandas().reset_index(drop=True)
    exp = pd.merge_asof(l.sort_values("t"), r,
                        on="t").reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)
    s = bpd.from_pandas(pd.DataFrame({"s": ["01/02/2024", None]}))["s"]
    out = bpd.to_datetime(s, format="%d/%m/%Y").to_pandas()
    assert out.iloc[0] == pd.Timestamp("2024-02-01") and pd.isna(out.iloc[1])
    s2 = bpd.from_pandas(pd.DataFrame({"s": ["2024-01-01", "bad"]}))["s"]
    out2 = bpd.to_datetime(s2, errors="coerce").to_pandas()
    assert out2.iloc[0] == pd.Timestamp("2024-01-01") and pd.isna(out2.iloc[1])


def test_rolling_center_timebased_ewm():
    rng = np.random.default_rng(3)
    df = pd.DataFrame({"y": np.where(rng.random(100) < 0.1, np.nan,
                                     rng.random(100)),
                       "t": pd.date_range("2024-01-01", periods=100,
                                          freq="6h")})
    b = bpd.from_pandas(df)
    got = b.rolling(7, center=True).mean().to_pandas()["y"]
    want = df[["y"]].rolling(7, center=True).mean()["y"]
    np.testing.assert_allclose(got.fillna(-9), want.fillna(-9))
    gs = b["y"].rolling(7, center=True).mean().to_pandas()
    np.testing.assert_allclose(gs.fillna(-9), want.fillna(-9))
    g2 = b.set_index("t").rolling("1D").mean()
    g2 = g2.to_pandas() if hasattr(g2, "to_pandas") else g2
    w2 = df.set_index("t")[["y"]].rolling("1D").mean()
    np.testing.assert_allclose(np.asarray(g2["y"]), w2["y"].to_numpy(),
                               equal_nan=True)
    ge = b["y"].ewm(alpha=0.3).mean()
    we = df["y"].ewm(alpha=0.3).mean()
    np.testing.assert_allclose(ge.fillna(-9), we.fillna(-9))


def test_series_rank_pct():
    rng = np.random.default_rng(4)
    df = pd.DataFrame({"x": rng.integers(0, 30, 100).astype(float)})
    df.loc[3, "x"] = np.nan
    b = bpd.from_pandas(df)
    got = b["x"].rank(method="min", pct=True).to_pandas()
    want = df["x"].rank(method="min", pct=True)
    np.testing.assert_allclose(got.fillna(-9), want.fillna(-9))


def test_series_ai_accessor():
    """Series.ai with local callables (reference: BodoSeriesAiMethods;
    offline build — endpoints are caller-provided functions)."""
    df = pd.DataFrame({"s": ["hello world", "foo", None]})
    b = bpd.from_pandas(df)
    out = b["s"].ai.tokenize(
        lambda t: [ord(c) % 97 for c in t[:4]]).to_pandas()
    assert list(out.iloc[0]) == [ord(c) % 97 for c in "hell"]
    assert out.iloc[2] is None or pd.isna(out.iloc[2])
    emb = b.head(2)["s"].ai.embed(
        lambda batch: [[float(len(x)), 1.0] for x in batch]).to_pandas()
    assert list(emb.iloc[0]) == [11.0, 1.0]
    gen = b["s"].ai.llm_generate(lambda p: p.upper()).to_pandas()
    assert gen.iloc[0] == "HELLO WORLD" and pd.isna(gen.iloc[2])


def test_str_encode_binary():
    """.str.encode returns BINARY columns (dict and plain layouts)."""
    df = pd.DataFrame({"s": ["Hello", None, "abc"]})
    b = bpd.from_pandas(df)
    got = b["s"].str.encode("utf-8").to_pandas()
    assert got.iloc[0] == b"Hello" and pd.isna(got.iloc[1]) \
        and got.iloc[2] == b"abc"


def test_corrwith_and_dot():
    rng = np.random.default_rng(1)
    df = pd.DataFrame({"x": rng.random(50), "y": rng.random(50),
                       "z": rng.random(50)})
    b = bpd.from_pandas(df)
    np.testing.assert_allclose(b[["x", "y"]].corrwith(b["z"]).values,
                               df[["x", "y"]].corrwith(df["z"]).values)
    assert abs(float(b["x"].dot(b["y"])) - df["x"].dot(df["y"])) < 1e-9


def test_value_counts_normalize_and_mask():
    df = pd.DataFrame({"g": ["a", "b", "a", "c"], "x": [1, 2, 3, 4]})
    b = bpd.from_pandas(df)
    vc = b["g"].value_counts(normalize=True)
    assert abs(vc.sum() - 1.0) < 1e-12 and max(vc) == 0.5
    m = b["x"].mask(b["x"] > 2, 0).to_pandas()
    assert m.tolist() == df["x"].mask(df["x"] > 2, 0).tolist()
