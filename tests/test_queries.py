"""End-to-end benchmark-query tests (reference: e2e-tests/tpch/test_q1.py,
benchmarks/nyc_taxi/bodo/nyc_taxi_precipitation.py)."""

import numpy as np
import pandas as pd
import pytest

from tests.utils import check_query


def make_taxi(n=20000, seed=0):
    rng = np.random.default_rng(seed)
    base = pd.Timestamp("2023-01-01").value
    pickup = base + rng.integers(0, 365 * 86400 * 10**9, n)
    trips = pd.DataFrame({
        "hvfhs_license_num": rng.choice(["HV0002", "HV0003", "HV0004", "HV0005"], n),
        "pickup_datetime": pd.to_datetime(pickup),
        "PULocationID": rng.integers(1, 266, n).astype(np.int64),
        "DOLocationID": rng.integers(1, 266, n).astype(np.int64),
        "trip_miles": rng.exponential(3.0, n),
    })
    dates = pd.date_range("2023-01-01", "2023-12-31")
    weather = pd.DataFrame({"DATE": dates,
                            "PRCP": rng.exponential(0.05, len(dates))})
    return trips, weather


def nyc_taxi_q1(m, trips, weather):
    w = weather.rename(columns={"DATE": "date", "PRCP": "precipitation"})
    t = trips
    w["date"] = w["date"].dt.date
    t["date"] = t["pickup_datetime"].dt.date
    t["month"] = t["pickup_datetime"].dt.month
    t["hour"] = t["pickup_datetime"].dt.hour
    t["weekday"] = t["pickup_datetime"].dt.dayofweek.isin([0, 1, 2, 3, 4])
    mt = t.merge(w, on="date", how="inner")
    mt["date_with_precipitation"] = mt["precipitation"] > 0.1

    def get_time_bucket(tt):
        if tt in (8, 9, 10):
            return "morning"
        if tt in (11, 12, 13, 14, 15):
            return "midday"
        if tt in (16, 17, 18):
            return "afternoon"
        if tt in (19, 20, 21):
            return "evening"
        return "other"

    mt["time_bucket"] = mt.hour.map(get_time_bucket)
    g = mt.groupby(
        ["PULocationID", "DOLocationID", "month", "weekday",
         "date_with_precipitation", "time_bucket"],
        as_index=False).agg({"hvfhs_license_num": "count", "trip_miles": "mean"})
    return g.sort_values(
        by=["PULocationID", "DOLocationID", "month", "weekday",
            "date_with_precipitation", "time_bucket"])


def test_nyc_taxi_q1():
    trips, weather = make_taxi()
    check_query(nyc_taxi_q1, {"trips": trips, "weather": weather})


def make_lineitem(n=30000, seed=1):
    rng = np.random.default_rng(seed)
    return pd.DataFrame({
        "L_ORDERKEY": rng.integers(0, 10000, n),
        "L_QUANTITY": rng.integers(1, 51, n).astype(np.float64),
        "L_EXTENDEDPRICE": rng.uniform(1000, 100000, n),
        "L_DISCOUNT": rng.uniform(0, 0.1, n).round(2),
        "L_TAX": rng.uniform(0, 0.08, n).round(2),
        "L_RETURNFLAG": rng.choice(["A", "N", "R"], n),
        "L_LINESTATUS": rng.choice(["O", "F"], n),
        "L_SHIPDATE": pd.to_datetime(
            pd.Timestamp("1992-01-01").value
            + rng.integers(0, 2500 * 86400 * 10**9, n)),
    })


def tpch_q1(m, lineitem):
    var1 = pd.Timestamp("1998-09-02")
    filt = lineitem[lineitem["L_SHIPDATE"] <= var1]
    filt["DISC_PRICE"] = filt.L_EXTENDEDPRICE * (1.0 - filt.L_DISCOUNT)
    filt["CHARGE"] = (filt.L_EXTENDEDPRICE * (1.0 - filt.L_DISCOUNT)
                      * (1.0 + filt.L_TAX))
    gb = filt.groupby(["L_RETURNFLAG", "L_LINESTATUS"], as_index=False)
    agg = gb.agg(
        SUM_QTY=m.NamedAgg("L_QUANTITY", "sum"),
        SUM_BASE_PRICE=m.NamedAgg("L_EXTENDEDPRICE", "sum"),
        SUM_DISC_PRICE=m.NamedAgg("DISC_PRICE", "sum"),
        SUM_CHARGE=m.NamedAgg("CHARGE", "sum"),
        AVG_QTY=m.NamedAgg("L_QUANTITY", "mean"),
        AVG_PRICE=m.NamedAgg("L_EXTENDEDPRICE", "mean"),
        AVG_DISC=m.NamedAgg("L_DISCOUNT", "mean"),
        COUNT_ORDER=m.NamedAgg("L_ORDERKEY", "size"))
    return agg.sort_values(["L_RETURNFLAG", "L_LINESTATUS"])


def test_tpch_q1():
    check_query(tpch_q1, {"lineitem": make_lineitem()})


def test_parquet_roundtrip(tmp_path):
    import bodo_amd.pandas as bpd

    df = make_lineitem(5000)
    p = str(tmp_path / "li.parquet")
    df.to_parquet(p)
    b = bpd.read_parquet(p)
    out = tpch_q1(bpd, b).to_pandas().reset_index(drop=True)
    exp = tpch_q1(pd, df).reset_index(drop=True)
    for c in ("L_RETURNFLAG", "L_LINESTATUS"):
        out[c] = out[c].astype(str)
        exp[c] = exp[c].astype(str)
    pd.testing.assert_frame_equal(out, exp, check_dtype=False)


def test_parquet_pushdown_pruning(tmp_path):
    """Column pruning + filter pushdown reach the scan node."""
    import bodo_amd.pandas as bpd
    from bodo_amd.engine.optimizer import optimize
    from bodo_amd.plan import nodes as pn

    df = make_lineitem(1000)
    p = str(tmp_path / "li2.parquet")
    df.to_parquet(p)
    b = bpd.read_parquet(p)
    q = b[b.L_QUANTITY > 10.0][["L_ORDERKEY", "L_QUANTITY"]]
    plan = optimize(q._lazy_plan)
    scans = [n for n in _walk(plan) if isinstance(n, pn.ParquetScan)]
    assert len(scans) == 1
    assert set(scans[0].columns) == {"L_ORDERKEY", "L_QUANTITY"}
    assert len(scans[0].filters) == 1
    out = q.to_pandas()
    exp = df[df.L_QUANTITY > 10.0][["L_ORDERKEY", "L_QUANTITY"]].reset_index(drop=True)
    pd.testing.assert_frame_equal(out, exp, check_dtype=False)


def _walk(n):
    yield n
    for c in n.children():
        yield from _walk(c)


def test_readme_20m_style(tmp_path):
    """Scaled version of the README quickstart config:
    read_parquet + apply + groupby + to_parquet via bodo_amd.pandas."""
    import bodo_amd.pandas as bpd

    n = 50_000
    rng = np.random.default_rng(2)
    df = pd.DataFrame({
        "A": rng.integers(0, 100, n),
        "B": rng.uniform(0, 1, n),
    })
    src = str(tmp_path / "in.parquet")
    dst = str(tmp_path / "out.parquet")
    df.to_parquet(src)
    b = bpd.read_parquet(src)
    b["C"] = b.A.map(lambda x: x * 2 + 1)
    res = b.groupby("A", as_index=False).agg(s=bpd.NamedAgg("C", "sum"),
                                             m=bpd.NamedAgg("B", "mean"))
    res.to_parquet(dst)
    out = pd.read_parquet(dst).sort_values("A").reset_index(drop=True)
    exp_c = df.A * 2 + 1
    exp = df.assign(C=exp_c).groupby("A", as_index=False).agg(
        s=("C", "sum"), m=("B", "mean")).sort_values("A").reset_index(drop=True)
    pd.testing.assert_frame_equal(out, exp, check_dtype=False)


def test_tpcxbb_q26_style(tmp_path):
    """TPCx-BB Q26 shape (reference: e2e-tests/tpcx-bb/TPCxBB_q26.py):
    csv read with sep/usecols/names/dtype + merge + custom-callable aggs."""
    import bodo_amd.pandas as bpd

    rng = np.random.default_rng(4)
    n = 20000
    ss = pd.DataFrame({
        "junk1": np.zeros(n, dtype=np.int64),
        "junk2": np.zeros(n, dtype=np.int64),
        "ss_item_sk": rng.integers(1, 500, n),
        "ss_customer_sk": rng.integers(1, 300, n),
    })
    item = pd.DataFrame({
        "i_item_sk": np.arange(1, 501),
        "i_class_id": rng.integers(1, 16, 500).astype(np.int32),
        "i_category": rng.choice(["Books", "Music", "Home"], 500),
    })
    ss_path = str(tmp_path / "ss.dat")
    ss.to_csv(ss_path, sep="|", header=False, index=False)

    def q26(m, ss_frame, item_frame, category, item_count):
        item2 = item_frame[item_frame["i_category"] == category]
        sale_items = ss_frame.merge(item2, left_on="ss_item_sk",
                                    right_on="i_item_sk")

        def id1(x):
            return (x == 1).sum()

        def id2(x):
            return (x == 2).sum()

        agg = sale_items.groupby("ss_customer_sk", as_index=False).agg(
            cnt=m.NamedAgg("ss_item_sk", "count"),
            c1=m.NamedAgg("i_class_id", id1),
            c2=m.NamedAgg("i_class_id", id2))
        agg = agg[agg.cnt > item_count]
        return agg.sort_values("ss_customer_sk")

    b_ss = bpd.read_csv(ss_path, sep="|",
                        names=["junk1", "junk2", "ss_item_sk", "ss_customer_sk"],
                        usecols=[2, 3],
                        dtype={"ss_item_sk": np.int64, "ss_customer_sk": np.int64})
    got = q26(bpd, b_ss, bpd.from_pandas(item), "Books", 5).to_pandas()
    exp = q26(pd, ss[["ss_item_sk", "ss_customer_sk"]].copy(), item.copy(),
              "Books", 5).reset_index(drop=True)
    pd.testing.assert_frame_equal(got.reset_index(drop=True), exp,
                                  check_dtype=False)


def test_streaming_aggregate(tmp_path, monkeypatch):
    """Morsel-wise streaming aggregate over parquet (larger-than-HBM path,
    reference: bodo/libs/streaming/_groupby.cpp incremental states)."""
    import bodo_amd.config as cfg
    import bodo_amd.pandas as bpd

    df = make_lineitem(40000)
    p = str(tmp_path / "s.parquet")
    df.to_parquet(p, row_group_size=2000)
    old_mode, old_batch = cfg.STREAMING, cfg.STREAM_BATCH_SIZE
    cfg.STREAMING = "1"
    cfg.STREAM_BATCH_SIZE = 5000
    try:
        b = bpd.read_parquet(p)
        got = tpch_q1(bpd, b).to_pandas().reset_index(drop=True)
        exp = tpch_q1(pd, df.copy()).reset_index(drop=True)
        for c in ("L_RETURNFLAG", "L_LINESTATUS"):
            got[c] = got[c].astype(str)
            exp[c] = exp[c].astype(str)
        pd.testing.assert_frame_equal(got, exp, check_dtype=False)
        # streaming reduce
        b2 = bpd.read_parquet(p)
        s = b2[b2.L_QUANTITY < 25].L_EXTENDEDPRICE.sum()
        exp_s = df[df.L_QUANTITY < 25].L_EXTENDEDPRICE.sum()
        assert abs(s - exp_s) < 1e-6
    finally:
        cfg.STREAMING, cfg.STREAM_BATCH_SIZE = old_mode, old_batch


def test_tpcxbb_q05_style():
    """TPCx-BB Q05 shape: per-user category-click pivot via SQL aggregation
    feeding a distributed logistic regression (BASELINE config 5)."""
    import bodo_amd.pandas as bpd
    from bodo_amd.ml import LogisticRegression
    from bodo_amd.sql import BodoSQLContext

    rng = np.random.default_rng(11)
    n = 30000
    clicks = pd.DataFrame({
        "wcs_user_sk": rng.integers(1, 2000, n),
        "i_category_id": rng.integers(1, 8, n),
    })
    users = pd.DataFrame({
        "c_customer_sk": np.arange(1, 2001),
        "c_education": rng.integers(0, 5, 2000),
    })
    bc = BodoSQLContext({"clicks": clicks, "users": users})
    feats = bc.sql("""
        SELECT wcs_user_sk,
               SUM(CASE WHEN i_category_id = 1 THEN 1 ELSE 0 END) AS cat1,
               SUM(CASE WHEN i_category_id = 2 THEN 1 ELSE 0 END) AS cat2,
               SUM(CASE WHEN i_category_id = 3 THEN 1 ELSE 0 END) AS cat3,
               COUNT(*) AS total
        FROM clicks GROUP BY wcs_user_sk
    """)
    joined = feats.merge(bpd.from_pandas(users), left_on="wcs_user_sk",
                         right_on="c_customer_sk")
    pdf = joined.to_pandas()
    X = pdf[["cat1", "cat2", "cat3", "total"]].to_numpy(dtype=np.float64)
    y = (pdf["c_education"] >= 3).to_numpy(dtype=np.float64)
    m = LogisticRegression(lr=0.5, max_iter=100).fit(X, y)
    proba = m.predict_proba(X)
    assert proba.shape[0] == len(pdf) and np.isfinite(proba).all()
    # sanity vs pandas-computed features
    exp = clicks.groupby("wcs_user_sk").size()
    got = pdf.set_index("wcs_user_sk")["total"].sort_index()
    assert (got == exp.loc[got.index]).all()


def test_streaming_join_pipeline(tmp_path):
    """Morsel-wise probe-side streaming through a join against a resident
    build table (TPC-H q3/q5 shape at SF1000; reference:
    streaming/_join.h probe loop inside _pipeline.h push batches)."""
    import bodo_amd.config as cfg
    import bodo_amd.engine.streaming as st
    import bodo_amd.pandas as bpd

    rng = np.random.default_rng(23)
    n = 60000
    orders = pd.DataFrame({
        "o_key": np.arange(2000),
        "o_flag": rng.choice(["A", "B"], 2000),
    })
    li = pd.DataFrame({
        "l_okey": rng.integers(0, 2000, n),
        "l_qty": rng.random(n) * 50,
        "l_price": rng.random(n) * 1000,
    })
    p = str(tmp_path / "li.parquet")
    li.to_parquet(p, row_group_size=4000)
    old_mode, old_batch = cfg.STREAMING, cfg.STREAM_BATCH_SIZE
    cfg.STREAMING = "1"
    cfg.STREAM_BATCH_SIZE = 7000
    calls = {"n": 0}
    real = st.exec_streaming

    def counted(node, ctx):
        calls["n"] += 1
        return real(node, ctx)

    st_exec = st.exec_streaming
    st.exec_streaming = counted
    import bodo_amd.engine.executor  # noqa: F401 (rebinding target module)

    try:
        b = bpd.read_parquet(p)
        o = bpd.from_pandas(orders)
        m = b[b.l_qty < 40].merge(o, left_on="l_okey", right_on="o_key",
                                  how="inner")
        got = m.groupby("o_flag", as_index=False).agg(
            s=bpd.NamedAgg("l_price", "sum"),
            c=bpd.NamedAgg("l_qty", "count")).sort_values(
            "o_flag").to_pandas().reset_index(drop=True)
        exp_m = li[li.l_qty < 40].merge(orders, left_on="l_okey",
                                        right_on="o_key", how="inner")
        exp = exp_m.groupby("o_flag", as_index=False).agg(
            s=("l_price", "sum"), c=("l_qty", "count")).sort_values(
            "o_flag").reset_index(drop=True)
        got["o_flag"] = got["o_flag"].astype(str)
        pd.testing.assert_frame_equal(got, exp, check_dtype=False)
        assert calls["n"] >= 1, "join pipeline did not stream"
    finally:
        st.exec_streaming = st_exec
        cfg.STREAMING, cfg.STREAM_BATCH_SIZE = old_mode, old_batch
