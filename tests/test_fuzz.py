"""Seeded differential fuzz: random small pipelines (filter/groupby/sort/
join/distinct) and random SQL queries compared against pandas exactly
(reference analog: the check_func parameter matrix; a fixed seed set keeps
CI deterministic — run more seeds ad hoc by raising N_SEEDS)."""

import numpy as np
import pandas as pd
import pytest

import bodo_amd.pandas as bpd
from bodo_amd.sql import BodoSQLContext

N_SEEDS = 40

AGGS = ["sum", "mean", "min", "max", "count", "size"]


def _decat(d):
    d = d.copy()
    for c in d.columns:
        if isinstance(d[c].dtype, pd.CategoricalDtype):
            d[c] = d[c].astype(object)
    return d


def _frame(rng, n):
    return pd.DataFrame({
        "a": rng.integers(-5, 10, n),
        "b": np.where(rng.random(n) < 0.15, np.nan, rng.random(n) * 20 - 5),
        "c": rng.choice(["p", "q", "r", "s"], n),
    })


@pytest.mark.parametrize("seed", range(N_SEEDS))
def test_fuzz_frames(seed):
    rng = np.random.default_rng(seed)
    df = _frame(rng, int(rng.integers(1, 500)))
    b = bpd.from_pandas(df)
    op = seed % 5
    if op == 0:
        lo, hi = sorted(rng.uniform(-5, 15, 2))
        got = _decat(b[(b.b > lo) & (b.b < hi)].to_pandas()).reset_index(
            drop=True)
        exp = df[(df.b > lo) & (df.b < hi)].reset_index(drop=True)
    elif op == 1:
        f1, f2 = rng.choice(AGGS, 2, replace=True)
        got = b.groupby(["a", "c"], as_index=False).agg(
            x=bpd.NamedAgg("b", f1), y=bpd.NamedAgg("b", f2)).to_pandas()
        got = _decat(got).sort_values(["a", "c"]).reset_index(drop=True)
        exp = df.groupby(["a", "c"], as_index=False).agg(
            x=("b", f1), y=("b", f2)).sort_values(["a", "c"]).reset_index(
            drop=True)
    elif op == 2:
        asc = bool(seed % 2)
        got = _decat(b.sort_values(["a", "b"], ascending=asc).to_pandas())
        got = got.reset_index(drop=True)
        exp = df.sort_values(["a", "b"], ascending=asc).reset_index(drop=True)
    elif op == 3:
        m = int(rng.integers(1, 12))
        right = pd.DataFrame({"a": rng.integers(-5, 10, m),
                              "w": rng.random(m)}).drop_duplicates("a")
        how = ["inner", "left"][seed % 2]
        got = _decat(b.merge(bpd.from_pandas(right), on="a",
                             how=how).to_pandas())
        got = got.sort_values(["a", "b"], na_position="last").reset_index(
            drop=True)
        exp = df.merge(right, on="a", how=how).sort_values(
            ["a", "b"], na_position="last").reset_index(drop=True)
    else:
        got = _decat(b.drop_duplicates(subset=["a", "c"]).to_pandas())
        got = got.sort_values(["a", "c", "b"]).reset_index(drop=True)
        exp = df.drop_duplicates(subset=["a", "c"]).sort_values(
            ["a", "c", "b"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False, atol=1e-9)


@pytest.mark.parametrize("seed", range(N_SEEDS // 2))
def test_fuzz_sql(seed):
    rng = np.random.default_rng(1000 + seed)
    n = int(rng.integers(5, 400))
    df = pd.DataFrame({"a": rng.integers(0, 8, n), "b": rng.random(n) * 10,
                       "c": rng.choice(["p", "q", "r"], n)})
    bc = BodoSQLContext({"t": df})
    thr = float(rng.uniform(0, 10))
    mode = seed % 3
    if mode == 0:
        got = bc.sql(f"select a, sum(b) as s, count(*) as n from t "
                     f"where b > {thr} group by a order by a").to_pandas()
        sub = df[df.b > thr]
        exp = sub.groupby("a", as_index=False).agg(
            s=("b", "sum"), n=("b", "size")).sort_values("a").reset_index(
            drop=True)
    elif mode == 1:
        got = bc.sql("select c, avg(b) as m from t group by c "
                     "having count(*) > 2 order by c").to_pandas()
        got["c"] = got["c"].astype(str)
        g = df.groupby("c").agg(m=("b", "mean"), n=("b", "size")).reset_index()
        exp = g[g.n > 2][["c", "m"]].sort_values("c").reset_index(drop=True)
    else:
        got = bc.sql(f"select a, b from t where c in ('p', 'q') "
                     f"and b < {thr} order by a, b limit 7").to_pandas()
        exp = df[(df.c.isin(["p", "q"])) & (df.b < thr)].sort_values(
            ["a", "b"]).head(7)[["a", "b"]].reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False, atol=1e-9)
