"""Multi-process distributed tests on gloo (world_size=2), CPU.

Exercises the same exchange logic (hash shuffle, two-phase agg, broadcast
join, range-partitioned sort) that runs over RCCL on MI355X.  Reference
analog: bodo/runtests.py running pytest under mpiexec -n N.
"""

import multiprocessing as mp
import os
import pickle
import sys
import traceback

import numpy as np
import pandas as pd
import pytest

pytestmark = pytest.mark.multi_rank

_PORT = [29600]


def _worker(rank, world, port, fn, payload, out_q):
    try:
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world)
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["BODO_AMD_DEVICE"] = "cpu"
        import warnings

        warnings.filterwarnings("ignore")
        import torch.distributed as dist

        import bodo_amd  # noqa: F401  (auto-inits process group)
        import bodo_amd.pandas as bpd

        res = fn(bpd, rank, payload)
        got = res.to_pandas() if hasattr(res, "to_pandas") else res
        if rank == 0:
            out_q.put(("ok", got))
        dist.barrier()
        dist.destroy_process_group()
    except Exception:
        out_q.put(("err", traceback.format_exc()))


def run_dist(fn, payload, world=2):
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    _PORT[0] += 1
    port = _PORT[0] + os.getpid() % 500
    procs = [ctx.Process(target=_worker, args=(r, world, port, fn, payload, out_q))
             for r in range(world)]
    for p in procs:
        p.start()
    status, result = out_q.get(timeout=180)
    for p in procs:
        p.join(timeout=60)
        if p.is_alive():
            p.terminate()
    assert status == "ok", result
    return result


# ----------------------------------------------------------------------
# module-level query functions (must be picklable)
# ----------------------------------------------------------------------

def _q_groupby(bpd, rank, payload):
    df = payload["df"]
    b = bpd.from_pandas(df)
    return b.groupby(["a", "c"], as_index=False).agg(
        s=bpd.NamedAgg("b", "sum"), m=bpd.NamedAgg("b", "mean"),
        n=bpd.NamedAgg("b", "count"), mx=bpd.NamedAgg("b", "max"),
    ).sort_values(["a", "c"])


def _q_join(bpd, rank, payload):
    l = bpd.from_pandas(payload["left"])
    r = bpd.from_pandas(payload["right"])
    return l.merge(r, on="k", how="inner").sort_values(["k", "v1", "v2"])


def _q_sort(bpd, rank, payload):
    return bpd.from_pandas(payload["df"]).sort_values(
        ["a", "b"], ascending=[True, False])


def _q_taxi(bpd, rank, payload):
    from tests.test_queries import nyc_taxi_q1

    t = bpd.from_pandas(payload["trips"])
    w = bpd.from_pandas(payload["weather"])
    return nyc_taxi_q1(bpd, t, w)


def _q_median(bpd, rank, payload):
    return bpd.from_pandas(payload["df"]).groupby("a", as_index=False).agg(
        md=bpd.NamedAgg("b", "median"), nu=bpd.NamedAgg("c", "nunique"),
    ).sort_values("a")


def _q_distinct(bpd, rank, payload):
    return bpd.from_pandas(payload["df"]).drop_duplicates(
        subset=["a"]).sort_values("a")[["a"]]


def _q_limit(bpd, rank, payload):
    return bpd.from_pandas(payload["df"]).sort_values("b").head(13)


def _df(n=3000, seed=0):
    rng = np.random.default_rng(seed)
    return pd.DataFrame({
        "a": rng.integers(0, 20, n),
        "b": rng.uniform(-1, 1, n),
        "c": rng.choice(["x", "y", "z", "w"], n),
    })


def test_dist_groupby():
    df = _df()
    got = run_dist(_q_groupby, {"df": df}).reset_index(drop=True)
    exp = df.groupby(["a", "c"], as_index=False).agg(
        s=("b", "sum"), m=("b", "mean"), n=("b", "count"), mx=("b", "max"),
    ).sort_values(["a", "c"]).reset_index(drop=True)
    got["c"] = got["c"].astype(str)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_dist_join():
    rng = np.random.default_rng(3)
    left = pd.DataFrame({"k": rng.integers(0, 50, 2000),
                         "v1": rng.uniform(0, 1, 2000)})
    right = pd.DataFrame({"k": np.arange(40), "v2": rng.uniform(0, 1, 40)})
    got = run_dist(_q_join, {"left": left, "right": right}).reset_index(drop=True)
    exp = left.merge(right, on="k", how="inner").sort_values(
        ["k", "v1", "v2"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_dist_sort():
    df = _df(2500, 5)
    got = run_dist(_q_sort, {"df": df}).reset_index(drop=True)
    exp = df.sort_values(["a", "b"], ascending=[True, False]).reset_index(drop=True)
    got["c"] = got["c"].astype(str)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_dist_taxi_q1():
    from tests.test_queries import make_taxi, nyc_taxi_q1

    trips, weather = make_taxi(8000, 7)
    got = run_dist(_q_taxi, {"trips": trips, "weather": weather}).reset_index(drop=True)
    exp = nyc_taxi_q1(pd, trips.copy(), weather.copy()).reset_index(drop=True)
    got["time_bucket"] = got["time_bucket"].astype(str)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_dist_median_nunique():
    df = _df(1500, 9)
    got = run_dist(_q_median, {"df": df}).reset_index(drop=True)
    exp = df.groupby("a", as_index=False).agg(
        md=("b", "median"), nu=("c", "nunique")).sort_values("a").reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_dist_distinct_limit():
    df = _df(1200, 11)
    got = run_dist(_q_distinct, {"df": df}).reset_index(drop=True)
    exp = df.drop_duplicates(subset=["a"]).sort_values("a")[["a"]].reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)
    got2 = run_dist(_q_limit, {"df": df}).reset_index(drop=True)
    exp2 = df.sort_values("b").head(13).reset_index(drop=True)
    got2["c"] = got2["c"].astype(str)
    pd.testing.assert_frame_equal(got2, exp2, check_dtype=False)


def _q_tpch(bpd, rank, payload):
    import os, sys

    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "benchmarks"))
    from tpch_data import gen_all
    from tpch_queries import ALL

    t = gen_all(0.01)
    frames = {k: bpd.from_pandas(v) for k, v in t.items()}
    qn = payload["q"]
    res = ALL[qn](bpd, frames)
    return res


@pytest.mark.parametrize("qn", [1, 4, 5, 13, 16, 21, 22])
def test_dist_tpch(qn):
    import os, sys

    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "benchmarks"))
    from tpch_data import gen_all
    from tpch_queries import ALL

    got = run_dist(_q_tpch, {"q": qn}).reset_index(drop=True)
    t = gen_all(0.01)
    from tests.test_tpch import _decat_df

    exp = ALL[qn](pd, {k: _decat_df(v) for k, v in t.items()}).reset_index(drop=True)
    for c in exp.columns:
        if exp[c].dtype == object or str(exp[c].dtype) == "category":
            exp[c] = exp[c].astype(str)
            got[c] = got[c].astype(str)
    cols = list(exp.columns)
    pd.testing.assert_frame_equal(
        got.sort_values(cols).reset_index(drop=True),
        exp.sort_values(cols).reset_index(drop=True),
        check_dtype=False, atol=1e-6, rtol=1e-6)


def _q_window(bpd, rank, payload):
    b = bpd.from_pandas(payload["df"])
    b["gs"] = b.groupby("a")["b"].transform("sum")
    b["cs"] = b.groupby("a")["b"].cumsum()
    b["r"] = b.groupby("a")["b"].rank(method="min")
    return b[["a", "b", "gs", "cs", "r"]]


def test_dist_window():
    df = _df(2000, 21)
    got = run_dist(_q_window, {"df": df}).reset_index(drop=True)
    exp = df.copy()
    exp["gs"] = exp.groupby("a")["b"].transform("sum")
    exp["cs"] = exp.groupby("a")["b"].cumsum()
    exp["r"] = exp.groupby("a")["b"].rank(method="min")
    exp = exp[["a", "b", "gs", "cs", "r"]].reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def _q_rolling(bpd, rank, payload):
    return bpd.from_pandas(payload["df"]).rolling(7).mean()


def test_dist_rolling():
    """Halo exchange must make shard boundaries invisible."""
    df = _df(1000, 11)
    got = run_dist(_q_rolling, {"df": df}).reset_index(drop=True)
    exp = df[["a", "b"]].rolling(7).mean().reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def _q_exists_noneq(bpd, rank, payload):
    from bodo_amd.sql import BodoSQLContext

    bc = BodoSQLContext({"t": bpd.from_pandas(payload["df"])})
    return bc.sql(
        "select a, count(*) as n from t t1 "
        "where exists (select * from t t2 "
        "              where t2.a = t1.a and t2.b <> t1.b) "
        "group by a order by a")


def test_dist_exists_rowid_decorrelation():
    """q21-shape EXISTS with non-equality correlation on 2 ranks (the RowId
    plan node must produce globally unique ids across shards)."""
    df = _df(800, 13)
    got = run_dist(_q_exists_noneq, {"df": df}).reset_index(drop=True)
    has_other = df.groupby("a")["b"].transform("nunique") > 1
    keep = df[has_other]
    exp = keep.groupby("a").size().reset_index(name="n").sort_values(
        "a").reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def _q_iceberg(bpd, rank, payload):
    b = bpd.from_pandas(payload["df"])
    b.to_iceberg(payload["path"])
    return bpd.read_iceberg(payload["path"])


def test_dist_iceberg_roundtrip(tmp_path):
    """2-rank transactional write then parallel read-back."""
    df = _df(600, 17)
    p = str(tmp_path / "ice_tbl")
    got = run_dist(_q_iceberg, {"df": df, "path": p})
    got = got.sort_values(["a", "b"]).reset_index(drop=True)
    got["c"] = got["c"].astype(str)
    exp = df.sort_values(["a", "b"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def _q_kmeans(bpd, rank, payload):
    import numpy as np

    from bodo_amd.ml import KMeans, StandardScaler

    X = payload["X"]
    n = len(X)
    # block-shard the rows like the engine does
    w, r = 2, rank
    base, rem = divmod(n, w)
    start = r * base + min(r, rem)
    stop = start + base + (1 if r < rem else 0)
    shard = X[start:stop]
    sc = StandardScaler().fit(shard)
    km = KMeans(n_clusters=4, random_state=1).fit(shard)
    return {"mean": sc.mean_, "scale": sc.scale_,
            "centers": np.sort(km.cluster_centers_.round(0), axis=0),
            "inertia": km.inertia_}


def test_dist_ml():
    rng = np.random.default_rng(21)
    centers = np.array([[0.0, 0.0], [10.0, 0.0], [0.0, 10.0], [10.0, 10.0]])
    X = np.concatenate([rng.normal(c, 0.5, size=(200, 2)) for c in centers])
    X = X[rng.permutation(len(X))].astype(np.float32)
    got = run_dist(_q_kmeans, {"X": X})
    np.testing.assert_allclose(got["mean"], X.mean(axis=0), rtol=1e-4)
    np.testing.assert_allclose(got["scale"], X.std(axis=0), rtol=1e-3)
    exp_centers = np.sort(centers, axis=0)
    np.testing.assert_allclose(got["centers"], exp_centers, atol=1.0)


def _q_cumsum(bpd, rank, payload):
    return bpd.from_pandas(payload["df"]).b.cumsum()


def test_dist_cumsum():
    """Global cumulative sum must be exact across shard boundaries
    (exscan of shard totals)."""
    df = _df(700, 23)
    got = run_dist(_q_cumsum, {"df": df})
    got = got if isinstance(got, pd.Series) else got.iloc[:, 0]
    exp = df.b.cumsum().reset_index(drop=True)
    pd.testing.assert_series_equal(got.reset_index(drop=True), exp,
                                   check_names=False, check_dtype=False)


def _q_shift(bpd, rank, payload):
    return bpd.from_pandas(payload["df"]).b.shift(3)


def test_dist_shift():
    df = _df(500, 29)
    got = run_dist(_q_shift, {"df": df})
    exp = df.b.shift(3).reset_index(drop=True)
    pd.testing.assert_series_equal(got.reset_index(drop=True), exp,
                                   check_names=False, check_dtype=False)


def _q_setops_window(bpd, rank, payload):
    from bodo_amd.sql import BodoSQLContext

    bc = BodoSQLContext({"t": bpd.from_pandas(payload["df"])})
    return bc.sql(
        "select a, rank() over (partition by a order by b) as rk from t "
        "where b > 0.5 union all "
        "select a, rank() over (partition by a order by b) as rk from t "
        "where b <= 0.5 order by a, rk")


def test_dist_setops_window():
    df = _df(600, 41)
    got = run_dist(_q_setops_window, {"df": df}).reset_index(drop=True)
    hi = df[df.b > 0.5].copy()
    hi["rk"] = hi.groupby("a")["b"].rank(method="min")
    lo = df[df.b <= 0.5].copy()
    lo["rk"] = lo.groupby("a")["b"].rank(method="min")
    exp = pd.concat([hi, lo])[["a", "rk"]].sort_values(
        ["a", "rk"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def _q_rollup(bpd, rank, payload):
    from bodo_amd.sql import BodoSQLContext

    bc = BodoSQLContext({"t": bpd.from_pandas(payload["df"])})
    return bc.sql("select c, sum(b) as s from t "
                  "group by rollup(c) order by c")


def test_dist_rollup():
    df = _df(500, 43)
    got = run_dist(_q_rollup, {"df": df}).reset_index(drop=True)
    per = df.groupby("c")["b"].sum().reset_index()
    per.columns = ["c", "s"]
    tot = pd.DataFrame({"c": [None], "s": [df.b.sum()]})
    exp = pd.concat([per, tot], ignore_index=True)
    got["c"] = got["c"].astype(object).where(lambda x: x.notna(), None)
    key = got["c"].astype(str)
    got = got.iloc[key.argsort().to_numpy()].reset_index(drop=True)
    key = exp["c"].astype(str)
    exp = exp.iloc[key.argsort().to_numpy()].reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def _q_topk(bpd, rank, payload):
    return bpd.from_pandas(payload["df"]).sort_values(
        "b", ascending=False).head(25)


def test_dist_topk():
    df = _df(3000, 47)
    got = run_dist(_q_topk, {"df": df}).reset_index(drop=True)
    exp = df.sort_values("b", ascending=False).head(25).reset_index(drop=True)
    got["c"] = got["c"].astype(str)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def _q_ffill(bpd, rank, payload):
    return bpd.from_pandas(payload["df"]).x.ffill()


def test_dist_ffill():
    """Boundary carry: a leading-null shard must fill from the previous
    rank's last valid value."""
    rng = np.random.default_rng(51)
    x = rng.random(600)
    x[rng.random(600) < 0.4] = np.nan
    df = pd.DataFrame({"x": x})
    got = run_dist(_q_ffill, {"df": df})
    exp = df.x.ffill().reset_index(drop=True)
    pd.testing.assert_series_equal(got.reset_index(drop=True), exp,
                                   check_names=False)


def _q_rtf_join(bpd, rank, payload):
    import bodo_amd.engine.executor as exm
    import bodo_amd.engine.join_filter as jf

    jf.MIN_PROBE_ROWS = 16  # force the runtime filter at test scale
    exm.BROADCAST_JOIN_THRESHOLD = 0  # force the shuffle path
    l = bpd.from_pandas(payload["left"])
    r = bpd.from_pandas(payload["right"])
    return l.merge(r, on="k", how=payload["how"]).sort_values(
        ["k", "v1"])


def test_dist_runtime_join_filter():
    """Bloom + min/max probe pruning must not change any join result."""
    rng = np.random.default_rng(53)
    left = pd.DataFrame({"k": rng.integers(0, 100_000, 5000),
                         "v1": rng.uniform(0, 1, 5000)})
    right = pd.DataFrame({"k": rng.permutation(100_000)[:40],
                          "v2": rng.uniform(0, 1, 40)})
    for how in ("inner", "left"):
        got = run_dist(_q_rtf_join,
                       {"left": left, "right": right, "how": how})
        got = got.reset_index(drop=True)
        exp = left.merge(right, on="k", how=how).sort_values(
            ["k", "v1"]).reset_index(drop=True)
        pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def _q_anyall(bpd, rank, payload):
    return bpd.from_pandas(payload["df"]).groupby("a", as_index=False).agg(
        an=bpd.NamedAgg("d", "any"), al=bpd.NamedAgg("d", "all"),
        sk=bpd.NamedAgg("b", "skew")).sort_values("a")


def test_dist_any_all_skew():
    df = _df(900, 61)
    df["d"] = df.b > 0
    got = run_dist(_q_anyall, {"df": df}).reset_index(drop=True)
    exp = df.groupby("a", as_index=False).agg(
        an=("d", "any"), al=("d", "all"),
        sk=("b", "skew")).sort_values("a").reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False, atol=1e-6)


def _q_dummies(bpd, rank, payload):
    b = bpd.from_pandas(payload["df"])
    return bpd.get_dummies(b.c, prefix="c")


def test_dist_get_dummies():
    """Category values come from a distributed unique() pass, so every
    rank must emit the same column set."""
    df = _df(400, 71)
    got = run_dist(_q_dummies, {"df": df}).reset_index(drop=True)
    exp = pd.get_dummies(df.c, prefix="c").astype(bool).reset_index(drop=True)
    exp = exp[sorted(exp.columns)]
    pd.testing.assert_frame_equal(got[sorted(got.columns)].astype(bool), exp,
                                  check_dtype=False)


def _q_distinct_keep(bpd, rank, payload):
    return bpd.from_pandas(payload["df"]).drop_duplicates(
        subset=["a"], keep=payload["keep"])


def test_dist_distinct_keep_variants():
    df = _df(800, 81)
    for keep in ["first", "last", False]:
        got = run_dist(_q_distinct_keep, {"df": df, "keep": keep})
        exp = df.drop_duplicates(subset=["a"], keep=keep)
        assert sorted(got.b.tolist()) == sorted(exp.b.tolist()), keep


def test_dist_distinct_keep_false_cross_rank_counts():
    """A key duplicated on one rank AND present on the other must still be
    dropped entirely under keep=False (regression: the old pre-shuffle local
    distinct destroyed per-key counts, leaving a lone survivor)."""
    df = pd.DataFrame({"a": [1, 2, 3, 5, 5, 4, 5, 6],
                       "b": [10, 20, 30, 50, 51, 40, 52, 60]})
    got = run_dist(_q_distinct_keep, {"df": df, "keep": False})
    exp = df.drop_duplicates(subset=["a"], keep=False)
    assert sorted(got.b.tolist()) == sorted(exp.b.tolist())
    for keep in ["first", "last"]:
        got = run_dist(_q_distinct_keep, {"df": df, "keep": keep})
        exp = df.drop_duplicates(subset=["a"], keep=keep)
        assert sorted(got.b.tolist()) == sorted(exp.b.tolist()), keep


def _q_offset(bpd, rank, payload):
    from bodo_amd.sql import BodoSQLContext

    bc = BodoSQLContext({"t": bpd.from_pandas(payload["df"])})
    return bc.sql("select b from t order by b limit 9 offset 4")


def test_dist_limit_offset():
    df = _df(300, 91)
    got = run_dist(_q_offset, {"df": df}).reset_index(drop=True)
    exp = df.sort_values("b").reset_index(drop=True).iloc[4:13][
        ["b"]].reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def _q_pack_collectives(bpd, rank, payload):
    """Exercise the packed-buffer allgather_table / gather_table directly
    (strings + dict + nulls + empty-on-one-rank)."""
    import pyarrow as pa
    import torch

    from bodo_amd.core.column import Column
    from bodo_amd.core.table import Table
    from bodo_amd.core import types as bt
    from bodo_amd.parallel import comm

    df = payload["df"]
    half = len(df) // 2
    shard = df.iloc[rank * half:(rank + 1) * half].reset_index(drop=True)
    if payload.get("empty_rank1") and rank == 1:
        shard = shard.iloc[:0]
    t = Table.from_pandas(shard)
    rep = comm.allgather_table(t)
    g = comm.gather_table(t, root=0)
    out = rep.to_pandas()
    out["gathered_rows"] = len(g) if g is not None else -1
    return out


def test_dist_packed_table_collectives():
    rng = np.random.default_rng(5)
    n = 300
    df = pd.DataFrame({
        "a": rng.integers(0, 50, n),
        "f": rng.random(n),
        "s": np.array(["v" + str(i % 37) for i in range(n)], dtype=object),
        "c": pd.Categorical(rng.choice(["x", "y", "z"], n)),
    })
    df.loc[rng.random(n) < 0.1, "f"] = np.nan
    got = run_dist(_q_pack_collectives, {"df": df})
    exp = df.iloc[:300 // 2 * 2].reset_index(drop=True)
    assert got["gathered_rows"].iloc[0] == len(exp)
    got2 = got.drop(columns=["gathered_rows"]).reset_index(drop=True)
    exp2 = exp.copy()
    for c in exp2.columns:
        if exp2[c].dtype == object or str(exp2[c].dtype) == "category":
            exp2[c] = exp2[c].astype(str)
            got2[c] = got2[c].astype(str)
    pd.testing.assert_frame_equal(got2, exp2, check_dtype=False)


def test_dist_packed_collectives_empty_shard():
    rng = np.random.default_rng(6)
    n = 100
    df = pd.DataFrame({"a": rng.integers(0, 5, n),
                       "s": np.array([f"w{i}" for i in range(n)],
                                     dtype=object)})
    got = run_dist(_q_pack_collectives, {"df": df, "empty_rank1": True})
    exp = df.iloc[:50].reset_index(drop=True)
    assert got["gathered_rows"].iloc[0] == len(exp)
    got2 = got.drop(columns=["gathered_rows"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got2, exp, check_dtype=False)


def _q_jit_numpy(bpd, rank, payload):
    import numpy as np

    import bodo_amd

    @bodo_amd.jit
    def f(a, k):
        b = a * 2.0 + k
        sel = b[b > 1.0]
        return sel.sum() + np.sum(np.arange(10000))

    return float(f(payload["a"], payload["k"]))


def test_dist_jit_numpy_world2():
    """@bodo_amd.jit numpy path at world=2: scattered args, distributed
    creation, 1D_Var selection and global reductions must match the serial
    result (reference: distributed_pass parfor semantics)."""
    import numpy as np

    rng = np.random.default_rng(8)
    a = rng.random(40_000)
    got = run_dist(_q_jit_numpy, {"a": a, "k": 0.25})
    b = a * 2.0 + 0.25
    exp = b[b > 1.0].sum() + np.arange(10000).sum()
    assert abs(got - exp) < 1e-6


def test_dist_world4_groupby_join_sort():
    """world=4 on gloo: the packed single-collective shuffle, broadcast
    joins and range-partitioned sort must hold beyond 2 ranks (the driver's
    8-GPU scale run exercises this path over RCCL)."""
    df = _df(4000, 23)
    got = run_dist(_q_groupby, {"df": df}, world=4).reset_index(drop=True)
    exp = df.groupby(["a", "c"], as_index=False).agg(
        s=("b", "sum"), m=("b", "mean"), n=("b", "count"), mx=("b", "max"),
    ).sort_values(["a", "c"]).reset_index(drop=True)
    got["c"] = got["c"].astype(str)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)

    rng = np.random.default_rng(29)
    left = pd.DataFrame({"k": rng.integers(0, 50, 3000),
                         "v1": rng.uniform(0, 1, 3000)})
    right = pd.DataFrame({"k": np.arange(40), "v2": rng.uniform(0, 1, 40)})
    got2 = run_dist(_q_join, {"left": left, "right": right},
                    world=4).reset_index(drop=True)
    exp2 = left.merge(right, on="k", how="inner").sort_values(
        ["k", "v1", "v2"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got2, exp2, check_dtype=False)

    got3 = run_dist(_q_sort, {"df": df}, world=4).reset_index(drop=True)
    exp3 = df.sort_values(["a", "b"], ascending=[True, False]).reset_index(
        drop=True)
    got3["c"] = got3["c"].astype(str)
    pd.testing.assert_frame_equal(got3, exp3, check_dtype=False)


def _q_csv_range(bpd, rank, payload):
    return bpd.read_csv(payload["path"]).groupby("k", as_index=False).agg(
        s=bpd.NamedAgg("v", "sum"), c=bpd.NamedAgg("s", "count")
    ).sort_values("k")


def test_dist_csv_byte_range_split(tmp_path):
    """Byte-range parallel CSV: ranks read disjoint newline-aligned slices
    (reference: _csv_json_reader.cpp byte division)."""
    rng = np.random.default_rng(33)
    n = 320_000
    df = pd.DataFrame({"k": rng.integers(0, 40, n),
                       "v": rng.random(n).round(6),
                       "s": rng.choice(["aa", "bb", "cc"], n)})
    p = str(tmp_path / "big.csv")
    df.to_csv(p, index=False)
    import os
    assert os.path.getsize(p) >= 4 << 20, "grow n: file under range threshold"
    got = run_dist(_q_csv_range, {"path": p}).reset_index(drop=True)
    exp = df.groupby("k", as_index=False).agg(
        s=("v", "sum"), c=("s", "count")).sort_values("k").reset_index(
        drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False, atol=1e-6)


def _q_gbdt(bpd, rank, payload):
    import numpy as np

    from bodo_amd.ml.gbdt import GradientBoostingRegressor

    X, y = payload["X"], payload["y"]
    half = len(X) // 2
    Xs = X[rank * half:(rank + 1) * half]
    ys = y[rank * half:(rank + 1) * half]
    m = GradientBoostingRegressor(n_estimators=25, max_depth=4)
    m.fit(Xs, ys)  # histograms all-reduce across the 2 ranks
    return float(m.score(payload["Xt"], payload["yt"]))


def test_dist_gbdt_histogram_allreduce():
    """2-rank GBDT must match the single-process fit (histogram sums are
    the only cross-rank state; reference: xgboost rabit AllReduce)."""
    import numpy as np

    from bodo_amd.ml.gbdt import GradientBoostingRegressor

    rng = np.random.default_rng(11)
    n = 12000
    X = rng.random((n, 5)).astype(np.float32)
    y = (2 * X[:, 0] - X[:, 1] ** 2
         + 0.05 * rng.standard_normal(n)).astype(np.float32)
    Xt = rng.random((2000, 5)).astype(np.float32)
    yt = (2 * Xt[:, 0] - Xt[:, 1] ** 2).astype(np.float32)
    r2_dist = run_dist(_q_gbdt, {"X": X, "y": y, "Xt": Xt, "yt": yt})
    m = GradientBoostingRegressor(n_estimators=25, max_depth=4)
    m.fit(X[:n // 2 * 2], y[:n // 2 * 2])
    r2_single = m.score(Xt, yt)
    assert abs(r2_dist - r2_single) < 0.02, (r2_dist, r2_single)
    assert r2_dist > 0.9


def _q_sort_float(bpd, rank, payload):
    return bpd.from_pandas(payload["df"]).sort_values(
        "v", ascending=payload["asc"], na_position=payload["na"])


def test_dist_sort_float_key_device_partition():
    """Single-float-key distributed sorts use the sortable-bits device
    range partition (was a host per-row comparison)."""
    rng = np.random.default_rng(71)
    n = 4000
    df = pd.DataFrame({"v": rng.standard_normal(n) * 100,
                       "k": np.arange(n)})
    df.loc[rng.random(n) < 0.05, "v"] = np.nan
    for asc in (True, False):
        for na in ("last", "first"):
            got = run_dist(_q_sort_float,
                           {"df": df, "asc": asc, "na": na}).reset_index(
                drop=True)
            exp = df.sort_values("v", ascending=asc,
                                 na_position=na).reset_index(drop=True)
            pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def _q_incremental_shuffle(bpd, rank, payload):
    import numpy as np

    from bodo_amd.core.table import Table
    from bodo_amd.parallel.incremental import IncrementalShuffle

    rng = np.random.default_rng(100 + rank)
    received = []
    st = IncrementalShuffle(["k"], received.append, threshold=1 << 12,
                            cadence=3)
    # UNEVEN batch counts across ranks: rank 0 sends 11 morsels, rank 1
    # sends 4 — termination consensus must drain both
    import pyarrow as pa

    n_batches = 11 if rank == 0 else 4
    total = 0
    for i in range(n_batches):
        at = pa.table({"k": rng.integers(0, 10, 50),
                       "v": rng.random(50)})
        st.append(Table.from_arrow(at))  # from_arrow: no collective vote
        total += 50
    st.finish()
    import pandas as pd2

    mine = pd2.concat([t.to_pandas() for t in received]) if received \
        else pd2.DataFrame({"k": [], "v": []})
    # every received key must hash to this rank
    from bodo_amd import ops as _ops
    from bodo_amd.core.column import Column

    if len(mine):
        h = _ops.hash_columns([Column.from_numpy(
            mine["k"].to_numpy())]).numpy()
        assert ((h % 2 + 2) % 2 == rank).all()
    counts = sorted(mine["k"].value_counts().to_dict().items())
    return {"rows": len(mine), "counts": counts, "sent": total}


def test_dist_incremental_shuffle_uneven_ranks():
    """Cadenced-round incremental shuffle with uneven morsel counts and
    threshold flushes (reference: IncrementalShuffleState + Ibarrier
    consensus, streaming/_shuffle.h:777)."""
    out = run_dist(_q_incremental_shuffle, {})
    assert out["rows"] > 0


def _q_list_shuffle(bpd, rank, payload):
    b = bpd.from_pandas(payload["df"])
    return b.groupby("k", as_index=False).agg(
        c=bpd.NamedAgg("l", "count")).sort_values("k")


def _q_list_explode_shuffle(bpd, rank, payload):
    b = bpd.from_pandas(payload["df"])
    return b.explode("l").groupby("k", as_index=False).agg(
        s=bpd.NamedAgg("l", "sum")).sort_values("k")


def test_dist_list_columns_shuffle():
    """LIST columns survive distributed shuffles (recursive varlen
    exchange; reference: nested-array shuffle in _shuffle.cpp)."""
    rng = np.random.default_rng(91)
    n = 600
    df = pd.DataFrame({"k": rng.integers(0, 12, n)})
    df["l"] = pd.Series([list(map(int, rng.integers(0, 9, rng.integers(0, 5))))
                         for _ in range(n)], dtype=object)
    # window/groupby path shuffles the frame incl. the list payload
    got = run_dist(_q_list_explode_shuffle, {"df": df}).reset_index(drop=True)
    exp = df.explode("l").groupby("k", as_index=False).agg(
        s=("l", "sum")).sort_values("k").reset_index(drop=True)
    assert got["k"].tolist() == exp["k"].tolist()
    assert [float(v) for v in got["s"]] == [float(v) for v in exp["s"]]


def _q_list_allgather(bpd, rank, payload):
    from bodo_amd.core.table import Table
    from bodo_amd.parallel import comm

    df = payload["df"]
    half = len(df) // 2
    t = Table.from_pandas(df.iloc[rank * half:(rank + 1) * half]
                          .reset_index(drop=True))
    rep = comm.allgather_table(t)
    return rep.to_pandas()


def test_dist_list_allgather():
    rng = np.random.default_rng(93)
    n = 100
    df = pd.DataFrame({"k": rng.integers(0, 5, n)})
    df["l"] = pd.Series([list(map(int, rng.integers(0, 9, rng.integers(0, 4))))
                         for _ in range(n)], dtype=object)
    got = run_dist(_q_list_allgather, {"df": df})
    exp = df.iloc[:n // 2 * 2].reset_index(drop=True)
    assert got["k"].tolist() == exp["k"].tolist()
    assert [list(v) for v in got["l"]] == [list(v) for v in exp["l"]]


def _q_kitchen_sink_shuffle(bpd, rank, payload):
    from bodo_amd.core.table import Table
    from bodo_amd.parallel import comm
    import torch

    df = payload["df"]
    if rank == 1 and payload.get("empty1"):
        df = df.iloc[:0]
    t = Table.from_pandas(df.reset_index(drop=True))
    n = len(t)
    part = torch.arange(n) % 2 if n else torch.zeros(0, dtype=torch.int64)
    out = comm.shuffle_table(t, part)
    back = out.to_pandas()
    total = sum(comm.allgather_obj(len(back)))
    return {"total": total, "cols": list(back.columns)}


def test_dist_shuffle_all_column_kinds():
    """Packed shuffle across every column kind (ints, floats with NaN,
    masked ints, bool, decimal, date, timestamp, dict, plain string, list)
    incl. a zero-row rank."""
    from decimal import Decimal

    rng = np.random.default_rng(101)
    n = 200
    df = pd.DataFrame({
        "i64": rng.integers(-5, 5, n),
        "f64": np.where(rng.random(n) < 0.2, np.nan, rng.random(n)),
        "b": rng.integers(0, 2, n).astype(bool),
        "s_plain": np.array([f"v{i}" for i in range(n)], dtype=object),
        "s_dict": rng.choice(["aa", "bb"], n),
        "ts": pd.to_datetime(1_600_000_000_000_000_000
                             + rng.integers(0, 10**15, n)),
    })
    df["mask_i"] = pd.array(rng.integers(0, 9, n), dtype="Int64")
    df.loc[rng.random(n) < 0.2, "mask_i"] = pd.NA
    df["dec"] = pd.Series([Decimal(int(v)) / 100
                           for v in rng.integers(-10**4, 10**4, n)])
    df["lst"] = pd.Series([list(map(int, rng.integers(0, 5,
                                                      rng.integers(0, 3))))
                           for _ in range(n)], dtype=object)
    for empty1 in (False, True):
        out = run_dist(_q_kitchen_sink_shuffle,
                       {"df": df, "empty1": empty1})
        exp_total = 2 * len(df) if not empty1 else len(df)
        assert out["total"] == exp_total, (empty1, out)


def _q_scalar_reduces(bpd, rank, payload):
    b = bpd.from_pandas(payload["df"])
    return {
        "median": float(b["b"].median()),
        "kurt": float(b["b"].kurt()),
        "skew": float(b["b"].skew()),
        "sem": float(b["b"].sem()),
        "min_d": bpd.to_pandas_scalar(b["d"].min())
        if hasattr(bpd, "to_pandas_scalar") else b["d"].min().value,
        "max_c": str(b["c"].max()),
    }


def test_dist_scalar_reduces():
    """Whole-column reductions added in round 2 (median/kurt/skew/sem, plus
    value semantics for dict-string max and timestamp min) across 2 ranks."""
    rng = np.random.default_rng(7)
    n = 101
    df = pd.DataFrame({
        "b": np.where(rng.random(n) < 0.2, np.nan, rng.random(n) * 9 - 3),
        "c": rng.choice(["xq", "yy", "zx", "wv"], n),
        "d": pd.to_datetime(1.6e18 + rng.integers(0, 9e16, n)),
    })
    got = run_dist(_q_scalar_reduces, {"df": df})
    assert abs(got["median"] - df["b"].median()) < 1e-9
    assert abs(got["kurt"] - df["b"].kurt()) < 1e-9
    assert abs(got["skew"] - df["b"].skew()) < 1e-9
    assert abs(got["sem"] - df["b"].sem()) < 1e-9
    assert pd.Timestamp(got["min_d"]) == df["d"].min()
    assert got["max_c"] == df["c"].max()


def _q_round2_sql(bpd, rank, payload):
    from bodo_amd.sql import BodoSQLContext

    bc = BodoSQLContext({"t": payload["df"], "o": payload["o"]})
    a = bc.sql(
        "with s as (select g, sum(y) as sy from t group by g) "
        "select s.g, s.sy, o.w from s join o on s.g = o.g").to_pandas()
    b = bc.sql(
        "select g, min(y) over (partition by g order by x) as mn "
        "from t").to_pandas()
    c = bc.sql(
        "select g, x, (select w from o where o.g = t.g) as w "
        "from t").to_pandas()
    return {"cte": a.sort_values("g").reset_index(drop=True),
            "win_rows": len(b), "subq_nulls": int(c["w"].isna().sum()),
            "subq_rows": len(c)}


def test_dist_round2_sql_features():
    """2-rank collective symmetry of the round-2 SQL plan shapes: CTE with
    shared subtree, running window frame, SELECT-list decorrelated
    subquery."""
    rng = np.random.default_rng(3)
    n = 400
    df = pd.DataFrame({"g": rng.choice(["a", "b", "c", "d", "e"], n),
                       "x": rng.integers(0, 50, n),
                       "y": rng.random(n) * 100})
    o = pd.DataFrame({"g": ["a", "b", "c", "d"], "w": [1.0, 2.0, 3.0, 4.0]})
    got = run_dist(_q_round2_sql, {"df": df, "o": o})
    exp_cte = df.groupby("g", as_index=False).agg(sy=("y", "sum")) \
        .merge(o, on="g").sort_values("g").reset_index(drop=True)
    g = got["cte"]
    assert np.allclose(g["sy"], exp_cte["sy"]) and np.allclose(
        g["w"], exp_cte["w"])
    assert got["win_rows"] == n
    assert got["subq_rows"] == n
    assert got["subq_nulls"] == int((df["g"] == "e").sum())


def _q_struct_shuffle(bpd, rank, payload):
    b = bpd.from_pandas(payload["df"])
    out = b.sort_values("k").to_pandas()  # range partition => shuffle
    return out.reset_index(drop=True)


def test_dist_struct_shuffle():
    """STRUCT columns survive the packed shuffle (2 ranks): per-field
    exchanges incl. a string field and the struct validity mask."""
    rng = np.random.default_rng(11)
    n = 60
    df = pd.DataFrame({
        "k": rng.permutation(n),
        "st": [None if i % 7 == 0 else
               {"x": int(i), "y": f"s{i}" if i % 3 else None}
               for i in range(n)],
    })
    got = run_dist(_q_struct_shuffle, {"df": df})
    exp = df.sort_values("k").reset_index(drop=True)
    assert got["k"].tolist() == exp["k"].tolist()
    assert got["st"].tolist() == exp["st"].tolist()


def _q_ml_metrics(bpd, rank, payload):
    import numpy as np

    from bodo_amd import ml

    X = payload["X"]
    y = payload["y"]
    # each rank takes its block (estimator inputs are per-rank shards)
    w = 2
    n = len(X)
    lo, hi = rank * n // w, (rank + 1) * n // w
    Xl, yl = X[lo:hi], y[lo:hi]
    mm = ml.MinMaxScaler().fit(Xl)
    acc = ml.accuracy_score(yl, np.zeros_like(yl))
    mse = ml.mean_squared_error(yl, np.zeros_like(yl, dtype=float))
    return {"min": mm.data_min_.tolist(), "max": mm.data_max_.tolist(),
            "acc": acc, "mse": mse}


def test_dist_ml_metrics_and_scaler():
    """MinMaxScaler min/max and metrics all-reduce across 2 ranks to the
    global values."""
    rng = np.random.default_rng(5)
    X = rng.random((101, 3)) * 4 - 2
    y = rng.integers(0, 2, 101)
    got = run_dist(_q_ml_metrics, {"X": X, "y": y})
    assert np.allclose(got["min"], X.min(axis=0))
    assert np.allclose(got["max"], X.max(axis=0))
    assert abs(got["acc"] - (y == 0).mean()) < 1e-12
    assert abs(got["mse"] - (y.astype(float) ** 2).mean()) < 1e-12


def _q_grouped_value_aggs(bpd, rank, payload):
    from bodo_amd.sql import BodoSQLContext

    bc = BodoSQLContext({"t": payload["df"]})
    out = bc.sql(
        "select c, min(c) as mc, max(d) as xd, approx_count_distinct(a) "
        "as ad, mode(a) as mo from t group by c order by c").to_pandas()
    return out.reset_index(drop=True)


def test_dist_grouped_value_aggs():
    """2-rank: single-phase grouped value aggs added in round 2 (dict-key
    min/max, approx_count_distinct, mode) keep collective symmetry."""
    rng = np.random.default_rng(9)
    n = 300
    df = pd.DataFrame({"a": rng.integers(0, 7, n),
                       "c": rng.choice(["x", "y", "z"], n),
                       "d": pd.to_datetime(1.6e18 + rng.integers(0, 1e16, n))})
    got = run_dist(_q_grouped_value_aggs, {"df": df})
    exp = df.groupby("c").agg(
        xd=("d", "max"), ad=("a", "nunique"),
        mo=("a", lambda s: s.mode().iloc[0])).reset_index()
    assert got["c"].astype(str).tolist() == exp["c"].tolist()
    assert got["mc"].astype(str).tolist() == exp["c"].tolist()
    assert [pd.Timestamp(v) for v in got["xd"]] == list(exp["xd"])
    assert got["ad"].tolist() == exp["ad"].tolist()
    assert got["mo"].tolist() == exp["mo"].tolist()


def test_dist_struct_shuffle_4rank():
    """4-rank struct shuffle incl. an empty-ish rank distribution."""
    rng = np.random.default_rng(13)
    n = 37  # odd count over 4 ranks -> uneven blocks
    df = pd.DataFrame({
        "k": rng.permutation(n),
        "st": [None if i % 5 == 0 else {"x": int(i), "y": f"v{i}"}
               for i in range(n)],
    })
    got = run_dist(_q_struct_shuffle, {"df": df}, world=4)
    exp = df.sort_values("k").reset_index(drop=True)
    assert got["k"].tolist() == exp["k"].tolist()
    assert got["st"].tolist() == exp["st"].tolist()


def _q_array_agg(bpd, rank, payload):
    from bodo_amd.sql import BodoSQLContext

    bc = BodoSQLContext({"t": payload["df"]})
    out = bc.sql("select g, array_agg(x) as ax from t group by g "
                 "order by g").to_pandas()
    return {"lists": [sorted(v) for v in out["ax"]],
            "g": out["g"].astype(str).tolist()}


def test_dist_array_agg():
    rng = np.random.default_rng(17)
    n = 120
    df = pd.DataFrame({"g": rng.choice(["a", "b", "c"], n),
                       "x": rng.integers(0, 50, n)})
    got = run_dist(_q_array_agg, {"df": df})
    exp = df.groupby("g")["x"].apply(lambda s: sorted(s)).to_dict()
    assert got["g"] == sorted(exp)
    for g, lst in zip(got["g"], got["lists"]):
        assert lst == exp[g]


def _q_global_listagg(bpd, rank, payload):
    from bodo_amd.sql import BodoSQLContext

    bc = BodoSQLContext({"t": payload["df"]})
    out = bc.sql("select listagg(s, ',') as l, "
                 "percentile_cont(0.5) within group (order by v) as p "
                 "from t").to_pandas()
    return {"l": out["l"][0], "p": float(out["p"][0])}


def test_dist_global_listagg_percentile():
    """2-rank callable reduces: LISTAGG / PERCENTILE_CONT combine the
    gathered values in rank order."""
    df = pd.DataFrame({"s": list("abcdef"), "v": [1.0, 5, 2, 4, 3, 6]})
    got = run_dist(_q_global_listagg, {"df": df})
    assert got["l"] == "a,b,c,d,e,f"
    assert got["p"] == 3.5
