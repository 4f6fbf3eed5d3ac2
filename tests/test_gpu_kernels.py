"""GPU kernel numerics tests: each HIP kernel vs the CPU (pandas/numpy)
reference of the same op.  All marked gpu; run via
``pytest tests -m gpu`` on an MI355X box."""

import numpy as np
import pandas as pd
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(autouse=True)
def _cuda_required():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")


def _to_gpu(col):
    return col.to_device("cuda")


def make_cols(n=10000, seed=0):
    from bodo_amd.core.column import Column

    rng = np.random.default_rng(seed)
    cols = {
        "i64": Column.from_numpy(rng.integers(-1000, 1000, n)),
        "i32": Column.from_numpy(rng.integers(-100, 100, n).astype(np.int32)),
        "f64": Column.from_numpy(rng.uniform(-1, 1, n)),
        "b": Column.from_numpy(rng.integers(0, 2, n).astype(bool)),
        "s": Column.from_numpy(rng.choice(["aa", "bb", "cc", "dddd"], n)),
    }
    import pyarrow as pa

    dict_arr = pa.array(rng.choice(["u", "vv", "www"], n)).dictionary_encode()
    cols["d"] = Column.from_arrow(dict_arr)
    return cols


def test_hash_matches_cpu():
    from bodo_amd import ops

    cols = make_cols()
    for name, c in cols.items():
        h_cpu = ops.hash_columns([c])
        h_gpu = ops.hash_columns([_to_gpu(c)]).cpu()
        assert torch.equal(h_cpu, h_gpu), f"hash mismatch for {name}"
    # multi-column combine
    combo = [cols["i64"], cols["s"], cols["f64"]]
    h_cpu = ops.hash_columns(combo)
    h_gpu = ops.hash_columns([_to_gpu(c) for c in combo]).cpu()
    assert torch.equal(h_cpu, h_gpu)


def test_hash_nulls_match():
    from bodo_amd import ops
    from bodo_amd.core.column import Column

    rng = np.random.default_rng(1)
    data = rng.integers(0, 50, 5000)
    mask = rng.random(5000) > 0.2
    c = Column.from_numpy(data, mask=mask)
    h_cpu = ops.hash_columns([c])
    h_gpu = ops.hash_columns([_to_gpu(c)]).cpu()
    assert torch.equal(h_cpu, h_gpu)


def test_dt_field_gpu():
    from bodo_amd.core.column import Column
    from bodo_amd.ops import gpu
    from bodo_amd.ops.evaluate import _dt_field_cpu

    rng = np.random.default_rng(2)
    ts = (pd.Timestamp("1995-01-01").value
          + rng.integers(0, 40 * 365 * 86400 * 10**9, 20000))
    c = Column.from_numpy(ts.view("datetime64[ns]"))
    for fld in ("year", "month", "day", "hour", "minute", "second",
                "dayofweek", "dayofyear", "quarter", "date", "normalize"):
        exp = _dt_field_cpu(c, fld)
        got = gpu.dt_field(_to_gpu(c), fld)
        assert torch.equal(exp.data, got.data.cpu()), fld


def test_gather_string_gpu():
    from bodo_amd.core.column import Column
    from bodo_amd import ops

    rng = np.random.default_rng(3)
    vals = np.array(["", "a", "xyz", "hello world", "qq"], dtype=object)
    c = Column.from_numpy(vals[rng.integers(0, 5, 3000)])
    idx = torch.from_numpy(rng.integers(0, 3000, 500))
    exp = ops.gather(c, idx)
    got = ops.gather(_to_gpu(c), idx.cuda())
    assert exp.to_pandas().tolist() == got.to_pandas().tolist()


def _gb_frames(n=20000, seed=4, with_nulls=False):
    rng = np.random.default_rng(seed)
    df = pd.DataFrame({
        "k1": rng.integers(0, 50, n),
        "k2": rng.choice(["x", "y", "z"], n),
        "v1": rng.uniform(-1, 1, n),
        "v2": rng.integers(0, 100, n),
    })
    if with_nulls:
        df.loc[rng.random(n) < 0.1, "v1"] = np.nan
    return df


@pytest.mark.parametrize("funcs", [
    {"s": ("v1", "sum"), "c": ("v1", "count"), "m": ("v1", "mean")},
    {"mi": ("v1", "min"), "mx": ("v1", "max"), "sz": ("v1", "size")},
    {"is": ("v2", "sum"), "imi": ("v2", "min"), "imx": ("v2", "max")},
    {"f": ("v2", "first"), "l": ("v2", "last")},
    {"md": ("v1", "median"), "nu": ("v2", "nunique")},
    {"vr": ("v1", "var"), "sd": ("v1", "std")},
])
def test_groupby_gpu_vs_pandas(funcs):
    from bodo_amd.core.table import Table
    from bodo_amd.ops import gpu

    df = _gb_frames(with_nulls=True)
    t = Table.from_pandas(df, device="cuda")
    aggs = [(out, src, f) for out, (src, f) in funcs.items()]
    got = gpu.groupby_local(t, ["k1", "k2"], aggs).to_pandas()
    got["k2"] = got["k2"].astype(str)
    exp = df.groupby(["k1", "k2"], sort=False, dropna=True).agg(
        **{out: (src, f) for out, (src, f) in funcs.items()}).reset_index()
    got = got.sort_values(["k1", "k2"]).reset_index(drop=True)
    exp = exp.sort_values(["k1", "k2"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False,
                                  check_exact=False, rtol=1e-9)


def test_groupby_gpu_high_cardinality():
    from bodo_amd.core.table import Table
    from bodo_amd.ops import gpu

    rng = np.random.default_rng(5)
    n = 200_000
    df = pd.DataFrame({"k": rng.integers(0, n // 2, n),
                       "v": rng.uniform(0, 1, n)})
    t = Table.from_pandas(df, device="cuda")
    got = gpu.groupby_local(t, ["k"], [("s", "v", "sum"), ("n", "v", "count")]) \
        .to_pandas().sort_values("k").reset_index(drop=True)
    exp = df.groupby("k", sort=True).agg(s=("v", "sum"), n=("v", "count")) \
        .reset_index()
    pd.testing.assert_frame_equal(got, exp, check_dtype=False,
                                  check_exact=False, rtol=1e-9)


@pytest.mark.parametrize("how", ["inner", "left", "semi", "anti", "outer"])
def test_join_gpu_vs_pandas(how):
    from bodo_amd.core.table import Table
    from bodo_amd.ops import gpu

    rng = np.random.default_rng(6)
    left = pd.DataFrame({"k": rng.integers(0, 40, 5000),
                         "v1": rng.uniform(0, 1, 5000)})
    right = pd.DataFrame({"k": rng.integers(0, 60, 30),
                          "v2": rng.uniform(0, 1, 30)}).drop_duplicates("k")
    lt = Table.from_pandas(left, device="cuda")
    rt = Table.from_pandas(right, device="cuda")
    got = gpu.join_local(lt, rt, ["k"], ["k"], how).to_pandas()
    if how in ("semi", "anti"):
        keys = set(right["k"])
        exp = left[left["k"].isin(keys)] if how == "semi" else left[~left["k"].isin(keys)]
        got = got.sort_values(["k", "v1"]).reset_index(drop=True)
        exp = exp.sort_values(["k", "v1"]).reset_index(drop=True)
        pd.testing.assert_frame_equal(got, exp, check_dtype=False)
        return
    exp = left.merge(right, on="k", how=how)
    sort_cols = ["k", "v1", "v2"]
    got = got.sort_values(sort_cols, na_position="last").reset_index(drop=True)
    exp = exp.sort_values(sort_cols, na_position="last").reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_distinct_gpu():
    from bodo_amd.core.table import Table
    from bodo_amd.ops import gpu

    rng = np.random.default_rng(7)
    df = pd.DataFrame({"a": rng.integers(0, 30, 2000),
                       "b": rng.integers(0, 3, 2000)})
    t = Table.from_pandas(df, device="cuda")
    got = gpu.distinct_local(t, ["a"]).to_pandas().reset_index(drop=True)
    exp = df.drop_duplicates(subset=["a"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_taxi_q1_gpu_end_to_end():
    import bodo_amd.config as cfg

    cfg.DEVICE = "cuda"
    import bodo_amd.pandas as bpd
    from tests.test_queries import make_taxi, nyc_taxi_q1

    trips, weather = make_taxi(50000, 17)
    got = nyc_taxi_q1(bpd, bpd.from_pandas(trips), bpd.from_pandas(weather))
    got = got.to_pandas().reset_index(drop=True)
    exp = nyc_taxi_q1(pd, trips.copy(), weather.copy()).reset_index(drop=True)
    got["time_bucket"] = got["time_bucket"].astype(str)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_tpch_q1_gpu_end_to_end():
    import bodo_amd.config as cfg

    cfg.DEVICE = "cuda"
    import bodo_amd.pandas as bpd
    from tests.test_queries import make_lineitem, tpch_q1

    li = make_lineitem(60000, 19)
    got = tpch_q1(bpd, bpd.from_pandas(li)).to_pandas().reset_index(drop=True)
    exp = tpch_q1(pd, li.copy()).reset_index(drop=True)
    for c in ("L_RETURNFLAG", "L_LINESTATUS"):
        got[c] = got[c].astype(str)
        exp[c] = exp[c].astype(str)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_gpu_parquet_decode(tmp_path):
    """On-GPU parquet decode (PLAIN + RLE_DICTIONARY, uncompressed) vs
    pyarrow host read."""
    import pyarrow.parquet as pq
    import pyarrow as pa
    from bodo_amd.io import parquet_gpu
    from bodo_amd.engine.executor import ExecutionContext

    rng = np.random.default_rng(9)
    n = 200_000
    df = pd.DataFrame({
        "i": rng.integers(0, 1000, n),
        "f": rng.uniform(-1, 1, n),
        "c": rng.choice(["aa", "bb", "cc", "dd"], n),
        "t": pd.to_datetime(pd.Timestamp("2020-01-01").value
                            + rng.integers(0, 10**15, n)),
    })
    p = str(tmp_path / "g.parquet")
    pq.write_table(pa.Table.from_pandas(df, preserve_index=False), p,
                   compression="NONE", use_dictionary=["c"])
    ctx = ExecutionContext("cuda")
    t = parquet_gpu.read_shard_gpu(p, None, ctx)
    assert t is not None, "GPU decode fell back"
    got = t.to_pandas()
    got["c"] = got["c"].astype(str)
    pd.testing.assert_frame_equal(got, df, check_dtype=False)


def test_gpu_parquet_dict_int_decode(tmp_path):
    import pyarrow.parquet as pq
    import pyarrow as pa
    from bodo_amd.io import parquet_gpu
    from bodo_amd.engine.executor import ExecutionContext

    rng = np.random.default_rng(11)
    n = 100_000
    df = pd.DataFrame({"k": rng.integers(0, 50, n),
                       "v": rng.uniform(0, 1, n).round(3)})
    p = str(tmp_path / "d.parquet")
    pq.write_table(pa.Table.from_pandas(df, preserve_index=False), p,
                   compression="NONE", use_dictionary=True)
    ctx = ExecutionContext("cuda")
    t = parquet_gpu.read_shard_gpu(p, None, ctx)
    assert t is not None
    pd.testing.assert_frame_equal(t.to_pandas(), df, check_dtype=False)


@pytest.mark.gpu
def test_gpu_cumulative_and_shift():
    import bodo_amd.config as cfg

    cfg.DEVICE = "cuda"
    import bodo_amd.pandas as bpd

    rng = np.random.default_rng(11)
    x = rng.random(200_000)
    x[rng.random(200_000) < 0.05] = np.nan
    df = pd.DataFrame({"x": x})
    b = bpd.from_pandas(df)
    for f in ["cumsum", "cummin", "cummax"]:
        got = getattr(b.x, f)().to_pandas()
        exp = getattr(df.x, f)().reset_index(drop=True)
        pd.testing.assert_series_equal(got, exp, check_names=False,
                                       check_dtype=False)
    pd.testing.assert_series_equal(
        b.x.shift(5).to_pandas(), df.x.shift(5).reset_index(drop=True),
        check_names=False, check_dtype=False)


@pytest.mark.gpu
def test_gpu_window_sql():
    import bodo_amd.config as cfg
    from bodo_amd.sql import BodoSQLContext

    cfg.DEVICE = "cuda"
    rng = np.random.default_rng(12)
    df = pd.DataFrame({"k": rng.integers(0, 50, 100_000),
                       "v": rng.random(100_000),
                       "o": rng.permutation(100_000)})
    bc = BodoSQLContext({"t": df})
    got = bc.sql("select k, sum(v) over (partition by k) as tot, "
                 "row_number() over (partition by k order by o) as rn "
                 "from t order by k, rn limit 1000").to_pandas()
    exp = df.copy()
    exp["tot"] = exp.groupby("k")["v"].transform("sum")
    exp["rn"] = exp.sort_values("o").groupby("k").cumcount() + 1
    exp = exp.sort_values(["k", "rn"]).reset_index(drop=True)[
        ["k", "tot", "rn"]].head(1000)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def _ctx(device):
    import torch

    class Ctx:
        world, rank = 1, 0

    Ctx.device = torch.device(device)
    return Ctx()


@pytest.mark.gpu
def test_parquet_gpu_snappy_dict_nulls():
    """GPU decode of snappy-compressed dictionary pages with nulls: the
    definition-level RLE expands to a device validity mask and dense codes
    scatter into place."""
    import tempfile

    import pyarrow as pa
    import pyarrow.parquet as pq

    from bodo_amd.io import parquet_gpu as g

    rng = np.random.default_rng(1)
    n = 200_000
    a = rng.integers(0, 50, n).astype("float64")
    a[rng.random(n) < 0.1] = np.nan
    s = rng.choice(["aa", "bb", "cc", "dd"], n)
    df = pd.DataFrame({"a": a, "s": s, "v": rng.random(n)})
    with tempfile.TemporaryDirectory() as d:
        fp = d + "/t.parquet"
        pq.write_table(pa.Table.from_pandas(df), fp, compression="snappy")
        t = g._read_row_group_gpu(fp, 0, None, _ctx("cuda"))
        assert t is not None, "GPU decode fell back"
        out = t.to_pandas()
        out["s"] = out["s"].astype(str)
        pd.testing.assert_frame_equal(out, df, check_dtype=False)


@pytest.mark.gpu
def test_parquet_gpu_snappy_e2e():
    """read_parquet end-to-end over snappy files on device."""
    import tempfile

    import bodo_amd.config as cfg

    cfg.DEVICE = "cuda"
    import bodo_amd.pandas as bpd

    rng = np.random.default_rng(2)
    df = pd.DataFrame({"k": rng.integers(0, 20, 300_000),
                       "v": rng.random(300_000)})
    with tempfile.TemporaryDirectory() as d:
        fp = d + "/t.parquet"
        df.to_parquet(fp, compression="snappy")
        b = bpd.read_parquet(fp)
        got = b.groupby("k", as_index=False).agg(
            s=bpd.NamedAgg("v", "sum")).sort_values("k").to_pandas()
        exp = df.groupby("k", as_index=False).agg(
            s=("v", "sum")).sort_values("k").reset_index(drop=True)
        pd.testing.assert_frame_equal(got.reset_index(drop=True), exp,
                                      check_dtype=False)


@pytest.mark.gpu
def test_gpu_groupby_any_all_skew():
    import bodo_amd.config as cfg

    cfg.DEVICE = "cuda"
    import bodo_amd.pandas as bpd

    rng = np.random.default_rng(34)
    df = pd.DataFrame({"k": rng.integers(0, 20, 100_000),
                       "f": rng.random(100_000) > 0.3,
                       "v": rng.random(100_000) * 5})
    b = bpd.from_pandas(df)
    got = b.groupby("k", as_index=False).agg(
        an=bpd.NamedAgg("f", "any"), al=bpd.NamedAgg("f", "all"),
        sk=bpd.NamedAgg("v", "skew")).to_pandas().sort_values("k")
    got = got.reset_index(drop=True)
    exp = df.groupby("k", as_index=False).agg(
        an=("f", "any"), al=("f", "all"),
        sk=("v", "skew")).sort_values("k").reset_index(drop=True)
    # raw-moment skew carries ~1e-8 cancellation error vs pandas' two-pass
    pd.testing.assert_frame_equal(got, exp, check_dtype=False, atol=1e-6)


@pytest.mark.gpu
def test_gpu_distinct_keep_variants():
    import bodo_amd.config as cfg

    cfg.DEVICE = "cuda"
    import bodo_amd.pandas as bpd

    rng = np.random.default_rng(35)
    df = pd.DataFrame({"a": rng.integers(0, 500, 50_000),
                       "b": np.arange(50_000)})
    b = bpd.from_pandas(df)
    for keep in ["first", "last", False]:
        got = b.drop_duplicates(subset=["a"], keep=keep).to_pandas()
        exp = df.drop_duplicates(subset=["a"], keep=keep)
        assert sorted(got.b.tolist()) == sorted(exp.b.tolist()), keep


@pytest.mark.gpu
def test_parquet_batched_decode_layouts(tmp_path):
    """The page-parallel batched device decode must handle every common
    layout without falling back: snappy/uncompressed x plain/dict x
    nulls/no-nulls x fixed/strings, multi-page chunks."""
    import pyarrow as pa
    import pyarrow.parquet as pq

    from bodo_amd.io import parquet_gpu as g

    rng = np.random.default_rng(21)
    n = 400_000
    base = {
        "i64": rng.integers(-10**9, 10**9, n).astype("int64"),
        "i32": rng.integers(0, 100, n).astype("int32"),
        "f64": rng.random(n) * 1e6,
        "f32": rng.random(n).astype("float32"),
        "s_plain": np.array(
            ["s" + str(v) for v in rng.integers(0, 10**9, n)], dtype=object),
        "s_dict": rng.choice(["alpha", "beta", "gamma", "delta"], n),
        "ts": (np.datetime64("2023-01-01")
               + rng.integers(0, 10**9, n).astype("timedelta64[s]")),
    }
    for comp in ("snappy", "none"):
        for with_nulls in (False, True):
            df = pd.DataFrame({k: v.copy() for k, v in base.items()})
            if with_nulls:
                df.loc[rng.random(n) < 0.08, "f64"] = np.nan
                df.loc[rng.random(n) < 0.08, "s_plain"] = None
                df.loc[rng.random(n) < 0.08, "s_dict"] = None
            fp = str(tmp_path / f"t_{comp}_{with_nulls}.parquet")
            pq.write_table(
                pa.Table.from_pandas(df), fp, compression=comp,
                use_dictionary=["s_dict", "i32"], data_page_size=128 * 1024)
            before = g.STATS["slow"]
            t = g._read_row_group_gpu(fp, 0, None, _ctx("cuda"))
            assert t is not None, (comp, with_nulls)
            assert g.STATS["slow"] == before, \
                f"fallback used for {comp} nulls={with_nulls}"
            out = t.to_pandas()
            for c in out.columns:
                if out[c].dtype.name == "category":
                    out[c] = out[c].astype(object)
            exp = df.copy()
            pd.testing.assert_frame_equal(out, exp, check_dtype=False)


@pytest.mark.gpu
def test_parquet_batched_decode_multi_row_group(tmp_path):
    import pyarrow as pa
    import pyarrow.parquet as pq

    import bodo_amd.config as cfg
    from bodo_amd.io import parquet_gpu as g

    cfg.DEVICE = "cuda"
    import bodo_amd.pandas as bpd

    rng = np.random.default_rng(22)
    n = 1_000_000
    df = pd.DataFrame({"k": rng.integers(0, 500, n),
                       "v": rng.random(n),
                       "s": rng.choice(["a", "bb", "ccc"], n)})
    fp = str(tmp_path / "big.parquet")
    pq.write_table(pa.Table.from_pandas(df), fp, compression="snappy",
                   row_group_size=200_000)
    b = bpd.read_parquet(fp)
    got = b.groupby("k", as_index=False).agg(
        s=bpd.NamedAgg("v", "sum"), c=bpd.NamedAgg("s", "count"))
    got = got.to_pandas().sort_values("k").reset_index(drop=True)
    exp = df.groupby("k", as_index=False).agg(
        s=("v", "sum"), c=("s", "count")).sort_values("k").reset_index(
        drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


@pytest.mark.gpu
def test_gather_multi_fused():
    """Fused one-launch multi-column gather == per-column gathers."""
    import bodo_amd_kernels as K
    from bodo_amd import ops
    from bodo_amd.core.table import Table

    rng = np.random.default_rng(44)
    n = 100_000
    df = pd.DataFrame({
        "a": rng.integers(-10**9, 10**9, n),
        "b": rng.random(n),
        "c": rng.integers(0, 100, n).astype(np.int32),
        "f": rng.random(n).astype(np.float32),
        "s": rng.choice(["x", "yy", "zzz"], n),
        "bo": rng.integers(0, 2, n).astype(bool),
    })
    df.loc[rng.random(n) < 0.1, "b"] = np.nan
    t = Table.from_pandas(df, device="cuda")
    idx = torch.from_numpy(rng.integers(0, n, 30_000)).cuda()
    fused = ops.take_table(t, idx)
    ref = Table(t.names, [ops.gather(c, idx) for c in t.columns],
                int(idx.numel()))
    pd.testing.assert_frame_equal(fused.to_pandas(), ref.to_pandas())
