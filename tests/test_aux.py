"""Auxiliary subsystem tests: tracing, query profile, user logging, plan
cache (reference: bodo/utils/tracing.pyx, _query_profile_collector,
user_logging, sql_plan_cache)."""

import json
import os

import numpy as np
import pandas as pd

import bodo_amd.pandas as bpd


def test_tracing(tmp_path):
    from bodo_amd.utils import tracing

    tracing.start_tracing()
    df = pd.DataFrame({"a": np.arange(100), "b": np.arange(100) * 0.5})
    b = bpd.from_pandas(df)
    b.groupby("a", as_index=False).agg(s=bpd.NamedAgg("b", "sum")).to_pandas()
    with tracing.Event("custom", foo=1):
        pass
    agg = tracing.aggregate_events()
    assert any(e["name"].startswith("exec.") for e in agg)
    p = str(tmp_path / "trace.json")
    tracing.dump(p)
    data = json.load(open(p))
    assert len(data["traceEvents"]) > 0
    tracing.stop_tracing()


def test_query_profile(tmp_path):
    from bodo_amd.utils import query_profile as qp

    qp.clear()
    qp.enable(str(tmp_path))
    df = pd.DataFrame({"a": np.arange(50), "b": np.arange(50) * 2.0})
    b = bpd.from_pandas(df)
    b[b.a > 10].to_pandas()
    recs = qp.get_records()
    assert any(r["operator"] == "Filter" for r in recs)
    qp.flush(str(tmp_path))
    assert os.path.exists(str(tmp_path / "query_profile_rank0.json"))
    qp.clear()


def test_user_logging(capsys):
    import logging

    from bodo_amd import user_logging

    user_logging.set_verbose_level(2)
    user_logging.log_message("Test", "hello %s", "world")
    user_logging.set_verbose_level(0)
    out = capsys.readouterr().out
    assert "hello world" in out


def test_sql_plan_cache():
    from bodo_amd.sql import BodoSQLContext

    df = pd.DataFrame({"a": np.arange(30), "b": np.arange(30) * 1.5})
    bc = BodoSQLContext({"t": df})
    r1 = bc.sql("SELECT a FROM t WHERE a > 5")
    n0 = len(BodoSQLContext._plan_cache)
    r2 = bc.sql("SELECT a FROM t WHERE a > 5")
    assert len(BodoSQLContext._plan_cache) == n0
    assert r1._lazy_plan is r2._lazy_plan


def test_distributed_api_single_rank():
    import bodo_amd

    df = pd.DataFrame({"a": np.arange(20), "b": np.arange(20) * 0.5})
    b = bpd.from_pandas(df)
    g = bodo_amd.gatherv(b)
    pd.testing.assert_frame_equal(g, df, check_dtype=False)
    ag = bodo_amd.allgatherv(b)
    pd.testing.assert_frame_equal(ag, df, check_dtype=False)
    sc = bodo_amd.scatterv(df)
    pd.testing.assert_frame_equal(sc, df, check_dtype=False)
    rb = bodo_amd.rebalance(b)
    assert len(rb) == 20
    assert bodo_amd.get_rank() == 0 and bodo_amd.get_size() == 1
    assert isinstance(bodo_amd.get_gpu_ranks(), list)


def test_jit_options_accepted():
    import bodo_amd

    @bodo_amd.jit(cache=True, distributed=["df"], spawn=True)
    def f(df):
        return df[df.a > 5]

    df = pd.DataFrame({"a": np.arange(10)})
    out = f(bpd.from_pandas(df)).to_pandas()
    assert len(out) == 4


def test_read_json(tmp_path):
    df = pd.DataFrame({"a": [1, 2, 3], "b": ["x", "y", "z"]})
    p = str(tmp_path / "d.json")
    df.to_json(p, orient="records", lines=True)
    out = bpd.read_json(p).to_pandas()
    out["b"] = out["b"].astype(str)
    pd.testing.assert_frame_equal(out, df, check_dtype=False)


def test_profile_aggregation_cli(tmp_path):
    import json as _json
    import subprocess
    import sys

    from bodo_amd.utils import query_profile as qp

    qp.clear()
    qp.enable(str(tmp_path))
    b = bpd.from_pandas(pd.DataFrame({"a": np.arange(40)}))
    b[b.a > 3].to_pandas()
    qp.flush(str(tmp_path))
    out = subprocess.run([sys.executable, "-m",
                          "bodo_amd.utils.aggregate_query_profiles",
                          str(tmp_path)], capture_output=True, text=True)
    assert out.returncode == 0, out.stderr
    data = _json.loads(out.stdout)
    assert data["n_ranks"] == 1 and data["operators"]
    qp.clear()


def test_nlargest():
    df = pd.DataFrame({"a": np.arange(100), "b": np.arange(100)[::-1]})
    b = bpd.from_pandas(df)
    got = b.nlargest(5, "b").to_pandas().reset_index(drop=True)
    exp = df.nlargest(5, "b").reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_allocation_tracking():
    from bodo_amd.utils.allocation_tracking import get_allocation_stats

    s = get_allocation_stats()
    assert set(s) == {"allocated_bytes", "reserved_bytes", "peak_bytes"}


def test_bodo_alias_package():
    import bodo
    import bodo.pandas as bpd2

    df = pd.DataFrame({"a": np.arange(10)})
    b = bpd2.from_pandas(df)
    out = b[b.a > 4].to_pandas()
    assert len(out) == 5

    @bodo.jit
    def f(d):
        return d[d.a > 7]

    assert len(f(b).to_pandas()) == 2


def test_jit_df_lib_substitution():
    """@jit rebinds pandas globals to the lazy engine for the call and
    converts DataFrame args to distributed frames (reference: df-lib mode,
    bodo/tests/utils.py check_func)."""
    import numpy as np
    import pandas as pd

    import bodo_amd

    @bodo_amd.jit
    def pipeline(df):
        f = df[df.a > 3]
        return f.groupby("c", as_index=False).agg(s=pd.NamedAgg("b", "sum"))

    rng = np.random.default_rng(0)
    df = pd.DataFrame({"a": rng.integers(0, 10, 500), "b": rng.random(500),
                       "c": rng.choice(["x", "y"], 500)})
    out = pipeline(df)
    assert type(out).__name__ == "BodoDataFrame"
    got = out.to_pandas()
    got["c"] = got["c"].astype(str)
    got = got.sort_values("c").reset_index(drop=True)
    exp = df[df.a > 3].groupby("c", as_index=False).agg(
        s=("b", "sum")).sort_values("c").reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)
    assert pd.__name__ == "pandas"  # globals restored after the call


def test_hll_approx_nunique():
    """HLL sketch accuracy: within ~3% at 2^14 registers (reference role:
    theta sketches / hyperloglog.hpp)."""
    import numpy as np

    from bodo_amd.core.column import Column
    from bodo_amd.utils import sketches

    rng = np.random.default_rng(9)
    for true_n in (100, 10_000, 1_000_000):
        vals = rng.integers(0, true_n, true_n * 3)
        c = Column.from_numpy(vals)
        est = sketches.approx_nunique([c], distributed=False)
        exact = len(np.unique(vals))
        assert abs(est - exact) / exact < 0.05, (true_n, est, exact)


def test_sql_approx_count_distinct():
    import numpy as np
    import pandas as pd

    from bodo_amd.sql import BodoSQLContext

    rng = np.random.default_rng(4)
    df = pd.DataFrame({"v": rng.integers(0, 50_000, 200_000)})
    bc = BodoSQLContext({"t": df})
    got = bc.sql("select approx_count_distinct(v) as a from t").to_pandas()
    exact = df["v"].nunique()
    assert abs(int(got["a"].iloc[0]) - exact) / exact < 0.05


def test_fs_uri_routing(tmp_path):
    """file:// URIs route through the arrow-FS abstraction (reference:
    bodo/libs/_fs_io.cpp path routing; s3:// takes the same code path)."""
    import numpy as np
    import pandas as pd

    import bodo_amd.pandas as bpd
    from bodo_amd.io import fs as bfs

    assert not bfs.is_remote("/tmp/x.parquet")
    assert not bfs.is_remote(str(tmp_path))  # local stays on OS path
    rng = np.random.default_rng(3)
    df = pd.DataFrame({"a": rng.integers(0, 10, 500), "b": rng.random(500)})
    p = tmp_path / "t.parquet"
    df.to_parquet(str(p))
    uri = "file://" + str(p)
    got = bpd.read_parquet(uri).groupby("a", as_index=False).agg(
        s=bpd.NamedAgg("b", "sum")).sort_values("a").to_pandas()
    exp = df.groupby("a", as_index=False).agg(
        s=("b", "sum")).sort_values("a").reset_index(drop=True)
    pd.testing.assert_frame_equal(got.reset_index(drop=True), exp,
                                  check_dtype=False)
    # write through a URI
    out_uri = "file://" + str(tmp_path / "out.parquet")
    bpd.from_pandas(df).to_parquet(out_uri)
    back = pd.read_parquet(str(tmp_path / "out.parquet"))
    assert len(back) == len(df)


def test_fft_and_json_extract():
    import numpy as np

    import bodo_amd
    from bodo_amd.sql import BodoSQLContext

    x = np.random.default_rng(2).random(4096)
    got = bodo_amd.fft.fft(x)
    np.testing.assert_allclose(got, np.fft.fft(x), rtol=1e-8, atol=1e-8)
    back = bodo_amd.fft.ifft(got)
    np.testing.assert_allclose(back.real, x, atol=1e-9)

    import pandas as pd

    df = pd.DataFrame({"j": ['{"a": {"b": 5}, "c": [1, 2]}',
                             '{"a": {"b": "x"}}', "not json", None]})
    bc = BodoSQLContext({"t": df})
    out = bc.sql("select json_extract_path_text(j, 'a.b') as v, "
                 "json_extract_path_text(j, 'c[1]') as e from t").to_pandas()
    assert out["v"].tolist()[:2] == ["5", "x"]
    assert out["e"].iloc[0] == "2"
    assert pd.isna(out["v"].iloc[2]) and pd.isna(out["v"].iloc[3])


def test_fft_distarray():
    import numpy as np

    import bodo_amd
    from bodo_amd.compiler.distarray import DistArray

    x = np.random.default_rng(3).random(8192)
    d = DistArray.from_numpy(x)
    got = bodo_amd.fft.rfft(d)
    np.testing.assert_allclose(got.to_numpy(), np.fft.rfft(x), rtol=1e-8,
                               atol=1e-8)


def test_roctx_disabled_noop_and_enabled_fallback(monkeypatch):
    import importlib

    from bodo_amd.utils import roctx

    # disabled: no library load, no-ops
    with roctx.Range("x"):
        pass
    # enabled: loads the ROCm roctx library (present in the image) and the
    # push/pop pair must not raise
    monkeypatch.setenv("BODO_AMD_ROCTX", "1")
    importlib.reload(roctx)
    with roctx.Range("bodo.test"):
        roctx.range_push("inner")
        roctx.range_pop()
