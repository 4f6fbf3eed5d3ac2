"""Spawn-mode tests: a plain python process with BODO_NUM_WORKERS=2 executes
frames on worker subprocesses (reference: bodo/spawn/ + e2e-tests
spawn-mode runs)."""

import os
import subprocess
import sys
import textwrap

import pytest

pytestmark = pytest.mark.multi_rank

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_spawn_script(body: str, n_workers=2, timeout=240) -> str:
    script = textwrap.dedent(body)
    env = dict(os.environ)
    env.update({"BODO_NUM_WORKERS": str(n_workers), "BODO_AMD_DEVICE": "cpu",
                "PYTHONPATH": REPO})
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    p = subprocess.run([sys.executable, "-c", script], env=env, cwd=REPO,
                       capture_output=True, text=True, timeout=timeout)
    assert p.returncode == 0, f"stdout:\n{p.stdout}\nstderr:\n{p.stderr}"
    return p.stdout


def test_spawn_groupby():
    out = run_spawn_script("""
        import warnings; warnings.filterwarnings("ignore")
        import numpy as np, pandas as pd
        import bodo_amd.pandas as bpd

        rng = np.random.default_rng(0)
        df = pd.DataFrame({"a": rng.integers(0, 10, 5000),
                           "b": rng.uniform(0, 1, 5000)})
        b = bpd.from_pandas(df)
        got = b.groupby("a", as_index=False).agg(
            s=bpd.NamedAgg("b", "sum")).sort_values("a").to_pandas()
        exp = df.groupby("a", as_index=False).agg(
            s=("b", "sum")).sort_values("a").reset_index(drop=True)
        pd.testing.assert_frame_equal(got.reset_index(drop=True), exp,
                                      check_dtype=False)
        print("SPAWN_GROUPBY_OK", len(got))
    """)
    assert "SPAWN_GROUPBY_OK" in out


def test_spawn_taxi_query():
    out = run_spawn_script("""
        import warnings; warnings.filterwarnings("ignore")
        import sys
        sys.path.insert(0, "tests")
        import numpy as np, pandas as pd
        import bodo_amd.pandas as bpd
        from tests.test_queries import make_taxi, nyc_taxi_q1

        trips, weather = make_taxi(5000, 3)
        got = nyc_taxi_q1(bpd, bpd.from_pandas(trips),
                          bpd.from_pandas(weather)).to_pandas()
        exp = nyc_taxi_q1(pd, trips.copy(), weather.copy()).reset_index(drop=True)
        got = got.reset_index(drop=True)
        got["time_bucket"] = got["time_bucket"].astype(str)
        pd.testing.assert_frame_equal(got, exp, check_dtype=False)
        print("SPAWN_TAXI_OK", len(got))
    """)
    assert "SPAWN_TAXI_OK" in out


def test_spawn_jit_and_scalars():
    out = run_spawn_script("""
        import warnings; warnings.filterwarnings("ignore")
        import numpy as np, pandas as pd
        import bodo_amd
        import bodo_amd.pandas as bpd

        rng = np.random.default_rng(1)
        df = pd.DataFrame({"x": rng.uniform(0, 1, 4000)})
        b = bpd.from_pandas(df)
        assert abs(b.x.sum() - df.x.sum()) < 1e-9
        assert abs(b.x.mean() - df.x.mean()) < 1e-12

        @bodo_amd.jit
        def f(d):
            d2 = d[d.x > 0.5]
            return d2

        res = f(b).to_pandas()
        exp = df[df.x > 0.5].reset_index(drop=True)
        pd.testing.assert_frame_equal(res.reset_index(drop=True), exp,
                                      check_dtype=False)
        print("SPAWN_JIT_OK", len(res))
    """)
    assert "SPAWN_JIT_OK" in out


def test_spawn_sql():
    """BodoSQLContext in spawn mode: the plan executes on the worker group
    and gathers through the control plane."""
    out = run_spawn_script("""
        import warnings; warnings.filterwarnings("ignore")
        import numpy as np, pandas as pd
        from bodo_amd.sql import BodoSQLContext

        rng = np.random.default_rng(15)
        df = pd.DataFrame({"k": rng.integers(0, 6, 400),
                           "v": rng.random(400)})
        bc = BodoSQLContext({"t": df})
        got = bc.sql("select k, sum(v) as s, count(*) as n from t "
                     "group by k order by k").to_pandas()
        exp = df.groupby("k", as_index=False).agg(
            s=("v", "sum"), n=("v", "size")).sort_values(
            "k").reset_index(drop=True)
        pd.testing.assert_frame_equal(got.reset_index(drop=True), exp,
                                      check_dtype=False)
        print("SPAWN_SQL_OK")
    """)
    assert "SPAWN_SQL_OK" in out


def test_spawn_window_rolling():
    out = run_spawn_script("""
        import warnings; warnings.filterwarnings("ignore")
        import numpy as np, pandas as pd
        import bodo_amd.pandas as bpd

        rng = np.random.default_rng(7)
        df = pd.DataFrame({"k": rng.integers(0, 6, 3000),
                           "v": rng.uniform(0, 1, 3000)})
        b = bpd.from_pandas(df)
        b["t"] = b.groupby("k")["v"].transform("sum")
        roll = b.v.rolling(5, min_periods=1).mean().to_pandas()
        exp_roll = df.v.rolling(5, min_periods=1).mean().reset_index(drop=True)
        pd.testing.assert_series_equal(roll, exp_roll, check_names=False)
        got = b.to_pandas()
        exp_t = df.groupby("k")["v"].transform("sum")
        assert np.allclose(got["t"].to_numpy(), exp_t.to_numpy())
        print("SPAWN_WINDOW_OK")
    """)
    assert "SPAWN_WINDOW_OK" in out
