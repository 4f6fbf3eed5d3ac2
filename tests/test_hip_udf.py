"""UDF -> HIP lowering tests (reference role: @bodo.jit UDF cfuncs,
physical/expression.h:1288; here lowered to gfx950 via hipRTC)."""

import math

import numpy as np
import pandas as pd
import pytest
import torch

from bodo_amd.compiler.hip_udf import translate_udf


def test_translate_lambda():
    e = translate_udf(lambda x: x * 2 + 1)
    assert e is not None and "x" in e and "*" in e


def test_translate_branches():
    def f(v):
        if v > 0.5:
            return v * 2
        elif v > 0.2:
            return v + 1
        return 0.0

    e = translate_udf(f)
    assert e is not None and "?" in e


def test_translate_membership_and_math():
    def g(t):
        if t in (1, 2, 3):
            return math.sqrt(t)
        return t ** 2

    e = translate_udf(g)
    assert e is not None and "sqrt" in e and "==" in e


def test_translate_unsupported_returns_none():
    assert translate_udf(lambda s: s.upper()) is None
    assert translate_udf(lambda s: "a" if s else "b") is None


@pytest.mark.gpu
def test_hip_udf_matches_pandas():
    from bodo_amd.compiler.hip_udf import try_hip_udf

    rng = np.random.default_rng(0)
    vals = rng.uniform(-2, 2, 100000)
    t = torch.from_numpy(vals).cuda()

    funcs = [
        lambda x: x * 2 + 1,
        lambda x: x * x - 0.5 * x,
        lambda x: x if x > 0 else -x,
        lambda x: min(x, 0.5) + max(x, -0.5),
    ]
    for f in funcs:
        got = try_hip_udf(f, t)
        assert got is not None
        exp = pd.Series(vals).map(f).to_numpy()
        assert np.allclose(got.cpu().numpy(), exp, atol=1e-12)


@pytest.mark.gpu
def test_series_map_uses_hip(monkeypatch):
    import bodo_amd.config as cfg

    cfg.DEVICE = "cuda"
    try:
        import bodo_amd.pandas as bpd

        rng = np.random.default_rng(1)
        df = pd.DataFrame({"x": rng.uniform(0, 1, 50000)})
        b = bpd.from_pandas(df)
        got = b.x.map(lambda v: v * 3.0 + 0.25).to_pandas().to_numpy()
        exp = (df.x * 3.0 + 0.25).to_numpy()
        assert np.allclose(got, exp)
    finally:
        cfg.DEVICE = ""
