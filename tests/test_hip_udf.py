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


def test_fuse_signature_and_emit():
    """Codegen smoke (host): fusable expr tree -> C source compiles the
    string path (no GPU launch)."""
    import pandas as pd

    from bodo_amd.core.table import Table
    from bodo_amd.compiler.expr_fuse import _Fuser
    from bodo_amd.plan.expr import BinOp, ColRef, Cmp, Const, DtField

    df = pd.DataFrame({"a": np.arange(5, dtype="int64"),
                       "b": np.arange(5) * 0.5,
                       "t": pd.to_datetime(["2020-01-0%d" % d for d in
                                            range(1, 6)])})
    t = Table.from_pandas(df)
    f = _Fuser(t)
    e1 = BinOp("mul", ColRef("b"), BinOp("sub", Const(1.0), ColRef("b")))
    c1, t1 = f.emit(e1)
    assert t1 == "double" and "*" in c1
    c2, t2 = f.emit(Cmp("gt", ColRef("a"), Const(3)))
    assert t2 == "bool"
    c3, t3 = f.emit(DtField(ColRef("t"), "month"))
    assert t3 == "i64" and "civil" in "".join(f.lines)


@pytest.mark.gpu
def test_fused_projection_matches_unfused():
    import bodo_amd.config as cfg
    import pandas as pd

    from bodo_amd.core.table import Table
    from bodo_amd.ops import evaluate as ev
    from bodo_amd.plan.expr import BinOp, BoolOp, Case, ColRef, Cmp, Const, \
        DtField, IsIn

    rng = np.random.default_rng(3)
    n = 200000
    df = pd.DataFrame({
        "x": rng.uniform(-2, 2, n),
        "y": rng.integers(-100, 100, n),
        "ts": pd.to_datetime(pd.Timestamp("1999-01-01").value
                             + rng.integers(0, 10**18, n)),
    })
    t = Table.from_pandas(df, device="cuda")
    exprs = [
        BinOp("mul", ColRef("x"), BinOp("add", Const(1.0), ColRef("x"))),
        Cmp("gt", ColRef("x"), Const(0.25)),
        BinOp("add", ColRef("y"), Const(7)),
        DtField(ColRef("ts"), "month"),
        DtField(ColRef("ts"), "hour"),
        IsIn(DtField(ColRef("ts"), "dayofweek"), (0, 1, 2, 3, 4)),
        Case((Cmp("lt", ColRef("x"), Const(0.0)),), (Const(-1.0),),
             ColRef("x")),
    ]
    names = [f"o{i}" for i in range(len(exprs))]
    cfg.FUSE_EXPR = True
    fused = ev.project(t, names, exprs).to_pandas()
    cfg.FUSE_EXPR = False
    try:
        unfused = ev.project(t, names, exprs).to_pandas()
    finally:
        cfg.FUSE_EXPR = True
    for c in names:
        a, b = fused[c].to_numpy(), unfused[c].to_numpy()
        assert np.allclose(a.astype(np.float64), b.astype(np.float64)), c
