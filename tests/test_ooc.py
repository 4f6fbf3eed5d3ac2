"""Out-of-core partition-splitting tests: a tiny forced budget makes every
groupby/join run the host-staged partition-at-a-time path and the results
must match pandas exactly (reference analog: JoinPartition/GroupbyPartition
top-bitmask recursion + spill, bodo/libs/streaming/_join.h:267)."""

import numpy as np
import pandas as pd
import pytest

import bodo_amd.config as cfg
import bodo_amd.pandas as bpd


@pytest.fixture()
def tiny_budget(monkeypatch):
    monkeypatch.setenv("BODO_AMD_OOC_BYTES", "4096")
    yield


def _df(n=20000, seed=3):
    rng = np.random.default_rng(seed)
    return pd.DataFrame({
        "k": rng.integers(0, 400, n),
        "v": rng.random(n),
        "c": rng.choice(["a", "b", "c"], n),
    })


def test_ooc_groupby(tiny_budget):
    df = _df()
    got = bpd.from_pandas(df).groupby("k", as_index=False).agg(
        s=bpd.NamedAgg("v", "sum"), m=bpd.NamedAgg("v", "mean"),
        n=bpd.NamedAgg("v", "count"),
    ).sort_values("k").to_pandas().reset_index(drop=True)
    exp = df.groupby("k", as_index=False).agg(
        s=("v", "sum"), m=("v", "mean"), n=("v", "count"),
    ).sort_values("k").reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_ooc_groupby_multikey(tiny_budget):
    df = _df()
    got = bpd.from_pandas(df).groupby(["k", "c"], as_index=False).agg(
        s=bpd.NamedAgg("v", "sum")).sort_values(["k", "c"]).to_pandas()
    got = got.reset_index(drop=True)
    got["c"] = got["c"].astype(str)
    exp = df.groupby(["k", "c"], as_index=False).agg(
        s=("v", "sum")).sort_values(["k", "c"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


@pytest.mark.parametrize("how", ["inner", "left", "semi"])
def test_ooc_join(tiny_budget, how):
    df = _df(15000, 5)
    rng = np.random.default_rng(6)
    right = pd.DataFrame({"k": rng.permutation(600)[:350],
                          "w": rng.random(350)})
    b = bpd.from_pandas(df)
    br = bpd.from_pandas(right)
    if how == "semi":
        got = b[b.k.isin(br.k)].to_pandas()
        exp = df[df.k.isin(set(right.k))]
    else:
        got = b.merge(br, on="k", how=how).to_pandas()
        exp = df.merge(right, on="k", how=how)
    got = got.sort_values(["k", "v"]).reset_index(drop=True)
    got["c"] = got["c"].astype(str)
    exp = exp.sort_values(["k", "v"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_budget_off_uses_single_pass(monkeypatch):
    """No budget -> partition path must not trigger on CPU."""
    from bodo_amd.engine import ooc

    monkeypatch.delenv("BODO_AMD_OOC_BYTES", raising=False)
    monkeypatch.setattr(cfg, "OOC_BYTES", 0, raising=False)
    import torch

    assert ooc.budget_bytes(torch.device("cpu")) is None


def test_ooc_sort(tiny_budget):
    df = _df(30000, 7)
    got = bpd.from_pandas(df).sort_values(["k", "v"]).to_pandas()
    got = got.reset_index(drop=True)
    got["c"] = got["c"].astype(str)
    exp = df.sort_values(["k", "v"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)


def test_ooc_sort_nans(tiny_budget):
    import numpy as np

    rng = np.random.default_rng(8)
    v = rng.random(20000)
    v[rng.random(20000) < 0.05] = np.nan
    df = pd.DataFrame({"v": v, "i": np.arange(20000)})
    got = bpd.from_pandas(df).sort_values("v").to_pandas().reset_index(
        drop=True)
    exp = df.sort_values("v").reset_index(drop=True)
    pd.testing.assert_frame_equal(got[["v"]], exp[["v"]], check_dtype=False)


def test_ooc_disk_spill(tiny_budget, tmp_path, monkeypatch):
    """With BODO_AMD_SPILL_DIR set, over-budget partitions stage on disk
    (the NVMe tier) and results stay exact."""
    monkeypatch.setenv("BODO_AMD_SPILL_DIR", str(tmp_path / "spill"))
    df = _df(15000, 9)
    got = bpd.from_pandas(df).groupby("k", as_index=False).agg(
        s=bpd.NamedAgg("v", "sum")).sort_values("k").to_pandas()
    got = got.reset_index(drop=True)
    exp = df.groupby("k", as_index=False).agg(
        s=("v", "sum")).sort_values("k").reset_index(drop=True)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)
    srt = bpd.from_pandas(df).sort_values(["k", "v"]).to_pandas()
    srt = srt.reset_index(drop=True)
    srt["c"] = srt["c"].astype(str)
    exp2 = df.sort_values(["k", "v"]).reset_index(drop=True)
    pd.testing.assert_frame_equal(srt, exp2, check_dtype=False)


def test_comptroller_divides_budget(monkeypatch):
    """Two live operators each see half the budget (reference:
    _memory_budget.h OperatorComptroller)."""
    from bodo_amd.engine import comptroller, ooc
    import torch

    monkeypatch.setenv("BODO_AMD_OOC_BYTES", "1000000")
    dev = torch.device("cpu")
    assert ooc.budget_bytes(dev) == 1000000
    with comptroller.operator():
        assert ooc.budget_bytes(dev) == 1000000  # self counts as the 1 live
        with comptroller.operator():
            assert ooc.budget_bytes(dev) == 500000
    assert comptroller.peak() >= 2


def test_streaming_join_with_ooc_budget(tmp_path, monkeypatch):
    """Forced morsel streaming THROUGH a join with a tiny OOC budget: the
    SF1000 execution shape (streamed probe + partition-split local ops)."""
    import bodo_amd.config as cfg

    monkeypatch.setenv("BODO_AMD_OOC_BYTES", "200000")
    rng = np.random.default_rng(7)
    n = 120_000
    li = pd.DataFrame({"okey": rng.integers(0, 3000, n),
                       "qty": rng.random(n) * 50,
                       "price": rng.random(n) * 1000})
    orders = pd.DataFrame({"o_key": np.arange(3000),
                           "flag": rng.choice(["A", "B", "C"], 3000)})
    p = str(tmp_path / "li.parquet")
    li.to_parquet(p, row_group_size=8000)
    old = cfg.STREAMING, cfg.STREAM_BATCH_SIZE
    cfg.STREAMING = "1"
    cfg.STREAM_BATCH_SIZE = 10_000
    try:
        b = bpd.read_parquet(p)
        o = bpd.from_pandas(orders)
        got = b[b.qty < 40].merge(o, left_on="okey", right_on="o_key") \
            .groupby("flag", as_index=False).agg(
            s=bpd.NamedAgg("price", "sum"),
            c=bpd.NamedAgg("qty", "count")).sort_values("flag").to_pandas()
        exp = li[li.qty < 40].merge(orders, left_on="okey",
                                    right_on="o_key").groupby(
            "flag", as_index=False).agg(
            s=("price", "sum"), c=("qty", "count")).sort_values(
            "flag").reset_index(drop=True)
        got["flag"] = got["flag"].astype(str)
        pd.testing.assert_frame_equal(got.reset_index(drop=True), exp,
                                      check_dtype=False, rtol=1e-9)
    finally:
        cfg.STREAMING, cfg.STREAM_BATCH_SIZE = old
