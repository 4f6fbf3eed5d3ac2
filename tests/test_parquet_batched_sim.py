"""CPU validation of the batched device decode flow: the real C++ host
header parser + bit-faithful Python simulators of each HIP kernel (with
bounds assertions) driven through the real `_decode_batched` orchestration.
Catches index/parse bugs that would be hardware faults on the MI355X."""

import sys
import types

import numpy as np
import pandas as pd
import pytest
import torch

from tests import kernel_sim as ks


def _metas(pages_blob, n_pages):
    from bodo_amd.io.parquet_gpu import _PAGEMETA_DT

    return np.frombuffer(pages_blob.cpu().numpy().tobytes(),
                         dtype=_PAGEMETA_DT, count=int(n_pages))


def _sim_module():
    import bodo_amd_kernels as K

    m = types.SimpleNamespace()
    m.pq_parse_headers = K.pq_parse_headers
    m.rle_expand = K.rle_expand

    def pq_decompress(src, pages_blob, n_pages, scratch):
        ks.sim_decompress(src.numpy(), _metas(pages_blob, n_pages),
                          scratch.numpy())

    def pq_def_levels(scratch, pages_blob, n_pages, bitwidth, max_def,
                      total_nv):
        mask, n_valid, vdo = ks.sim_def_levels(
            scratch.numpy(), _metas(pages_blob, n_pages), int(bitwidth),
            int(max_def), int(total_nv))
        return (torch.from_numpy(mask), torch.from_numpy(n_valid),
                torch.from_numpy(vdo))

    def pq_expand_codes(scratch, pages_blob, n_pages, vdo, dense_off,
                        n_valid, dense_total):
        out = ks.sim_expand_codes(
            scratch.numpy(), _metas(pages_blob, n_pages),
            None if vdo is None else vdo.numpy(), dense_off.numpy(),
            None if n_valid is None else n_valid.numpy(), int(dense_total))
        return torch.from_numpy(out)

    def pq_copy_fixed(scratch, pages_blob, n_pages, vdo, dense_off, n_valid,
                      esize, dense_total):
        out = ks.sim_copy_fixed(
            scratch.numpy(), _metas(pages_blob, n_pages),
            None if vdo is None else vdo.numpy(), dense_off.numpy(),
            None if n_valid is None else n_valid.numpy(), int(esize),
            int(dense_total))
        return torch.from_numpy(out)

    def pq_byte_array_lengths(scratch, pages_blob, n_pages, vdo, dense_off,
                              n_valid, dense_total):
        lens, src_abs = ks.sim_byte_array(
            scratch.numpy(), _metas(pages_blob, n_pages),
            None if vdo is None else vdo.numpy(), dense_off.numpy(),
            None if n_valid is None else n_valid.numpy(), int(dense_total))
        return torch.from_numpy(lens), torch.from_numpy(src_abs)

    def pq_copy_strings(scratch, src_abs, dst_off, lengths, n, total_bytes):
        out = np.zeros(max(int(total_bytes), 1), dtype=np.uint8)
        s = scratch.numpy()
        sa, do, ln = src_abs.numpy(), dst_off.numpy(), lengths.numpy()
        for i in range(int(n)):
            assert sa[i] + ln[i] <= len(s), f"string src overrun row {i}"
            assert do[i] + ln[i] <= len(out), f"string dst overrun row {i}"
            out[do[i]:do[i] + ln[i]] = s[sa[i]:sa[i] + ln[i]]
        return torch.from_numpy(out)

    m.pq_decompress = pq_decompress
    m.pq_def_levels = pq_def_levels
    m.pq_expand_codes = pq_expand_codes
    m.pq_copy_fixed = pq_copy_fixed
    m.pq_byte_array_lengths = pq_byte_array_lengths
    m.pq_copy_strings = pq_copy_strings
    return m


@pytest.fixture
def sim_kernels(monkeypatch):
    from bodo_amd.io import parquet_gpu as g

    monkeypatch.setitem(sys.modules, "bodo_amd_kernels", _sim_module())
    monkeypatch.setattr(g, "FORCE_BATCHED", True)
    yield g


def _ctx(device="cpu"):
    class Ctx:
        world, rank = 1, 0

    Ctx.device = torch.device(device)
    return Ctx()


def _roundtrip(g, df, tmp_path, name, **write_kw):
    import pyarrow as pa
    import pyarrow.parquet as pq

    fp = str(tmp_path / f"{name}.parquet")
    pq.write_table(pa.Table.from_pandas(df, preserve_index=False), fp,
                   **write_kw)
    before = g.STATS["slow"]
    t = g._read_row_group_gpu(fp, 0, None, _ctx())
    assert t is not None, "decode fell back to host"
    assert g.STATS["slow"] == before, "batched path fell back"
    out = t.to_pandas()
    for c in out.columns:
        if out[c].dtype.name == "category":
            out[c] = out[c].astype(object)
    pd.testing.assert_frame_equal(out, df, check_dtype=False)


def test_sim_uncompressed_plain_and_dict(tmp_path, sim_kernels):
    """Exact shape of the GPU test that faulted on hardware."""
    rng = np.random.default_rng(9)
    n = 200_000
    df = pd.DataFrame({
        "i": rng.integers(0, 1000, n),
        "f": rng.uniform(-1, 1, n),
        "c": rng.choice(["aa", "bb", "cc", "dd"], n),
        "t": pd.to_datetime(pd.Timestamp("2020-01-01").value
                            + rng.integers(0, 10**15, n)),
    })
    _roundtrip(sim_kernels, df, tmp_path, "g", compression="NONE",
               use_dictionary=["c"])


def test_sim_snappy_layout_matrix(tmp_path, sim_kernels):
    rng = np.random.default_rng(21)
    n = 50_000
    base = {
        "i64": rng.integers(-10**9, 10**9, n).astype("int64"),
        "i32": rng.integers(0, 100, n).astype("int32"),
        "f64": rng.random(n) * 1e6,
        "f32": rng.random(n).astype("float32"),
        "s_plain": np.array(
            ["s" + str(v) for v in rng.integers(0, 10**9, n)], dtype=object),
        "s_dict": rng.choice(["alpha", "beta", "gamma", "delta"], n),
    }
    for comp in ("snappy", "none"):
        for with_nulls in (False, True):
            df = pd.DataFrame({k: v.copy() for k, v in base.items()})
            if with_nulls:
                df.loc[rng.random(n) < 0.08, "f64"] = np.nan
                df.loc[rng.random(n) < 0.08, "s_plain"] = None
                df.loc[rng.random(n) < 0.08, "s_dict"] = None
            _roundtrip(sim_kernels, df, tmp_path,
                       f"m_{comp}_{with_nulls}", compression=comp,
                       use_dictionary=["s_dict", "i32"],
                       data_page_size=64 * 1024)


def test_sim_multi_page_small_pages(tmp_path, sim_kernels):
    rng = np.random.default_rng(5)
    n = 120_000
    df = pd.DataFrame({"a": rng.integers(0, 10**6, n),
                       "b": rng.random(n)})
    df.loc[rng.random(n) < 0.2, "b"] = np.nan
    _roundtrip(sim_kernels, df, tmp_path, "mp", compression="snappy",
               use_dictionary=False, data_page_size=8 * 1024)
