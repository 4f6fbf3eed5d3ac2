"""CPU-verifiable pieces of the GPU parquet decoder (snappy page
decompression + PLAIN decode are device-agnostic)."""

import numpy as np
import pandas as pd
import torch


def _ctx(device):
    class Ctx:
        world, rank = 1, 0

    Ctx.device = torch.device(device)
    return Ctx()


def test_parquet_snappy_plain_decode(tmp_path):
    import pyarrow as pa
    import pyarrow.parquet as pq

    from bodo_amd.io import parquet_gpu as g

    rng = np.random.default_rng(0)
    df = pd.DataFrame({"a": rng.integers(0, 1000, 5000).astype("int64"),
                       "b": rng.random(5000)})
    fp = str(tmp_path / "t.parquet")
    pq.write_table(pa.Table.from_pandas(df), fp, compression="snappy",
                   use_dictionary=False)
    t = g._read_row_group_gpu(fp, 0, None, _ctx("cpu"))
    assert t is not None
    pd.testing.assert_frame_equal(t.to_pandas(), df, check_dtype=False)


def test_parquet_uncompressed_regression(tmp_path):
    import pyarrow as pa
    import pyarrow.parquet as pq

    from bodo_amd.io import parquet_gpu as g

    rng = np.random.default_rng(1)
    df = pd.DataFrame({"a": rng.integers(0, 9, 4000).astype("int32"),
                       "b": rng.random(4000)})
    fp = str(tmp_path / "t.parquet")
    pq.write_table(pa.Table.from_pandas(df), fp, compression="none",
                   use_dictionary=False)
    t = g._read_row_group_gpu(fp, 0, None, _ctx("cpu"))
    assert t is not None
    pd.testing.assert_frame_equal(t.to_pandas(), df, check_dtype=False)


def _rle_expand_ref(blob, nruns, dev_buf, bitwidth, nv):
    """Python reference of csrc rle_expand (bit-exact); lets the whole
    device decode flow verify on CPU."""
    runs = np.frombuffer(
        bytes(blob.cpu().numpy()),
        dtype=[("out_start", np.int64), ("count", np.int32),
               ("kind", np.int32), ("val", np.int64)], count=nruns)
    buf = dev_buf.cpu().numpy().tobytes()
    out = np.zeros(nv, dtype=np.int32)
    for o, c, k, v in runs:
        if k == 0:
            out[o:o + c] = v
        else:
            for i in range(c):
                start = int(v) + i * bitwidth
                byte, bit = divmod(start, 8)
                word = int.from_bytes(buf[byte:byte + 8], "little")
                out[o + i] = (word >> bit) & ((1 << bitwidth) - 1)
    return torch.from_numpy(out)


def test_parquet_snappy_dict_nulls_mixed(tmp_path, monkeypatch):
    """Snappy + dictionary pages + nulls + dict->PLAIN page switches, the
    full decode flow on CPU with the reference RLE expander."""
    import sys
    import types

    import pyarrow as pa
    import pyarrow.parquet as pq

    monkeypatch.setitem(sys.modules, "bodo_amd_kernels",
                        types.SimpleNamespace(rle_expand=_rle_expand_ref))
    from bodo_amd.io import parquet_gpu as g

    rng = np.random.default_rng(1)
    n = 60_000
    a = rng.integers(0, 50, n).astype("float64")
    a[rng.random(n) < 0.1] = np.nan
    df = pd.DataFrame({
        "a": a,                                   # dict + nulls
        "s": rng.choice(["aa", "bb", "cc"], n),   # string dict
        "v": rng.random(n),                       # high-card: dict->PLAIN mix
        "i": rng.integers(0, 10, n),              # int dict
    })
    fp = str(tmp_path / "t.parquet")
    pq.write_table(pa.Table.from_pandas(df), fp, compression="snappy")
    t = g._read_row_group_gpu(fp, 0, None, _ctx("cpu"))
    assert t is not None, "decode fell back"
    out = t.to_pandas()
    for c in out.columns:
        if out[c].dtype.name == "category":
            out[c] = out[c].astype(str)
    pd.testing.assert_frame_equal(out, df, check_dtype=False)


def test_pq_parse_headers_matches_python(tmp_path):
    """C++ host page-header parser vs the Python ThriftCompact walk."""
    import pyarrow as pa
    import pyarrow.parquet as pq

    import bodo_amd_kernels as K
    from bodo_amd.io import parquet_gpu as g

    rng = np.random.default_rng(3)
    n = 200_000
    df = pd.DataFrame({
        "a": rng.integers(0, 50, n).astype("int64"),
        "b": rng.random(n),
        "s": rng.choice(["xx", "yyy", "zzzz"], n),
    })
    df.loc[rng.random(n) < 0.05, "b"] = np.nan
    fp = str(tmp_path / "t.parquet")
    pq.write_table(pa.Table.from_pandas(df), fp, compression="snappy",
                   data_page_size=64 * 1024)
    pf = pq.ParquetFile(fp)
    rgm = pf.metadata.row_group(0)
    with open(fp, "rb") as f:
        for ci in range(rgm.num_columns):
            cm = rgm.column(ci)
            r = g._ChunkReader(f, cm, cm.physical_type)
            got = K.pq_parse_headers(
                torch.from_numpy(
                    np.frombuffer(r.buf, dtype=np.uint8).copy())).numpy()
            # python reference walk
            pos, rows = 0, []
            while pos < len(r.buf):
                t = g.ThriftCompact(r.buf, pos)
                hdr = t.read_struct()
                body = t.pos
                ptype = hdr.get(1)
                dph = hdr.get(5 if ptype == g.PAGE_DATA else 7, {})
                rows.append((ptype, hdr.get(2), hdr.get(3), body,
                             dph.get(1, 0), dph.get(2, -1),
                             dph.get(3, -1) if ptype == g.PAGE_DATA else -1))
                pos = body + hdr.get(3)
            exp = np.array(rows, dtype=np.int64)
            assert got.shape == exp.shape, (got.shape, exp.shape)
            assert (got == exp).all(), cm.path_in_schema
