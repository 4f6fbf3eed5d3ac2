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
