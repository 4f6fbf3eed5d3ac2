"""README quickstart runs end-to-end (reference:
bodo/tests/test_quickstart_docs.py)."""

import os

import numpy as np
import pandas as pd


def test_quickstart(tmp_path, monkeypatch):
    rng = np.random.default_rng(0)
    src = pd.DataFrame({"driver": rng.integers(0, 50, 10000),
                        "miles": rng.random(10000) * 30,
                        "hours": rng.random(10000) + 0.1})
    src.to_parquet(str(tmp_path / "trips.parquet"))
    monkeypatch.chdir(tmp_path)

    import bodo_amd.pandas as bpd

    df = bpd.read_parquet("trips.parquet")
    df["speed"] = df.miles / df.hours
    out = df[df.speed > 1.0].groupby("driver", as_index=False).agg(
        trips=bpd.NamedAgg("speed", "count"),
        avg=bpd.NamedAgg("speed", "mean"))
    out.to_parquet("out.parquet")
    assert os.path.exists("out.parquet")

    from bodo_amd.sql import BodoSQLContext

    bc = BodoSQLContext({"lineitem": df})
    res = bc.sql("SELECT driver, SUM(miles) AS m FROM lineitem "
                 "GROUP BY driver ORDER BY driver").to_pandas()
    assert len(res) == 50

    # the drop-in alias spelled exactly as the reference documents it
    import bodo.pandas as bp

    assert bp.read_parquet is bpd.read_parquet
