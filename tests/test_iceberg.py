"""Filesystem Iceberg table tests: transactional writes, append snapshots,
time travel (reference surface: bodo/pandas/base.py read_iceberg,
frame.py to_iceberg, bodo/io/iceberg/)."""

import os

import pandas as pd
import pytest

import bodo_amd.pandas as bpd
from bodo_amd.io import iceberg as ib


@pytest.fixture()
def df():
    return pd.DataFrame({
        "a": [1, 2, 3, 4, 5],
        "b": ["x", "y", "x", "z", "y"],
        "c": [1.5, 2.5, 3.5, 4.5, 5.5],
    })


def _read_sorted(path, **kw):
    out = bpd.read_iceberg(path, **kw).to_pandas()
    for c in out.columns:  # low-cardinality strings read back dict-encoded
        if isinstance(out[c].dtype, pd.CategoricalDtype):
            out[c] = out[c].astype(str)
    return out.sort_values("a").reset_index(drop=True)


def test_roundtrip(tmp_path, df):
    p = str(tmp_path / "tbl")
    bpd.from_pandas(df).to_iceberg(p)
    assert ib.is_iceberg_dir(p)
    pd.testing.assert_frame_equal(_read_sorted(p), df, check_dtype=False)


def test_create_exists_raises(tmp_path, df):
    p = str(tmp_path / "tbl")
    bpd.from_pandas(df).to_iceberg(p)
    with pytest.raises(FileExistsError):
        bpd.from_pandas(df).to_iceberg(p, mode="create")


def test_append_and_time_travel(tmp_path, df):
    p = str(tmp_path / "tbl")
    bpd.from_pandas(df).to_iceberg(p)
    snap1 = ib.snapshots(p)[-1]["snapshot-id"]
    df2 = pd.DataFrame({"a": [6, 7], "b": ["q", "q"], "c": [6.5, 7.5]})
    bpd.from_pandas(df2).to_iceberg(p, mode="append")
    snaps = ib.snapshots(p)
    assert len(snaps) == 2 and snaps[-1]["operation"] == "append"
    both = pd.concat([df, df2], ignore_index=True)
    pd.testing.assert_frame_equal(_read_sorted(p), both, check_dtype=False)
    # time travel back to the first snapshot
    pd.testing.assert_frame_equal(_read_sorted(p, snapshot_id=snap1), df,
                                  check_dtype=False)


def test_replace(tmp_path, df):
    p = str(tmp_path / "tbl")
    bpd.from_pandas(df).to_iceberg(p)
    df2 = pd.DataFrame({"a": [9], "b": ["r"], "c": [9.5]})
    bpd.from_pandas(df2).to_iceberg(p, mode="replace")
    pd.testing.assert_frame_equal(_read_sorted(p), df2, check_dtype=False)
    assert len(ib.snapshots(p)) == 1


def test_query_over_iceberg(tmp_path, df):
    p = str(tmp_path / "tbl")
    bpd.from_pandas(df).to_iceberg(p)
    t = bpd.read_iceberg(p)
    out = t[t.a > 2].groupby("b", as_index=False).agg(
        s=bpd.NamedAgg("c", "sum")).to_pandas()
    if isinstance(out["b"].dtype, pd.CategoricalDtype):
        out["b"] = out["b"].astype(str)
    exp = df[df.a > 2].groupby("b", as_index=False).agg(
        s=pd.NamedAgg("c", "sum"))
    pd.testing.assert_frame_equal(
        out.sort_values("b").reset_index(drop=True),
        exp.sort_values("b").reset_index(drop=True), check_dtype=False)


def test_unknown_snapshot_raises(tmp_path, df):
    p = str(tmp_path / "tbl")
    bpd.from_pandas(df).to_iceberg(p)
    with pytest.raises(ValueError):
        bpd.read_iceberg(p, snapshot_id=12345)


def test_commit_is_atomic_visibility(tmp_path, df):
    """A data file written without a committed snapshot is invisible."""
    p = str(tmp_path / "tbl")
    bpd.from_pandas(df).to_iceberg(p)
    stray = os.path.join(p, "data", "part-9999-00099.parquet")
    with open(stray, "wb") as f:
        f.write(b"not parquet")
    pd.testing.assert_frame_equal(_read_sorted(p), df, check_dtype=False)


def test_iceberg_merge_into_upsert(tmp_path):
    """MERGE INTO (update + insert): upsert semantics with a transactional
    snapshot (reference: bodo/io/iceberg/merge_into.py)."""
    import bodo_amd.pandas as bpd
    from bodo_amd.io import iceberg as ib

    p = str(tmp_path / "tbl")
    base = pd.DataFrame({"k": [1, 2, 3, 4], "v": [10.0, 20.0, 30.0, 40.0],
                         "s": ["a", "b", "c", "d"]})
    bpd.from_pandas(base).to_iceberg(p, mode="create")
    src = pd.DataFrame({"k": [3, 4, 5], "v": [33.0, 44.0, 55.0],
                        "s": ["cc", "dd", "ee"]})
    ib.merge_into(p, src, on="k")
    got = bpd.read_iceberg(p).to_pandas().sort_values("k").reset_index(
        drop=True)
    exp = pd.DataFrame({"k": [1, 2, 3, 4, 5],
                        "v": [10.0, 20.0, 33.0, 44.0, 55.0],
                        "s": ["a", "b", "cc", "dd", "ee"]})
    got["s"] = got["s"].astype(str)
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)
    # two snapshots exist; the pre-merge snapshot is still readable
    snaps = ib.snapshots(p)
    assert len(snaps) >= 1


def test_iceberg_merge_into_delete(tmp_path):
    import bodo_amd.pandas as bpd
    from bodo_amd.io import iceberg as ib

    p = str(tmp_path / "tbl2")
    base = pd.DataFrame({"k": [1, 2, 3], "v": [1.0, 2.0, 3.0]})
    bpd.from_pandas(base).to_iceberg(p, mode="create")
    src = pd.DataFrame({"k": [2, 9], "v": [0.0, 9.0]})
    ib.merge_into(p, src, on="k", when_matched="delete",
                  when_not_matched="insert")
    got = bpd.read_iceberg(p).to_pandas().sort_values("k").reset_index(
        drop=True)
    exp = pd.DataFrame({"k": [1, 3, 9], "v": [1.0, 3.0, 9.0]})
    pd.testing.assert_frame_equal(got, exp, check_dtype=False)
