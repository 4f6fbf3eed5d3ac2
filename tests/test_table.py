"""CPU-exercisable regression tests for Table device movement.

Round-1 postmortem: ``Table.__slots__`` omitted ``_dev_cache`` so the first
``to_device`` to a different device raised AttributeError — but only on a GPU
box, because ``to_device("cpu")`` on a cpu table returns early.  These tests
force the cross-device cache path on CPU via torch's ``meta`` device so the
plain pytest run catches any recurrence.
"""

import pandas as pd
import torch

from bodo_amd.core.table import Table, _normalize_device


def _mk():
    return Table.from_pandas(pd.DataFrame({
        "a": [1, 2, 3], "b": [1.5, 2.5, None], "s": ["x", "yy", "zzz"]}))


def test_to_device_cross_device_cache_path():
    t = _mk()
    # meta is a real "different device" on a CPU-only box: exercises the
    # slotted _dev_cache attribute exactly like a cuda upload would
    m = t.to_device("meta")
    assert m is not t
    assert m.columns[0].data.device.type == "meta"
    assert len(m) == len(t)
    # second request must hit the cache (no re-copy)
    assert t.to_device("meta") is m


def test_to_device_same_device_returns_self():
    t = _mk()
    assert t.to_device("cpu") is t


def test_to_device_empty_table():
    t = Table([], [], length=5)
    assert t.to_device("meta") is not None
    assert len(t.to_device("meta")) == 5


def test_slots_has_dev_cache():
    # the literal round-1 failure mode: slotted class must carry the cache
    assert "_dev_cache" in Table.__slots__
    t = _mk()
    assert t._dev_cache == {}


def test_normalize_device_cuda_index():
    d = _normalize_device("cuda")
    assert d.type == "cuda" and d.index is not None
    assert _normalize_device("cpu") == torch.device("cpu")


def test_struct_column_roundtrip_and_ops():
    """STRUCT columns: arrow roundtrip, gather, concat, field extraction
    (reference: struct_arr_ext.py layout)."""
    import pyarrow as pa
    import torch

    from bodo_amd import ops
    from bodo_amd.core.column import Column

    a = pa.array([{"x": 1, "y": "p"}, {"x": 2, "y": "q"}, None,
                  {"x": 4, "y": None}])
    c = Column.from_arrow(a)
    assert len(c) == 4 and c.dtype.fields == ("x", "y")
    assert c.to_arrow().to_pylist() == a.to_pylist()
    g = ops.gather(c, torch.tensor([3, 1, 2]))
    assert g.to_arrow().to_pylist() == [a[3].as_py(), a[1].as_py(), None]
    cc = ops.concat_columns([c, g])
    assert len(cc) == 7 and cc.to_arrow().to_pylist()[4] == a[3].as_py()


def test_struct_frontend_and_sql():
    import bodo_amd.pandas as bpd
    from bodo_amd.sql import BodoSQLContext

    df = pd.DataFrame({"k": [1, 2, 3, 4],
                       "st": [{"x": 1, "y": "p"}, {"x": 2, "y": "q"},
                              None, {"x": 4, "y": None}]})
    b = bpd.from_pandas(df)
    assert b["st"].struct.field("x").to_pandas().fillna(-1).tolist() == \
        [1, 2, -1, 4]
    bc = BodoSQLContext({"t": df})
    o = bc.sql("select k, get(st, 'y') as y from t order by k").to_pandas()
    assert o["y"].where(o["y"].notna(), None).tolist() == \
        ["p", "q", None, None]
    o2 = bc.sql("select k, st from t where k >= 2 order by k desc") \
        .to_pandas()
    assert o2["st"].tolist()[0] == {"x": 4, "y": None}


def test_map_column_via_list_struct():
    """MAP columns load as list<struct<key,value>> (the physical arrow
    layout; reference: map_arr_ext.py)."""
    import pyarrow as pa

    from bodo_amd.core.column import Column

    m = pa.array([[("a", 1), ("b", 2)], None, [("c", 3)]],
                 type=pa.map_(pa.string(), pa.int64()))
    c = Column.from_arrow(m)
    assert len(c) == 3
    assert c.to_arrow().to_pylist() == [
        [{"key": "a", "value": 1}, {"key": "b", "value": 2}], None,
        [{"key": "c", "value": 3}]]


def test_binary_columns():
    """BINARY columns share the STRING offsets+bytes layout (precision=1
    flags arrow large_binary round-trip; reference: binary_arr_ext.py)."""
    import bodo_amd.pandas as bpd

    df = pd.DataFrame({"b": [b"ab", b"c", None, b"\x00\xffbin"],
                       "k": [1, 2, 3, 4]})
    b = bpd.from_pandas(df)
    assert b.to_pandas()["b"].tolist() == df["b"].tolist()
    assert b[b["k"] > 1].to_pandas()["b"].tolist() == df["b"].tolist()[1:]
    out = b.sort_values("k", ascending=False).to_pandas()["b"].tolist()
    assert out == df["b"].tolist()[::-1]


def test_timedelta_columns_and_arith():
    """DURATION_NS: timedelta64 round trip; ts - ts yields timedelta
    (value semantics, not raw ns) (reference: pd_timedelta_ext)."""
    import bodo_amd.pandas as bpd

    df = pd.DataFrame({"d1": pd.to_datetime(["2024-01-05", "2024-02-01"]),
                       "d2": pd.to_datetime(["2024-01-01", "2024-01-15"])})
    b = bpd.from_pandas(df)
    delta = (b["d1"] - b["d2"]).to_pandas()
    assert str(delta.dtype) == "timedelta64[ns]"
    assert delta.tolist() == (df["d1"] - df["d2"]).tolist()
    df2 = pd.DataFrame({"td": pd.to_timedelta(["1 days", "3 hours", None]),
                        "k": [1, 2, 3]})
    b2 = bpd.from_pandas(df2)
    assert b2.to_pandas()["td"].tolist()[:2] == df2["td"].tolist()[:2]
    assert pd.Timedelta(b2["td"].max().value) == pd.Timedelta("1 days")
