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
