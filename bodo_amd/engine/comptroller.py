"""Operator memory comptroller (reference: bodo/libs/_memory_budget.h:126
OperatorComptroller + _operator_pool.h): operators that hold large device
state (join builds, groupby tables, sort buffers, streaming build sides)
register while live; the out-of-core budget divides the free-HBM allowance
across the live set so concurrently-held states cannot each claim the whole
device.  MI355X note: the torch caching allocator owns physical HBM; this
layer only governs when operators choose the partition-split/spill path."""

from __future__ import annotations

import threading
from contextlib import contextmanager

_LOCK = threading.Lock()
_LIVE = 0
_PEAK = 0


def live_count() -> int:
    return max(1, _LIVE)


def peak() -> int:
    return _PEAK


@contextmanager
def operator():
    """Mark one budget-consuming operator live for the duration."""
    global _LIVE, _PEAK
    with _LOCK:
        _LIVE += 1
        _PEAK = max(_PEAK, _LIVE)
    try:
        yield
    finally:
        with _LOCK:
            _LIVE -= 1


def share(total_budget: int) -> int:
    """This operator's slice of the device budget."""
    return max(1, total_budget // live_count())
