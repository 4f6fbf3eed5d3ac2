"""Event tracing in Chrome trace-event format (reference:
bodo/utils/tracing.pyx — per-rank event list, rank-0 aggregated dump).

Enable with BODO_AMD_TRACE=1; ``dump(path)`` gathers all ranks' events to
rank 0 and writes a chrome://tracing -compatible JSON file.
"""

from __future__ import annotations

import json
import os
import time
from typing import List, Optional

_EVENTS: List[dict] = []
_ENABLED = os.environ.get("BODO_AMD_TRACE", "0").lower() in ("1", "true")


def is_tracing() -> bool:
    return _ENABLED


def start_tracing():
    global _ENABLED
    _ENABLED = True
    _EVENTS.clear()


def stop_tracing():
    global _ENABLED
    _ENABLED = False


class Event:
    """with tracing.Event("shuffle", bytes=123): ..."""

    def __init__(self, name: str, is_batchable: bool = True, sync: bool = False,
                 **args):
        self.name = name
        self.args = args
        self._t0 = None

    def __enter__(self):
        if _ENABLED:
            self._t0 = time.perf_counter_ns()
        return self

    def add_attribute(self, key, value):
        self.args[key] = value

    def finalize(self):
        self.__exit__(None, None, None)

    def __exit__(self, *exc):
        if _ENABLED and self._t0 is not None:
            from ..parallel import comm

            _EVENTS.append({
                "name": self.name, "ph": "X", "pid": comm.get_rank(),
                "tid": 0, "ts": self._t0 / 1000.0,
                "dur": (time.perf_counter_ns() - self._t0) / 1000.0,
                "args": self.args,
            })
            self._t0 = None
        return False


def event(name, **args):
    return Event(name, **args)


def dump(path: str = "bodo_trace.json"):
    """Gather per-rank events to rank 0 and write one chrome trace file."""
    from ..parallel import comm

    parts = comm.gather_obj(_EVENTS, root=0)
    if comm.get_rank() == 0 and parts is not None:
        all_events = [e for p in parts for e in p]
        with open(path, "w") as f:
            json.dump({"traceEvents": all_events}, f)
    _EVENTS.clear()


def aggregate_events() -> List[dict]:
    """Min/max/avg duration per event name across ranks (reference:
    tracing.pyx aggregate_events)."""
    from collections import defaultdict

    from ..parallel import comm

    parts = comm.allgather_obj(_EVENTS)
    agg = defaultdict(list)
    for p in parts:
        for e in p:
            agg[e["name"]].append(e["dur"])
    return [{"name": k, "count": len(v), "avg_us": sum(v) / len(v),
             "min_us": min(v), "max_us": max(v)} for k, v in agg.items()]
