"""Query profile collector (reference: bodo/libs/_query_profile_collector.h:
per-pipeline/operator timers, row counts and metrics dumped as JSON per
rank).  Enabled with BODO_AMD_PROFILE=<dir> or programmatically."""

from __future__ import annotations

import json
import os
import time
from typing import Dict, List, Optional

_DIR = os.environ.get("BODO_AMD_PROFILE", "")
_RECORDS: List[dict] = []
_QUERY_SEQ = [0]


def enabled() -> bool:
    return bool(_DIR) or bool(_RECORDS is not None and _FORCED[0])


_FORCED = [False]


def enable(directory: Optional[str] = None):
    global _DIR
    _FORCED[0] = True
    if directory:
        _DIR = directory


def record_operator(op_name: str, duration_s: float, rows_in: int,
                    rows_out: int, **metrics):
    _RECORDS.append({
        "query": _QUERY_SEQ[0], "operator": op_name,
        "duration_s": duration_s, "rows_in": rows_in, "rows_out": rows_out,
        "metrics": metrics,
    })


class OpTimer:
    def __init__(self, name: str, rows_in: int = -1):
        self.name = name
        self.rows_in = rows_in
        self.rows_out = -1

    def __enter__(self):
        self.t0 = time.perf_counter()
        self.mem0 = _device_allocated() if (_FORCED[0] or _DIR) else 0
        return self

    def __exit__(self, *exc):
        if _FORCED[0] or _DIR:
            record_operator(self.name, time.perf_counter() - self.t0,
                            self.rows_in, self.rows_out,
                            hbm_delta_bytes=_device_allocated() - self.mem0)
        return False


def _device_allocated() -> int:
    import torch

    return torch.cuda.memory_allocated() if torch.cuda.is_available() else 0


def finish_query():
    _QUERY_SEQ[0] += 1
    if _DIR and _RECORDS:
        flush()


def flush(directory: Optional[str] = None):
    from ..parallel import comm

    d = directory or _DIR or "."
    os.makedirs(d, exist_ok=True)
    path = os.path.join(d, f"query_profile_rank{comm.get_rank()}.json")
    with open(path, "w") as f:
        json.dump({"records": _RECORDS}, f, indent=1)


def get_records() -> List[dict]:
    return list(_RECORDS)


def clear():
    _RECORDS.clear()
