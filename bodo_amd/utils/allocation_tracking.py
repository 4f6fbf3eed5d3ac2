"""Device-allocation tracking (reference: bodo/utils/allocation_tracking.py).
Thin wrapper over the torch caching allocator (our HBM BufferPool analog)."""

from __future__ import annotations

import torch


def get_allocation_stats() -> dict:
    if not torch.cuda.is_available():
        return {"allocated_bytes": 0, "reserved_bytes": 0, "peak_bytes": 0}
    return {
        "allocated_bytes": torch.cuda.memory_allocated(),
        "reserved_bytes": torch.cuda.memory_reserved(),
        "peak_bytes": torch.cuda.max_memory_allocated(),
    }


def reset_peak():
    if torch.cuda.is_available():
        torch.cuda.reset_peak_memory_stats()


def print_allocation_stats(prefix: str = ""):  # pragma: no cover
    s = get_allocation_stats()
    print(f"{prefix}allocated={s['allocated_bytes']/2**30:.2f}GiB "
          f"reserved={s['reserved_bytes']/2**30:.2f}GiB "
          f"peak={s['peak_bytes']/2**30:.2f}GiB")
