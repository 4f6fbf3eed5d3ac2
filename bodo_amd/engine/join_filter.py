"""Runtime join filters: prune the probe side by the build side's key
range and a byte-table bloom BEFORE the expensive shuffle/probe
(reference: RuntimeJoinFilterPushdownOptimizer +
bodo/libs/gpu_bloom_filter.cu set/test kernels — here the filter runs as
plain tensor ops so the same code serves CPU and HBM tensors, and the
bitset exchange is one all-reduce over RCCL).

Safety: a runtime filter may only drop rows that can never match.
- inner: both sides prunable.
- left join: the RIGHT side is prunable (unmatched right rows produce
  nothing); the preserved left side is NOT.
- right join: mirror.
- semi: the left (output) side is prunable by right keys, and vice versa.
- anti: the LEFT side must NOT be pruned (non-matching rows are the
  output); the right side is prunable.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from ..core.table import Table
from .. import ops
from ..parallel import comm

#: probe side must have this many rows before a filter pays for itself
MIN_PROBE_ROWS = 1 << 16
#: byte-table bloom slots (1 byte each -> 4 MiB exchanged once)
BLOOM_SLOTS = 1 << 22


def _prunable_sides(how: str) -> Tuple[bool, bool]:
    """(left_prunable, right_prunable)."""
    return {
        "inner": (True, True),
        "left": (False, True),
        "right": (True, False),
        "semi": (True, True),
        "anti": (False, True),
    }.get(how, (False, False))


def _key_hash(tbl: Table, keys) -> torch.Tensor:
    return ops.hash_columns([tbl.column(k) for k in keys])


def _numeric_key(tbl: Table, key: str) -> Optional[torch.Tensor]:
    col = tbl.column(key)
    from ..core.types import TypeKind

    if col.dtype.kind in (TypeKind.STRING,):
        return None
    if col.mask is not None:
        return None
    return col.data


def apply_runtime_filters(left: Table, right: Table, left_on, right_on,
                          how: str, ctx) -> Tuple[Table, Table]:
    """Prune whichever sides are safely prunable; returns (left, right)."""
    lp, rp = _prunable_sides(how)
    if not (lp or rp) or not left_on:
        return left, right
    ln = sum(comm.allgather_obj(len(left))) if ctx.world > 1 else len(left)
    rn = sum(comm.allgather_obj(len(right))) if ctx.world > 1 else len(right)
    # filter the BIG side using the SMALL side's keys (4x imbalance so the
    # bitset build + exchange pays for itself)
    if lp and ln >= MIN_PROBE_ROWS and rn * 4 <= ln:
        left = _filter_by(left, left_on, right, right_on, ctx)
    elif rp and rn >= MIN_PROBE_ROWS and ln * 4 <= rn:
        right = _filter_by(right, right_on, left, left_on, ctx)
    return left, right


def _filter_by(probe: Table, probe_on, build: Table, build_on,
               ctx) -> Table:
    keep = None
    # min/max range filter on single numeric keys (nearly free)
    if len(probe_on) == 1:
        pk = _numeric_key(probe, probe_on[0])
        bk = _numeric_key(build, build_on[0])
        if pk is not None and bk is not None:
            has = len(bk) > 0
            lo = bk.min().item() if has else float("inf")
            hi = bk.max().item() if has else float("-inf")
            if ctx.world > 1:
                bounds = comm.allgather_obj((lo, hi))
                lo = min(b[0] for b in bounds)
                hi = max(b[1] for b in bounds)
            keep = (pk >= lo) & (pk <= hi)
    # byte-table bloom on the row hash (exact ops, ~n/4M false positives)
    bh = _key_hash(build, build_on)
    slots = torch.remainder(bh, BLOOM_SLOTS)
    table = torch.zeros(BLOOM_SLOTS, dtype=torch.int32, device=probe.device)
    if len(slots):
        table[slots] = 1
    if ctx.world > 1:
        comm.allreduce_max_(table)
    ph = _key_hash(probe, probe_on)
    hit = table[torch.remainder(ph, BLOOM_SLOTS)] != 0
    keep = hit if keep is None else (keep & hit)
    if bool(keep.all().item()):
        return probe
    from ..user_logging import log_message

    idx = torch.nonzero(keep, as_tuple=False).reshape(-1)
    log_message("Runtime Join Filter",
                f"pruned probe side {len(probe)} -> {int(idx.numel())} rows")
    return ops.take_table(probe, idx)
