"""Out-of-core tier: partition-splitting hash operators with host staging.

When a local groupby/join working set exceeds the device budget, rows are
hash-partitioned on the SAME key hash the distributed shuffle uses but on
disjoint (high) bits, each partition is staged in host DRAM (288 GB HBM3E
on-device, ~TB host), and the operator runs partition-at-a-time on the
device.  Groups / join matches land in exactly one partition, so results
concatenate without a combine pass.

Reference analog: partition-splitting hash tables + spill in
bodo/libs/streaming/_join.h:267 (JoinPartition top-bitmask recursion),
streaming/_groupby.h:243, _storage_manager.h (spill tiers).  MI355X design:
the budget defaults to a fraction of free HBM; host DRAM is the spill tier
(NVMe would be next).
"""

from __future__ import annotations

import math
import os
from typing import List, Optional

import torch

from ..core.table import Table
from .. import config as cfg
from ..ops import hash_columns, take_table

_ENV = "BODO_AMD_OOC_BYTES"
_SPILL_ENV = "BODO_AMD_SPILL_DIR"


class _Staged:
    """A partition staged in host DRAM or, when BODO_AMD_SPILL_DIR is set,
    spilled to disk (the NVMe tier below host memory; reference:
    _storage_manager.h spill tiers).  Tables pickle cleanly: tensors via
    torch serialization, dictionaries via arrow pickling."""

    def __init__(self, tbl: Table):
        self._tbl = None
        self._path = None
        spill = os.environ.get(_SPILL_ENV, "")
        if spill:
            import tempfile

            os.makedirs(spill, exist_ok=True)
            fd, self._path = tempfile.mkstemp(dir=spill, suffix=".spill")
            os.close(fd)
            torch.save(tbl, self._path)
        else:
            self._tbl = tbl

    def load(self, device) -> Table:
        if self._path is not None:
            tbl = torch.load(self._path, weights_only=False)
            os.unlink(self._path)
            self._path = None
            return tbl.to_device(device)
        return self._tbl.to_device(device)


def budget_bytes(device) -> Optional[int]:
    """Per-operator device working-set budget; None = unlimited.  The free
    allowance divides across concurrently-live operators (comptroller,
    reference: _memory_budget.h OperatorComptroller)."""
    from . import comptroller

    v = os.environ.get(_ENV, "")
    if v:
        return comptroller.share(int(v))
    if getattr(cfg, "OOC_BYTES", None):
        return comptroller.share(int(cfg.OOC_BYTES))
    if device.type == "cuda":
        # available = device-free + blocks the caching allocator holds but
        # has not handed out (mem_get_info alone under-reports after the
        # first big query and sent steady-state runs into the spill path)
        free, _total = torch.cuda.mem_get_info()
        cached = torch.cuda.memory_reserved() - torch.cuda.memory_allocated()
        return comptroller.share(int((free + max(cached, 0)) * 0.5))
    return None


def _npartitions(total_bytes: int, budget: int) -> int:
    n = max(2, math.ceil(total_bytes / max(budget, 1)))
    return min(1 << (n - 1).bit_length(), 256)  # next pow2, capped


def partition_table(tbl: Table, keys, nparts: int) -> List[Table]:
    """Hash-partition on the HIGH bits of the row hash (the distributed
    shuffle consumes the low bits via h % world, so the two partitionings
    stay independent) and stage each part in host memory."""
    h = hash_columns([tbl.column(k) for k in keys])
    part = torch.remainder(h >> 32, nparts)
    part = torch.where(part < 0, part + nparts, part)
    out = []
    for p in range(nparts):
        idx = torch.nonzero(part == p, as_tuple=False).reshape(-1)
        out.append(_Staged(take_table(tbl, idx).to_device("cpu")))
    return out


def groupby_local(child: Table, keys, aggs, dropna) -> Table:
    """rel.groupby_local with partition-splitting when over budget."""
    from ..ops import concat_tables
    from ..ops import relational as rel
    from . import comptroller

    with comptroller.operator():
        return _groupby_local_inner(child, keys, aggs, dropna,
                                    concat_tables, rel)


def _groupby_local_inner(child, keys, aggs, dropna, concat_tables, rel):
    budget = budget_bytes(child.device)
    if not keys or budget is None or child.nbytes() <= budget:
        return rel.groupby_local(child, keys, aggs, dropna)
    nparts = _npartitions(child.nbytes(), budget)
    device = child.device
    parts = partition_table(child, keys, nparts)
    del child
    outs = []
    for p in parts:
        res = rel.groupby_local(p.load(device), keys, aggs, dropna)
        outs.append(res.to_device("cpu"))
    return concat_tables([o.to_device(device) for o in outs])


def join_local(left: Table, right: Table, left_on, right_on, how,
               suffixes) -> Table:
    """rel.join_local with partition-splitting when over budget (keyed joins
    only; each key lands in one partition so inner/left/right/semi/anti all
    decompose row-exactly)."""
    from ..ops import concat_tables
    from ..ops import relational as rel
    from . import comptroller

    with comptroller.operator():
        return _join_local_inner(left, right, left_on, right_on, how,
                                 suffixes, concat_tables, rel)


def _join_local_inner(left, right, left_on, right_on, how, suffixes,
                      concat_tables, rel):
    budget = budget_bytes(left.device)
    total = left.nbytes() + right.nbytes()
    if not left_on or budget is None or total <= budget or how == "cross":
        return rel.join_local(left, right, left_on, right_on, how, suffixes)
    nparts = _npartitions(total, budget)
    device = left.device
    lparts = partition_table(left, list(left_on), nparts)
    rparts = partition_table(right, list(right_on), nparts)
    del left, right
    outs = []
    for lp, rp in zip(lparts, rparts):
        res = rel.join_local(lp.load(device), rp.load(device),
                             left_on, right_on, how, suffixes)
        outs.append(res.to_device("cpu"))
    return concat_tables([o.to_device(device) for o in outs])


def sort_local(tbl: Table, keys, ascending, na_position) -> Table:
    """Local sort with partition-splitting when over budget: rows range-
    partition by sampled key bounds into host-staged runs, each run sorts
    on device, and runs concatenate in bound order — no merge pass needed
    (reference analog: external k-way merge sort, streaming/_sort.h:237;
    the range split replaces the merge)."""
    from ..ops import concat_tables, sort_indices, take_table

    budget = budget_bytes(tbl.device)
    if budget is None or tbl.nbytes() <= budget or not keys:
        idx = sort_indices([tbl.column(k) for k in keys], ascending,
                           na_position)
        return take_table(tbl, idx)
    nparts = _npartitions(tbl.nbytes(), budget)
    device = tbl.device
    # sample bounds on the first key only (ties stay within a run and are
    # resolved by the per-run full-key sort; equal-key spans across run
    # boundaries remain key-sorted because runs are contiguous ranges)
    first = tbl.column(keys[0])
    n = len(tbl)
    take = min(65536, n)
    step = max(1, n // take)
    sample = first.data[::step]
    order = torch.argsort(sample, stable=True)
    svals = sample[order]
    qs = [svals[int(i * (len(svals) - 1) / nparts)] for i in
          range(1, nparts)]
    asc0 = ascending[0] if ascending else True
    part = torch.zeros(n, dtype=torch.int64, device=device)
    for b in qs:
        part += (first.data > b).to(torch.int64) if asc0 else \
            (first.data < b).to(torch.int64)
    # null/NaN first keys route to the na_position run so the per-run sort
    # places them globally first/last
    invalid = None
    if first.dtype.is_float:
        invalid = torch.isnan(first.data)
    if first.mask is not None:
        miss = ~first.mask
        invalid = miss if invalid is None else (invalid | miss)
    if invalid is not None and bool(invalid.any().item()):
        na_run = nparts - 1 if na_position == "last" else 0
        part = torch.where(invalid, torch.tensor(na_run, device=device),
                           part)
    runs = []
    for p in range(nparts):
        idx = torch.nonzero(part == p, as_tuple=False).reshape(-1)
        runs.append(_Staged(take_table(tbl, idx).to_device("cpu")))
    del tbl
    outs = []
    for run in runs:
        dev_run = run.load(device)
        idx = sort_indices([dev_run.column(k) for k in keys], ascending,
                           na_position)
        outs.append(take_table(dev_run, idx).to_device("cpu"))
    return concat_tables([o.to_device(device) for o in outs])
