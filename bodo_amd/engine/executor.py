"""Plan executor: interprets an optimized logical plan against the rank-local
shard, inserting distributed exchanges (RCCL) where operators need them.

Reference role: bodo/pandas/_executor.cpp + _physical_conv.cpp + the streaming
operator states — redesigned for MI355X as a materialized (whole-shard)
columnar executor first: with 288 GB HBM3E per GPU the SF100-class working
sets fit resident, so operators run one fused device pass per shard instead
of 32k-row host morsels; out-of-core streaming is layered on top for
larger-than-HBM inputs (see engine/streaming notes).
"""

from __future__ import annotations

import uuid
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import pandas as pd
import torch

from .. import ops
from ..core import types as bt
from ..core.column import Column
from ..core.table import Table
from ..core.types import TypeKind
from ..ops import evaluate as ev
from ..ops import relational as rel
from ..parallel import comm
from ..plan import nodes as pn
from ..plan.expr import BinOp, ColRef, Const, Expr

# registry of in-memory source objects (host DataFrames or distributed shards)
_OBJECT_REGISTRY: Dict[str, object] = {}


def register_object(obj, data_id: Optional[str] = None) -> str:
    key = data_id or uuid.uuid4().hex
    _OBJECT_REGISTRY[key] = obj
    return key


def get_object(data_id: str):
    return _OBJECT_REGISTRY[data_id]


def delete_object(data_id: str):
    _OBJECT_REGISTRY.pop(data_id, None)


class ExecutionContext:
    def __init__(self, device: Optional[str] = None):
        from .. import config

        if device is None:
            device = config.default_device()
        self.device = torch.device(device)
        self.rank = comm.get_rank()
        self.world = comm.get_world_size()
        # per-execution memo of repeated identical subtrees (reference
        # analog: LogicalMaterializedCTE / bfs_duplicate dedup,
        # bodo/pandas/plan.py:103): q21-style self-join queries re-derive
        # the same scan+filter chain several times otherwise
        self.memo: Dict[object, Table] = {}


def execute(plan: pn.PlanNode, ctx: Optional[ExecutionContext] = None) -> Table:
    """Execute the plan; returns the rank-local shard of the result."""
    from .optimizer import optimize

    ctx = ctx or ExecutionContext()
    plan = optimize(plan)
    out = _exec(plan, ctx)
    from ..utils import query_profile as qp

    qp.finish_query()
    return out


_NO_MEMO = (pn.Sample, pn.MapPartitions, pn.ParquetWrite, pn.IcebergWrite)


def _memo_key(node: pn.PlanNode):
    if isinstance(node, _NO_MEMO):
        return None
    try:
        hash(node)
    except TypeError:
        return None
    return node


def _exec(node: pn.PlanNode, ctx: ExecutionContext) -> Table:
    h = _HANDLERS.get(type(node))
    if h is None:
        raise NotImplementedError(f"no executor for {type(node).__name__}")
    key = _memo_key(node)
    memo = getattr(ctx, "memo", None)
    if memo is not None and key is not None and key in memo:
        return memo[key]
    from ..utils import query_profile as qp
    from ..utils import tracing

    from ..utils import roctx

    name = type(node).__name__
    with tracing.Event(f"exec.{name}"), roctx.Range(f"bodo.{name}"), \
            qp.OpTimer(name) as t:
        out = h(node, ctx)
        from .. import config as _cfg

        if _cfg.DEBUG_SYNC and ctx.device.type == "cuda":
            # surface async HIP faults at the operator that queued them
            torch.cuda.synchronize()
        t.rows_out = len(out) if out is not None else -1
    if memo is not None and key is not None:
        memo[key] = out
    return out


# ---------------------------------------------------------------- sources

def _exec_parquet(node: pn.ParquetScan, ctx) -> Table:
    from ..io import parquet as pio

    return pio.read_shard(node.path, node.columns, node.filters, ctx)


def _exec_csv(node: pn.CsvScan, ctx) -> Table:
    from ..io import csv as cio

    return cio.read_shard(node.path, dict(node.options), node.columns, ctx)


def _exec_pandas_scan(node: pn.PandasScan, ctx) -> Table:
    obj = get_object(node.data_id)
    if node.distributed:
        assert isinstance(obj, Table)
        return obj.to_device(ctx.device)
    if isinstance(obj, Table):
        tbl = obj
    else:
        # convert once: repeated executions over the same registered source
        # (one per query run otherwise) were ~50ms of arrow round-trips
        tbl = Table.from_pandas(obj)
        _OBJECT_REGISTRY[node.data_id] = tbl
    # replicated host data: slice this rank's block (1D distribution)
    n = len(tbl)
    w, r = ctx.world, ctx.rank
    start, stop = _block_bounds(n, w, r)
    local = ops.slice_table(tbl, start, stop) if w > 1 else tbl
    return local.to_device(ctx.device)


def _block_bounds(n: int, w: int, r: int) -> Tuple[int, int]:
    base, rem = divmod(n, w)
    start = r * base + min(r, rem)
    stop = start + base + (1 if r < rem else 0)
    return start, stop


# ---------------------------------------------------------------- unary

def _exec_projection(node: pn.Projection, ctx) -> Table:
    child = _exec(node.child, ctx)
    return ev.project(child, node.names, node.exprs)


def _exec_filter(node: pn.Filter, ctx) -> Table:
    child = _exec(node.child, ctx)
    return ev.eval_filter(node.cond, child)


SINGLE_PHASE_AGGS = {"median", "nunique", "approx_nunique", "var", "std",
                     "quantile", "skew", "kurt", "sem", "mode",
                     "array_agg"}


def _exec_aggregate(node: pn.Aggregate, ctx) -> Table:
    from .streaming import exec_streaming, want_streaming

    if want_streaming(node, ctx):
        return exec_streaming(node, ctx)
    child = _exec(node.child, ctx)
    keys = list(node.keys)
    aggs = list(node.aggs)
    from . import ooc

    if ctx.world == 1:
        return ooc.groupby_local(child, keys, aggs, node.dropna)
    if any(a[2] in SINGLE_PHASE_AGGS or callable(a[2]) for a in aggs):
        # shuffle raw rows by key hash, then single local groupby
        shuffled = _shuffle_by_keys(child, keys, ctx)
        return ooc.groupby_local(shuffled, keys, aggs, node.dropna)
    # two-phase: local pre-agg -> shuffle groups -> combine
    partial_aggs, final_map = _decompose_aggs(aggs)
    local = ooc.groupby_local(child, keys, partial_aggs, node.dropna)
    shuffled = _shuffle_by_keys(local, keys, ctx)
    combine_aggs = [(name, name, rel.COMBINE_FUNC[f]) for name, _, f in partial_aggs]
    combined = rel.groupby_local(shuffled, keys, combine_aggs, node.dropna)
    # final projection (mean = sum/count etc.)
    names, exprs = list(keys), [ColRef(k) for k in keys]
    for out_name, expr in final_map:
        names.append(out_name)
        exprs.append(expr)
    return ev.project(combined, names, exprs)


def _decompose_aggs(aggs):
    """Split each agg into partial physical aggs + a final expression."""
    partials: List[Tuple[str, str, str]] = []
    finals: List[Tuple[str, Expr]] = []
    seen = {}

    def add_partial(in_name, func):
        key = (in_name, func)
        if key not in seen:
            nm = f"__p{len(partials)}_{func}"
            seen[key] = nm
            partials.append((nm, in_name, func))
        return seen[key]

    for out_name, in_name, func in aggs:
        if func == "mean":
            s = add_partial(in_name, "sum")
            c = add_partial(in_name, "count")
            finals.append((out_name, BinOp("div", ColRef(s), ColRef(c))))
        elif func in ("sum", "min", "max", "count", "size", "first", "last",
                      "prod", "any", "all"):
            p = add_partial(in_name, func)
            finals.append((out_name, ColRef(p)))
        else:
            raise NotImplementedError(f"two-phase agg {func}")
    return partials, finals


def _shuffle_by_keys(tbl: Table, keys: Sequence[str], ctx) -> Table:
    cols = [tbl.column(k) for k in keys]
    h = ops.hash_columns(cols)
    part = torch.remainder(h, ctx.world)
    part = torch.where(part < 0, part + ctx.world, part)
    return comm.shuffle_table(tbl, part)


def _exec_sort(node: pn.Sort, ctx) -> Table:
    child = _exec(node.child, ctx)
    asc = list(node.ascending) or [True] * len(node.keys)
    if ctx.world > 1:
        child = _range_partition(child, list(node.keys), asc, node.na_position, ctx)
    from . import ooc

    return ooc.sort_local(child, list(node.keys), asc, node.na_position)


def _float_sort_key(col: Column, asc: bool, na_position: str):
    """Order-preserving int64 view of a float sort key (sortable-bits
    transform) so single-float-key range partitions stay on device
    (round-1 finding: non-packable splitters compared per row in host
    pandas).  NaN maps to the na_position end; descending inverts."""
    if not col.dtype.is_float or col.mask is not None:
        return None
    d = col.data
    x = d.double() if d.dtype != torch.float64 else d
    bits = x.view(torch.int64)
    # IEEE-754 total order: positive floats flip the sign bit; negatives
    # flip all bits
    sortable = torch.where(bits >= 0, bits ^ torch.tensor(
        -0x8000000000000000, device=d.device), ~bits)
    if not asc:
        sortable = ~sortable  # order-reversing, overflow-free
    nan = torch.isnan(x)
    sent = torch.tensor(
        (2 ** 62) if na_position == "last" else -(2 ** 62),
        device=d.device)
    return torch.where(nan, sent, sortable)


def _sample_positions(n: int, k: int, ctx, device) -> torch.Tensor:
    """Seeded per-rank sample positions so distributed sorts are reproducible
    across runs (splitters otherwise change run to run)."""
    if n <= k:
        return torch.arange(n, device=device)
    g = torch.Generator(device="cpu")
    g.manual_seed(0x5EED5 + 1315423911 * (ctx.rank + 1))
    return torch.randint(0, n, (k,), generator=g).to(device)


def _range_partition(tbl: Table, keys, asc, na_position, ctx) -> Table:
    """Sample-based range partitioning so rank r holds globally-contiguous key
    range r (PSRS step 1-3; reference: streaming/_sort.h reservoir sampling +
    bounds).  Packable key sets partition entirely on device (sampled packed
    keys -> allgathered splitters -> device searchsorted)."""
    w = ctx.world
    n = len(tbl)
    packed = ops.pack_ordered_keys([tbl.column(k) for k in keys], asc)
    if packed is None and len(keys) == 1:
        packed = _float_sort_key(tbl.column(keys[0]), asc[0], na_position)
    if packed is not None:
        k = min(n, 64 * w)
        if n > 0:
            pos = _sample_positions(n, k, ctx, tbl.device)
            sample = packed[pos].cpu().numpy()
        else:
            sample = np.zeros(0, dtype=np.int64)
        merged = np.sort(np.concatenate(comm.allgather_obj(sample)))
        if len(merged) == 0:
            return tbl
        q = [min(int(len(merged) * (i + 1) / w), len(merged) - 1)
             for i in range(w - 1)]
        splitters = torch.from_numpy(merged[q].copy()).to(tbl.device)
        part = torch.searchsorted(splitters, packed, right=True)
        return comm.shuffle_table(tbl, part)
    # sample up to 64*w rows of the key columns
    k = min(n, 64 * w)
    if k > 0:
        pos = _sample_positions(n, k, ctx, tbl.device)
        sample = ops.take_table(tbl.select(list(keys)), pos).to_device("cpu").to_pandas()
    else:
        sample = tbl.select(list(keys)).to_device("cpu").to_pandas()
    all_samples = comm.allgather_obj(sample)
    merged = _decat(pd.concat(all_samples, ignore_index=True))
    merged = merged.sort_values(list(keys), ascending=asc, na_position=na_position)
    if len(merged) == 0:
        return tbl
    # pick w-1 splitters
    q = [int(len(merged) * (i + 1) / w) for i in range(w - 1)]
    splitters = merged.iloc[[min(x, len(merged) - 1) for x in q]]
    # assign each local row a partition by comparing against splitters.
    # host comparison on the key sample of local rows (vectorized via
    # pandas merge_sorted rank); for performance the GPU path will move to a
    # device searchsorted on normalized keys.
    local_keys = _decat(tbl.select(list(keys)).to_device("cpu").to_pandas())
    part = np.zeros(n, dtype=np.int64)
    for i in range(w - 1):
        row = splitters.iloc[i]
        gt = _row_greater(local_keys, row, keys, asc, na_position)
        part = np.where(gt, i + 1, part)
    return comm.shuffle_table(tbl, torch.from_numpy(part).to(tbl.device))


def _decat(df: pd.DataFrame) -> pd.DataFrame:
    """Categoricals -> strings so splitter comparisons are well-defined."""
    out = df
    for c in df.columns:
        if isinstance(df[c].dtype, pd.CategoricalDtype):
            if out is df:
                out = df.copy()
            out[c] = out[c].astype("string")
    return out


def _row_greater(df: pd.DataFrame, row, keys, asc, na_position) -> np.ndarray:
    """True where df-row sorts strictly after `row` under keys/asc."""
    n = len(df)
    gt = np.zeros(n, dtype=bool)
    eq = np.ones(n, dtype=bool)
    for k, a in zip(keys, asc):
        col = df[k]
        rv = row[k]
        colna = pd.isna(col).to_numpy()
        rvna = pd.isna(rv)
        if rvna:
            c_gt = np.zeros(n, dtype=bool) if na_position == "last" else ~colna
            c_eq = colna
        else:
            with np.errstate(invalid="ignore"):
                raw_gt = (col > rv).to_numpy(dtype=bool, na_value=False) if a else (col < rv).to_numpy(dtype=bool, na_value=False)
                c_eq = (col == rv).to_numpy(dtype=bool, na_value=False)
            c_gt = np.where(colna, na_position == "last", raw_gt)
        gt |= eq & c_gt
        eq &= c_eq
    return gt


_TOPK_MAX = 1_000_000


def _exec_limit(node: pn.Limit, ctx) -> Table:
    # top-k fast path: Limit(Sort) with small n computes each rank's local
    # top-k and resolves globally on rank-replicated candidates — no
    # range-partition exchange of the full table (reference analog: TopN,
    # bodo/pandas/plan.py LogicalTopN:474)
    if isinstance(node.child, pn.Sort) and not node.tail and \
            node.offset == 0 and 0 < node.n <= _TOPK_MAX and ctx.world > 1:
        srt = node.child
        child = _exec(srt.child, ctx)
        asc = list(srt.ascending) or [True] * len(srt.keys)
        cols = [child.column(k) for k in srt.keys]
        idx = ops.sort_indices(cols, asc, srt.na_position)
        local_top = ops.take_table(child, idx[:node.n])
        full = comm.allgather_table(local_top)
        cols = [full.column(k) for k in srt.keys]
        idx = ops.sort_indices(cols, asc, srt.na_position)
        merged = ops.take_table(full, idx[:node.n])
        # keep the result distributed: this rank's block of the k rows
        start, stop = _block_bounds(len(merged), ctx.world, ctx.rank)
        return ops.slice_table(merged, start, stop)
    child = _exec(node.child, ctx)
    n = node.n
    if ctx.world == 1:
        if node.tail:
            return ops.slice_table(child, max(0, len(child) - n), len(child))
        return ops.slice_table(child, node.offset, node.offset + n)
    lengths = comm.allgather_obj(len(child))
    if node.tail:
        suffix_after = sum(lengths[ctx.rank + 1:])
        take = max(0, min(len(child), n - suffix_after))
        return ops.slice_table(child, len(child) - take, len(child))
    prefix = sum(lengths[:ctx.rank])
    start = max(0, node.offset - prefix)
    stop = max(0, node.offset + n - prefix)
    return ops.slice_table(child, start, min(stop, len(child)))


def _exec_distinct(node: pn.Distinct, ctx) -> Table:
    child = _exec(node.child, ctx)
    if ctx.world == 1:
        return rel.distinct_local(child, node.subset, node.keep)
    # keep=False must see every occurrence of a key to know whether it is
    # duplicated anywhere: a pre-shuffle local distinct destroys per-key
    # counts (a key with 2 rows here + 1 elsewhere would wrongly survive).
    # Shuffle the full table by key first, then drop once.
    if node.keep is False:
        keys = list(node.subset) if node.subset else list(child.names)
        shuffled = _shuffle_by_keys(child, keys, ctx)
        return rel.distinct_local(shuffled, node.subset, node.keep)
    # keep='first'/'last': local pass is a safe reducer because the shuffle
    # concatenates inbound shards in source-rank order, preserving global
    # row order per key across the two passes.
    local = rel.distinct_local(child, node.subset, node.keep)
    keys = list(node.subset) if node.subset else list(local.names)
    shuffled = _shuffle_by_keys(local, keys, ctx)
    return rel.distinct_local(shuffled, node.subset, node.keep)


def _exec_explode(node: pn.Explode, ctx) -> Table:
    child = _exec(node.child, ctx)
    return ops.explode_table(child, node.column, pos_name=node.pos)


def _exec_sample(node: pn.Sample, ctx) -> Table:
    child = _exec(node.child, ctx)
    n_local = len(child)
    if node.frac is not None:
        k = int(round(n_local * node.frac))
    else:
        total = sum(comm.allgather_obj(n_local))
        share = (node.n or 0) * n_local / max(total, 1)
        k = int(round(share))
    g = torch.Generator(device="cpu")
    if node.seed is not None:
        g.manual_seed(node.seed + ctx.rank)
    pos = torch.randperm(n_local, generator=g)[:k].to(child.device)
    return ops.take_table(child, torch.sort(pos).values)


def _exec_window(node: pn.Window, ctx) -> Table:
    from .window import exec_window

    child = _exec(node.child, ctx)
    return exec_window(node, ctx, child)


def _exec_shuffle_by_key(node: pn.ShuffleByKey, ctx) -> Table:
    child = _exec(node.child, ctx)
    if ctx.world == 1:
        return child
    return _shuffle_by_keys(child, list(node.keys), ctx)


def _exec_rowid(node: pn.RowId, ctx) -> Table:
    child = _exec(node.child, ctx)
    n = len(child)
    base = 0
    if ctx.world > 1:
        lengths = comm.allgather_obj(n)
        base = sum(lengths[:ctx.rank])
    rid = torch.arange(base, base + n, dtype=torch.int64, device=child.device)
    from ..core import types as bt
    from ..core.column import Column

    return child.with_column(node.name, Column(bt.int64, rid))


def _gather_halo(data: torch.Tensor, mask, k: int, ctx):
    """Collect |k| boundary rows from neighbor ranks as tensors: for k>0
    the tail of preceding ranks, for k<0 the head of following ranks
    (reference: border exchange in the distributed pass for shifts)."""
    n = int(data.numel())
    if k > 0:
        edge = data[-min(k, n):]
        emask = None if mask is None else mask[-min(k, n):]
    else:
        edge = data[:min(-k, n)]
        emask = None if mask is None else mask[:min(-k, n)]
    edges = comm.allgather_obj((edge.cpu(), None if emask is None
                                else emask.cpu()))
    pieces = []
    need = abs(k)
    have = 0
    if k > 0:
        i = ctx.rank - 1
        while have < need and i >= 0:
            t, m = edges[i]
            take = min(need - have, int(t.numel()))
            pieces.insert(0, (t[-take:] if take else t[:0],
                              None if m is None else m[-take:]))
            have += take
            if int(t.numel()) >= need:
                break
            i -= 1
    else:
        i = ctx.rank + 1
        while have < need and i < ctx.world:
            t, m = edges[i]
            take = min(need - have, int(t.numel()))
            pieces.append((t[:take], None if m is None else m[:take]))
            have += take
            if int(t.numel()) >= need:
                break
            i += 1
    if not pieces:
        return data[:0], None if mask is None else mask[:0]
    hd = torch.cat([p[0] for p in pieces]).to(data.device)
    if mask is not None or any(p[1] is not None for p in pieces):
        hm = torch.cat([
            p[1] if p[1] is not None
            else torch.ones(int(p[0].numel()), dtype=torch.bool)
            for p in pieces]).to(data.device)
    else:
        hm = None
    return hd, hm


def _shift_column_device(col: Column, k: int, ctx) -> Optional[Column]:
    """Device shift for fixed-width columns: slice + halo, null-padded ends
    (no host round trip)."""
    if col.dtype.kind in (TypeKind.STRING,):
        return None
    n = len(col)
    dev = col.device
    data, mask = col.data, col.mask
    if k == 0:
        return col
    if n == 0:
        # still participate in the halo allgather (SPMD sequence)
        if ctx.world > 1:
            _gather_halo(data, mask, k, ctx)
        return col
    halo_d, halo_m = (data[:0], None)
    if ctx.world > 1:
        halo_d, halo_m = _gather_halo(data, mask, k, ctx)
    h = int(halo_d.numel())
    out = torch.zeros(n, dtype=data.dtype, device=dev)
    valid = torch.zeros(n, dtype=torch.bool, device=dev)
    if k > 0:
        pad = k - h  # rows with no predecessor anywhere: null
        if h:
            out[pad:pad + h] = halo_d
            valid[pad:pad + h] = halo_m if halo_m is not None else True
        m = n - k
        if m > 0:
            out[k:] = data[:m]
            valid[k:] = mask[:m] if mask is not None else True
    else:
        kk = -k
        m = n - kk
        if m > 0:
            out[:m] = data[kk:]
            valid[:m] = mask[kk:] if mask is not None else True
        if h:
            out[m:m + h] = halo_d
            valid[m:m + h] = halo_m if halo_m is not None else True
    if col.dtype.kind == TypeKind.DICT:
        res = Column(col.dtype, out.to(torch.int32), valid,
                     dictionary=col.dictionary, length=n)
        return res
    if col.dtype.is_float:
        out = torch.where(valid, out, torch.tensor(
            float("nan"), dtype=out.dtype, device=dev))
        return Column(col.dtype, out)
    return Column(col.dtype, out, valid)


def _exec_shift(node: pn.Shift, ctx) -> Table:
    child = _exec(node.child, ctx)
    n = len(child)
    k = int(node.periods)
    names, cols = [], []
    for out_name, in_name in node.specs:
        col = child.column(in_name)
        dcol = _shift_column_device(col, k, ctx)
        if dcol is not None:
            names.append(out_name)
            cols.append(dcol)
            continue
        ser = col.to_pandas()
        if ctx.world > 1 and k != 0:
            # exchange |k| boundary rows: for k>0 take the tail of preceding
            # ranks, for k<0 the head of following ranks
            if k > 0:
                edge = ser.tail(min(k, n))
            else:
                edge = ser.head(min(-k, n))
            edges = comm.allgather_obj(edge)
            if k > 0:
                pieces, have, i = [], 0, ctx.rank - 1
                while have < k and i >= 0:
                    t = edges[i].tail(k - have)
                    pieces.insert(0, t)
                    have += len(t)
                    if len(edges[i]) >= k:
                        break
                    i -= 1
                halo = pd.concat(pieces) if pieces else ser.iloc[0:0]
                full = pd.concat([halo, ser], ignore_index=True)
                res = full.shift(k).iloc[len(halo):].reset_index(drop=True)
            else:
                m = -k
                pieces, have, i = [], 0, ctx.rank + 1
                while have < m and i < ctx.world:
                    t = edges[i].head(m - have)
                    pieces.append(t)
                    have += len(t)
                    if len(edges[i]) >= m:
                        break
                    i += 1
                halo = pd.concat(pieces) if pieces else ser.iloc[0:0]
                full = pd.concat([ser, halo], ignore_index=True)
                res = full.shift(k).iloc[:n].reset_index(drop=True)
        else:
            res = ser.shift(k)
        names.append(out_name)
        cols.append(Column.from_numpy(res.to_numpy(), ctx.device))
    out = child
    for nm, c in zip(names, cols):
        out = out.with_column(nm, c)
    return out


def _fill_column_device(col: Column, forward: bool, ctx) -> Optional[Column]:
    """Device ffill/bfill: last-valid-index via cummax gather; cross-rank
    carry as one small object round (reference: array_kernels fillna
    forward/backward with border exchange)."""
    if col.dtype.kind in (TypeKind.STRING, TypeKind.DICT):
        return None
    n = len(col)
    dev = col.device
    data, mask = col.data, col.mask
    valid = torch.ones(n, dtype=torch.bool, device=dev)
    if mask is not None:
        valid = mask.clone()
    if col.dtype.is_float:
        valid &= ~torch.isnan(data)
    pos = torch.arange(n, device=dev)
    if forward:
        marked = torch.where(valid, pos, torch.full_like(pos, -1))
        last = torch.cummax(marked, 0).values
    else:
        # nearest valid at-or-after i == forward pass in the flipped domain
        v_f = torch.flip(valid, [0])
        m_f = torch.where(v_f, pos, torch.full_like(pos, -1))
        last_f = torch.cummax(m_f, 0).values
        last = torch.flip(torch.where(last_f >= 0, (n - 1) - last_f,
                                      torch.full_like(last_f, -1)), [0])
    have = last >= 0
    out = data[last.clamp(min=0)]
    out_valid = valid | have
    # cross-rank carry of the edge value
    if ctx.world > 1:
        if forward:
            edge = None
            if n and bool(have[-1].item()):
                edge = data[last[-1]].item()
            edges = comm.allgather_obj(edge)
            carry = None
            for v in edges[:ctx.rank]:
                if v is not None:
                    carry = v
        else:
            edge = None
            if n and bool(have[0].item()):
                edge = data[last[0]].item()
            edges = comm.allgather_obj(edge)
            carry = None
            for v in edges[ctx.rank + 1:]:
                if v is not None:
                    carry = v
                    break
        if carry is not None:
            fillv = torch.tensor(carry, dtype=out.dtype, device=dev)
            out = torch.where(out_valid, out, fillv)
            out_valid = torch.ones_like(out_valid)
    if col.dtype.is_float:
        out = torch.where(out_valid, out, torch.tensor(
            float("nan"), dtype=out.dtype, device=dev))
        return Column(col.dtype, out)
    if bool(out_valid.all().item()):
        return Column(col.dtype, out)
    return Column(col.dtype, out, out_valid)


def _exec_fill(node: "pn.Fill", ctx) -> Table:
    """ffill/bfill over the global row order: local fill plus the nearest
    valid value carried across rank boundaries (reference: array_kernels
    fillna forward/backward with border exchange)."""
    child = _exec(node.child, ctx)
    names, cols = [], []
    for out_name, in_name in node.specs:
        dcol = _fill_column_device(child.column(in_name), node.forward, ctx)
        if dcol is not None:
            names.append(out_name)
            cols.append(dcol)
            continue
        ser = child.column(in_name).to_pandas()
        filled = ser.ffill() if node.forward else ser.bfill()
        if ctx.world > 1:
            if node.forward:
                last = filled.iloc[-1] if len(filled) else None
                edges = comm.allgather_obj(
                    None if last is None or pd.isna(last) else last)
                carry = None
                for v in edges[:ctx.rank]:
                    if v is not None:
                        carry = v
                if carry is not None:
                    filled = filled.fillna(carry)
            else:
                first = filled.iloc[0] if len(filled) else None
                edges = comm.allgather_obj(
                    None if first is None or pd.isna(first) else first)
                carry = None
                for v in edges[ctx.rank + 1:]:
                    if v is not None:
                        carry = v
                        break
                if carry is not None:
                    filled = filled.fillna(carry)
        names.append(out_name)
        cols.append(Column.from_numpy(filled.to_numpy(), ctx.device))
    out = child
    for nm, c in zip(names, cols):
        out = out.with_column(nm, c)
    return out


_CUM_IDENT = {"cumsum": 0.0, "cumprod": 1.0,
              "cummin": float("inf"), "cummax": float("-inf")}


def _exec_cumulative(node: pn.Cumulative, ctx) -> Table:
    child = _exec(node.child, ctx)
    n = len(child)
    names, cols = [], []
    for out_name, in_name, func in node.specs:
        col = child.column(in_name)
        data = col.data
        invalid = None
        if col.dtype.is_float:
            invalid = torch.isnan(data)
        if col.mask is not None:
            miss = ~col.mask
            invalid = miss if invalid is None else (invalid | miss)
        work = data.to(torch.float64) if invalid is not None else data
        ident = _CUM_IDENT[func]
        if invalid is not None:
            work = torch.where(invalid, torch.tensor(
                ident, dtype=work.dtype, device=work.device), work)
        if func == "cumsum":
            local = torch.cumsum(work, 0)
        elif func == "cumprod":
            local = torch.cumprod(work, 0)
        elif func == "cummin":
            local = torch.cummin(work, 0).values
        else:
            local = torch.cummax(work, 0).values
        if ctx.world > 1:
            total = local[-1].item() if n else ident
            totals = comm.allgather_obj(total)
            prefix = ident
            for t in totals[:ctx.rank]:
                if func == "cumsum":
                    prefix += t
                elif func == "cumprod":
                    prefix *= t
                elif func == "cummin":
                    prefix = min(prefix, t)
                else:
                    prefix = max(prefix, t)
            if prefix != ident:
                pt = torch.tensor(prefix, dtype=local.dtype,
                                  device=local.device)
                if func == "cumsum":
                    local = local + pt
                elif func == "cumprod":
                    local = local * pt
                elif func == "cummin":
                    local = torch.minimum(local, pt)
                else:
                    local = torch.maximum(local, pt)
        out_dtype = col.dtype
        if invalid is not None:
            local = torch.where(invalid, torch.tensor(
                float("nan"), dtype=local.dtype, device=local.device), local)
            out_dtype = bt.float64
        names.append(out_name)
        cols.append(Column(out_dtype, local))
    return Table(names, cols, n)


_ROLL_DEV_FUNCS = {"sum", "mean", "count", "min", "max"}


def _exec_rolling_device(node: pn.Rolling, ctx, child: Table) -> Optional[Table]:
    """Global rolling via prefix sums on device with tensor halo rows
    (reference role: window frame aggregation, _window_aggfuncs.cpp)."""
    w = int(node.window)
    minp = node.min_periods if node.min_periods is not None else w
    if any(f not in _ROLL_DEV_FUNCS for _, _, f in node.specs):
        return None
    if w > 512:
        return None
    in_cols = {inn: child.column(inn) for _, inn, _ in node.specs}
    if any(c.dtype.kind in (TypeKind.STRING, TypeKind.DICT)
           for c in in_cols.values()):
        return None
    n = len(child)
    dev = child.device
    names, cols = [], []
    halo_cache = {}
    for out_name, in_name, func in node.specs:
        c = in_cols[in_name]
        if in_name not in halo_cache:
            hd, hm = (c.data[:0], None)
            if ctx.world > 1 and w > 1:
                hd, hm = _gather_halo(c.data, c.mask, w - 1, ctx)
            halo_cache[in_name] = (hd, hm)
        hd, hm = halo_cache[in_name]
        h = int(hd.numel())
        x = torch.cat([hd, c.data]).double()
        valid = torch.ones(h + n, dtype=torch.bool, device=dev)
        if c.mask is not None:
            valid = torch.cat([hm if hm is not None else torch.ones(
                h, dtype=torch.bool, device=dev), c.mask])
        if c.dtype.is_float:
            valid = valid & ~torch.isnan(torch.cat([hd, c.data]))
        xf = torch.where(valid, x, torch.zeros((), dtype=x.dtype, device=dev))
        pos = torch.arange(h + n, device=dev)
        lo = (pos - (w - 1)).clamp(min=0)
        cs = torch.cumsum(xf, 0)
        excl = cs - xf
        cv = torch.cumsum(valid.double(), 0)
        excl_v = cv - valid.double()
        rcount = cv - excl_v[lo]
        if func == "count":
            res = rcount
        elif func in ("sum", "mean"):
            rsum = cs - excl[lo]
            res = rsum if func == "sum" else rsum / rcount
        else:
            sent = float("inf") if func == "min" else float("-inf")
            cur = torch.where(valid, x, torch.full(
                (), sent, dtype=x.dtype, device=dev))
            acc = cur.clone()
            for t in range(1, w):
                sh = torch.empty_like(cur)
                sh[t:] = cur[:h + n - t]
                sh[:t] = sent
                inwin = pos - t >= lo
                sh = torch.where(inwin, sh, torch.full(
                    (), sent, dtype=x.dtype, device=dev))
                acc = torch.minimum(acc, sh) if func == "min" else \
                    torch.maximum(acc, sh)
            res = acc
        # pandas min_periods: non-null observations for value aggs, but
        # window ROW completeness for count
        rows_in_win = pos - lo + 1
        if func == "count":
            bad = rows_in_win < minp
        else:
            bad = (rcount < minp) | torch.isinf(res)
        res = torch.where(bad, torch.full((), float("nan"), dtype=res.dtype,
                                          device=dev), res)
        names.append(out_name)
        cols.append(Column(bt.float64, res[h:].contiguous()))
    return Table(names, cols, n)


def _exec_rolling(node: pn.Rolling, ctx) -> Table:
    child = _exec(node.child, ctx)
    dev_out = None
    try:
        dev_out = _exec_rolling_device(node, ctx, child)
    except Exception:
        dev_out = None
    if dev_out is not None:
        return dev_out
    w = int(node.window)
    pdf = child.to_pandas()
    halo = 0
    if ctx.world > 1 and w > 1:
        # halo exchange: each rank publishes its last (w-1) rows; a rank
        # prepends rows from preceding ranks until it holds w-1 of them
        # (walking further back when a predecessor shard is shorter)
        tail = pdf.tail(w - 1)
        tails = comm.allgather_obj(tail)
        pieces, have, i = [], 0, ctx.rank - 1
        while have < w - 1 and i >= 0:
            t = tails[i]
            take = t.tail(w - 1 - have)
            pieces.insert(0, take)
            have += len(take)
            if len(t) >= w - 1:
                break
            i -= 1
        if pieces:
            halo = sum(len(p) for p in pieces)
            pdf = pd.concat(pieces + [pdf], ignore_index=True)
    out = {}
    for out_name, in_name, func in node.specs:
        r = pdf[in_name].rolling(w, min_periods=node.min_periods)
        out[out_name] = getattr(r, func)()
    res = pd.DataFrame(out).iloc[halo:].reset_index(drop=True)
    return Table.from_pandas(res, ctx.device)


def _exec_map_partitions(node: pn.MapPartitions, ctx) -> Table:
    child = _exec(node.child, ctx)
    pdf = child.to_pandas()
    res = node.func(pdf, *node.args)
    if isinstance(res, pd.Series):
        res = res.to_frame(name=res.name if res.name is not None else "0")
    return Table.from_pandas(res.reset_index(drop=True), ctx.device)


# ---------------------------------------------------------------- binary

BROADCAST_JOIN_THRESHOLD = 256 * 1024 * 1024  # bytes (reference: gpu_join.h:92)


def _exec_join(node: pn.Join, ctx) -> Table:
    left = _exec(node.left, ctx)
    right = _exec(node.right, ctx)
    from . import ooc

    if ctx.world == 1 or node.how == "cross":
        if ctx.world > 1 and node.how == "cross":
            right = comm.allgather_table(right)
        return ooc.join_local(left, right, node.left_on, node.right_on,
                              node.how, node.suffixes)
    # broadcast the smaller side when cheap and semantics allow
    lsize = sum(comm.allgather_obj(left.nbytes()))
    rsize = sum(comm.allgather_obj(right.nbytes()))
    if rsize <= BROADCAST_JOIN_THRESHOLD and node.how in ("inner", "left", "semi", "anti"):
        right_full = comm.allgather_table(right)
        return rel.join_local(left, right_full, node.left_on, node.right_on,
                              node.how, node.suffixes)
    if lsize <= BROADCAST_JOIN_THRESHOLD and node.how in ("inner", "right"):
        left_full = comm.allgather_table(left)
        out = rel.join_local(left_full, right, node.left_on, node.right_on,
                             node.how, node.suffixes)
        return out
    # hash-shuffle both sides by join keys; runtime filters prune the probe
    # side first so non-matching rows never enter the exchange (broadcast
    # joins above skip this: they have no shuffle to save)
    from . import join_filter

    left, right = join_filter.apply_runtime_filters(
        left, right, node.left_on, node.right_on, node.how, ctx)
    lh = ops.hash_columns([left.column(k) for k in node.left_on])
    rh = ops.hash_columns([right.column(k) for k in node.right_on])
    lp = torch.remainder(lh, ctx.world)
    lp = torch.where(lp < 0, lp + ctx.world, lp)
    rp = torch.remainder(rh, ctx.world)
    rp = torch.where(rp < 0, rp + ctx.world, rp)
    lshuf = comm.shuffle_table(left, lp)
    rshuf = comm.shuffle_table(right, rp)
    return ooc.join_local(lshuf, rshuf, node.left_on, node.right_on,
                          node.how, node.suffixes)


def _exec_union(node: pn.Union, ctx) -> Table:
    parts = [_exec(c, ctx) for c in node.inputs]
    # align columns by name of the first input
    names = parts[0].names
    aligned = [p.select([n for n in names]) if p.names != names else p for p in parts]
    out = ops.concat_tables(aligned)
    if node.distinct:
        out = rel.distinct_local(out)
        if ctx.world > 1:
            out = _shuffle_by_keys(out, list(out.names), ctx)
            out = rel.distinct_local(out)
    return out


# ---------------------------------------------------------------- sinks

def _exec_parquet_write(node: pn.ParquetWrite, ctx) -> Table:
    child = _exec(node.child, ctx)
    from ..io import parquet as pio

    pio.write_shard(child, node.path, node.compression, ctx,
                    getattr(node, "partition_cols", ()))
    return Table([], [], 0)


def _exec_iceberg_write(node: pn.IcebergWrite, ctx) -> Table:
    child = _exec(node.child, ctx)
    from ..io import iceberg

    iceberg.write_iceberg(child, node.path, node.mode, ctx)
    return Table([], [], 0)


def _exec_reduce(node: pn.Reduce, ctx) -> Table:
    from .streaming import exec_streaming, want_streaming

    if want_streaming(node, ctx):
        return exec_streaming(node, ctx)
    child = _exec(node.child, ctx)
    partials = {}
    for out, in_name, func in node.aggs:
        if not in_name:  # COUNT(*) / size over the whole table
            partials[out] = {"count": len(child)}
        else:
            partials[out] = ops.reduce_column(child.column(in_name), func)
    return _finish_reduce(node, ctx, partials)


def _finish_reduce(node: pn.Reduce, ctx, partials: dict) -> Table:
    out_names, out_cols = [], []
    for out_name, in_name, func in node.aggs:
        parts = comm.allgather_obj(partials[out_name])
        val = _combine_reduce(parts, func)
        out_names.append(out_name)
        ser = pd.Series([val])
        if ser.dtype == object:  # strings / None: go through arrow
            import pyarrow as pa

            arr = pa.Array.from_pandas(ser)
            if pa.types.is_string(arr.type):
                arr = arr.cast(pa.large_string())
            elif pa.types.is_null(arr.type):
                arr = arr.cast(pa.float64())
            out_cols.append(Column.from_arrow(arr, ctx.device))
        else:
            out_cols.append(Column.from_numpy(ser.to_numpy(), ctx.device))
    return Table(out_names, out_cols, 1)


def _combine_reduce(partials: List[dict], func: str):
    if callable(func):
        ser = pd.Series([v for p in partials for v in p["vals_list"]])
        return func(ser)
    if func in ("sum",):
        return sum(p["sum"] for p in partials)
    if func in ("count", "size"):
        return sum(p["count"] for p in partials)
    if func == "mean":
        c = sum(p["count"] for p in partials)
        return (sum(p["sum"] for p in partials) / c) if c else float("nan")
    if func == "min":
        vals = [p["min"] for p in partials if p["min"] is not None]
        return min(vals) if vals else None
    if func == "max":
        vals = [p["max"] for p in partials if p["max"] is not None]
        return max(vals) if vals else None
    if func in ("var", "std", "sem"):
        c = sum(p["count"] for p in partials)
        s = sum(p["sum"] for p in partials)
        ss = sum(p["sumsq"] for p in partials)
        if c < 2:
            return float("nan")
        var = (ss - s * s / c) / (c - 1)
        if func == "var":
            return var
        return var ** 0.5 if func == "std" else (var / c) ** 0.5
    if func in ("kurt", "skew"):
        # pandas-compatible bias-adjusted central moments from raw sums
        c = sum(p["count"] for p in partials)
        s1 = sum(p["sum"] for p in partials)
        s2 = sum(p["sumsq"] for p in partials)
        s3 = sum(p["sum3"] for p in partials)
        s4 = sum(p["sum4"] for p in partials)
        if (func == "skew" and c < 3) or (func == "kurt" and c < 4):
            return float("nan")
        mu = s1 / c
        d2 = s2 - c * mu * mu
        if d2 <= 0:
            return float("nan")
        m2 = d2 / c
        if func == "skew":
            d3 = s3 - 3 * mu * s2 + 2 * c * mu ** 3
            g1 = (d3 / c) / m2 ** 1.5
            return (c * (c - 1)) ** 0.5 / (c - 2) * g1
        d4 = s4 - 4 * mu * s3 + 6 * mu * mu * s2 - 3 * c * mu ** 4
        g2 = (d4 / c) / (m2 * m2) - 3.0
        return ((c + 1) * g2 + 6) * (c - 1) / ((c - 2) * (c - 3))
    if func in ("first", "last"):
        seq = partials if func == "first" else list(reversed(partials))
        for p in seq:
            if p["has"]:
                return p["val"]
        return None
    if func == "median":
        import numpy as np

        vs = [p["vals"] for p in partials if p["vals"] is not None]
        if not vs:
            return float("nan")
        return float(np.median(np.concatenate(vs)))
    if func == "array_agg":
        out = []
        for p_ in partials:
            out.extend(p_["vals_list"])
        return out
    if func == "mode":
        counts: dict = {}
        for p in partials:
            for k, v in p["counts"].items():
                counts[k] = counts.get(k, 0) + v
        if not counts:
            return None
        best = max(counts.values())
        return min(k for k, v in counts.items() if v == best)
    if func == "any":
        return any(p["any"] for p in partials)
    if func == "all":
        return all(p["all"] for p in partials)
    if func == "prod":
        out = 1.0
        for p in partials:
            out *= p["prod"]
        return out
    if func == "nunique":
        u = set()
        for p in partials:
            u |= p["uniq"]
        return len(u)
    if func == "approx_nunique":
        from ..utils import sketches

        regs = partials[0]["hll"]
        for p in partials[1:]:
            regs = torch.maximum(regs, p["hll"])
        return int(round(sketches.hll_estimate(regs)))
    raise NotImplementedError(func)


_HANDLERS = {
    pn.ParquetScan: _exec_parquet,
    pn.CsvScan: _exec_csv,
    pn.PandasScan: _exec_pandas_scan,
    pn.Projection: _exec_projection,
    pn.Filter: _exec_filter,
    pn.Aggregate: _exec_aggregate,
    pn.Sort: _exec_sort,
    pn.Limit: _exec_limit,
    pn.Distinct: _exec_distinct,
    pn.Sample: _exec_sample,
    pn.Explode: _exec_explode,
    pn.MapPartitions: _exec_map_partitions,
    pn.ShuffleByKey: _exec_shuffle_by_key,
    pn.RowId: _exec_rowid,
    pn.Rolling: _exec_rolling,
    pn.Cumulative: _exec_cumulative,
    pn.Shift: _exec_shift,
    pn.Fill: _exec_fill,
    pn.Window: _exec_window,
    pn.Join: _exec_join,
    pn.Union: _exec_union,
    pn.ParquetWrite: _exec_parquet_write,
    pn.IcebergWrite: _exec_iceberg_write,
    pn.Reduce: _exec_reduce,
}
