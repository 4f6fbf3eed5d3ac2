"""Window / transform execution (reference: bodo/libs/window/ +
bodo/libs/streaming/_window.cpp).

Two strategies:
* transform-aggs (groupby.transform sum/mean/min/max/count/size): compute the
  (small) aggregate table via the regular two-phase machinery, replicate it,
  and map values back with an order-preserving left probe join — no shuffle
  of the big table, no reordering.
* ordered funcs (rank/row_number/shift/cumsum/cumcount): tag rows with their
  global row-id + home rank, hash-shuffle by partition keys, compute locally
  (sorted by partition then order keys), shuffle back and restore order.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import numpy as np
import pandas as pd
import torch

from .. import ops
from ..core import types as bt
from ..core.column import Column
from ..core.table import Table
from ..ops import relational as rel
from ..parallel import comm
from ..plan import nodes as pn

TRANSFORM_AGGS = {"transform_sum": "sum", "transform_mean": "mean",
                  "transform_min": "min", "transform_max": "max",
                  "transform_count": "count", "transform_size": "size",
                  "transform_nunique": "nunique"}

ORDERED_FUNCS = {"row_number", "rank", "dense_rank", "percent_rank",
                 "cume_dist", "nth_value", "shift", "cumsum",
                 "cumcount", "cummin", "cummax", "cummean", "cumcount_v",
                 "first_value", "last_value", "first_value_ig", "last_value_ig",
                 "ntile",
                 "rolling_sum", "rolling_mean", "rolling_min", "rolling_max",
                 "rolling_count"}


def exec_window(node: pn.Window, ctx, child: Table) -> Table:
    keys = list(node.keys)
    agg_specs = [s for s in node.specs if s[2] in TRANSFORM_AGGS]
    ord_specs = [s for s in node.specs if s[2] in ORDERED_FUNCS]
    out = child
    if agg_specs:
        out = _exec_transform_aggs(out, keys, agg_specs, ctx)
    if ord_specs:
        out = _exec_ordered(out, keys, list(node.order_by),
                            list(node.ascending), ord_specs, ctx)
    return out


def _exec_transform_aggs(child: Table, keys, specs, ctx) -> Table:
    from .executor import _exec_aggregate

    tmp_names = [f"__w{i}" for i in range(len(specs))]
    aggs = tuple((tmp, s[1], TRANSFORM_AGGS[s[2]])
                 for tmp, s in zip(tmp_names, specs))
    # aggregate the child table (already materialized) via a synthetic plan
    from . import executor as ex

    key_id = ex.register_object(child)
    agg_plan = pn.Aggregate(
        pn.PandasScan(key_id, tuple(child.names), distributed=True),
        tuple(keys), aggs)
    agg_shard = _exec_aggregate(agg_plan, ctx)
    ex.delete_object(key_id)
    agg_full = comm.allgather_table(agg_shard) if ctx.world > 1 else agg_shard
    # order-preserving left probe join against the (unique-key) agg table
    joined = rel.join_local(child, agg_full, keys, keys, "left",
                            suffixes=("", "__aggdup"))
    names, cols = list(child.names), list(child.columns)
    for tmp, s in zip(tmp_names, specs):
        names.append(s[0])
        cols.append(joined.column(tmp))
    return Table(names, cols, len(child))


def _exec_ordered(child: Table, keys, order_by, ascending, specs, ctx) -> Table:
    n = len(child)
    dev = child.device
    work = child
    if ctx.world > 1:
        lengths = comm.allgather_obj(n)
        base = sum(lengths[:ctx.rank])
        rid = torch.arange(base, base + n, dtype=torch.int64, device=dev)
        src = torch.full((n,), ctx.rank, dtype=torch.int64, device=dev)
        work = work.with_column("__rid", Column(bt.int64, rid))
        work = work.with_column("__src", Column(bt.int64, src))
        h = ops.hash_columns([work.column(k) for k in keys])
        part = torch.remainder(h, ctx.world)
        part = torch.where(part < 0, part + ctx.world, part)
        work = comm.shuffle_table(work, part)
    out = _ordered_local(work, keys, order_by, ascending, specs)
    if ctx.world > 1:
        out = comm.shuffle_table(out, out.column("__src").data)
        order = torch.argsort(out.column("__rid").data)
        out = ops.take_table(out, order)
        keep = [c for c in out.names if c not in ("__rid", "__src")]
        out = out.select(keep)
    return out


_ROLLING_MINMAX_CAP = 128


def _adj_eq(c: Column, idx: torch.Tensor) -> Optional[torch.Tensor]:
    """Bool tensor of length n-1: sorted-adjacent rows equal under pandas
    groupby(dropna=False) semantics (null == null, NaN == NaN).  None when
    the column kind has no device comparison (plain strings)."""
    from ..core.types import TypeKind as TK

    k = c.dtype.kind
    if k == TK.STRING:
        return None
    s = c.data[idx]
    if c.dtype.is_float:
        eq = (s[1:] == s[:-1]) | (torch.isnan(s[1:]) & torch.isnan(s[:-1]))
    else:
        eq = s[1:] == s[:-1]
    if c.mask is not None:
        m = c.mask[idx]
        eq = (eq & m[1:] & m[:-1]) | (~m[1:] & ~m[:-1])
    return eq


def _ordered_local_device(tbl: Table, keys, order_by, ascending,
                          specs) -> Optional[Table]:
    """Segmented device window calculator: one stable sort by
    (partition keys, order keys), segment ids from adjacent key equality,
    every function as segmented prefix ops — no host pandas (reference role:
    bodo/libs/window/_window_calculator.cpp:2106, redesigned as whole-shard
    tensor passes instead of chunked streaming)."""
    from ..core.types import TypeKind as TK

    n = len(tbl)
    dev = tbl.device
    if n == 0:
        return None
    key_cols = [tbl.column(k) for k in keys]
    order_cols = [tbl.column(c) for c in order_by]
    asc = list(ascending) or [True] * len(order_by)
    if any(c.dtype.kind == TK.STRING for c in key_cols + order_cols):
        return None
    for _, in_name, func, arg in specs:
        if func in ("first_value_ig", "last_value_ig"):
            return None  # ignore-nulls variants: host path
        if func.startswith("rolling_") and func[8:] in ("min", "max") \
                and int(arg or 1) > _ROLLING_MINMAX_CAP:
            return None
        if in_name and tbl.has_column(in_name):
            ic = tbl.column(in_name)
            if ic.dtype.kind in (TK.STRING, TK.DICT) and func in (
                    "shift", "first_value", "last_value"):
                return None  # string payloads: host path
    idx = ops.sort_indices(key_cols + order_cols,
                           [True] * len(keys) + asc)
    seg_new = torch.zeros(n, dtype=torch.bool, device=dev)
    seg_new[0] = True
    for c in key_cols:
        eq = _adj_eq(c, idx)
        if eq is None:
            return None
        seg_new[1:] |= ~eq
    segid = torch.cumsum(seg_new.long(), 0) - 1
    start_pos = torch.nonzero(seg_new).squeeze(1)           # [n_segs]
    n_segs = int(start_pos.numel())
    pos = torch.arange(n, dtype=torch.int64, device=dev)
    seg_start = start_pos[segid]                            # [n] start of my seg
    row_in_seg = pos - seg_start
    end_pos = torch.cat([start_pos[1:] - 1,
                         torch.tensor([n - 1], device=dev)])
    seg_size = end_pos[segid] - seg_start + 1

    def sorted_vals(name, fill0=False):
        c = tbl.column(name)
        v = c.data[idx]
        invalid = None
        if c.mask is not None:
            invalid = ~c.mask[idx]
        if c.dtype.is_float:
            nanv = torch.isnan(v)
            invalid = nanv if invalid is None else (invalid | nanv)
        if fill0 and invalid is not None:
            v = torch.where(invalid, torch.zeros((), dtype=v.dtype,
                                                 device=dev), v)
        return v, invalid

    def seg_cumsum(vf):
        cs = torch.cumsum(vf.double() if vf.dtype != torch.float64 else vf, 0)
        excl = cs - (vf.double() if vf.dtype != torch.float64 else vf)
        return cs - excl[seg_start]

    out_sorted = {}
    for out_name, in_name, func, arg in specs:
        if func == "row_number":
            res, res_inv = row_in_seg + 1, None
        elif func == "cumcount":
            res, res_inv = row_in_seg, None
        elif func in ("rank", "dense_rank"):
            method = "dense" if func == "dense_rank" else (arg or "min")
            if method not in ("dense", "min"):
                return None
            if order_cols:
                r_idx, r_seg_new, r_seg_start = idx, seg_new, seg_start
                ocol = order_cols[0]
            else:
                # value rank (groupby().rank()): needs its own sort by the
                # ranked column within each partition
                ocol = tbl.column(in_name)
                if ocol.dtype.kind == TK.STRING:
                    return None
                rasc = ascending[0] if ascending else True
                r_idx = ops.sort_indices(key_cols + [ocol],
                                         [True] * len(keys) + [rasc])
                r_seg_new = torch.zeros(n, dtype=torch.bool, device=dev)
                r_seg_new[0] = True
                for c in key_cols:
                    eq = _adj_eq(c, r_idx)
                    if eq is None:
                        return None
                    r_seg_new[1:] |= ~eq
                r_segid = torch.cumsum(r_seg_new.long(), 0) - 1
                r_seg_start = torch.nonzero(r_seg_new).squeeze(1)[r_segid]
            eq = _adj_eq(ocol, r_idx)
            if eq is None:
                return None
            run_new = r_seg_new.clone()
            run_new[1:] |= ~eq
            if func == "dense_rank":
                rg = torch.cumsum(run_new.long(), 0)
                res = rg - rg[r_seg_start] + 1
            else:  # rank(min): row_number of the run start
                run_start = torch.nonzero(run_new).squeeze(1)
                runid = torch.cumsum(run_new.long(), 0) - 1
                res = run_start[runid] - r_seg_start + 1
            # pandas rank: NaN order value -> NaN rank
            ov = ocol.data[r_idx]
            oinv = None
            if ocol.dtype.is_float:
                oinv = torch.isnan(ov)
            if ocol.mask is not None:
                om = ~ocol.mask[r_idx]
                oinv = om if oinv is None else (oinv | om)
            res_inv = oinv
            if r_idx is not idx:
                # map through this spec's own sort back to original order
                r_inv = torch.empty_like(r_idx)
                r_inv[r_idx] = pos
                res = res[r_inv][idx]
                if res_inv is not None:
                    res_inv = res_inv[r_inv][idx]
        elif func == "cumsum":
            v, inv = sorted_vals(in_name, fill0=True)
            res = seg_cumsum(v)
            if tbl.column(in_name).dtype.is_integer and inv is None:
                res = res.long()
            res_inv = inv
        elif func in ("cummin", "cummax", "cummean", "cumcount_v"):
            # running frame: ROWS UNBOUNDED PRECEDING .. CURRENT ROW
            v, inv = sorted_vals(in_name, fill0=True)
            valid = (torch.ones(n, dtype=torch.float64, device=dev)
                     if inv is None else (~inv).double())
            if func == "cummean":
                cnt = seg_cumsum(valid)
                res = seg_cumsum(v.double()) / cnt
                res_inv = cnt == 0
            elif func == "cumcount_v":
                res = seg_cumsum(valid).long()
                res_inv = None
            else:
                # segmented Hillis–Steele scan: log2(n) whole-tensor passes;
                # the segid guard at distance `off` is exact for contiguous
                # segments and an idempotent combine
                pad = float("inf") if func == "cummin" else float("-inf")
                cur = torch.where(valid.bool(), v.double(),
                                  torch.full((), pad, device=dev))
                off = 1
                while off < n:
                    prev = cur
                    cur = prev.clone()
                    samseg = segid[off:] == segid[:n - off]
                    comb = (torch.minimum if func == "cummin"
                            else torch.maximum)(prev[off:], prev[:n - off])
                    cur[off:] = torch.where(samseg, comb, prev[off:])
                    off *= 2
                res = cur
                res_inv = torch.isinf(cur)
        elif func == "shift":
            k, dflt = (arg if isinstance(arg, tuple)
                       else (int(arg if arg is not None else 1), None))
            if dflt is not None and isinstance(dflt, str):
                return None  # string default: host path
            k = int(k)
            v, inv = sorted_vals(in_name)
            res = torch.full_like(v, 0)
            res_inv = torch.ones(n, dtype=torch.bool, device=dev)
            # outside = offset crosses the partition edge (LAG default
            # applies there; a NULL source value stays NULL)
            outside = torch.ones(n, dtype=torch.bool, device=dev)
            if k >= 0:
                if k < n:
                    res[k:] = v[:n - k]
                    inwin = segid[k:] == segid[:n - k]
                    outside[k:] = ~inwin
                    ok = inwin if inv is None else (inwin & ~inv[:n - k])
                    res_inv[k:] = ~ok
            else:
                kk = -k
                if kk < n:
                    res[:n - kk] = v[kk:]
                    inwin = segid[:n - kk] == segid[kk:]
                    outside[:n - kk] = ~inwin
                    ok = inwin if inv is None else (inwin & ~inv[kk:])
                    res_inv[:n - kk] = ~ok
            if dflt is not None:
                fill = torch.full((), dflt, dtype=(
                    v.dtype if v.dtype.is_floating_point
                    else torch.float64), device=dev)
                if not v.dtype.is_floating_point:
                    res = res.to(torch.float64)
                res = torch.where(outside, fill, res)
                res_inv = res_inv & ~outside
        elif func in ("first_value", "last_value"):
            v, inv = sorted_vals(in_name)
            at = seg_start if func == "first_value" else end_pos[segid]
            res = v[at]
            res_inv = inv[at] if inv is not None else None
        elif func == "percent_rank":
            # (rank-1)/(size-1); single-row partitions -> 0
            ocol = order_cols[0] if order_cols else None
            if ocol is None:
                return None
            eq = _adj_eq(ocol, idx)
            if eq is None:
                return None
            run_new = seg_new.clone()
            run_new[1:] |= ~eq
            run_start = torch.nonzero(run_new).squeeze(1)
            runid = torch.cumsum(run_new.long(), 0) - 1
            rk = (run_start[runid] - seg_start + 1).double()
            denom = (seg_size - 1).double().clamp(min=1)
            res = (rk - 1) / denom
            res_inv = None
        elif func == "cume_dist":
            # rows with order value <= current / partition size: position
            # of this value-run's END + 1, over seg_size
            ocol = order_cols[0] if order_cols else None
            if ocol is None:
                return None
            eq = _adj_eq(ocol, idx)
            if eq is None:
                return None
            run_new = seg_new.clone()
            run_new[1:] |= ~eq
            run_start = torch.nonzero(run_new).squeeze(1)
            runid = torch.cumsum(run_new.long(), 0) - 1
            n_runs = int(run_start.numel())
            run_end = torch.cat([run_start[1:] - 1,
                                 torch.tensor([n - 1], device=dev)])
            res = (run_end[runid] - seg_start + 1).double() / \
                seg_size.double()
            res_inv = None
        elif func == "nth_value":
            k = int(arg or 1)
            v, inv = sorted_vals(in_name)
            at = (seg_start + (k - 1)).clamp(max=n - 1)
            ok = (row_in_seg >= k - 1) & (seg_start + (k - 1) <= end_pos[segid])
            res = v[at]
            res_inv = ~ok if inv is None else (~ok | inv[at])
        elif func == "ntile":
            k = int(arg or 1)
            res = (row_in_seg * k) // seg_size + 1
            res_inv = None
        elif func.startswith("rolling_"):
            base = func[len("rolling_"):]
            w = int(arg)
            v, inv = sorted_vals(in_name, fill0=True)
            valid = torch.ones(n, dtype=torch.float64, device=dev)
            if inv is not None:
                valid = (~inv).double()
            lo = torch.maximum(pos - (w - 1), seg_start)
            if base in ("sum", "mean", "count"):
                vf = v.double()
                cs = torch.cumsum(vf, 0)
                excl = cs - vf
                rsum = cs - excl[lo]
                cv = torch.cumsum(valid, 0)
                excl_v = cv - valid
                rcount = cv - excl_v[lo]
                if base == "sum":
                    res = rsum
                elif base == "count":
                    res = rcount
                else:
                    res = rsum / rcount
                res_inv = rcount == 0 if base != "count" else None
            else:  # min / max over small windows: log-free shift loop
                cur = torch.where(valid.bool(), v.double(),
                                  torch.full((), float("inf") if base == "min"
                                             else float("-inf"), device=dev))
                acc = cur.clone()
                for t in range(1, w):
                    sh = torch.empty_like(cur)
                    sh[t:] = cur[:n - t]
                    sh[:t] = float("inf") if base == "min" else float("-inf")
                    inwin = pos - t >= lo
                    sh = torch.where(inwin, sh,
                                     torch.full((), float("inf")
                                                if base == "min"
                                                else float("-inf"),
                                                device=dev))
                    acc = torch.minimum(acc, sh) if base == "min" \
                        else torch.maximum(acc, sh)
                res = acc
                res_inv = torch.isinf(res)
        else:
            return None
        out_sorted[out_name] = (res, res_inv)

    out = tbl
    inv_perm = torch.empty_like(idx)
    inv_perm[idx] = pos
    for name, (res, res_inv) in out_sorted.items():
        orig = res[inv_perm]
        mask = None
        if res_inv is not None:
            mask = ~res_inv[inv_perm]
            if orig.dtype.is_floating_point:
                orig = torch.where(mask, orig,
                                   torch.full((), float("nan"), device=dev,
                                              dtype=orig.dtype))
                mask = None
        kind = bt.from_numpy_dtype(np.dtype(str(orig.dtype).replace(
            "torch.", "")))
        col = Column(kind, orig, mask)
        out = out.with_column(name, col)
    return out


def _ordered_local(tbl: Table, keys, order_by, ascending, specs) -> Table:
    """Compute ordered window funcs on co-located partitions; preserves the
    input row order of `tbl`.  Device path first; host pandas only for
    layouts the tensor calculator does not cover."""
    try:
        dev_out = _ordered_local_device(tbl, keys, order_by, ascending, specs)
    except Exception:
        dev_out = None
    if dev_out is not None:
        return dev_out
    need = list(dict.fromkeys(
        keys + order_by + [s[1] for s in specs if s[1] and tbl.has_column(s[1])]))
    pdf = tbl.select(need).to_pandas()
    if order_by:
        asc = list(ascending) or [True] * len(order_by)
        spdf = pdf.sort_values(order_by, ascending=asc, kind="stable")
    else:
        spdf = pdf
    gb = pdf.groupby(keys, dropna=False, sort=False, observed=True)
    gbs = spdf.groupby(keys, dropna=False, sort=False, observed=True)
    out_cols = {}
    for out_name, in_name, func, arg in specs:
        if func == "row_number":
            res = (gbs.cumcount() + 1).reindex(pdf.index)
        elif func == "cumcount":
            res = gbs.cumcount().reindex(pdf.index)
        elif func in ("rank", "dense_rank"):
            method = "dense" if func == "dense_rank" else (arg or "min")
            col = order_by[0] if order_by else in_name
            res = gb[col].rank(method=method,
                               ascending=ascending[0] if ascending else True)
        elif func == "shift":
            if isinstance(arg, tuple):
                k, dflt = arg
                res = gbs[in_name].shift(int(k), fill_value=dflt)
            else:
                res = gbs[in_name].shift(arg if arg is not None else 1)
            res = res.reindex(pdf.index)
        elif func == "cumsum":
            res = gbs[in_name].cumsum().reindex(pdf.index)
        elif func == "cummin":
            # expanding (not pandas cummin): SQL MIN OVER reports the frame
            # min even at rows whose own value is NULL
            res = gbs[in_name].transform(
                lambda s: s.expanding(min_periods=1).min())
            res = res.reindex(pdf.index)
        elif func == "cummax":
            res = gbs[in_name].transform(
                lambda s: s.expanding(min_periods=1).max())
            res = res.reindex(pdf.index)
        elif func == "cummean":
            res = gbs[in_name].transform(
                lambda s: s.expanding(min_periods=1).mean())
            res = res.reindex(pdf.index)
        elif func == "cumcount_v":
            res = gbs[in_name].transform(
                lambda s: s.notna().cumsum()).reindex(pdf.index)
        elif func in ("first_value", "last_value"):
            which = "first" if func == "first_value" else "last"
            res = gbs[in_name].transform(which).reindex(pdf.index)
        elif func in ("first_value_ig", "last_value_ig"):
            # IGNORE NULLS: partition first/last non-null value
            take_first = func.startswith("first")
            res = gbs[in_name].transform(
                lambda s, _f=take_first: (
                    (s.dropna().iloc[0] if _f else s.dropna().iloc[-1])
                    if s.notna().any() else np.nan))
            res = res.reindex(pdf.index)
        elif func.startswith("rolling_"):
            base = func[len("rolling_"):]
            r = gbs[in_name].rolling(int(arg), min_periods=1)
            res = getattr(r, base)()
            res = res.droplevel(list(range(len(keys))))
            res = res.reindex(pdf.index)
        elif func == "percent_rank":
            col = order_by[0] if order_by else in_name
            r = gb[col].rank(method="min")
            size = gb[keys[0]].transform("size")
            res = (r - 1) / (size - 1).clip(lower=1)
        elif func == "cume_dist":
            col = order_by[0] if order_by else in_name
            r = gb[col].rank(method="max")
            size = gb[keys[0]].transform("size")
            res = r / size
        elif func == "nth_value":
            k = int(arg or 1)
            res = gbs[in_name].transform(
                lambda s, _k=k: s.iloc[_k - 1] if len(s) >= _k else np.nan)
            res = res.reindex(pdf.index)
        elif func == "ntile":
            k = int(arg or 1)
            rn = gbs.cumcount().reindex(pdf.index)
            size = gb[keys[0]].transform("size")
            res = (rn * k) // size + 1
        else:
            raise NotImplementedError(func)
        out_cols[out_name] = res
    out = tbl
    for name, ser in out_cols.items():
        col = Column.from_numpy(ser.to_numpy(), tbl.device)
        out = out.with_column(name, col)
    return out
