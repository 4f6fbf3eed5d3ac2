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

from typing import List, Tuple

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
                  "transform_count": "count", "transform_size": "size"}

ORDERED_FUNCS = {"row_number", "rank", "dense_rank", "shift", "cumsum",
                 "cumcount", "first_value", "last_value", "ntile",
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


def _ordered_local(tbl: Table, keys, order_by, ascending, specs) -> Table:
    """Compute ordered window funcs on co-located partitions; preserves the
    input row order of `tbl`."""
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
            res = gbs[in_name].shift(arg if arg is not None else 1)
            res = res.reindex(pdf.index)
        elif func == "cumsum":
            res = gbs[in_name].cumsum().reindex(pdf.index)
        elif func in ("first_value", "last_value"):
            which = "first" if func == "first_value" else "last"
            res = gbs[in_name].transform(which).reindex(pdf.index)
        elif func.startswith("rolling_"):
            base = func[len("rolling_"):]
            r = gbs[in_name].rolling(int(arg), min_periods=1)
            res = getattr(r, base)()
            res = res.droplevel(list(range(len(keys))))
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
