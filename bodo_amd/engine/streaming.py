"""Streaming (morsel-wise) execution for larger-than-HBM inputs.

Reference: the push-based batch pipelines of bodo/pandas/_pipeline.h and the
incremental operator states in bodo/libs/streaming/.  MI355X redesign: with
288 GB HBM3E per GPU the morsel is sized in the hundreds of MB–GB range
(config.STREAM_BATCH_SIZE rows), and only the FIRST pipeline (scan side) is
streamed — the blocking operator keeps a bounded device-resident state:

* Aggregate: per-morsel local hash groupby, partial states concatenated and
  re-combined whenever they exceed a bound (monotone shrink per merge).
* Reduce: scalar partials folded per morsel.
* Join probe: the build side is materialized (small side), each probe morsel
  joins and feeds the downstream incremental consumer.

Activated when a plan bottoms out in a ParquetScan whose estimated bytes
exceed config.STREAM_THRESHOLD_BYTES, or explicitly via
BODO_AMD_STREAMING=1.
"""

from __future__ import annotations

from typing import Iterator, List, Optional

import torch

from .. import ops
from ..core.table import Table
from ..ops import relational as rel
from ..plan import nodes as pn
from ..plan.expr import ColRef


def _host_scan_batches(node: pn.ParquetScan, ctx,
                       batch_rows: int) -> Iterator["object"]:
    """Yield this rank's shard of a parquet scan as HOST arrow morsels
    (row-group granularity grouped into ~batch_rows chunks); runs on the
    prefetch thread so file IO + arrow decode overlap device compute."""
    import pyarrow as pa

    from ..core.table import dict_encode_strings
    from ..io.parquet import _dataset, _expr_to_arrow

    d = _dataset(node.path)
    pieces = []
    for f in d.get_fragments():
        try:
            pieces.extend(f.split_by_row_group())
        except Exception:
            pieces.append(f)
    w, r = ctx.world, ctx.rank
    base, rem = divmod(len(pieces), w)
    start = r * base + min(r, rem)
    my = pieces[start:start + base + (1 if r < rem else 0)]
    filt = None
    for e in node.filters:
        ae = _expr_to_arrow(e)
        filt = ae if filt is None else (filt & ae)
    cols = list(node.columns) if node.columns else None
    buf: List[pa.Table] = []
    rows = 0
    got_any = False
    for piece in my:
        t = piece.to_table(columns=cols, filter=filt)
        buf.append(t)
        rows += t.num_rows
        if rows >= batch_rows:
            merged = pa.concat_tables(buf).combine_chunks() \
                if len(buf) > 1 else buf[0]
            yield dict_encode_strings(merged)
            got_any = True
            buf, rows = [], 0
    if buf:
        merged = pa.concat_tables(buf).combine_chunks() \
            if len(buf) > 1 else buf[0]
        yield dict_encode_strings(merged)
    elif not got_any:
        schema = d.schema
        if cols:
            schema = pa.schema([schema.field(c) for c in cols])
        yield schema.empty_table()


def scan_batches(node: pn.ParquetScan, ctx, batch_rows: int) -> Iterator[Table]:
    """Device morsels with one-batch read-ahead: a prefetch thread does the
    file IO + arrow decode for batch N+1 while the engine computes on batch
    N (reference analog: the dedicated IO pool in
    bodo/io/_io_cpu_thread_pool.cpp overlapping reads with compute)."""
    import queue
    import threading

    q: "queue.Queue" = queue.Queue(maxsize=2)

    def producer():
        try:
            for t in _host_scan_batches(node, ctx, batch_rows):
                q.put(("ok", t))
            q.put(("done", None))
        except BaseException as e:  # surfaced on the consumer thread
            q.put(("err", e))

    th = threading.Thread(target=producer, daemon=True)
    th.start()
    while True:
        kind, t = q.get()
        if kind == "err":
            raise t
        if kind == "done":
            break
        yield Table.from_arrow(t, ctx.device)
    th.join()


def _to_device_batch(bufs, ctx) -> Table:
    import pyarrow as pa

    from ..core.table import dict_encode_strings

    t = pa.concat_tables(bufs).combine_chunks() if len(bufs) > 1 else bufs[0]
    t = dict_encode_strings(t)
    return Table.from_arrow(t, ctx.device)


def want_streaming(node, ctx) -> bool:
    """Stream when forced (BODO_AMD_STREAMING=1) or when the scan's file
    bytes exceed the threshold (auto mode)."""
    from .. import config

    mode = config.STREAMING
    if mode == "0" or not streamable(node):
        return False
    if mode == "1":
        return True
    # auto: compare file sizes against threshold
    import glob
    import os

    walked = _walk_stream_chain(node)
    if walked is None:
        return False
    path = walked[1].path
    try:
        if os.path.isdir(path):
            total = sum(os.path.getsize(p)
                        for p in glob.glob(os.path.join(path, "*")))
        else:
            total = os.path.getsize(path)
    except OSError:
        return False
    return total > config.STREAM_THRESHOLD_BYTES


_STREAM_JOIN_HOWS = {"inner", "left", "semi", "anti"}


def _walk_stream_chain(node: pn.PlanNode):
    """Follow the streamed (probe) spine below an Aggregate/Reduce:
    Filter/Projection pass through; a Join continues down the child whose
    subtree holds the ParquetScan while the other side becomes a resident
    build table (reference: push pipelines with resident build sides,
    bodo/pandas/_pipeline.h:108 + streaming/_join.h probe loop).
    Returns (chain ops root-to-scan order reversed later, scan) or None."""

    def has_scan(nd) -> bool:
        if isinstance(nd, pn.ParquetScan):
            return True
        return any(has_scan(c) for c in nd.children())

    chain = []
    cur = node.children()[0]
    depth = 0
    while depth < 64:
        depth += 1
        if isinstance(cur, (pn.Filter, pn.Projection)):
            chain.append(("op", cur))
            cur = cur.children()[0]
            continue
        if isinstance(cur, pn.Join) and cur.how in _STREAM_JOIN_HOWS:
            left, right = cur.children()
            ls, rs = has_scan(left), has_scan(right)
            if ls and not rs:
                chain.append(("join_left", cur))
                cur = left
                continue
            # streaming the right side only safe for inner joins
            if rs and not ls and cur.how == "inner":
                chain.append(("join_right", cur))
                cur = right
                continue
            return None
        break
    if not isinstance(cur, pn.ParquetScan):
        return None
    return chain, cur


def streamable(node: pn.PlanNode) -> bool:
    """True when `node` is Aggregate/Reduce over a chain of
    Filter/Projection/Join-with-resident-build over a ParquetScan."""
    if not isinstance(node, (pn.Aggregate, pn.Reduce)):
        return False
    if isinstance(node, pn.Aggregate):
        ok = {"sum", "count", "size", "min", "max", "mean", "first", "last",
              "prod"}
        if any((a[2] not in ok) if not callable(a[2]) else True
               for a in node.aggs):
            return False
    return _walk_stream_chain(node) is not None


def exec_streaming(node, ctx):
    """Execute Aggregate/Reduce(…chain with resident-build joins…
    (ParquetScan)) morsel-wise: build sides materialize once (replicated
    across ranks for local probes), probe batches flow through
    filter/project/join without ever materializing the fact table
    (reference: streaming/_join.h probe loop + _pipeline.h push batches)."""
    from .. import config
    from ..ops import evaluate as ev
    from ..parallel import comm
    from . import executor as ex

    walked = _walk_stream_chain(node)
    assert walked is not None
    raw_chain, scan = walked
    # materialize every resident build side once, up front
    chain = []
    for kind, op in raw_chain:
        if kind == "op":
            chain.append((kind, op, None))
            continue
        build_node = op.right if kind == "join_left" else op.left
        build = ex._exec(build_node, ctx)
        if ctx.world > 1:
            build = comm.allgather_table(build)
        chain.append((kind, op, build))
    chain.reverse()  # scan-side first

    batch_rows = config.STREAM_BATCH_SIZE

    def apply_chain(batch: Table) -> Table:
        for kind, op, build in chain:
            if kind == "op":
                if isinstance(op, pn.Filter):
                    batch = ev.eval_filter(op.cond, batch)
                else:
                    batch = ev.project(batch, op.names, op.exprs)
            elif kind == "join_left":
                batch = rel.join_local(batch, build, op.left_on,
                                       op.right_on, op.how, op.suffixes)
            else:  # join_right: resident left, streamed right (inner)
                batch = rel.join_local(build, batch, op.left_on,
                                       op.right_on, op.how, op.suffixes)
        return batch

    if isinstance(node, pn.Reduce):
        partials = None
        for batch in scan_batches(scan, ctx, batch_rows):
            batch = apply_chain(batch)
            cur_p = {out: ops.reduce_column(batch.column(in_name), func)
                     for out, in_name, func in node.aggs}
            partials = cur_p if partials is None else _merge_reduce(
                partials, cur_p, node.aggs)
        # reuse the executor's distributed combine
        return ex._finish_reduce(node, ctx, partials)

    # Aggregate: incremental partial-groupby state with bounded re-combine
    assert isinstance(node, pn.Aggregate)
    keys = list(node.keys)
    partial_aggs, final_map = ex._decompose_aggs(list(node.aggs))
    state: Optional[Table] = None
    combine_aggs = [(name, name, rel.COMBINE_FUNC[f])
                    for name, _, f in partial_aggs]
    for batch in scan_batches(scan, ctx, batch_rows):
        batch = apply_chain(batch)
        part = rel.groupby_local(batch, keys, partial_aggs, node.dropna)
        if state is None:
            state = part
        else:
            both = ops.concat_tables([state, part])
            state = rel.groupby_local(both, keys, combine_aggs, node.dropna)
    if state is None:
        return ex._exec_aggregate(node, ctx)  # no batches: fall back
    # distributed final combine (same as two-phase tail)
    if ctx.world > 1:
        state = ex._shuffle_by_keys(state, keys, ctx)
        state = rel.groupby_local(state, keys, combine_aggs, node.dropna)
    from ..ops import evaluate as ev2

    names, exprs = list(keys), [ColRef(k) for k in keys]
    for out_name, expr in final_map:
        names.append(out_name)
        exprs.append(expr)
    return ev2.project(state, names, exprs)


def _merge_reduce(a: dict, b: dict, aggs) -> dict:
    out = {}
    for out_name, in_name, func in aggs:
        pa_, pb = a[out_name], b[out_name]
        m = {}
        for k in set(pa_) | set(pb):
            va, vb = pa_.get(k), pb.get(k)
            if k in ("sum", "count", "sumsq"):
                m[k] = (va or 0) + (vb or 0)
            elif k == "min":
                vals = [v for v in (va, vb) if v is not None]
                m[k] = min(vals) if vals else None
            elif k == "max":
                vals = [v for v in (va, vb) if v is not None]
                m[k] = max(vals) if vals else None
            elif k == "any":
                m[k] = bool(va) or bool(vb)
            elif k == "all":
                m[k] = bool(va) and bool(vb)
        out[out_name] = m
    return out
