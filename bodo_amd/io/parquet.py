"""Parquet IO: rank-parallel read (row-group granularity, Arrow decode on the
host IO path; on-GPU decode for plain/dict-encoded columns is the csrc
upgrade path) and distributed part-file writes.

Reference roles: bodo/io/parquet_reader.cpp (piece assignment, filter
pushdown), bodo/io/parquet_write.cpp (part-NNN.parquet layout).
"""

from __future__ import annotations

import glob
import os
from typing import Optional, Sequence, Tuple

import pyarrow as pa
import pyarrow.dataset as pads
import pyarrow.parquet as pq

from ..core.table import Table
from ..plan.expr import BoolOp, ColRef, Cmp, Const, Expr, IsIn, IsNull


def schema_names(path: str) -> Tuple[str, ...]:
    d = _dataset(path)
    return tuple(d.schema.names)


def _dataset(path: str) -> pads.Dataset:
    from . import fs as bfs
    from . import iceberg

    files = iceberg.resolve_scan_path(path)
    if files is not None:
        return pads.dataset(files, format="parquet")
    pafs, p = bfs.resolve(path)
    if pafs is not None:
        return pads.dataset(p, format="parquet", filesystem=pafs)
    return pads.dataset(path, format="parquet")


def _expr_to_arrow(e: Expr):
    import pyarrow.compute as pc

    if isinstance(e, Cmp):
        l, r = e.left, e.right
        flip = {"lt": "gt", "le": "ge", "gt": "lt", "ge": "le", "eq": "eq", "ne": "ne"}
        if isinstance(l, Const) and isinstance(r, ColRef):
            l, r = r, l
            op = flip[e.op]
        else:
            op = e.op
        assert isinstance(l, ColRef) and isinstance(r, Const)
        field = pc.field(l.name)
        val = r.value
        import pandas as pd

        if isinstance(val, pd.Timestamp):
            val = val.to_pydatetime()
        return {
            "lt": field < val, "le": field <= val, "gt": field > val,
            "ge": field >= val, "eq": field == val, "ne": field != val,
        }[op]
    if isinstance(e, IsIn):
        return pc.field(e.operand.name).isin(list(e.values))
    if isinstance(e, IsNull):
        f = pc.field(e.operand.name)
        return f.is_valid() if e.negate else ~f.is_valid()
    if isinstance(e, BoolOp):
        a, b = _expr_to_arrow(e.left), _expr_to_arrow(e.right)
        return (a & b) if e.op == "and" else (a | b)
    raise NotImplementedError(type(e).__name__)


def read_shard(path: str, columns: Optional[Sequence[str]],
               filters: Sequence[Expr], ctx) -> Table:
    from . import iceberg

    from . import fs as bfs

    if ctx.device.type == "cuda" and not filters and \
            iceberg.resolve_scan_path(path) is None and \
            not bfs.is_remote(path):
        # on-GPU decode fast path (uncompressed PLAIN / RLE_DICTIONARY)
        from . import parquet_gpu

        try:
            t = parquet_gpu.read_shard_gpu(path, list(columns) if columns
                                           else None, ctx)
        except Exception:
            t = None
        if t is not None:
            return t
    d = _dataset(path)
    frags = list(d.get_fragments())
    # explode into row-group pieces and block-assign to ranks
    # (reference: parquet_reader.h:50-60 balances avg pieces per rank)
    pieces = []
    for f in frags:
        try:
            rgs = f.split_by_row_group()
            pieces.extend(rgs)
        except Exception:
            pieces.append(f)
    w, r = ctx.world, ctx.rank
    n = len(pieces)
    base, rem = divmod(n, w)
    start = r * base + min(r, rem)
    stop = start + base + (1 if r < rem else 0)
    my = pieces[start:stop]
    filt = None
    for e in filters:
        ae = _expr_to_arrow(e)
        filt = ae if filt is None else (filt & ae)
    cols = list(columns) if columns else None
    tables = []
    for piece in my:
        t = piece.to_table(columns=cols, filter=filt)
        tables.append(t)
    if tables:
        out = pa.concat_tables(tables).combine_chunks()
    else:
        schema = d.schema
        if cols:
            schema = pa.schema([schema.field(c) for c in cols])
        out = schema.empty_table()
    # auto dict-encode low-cardinality string columns for device residency
    from ..core.table import dict_encode_strings

    out = dict_encode_strings(out)
    return Table.from_arrow(out, ctx.device)


def _dict_encode_strings(tbl: pa.Table, threshold: float = 0.5,
                         sample: int = 8192) -> pa.Table:
    import pyarrow.compute as pc

    new_cols = []
    changed = False
    for i, f in enumerate(tbl.schema):
        col = tbl.column(i)
        if pa.types.is_string(f.type) or pa.types.is_large_string(f.type):
            head = col.slice(0, min(sample, len(col)))
            try:
                nuniq = len(pc.unique(head.combine_chunks()))
            except Exception:
                nuniq = sample
            if len(head) > 0 and nuniq <= max(1, int(len(head) * threshold)):
                col = pc.dictionary_encode(col.combine_chunks())
                changed = True
        new_cols.append(col)
    if not changed:
        return tbl
    return pa.table(dict(zip(tbl.column_names, new_cols)))


def write_shard(tbl: Table, path: str, compression, ctx,
                partition_cols=()) -> None:
    at = tbl.to_device("cpu").to_arrow()
    if partition_cols:
        # hive-style partition directories; per-rank unique basenames
        import pyarrow.dataset as pads

        pads.write_dataset(
            at, path, format="parquet", partitioning_flavor="hive",
            partitioning=list(partition_cols),
            basename_template=f"part-{ctx.rank:05d}-{{i}}.parquet",
            existing_data_behavior="overwrite_or_ignore")
        return
    from . import fs as bfs

    pafs, p = bfs.resolve(path)
    if pafs is not None:
        if ctx.world == 1 and p.endswith(".parquet"):
            with pafs.open_output_stream(p) as f:
                pq.write_table(at, f, compression=compression)
            return
        pafs.create_dir(p, recursive=True)
        with pafs.open_output_stream(
                p.rstrip("/") + f"/part-{ctx.rank:05d}.parquet") as f:
            pq.write_table(at, f, compression=compression)
        return
    if ctx.world == 1 and path.endswith(".parquet") and not os.path.isdir(path):
        pq.write_table(at, path, compression=compression)
        return
    os.makedirs(path, exist_ok=True)
    fname = os.path.join(path, f"part-{ctx.rank:05d}.parquet")
    pq.write_table(at, fname, compression=compression)
