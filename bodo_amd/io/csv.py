"""CSV IO (reference: bodo/io/_csv_json_reader.cpp).  Round-1: host pandas
reader with rank block-slicing; byte-range parallel reader is the upgrade."""

from __future__ import annotations

from typing import Optional, Sequence

import pandas as pd

from ..core.table import Table


def schema_names(path: str, options: dict):
    head = pd.read_csv(path, nrows=10, **_reader_opts(options))
    return tuple(head.columns)


def _reader_opts(options: dict) -> dict:
    allowed = {"sep", "header", "names", "dtype", "parse_dates", "usecols",
               "skiprows", "na_values", "delimiter", "compression"}
    return {k: v for k, v in options.items() if k in allowed}


def read_shard(path: str, options: dict, columns: Optional[Sequence[str]], ctx) -> Table:
    opts = _reader_opts(options)
    if columns:
        use = opts.get("usecols")
        if use is None:
            opts["usecols"] = list(columns)
        elif all(isinstance(c, str) for c in use):
            opts["usecols"] = [c for c in use if c in columns]
        # integer usecols: keep the user's selection; prune after reading
    df = pd.read_csv(path, **opts)
    if columns:
        keep = [c for c in df.columns if c in set(columns)]
        if keep and len(keep) < len(df.columns):
            df = df[keep]
    n = len(df)
    w, r = ctx.world, ctx.rank
    base, rem = divmod(n, w)
    start = r * base + min(r, rem)
    stop = start + base + (1 if r < rem else 0)
    shard = df.iloc[start:stop].reset_index(drop=True)
    return Table.from_pandas(shard, ctx.device)


def write_shard(tbl: Table, path: str, ctx, **kwargs) -> None:
    import os

    df = tbl.to_pandas()
    if ctx.world == 1:
        df.to_csv(path, index=False, **kwargs)
        return
    os.makedirs(path, exist_ok=True)
    df.to_csv(os.path.join(path, f"part-{ctx.rank:05d}.csv"), index=False, **kwargs)
