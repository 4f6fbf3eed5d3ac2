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


_RANGE_MIN_BYTES = 1 << 22


def _aligned_offset(f, pos: int, data_start: int) -> int:
    """First byte after the next newline at/after pos (deterministic: every
    rank computes any boundary identically)."""
    if pos <= data_start:
        return data_start
    f.seek(pos - 1)
    f.readline()  # consume up to and including the next newline
    return f.tell()


def read_shard(path: str, options: dict, columns: Optional[Sequence[str]], ctx) -> Table:
    opts = _reader_opts(options)
    if columns:
        use = opts.get("usecols")
        if use is None:
            opts["usecols"] = list(columns)
        elif all(isinstance(c, str) for c in use):
            opts["usecols"] = [c for c in use if c in columns]
        # integer usecols: keep the user's selection; prune after reading
    df = _read_local(path, opts, ctx)
    if columns:
        keep = [c for c in df.columns if c in set(columns)]
        if keep and len(keep) < len(df.columns):
            df = df[keep]
    return Table.from_pandas(df.reset_index(drop=True), ctx.device)


def _read_local(path: str, opts: dict, ctx) -> pd.DataFrame:
    """This rank's rows.  Large plain files split by newline-aligned byte
    ranges so each rank reads ONLY its slice (reference:
    bodo/io/_csv_json_reader.cpp byte-range division); everything else
    falls back to whole-file + row-slice.  Limitation shared with the
    reference's splitter: quoted fields containing newlines need the
    fallback."""
    import io
    import os

    w, r = ctx.world, ctx.rank
    simple = (w > 1 and os.path.isfile(path)
              and opts.get("compression") in (None, "infer")
              and not str(path).endswith((".gz", ".bz2", ".zip", ".xz"))
              and opts.get("skiprows") in (None, 0)
              and opts.get("header", "infer") in ("infer", 0))
    if simple:
        size = os.path.getsize(path)
        if size >= _RANGE_MIN_BYTES:
            with open(path, "rb") as f:
                header = f.readline()
                data_start = f.tell()
                span = size - data_start
                lo = _aligned_offset(f, data_start + r * span // w,
                                     data_start)
                hi = _aligned_offset(f, data_start + (r + 1) * span // w,
                                     data_start) if r + 1 < w else size
                f.seek(lo)
                chunk = f.read(max(hi - lo, 0))
            buf = io.BytesIO(header + chunk)
            return pd.read_csv(buf, **opts)
    df = pd.read_csv(path, **opts)
    n = len(df)
    base, rem = divmod(n, w)
    start = r * base + min(r, rem)
    stop = start + base + (1 if r < rem else 0)
    return df.iloc[start:stop]


def write_shard(tbl: Table, path: str, ctx, **kwargs) -> None:
    import os

    df = tbl.to_pandas()
    if ctx.world == 1:
        df.to_csv(path, index=False, **kwargs)
        return
    os.makedirs(path, exist_ok=True)
    df.to_csv(os.path.join(path, f"part-{ctx.rank:05d}.csv"), index=False, **kwargs)
