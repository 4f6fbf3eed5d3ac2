"""Physical kernel layer: dispatches each op to the HIP extension on HBM-resident
columns or to host (numpy/pandas/pyarrow) kernels on CPU columns.

The GPU implementations live in csrc/ (hand-written gfx950 HIP) and are
mandatory on a GPU box: if a column is on CUDA and the extension is missing
we raise rather than silently falling back (round-end native check).
"""

from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import numpy as np
import torch

from ..core import types as bt
from ..core.column import Column
from ..core.table import Table
from ..core.types import TypeKind

from .evaluate import eval_expr, eval_filter, project  # noqa: F401


# ----------------------------------------------------------------------
# gather / take
# ----------------------------------------------------------------------

def gather(col: Column, idx: torch.Tensor) -> Column:
    n = int(idx.shape[0])
    if col.dtype.kind == TypeKind.LIST:
        return _gather_list(col, idx)
    if col.dtype.kind == TypeKind.STRUCT:
        out = Column(col.dtype, None,
                     col.mask[idx] if col.mask is not None else None,
                     length=n)
        out.children = [gather(c, idx) for c in col.children]
        return out
    if len(col) == 0 and n > 0:
        # gathering from an empty column only happens for null-padded rows
        # (outer-join unmatched side): produce an all-null column
        if col.dtype.kind == TypeKind.STRING:
            return Column(col.dtype,
                          torch.zeros(0, dtype=torch.uint8, device=idx.device),
                          torch.zeros(n, dtype=torch.bool, device=idx.device),
                          offsets=torch.zeros(n + 1, dtype=torch.int64,
                                              device=idx.device), length=n)
        from ..core import types as _bt

        store = _bt.torch_storage_dtype(col.dtype)
        data = torch.zeros(n, dtype=store, device=idx.device)
        mask = torch.zeros(n, dtype=torch.bool, device=idx.device)
        return Column(col.dtype, data, mask, dictionary=col.dictionary, length=n)
    if col.dtype.kind == TypeKind.STRING:
        if col.is_cuda:
            from . import gpu

            return gpu.gather_string(col, idx)
        return _gather_string_arrow(col, idx)
    data = col.data[idx]
    mask = col.mask[idx] if col.mask is not None else None
    out = Column(col.dtype, data, mask, dictionary=col.dictionary, length=n)
    out.val_range = col.val_range
    return out


def _gather_string_arrow(col: Column, idx: torch.Tensor) -> Column:
    """CPU string gather through Arrow take (C++ kernel)."""
    import pyarrow.compute as pc

    arr = col.to_arrow()
    taken = pc.take(arr, idx.numpy())
    return Column.from_arrow(taken, col.device)


def _gather_list(col: Column, idx: torch.Tensor) -> Column:
    """LIST gather: lengths gather + child range expansion (reference:
    array_item array gather in _array_utils.cpp)."""
    from ..core import types as _bt

    dev = col.device
    off = col.offsets
    lens = (off[1:] - off[:-1])[idx]
    new_off = torch.zeros(int(idx.numel()) + 1, dtype=torch.int64, device=dev)
    torch.cumsum(lens, 0, out=new_off[1:])
    starts = off[:-1][idx]
    total = int(new_off[-1].item())
    reps = torch.repeat_interleave(starts, lens)
    seq = torch.arange(total, device=dev) - torch.repeat_interleave(
        new_off[:-1], lens)
    child = gather(col.child, reps + seq) if total or len(col.child) else \
        gather(col.child, torch.zeros(0, dtype=torch.int64, device=dev))
    mask = col.mask[idx] if col.mask is not None else None
    out = Column(_bt.list_, None, mask, offsets=new_off,
                 length=int(idx.numel()))
    out.child = child
    return out


def explode_table(tbl: Table, column: str, pos_name=None) -> Table:
    """pandas explode semantics: each list element becomes a row; empty or
    null lists produce one row with a null value (reference role:
    bodo/libs/_lateral.cpp FLATTEN)."""
    from ..core import types as _bt

    col = tbl.column(column)
    if col.dtype.kind != TypeKind.LIST:
        return tbl  # non-list explode is the identity on scalars
    dev = tbl.device
    off = col.offsets
    lens = off[1:] - off[:-1]
    valid_list = col.mask if col.mask is not None else torch.ones(
        len(col), dtype=torch.bool, device=dev)
    out_lens = torch.where((lens == 0) | ~valid_list,
                           torch.ones_like(lens), lens)
    row_idx = torch.repeat_interleave(
        torch.arange(len(col), dtype=torch.int64, device=dev), out_lens)
    new_off = torch.zeros(len(col) + 1, dtype=torch.int64, device=dev)
    torch.cumsum(out_lens, 0, out=new_off[1:])
    total = int(new_off[-1].item())
    pos_in_row = torch.arange(total, device=dev) - new_off[:-1][row_idx]
    real = (lens[row_idx] > 0) & valid_list[row_idx] & \
        (pos_in_row < lens[row_idx])
    child_idx = (off[:-1][row_idx] + pos_in_row).clamp(
        min=0, max=max(len(col.child) - 1, 0))
    vals = gather(col.child, child_idx)
    vmask = real if vals.mask is None else (vals.mask & real)
    vals = Column(vals.dtype, vals.data, vmask, vals.offsets,
                  vals.dictionary, total)
    if vals.dtype.kind == TypeKind.LIST:
        vals.child = gather(col.child, child_idx).child
    names, cols = [], []
    for nm, c in zip(tbl.names, tbl.columns):
        if nm == column:
            names.append(nm)
            cols.append(vals)
        else:
            names.append(nm)
            cols.append(gather(c, row_idx))
    if pos_name is not None:
        names.append(pos_name)
        pc = Column(_bt.int64, pos_in_row,
                    real.clone() if (~real).any() else None)
        cols.append(pc)
    return Table(names, cols, total)


def take_table(tbl: Table, idx: torch.Tensor) -> Table:
    if tbl.device.type == "cuda" and len(tbl.columns) > 1:
        from . import gpu

        fused = gpu.take_table_fused(tbl, idx)
        if fused is not None:
            return fused
    return Table(tbl.names, [gather(c, idx) for c in tbl.columns], int(idx.shape[0]))


def slice_table(tbl: Table, start: int, stop: int) -> Table:
    n = len(tbl)
    start = max(0, min(start, n))
    stop = max(start, min(stop, n))
    dev = tbl.device
    idx = torch.arange(start, stop, dtype=torch.int64, device=dev)
    return take_table(tbl, idx)


# ----------------------------------------------------------------------
# concat
# ----------------------------------------------------------------------

def _is_all_null(c: Column) -> bool:
    if len(c) == 0:
        return True
    if c.mask is not None:
        return not bool(c.mask.any().item())
    if c.dtype.is_float:
        return bool(torch.isnan(c.data).all().item())
    return False


def _null_like(templ: Column, n: int) -> Column:
    """An all-null column with the template's dtype/shape (grouping-set
    arms where a key is rolled up)."""
    dev = templ.device
    mask = torch.zeros(n, dtype=torch.bool, device=dev)
    if templ.dtype.kind == TypeKind.STRING:
        return Column(bt.string, torch.zeros(0, dtype=torch.uint8, device=dev),
                      mask, offsets=torch.zeros(n + 1, dtype=torch.int64,
                                                device=dev), length=n)
    if templ.dtype.kind == TypeKind.DICT:
        return Column(bt.dictionary,
                      torch.zeros(n, dtype=torch.int32, device=dev), mask,
                      dictionary=templ.dictionary, length=n)
    return Column(templ.dtype, torch.zeros(n, dtype=templ.data.dtype,
                                           device=dev), mask)


def concat_columns(cols: Sequence[Column]) -> Column:
    cols = [c for c in cols]
    kinds = {c.dtype.kind for c in cols}
    if len(kinds) > 1:
        # mixed kinds: only sensible when the odd ones are all-null
        # (UNION of grouping-set arms) — replace them with typed nulls
        templ = next((c for c in cols if not _is_all_null(c)), cols[0])
        cols = [c if c.dtype.kind == templ.dtype.kind
                else _null_like(templ, len(c)) for c in cols]
    first = cols[0]
    if len(cols) == 1:
        return first
    n = sum(len(c) for c in cols)
    masks = None
    if any(c.mask is not None for c in cols):
        masks = torch.cat([
            c.mask if c.mask is not None
            else torch.ones(len(c), dtype=torch.bool, device=first.device)
            for c in cols])
    if first.dtype.kind == TypeKind.LIST:
        from ..core import types as _bt

        offs = [torch.zeros(1, dtype=torch.int64, device=first.device)]
        base = 0
        for c in cols:
            offs.append(c.offsets[1:] + base)
            base += int(c.offsets[-1].item())
        out = Column(_bt.list_, None, masks, offsets=torch.cat(offs),
                     length=n)
        out.child = concat_columns([c.child for c in cols])
        return out
    if first.dtype.kind == TypeKind.STRUCT:
        out = Column(first.dtype, None, masks, length=n)
        out.children = [
            concat_columns([c.children[i] for c in cols])
            for i in range(len(first.children))]
        return out
    if first.dtype.kind == TypeKind.STRING:
        datas, offs, base = [], [torch.zeros(1, dtype=torch.int64, device=first.device)], 0
        for c in cols:
            datas.append(c.data)
            offs.append(c.offsets[1:] + base)
            base += int(c.offsets[-1].item())
        return Column(bt.string, torch.cat(datas), masks,
                      offsets=torch.cat(offs), length=n)
    if first.dtype.kind == TypeKind.DICT:
        # unify dictionaries
        import pyarrow as pa

        dicts = [c.dictionary for c in cols]
        if all(d.equals(dicts[0]) for d in dicts[1:]):
            return Column(bt.dictionary, torch.cat([c.data for c in cols]), masks,
                          dictionary=dicts[0], length=n)
        merged = pa.concat_arrays([d.cast(pa.large_string()) for d in dicts])
        uniq = merged.unique()
        lut = {v: i for i, v in enumerate(uniq.to_pylist())}
        remapped = []
        for c in cols:
            rm = np.array([lut[v] for v in c.dictionary.to_pylist()], dtype=np.int32)
            rm_t = torch.from_numpy(rm).to(c.device)
            remapped.append(rm_t[c.data.long()])
        return Column(bt.dictionary, torch.cat(remapped).to(torch.int32), masks,
                      dictionary=uniq, length=n)
    datas = [c.data for c in cols]
    if len({d.dtype for d in datas}) > 1:
        t = datas[0].dtype
        for d in datas[1:]:
            t = torch.promote_types(t, d.dtype)
        datas = [d.to(t) for d in datas]
    out = Column(first.dtype, torch.cat(datas), masks,
                 dictionary=first.dictionary, length=n)
    if all(c.val_range is not None for c in cols):
        out.val_range = (min(c.val_range[0] for c in cols),
                         max(c.val_range[1] for c in cols))
    return out


def concat_tables(tables: Sequence[Table]) -> Table:
    tables = [t for t in tables if t is not None]
    first = tables[0]
    if len(tables) == 1:
        return first
    cols = []
    for i, name in enumerate(first.names):
        cols.append(concat_columns([t.columns[t.names.index(name)] for t in tables]))
    return Table(first.names, cols, sum(len(t) for t in tables))


# ----------------------------------------------------------------------
# hashing / partitioning
# ----------------------------------------------------------------------

def hash_columns(cols: Sequence[Column], seed: int = 0) -> torch.Tensor:
    """64-bit row hash over multiple key columns (int64 tensor, bit-identical
    across CPU and GPU so shuffles agree).  Reference role:
    bodo/libs/_array_hash.cpp."""
    if cols and cols[0].is_cuda:
        from . import gpu

        return gpu.hash_columns(cols, seed)
    return _hash_columns_cpu(cols, seed)


_M1 = np.uint64(0xff51afd7ed558ccd)
_M2 = np.uint64(0xc4ceb9fe1a85ec53)


def _mix64_np(x: np.ndarray) -> np.ndarray:
    # splitmix64-style finalizer; must match csrc/common.h mix64()
    with np.errstate(over="ignore"):
        x = x.astype(np.uint64, copy=True)
        x ^= x >> np.uint64(33)
        x *= _M1
        x ^= x >> np.uint64(33)
        x *= _M2
        x ^= x >> np.uint64(33)
    return x


def _col_hash_cpu(c: Column, seed: int) -> np.ndarray:
    k = c.dtype.kind
    if k == TypeKind.STRING:
        # hash the bytes per row with FNV-1a (matches csrc string hash)
        off = c.offsets.numpy()
        data = c.data.numpy()
        out = np.empty(len(c), dtype=np.uint64)
        _fnv_rows(data, off, out)
        h = out
    elif k == TypeKind.DICT:
        # hash the dictionary VALUES (not codes) so hashes agree across ranks
        # with different dictionaries
        dvals = c.dictionary.to_pylist()
        lut = np.array([_fnv_bytes(v.encode() if v is not None else b"")
                        for v in dvals], dtype=np.uint64)
        h = lut[c.data.numpy()]
    elif k == TypeKind.BOOL:
        h = _mix64_np(c.data.numpy().astype(np.uint64))
    elif c.dtype.is_float:
        f = c.data.numpy()
        if f.dtype == np.float32:
            f = f.astype(np.float64)
        bits = f.view(np.uint64).copy()
        bits[f == 0.0] = 0  # -0.0 == 0.0
        bits[np.isnan(f)] = np.uint64(0x7FF8000000000000)
        h = _mix64_np(bits)
    else:
        h = _mix64_np(c.data.numpy().astype(np.int64).view(np.uint64))
    if c.mask is not None:
        h = h.copy()
        h[~c.mask.numpy()] = np.uint64(0x9E3779B97F4A7C15)
    if seed:
        h = _mix64_np(h ^ np.uint64(seed))
    return h


def _fnv_bytes(b: bytes) -> int:
    h = 0xcbf29ce484222325
    for ch in b:
        h = ((h ^ ch) * 0x100000001b3) & 0xFFFFFFFFFFFFFFFF
    return h


def _fnv_rows(data: np.ndarray, off: np.ndarray, out: np.ndarray):
    FNV = np.uint64(0x100000001b3)
    with np.errstate(over="ignore"):
        for i in range(len(out)):
            h = np.uint64(0xcbf29ce484222325)
            for j in range(off[i], off[i + 1]):
                h = (h ^ np.uint64(data[j])) * FNV
            out[i] = h


def _hash_columns_cpu(cols, seed):
    acc = None
    with np.errstate(over="ignore"):
        for c in cols:
            h = _col_hash_cpu(c, seed)
            if acc is None:
                acc = h.copy()
            else:
                # boost-style hash combine, must match csrc combine
                acc = _mix64_np(acc * np.uint64(0x9E3779B97F4A7C15) + h)
    if acc is None:
        acc = np.zeros(0, dtype=np.uint64)
    return torch.from_numpy(acc.view(np.int64).copy())


def partition_indices(hashes: torch.Tensor, nparts: int) -> Tuple[torch.Tensor, torch.Tensor]:
    """Stable partition permutation: returns (perm, counts) such that rows
    perm[counts[:p].sum():counts[:p+1].sum()] belong to partition p."""
    p = torch.remainder(hashes, nparts)
    p = torch.where(p < 0, p + nparts, p)
    perm = torch.argsort(p, stable=True)
    counts = torch.bincount(p, minlength=nparts)
    return perm, counts


# ----------------------------------------------------------------------
# sort
# ----------------------------------------------------------------------

def sort_indices(cols: Sequence[Column], ascending: Sequence[bool],
                 na_position: str = "last") -> torch.Tensor:
    """Stable lexicographic argsort (last key first, stable iterations).
    Small-range key sets pack into ONE order-preserving u64 -> a single
    argsort instead of one stable pass per key."""
    device = cols[0].device if cols else torch.device("cpu")
    n = len(cols[0]) if cols else 0
    packed = pack_ordered_keys(cols, ascending)
    if packed is not None:
        return torch.argsort(packed, stable=True)
    idx = torch.arange(n, dtype=torch.int64, device=device)
    for c, asc in reversed(list(zip(cols, ascending))):
        keys = _sort_key_tensor(c, asc, na_position)
        k = keys[idx]
        order = torch.argsort(k, stable=True)
        idx = idx[order]
    return idx


def pack_ordered_keys(cols: Sequence[Column],
                      ascending: Sequence[bool]) -> Optional[torch.Tensor]:
    """Pack sort keys into one int64 whose numeric order equals the
    lexicographic (keys, ascending) order.  First key occupies the most
    significant bits; DICT codes are rank-remapped by dictionary value
    order; descending keys are bit-flipped within their range.  Returns
    None when any key is unsuitable (mask, float, wide range, strings)."""
    if not cols:
        return None
    widths = []
    parts = []
    for c, asc in zip(cols, ascending):
        if c.mask is not None:
            return None
        k = c.dtype.kind
        if k == TypeKind.DICT:
            import pyarrow.compute as pc

            order = pc.array_sort_indices(c.dictionary).to_numpy()
            rank = np.empty(len(order), dtype=np.int64)
            rank[order] = np.arange(len(order))
            vals = torch.from_numpy(rank).to(c.device)[c.data.long()]
            lo, hi = 0, max(0, len(order) - 1)
        elif k == TypeKind.BOOL:
            vals = c.data.long()
            lo, hi = 0, 1
        elif (c.dtype.is_integer or k == TypeKind.DATE32) and c.val_range:
            lo, hi = c.val_range
            vals = c.data.long()
        else:
            return None
        width = max(1, int(hi - lo + 1).bit_length())
        v = vals - lo
        if not asc:
            v = (hi - lo) - v
        parts.append(v)
        widths.append(width)
    if sum(widths) > 63:
        return None
    packed = None
    for v, w in zip(parts, widths):
        packed = v if packed is None else ((packed << w) | v)
    return packed


def _sort_key_tensor(c: Column, asc: bool, na_position: str) -> torch.Tensor:
    k = c.dtype.kind
    if k == TypeKind.DICT:
        # dense rank: order codes by dictionary value order (equal values
        # MUST get equal ranks for multi-key lexsort stability)
        import pyarrow.compute as pc

        order = pc.array_sort_indices(c.dictionary).to_numpy()
        rank = np.empty(len(order), dtype=np.int64)
        rank[order] = np.arange(len(order))
        data = torch.from_numpy(rank).to(c.device)[c.data.long()]
    elif k == TypeKind.STRING:
        import pyarrow as pa
        import pyarrow.compute as pc

        arr = c.to_device("cpu").to_arrow()
        denc = pc.dictionary_encode(arr)
        order = pc.array_sort_indices(denc.dictionary).to_numpy()
        rank = np.empty(len(order), dtype=np.int64)
        rank[order] = np.arange(len(order))
        codes = denc.indices.to_numpy(zero_copy_only=False)
        codes = np.where(np.isnan(codes.astype(np.float64)), 0, codes).astype(np.int64) \
            if codes.dtype == object else codes.astype(np.int64)
        data = torch.from_numpy(rank[codes]).to(c.device)
        if c.mask is None and arr.null_count:
            nullmask = torch.from_numpy(
                pc.is_valid(arr).to_numpy(zero_copy_only=False).astype(bool)
            ).to(c.device)
            sent = np.iinfo(np.int64).max if na_position == "last" else np.iinfo(np.int64).min
            data = torch.where(nullmask, data, torch.full_like(data, sent))
    elif c.dtype.is_float:
        data = c.data.clone()
        nan = torch.isnan(data)
        big = torch.finfo(data.dtype).max if na_position == "last" else torch.finfo(data.dtype).min
        if not asc:
            big = -big
        data = torch.where(nan, torch.full_like(data, big), data)
    elif c.data.dtype == torch.bool:
        data = c.data.to(torch.int8)
    else:
        data = c.data
    if not asc:
        if data.dtype == torch.bool:
            data = data.to(torch.int8)
        data = -data.to(torch.float64) if data.dtype.is_floating_point else -data
    if c.mask is not None:
        data = data.clone()
        if data.dtype.is_floating_point:
            sentinel = torch.finfo(data.dtype).max if na_position == "last" else torch.finfo(data.dtype).min
        else:
            sentinel = torch.iinfo(data.dtype).max if na_position == "last" else torch.iinfo(data.dtype).min
        data[~c.mask] = sentinel
    return data


# ----------------------------------------------------------------------
# reductions
# ----------------------------------------------------------------------

def _reduce_wrap_val(col: Column, v):
    """Storage scalar -> user value (ns int -> Timestamp, days -> Timestamp);
    reductions report VALUES, never storage encodings."""
    if v is None:
        return None
    import pandas as pd

    if col.dtype.kind == TypeKind.TIMESTAMP_NS:
        return pd.Timestamp(int(v))
    if col.dtype.kind == TypeKind.DATE32:
        return pd.Timestamp(int(v) * 86_400_000_000_000)
    if col.dtype.kind == TypeKind.DURATION_NS:
        return pd.Timedelta(int(v))
    return v


# reductions where the partial depends on actual VALUES (strings must not
# reduce over dictionary codes)
_VALUE_REDUCES = {"min", "max", "first", "last", "mode", "median",
                  "nunique", "array_agg"}


def reduce_column(col: Column, func: str):
    """Local partial reduction -> dict of partials (combined across ranks by
    the executor's _combine_reduce).  Reference role: the parallel agg
    combine in bodo/libs/_groupby_ftypes + distributed_api dist_reduce."""
    if callable(func):
        # custom python agg (LISTAGG / PERCENTILE_*): ship local values,
        # the combine applies the callable to the concatenation
        return {"vals_list": col.to_arrow().to_pylist()}
    if col.dtype.kind == TypeKind.DECIMAL128:
        from .evaluate import decimal_to_float

        col = decimal_to_float(col)  # scalar reductions report values
    if col.dtype.kind in (TypeKind.STRING, TypeKind.DICT) \
            and func in _VALUE_REDUCES:
        import pandas as pd

        vals = pd.Series(
            [v for v in col.to_arrow().to_pylist() if v is not None],
            dtype=object)
        n = len(vals)
        if func == "min":
            return {"min": vals.min() if n else None}
        if func == "max":
            return {"max": vals.max() if n else None}
        if func == "first":
            return {"has": n > 0, "val": vals.iloc[0] if n else None}
        if func == "last":
            return {"has": n > 0, "val": vals.iloc[-1] if n else None}
        if func == "mode":
            return {"counts": vals.value_counts().to_dict()}
        if func == "nunique":
            return {"uniq": set(vals)}
        raise NotImplementedError(f"reduce {func} on strings")
    data, mask = col.data, col.mask
    if col.dtype.is_float:
        valid = ~torch.isnan(data)
        if mask is not None:
            valid &= mask
    elif mask is not None:
        valid = mask
    else:
        valid = None
    if valid is not None:
        data = data[valid]
    n = int(data.numel())
    if func in ("sum", "mean", "var", "std", "sem", "kurt", "skew"):
        acc = data.to(torch.float64) if not col.dtype.is_integer else data.to(torch.int64)
        s = acc.sum().item() if n else (0.0 if not col.dtype.is_integer else 0)
        out = {"sum": s, "count": n}
        if func in ("var", "std", "sem", "kurt", "skew"):
            f = acc.to(torch.float64)
            out["sumsq"] = float((f ** 2).sum().item()) if n else 0.0
        if func in ("kurt", "skew"):
            out["sum3"] = float((f ** 3).sum().item()) if n else 0.0
            out["sum4"] = float((f ** 4).sum().item()) if n else 0.0
        return out
    if func in ("count",):
        return {"count": n}
    if func == "size":
        return {"count": len(col)}
    if func == "min":
        return {"min": _reduce_wrap_val(col, data.min().item() if n else None)}
    if func == "max":
        return {"max": _reduce_wrap_val(col, data.max().item() if n else None)}
    if func == "first":
        return {"has": n > 0,
                "val": _reduce_wrap_val(col, data[0].item() if n else None)}
    if func == "last":
        return {"has": n > 0,
                "val": _reduce_wrap_val(col, data[-1].item() if n else None)}
    if func == "any":
        return {"any": bool(data.any().item()) if n else False}
    if func == "all":
        return {"all": bool(data.all().item()) if n else True}
    if func == "prod":
        return {"prod": data.to(torch.float64).prod().item() if n else 1.0}
    if func == "median":
        # exact: ship local sorted values (scalar result; the grouped median
        # runs co-located on shuffled groups instead, relational.py)
        v = torch.sort(data.to(torch.float64))[0].cpu().numpy() if n else None
        return {"vals": v}
    if func == "mode":
        u, c = torch.unique(data, return_counts=True)
        return {"counts": {
            _reduce_wrap_val(col, k): int(v)
            for k, v in zip(u.cpu().tolist(), c.cpu().tolist())}}
    if func == "nunique":
        return {"uniq": set(torch.unique(data).cpu().tolist()) if n else set()}
    if func == "approx_nunique":
        from ..utils import sketches

        return {"hll": sketches.hll_registers([col]).cpu()}
    if func == "array_agg":
        return {"vals_list": [v for v in col.to_arrow().to_pylist()
                              if v is not None]}
    raise NotImplementedError(f"reduce {func}")
