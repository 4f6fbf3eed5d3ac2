"""GPU op implementations: thin wrappers over the hand-written gfx950 HIP
kernels in csrc/ (built in-tree as ``bodo_amd_kernels``).

Fails loudly if the native extension is missing on a CUDA device (no silent
eager fallback; round-end native check).
"""

from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import numpy as np
import torch

from ..core import types as bt
from ..core.column import Column
from ..core.table import Table
from ..core.types import TypeKind

try:
    import bodo_amd_kernels as _K
except ImportError as e:  # pragma: no cover
    _K = None
    _IMPORT_ERR = e


def kernels():
    if _K is None:
        raise ImportError(
            "bodo_amd_kernels HIP extension not built; run "
            "`python setup.py build_ext --inplace` "
            f"(import error: {_IMPORT_ERR})")
    return _K


def _fnv_bytes(b: bytes) -> int:
    h = 0xcbf29ce484222325
    for ch in b:
        h = ((h ^ ch) * 0x100000001b3) & 0xFFFFFFFFFFFFFFFF
    return h


_DICT_HASH_CACHE: dict = {}


def _dict_hash_lut(col: Column) -> torch.Tensor:
    key = (id(col.dictionary), str(col.device))
    lut = _DICT_HASH_CACHE.get(key)
    if lut is None:
        vals = col.dictionary.to_pylist()
        arr = np.array(
            [_fnv_bytes(v.encode() if v is not None else b"") for v in vals],
            dtype=np.uint64)
        lut = torch.from_numpy(arr.view(np.int64)).to(col.device)
        _DICT_HASH_CACHE[key] = lut
        if len(_DICT_HASH_CACHE) > 256:
            _DICT_HASH_CACHE.clear()
            _DICT_HASH_CACHE[key] = lut
    return lut


def _col_args(cols: Sequence[Column]):
    datas, masks, offsets, auxs, dtypes = [], [], [], [], []
    for c in cols:
        if c.dtype.kind == TypeKind.STRING:
            datas.append(c.data if c.data is not None else
                         torch.zeros(0, dtype=torch.uint8, device=c.device))
            offsets.append(c.offsets)
            auxs.append(None)
        elif c.dtype.kind == TypeKind.DICT:
            datas.append(c.data)
            offsets.append(None)
            auxs.append(_dict_hash_lut(c))
        else:
            data = c.data
            if data.dtype == torch.bool:
                data = data.view(torch.uint8)
            datas.append(data)
            offsets.append(None)
            auxs.append(None)
        masks.append(None if c.mask is None else c.mask.view(torch.uint8))
        dtypes.append(int(c.dtype.kind))
    return datas, masks, offsets, auxs, dtypes


def hash_columns(cols: Sequence[Column], seed: int = 0) -> torch.Tensor:
    K = kernels()
    n = len(cols[0])
    datas, masks, offsets, auxs, dtypes = _col_args(cols)
    return K.hash_columns(datas, masks, offsets, auxs, dtypes, n, seed)


_DT_FIELD_ID = {
    "year": 0, "month": 1, "day": 2, "hour": 3, "minute": 4, "second": 5,
    "dayofweek": 6, "weekday": 6, "dayofyear": 7, "quarter": 8, "date": 9,
    "normalize": 10, "floor_day": 10,
}

_DT_RANGE = {
    "month": (1, 12), "day": (1, 31), "hour": (0, 23), "minute": (0, 59),
    "second": (0, 59), "dayofweek": (0, 6), "weekday": (0, 6),
    "dayofyear": (1, 366), "quarter": (1, 4),
}

_DT_OUT_KIND = {
    "year": (0, bt.int16), "month": (0, bt.int8), "day": (0, bt.int8),
    "hour": (0, bt.int8), "minute": (0, bt.int8), "second": (0, bt.int8),
    "dayofweek": (0, bt.int8), "weekday": (0, bt.int8),
    "dayofyear": (0, bt.int16), "quarter": (0, bt.int8),
    "date": (1, bt.date32), "normalize": (2, bt.timestamp_ns),
    "floor_day": (2, bt.timestamp_ns),
}


def dt_field(col: Column, fld: str) -> Column:
    K = kernels()
    is_date32 = 1 if col.dtype.kind == TypeKind.DATE32 else 0
    out_kind, out_dtype = _DT_OUT_KIND[fld]
    res = K.dt_field(col.data, is_date32, _DT_FIELD_ID[fld], out_kind)
    # int16 kernel output narrowed per field on the python side
    store = bt.torch_storage_dtype(out_dtype)
    if res.dtype != store:
        res = res.to(store)
    out = Column(out_dtype, res, col.mask)
    out.val_range = _DT_RANGE.get(fld)
    return out


def gather_string(col: Column, idx: torch.Tensor) -> Column:
    K = kernels()
    data = col.data if col.data is not None else torch.zeros(
        0, dtype=torch.uint8, device=col.device)
    out, new_off = K.gather_string(data, col.offsets, idx.to(torch.int64))
    mask = col.mask[idx] if col.mask is not None else None
    return Column(bt.string, out, mask, offsets=new_off, length=int(idx.numel()))


# ----------------------------------------------------------------------
# groupby
# ----------------------------------------------------------------------

_AGG_OP = {
    "sum_f64": 0, "sum_i64": 1, "count": 2, "min_f64": 3, "max_f64": 4,
    "min_i64": 5, "max_i64": 6, "size": 7, "first_row": 8, "last_row": 9,
    "prod_f64": 10,
}

_F64_MAX = float(np.finfo(np.float64).max)
_I64_MAX = int(np.iinfo(np.int64).max)
_I64_MIN = int(np.iinfo(np.int64).min)


def _keys_valid_mask(cols: Sequence[Column]) -> Optional[torch.Tensor]:
    m = None
    for c in cols:
        v = None
        if c.mask is not None:
            v = c.mask
        if c.dtype.is_float:
            nn = ~torch.isnan(c.data)
            v = nn if v is None else (v & nn)
        if v is not None:
            m = v if m is None else (m & v)
    return m


def _try_pack_keys(key_cols: Sequence[Column]) -> Optional[Column]:
    """Pack small-range integer/dict/bool keys into ONE int64 column so the
    hash-table equality check reads 8 bytes instead of one random read per
    key column (reference analog: key normalization in _hash_join.cpp)."""
    shift = 0
    packed = None
    for c in key_cols:
        if c.mask is not None:
            return None
        k = c.dtype.kind
        if k == TypeKind.DICT:
            lo, hi = 0, max(0, len(c.dictionary) - 1)
        elif k == TypeKind.BOOL:
            lo, hi = 0, 1
        elif (c.dtype.is_integer or k == TypeKind.DATE32) and c.val_range:
            lo, hi = c.val_range
        else:
            return None
        width = max(1, int(hi - lo + 1).bit_length())
        if shift + width > 63:
            return None
        part = (c.data.long() - lo) << shift
        packed = part if packed is None else (packed | part)
        shift += width
    if packed is None:
        return None
    return Column(bt.int64, packed)


def groupby_build(key_cols: Sequence[Column]) -> Tuple[torch.Tensor, torch.Tensor]:
    K = kernels()
    n = len(key_cols[0])
    pk = _try_pack_keys(key_cols)
    if pk is not None:
        return K.groupby_build_packed(pk.data)
    h = hash_columns(key_cols)
    datas, masks, offsets, auxs, dtypes = _col_args(key_cols)
    row_gid, uniq_rows = K.groupby_build(datas, masks, offsets, auxs, dtypes, n, h)
    return row_gid, uniq_rows


def groupby_local(tbl: Table, keys: Sequence[str],
                  aggs: Sequence[Tuple[str, str, str]], dropna: bool = True) -> Table:
    from . import take_table

    key_cols = [tbl.column(k) for k in keys]
    n = len(tbl)
    if dropna and n:
        valid = _keys_valid_mask(key_cols)
        if valid is not None:
            idx = torch.nonzero(valid, as_tuple=False).reshape(-1)
            if int(idx.numel()) != n:
                tbl = take_table(tbl, idx)
                key_cols = [tbl.column(k) for k in keys]
                n = len(tbl)
    if n == 0:
        return _empty_gb_result(tbl, keys, aggs)
    row_gid, uniq_rows = groupby_build(key_cols)
    ngroups = int(uniq_rows.numel())
    out_names = list(keys)
    out_cols = [_gather_any(tbl.column(k), uniq_rows) for k in keys]
    # fused path: batch simple aggs (count/size/sum/mean/min/max) into one
    # kernel pass sharing the row_gid read; complex aggs go one-by-one
    FUSABLE = {"count", "size", "sum", "mean", "min", "max"}
    fused_batch = []
    results = {}

    def flush_fused():
        if not fused_batch:
            return
        _run_fused(fused_batch, row_gid, ngroups, results)
        fused_batch.clear()

    for out_name, in_name, func in aggs:
        col = tbl.column(in_name) if in_name and tbl.has_column(in_name) else None
        if func in FUSABLE and (col is None or col.dtype.kind != TypeKind.STRING):
            fused_batch.append((out_name, col, func))
            if len(fused_batch) == 4:
                flush_fused()
        else:
            results[out_name] = _agg_one(col, row_gid, ngroups, func,
                                         uniq_rows, tbl)
    flush_fused()
    for out_name, in_name, func in aggs:
        out_cols.append(results[out_name])
        out_names.append(out_name)
    return Table(out_names, out_cols, ngroups)


def _run_fused(batch, row_gid, ngroups, results):
    """Launch one agg_update_fused for up to 4 (out_name, col, func)."""
    K = kernels()
    # means over provably-NaN-free unmasked float columns can reuse a shared
    # group-size count instead of carrying their own valid-count atomics
    clean_float = {}
    for _, col, func in batch:
        if func == "mean" and col is not None and col.mask is None \
                and col.dtype.is_float and id(col) not in clean_float:
            clean_float[id(col)] = not bool(torch.isnan(col.data).any().item())
    share_size = any(clean_float.values()) and any(
        f == "size" or (f == "count" and c is not None and c.mask is None
                        and not c.dtype.is_float)
        for _, c, f in batch)
    datas, masks, dtypes, ops_, init_fs, init_is, wants, posts = \
        [], [], [], [], [], [], [], []
    for out_name, col, func in batch:
        is_float = col is not None and col.dtype.is_float
        nullable = col is not None and ((col.mask is not None) or is_float)
        col_for_data = col
        if func == "size" or col is None or (
                func == "count" and col.mask is None and not is_float):
            # count over a non-nullable column == size: no column read
            ops_.append(_AGG_OP["size"])
            init_fs.append(0.0)
            init_is.append(0)
            wants.append(1)
            posts.append(("cnt_i64", None))
            col_for_data = None
        elif func == "count":
            ops_.append(_AGG_OP["count"])
            init_fs.append(0.0)
            init_is.append(0)
            wants.append(1)
            posts.append(("cnt_i64", None))
        elif func in ("sum", "mean"):
            if is_float or func == "mean":
                ops_.append(_AGG_OP["sum_f64"])
                init_fs.append(0.0)
                init_is.append(0)
                shared = (func == "mean" and share_size
                          and clean_float.get(id(col), False))
                wants.append(0 if shared else 1)
                posts.append(("mean_shared" if shared else
                              ("mean" if func == "mean" else "acc_f64"), None))
            else:
                ops_.append(_AGG_OP["sum_i64"])
                init_fs.append(0.0)
                init_is.append(0)
                wants.append(0)
                posts.append(("acc_i64", None))
        else:  # min / max
            if is_float:
                ops_.append(_AGG_OP[f"{func}_f64"])
                init_fs.append(_F64_MAX if func == "min" else -_F64_MAX)
                init_is.append(0)
                wants.append(1)
                posts.append(("minmax_f64", None))
            else:
                ops_.append(_AGG_OP[f"{func}_i64"])
                init_fs.append(0.0)
                init_is.append(_I64_MAX if func == "min" else _I64_MIN)
                wants.append(1)
                posts.append(("minmax_i64", col))
        if col_for_data is None:
            data = torch.zeros(1, dtype=torch.int8, device=row_gid.device)
            masks.append(None)
            dtypes.append(int(TypeKind.INT8))
        else:
            data = col_for_data.data
            if data.dtype == torch.bool:
                data = data.view(torch.uint8)
            masks.append(None if col_for_data.mask is None
                         else col_for_data.mask.view(torch.uint8))
            dtypes.append(int(col_for_data.dtype.kind))
        datas.append(data)
    flat = K.agg_update_fused(datas, masks, dtypes, ops_, init_fs, init_is,
                              wants, row_gid, ngroups)
    shared_cnt = None
    for k in range(len(batch)):
        if posts[k][0] == "cnt_i64":
            shared_cnt = flat[2 * k + 1]
            break
    for k, (out_name, col, func) in enumerate(batch):
        acc, cnt = flat[2 * k], flat[2 * k + 1]
        kind, extra = posts[k]
        nullable = col is not None and ((col.mask is not None)
                                        or col.dtype.is_float)
        if kind == "cnt_i64":
            results[out_name] = Column(bt.int64, cnt)
        elif kind == "acc_f64":
            results[out_name] = Column(bt.float64, acc)
        elif kind == "acc_i64":
            results[out_name] = Column(bt.int64, acc)
        elif kind == "mean_shared":
            results[out_name] = Column(
                bt.float64, acc / shared_cnt.to(torch.float64))
        elif kind == "mean" and func == "mean":
            results[out_name] = Column(bt.float64, acc / cnt.to(torch.float64))
        elif kind == "mean":
            results[out_name] = Column(bt.float64, acc)
        elif kind == "minmax_f64":
            empty = cnt == 0
            out = torch.where(empty, torch.full_like(acc, float("nan")), acc)
            results[out_name] = Column(bt.float64, out)
        else:  # minmax_i64
            if nullable:
                results[out_name] = Column(bt.float64, torch.where(
                    cnt == 0,
                    torch.full((ngroups,), float("nan"), dtype=torch.float64,
                               device=acc.device),
                    acc.to(torch.float64)))
            elif col.dtype.kind == TypeKind.TIMESTAMP_NS:
                results[out_name] = Column(bt.timestamp_ns, acc)
            elif col.dtype.kind == TypeKind.DATE32:
                results[out_name] = Column(bt.date32, acc.to(torch.int32))
            elif col.dtype.kind == TypeKind.BOOL:
                results[out_name] = Column(bt.boolean, acc.to(torch.bool))
            else:
                results[out_name] = Column(col.dtype, acc.to(col.data.dtype))


def _gather_any(col: Column, idx: torch.Tensor) -> Column:
    from . import gather

    return gather(col, idx)


def _agg_one(col: Optional[Column], row_gid: torch.Tensor, ngroups: int,
             func: str, uniq_rows: torch.Tensor, tbl: Table) -> Column:
    K = kernels()

    def upd(op, init_f=0.0, init_i=0, want_count=False):
        data = col.data if col is not None else torch.zeros(
            int(row_gid.numel()), dtype=torch.int8, device=row_gid.device)
        if data.dtype == torch.bool:
            data = data.view(torch.uint8)
        mask = None if (col is None or col.mask is None) else col.mask.view(torch.uint8)
        offsets = col.offsets if (col is not None and col.dtype.kind == TypeKind.STRING) else None
        dtype = int(col.dtype.kind) if col is not None else int(TypeKind.INT8)
        return K.agg_update(data, mask, offsets, dtype, row_gid, ngroups,
                            _AGG_OP[op], init_f, init_i, want_count)

    if func == "size":
        (acc, cnt) = upd("size")
        return Column(bt.int64, cnt)
    if func == "count":
        acc, cnt = upd("count")
        return Column(bt.int64, cnt)
    is_float = col.dtype.is_float
    nullable = (col.mask is not None) or is_float
    if func == "sum":
        if is_float:
            acc, cnt = upd("sum_f64", want_count=True)
            return Column(bt.float64, acc)
        acc = upd("sum_i64")[0]
        if col.dtype.kind == TypeKind.BOOL or col.mask is not None:
            # pandas: bool sum -> int64; masked int sums become float on
            # to_pandas path anyway
            pass
        return Column(bt.int64, acc)
    if func in ("min", "max"):
        if is_float:
            init = _F64_MAX if func == "min" else -_F64_MAX
            acc, cnt = upd(f"{func}_f64", init_f=init, want_count=True)
            empty = cnt == 0
            acc = torch.where(empty, torch.full_like(acc, float("nan")), acc)
            return Column(bt.float64, acc)
        init = _I64_MAX if func == "min" else _I64_MIN
        acc, cnt = upd(f"{func}_i64", init_i=init, want_count=True)
        if nullable:
            return Column(bt.float64, torch.where(
                cnt == 0, torch.full((ngroups,), float("nan"),
                                     dtype=torch.float64, device=acc.device),
                acc.to(torch.float64)))
        out = acc
        if col.dtype.kind in (TypeKind.TIMESTAMP_NS,):
            return Column(bt.timestamp_ns, out)
        if col.dtype.kind == TypeKind.DATE32:
            return Column(bt.date32, out.to(torch.int32))
        if col.dtype.kind == TypeKind.BOOL:
            return Column(bt.boolean, out.to(torch.bool))
        return Column(col.dtype, out.to(col.data.dtype))
    if func == "mean":
        acc, cnt = upd("sum_f64", want_count=True)
        out = acc / cnt.to(torch.float64)
        return Column(bt.float64, out)
    if func in ("first", "last"):
        op = "first_row" if func == "first" else "last_row"
        init = _I64_MAX if func == "first" else _I64_MIN
        acc, cnt = upd(op, init_i=init, want_count=True)
        has = cnt > 0
        rows = torch.where(has, acc, uniq_rows[:ngroups] if uniq_rows.numel() >= ngroups else acc)
        got = _gather_any(col, rows.clamp(min=0))
        if col.mask is None and not is_float and (~has).any():
            got = Column(got.dtype, got.data, has.clone(), got.offsets,
                         got.dictionary, len(got))
        return got
    if func == "prod":
        if is_float:
            acc, cnt = upd("prod_f64", init_f=1.0, want_count=True)
            return Column(bt.float64, acc)
        acc, cnt = upd("prod_f64", init_f=1.0, want_count=True)
        return Column(bt.int64, acc.to(torch.int64))
    if func in ("var", "std"):
        acc, cnt = upd("sum_f64", want_count=True)
        # sum of squares via squared input
        sq = Column(bt.float64, col.data.to(torch.float64) ** 2, col.mask)
        col2 = sq
        data = col2.data
        acc2 = K.agg_update(data, None if col2.mask is None else col2.mask.view(torch.uint8),
                            None, int(TypeKind.FLOAT64), row_gid, ngroups,
                            _AGG_OP["sum_f64"], 0.0, 0, False)[0]
        c = cnt.to(torch.float64)
        var = (acc2 - acc * acc / c) / (c - 1)
        var = torch.where(cnt < 2, torch.full_like(var, float("nan")), var)
        return Column(bt.float64, var if func == "var" else var.sqrt())
    if func in ("any", "all"):
        nz = (col.data != 0).to(torch.int64)
        mask8 = None if col.mask is None else col.mask.view(torch.uint8)
        op = "max_i64" if func == "any" else "min_i64"
        init = _I64_MIN if func == "any" else _I64_MAX
        acc, cnt = K.agg_update(nz, mask8, None, int(TypeKind.INT64),
                                row_gid, ngroups, _AGG_OP[op], 0.0, init,
                                True)
        default = func == "all"  # empty/all-null group: any=False, all=True
        out = torch.where(cnt == 0,
                          torch.tensor(default, device=acc.device),
                          acc != 0 if func == "any" else acc != 0)
        return Column(bt.boolean, out.to(torch.bool))
    if func == "skew":
        x = col.data.to(torch.float64)
        mask8 = None if col.mask is None else col.mask.view(torch.uint8)
        s1, cnt = K.agg_update(x, mask8, None, int(TypeKind.FLOAT64),
                               row_gid, ngroups, _AGG_OP["sum_f64"], 0.0, 0,
                               True)
        s2 = K.agg_update(x * x, mask8, None, int(TypeKind.FLOAT64), row_gid,
                          ngroups, _AGG_OP["sum_f64"], 0.0, 0, False)[0]
        s3 = K.agg_update(x * x * x, mask8, None, int(TypeKind.FLOAT64),
                          row_gid, ngroups, _AGG_OP["sum_f64"], 0.0, 0,
                          False)[0]
        n = cnt.to(torch.float64)
        mean = s1 / n
        m2 = s2 / n - mean * mean
        m3 = s3 / n - 3 * mean * s2 / n + 2 * mean ** 3
        # pandas adjusted Fisher-Pearson
        g = torch.sqrt(n * (n - 1)) / (n - 2) * m3 / m2.clamp(min=0) ** 1.5
        g = torch.where(n < 3, torch.full_like(g, float("nan")), g)
        return Column(bt.float64, g)
    if func == "kurt":
        x = col.data.to(torch.float64)
        mask8 = None if col.mask is None else col.mask.view(torch.uint8)
        s1, cnt = K.agg_update(x, mask8, None, int(TypeKind.FLOAT64),
                               row_gid, ngroups, _AGG_OP["sum_f64"], 0.0, 0,
                               True)
        s2 = K.agg_update(x * x, mask8, None, int(TypeKind.FLOAT64), row_gid,
                          ngroups, _AGG_OP["sum_f64"], 0.0, 0, False)[0]
        s3 = K.agg_update(x * x * x, mask8, None, int(TypeKind.FLOAT64),
                          row_gid, ngroups, _AGG_OP["sum_f64"], 0.0, 0,
                          False)[0]
        s4 = K.agg_update(x * x * x * x, mask8, None, int(TypeKind.FLOAT64),
                          row_gid, ngroups, _AGG_OP["sum_f64"], 0.0, 0,
                          False)[0]
        n = cnt.to(torch.float64)
        mean = s1 / n
        m2 = s2 / n - mean ** 2
        m4 = (s4 - 4 * mean * s3 + 6 * mean ** 2 * s2) / n - 3 * mean ** 4
        # pandas adjusted kurtosis (Fisher, bias-corrected)
        g = (n - 1) / ((n - 2) * (n - 3)) * (
            (n + 1) * (n * m4 / m2.clamp(min=0) ** 2 - 3) + 6)
        g = torch.where(n < 4, torch.full_like(g, float("nan")), g)
        return Column(bt.float64, g)
    if func == "sem":
        acc, cnt = upd("sum_f64", want_count=True)
        sq = col.data.to(torch.float64) ** 2
        mask8 = None if col.mask is None else col.mask.view(torch.uint8)
        acc2 = K.agg_update(sq, mask8, None, int(TypeKind.FLOAT64), row_gid,
                            ngroups, _AGG_OP["sum_f64"], 0.0, 0, False)[0]
        c = cnt.to(torch.float64)
        var = (acc2 - acc * acc / c) / (c - 1)
        var = torch.where(cnt < 2, torch.full_like(var, float("nan")), var)
        return Column(bt.float64, (var / c).sqrt())
    if func == "median":
        return _median_by_group(col, row_gid, ngroups)
    if func == "nunique":
        return _nunique_by_group(col, row_gid, ngroups, tbl)
    raise NotImplementedError(f"gpu agg {func}")


def _median_by_group(col: Column, row_gid: torch.Tensor, ngroups: int) -> Column:
    # sort (gid, value) pairs; segmented median (device, torch-composable)
    vals = col.data.to(torch.float64)
    valid = ~torch.isnan(vals) if col.dtype.is_float else torch.ones_like(vals, dtype=torch.bool)
    if col.mask is not None:
        valid &= col.mask
    idx = torch.nonzero(valid, as_tuple=False).reshape(-1)
    v = vals[idx]
    g = row_gid[idx].to(torch.int64)
    order = torch.argsort(v, stable=True)
    g2, v2 = g[order], v[order]
    order2 = torch.argsort(g2, stable=True)
    gs, vs = g2[order2], v2[order2]
    cnt = torch.bincount(gs, minlength=ngroups)
    start = torch.zeros(ngroups, dtype=torch.int64, device=vals.device)
    torch.cumsum(cnt, 0, out=start)
    start = start - cnt
    mid = start + (cnt - 1) // 2
    mid2 = start + cnt // 2
    n_tot = int(vs.numel())
    safe_mid = mid.clamp(0, max(n_tot - 1, 0))
    safe_mid2 = mid2.clamp(0, max(n_tot - 1, 0))
    if n_tot == 0:
        return Column(bt.float64, torch.full((ngroups,), float("nan"),
                                             dtype=torch.float64, device=vals.device))
    med = (vs[safe_mid] + vs[safe_mid2]) / 2
    med = torch.where(cnt == 0, torch.full_like(med, float("nan")), med)
    return Column(bt.float64, med)


def _nunique_by_group(col: Column, row_gid: torch.Tensor, ngroups: int,
                      tbl: Table) -> Column:
    # distinct (gid, value) pairs via a second hash groupby
    gid_col = Column(bt.int32, row_gid)
    pair_gid, pair_rows = _groupby_pairs(gid_col, col)
    # count valid values per group among distinct pairs
    rows = pair_rows
    g = row_gid[rows].to(torch.int64)
    valid = torch.ones(int(rows.numel()), dtype=torch.bool, device=rows.device)
    if col.mask is not None:
        valid &= col.mask[rows]
    if col.dtype.is_float:
        valid &= ~torch.isnan(col.data[rows])
    cnt = torch.bincount(g[valid], minlength=ngroups)
    return Column(bt.int64, cnt)


def _groupby_pairs(gid_col: Column, col: Column):
    return groupby_build([gid_col, col])


# ----------------------------------------------------------------------
# join
# ----------------------------------------------------------------------

def join_local(left: Table, right: Table, left_on: Sequence[str],
               right_on: Sequence[str], how: str, suffixes=("_x", "_y")) -> Table:
    from . import take_table
    from .relational import _merge_joined, _null_out

    K = kernels()
    if how == "cross":
        nl, nr = len(left), len(right)
        li = torch.arange(nl, dtype=torch.int64, device=left.device).repeat_interleave(nr)
        ri = torch.arange(nr, dtype=torch.int64, device=left.device).repeat(nl)
        lt, rt = take_table(left, li), take_table(right, ri)
        return _merge_joined(lt, rt, [], [], suffixes, how)
    # build on the right side for inner/left/semi/anti; for right join swap
    swap = how == "right"
    if swap:
        left, right = right, left
        left_on, right_on = right_on, left_on
    build, probe = right, left
    bkeys = [build.column(k) for k in right_on]
    pkeys = [probe.column(k) for k in left_on]
    n_build, n_probe = len(build), len(probe)
    bh = hash_columns(bkeys) if n_build else torch.zeros(0, dtype=torch.int64, device=build.device)
    ph = hash_columns(pkeys) if n_probe else torch.zeros(0, dtype=torch.int64, device=build.device)
    heads, nxt = K.join_build(bh, n_build)
    how_id = {"inner": 0, "left": 1, "semi": 2, "anti": 3, "outer": 1,
              "right": 1}[how if not swap else "left"]
    bdatas, bmasks, boffs, bauxs, bdtypes = _col_args(bkeys)
    pdatas, pmasks, poffs, pauxs, pdtypes = _col_args(pkeys)
    track = how == "outer"
    res = K.join_probe(bdatas, bmasks, boffs, bauxs, bdtypes, n_build, bh,
                       pdatas, pmasks, poffs, pauxs, pdtypes, n_probe, ph,
                       heads, nxt, how_id, track)
    out_probe, out_build = res[0], res[1]
    if how == "semi":
        return take_table(probe, out_probe)
    if how == "anti":
        return take_table(probe, out_probe)
    if how == "outer":
        matched = res[2]
        unmatched = torch.nonzero(matched == 0, as_tuple=False).reshape(-1)
        out_probe = torch.cat([out_probe,
                               torch.full((int(unmatched.numel()),), -1,
                                          dtype=torch.int64, device=out_probe.device)])
        out_build = torch.cat([out_build, unmatched])
    # materialize (inner joins produce only valid pairs by construction;
    # skip the 1B-row .all() reductions/syncs there)
    if how == "inner":
        elide = False
        if int(out_probe.numel()) == n_probe and n_probe > 0:
            # identity iff strictly increasing (out_probe is sorted runs of
            # repeated probe ids; strict monotonic + length n => arange(n))
            elide = bool((out_probe[1:] > out_probe[:-1]).all().item())
        if elide:
            pt = probe  # every probe row matched exactly once: skip gather
        else:
            pt = take_table(probe, out_probe)
        btb = take_table(build, out_build)
    else:
        p_valid = out_probe >= 0
        b_valid = out_build >= 0
        pt = take_table(probe, out_probe.clamp(min=0))
        btb = take_table(build, out_build.clamp(min=0))
        if not bool(p_valid.all().item()):
            pt = _null_out(pt, p_valid)
        if not bool(b_valid.all().item()):
            btb = _null_out(btb, b_valid)
    if swap:
        # right join: probe side was the right table
        return _merge_joined(btb, pt, list(right_on), list(left_on), suffixes, how)
    return _merge_joined(pt, btb, list(left_on), list(right_on), suffixes, how)


def distinct_local(tbl: Table, subset=None, keep: str = "first") -> Table:
    from . import take_table

    keys = list(subset) if subset else list(tbl.names)
    key_cols = [tbl.column(k) for k in keys]
    if len(tbl) == 0:
        return tbl
    row_gid, uniq_rows = groupby_build(key_cols)
    if keep == "first":
        # uniq_rows is the first CLAIMER (race order); recompute true min row
        K = kernels()
        acc = K.agg_update(
            torch.zeros(len(tbl), dtype=torch.int8, device=tbl.device), None,
            None, int(TypeKind.INT8), row_gid, int(uniq_rows.numel()),
            _AGG_OP["first_row"], 0.0, _I64_MAX, False)[0]
        rows = acc
    elif keep == "last":
        K = kernels()
        acc = K.agg_update(
            torch.zeros(len(tbl), dtype=torch.int8, device=tbl.device), None,
            None, int(TypeKind.INT8), row_gid, int(uniq_rows.numel()),
            _AGG_OP["last_row"], 0.0, _I64_MIN, False)[0]
        rows = acc
    elif keep is False:
        # drop every member of any duplicated group
        K = kernels()
        cnt = K.agg_update(
            torch.zeros(len(tbl), dtype=torch.int8, device=tbl.device), None,
            None, int(TypeKind.INT8), row_gid, int(uniq_rows.numel()),
            _AGG_OP["size"], 0.0, 0, True)[1]
        singles = cnt[row_gid.long()] == 1
        rows = torch.nonzero(singles, as_tuple=False).reshape(-1)
    else:
        rows = uniq_rows
    rows = torch.sort(rows).values  # preserve original row order
    return take_table(tbl, rows)


def _empty_gb_result(tbl: Table, keys, aggs) -> Table:
    from ..core.table import Table as T

    base = T.empty_like(tbl.select([k for k in keys]))
    names = list(keys)
    cols = list(base.columns)
    for out_name, in_name, func in aggs:
        dt = bt.int64 if func in ("count", "size", "nunique") else bt.float64
        cols.append(Column(dt, torch.zeros(0, dtype=bt.torch_storage_dtype(dt),
                                           device=tbl.device), length=0))
        names.append(out_name)
    return T(names, cols, 0)


def take_table_fused(tbl: Table, idx: torch.Tensor) -> Optional[Table]:
    """All fixed-width columns (+masks) of a take materialize in ONE kernel
    launch (join-heavy queries were launch-bound on per-column
    index_select; reference role: cudf::gather single pass)."""
    from . import gather as _gather

    K = kernels()
    srcs = []
    plan = []  # (col_index, has_mask)
    for i, c in enumerate(tbl.columns):
        if c.dtype.kind == TypeKind.STRING:
            plan.append((i, None))
            continue
        srcs.append(c.data)
        if c.mask is not None:
            srcs.append(c.mask)
            plan.append((i, True))
        else:
            plan.append((i, False))
    if len(srcs) < 2:
        return None
    outs = K.gather_multi(srcs, idx)
    n = int(idx.numel())
    cols = []
    k = 0
    for i, has_mask in plan:
        c = tbl.columns[i]
        if has_mask is None:
            cols.append(_gather(c, idx))
            continue
        data = outs[k]
        k += 1
        mask = None
        if has_mask:
            mask = outs[k]
            k += 1
        out = Column(c.dtype, data, mask, dictionary=c.dictionary, length=n)
        out.val_range = c.val_range
        cols.append(out)
    return Table(tbl.names, cols, n)
