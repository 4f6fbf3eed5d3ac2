"""Local (single-rank) relational kernels: hash groupby, hash join, distinct.

CPU: pandas-backed (defines the semantics; the engine's test contract is
"matches pandas").  GPU: hand-written gfx950 HIP kernels via csrc/ — see
bodo_amd/ops/gpu.py.  Reference roles: bodo/libs/streaming/_groupby.cpp,
_join.cpp, groupby/ kernels.
"""

from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import numpy as np
import pandas as pd
import torch

from ..core import types as bt
from ..core.column import Column
from ..core.table import Table
from ..core.types import TypeKind

# physical agg funcs every backend implements
PHYSICAL_AGGS = {
    "sum", "count", "min", "max", "size", "first", "last", "prod",
    # single-phase only (require co-located full groups):
    "median", "nunique",
}

# funcs with no device kernel yet: run in host pandas even on GPU shards
HOST_ONLY_AGGS = {"mode", "array_agg"}


def _normalize_decimal_aggs(tbl: Table, aggs):
    """Aggregate decimal columns exactly on their scaled-int64 storage
    (sum/min/max/first/last/nunique) or on a float64 view (mean/var/...),
    then re-wrap exact results as decimal.  Reference role:
    bodo/libs/_decimal_ext.cpp decimal agg combine."""
    DEC = TypeKind.DECIMAL128
    if not any(a[1] and tbl.has_column(a[1])
               and tbl.column(a[1]).dtype.kind == DEC for a in aggs):
        return tbl, aggs, []
    new_cols = {}
    out_fix = []
    aggs2 = []
    for out, inn, func in aggs:
        col = tbl.column(inn) if inn and tbl.has_column(inn) else None
        if col is None or col.dtype.kind != DEC or callable(func):
            aggs2.append((out, inn, func))
            continue
        if func in ("sum", "min", "max", "first", "last", "prod"):
            nm = f"__deci_{inn}"
            if nm not in new_cols:
                new_cols[nm] = Column(bt.int64, col.data, col.mask,
                                      length=len(col))
            aggs2.append((out, nm, func))
            if func != "prod":
                out_fix.append((out, col.dtype))
        elif func in ("count", "size", "nunique", "any", "all"):
            nm = f"__deci_{inn}"
            if nm not in new_cols:
                new_cols[nm] = Column(bt.int64, col.data, col.mask,
                                      length=len(col))
            aggs2.append((out, nm, func))
        else:  # mean/var/std/median/skew/quantile: float64 view
            from .evaluate import decimal_to_float

            nm = f"__decf_{inn}"
            if nm not in new_cols:
                new_cols[nm] = decimal_to_float(col)
            aggs2.append((out, nm, func))
    work = tbl
    for nm, c in new_cols.items():
        work = work.with_column(nm, c)
    return work, aggs2, out_fix


def groupby_local(tbl: Table, keys: Sequence[str],
                  aggs: Sequence[Tuple[str, str, str]],
                  dropna: bool = True) -> Table:
    """Group rows of the local shard; aggs = (out_name, in_name, func)."""
    # grouped approx_nunique runs single-phase on co-located groups, where
    # the exact per-group count is the best possible approximation
    aggs = [(o, i, "nunique") if f == "approx_nunique" else (o, i, f)
            for o, i, f in aggs]
    tbl, aggs, dec_fix = _normalize_decimal_aggs(tbl, aggs)
    if tbl.device.type == "cuda":
        if any(callable(a[2]) or a[2] in HOST_ONLY_AGGS for a in aggs):
            # custom python agg: host pandas per co-located shard (the
            # @jit-to-HIP lowering is the native path for these)
            out = _groupby_pandas(tbl, keys, aggs, dropna)
        else:
            from . import gpu

            out = gpu.groupby_local(tbl, keys, aggs, dropna)
    else:
        out = _groupby_pandas(tbl, keys, aggs, dropna)
    for out_name, dec_dtype in dec_fix:
        c = out.column(out_name)
        if c.data.dtype == torch.int64:
            out = out.with_column(out_name,
                                  Column(dec_dtype, c.data, c.mask,
                                         length=len(c)))
        elif c.data.dtype.is_floating_point:
            # backend widened for empty-group NaN: value = scaled/10^s
            out = out.with_column(out_name, Column(
                bt.float64, c.data / float(10 ** dec_dtype.scale), c.mask,
                length=len(c)))
    return out


def _groupby_pandas(tbl: Table, keys, aggs, dropna) -> Table:
    need = list(dict.fromkeys(list(keys) + [a[1] for a in aggs if a[1]]))
    df = tbl.select([c for c in need if tbl.has_column(c)]).to_pandas()
    # dict-encoded inputs arrive as unordered Categorical: value aggs
    # (min/max/...) must see the string VALUES, so decode agg inputs (a
    # column can be key AND agg input at once; object-key groupby stays
    # correct, and keys are re-dict-encoded on the way out)
    agg_ins = {a[1] for a in aggs if a[1]}
    for c in agg_ins:
        if c in df.columns and isinstance(df[c].dtype, pd.CategoricalDtype):
            df[c] = df[c].astype(object)
    gb = df.groupby(list(keys), dropna=dropna, sort=False, observed=True)
    named = {}
    for out_name, in_name, func in aggs:
        if func == "approx_nunique":  # host: exact nunique qualifies
            named[out_name] = pd.NamedAgg(column=in_name, aggfunc="nunique")
        elif func == "mode":
            named[out_name] = pd.NamedAgg(
                column=in_name,
                aggfunc=lambda s: s.mode().iloc[0] if len(s.mode()) else None)
        elif func == "kurt":
            named[out_name] = pd.NamedAgg(column=in_name,
                                          aggfunc=lambda s: s.kurt())
        elif func == "array_agg":
            named[out_name] = pd.NamedAgg(
                column=in_name,
                aggfunc=lambda s: list(s.dropna()))
        elif func == "size":
            named[out_name] = pd.NamedAgg(column=df.columns[0] if not in_name or in_name not in df.columns else in_name, aggfunc="size")
        else:
            named[out_name] = pd.NamedAgg(column=in_name, aggfunc=func)
    res = gb.agg(**named).reset_index()
    # SPMD type stability: an empty shard must produce the same column kinds
    # as non-empty ranks (pandas infers `null` for empty object columns and
    # the shuffle collective sequences would then diverge) — enforce the
    # input column's type for keys and string-typed agg outputs.
    import pyarrow as pa

    cols, names = [], []
    for k in keys:
        cols.append(_col_from_pandas_typed(res[k], tbl.column(k), tbl.device))
        names.append(k)
    for out_name, in_name, func in aggs:
        src = tbl.column(in_name) if in_name and tbl.has_column(in_name) else None
        if (src is not None and src.dtype.is_string_like
                and func in ("min", "max", "first", "last")):
            cols.append(_col_from_pandas_typed(res[out_name], src, tbl.device))
        else:
            arr = pa.Array.from_pandas(res[out_name])
            if pa.types.is_null(arr.type):
                arr = arr.cast(pa.float64())
            cols.append(Column.from_arrow(arr, tbl.device))
        names.append(out_name)
    return Table(names, cols, len(res))


def _col_from_pandas_typed(ser: pd.Series, like: Column, device) -> Column:
    """Convert a pandas Series enforcing the arrow type of `like`."""
    import pyarrow as pa

    k = like.dtype.kind
    if k == TypeKind.DICT:
        arr = pa.Array.from_pandas(ser)
        if not pa.types.is_dictionary(arr.type):
            arr = arr.cast(pa.large_string()).dictionary_encode()
        return Column.from_arrow(arr, device)
    target = _PA_TYPE_OF.get(k)
    arr = pa.Array.from_pandas(ser, type=target)
    return Column.from_arrow(arr, device)


_PA_TYPE_OF = {}


def _init_pa_types():
    import pyarrow as pa

    global _PA_TYPE_OF
    _PA_TYPE_OF = {
        TypeKind.INT8: pa.int8(), TypeKind.INT16: pa.int16(),
        TypeKind.INT32: pa.int32(), TypeKind.INT64: pa.int64(),
        TypeKind.UINT8: pa.uint8(), TypeKind.UINT16: pa.uint16(),
        TypeKind.UINT32: pa.uint32(), TypeKind.UINT64: pa.uint64(),
        TypeKind.FLOAT32: pa.float32(), TypeKind.FLOAT64: pa.float64(),
        TypeKind.BOOL: pa.bool_(), TypeKind.DATE32: pa.date32(),
        TypeKind.TIMESTAMP_NS: pa.timestamp("ns"),
        TypeKind.DURATION_NS: pa.duration("ns"),
        TypeKind.STRING: pa.large_string(),
    }


_init_pa_types()


# combiner for two-phase aggregation: how to merge partial results
COMBINE_FUNC = {
    "sum": "sum", "count": "sum", "size": "sum", "min": "min", "max": "max",
    "first": "first", "last": "last", "prod": "prod",
    "any": "any", "all": "all",
}


def join_local(left: Table, right: Table, left_on: Sequence[str],
               right_on: Sequence[str], how: str,
               suffixes=("_x", "_y")) -> Table:
    if left.device.type == "cuda":
        from . import gpu

        return gpu.join_local(left, right, left_on, right_on, how, suffixes)
    return _join_pandas(left, right, left_on, right_on, how, suffixes)


def _join_pandas(left: Table, right: Table, left_on, right_on, how, suffixes) -> Table:
    from . import take_table

    if how == "cross":
        nl, nr = len(left), len(right)
        li = torch.arange(nl, dtype=torch.int64).repeat_interleave(nr)
        ri = torch.arange(nr, dtype=torch.int64).repeat(nl)
        lt, rt = take_table(left, li), take_table(right, ri)
        return _merge_joined(lt, rt, [], [], suffixes, how)
    ldf = left.select(list(left_on)).to_pandas()
    rdf = right.select(list(right_on)).to_pandas()
    ldf.columns = [f"k{i}" for i in range(len(left_on))]
    rdf.columns = [f"k{i}" for i in range(len(right_on))]
    ldf["__li"] = np.arange(len(ldf), dtype=np.int64)
    rdf["__ri"] = np.arange(len(rdf), dtype=np.int64)
    pd_how = {"semi": "inner", "anti": "left"}.get(how, how)
    m = ldf.merge(rdf, on=[f"k{i}" for i in range(len(left_on))], how=pd_how)
    if how == "semi":
        li = np.unique(m["__li"].to_numpy())
        return take_table(left, torch.from_numpy(li).to(left.device))
    if how == "anti":
        miss = m[m["__ri"].isna()]["__li"].to_numpy().astype(np.int64)
        return take_table(left, torch.from_numpy(np.unique(miss)).to(left.device))
    li = m["__li"].to_numpy()
    ri = m["__ri"].to_numpy()
    return _materialize_join(left, right, li, ri, left_on, right_on, how, suffixes)


def _materialize_join(left: Table, right: Table, li: np.ndarray, ri: np.ndarray,
                      left_on, right_on, how, suffixes) -> Table:
    from . import gather, take_table

    dev = left.device
    l_valid = ~pd.isna(li)
    r_valid = ~pd.isna(ri)
    li_t = torch.from_numpy(np.where(l_valid, li, 0).astype(np.int64)).to(dev)
    ri_t = torch.from_numpy(np.where(r_valid, ri, 0).astype(np.int64)).to(dev)
    l_valid_t = torch.from_numpy(l_valid.astype(bool)).to(dev)
    r_valid_t = torch.from_numpy(r_valid.astype(bool)).to(dev)
    lt = take_table(left, li_t)
    rt = take_table(right, ri_t)
    if not bool(l_valid.all()):
        lt = _null_out(lt, l_valid_t)
    if not bool(r_valid.all()):
        rt = _null_out(rt, r_valid_t)
    out = _merge_joined(lt, rt, left_on, right_on, suffixes, how)
    # outer join: coalesce key columns
    if how in ("outer", "right", "left"):
        pass
    return out


def _null_out(tbl: Table, valid: torch.Tensor) -> Table:
    cols = []
    for c in tbl.columns:
        mask = valid.clone() if c.mask is None else (c.mask & valid)
        if c.dtype.is_float:
            # fold BOTH the unmatched-row mask and the column's own null
            # mask into NaN (dropping c.mask here turned nulls into storage
            # fill values)
            data = c.data.clone()
            data[~mask] = float("nan")
            cols.append(Column(c.dtype, data, None, c.offsets, c.dictionary, len(c)))
        else:
            cols.append(Column(c.dtype, c.data, mask, c.offsets, c.dictionary, len(c)))
    return Table(tbl.names, cols, len(tbl))


def _merge_joined(lt: Table, rt: Table, left_on, right_on, suffixes, how) -> Table:
    """Column naming like pandas merge: shared key names appear once (from
    left, coalesced for outer); clashing non-key names get suffixes."""
    names, cols = [], []
    shared_keys = [k for k, rk in zip(left_on, right_on) if k == rk]
    lnames = set(lt.names)
    rnames = set(rt.names)
    for n, c in zip(lt.names, lt.columns):
        if n in rnames and n not in shared_keys:
            names.append(n + suffixes[0])
        else:
            names.append(n)
        cols.append(c)
    for n, c in zip(rt.names, rt.columns):
        if n in shared_keys:
            if how in ("outer", "right"):
                # coalesce into the left key column
                i = names.index(n)
                cols[i] = _coalesce(cols[i], c)
            continue
        if n in lnames:
            names.append(n + suffixes[1])
        else:
            names.append(n)
        cols.append(c)
    return Table(names, cols, len(lt))


def _coalesce(a: Column, b: Column) -> Column:
    """Outer-join key coalescing: a where a is non-null, else b.  Null =
    mask OR NaN (masked storage holds arbitrary fill, never trust it); a
    mixed int/float key pair coalesces in float64 like pandas merge."""
    if a.dtype.kind in (TypeKind.STRING, TypeKind.DICT) \
            or b.dtype.kind in (TypeKind.STRING, TypeKind.DICT):
        # host path (decode Categorical: the two dictionaries differ).
        # `a` can be a null-typed/empty-shard schema degenerate (float64
        # all-null) while `b` is the real string column — never fall into
        # the numeric branch then, it would emit dictionary CODES
        sa, sb = a.to_pandas(), b.to_pandas()
        if isinstance(sa.dtype, pd.CategoricalDtype):
            sa = sa.astype(object)
        if isinstance(sb.dtype, pd.CategoricalDtype):
            sb = sb.astype(object)
        out = sa.where(~pd.isna(sa), sb)
        import pyarrow as pa

        return Column.from_arrow(pa.Array.from_pandas(out), a.device)

    def nulls(c: Column) -> torch.Tensor:
        nn = torch.zeros(len(c), dtype=torch.bool, device=c.data.device)
        if c.mask is not None:
            nn |= ~c.mask
        if c.dtype.is_float:
            nn |= torch.isnan(c.data)
        return nn

    a_null = nulls(a)
    if not bool(a_null.any()):
        return Column(a.dtype, a.data, None, length=len(a))
    b_null = nulls(b)
    both = a_null & b_null
    if a.dtype.is_float or b.dtype.is_float:
        ad = a.data.to(torch.float64)
        data = torch.where(a_null, b.data.to(torch.float64), ad)
        data = torch.where(both, torch.full((), float("nan"),
                                            dtype=torch.float64,
                                            device=data.device), data)
        out_dt = a.dtype if a.dtype.is_float else bt.float64
        if out_dt.kind == TypeKind.FLOAT32:
            data = data.to(torch.float32)
        return Column(out_dt, data, None, length=len(a))
    # int-like (incl. timestamps): keep exact storage, mask the both-null rows
    data = torch.where(a_null, b.data.to(a.data.dtype), a.data)
    mask = ~both if bool(both.any()) else None
    return Column(a.dtype, data, mask, length=len(a))


def distinct_local(tbl: Table, subset: Optional[Sequence[str]] = None,
                   keep: str = "first") -> Table:
    if tbl.device.type == "cuda":
        from . import gpu

        return gpu.distinct_local(tbl, subset, keep)
    df = tbl.to_pandas()
    res = df.drop_duplicates(subset=list(subset) if subset else None, keep=keep)
    return Table.from_pandas(res.reset_index(drop=True), tbl.device)
