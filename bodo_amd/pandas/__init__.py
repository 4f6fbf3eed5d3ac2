"""``bodo_amd.pandas``: lazy drop-in pandas replacement (reference:
bodo/pandas/__init__.py + base.py).  Unimplemented attributes fall back to
real pandas with a warning (BODO_AMD_PANDAS_FALLBACK)."""

from __future__ import annotations

import warnings

import pandas as _pd

from .. import config
from ..engine import executor as _ex
from ..io import csv as _csv
from ..io import parquet as _pq
from ..plan import nodes as _pn
from .frame import BodoDataFrame, from_pandas_df
from .series import BodoSeries

# pandas passthroughs commonly used together with the frame API
Timestamp = _pd.Timestamp
Timedelta = _pd.Timedelta
NamedAgg = _pd.NamedAgg
NA = _pd.NA
NaT = _pd.NaT
isna = _pd.isna
notna = _pd.notna
Series = BodoSeries


def DataFrame(data=None, *args, **kwargs):
    """Constructor-compatible factory: builds a BodoDataFrame from pandas
    DataFrame constructor arguments (reference: bodo.pandas.DataFrame)."""
    if isinstance(data, BodoDataFrame):
        return data.copy()
    return from_pandas_df(_pd.DataFrame(data, *args, **kwargs))


def read_parquet(path, columns=None, filters=None,
                 **kwargs) -> BodoDataFrame:
    """filters accepts the pandas/pyarrow DNF-lite list form
    [(col, op, value), ...] (ANDed) and pushes into the scan."""
    from ..plan.expr import Cmp, ColRef, Const, IsIn

    names = _pq.schema_names(path)
    push = []
    for f in (filters or []):
        col, op, val = f
        opmap = {"<": "lt", "<=": "le", ">": "gt", ">=": "ge", "==": "eq",
                 "=": "eq", "!=": "ne"}
        if op in opmap:
            push.append(Cmp(opmap[op], ColRef(col), Const(val)))
        elif op == "in":
            push.append(IsIn(ColRef(col), tuple(val)))
        else:
            raise NotImplementedError(f"read_parquet filter op {op!r}")
    plan = _pn.ParquetScan(str(path), tuple(columns) if columns else None,
                           tuple(push), tuple(names))
    return BodoDataFrame(plan, list(columns) if columns else list(names))


def read_iceberg(table_uri, snapshot_id=None, columns=None,
                 **kwargs) -> BodoDataFrame:
    """Read a filesystem Iceberg table, optionally time-travelling to an
    older snapshot (reference: bodo/pandas/base.py read_iceberg:313)."""
    from ..io import iceberg as _ib

    path = str(table_uri)
    if snapshot_id is not None:
        path = f"{path}@{_ib.SNAP_PREFIX}{int(snapshot_id)}"
    names = _pq.schema_names(path)
    plan = _pn.ParquetScan(path, tuple(columns) if columns else None,
                           (), tuple(names))
    return BodoDataFrame(plan, list(columns) if columns else list(names))


def read_sql(sql, con, **kwargs) -> BodoDataFrame:
    """Database read through any DB-API connection or a sqlite path/URI
    (reference: bodo/ir/sql_ext.py SqlReader; the Snowflake fast path is a
    connector away — no network in this build).  The result distributes by
    block-slicing rows across ranks."""
    import sqlite3

    close = None
    if isinstance(con, str):
        path = con[len("sqlite://"):] if con.startswith("sqlite://") else con
        con = sqlite3.connect(path)
        close = con
    try:
        df = _pd.read_sql_query(sql, con, **kwargs) \
            if hasattr(_pd, "read_sql_query") else _pd.read_sql(sql, con)
    finally:
        if close is not None:
            close.close()
    return from_pandas(df)


def read_csv(path, **kwargs) -> BodoDataFrame:
    options = tuple(sorted(kwargs.items(), key=lambda kv: kv[0]))
    names = _csv.schema_names(str(path), dict(options))
    plan = _pn.CsvScan(str(path), options, None, tuple(names))
    return BodoDataFrame(plan, list(names))


def read_json(path, orient="records", lines=True, **kwargs) -> BodoDataFrame:
    """JSON reader (reference: bodo/io json reader).  Host parse with rank
    block-slicing, same model as read_csv."""
    df = _pd.read_json(path, orient=orient, lines=lines, **kwargs)
    return from_pandas_df(df)


def from_pandas(df: _pd.DataFrame) -> BodoDataFrame:
    return from_pandas_df(df)


def merge(left, right, **kwargs) -> BodoDataFrame:
    if not isinstance(left, BodoDataFrame):
        left = from_pandas_df(left)
    return left.merge(right, **kwargs)


def concat(objs, axis=0, ignore_index=False, **kwargs):
    objs = list(objs)
    if axis in (0, "index") and all(isinstance(o, BodoDataFrame) for o in objs) \
            and all(set(o._columns) == set(objs[0]._columns) for o in objs):
        plans = tuple(o._lazy_plan for o in objs)
        return BodoDataFrame(_pn.Union(plans), objs[0]._columns)
    warnings.warn("concat: falling back to pandas")
    mats = [o.to_pandas() if isinstance(o, (BodoDataFrame, BodoSeries)) else o
            for o in objs]
    return from_pandas_df(_pd.concat(mats, axis=axis, ignore_index=ignore_index,
                                     **kwargs))


def get_dummies(data, prefix=None, columns=None, dtype=None,
                **kwargs) -> BodoDataFrame:
    """One-hot encode (reference: pd_dataframe_ext get_dummies overload):
    category values come from a distributed unique() pass, then each output
    column is one lazy equality projection — the frame never leaves the
    device."""
    from ..plan.expr import Cmp, ColRef, Const
    from .series import BodoSeries

    if isinstance(data, BodoSeries):
        vals = [v for v in data.unique().tolist() if v is not None
                and v == v]
        vals = sorted(map(str, vals))
        names = [f"{prefix}_{v}" if prefix else str(v) for v in vals]
        exprs = tuple(Cmp("eq", data._expr, Const(v)) for v in vals)
        plan = _pn.Projection(data._plan, tuple(names), exprs)
        return BodoDataFrame(plan, list(names))
    if isinstance(data, BodoDataFrame):
        cats = list(columns) if columns is not None else [
            c for c in data._columns
            if not _pd.api.types.is_numeric_dtype(data.head(1)[c].dtype)]
        names, exprs = [], []
        for c in data._columns:
            if c not in cats:
                names.append(c)
                exprs.append(ColRef(c))
        for c in cats:
            ser = data[c]
            vals = sorted(str(v) for v in ser.unique().tolist()
                          if v is not None and v == v)
            for v in vals:
                names.append(f"{c}_{v}")
                exprs.append(Cmp("eq", ser._expr, Const(v)))
        plan = _pn.Projection(data._plan, tuple(names), tuple(exprs))
        return BodoDataFrame(plan, list(names))
    return _pd.get_dummies(data, prefix=prefix, columns=columns,
                           dtype=dtype, **kwargs)


def cut(x, bins, labels=None, right=True, **kwargs):
    """Binning.  labels=False stays lazy (bin indices from a comparison
    chain); labelled output falls back to pandas (interval categoricals)."""
    from ..plan.expr import Case, Cmp, Const
    from .series import BodoSeries

    if isinstance(x, BodoSeries) and labels is False \
            and not isinstance(bins, int):
        edges = list(bins)
        op = "le" if right else "lt"
        conds, thens = [], []
        # below the first edge (or on it for right=False) -> NaN
        conds.append(Cmp("le" if right else "lt", x._expr, Const(edges[0])))
        thens.append(Const(None, None))
        for i in range(1, len(edges)):
            conds.append(Cmp(op, x._expr, Const(edges[i])))
            thens.append(Const(i - 1))
        return x._wrap(Case(tuple(conds), tuple(thens), Const(None, None)),
                       None)
    if isinstance(x, BodoSeries):
        return _pd.cut(x.to_pandas(), bins, labels=labels, right=right,
                       **kwargs)
    return _pd.cut(x, bins, labels=labels, right=right, **kwargs)


def qcut(x, q, labels=None, **kwargs):
    """Quantile binning: edges from the exact distributed quantile pass,
    then cut()."""
    from .series import BodoSeries

    if isinstance(x, BodoSeries):
        n = q if isinstance(q, int) else len(q) - 1
        probs = [i / n for i in range(n + 1)] if isinstance(q, int) else list(q)
        edges = [x.quantile(p) for p in probs]
        edges[0] = edges[0] - 1e-9  # include the minimum (pandas semantics)
        if labels is False:
            return cut(x, edges, labels=False)
        return _pd.qcut(x.to_pandas(), q, labels=labels, **kwargs)
    return _pd.qcut(x, q, labels=labels, **kwargs)


def crosstab(index, columns, values=None, aggfunc=None,
             rownames=None, colnames=None):
    """Distributed crosstab: grouped count (or aggfunc) on the pair of
    series, host pivot of the small result (reference role:
    pd.crosstab via BodoDataFrame.pivot_table)."""
    from .series import BodoSeries

    if not isinstance(index, BodoSeries) and not isinstance(columns,
                                                            BodoSeries):
        return _pd.crosstab(index, columns, values=values, aggfunc=aggfunc,
                            rownames=rownames, colnames=colnames)
    rname = (rownames[0] if rownames else
             getattr(index, "name", None) or "row_0")
    cname = (colnames[0] if colnames else
             getattr(columns, "name", None) or "col_0")
    if cname == rname:
        cname = f"{cname}_col"
    df = index.to_frame(rname)
    df[cname] = columns
    if values is None:
        small = df.groupby([rname, cname], as_index=False).agg(
            __n=_pd.NamedAgg(rname, "size"))
        pdf = small.to_pandas()
        for c in pdf.columns:
            if isinstance(pdf[c].dtype, _pd.CategoricalDtype):
                pdf[c] = pdf[c].astype(object)
        out = pdf.pivot_table(values="__n", index=rname, columns=cname,
                              aggfunc="first", fill_value=0)
        return out.astype("int64").sort_index()
    df["__v"] = values
    small = df.groupby([rname, cname], as_index=False).agg(
        __v=_pd.NamedAgg("__v", aggfunc or "mean"))
    pdf = small.to_pandas()
    for c in pdf.columns:
        if isinstance(pdf[c].dtype, _pd.CategoricalDtype):
            pdf[c] = pdf[c].astype(object)
    return pdf.pivot_table(values="__v", index=rname, columns=cname,
                           aggfunc="first").sort_index()


def to_datetime(arg, format=None, errors=None, **kwargs):
    if isinstance(arg, BodoSeries):
        from ..core import types as bt
        from ..plan.expr import Cast, StrOp

        if format is not None or errors is not None:
            return arg._wrap(StrOp(arg._expr, "to_datetime",
                                   (format, errors)), arg.name)
        return arg._wrap(Cast(arg._expr, bt.timestamp_ns), arg.name)
    if format is not None:
        kwargs["format"] = format
    if errors is not None:
        kwargs["errors"] = errors
    return _pd.to_datetime(arg, **kwargs)


def merge_asof(left, right, on=None, by=None, direction="backward",
               suffixes=("_x", "_y")):
    """Distributed merge_asof: the (small) right side replicates to every
    rank; each locally sorted left shard then matches exactly with pandas
    merge_asof (reference: pd.merge_asof via bodo's sorted-merge path)."""
    from .frame import BodoDataFrame
    from ..plan import nodes as pn

    if not isinstance(left, BodoDataFrame):
        return _pd.merge_asof(left, right, on=on, by=by,
                              direction=direction, suffixes=suffixes)
    r_pd = right.to_pandas() if isinstance(right, BodoDataFrame) else right
    r_pd = r_pd.sort_values(on).reset_index(drop=True)
    sorted_left = left.sort_values(on)

    def _part(pdf, _r=r_pd):
        if len(pdf) == 0:
            out = _pd.merge_asof(pdf, _r, on=on, by=by,
                                 direction=direction, suffixes=suffixes)
            return out
        return _pd.merge_asof(pdf, _r, on=on, by=by, direction=direction,
                              suffixes=suffixes)

    probe = _part(left.head(0).to_pandas())
    names = list(probe.columns)
    plan = pn.MapPartitions(sorted_left._lazy_plan, _part, (), tuple(names))
    return BodoDataFrame(plan, names)


def __getattr__(name):
    if hasattr(_pd, name):
        if config.PANDAS_FALLBACK:
            return getattr(_pd, name)
    raise AttributeError(f"module 'bodo_amd.pandas' has no attribute {name!r}")
