"""Lazy groupby (reference: bodo/pandas/groupby.py)."""

from __future__ import annotations

from typing import Dict, List, Optional

import pandas as pd

from ..plan import nodes as pn


class DataFrameGroupBy:
    def __init__(self, frame, keys: List[str], as_index=True, dropna=True,
                 sort=False, selection: Optional[List[str]] = None):
        self._frame = frame
        self._keys = keys
        self._as_index = as_index
        self._dropna = dropna
        self._sort = sort
        self._selection = selection

    def __getitem__(self, key):
        if isinstance(key, str):
            return SeriesGroupBy(self._frame, self._keys, key,
                                 self._as_index, self._dropna, self._sort)
        return DataFrameGroupBy(self._frame, self._keys, self._as_index,
                                self._dropna, self._sort, list(key))

    # ------------------------------------------------------------------
    def _value_columns(self):
        if self._selection is not None:
            return self._selection
        return [c for c in self._frame._columns if c not in self._keys]

    def _build(self, aggs):
        plan = pn.Aggregate(self._frame._lazy_plan, tuple(self._keys),
                            tuple(aggs), self._as_index, self._dropna)
        if self._sort:
            plan = pn.Sort(plan, tuple(self._keys),
                           tuple([True] * len(self._keys)))
        from .frame import BodoDataFrame

        cols = list(self._keys) + [a[0] for a in aggs]
        out = BodoDataFrame(plan, cols)
        if self._as_index:
            return _IndexedAggResult(out, self._keys)
        return out

    def agg(self, arg=None, **kwargs):
        aggs = []
        if arg is None and kwargs:
            # named aggregation: out=NamedAgg(column=..., aggfunc=...) or tuples
            for out_name, spec in kwargs.items():
                if isinstance(spec, pd.NamedAgg) or (isinstance(spec, tuple) and len(spec) == 2):
                    col, func = spec
                else:
                    raise TypeError(f"bad agg spec {spec}")
                aggs.append((out_name, col, _norm_func(func)))
        elif isinstance(arg, dict):
            for col, func in arg.items():
                if isinstance(func, (list, tuple)):
                    for f in func:
                        aggs.append((f"{col}_{f}", col, _norm_func(f)))
                else:
                    aggs.append((col, col, _norm_func(func)))
        elif isinstance(arg, str):
            for col in self._value_columns():
                aggs.append((col, col, _norm_func(arg)))
        elif isinstance(arg, (list, tuple)):
            for col in self._value_columns():
                for f in arg:
                    aggs.append((f"{col}_{f}", col, _norm_func(f)))
        elif callable(arg):
            for col in self._value_columns():
                aggs.append((col, col, arg))
        else:
            raise NotImplementedError(f"agg({arg!r})")
        return self._build(aggs)

    aggregate = agg

    def _simple(self, func):
        aggs = [(c, c, func) for c in self._value_columns()]
        return self._build(aggs)

    def sum(self, numeric_only=False):
        return self._simple("sum")

    def count(self):
        return self._simple("count")

    def mean(self, numeric_only=False):
        return self._simple("mean")

    def min(self, numeric_only=False):
        return self._simple("min")

    def max(self, numeric_only=False):
        return self._simple("max")

    def median(self, numeric_only=False):
        return self._simple("median")

    def first(self):
        return self._simple("first")

    def last(self):
        return self._simple("last")

    def nunique(self):
        return self._simple("nunique")

    def var(self, ddof=1):
        return self._simple("var")

    def std(self, ddof=1):
        return self._simple("std")

    def any(self):
        return self._simple("any")

    def all(self):
        return self._simple("all")

    def skew(self, numeric_only=False):
        return self._simple("skew")

    def size(self):
        out = self._build([("size", "", "size")])
        if isinstance(out, _IndexedAggResult):
            ser = _decat(out._frame.to_pandas(), self._keys).set_index(self._keys)["size"]
            return ser
        return out

    def _rn_filter(self, cond_builder):
        """Shared row_number/size window + filter + column restore."""
        from .frame import BodoDataFrame
        from ..plan.expr import ColRef

        w = pn.Window(self._frame._lazy_plan, tuple(self._keys), (), (),
                      (("__rn", "", "row_number", None),
                       ("__sz", "", "transform_size", None)))
        filt = pn.Filter(w, cond_builder())
        cols = list(self._frame._columns)
        proj = pn.Projection(filt, tuple(cols),
                             tuple(ColRef(c) for c in cols))
        return BodoDataFrame(proj, cols)

    def head(self, n=5):
        """First n rows per group in encounter order (row_number window +
        filter; reference: groupby.head)."""
        from ..plan.expr import Cmp, ColRef, Const

        return self._rn_filter(
            lambda: Cmp("le", ColRef("__rn"), Const(int(n))))

    def tail(self, n=5):
        """Last n rows per group in encounter order."""
        from ..plan.expr import BinOp, Cmp, ColRef, Const

        return self._rn_filter(lambda: Cmp(
            "gt", ColRef("__rn"),
            BinOp("sub", ColRef("__sz"), Const(int(n)))))

    def nth(self, n):
        """The n-th row of each group (0-based, encounter order)."""
        from ..plan.expr import Cmp, ColRef, Const

        return self._rn_filter(
            lambda: Cmp("eq", ColRef("__rn"), Const(int(n) + 1)))

    def filter(self, func, *args):
        """Keep rows of groups where func(group_frame) is True (reference:
        groupby.filter): groups co-locate by key hash, pandas filter runs
        per shard."""
        from .frame import BodoDataFrame

        keys = list(self._keys)
        dropna = self._dropna

        def _part(pdf, *a):
            if len(pdf) == 0:
                return pdf
            return pdf.groupby(keys, dropna=dropna,
                               sort=False).filter(func, *a)

        plan = pn.MapPartitions(
            pn.ShuffleByKey(self._frame._lazy_plan, tuple(keys)),
            _part, tuple(args), tuple(self._frame._columns))
        return BodoDataFrame(plan, list(self._frame._columns))

    def _window(self, specs, order_by=(), ascending=()):
        from .frame import BodoDataFrame

        plan = pn.Window(self._frame._lazy_plan, tuple(self._keys),
                         tuple(order_by), tuple(ascending), tuple(specs))
        cols = list(self._frame._columns) + [s[0] for s in specs]
        return BodoDataFrame(plan, cols)

    def transform(self, func):
        from .series import BodoSeries
        from ..plan.expr import ColRef

        cols = self._value_columns()
        specs = [(f"__t_{c}", c, f"transform_{func}", None) for c in cols]
        out = self._window(specs)
        proj = out[[f"__t_{c}" for c in cols]]
        return proj.rename(columns={f"__t_{c}": c for c in cols})

    def cumcount(self):
        out = self._window([("__cc", "", "cumcount", None)])
        return out["__cc"]

    def ngroup(self):
        """Group number in sorted-key order (pandas sort=True default):
        row-id over the sorted distinct keys, merged back."""
        from .frame import BodoDataFrame

        d = self._frame[list(self._keys)].drop_duplicates() \
            .sort_values(list(self._keys))
        rid = "__ng"
        plan = pn.RowId(d._lazy_plan, rid)
        dk = BodoDataFrame(plan, list(self._keys) + [rid])
        out = self._frame.merge(dk, on=list(self._keys), how="left")
        return out[rid]

    def sample(self, n=None, frac=None, random_state=None):
        keys, dropna = self._keys, self._dropna

        def _part(pdf):
            if len(pdf) == 0:
                return pdf
            return pdf.groupby(keys, dropna=dropna, sort=False,
                               group_keys=False).sample(
                n=n, frac=frac, random_state=random_state)

        from .frame import BodoDataFrame

        shuffled = pn.ShuffleByKey(self._frame._lazy_plan, tuple(keys))
        plan = pn.MapPartitions(shuffled, _part, (),
                                tuple(self._frame._columns))
        return BodoDataFrame(plan, list(self._frame._columns))

    def apply(self, func, *args, **kwargs):
        # general groupby-apply: shuffle rows by key then run pandas apply
        # per shard (keys co-located so results are exact)
        frame = self._frame
        keys = self._keys

        def _part(pdf, *a):
            if len(pdf) == 0:
                return pd.DataFrame()
            res = pdf.groupby(keys, dropna=self._dropna,
                              sort=False).apply(func, *a, **kwargs)
            if isinstance(res, pd.Series):
                return res.reset_index()
            # frame-returning funcs: rows already carry the key columns
            # (pandas group_keys flattening)
            try:
                return res.reset_index()
            except ValueError:
                return res.reset_index(drop=True)

        from .frame import BodoDataFrame

        shuffled = pn.ShuffleByKey(frame._lazy_plan, tuple(keys))
        try:
            import pandas as _pd

            probe = _part(frame.head(0).to_pandas())
            names = list(probe.columns)
        except Exception:
            names = []
        plan = pn.MapPartitions(shuffled, _part, (), tuple(names))
        return BodoDataFrame(plan, names)


class SeriesGroupBy:
    def __init__(self, frame, keys, column, as_index, dropna, sort):
        self._frame = frame
        self._keys = keys
        self._col = column
        self._as_index = as_index
        self._dropna = dropna
        self._sort = sort

    def _agg1(self, func):
        gb = DataFrameGroupBy(self._frame, self._keys, self._as_index,
                              self._dropna, self._sort, [self._col])
        out = gb._build([(self._col, self._col, func)])
        if isinstance(out, _IndexedAggResult):
            pdf = _decat(out._frame.to_pandas(), self._keys).set_index(self._keys)[self._col]
            return pdf
        # as_index=False: pandas returns a DataFrame of keys + the column
        return out

    def sum(self):
        return self._agg1("sum")

    def count(self):
        return self._agg1("count")

    def mean(self):
        return self._agg1("mean")

    def min(self):
        return self._agg1("min")

    def max(self):
        return self._agg1("max")

    def nunique(self):
        return self._agg1("nunique")

    def agg(self, func):
        if isinstance(func, str):
            return self._agg1(func)
        if isinstance(func, (list, tuple)):
            from .frame import BodoDataFrame

            aggs = tuple((f, self._col, _norm_func(f)) for f in func)
            plan = pn.Aggregate(self._frame._lazy_plan, tuple(self._keys),
                                aggs)
            return BodoDataFrame(plan, list(self._keys) + list(func))
        if callable(func):
            return self._agg1(func)
        raise NotImplementedError

    def _window1(self, func, in_name=None, arg=None, order_by=(), ascending=()):
        from .frame import BodoDataFrame

        plan = pn.Window(self._frame._lazy_plan, tuple(self._keys),
                         tuple(order_by), tuple(ascending),
                         ((f"__w", in_name if in_name is not None else self._col,
                           func, arg),))
        cols = list(self._frame._columns) + ["__w"]
        return BodoDataFrame(plan, cols)["__w"]

    def transform(self, func):
        return self._window1(f"transform_{func}")

    def shift(self, periods=1):
        return self._window1("shift", arg=periods)

    def cumsum(self):
        return self._window1("cumsum")

    def cumcount(self):
        return self._window1("cumcount")

    def rank(self, method="min", ascending=True):
        return self._window1("rank", arg=method,
                             ascending=(ascending,))

    def _cum_skipna(self, func):
        # pandas cummin/cummax: running extreme skipping NaN, but the
        # output at a NaN input row is NaN (SQL reports the frame value
        # there — mask it back)
        from .frame import BodoDataFrame

        plan = pn.Window(self._frame._lazy_plan, tuple(self._keys),
                         (), (), (("__w", self._col, func, None),))
        fr = BodoDataFrame(plan, list(self._frame._columns) + ["__w"])
        return fr["__w"].where(fr[self._col].notna(), float("nan"))

    def cummin(self):
        return self._cum_skipna("cummin")

    def cummax(self):
        return self._cum_skipna("cummax")

    def pct_change(self, periods=1):
        # fill_method=None semantics (the pandas-3.0 default): NaN inputs
        # propagate instead of being padded forward
        from .frame import BodoDataFrame

        plan = pn.Window(self._frame._lazy_plan, tuple(self._keys),
                         (), (), (("__w", self._col, "shift", periods),))
        fr = BodoDataFrame(plan, list(self._frame._columns) + ["__w"])
        return fr[self._col] / fr["__w"] - 1

    def _idx_of(self, which):
        """Global row position of the group max/min (the drop-in contract:
        frames carry positional range indexes)."""
        from .frame import BodoDataFrame

        rid = "__rid"
        tagged = pn.RowId(self._frame._lazy_plan, rid)
        w = pn.Window(tagged, tuple(self._keys), (), (),
                      (("__m", self._col, f"transform_{which}", None),))
        fr = BodoDataFrame(w, list(self._frame._columns) + [rid, "__m"])
        sel = fr[fr[self._col] == fr["__m"]]
        gb = SeriesGroupBy(sel, self._keys, rid, self._as_index,
                           self._dropna, self._sort)
        out = gb.min()  # ties: first occurrence, like pandas
        return out.sort_index() if isinstance(out, pd.Series) else out

    def idxmax(self):
        return self._idx_of("max")

    def idxmin(self):
        return self._idx_of("min")

    def expanding(self, min_periods=1):
        return _GroupExpanding(self)

    def describe(self):
        gb = DataFrameGroupBy(self._frame, self._keys, self._as_index,
                              self._dropna, self._sort, [self._col])
        return gb.apply(lambda d, _c=self._col: d[_c].describe())


class _GroupExpanding:
    """groupby(...).col.expanding(): running aggregates in row order (the
    pandas MultiIndex wrapper is flattened to frame row order)."""

    def __init__(self, sgb: "SeriesGroupBy"):
        self._g = sgb

    def mean(self):
        return self._g._window1("cummean")

    def sum(self):
        return self._g._window1("cumsum")

    def min(self):
        return self._g._window1("cummin")

    def max(self):
        return self._g._window1("cummax")

    def count(self):
        return self._g._window1("cumcount_v")


class _IndexedAggResult:
    """as_index=True result: behaves like the BodoDataFrame but materializes
    with keys as the index (collected path only)."""

    def __init__(self, frame, keys):
        self._frame = frame
        self._keys = keys

    def to_pandas(self):
        return _decat(self._frame.to_pandas(), self._keys).set_index(self._keys)

    def reset_index(self):
        return self._frame

    def sort_values(self, *args, **kwargs):
        return _IndexedAggResult(self._frame.sort_values(*args, **kwargs),
                                 self._keys)

    def __getattr__(self, name):
        return getattr(self._frame, name)

    def __getitem__(self, k):
        return self._frame[k]

    def __repr__(self):
        return repr(self.to_pandas().head(10))


def _norm_func(f):
    if callable(f) and hasattr(f, "__name__"):
        name = f.__name__
        if name in ("sum", "mean", "min", "max", "count", "size", "median",
                    "var", "std", "prod"):
            return name
        return f  # custom callable: single-phase pandas agg on shuffled groups
    m = {"average": "mean", "nunique": "nunique"}
    return m.get(f, f)

def _decat(pdf, keys):
    """Dict-encoded keys arrive as Categorical; a CategoricalIndex sorts by
    category order, so decode key columns to values for pandas parity."""
    for k in keys:
        if k in pdf.columns and isinstance(pdf[k].dtype, pd.CategoricalDtype):
            pdf = pdf.copy()
            pdf[k] = pdf[k].astype(object)
    return pdf
