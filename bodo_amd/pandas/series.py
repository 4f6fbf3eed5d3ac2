"""BodoSeries: lazy Series with .dt/.str accessors (reference:
bodo/pandas/series.py:99).  A series is an expression over a source plan."""

from __future__ import annotations

import warnings
from typing import Optional

import pandas as pd

from .. import config
from ..core import types as bt
from ..engine import executor as ex
from ..parallel import comm
from ..plan import nodes as pn
from ..plan.expr import (
    BinOp, BoolOp, Case, Cast, Cmp, ColRef, Const, DtField, Expr, IsIn,
    IsNull, ListOp, Not, StrOp, UdfMap, as_expr,
)


class BodoSeries:
    def __init__(self, plan, expr: Expr, name: Optional[str], frame=None):
        self._plan = plan
        self._expr = expr
        self.name = name
        self._frame = frame

    # ------------------------------------------------------------------
    def _wrap(self, expr: Expr, name=None) -> "BodoSeries":
        return BodoSeries(self._plan, expr, name or self.name, self._frame)

    def _bin(self, op, other, reflect=False) -> "BodoSeries":
        if isinstance(other, BodoSeries) and other._plan is not self._plan:
            return self._host_combine(op, other, reflect)
        oe = other._expr if isinstance(other, BodoSeries) else as_expr(other)
        e = BinOp(op, oe, self._expr) if reflect else BinOp(op, self._expr, oe)
        return self._wrap(e, None)

    def _host_combine(self, op, other, reflect=False) -> "BodoSeries":
        """Series from DIFFERENT source plans: positional host alignment
        (expressions can only reference one plan's columns)."""
        import operator as _op

        f = getattr(_op, {"add": "add", "sub": "sub", "mul": "mul",
                          "div": "truediv", "floordiv": "floordiv",
                          "mod": "mod", "pow": "pow", "lt": "lt", "le": "le",
                          "gt": "gt", "ge": "ge", "eq": "eq",
                          "ne": "ne"}[op])
        a = self.to_pandas().reset_index(drop=True)
        b2 = other.to_pandas().reset_index(drop=True)
        res = f(b2, a) if reflect else f(a, b2)
        from .frame import from_pandas_df

        fr = from_pandas_df(res.to_frame(name="v"))
        return BodoSeries(fr._lazy_plan, ColRef("v"), self.name)

    def _cmp(self, op, other) -> "BodoSeries":
        if isinstance(other, BodoSeries) and other._plan is not self._plan:
            return self._host_combine(op, other)
        oe = other._expr if isinstance(other, BodoSeries) else as_expr(other)
        return self._wrap(Cmp(op, self._expr, oe), None)

    __add__ = lambda s, o: s._bin("add", o)
    __radd__ = lambda s, o: s._bin("add", o, True)
    __sub__ = lambda s, o: s._bin("sub", o)
    __rsub__ = lambda s, o: s._bin("sub", o, True)
    __mul__ = lambda s, o: s._bin("mul", o)
    __rmul__ = lambda s, o: s._bin("mul", o, True)
    __truediv__ = lambda s, o: s._bin("div", o)
    __rtruediv__ = lambda s, o: s._bin("div", o, True)
    __floordiv__ = lambda s, o: s._bin("floordiv", o)
    __mod__ = lambda s, o: s._bin("mod", o)
    __pow__ = lambda s, o: s._bin("pow", o)
    __lt__ = lambda s, o: s._cmp("lt", o)
    __le__ = lambda s, o: s._cmp("le", o)
    __gt__ = lambda s, o: s._cmp("gt", o)
    __ge__ = lambda s, o: s._cmp("ge", o)
    __eq__ = lambda s, o: s._cmp("eq", o)
    __ne__ = lambda s, o: s._cmp("ne", o)

    def __and__(self, o):
        oe = o._expr if isinstance(o, BodoSeries) else as_expr(o)
        return self._wrap(BoolOp("and", self._expr, oe), None)

    def __or__(self, o):
        oe = o._expr if isinstance(o, BodoSeries) else as_expr(o)
        return self._wrap(BoolOp("or", self._expr, oe), None)

    def __invert__(self):
        return self._wrap(Not(self._expr), None)

    def __neg__(self):
        return self._wrap(BinOp("sub", Const(0), self._expr), None)

    def __abs__(self):
        return self.abs()

    def __hash__(self):
        return id(self)

    # ------------------------------------------------------------------
    def isin(self, values) -> "BodoSeries":
        if isinstance(values, BodoSeries):
            # semi-join filter: df[a.isin(other_col)] becomes a SEMI join;
            # in general expression positions it evaluates via the other
            # side's distinct values (reference analog: runtime join filters)
            from ..plan.expr import SemiJoinIn

            return self._wrap(
                SemiJoinIn(self._expr, values._plan, values._expr), None)
        if isinstance(values, pd.Series):
            values = values.tolist()
        return self._wrap(IsIn(self._expr, tuple(values)), None)

    def isna(self) -> "BodoSeries":
        return self._wrap(IsNull(self._expr), None)

    isnull = isna

    def notna(self) -> "BodoSeries":
        return self._wrap(IsNull(self._expr, negate=True), None)

    notnull = notna

    def fillna(self, value) -> "BodoSeries":
        if isinstance(value, BodoSeries):
            if value._plan is self._plan:
                return self._wrap(Case((IsNull(self._expr),),
                                       (value._expr,), self._expr), None)
            res = self.to_pandas().reset_index(drop=True).fillna(
                value.to_pandas().reset_index(drop=True))
            from .frame import from_pandas_df

            fr = from_pandas_df(res.to_frame(name="v"))
            return BodoSeries(fr._lazy_plan, ColRef("v"), self.name)
        return self._wrap(
            Case((IsNull(self._expr),), (as_expr(value),), self._expr), None)

    def map(self, func, na_action=None) -> "BodoSeries":
        if isinstance(func, dict):
            d = dict(func)
            return self.map(lambda v: d.get(v), na_action=na_action)
        return self._wrap(UdfMap(self._expr, func, na_action), None)

    apply = map

    def astype(self, dtype) -> "BodoSeries":
        target = _pd_dtype_to_bodo(dtype)
        return self._wrap(Cast(self._expr, target), None)

    def replace(self, to_replace=None, value=None) -> "BodoSeries":
        """Value replacement (reference: series replace overloads); lowers
        to the UdfMap dictionary/LUT machinery."""
        if isinstance(to_replace, dict):
            d = dict(to_replace)
            return self.map(lambda v: d.get(v, v))
        if isinstance(to_replace, (list, tuple)):
            rep = set(to_replace)
            return self.map(lambda v, _r=rep, _val=value:
                            _val if v in _r else v)
        return self.map(lambda v, _t=to_replace, _val=value:
                        _val if v == _t else v)

    def abs(self):
        zero = Const(0)
        return self._wrap(
            Case((Cmp("lt", self._expr, zero),),
                 (BinOp("sub", zero, self._expr),), self._expr), None)

    def round(self, decimals=0):
        from ..plan.expr import RoundExpr

        return self._wrap(RoundExpr(self._expr, decimals), None)

    def where(self, cond, other) -> "BodoSeries":
        if any(isinstance(x, BodoSeries) and x._plan is not self._plan
               for x in (cond, other)):
            # cross-plan operands: positional host alignment
            cv = cond.to_pandas().reset_index(drop=True)                 if isinstance(cond, BodoSeries) else cond
            ov = other.to_pandas().reset_index(drop=True)                 if isinstance(other, BodoSeries) else other
            res = self.to_pandas().reset_index(drop=True).where(cv, ov)
            from .frame import from_pandas_df

            fr = from_pandas_df(res.to_frame(name="v"))
            return BodoSeries(fr._lazy_plan, ColRef("v"), self.name)
        ce = cond._expr if isinstance(cond, BodoSeries) else as_expr(cond)
        oe = other._expr if isinstance(other, BodoSeries) else as_expr(other)
        return self._wrap(Case((ce,), (self._expr,), oe), None)

    def __getitem__(self, key):
        if isinstance(key, BodoSeries):
            # boolean-mask filter: s[s > 0]
            filt = pn.Filter(
                pn.Projection(self._plan, ("v", "__m"),
                              (self._expr, key._expr)),
                ColRef("__m"))
            proj = pn.Projection(filt, ("v",), (ColRef("v"),))
            return BodoSeries(proj, ColRef("v"), self.name)
        if isinstance(key, slice) and key.start in (None, 0) and \
                key.step in (None, 1) and key.stop is not None:
            plan = pn.Limit(self._as_projection_plan(), key.stop)
            return BodoSeries(plan, ColRef("v"), self.name)
        return self.to_pandas()[key]

    @property
    def dt(self):
        return _DtAccessor(self)

    @property
    def str(self):
        return _StrAccessor(self)

    @property
    def list(self):
        return _ListAccessor(self)

    @property
    def struct(self):
        return _StructAccessor(self)

    @property
    def ai(self):
        return _AiAccessor(self)

    # ------------------------------------------------------------------
    # reductions (lazy scalars: the plan executes on first value access —
    # reference: BodoScalar, bodo/pandas/scalar.py:14)
    # ------------------------------------------------------------------
    def _reduce(self, func: str):
        from .scalar import BodoScalar

        plan = pn.Reduce(self._as_projection_plan(), (("r", "v", func),))
        return BodoScalar(plan, "r")

    def sum(self):
        return self._reduce("sum")

    def mean(self):
        return self._reduce("mean")

    def min(self):
        return self._reduce("min")

    def max(self):
        return self._reduce("max")

    def count(self):
        return self._reduce("count")

    def var(self, ddof=1):
        return self._reduce("var")

    def std(self, ddof=1):
        return self._reduce("std")

    def any(self):
        return self._reduce("any")

    def all(self):
        return self._reduce("all")

    def nunique(self):
        from ..engine import api

        plan = pn.Reduce(pn.Distinct(self._as_projection_plan(), ("v",)),
                         (("n", "v", "count"),))
        df = api.collect(plan)
        return int(df["n"].iloc[0])

    def unique(self):
        from ..engine import api

        plan = pn.Distinct(self._as_projection_plan(), ("v",))
        return api.collect(plan)["v"].to_numpy()

    def value_counts(self, normalize=False, ascending=False, dropna=True):
        from ..engine import api

        plan = pn.Aggregate(self._as_projection_plan(), ("v",),
                            (("count", "v", "size"),), dropna=dropna)
        # deterministic tie order: value ascending within equal counts
        plan = pn.Sort(plan, ("count", "v"), (ascending, True))
        pdf = api.collect(plan)
        # ties: pandas orders by value; match roughly
        vals = pdf["count"].to_numpy()
        name = "count"
        if normalize:
            vals = vals / vals.sum() if vals.sum() else vals.astype(float)
            name = "proportion"
        out = pd.Series(vals, index=pdf["v"].to_numpy(), name=name)
        out.index.name = self.name
        return out

    def agg(self, func):
        if isinstance(func, str):
            return getattr(self, func)()
        if isinstance(func, (list, tuple)):
            return pd.Series({f: getattr(self, f)() for f in func},
                             name=self.name)
        raise NotImplementedError(f"Series.agg({func!r})")

    aggregate = agg

    def prod(self):
        return self._reduce("prod")

    product = prod

    def quantile(self, q=0.5, interpolation="linear"):
        """Exact quantile: gathers the projected column (distributed shards
        allgather) and interpolates with numpy.  For terabyte-scale inputs
        prefer .sample(...)-based estimates; groupby('...').median() stays
        fully distributed."""
        ser = self.dropna().to_pandas()
        from decimal import Decimal as _D

        if len(ser) and isinstance(ser.iloc[0], _D):
            ser = ser.astype(float)
        if hasattr(q, "__len__"):
            return pd.Series([ser.quantile(x, interpolation=interpolation)
                              for x in q], index=list(q), name=self.name)
        return ser.quantile(q, interpolation=interpolation)

    def median(self):
        return self.quantile(0.5)

    def describe(self):
        from .scalar import _unwrap

        cnt = self.count()
        qs = self.quantile([0.25, 0.5, 0.75])
        vals = [_unwrap(cnt), _unwrap(self.mean()), _unwrap(self.std()),
                _unwrap(self.min()),
                qs.iloc[0], qs.iloc[1], qs.iloc[2], _unwrap(self.max())]
        return pd.Series(vals, index=["count", "mean", "std", "min", "25%",
                                      "50%", "75%", "max"], name=self.name)

    def mode(self):
        vc = self.value_counts(ascending=False)
        if not len(vc):
            return pd.Series([], dtype="object", name=self.name)
        top = vc[vc == vc.max()].index.to_numpy()
        return pd.Series(sorted(top), name=self.name)

    def between(self, left, right, inclusive="both"):
        lo = Cmp("ge" if inclusive in ("both", "left") else "gt",
                 self._expr, as_expr(left))
        hi = Cmp("le" if inclusive in ("both", "right") else "lt",
                 self._expr, as_expr(right))
        return self._wrap(BoolOp("and", lo, hi), None)

    def clip(self, lower=None, upper=None):
        e = self._expr
        conds, thens = [], []
        if lower is not None:
            conds.append(Cmp("lt", e, as_expr(lower)))
            thens.append(as_expr(lower))
        if upper is not None:
            conds.append(Cmp("gt", e, as_expr(upper)))
            thens.append(as_expr(upper))
        if not conds:
            return self
        return self._wrap(Case(tuple(conds), tuple(thens), e), None)

    def dropna(self) -> "BodoSeries":
        filt = pn.Filter(self._as_projection_plan(),
                         IsNull(ColRef("v"), negate=True))
        return BodoSeries(filt, ColRef("v"), self.name)

    def sort_values(self, ascending=True) -> "BodoSeries":
        plan = pn.Sort(self._as_projection_plan(), ("v",), (bool(ascending),))
        return BodoSeries(plan, ColRef("v"), self.name)

    def nlargest(self, n=5) -> pd.Series:
        from ..engine import api

        plan = pn.Limit(pn.Sort(self._as_projection_plan(), ("v",), (False,)),
                        n)
        ser = api.collect(plan)["v"]
        ser.name = self.name
        return ser

    def nsmallest(self, n=5) -> pd.Series:
        from ..engine import api

        plan = pn.Limit(pn.Sort(self._as_projection_plan(), ("v",), (True,)),
                        n)
        ser = api.collect(plan)["v"]
        ser.name = self.name
        return ser

    def _pair_moments(self, other: "BodoSeries"):
        """Distributed co-moments in ONE pass: n, Σx, Σy, Σxy, Σx², Σy²
        (reference analog: sklearn_ext/array_kernels corr via allreduce).
        Requires both series to share a source plan (the common case:
        columns of one frame); mixed-plan pairs fall back to host."""
        from ..engine import api

        if other._plan is not self._plan:
            return None
        x, y = self._expr, other._expr
        # pairwise-complete observations (pandas semantics): drop rows where
        # either side is null before the moment sums
        flt = pn.Filter(self._plan, BoolOp(
            "and", IsNull(x, negate=True), IsNull(y, negate=True)))
        proj = pn.Projection(flt, ("x", "y", "xy", "xx", "yy"),
                             (x, y, BinOp("mul", x, y), BinOp("mul", x, x),
                              BinOp("mul", y, y)))
        red = pn.Reduce(proj, (("n", "x", "count"), ("sx", "x", "sum"),
                               ("sy", "y", "sum"), ("sxy", "xy", "sum"),
                               ("sxx", "xx", "sum"), ("syy", "yy", "sum")))
        row = api.collect(red)
        return (float(row["n"].iloc[0]), float(row["sx"].iloc[0]),
                float(row["sy"].iloc[0]), float(row["sxy"].iloc[0]),
                float(row["sxx"].iloc[0]), float(row["syy"].iloc[0]))

    def cov(self, other: "BodoSeries", ddof: int = 1) -> float:
        m = self._pair_moments(other)
        if m is None:  # different source plans: align on host
            return self.to_pandas().cov(other.to_pandas(), ddof=ddof)
        n, sx, sy, sxy, _, _ = m
        if n <= ddof:
            return float("nan")
        return (sxy - sx * sy / n) / (n - ddof)

    def corr(self, other: "BodoSeries") -> float:
        m = self._pair_moments(other)
        if m is None:  # different source plans: align on host
            return self.to_pandas().corr(other.to_pandas())
        n, sx, sy, sxy, sxx, syy = m
        if n < 2:
            return float("nan")
        cov = sxy - sx * sy / n
        vx = sxx - sx * sx / n
        vy = syy - sy * sy / n
        if vx <= 0 or vy <= 0:
            return float("nan")
        return cov / (vx ** 0.5 * vy ** 0.5)

    def ffill(self) -> "BodoSeries":
        plan = pn.Fill(self._as_projection_plan(), True, (("v", "v"),))
        return BodoSeries(plan, ColRef("v"), self.name)

    def bfill(self) -> "BodoSeries":
        plan = pn.Fill(self._as_projection_plan(), False, (("v", "v"),))
        return BodoSeries(plan, ColRef("v"), self.name)

    def pct_change(self, periods=1) -> "BodoSeries":
        plan = pn.Shift(self._as_projection_plan(), int(periods),
                        (("__sh", "v"),))
        expr = BinOp("div", BinOp("sub", ColRef("v"), ColRef("__sh")),
                     ColRef("__sh"))
        return BodoSeries(plan, expr, self.name)

    def duplicated(self, keep="first") -> "BodoSeries":
        """True for repeats of an earlier (keep='first') occurrence —
        per-value running count via the window machinery."""
        if keep != "first":
            raise NotImplementedError("duplicated(keep!='first')")
        plan = pn.Window(self._as_projection_plan(), ("v",), (), (),
                         (("__cc", "", "cumcount", None),))
        return BodoSeries(plan, Cmp("gt", ColRef("__cc"), Const(0)),
                          self.name)

    def to_frame(self, name=None):
        from .frame import BodoDataFrame

        out_name = name or self.name or "0"
        plan = pn.Projection(self._plan, (out_name,), (self._expr,))
        return BodoDataFrame(plan, [out_name])

    def rename(self, name, **kwargs) -> "BodoSeries":
        return BodoSeries(self._plan, self._expr, name, self._frame)

    def tail(self, n=5) -> pd.Series:
        from ..engine import api

        plan = pn.Limit(self._as_projection_plan(), n, tail=True)
        ser = api.collect(plan)["v"]
        ser.name = self.name
        return ser

    def sample(self, n=None, frac=None, random_state=None) -> "BodoSeries":
        plan = pn.Sample(self._as_projection_plan(), n, frac, random_state)
        return BodoSeries(plan, ColRef("v"), self.name)

    def _idx_of(self, func: str) -> int:
        from ..engine import api

        extreme = self.max() if func == "max" else self.min()
        plan = pn.RowId(self._as_projection_plan(), "__rid")
        flt = pn.Filter(plan, Cmp("eq", ColRef("v"), Const(extreme)))
        red = pn.Reduce(flt, (("i", "__rid", "min"),))
        return int(api.collect(red)["i"].iloc[0])

    def idxmax(self) -> int:
        """Positional index of the first maximum (frames are positionally
        indexed; the global row id comes from a RowId tag)."""
        return self._idx_of("max")

    def idxmin(self) -> int:
        return self._idx_of("min")

    def map_partitions(self, func, *args) -> "BodoSeries":
        """Run a python function over each rank's shard of this series
        (reference: series.py map_partitions)."""

        def _part(pdf, *a):
            res = func(pdf["v"], *a)
            return res.to_frame(name="v") if isinstance(res, pd.Series) \
                else pd.DataFrame({"v": res})

        plan = pn.MapPartitions(self._as_projection_plan(), _part,
                                tuple(args), ("v",))
        return BodoSeries(plan, ColRef("v"), self.name)

    def convert_dtypes(self, **kwargs) -> "BodoSeries":
        return self  # columns are already arrow-typed

    def autocorr(self, lag=1) -> float:
        """Pearson autocorrelation at the given lag (distributed shift +
        shared-plan co-moments)."""
        plan = pn.Shift(self._as_projection_plan(), int(lag),
                        (("__sh", "v"),))
        a = BodoSeries(plan, ColRef("v"), None)
        b = BodoSeries(plan, ColRef("__sh"), None)
        return a.corr(b)

    def dot(self, other):
        return (self * other).sum()

    def rolling(self, window, min_periods=None, center=False, **kwargs):
        if isinstance(window, str):
            return _RollingSeriesHost(self, window, min_periods)
        return _RollingSeries(self, int(window), min_periods, center)

    def ewm(self, com=None, span=None, alpha=None, adjust=True):
        return _EwmSeries(self, com, span, alpha, adjust)

    def _cum(self, func) -> "BodoSeries":
        plan = pn.Cumulative(self._as_projection_plan(), (("v", "v", func),))
        return BodoSeries(plan, ColRef("v"), self.name)

    def cumsum(self):
        return self._cum("cumsum")

    def cumprod(self):
        return self._cum("cumprod")

    def cummin(self):
        return self._cum("cummin")

    def cummax(self):
        return self._cum("cummax")

    def mask(self, cond, other=float("nan")) -> "BodoSeries":
        return self.where(~cond, other)

    def rank(self, method="average", ascending=True,
             pct=False) -> "BodoSeries":
        """Global rank via the window machinery with a constant partition
        key (reference: array_kernels rank); pct divides by the non-null
        count like pandas."""
        proj = pn.Projection(self._plan, ("v", "__k"),
                             (self._expr, Const(1)))
        plan = pn.Window(proj, ("__k",), ("v",), (bool(ascending),),
                         (("__r", "v", "rank", method),))
        r = BodoSeries(plan, ColRef("__r"), self.name)
        if pct:
            return r / float(self.count())
        return r

    def shift(self, periods=1) -> "BodoSeries":
        plan = pn.Shift(self._as_projection_plan(), int(periods),
                        (("__sh", "v"),))
        return BodoSeries(plan, ColRef("__sh"), self.name)

    def diff(self, periods=1) -> "BodoSeries":
        plan = pn.Shift(self._as_projection_plan(), int(periods),
                        (("__sh", "v"),))
        return BodoSeries(plan, BinOp("sub", ColRef("v"), ColRef("__sh")),
                          self.name)

    def _as_projection_plan(self):
        return pn.Projection(self._plan, ("v",), (self._expr,))

    # ------------------------------------------------------------------
    def to_pandas(self) -> pd.Series:
        from ..engine import api

        plan = pn.Projection(self._plan, ("v",), (self._expr,))
        ser = api.collect(plan)["v"]
        ser.name = self.name
        return ser

    def head(self, n=5):
        from ..engine import api

        plan = pn.Limit(pn.Projection(self._plan, ("v",), (self._expr,)), n)
        ser = api.collect(plan)["v"]
        ser.name = self.name
        return ser

    def __len__(self):
        return len(self.to_pandas())

    def __repr__(self):
        return repr(self.head(10))

    def __getattr__(self, name):
        if name.startswith("_"):
            raise AttributeError(name)
        if hasattr(pd.Series, name) and config.PANDAS_FALLBACK:
            attr = getattr(pd.Series, name)
            if callable(attr):
                def method(*args, **kwargs):
                    warnings.warn(f"BodoSeries.{name}: falling back to pandas",
                                  stacklevel=2)
                    return getattr(self.to_pandas(), name)(*args, **kwargs)

                return method
            warnings.warn(f"BodoSeries.{name}: falling back to pandas",
                          stacklevel=2)
            return getattr(self.to_pandas(), name)
        raise AttributeError(name)


class _DtAccessor:
    def __init__(self, s: BodoSeries):
        self._s = s

    def _f(self, fld):
        return self._s._wrap(DtField(self._s._expr, fld), None)

    @property
    def date(self):
        return self._f("date")

    @property
    def year(self):
        return self._f("year")

    @property
    def month(self):
        return self._f("month")

    @property
    def day(self):
        return self._f("day")

    @property
    def hour(self):
        return self._f("hour")

    @property
    def minute(self):
        return self._f("minute")

    @property
    def second(self):
        return self._f("second")

    @property
    def dayofweek(self):
        return self._f("dayofweek")

    weekday = dayofweek

    @property
    def dayofyear(self):
        return self._f("dayofyear")

    @property
    def quarter(self):
        return self._f("quarter")

    def normalize(self):
        return self._f("normalize")

    def floor(self, freq):
        assert freq in ("D", "d"), "only day floor supported"
        return self._f("floor_day")

    @property
    def is_month_start(self):
        return self._f("is_month_start")

    @property
    def is_month_end(self):
        return self._f("is_month_end")

    @property
    def is_quarter_start(self):
        return self._f("is_quarter_start")

    @property
    def is_quarter_end(self):
        return self._f("is_quarter_end")

    @property
    def is_year_start(self):
        return self._f("is_year_start")

    @property
    def is_year_end(self):
        return self._f("is_year_end")

    @property
    def days_in_month(self):
        return self._f("days_in_month")

    @property
    def is_leap_year(self):
        return self._f("is_leap_year")

    daysinmonth = days_in_month

    def month_name(self, locale=None):
        # 12-value LUT over the month field (dense-domain UdfMap builds a
        # dictionary column, so this stays device-resident)
        import calendar

        return self._s._wrap(
            UdfMap(DtField(self._s._expr, "month"),
                   lambda m: calendar.month_name[int(m)], None), None)

    def day_name(self, locale=None):
        import calendar

        return self._s._wrap(
            UdfMap(DtField(self._s._expr, "dayofweek"),
                   lambda d: calendar.day_name[int(d)], None), None)

    def to_period(self, freq):
        fld = {"M": "trunc_month", "Y": "trunc_year", "Q": "trunc_quarter",
               "W": "trunc_week", "D": "floor_day"}.get(freq)
        if fld is None:
            raise NotImplementedError(f"to_period({freq!r})")
        return self._f(fld)

    def __getattr__(self, name):
        """Anything else materializes through the pandas .dt accessor with
        a fallback warning (mirrors the Series-level fallback)."""
        if name.startswith("_"):
            raise AttributeError(name)
        import warnings

        host = self._s.to_pandas().dt
        if not hasattr(host, name):
            raise AttributeError(name)
        warnings.warn(f"Series.dt.{name}: falling back to pandas",
                      stacklevel=2)
        return getattr(host, name)


class _StrAccessor:
    def __init__(self, s: BodoSeries):
        self._s = s

    def _f(self, op, *args):
        return self._s._wrap(StrOp(self._s._expr, op, tuple(args)), None)

    def lower(self):
        return self._f("lower")

    def upper(self):
        return self._f("upper")

    def strip(self):
        return self._f("strip")

    def title(self):
        return self._f("title")

    def capitalize(self):
        return self._f("capitalize")

    def len(self):
        return self._f("len")

    def contains(self, pat, regex=True, **kw):
        if regex and any(ch in pat for ch in ".*+?[](){}|\\^$"):
            return self._f("contains_re", pat)
        return self._f("contains", pat)

    def match(self, pat):
        return self._f("match", pat)

    def startswith(self, pat):
        return self._f("startswith", pat)

    def endswith(self, pat):
        return self._f("endswith", pat)

    def slice(self, start=None, stop=None, step=None):
        return self._f("slice", start or 0, stop, step or 1)

    def replace(self, pat, repl, regex=True, **kw):
        return self._s._wrap(
            StrOp(self._s._expr, "replace", (pat, repl),
                  (("regex", bool(regex)),)), None)

    def split(self, pat=None, n=-1, regex=None):
        """Returns a handle supporting .str.get(i) / .str[i], lowered to a
        single fused split+get pass (list-valued intermediates never
        materialize; reference: BodoSQL split_part kernel)."""
        return _SplitResult(self._s, pat)

    def findall(self, pat, flags=0):
        """LIST<string> result (reference: BodoSQL regexp kernels +
        array_item results)."""
        return self._s._wrap(StrOp(self._s._expr, "findall", (pat,),
                                   (("flags", flags),)), None)

    def extract(self, pat, flags=0, expand=True):
        """Single-group extract lowers to a host regex map; multi-group /
        expand results fall back (frame-shaped)."""
        import re as _re

        if _re.compile(pat, flags).groups != 1 or expand:
            raise AttributeError("extract")  # series-level pandas fallback
        rx = _re.compile(pat, flags)

        def _ex(v, _rx=rx):
            if v is None:
                return None
            m = _rx.search(v)
            return m.group(1) if m else None

        return self._s._wrap(UdfMap(self._s._expr, _ex, "ignore"), None)

    # methods whose pandas result is frame-shaped fall back via the
    # Series-level pandas fallback
    _UNSUPPORTED = {"extractall", "get_dummies", "partition", "rpartition",
                    "cat"}

    def __getattr__(self, op):
        """Any other pandas .str method lowers to a StrOp evaluated on the
        dictionary (DICT columns: one host call over the unique values) or
        the host string array (reference: BodoSeries str accessor,
        bodo/pandas/series.py)."""
        if op.startswith("_") or op in self._UNSUPPORTED or \
                not hasattr(pd.Series([], dtype="object").str, op):
            raise AttributeError(op)

        def method(*args, **kwargs):
            return self._s._wrap(
                StrOp(self._s._expr, op, tuple(args),
                      tuple(sorted(kwargs.items()))), None)

        return method


def _pd_dtype_to_bodo(dtype):
    import numpy as np

    m = {
        "int8": bt.int8, "int16": bt.int16, "int32": bt.int32, "int64": bt.int64,
        "float32": bt.float32, "float64": bt.float64, "bool": bt.boolean,
        "str": bt.string, str: bt.string, int: bt.int64, float: bt.float64,
        bool: bt.boolean, "category": bt.dictionary, "string": bt.string,
    }
    if dtype in m:
        return m[dtype]
    nd = np.dtype(dtype)
    if nd.kind == "M":
        return bt.timestamp_ns
    return bt.from_numpy_dtype(nd)


class _ListAccessor:
    """Series.list over LIST columns (pandas ArrowDtype .list surface;
    reference role: array_item element kernels)."""

    def __init__(self, s: BodoSeries):
        self._s = s

    def len(self):
        return self._s._wrap(ListOp(self._s._expr, "len"), None)

    def get(self, i):
        return self._s._wrap(ListOp(self._s._expr, "get", int(i)), None)

    __getitem__ = get


class _RollingSeriesHost:
    """Offset-string rolling windows on a series: host pandas fallback."""

    _FUNCS = ("sum", "mean", "min", "max", "count", "std", "var", "median")

    def __init__(self, s, window, min_periods):
        self._s = s
        self._window = window
        self._min_periods = min_periods

    def __getattr__(self, name):
        if name in self._FUNCS:
            return lambda: getattr(self._s.to_pandas().rolling(
                self._window, min_periods=self._min_periods or 1), name)()
        raise AttributeError(name)


class _EwmSeries:
    """s.ewm(...).mean()/var()/std(): host pandas on the gathered series
    (the recurrence needs cross-shard carry; replicated exact result)."""

    def __init__(self, s, com, span, alpha, adjust):
        self._s = s
        self._kw = dict(com=com, span=span, alpha=alpha, adjust=adjust)

    def _run(self, func):
        kw = {k: v for k, v in self._kw.items() if v is not None or
              k == "adjust"}
        return getattr(self._s.to_pandas().ewm(**kw), func)()

    def mean(self):
        return self._run("mean")

    def var(self):
        return self._run("var")

    def std(self):
        return self._run("std")


class _AiAccessor:
    """Series.ai (reference: BodoSeriesAiMethods, bodo/pandas/series.py:1961).
    Offline-first: every method takes a local callable/tokenizer — there is
    no network in this environment, so remote endpoints are the caller's
    responsibility (pass a client function)."""

    def __init__(self, s: BodoSeries):
        self._s = s

    def tokenize(self, tokenizer):
        """Token ids per row as LIST<int64>; `tokenizer` is a HuggingFace
        tokenizer (has .encode) or any str -> list[int] callable."""
        enc = tokenizer.encode if hasattr(tokenizer, "encode") else tokenizer
        from ..plan.expr import UdfMap

        return self._s._wrap(
            UdfMap(self._s._expr, lambda v: list(enc(str(v))), "ignore"),
            self._s.name)

    def embed(self, model, batch_size: int = 256):
        """Embeddings per row as LIST<float>; `model` maps a list[str] to a
        list of vectors (batched on each shard)."""
        def _map(pdf_col):
            out = []
            vals = pdf_col.tolist()
            for i in range(0, len(vals), batch_size):
                out.extend(model([str(v) for v in vals[i:i + batch_size]]))
            return [list(map(float, e)) for e in out]

        ser = self._s.to_pandas()
        import pandas as pd_

        res = pd_.Series(_map(ser), index=ser.index, name=self._s.name)
        from . import from_pandas

        return from_pandas(res.to_frame("v"))["v"]

    def llm_generate(self, fn, **kwargs):
        """Row-wise generation through a caller-provided client callable
        (prompt -> completion)."""
        from ..plan.expr import UdfMap

        return self._s._wrap(
            UdfMap(self._s._expr,
                   lambda v, _f=fn, _k=kwargs: _f(str(v), **_k), "ignore"),
            self._s.name)


class _StructAccessor:
    """Series.struct over STRUCT columns (pandas ArrowDtype .struct
    surface; reference role: struct_arr_ext field access)."""

    def __init__(self, s: BodoSeries):
        self._s = s

    def field(self, name: str):
        return self._s._wrap(ListOp(self._s._expr, "get", str(name)), name)

    __getitem__ = field


class _SplitResult:
    """Lazy result of .str.split(pat): element access fuses into one
    split_get pass; any other use materializes the real LIST<string>
    series (reference: str split -> array_item array)."""

    def __init__(self, s: BodoSeries, pat):
        object.__setattr__(self, "_s", s)
        object.__setattr__(self, "_pat", pat)

    @property
    def str(self):
        return self

    def get(self, i):
        return self._s._wrap(
            StrOp(self._s._expr, "split_get", (self._pat, int(i))), None)

    __getitem__ = get

    def _series(self) -> "BodoSeries":
        return self._s._wrap(
            StrOp(self._s._expr, "split_list", (self._pat,)), None)

    def __getattr__(self, name):
        return getattr(self._series(), name)


class _RollingSeries:
    """s.rolling(w): rolling aggregation over the series (Rolling plan node
    with distributed halo exchange)."""

    _FUNCS = ("sum", "mean", "min", "max", "count", "std", "var", "median")

    def __init__(self, s: BodoSeries, window: int, min_periods,
                 center: bool = False):
        self._s = s
        self._window = window
        self._min_periods = min_periods
        self._center = center

    def _agg(self, func):
        plan = pn.Rolling(self._s._as_projection_plan(), self._window,
                          self._min_periods, (("v", "v", func),))
        if self._center and self._window > 1:
            plan = pn.Shift(plan, -(self._window // 2), (("v", "v"),))
        return BodoSeries(plan, ColRef("v"), self._s.name)

    def __getattr__(self, name):
        if name in self._FUNCS:
            return lambda: self._agg(name)
        raise AttributeError(name)
