"""BodoDataFrame: lazy drop-in pandas DataFrame (reference:
bodo/pandas/frame.py:117).  Methods build logical plan nodes; materialization
(executed on every rank's GPU shard, gathered on collect) is triggered by
len/repr/to_pandas/fallback."""

from __future__ import annotations

import warnings
from typing import Dict, List, Optional, Sequence

import numpy as np
import pandas as pd

from .. import config
from ..core.table import Table
from ..engine import executor as ex
from ..parallel import comm
from ..plan import nodes as pn
from ..plan.expr import BoolOp, ColRef, Expr, Not


class BodoDataFrame:
    def __setattr__(self, name, value):
        if name == "columns":
            self._set_columns(value)
            return
        object.__setattr__(self, name, value)

    def __init__(self, plan: pn.PlanNode, columns: Sequence[str]):
        object.__setattr__(self, "_plan", plan)
        object.__setattr__(self, "_columns", list(columns))
        object.__setattr__(self, "_result", None)  # cached local shard Table
        object.__setattr__(self, "_remote", None)  # spawn-mode RemoteResult

    # ------------------------------------------------------------------
    # plan / execution
    # ------------------------------------------------------------------
    @property
    def _lazy_plan(self) -> pn.PlanNode:
        return self._plan

    def execute(self) -> Table:
        """Execute the plan; returns this rank's shard and caches it
        (SPMD/local mode only — spawn mode goes through _materialize)."""
        self._materialize()
        return self._result

    def _materialize(self):
        """Execute the plan locally or on spawn-mode workers; re-roots the
        plan on the materialized result (ExecState.DISTRIBUTED)."""
        if self._result is not None or getattr(self, "_remote", None) is not None:
            return
        from ..engine import api

        kind, res = api.materialize(self._plan)
        if kind == "remote":
            object.__setattr__(self, "_remote", res)
            object.__setattr__(
                self, "_plan",
                pn.PandasScan(res.res_id, tuple(res.names), distributed=True))
            object.__setattr__(self, "_columns", list(res.names))
            return
        tbl = res
        object.__setattr__(self, "_result", tbl)
        key = ex.register_object(tbl)
        object.__setattr__(
            self, "_plan",
            pn.PandasScan(key, tuple(tbl.names), distributed=True))
        object.__setattr__(self, "_columns", list(tbl.names))

    def to_pandas(self) -> pd.DataFrame:
        self._materialize()
        remote = getattr(self, "_remote", None)
        if remote is not None:
            from ..parallel import spawn

            out = spawn.get_spawner().gather(remote.res_id).to_pandas()
        else:
            full = comm.allgather_table(self._result)
            out = full.to_pandas()
        idx = list(getattr(self, "_index_cols", []) or [])
        if idx:
            out = out.set_index(idx)
        return out

    # aliases used by tests / fallback
    collect = to_pandas

    # ------------------------------------------------------------------
    # basic introspection
    # ------------------------------------------------------------------
    @property
    def columns(self):
        return pd.Index(self._columns)

    # (setter defined below via _set_columns)

    @property
    def shape(self):
        n = len(self)
        return (n, len(self._columns))

    def __len__(self) -> int:
        self._materialize()
        remote = getattr(self, "_remote", None)
        if remote is not None:
            return remote.length
        return int(sum(comm.allgather_obj(len(self._result))))

    @property
    def empty(self) -> bool:
        return len(self) == 0

    @property
    def dtypes(self):
        return self.head(1).to_pandas().dtypes

    def __repr__(self) -> str:
        head = self.head(10).to_pandas()
        return repr(head)

    # ------------------------------------------------------------------
    # column access
    # ------------------------------------------------------------------
    def __getitem__(self, key):
        from .series import BodoSeries

        if isinstance(key, str):
            if key not in self._columns:
                raise KeyError(key)
            return BodoSeries(self._plan, ColRef(key), key, frame=self)
        if isinstance(key, list):
            missing = [k for k in key if k not in self._columns]
            if missing:
                raise KeyError(missing)
            exprs = tuple(ColRef(k) for k in key)
            return BodoDataFrame(pn.Projection(self._plan, tuple(key), exprs), key)
        if isinstance(key, BodoSeries):
            from ..plan.expr import Not as _Not
            from ..plan.expr import SemiJoinIn as _SJ

            e = key._expr
            if isinstance(e, _SJ) and isinstance(e.operand, ColRef):
                return self._semi_filter(e, anti=False)
            if isinstance(e, _Not) and isinstance(e.operand, _SJ)                     and isinstance(e.operand.operand, ColRef):
                return self._semi_filter(e.operand, anti=True)
            if key._plan is not self._plan:
                # mask from a different frame: positional host alignment
                return self._fallback(
                    "__getitem__",
                    key.to_pandas().reset_index(drop=True))
            return BodoDataFrame(pn.Filter(self._plan, key._expr), self._columns)
        if isinstance(key, pd.Series) and key.dtype == bool:
            # host boolean mask: materialize path
            return self._fallback("__getitem__", key)
        raise TypeError(f"unsupported key {type(key)}")

    def _semi_filter(self, e, anti: bool):
        other = pn.Distinct(
            pn.Projection(e.other_plan, ("__in_v",), (e.other_expr,)),
            ("__in_v",))
        plan = pn.Join(self._plan, other, (e.operand.name,), ("__in_v",),
                       "anti" if anti else "semi")
        return BodoDataFrame(plan, self._columns)

    @property
    def loc(self):
        return _LocIndexer(self)

    def _set_columns(self, new_cols):
        new_cols = list(new_cols)
        assert len(new_cols) == len(self._columns)
        exprs = tuple(ColRef(c) for c in self._columns)
        plan = pn.Projection(self._plan, tuple(new_cols), exprs)
        object.__setattr__(self, "_plan", plan)
        object.__setattr__(self, "_columns", new_cols)
        object.__setattr__(self, "_result", None)
        object.__setattr__(self, "_remote", None)

    def __setitem__(self, key: str, value):
        from .series import BodoSeries

        if isinstance(value, BodoSeries):
            vplan = value._plan
            if (isinstance(vplan, pn.MapPartitions)
                    and vplan.child is self._plan):
                # df["s"] = df.apply(f, axis=1): fuse into one partition pass
                inner, expr_v = vplan.func, value._expr
                col = expr_v.name if isinstance(expr_v, ColRef) else None

                def _fused(pdf, *a, _inner=inner, _col=col, _key=key):
                    res = _inner(pdf, *a)
                    ser = res[_col] if _col is not None else res.iloc[:, 0]
                    out = pdf.copy()
                    out[_key] = ser.to_numpy()
                    return out

                names = list(self._columns)
                if key not in names:
                    names.append(key)
                new_plan = pn.MapPartitions(self._plan, _fused, vplan.args,
                                            tuple(names))
                object.__setattr__(self, "_plan", new_plan)
                object.__setattr__(self, "_columns", names)
                object.__setattr__(self, "_result", None)
                object.__setattr__(self, "_remote", None)
                return
            if (isinstance(vplan, pn.Window) and vplan.child is self._plan
                    and isinstance(value._expr, ColRef)
                    and value._expr.name == vplan.specs[0][0]):
                # df["r"] = df.groupby(k)[c].shift()/rank()/...: rename the
                # window output column to the assignment target
                spec = vplan.specs[0]
                new_specs = ((key,) + spec[1:],) + vplan.specs[1:]
                names = list(self._columns)
                if key not in names:
                    names.append(key)
                    new_plan = pn.Window(self._plan, vplan.keys,
                                         vplan.order_by, vplan.ascending,
                                         new_specs)
                else:
                    # overwrite existing column: window then project
                    w_plan = pn.Window(self._plan, vplan.keys, vplan.order_by,
                                       vplan.ascending, vplan.specs)
                    exprs = [ColRef(c) if c != key else ColRef(spec[0])
                             for c in names]
                    new_plan = pn.Projection(w_plan, tuple(names),
                                             tuple(exprs))
                object.__setattr__(self, "_plan", new_plan)
                object.__setattr__(self, "_columns", names)
                object.__setattr__(self, "_result", None)
                object.__setattr__(self, "_remote", None)
                return
            if vplan is not self._plan:
                # allow setting from a series derived from the same frame
                # lineage after assignments: rebuild on current plan if the
                # referenced columns still exist
                from ..plan.expr import expr_columns

                refs = expr_columns(value._expr)
                if not refs.issubset(set(self._columns)):
                    # series from an unrelated frame: positional host align
                    pdf = self.to_pandas().reset_index(drop=True)
                    pdf[key] = value.to_pandas().reset_index(
                        drop=True).to_numpy()
                    nf = from_pandas_df(pdf)
                    object.__setattr__(self, "_plan", nf._plan)
                    object.__setattr__(self, "_columns",
                                       list(nf._columns))
                    object.__setattr__(self, "_result", None)
                    object.__setattr__(self, "_remote", None)
                    return
            expr = value._expr
        else:
            from ..plan.expr import Const

            expr = Const(value)
        names = list(self._columns)
        exprs = [ColRef(c) for c in names]
        if key in names:
            exprs[names.index(key)] = expr
        else:
            names.append(key)
            exprs.append(expr)
        new_plan = pn.Projection(self._plan, tuple(names), tuple(exprs))
        object.__setattr__(self, "_plan", new_plan)
        object.__setattr__(self, "_columns", names)
        object.__setattr__(self, "_result", None)
        object.__setattr__(self, "_remote", None)

    def assign(self, **kwargs) -> "BodoDataFrame":
        out = BodoDataFrame(self._plan, self._columns)
        for k, v in kwargs.items():
            out[k] = v(out) if callable(v) else v
        return out

    def rename(self, columns: Optional[Dict[str, str]] = None, copy=None,
               inplace=False, **kwargs) -> "BodoDataFrame":
        assert columns is not None
        names = [columns.get(c, c) for c in self._columns]
        exprs = tuple(ColRef(c) for c in self._columns)
        plan = pn.Projection(self._plan, tuple(names), exprs)
        if inplace:
            object.__setattr__(self, "_plan", plan)
            object.__setattr__(self, "_columns", names)
            object.__setattr__(self, "_result", None)
            object.__setattr__(self, "_remote", None)
            return None
        return BodoDataFrame(plan, names)

    def drop(self, labels=None, columns=None, axis=0, inplace=False, **kw):
        if columns is None and axis in (1, "columns"):
            columns = labels
        if columns is None:
            return self._fallback("drop", labels=labels, axis=axis, **kw)
        if isinstance(columns, str):
            columns = [columns]
        keep = [c for c in self._columns if c not in set(columns)]
        plan = pn.Projection(self._plan, tuple(keep),
                             tuple(ColRef(c) for c in keep))
        if inplace:
            object.__setattr__(self, "_plan", plan)
            object.__setattr__(self, "_columns", keep)
            object.__setattr__(self, "_result", None)
            object.__setattr__(self, "_remote", None)
            return None
        return BodoDataFrame(plan, keep)

    def __getattr__(self, name: str):
        # column attribute access
        cols = object.__getattribute__(self, "_columns")
        if name in cols:
            return self[name]
        if name.startswith("_"):
            raise AttributeError(name)
        if hasattr(pd.DataFrame, name) and config.PANDAS_FALLBACK:
            attr = getattr(pd.DataFrame, name)
            if callable(attr):
                def method(*args, **kwargs):
                    return self._fallback(name, *args, **kwargs)

                return method
            warnings.warn(f"BodoDataFrame.{name}: falling back to pandas "
                          "(materializes the result)", stacklevel=2)
            return getattr(self.to_pandas(), name)
        raise AttributeError(name)

    def _fallback(self, name, *args, **kwargs):
        warnings.warn(f"BodoDataFrame.{name}: falling back to pandas "
                      "(materializes the result)", stacklevel=3)
        pdf = self.to_pandas()
        res = getattr(pdf, name)(*args, **kwargs)
        return wrap_result(res)

    # ------------------------------------------------------------------
    # relational methods
    # ------------------------------------------------------------------
    def head(self, n: int = 5) -> "BodoDataFrame":
        return BodoDataFrame(pn.Limit(self._plan, n), self._columns)

    def tail(self, n: int = 5) -> "BodoDataFrame":
        return BodoDataFrame(pn.Limit(self._plan, n, tail=True), self._columns)

    def merge(self, right: "BodoDataFrame", how="inner", on=None, left_on=None,
              right_on=None, suffixes=("_x", "_y"), copy=None, **kwargs):
        if how == "cross":
            left_on = right_on = []
        elif on is not None:
            if isinstance(on, str):
                on = [on]
            left_on = right_on = list(on)
        else:
            if isinstance(left_on, str):
                left_on = [left_on]
            if isinstance(right_on, str):
                right_on = [right_on]
        if not isinstance(right, BodoDataFrame):
            right = from_pandas_df(right)
        plan = pn.Join(self._plan, right._plan, tuple(left_on), tuple(right_on),
                       how, tuple(suffixes))
        out_cols = _join_out_columns(self._columns, right._columns,
                                     left_on, right_on, suffixes, how)
        return BodoDataFrame(plan, out_cols)

    def groupby(self, by, as_index: bool = True, dropna: bool = True,
                sort: bool = False, observed=True):
        from .groupby import DataFrameGroupBy

        if isinstance(by, str):
            by = [by]
        return DataFrameGroupBy(self, list(by), as_index=as_index,
                                dropna=dropna, sort=sort)

    def sort_values(self, by, ascending=True, na_position="last",
                    ignore_index=False, **kwargs) -> "BodoDataFrame":
        if isinstance(by, str):
            by = [by]
        if isinstance(ascending, bool):
            ascending = [ascending] * len(by)
        plan = pn.Sort(self._plan, tuple(by), tuple(ascending), na_position)
        return BodoDataFrame(plan, self._columns)

    def drop_duplicates(self, subset=None, keep="first", **kwargs):
        if isinstance(subset, str):
            subset = [subset]
        plan = pn.Distinct(self._plan, tuple(subset) if subset else None, keep)
        return BodoDataFrame(plan, self._columns)

    def nlargest(self, n, columns, keep="first"):
        if isinstance(columns, str):
            columns = [columns]
        plan = pn.Limit(pn.Sort(self._plan, tuple(columns),
                                tuple([False] * len(columns))), n)
        return BodoDataFrame(plan, self._columns)

    def nsmallest(self, n, columns, keep="first"):
        if isinstance(columns, str):
            columns = [columns]
        plan = pn.Limit(pn.Sort(self._plan, tuple(columns),
                                tuple([True] * len(columns))), n)
        return BodoDataFrame(plan, self._columns)

    def sample(self, n=None, frac=None, random_state=None, **kwargs):
        plan = pn.Sample(self._plan, n, frac, random_state)
        return BodoDataFrame(plan, self._columns)

    def map_partitions(self, func, *args) -> "BodoDataFrame":
        plan = pn.MapPartitions(self._plan, func, tuple(args))
        # run on empty frame to infer schema
        try:
            probe = func(pd.DataFrame(columns=self._columns), *args)
            names = list(probe.columns)
        except Exception:
            names = list(self._columns)
        plan = pn.MapPartitions(self._plan, func, tuple(args), tuple(names))
        return BodoDataFrame(plan, names)

    def apply(self, func, axis=0, args=(), **kwargs):
        if axis in (1, "columns"):
            def _part(pdf, *a):
                res = pdf.apply(func, axis=1, args=a, **kwargs)
                if isinstance(res, pd.Series):
                    res = res.to_frame(name="0")
                return res

            plan = pn.MapPartitions(self._plan, _part, tuple(args), ("0",))
            from .series import BodoSeries

            return BodoSeries(plan, ColRef("0"), None)
        return self._fallback("apply", func, axis=axis, args=args, **kwargs)

    def isna(self):
        from ..plan.expr import IsNull

        names = list(self._columns)
        exprs = tuple(IsNull(ColRef(c)) for c in names)
        return BodoDataFrame(pn.Projection(self._plan, tuple(names), exprs), names)

    def notna(self):
        from ..plan.expr import IsNull

        names = list(self._columns)
        exprs = tuple(IsNull(ColRef(c), negate=True) for c in names)
        return BodoDataFrame(pn.Projection(self._plan, tuple(names), exprs), names)

    def dropna(self, subset=None, how="any", **kwargs):
        from ..plan.expr import IsNull

        cols = list(subset) if subset is not None else list(self._columns)
        conds = [IsNull(ColRef(c), negate=True) for c in cols]
        out = conds[0]
        for c in conds[1:]:
            out = BoolOp("and" if how == "any" else "or", out, c)
        if how == "all":
            # keep row if any column non-null
            out = conds[0]
            for c in conds[1:]:
                out = BoolOp("or", out, c)
        return BodoDataFrame(pn.Filter(self._plan, out), self._columns)

    def astype(self, dtype) -> "BodoDataFrame":
        from .series import _pd_dtype_to_bodo
        from ..plan.expr import Cast

        per_col = dtype if isinstance(dtype, dict) else \
            {c: dtype for c in self._columns}
        exprs = tuple(
            Cast(ColRef(c), _pd_dtype_to_bodo(per_col[c]))
            if c in per_col else ColRef(c) for c in self._columns)
        return BodoDataFrame(
            pn.Projection(self._plan, tuple(self._columns), exprs),
            list(self._columns))

    def fillna(self, value) -> "BodoDataFrame":
        from ..plan.expr import Case, IsNull, as_expr

        per_col = value if isinstance(value, dict) else \
            {c: value for c in self._columns}
        exprs = tuple(
            Case((IsNull(ColRef(c)),), (as_expr(per_col[c]),), ColRef(c))
            if c in per_col else ColRef(c) for c in self._columns)
        return BodoDataFrame(
            pn.Projection(self._plan, tuple(self._columns), exprs),
            list(self._columns))

    def replace(self, to_replace=None, value=None) -> "BodoDataFrame":
        """Per-column value replacement (scalar or mapping applied to every
        column; reference: frame replace overloads)."""
        exprs = tuple(self[c].replace(to_replace, value)._expr
                      for c in self._columns)
        return BodoDataFrame(
            pn.Projection(self._plan, tuple(self._columns), exprs),
            list(self._columns))

    def round(self, decimals=0) -> "BodoDataFrame":
        from ..plan.expr import RoundExpr

        per = decimals if isinstance(decimals, dict) else             {c: decimals for c in self._columns}
        exprs = tuple(RoundExpr(ColRef(c), int(per[c])) if c in per
                      else ColRef(c) for c in self._columns)
        return BodoDataFrame(
            pn.Projection(self._plan, tuple(self._columns), exprs),
            list(self._columns))

    def value_counts(self, subset=None, ascending=False):
        cols = list(subset) if subset is not None else list(self._columns)
        g = self.groupby(cols, as_index=False).agg(
            count=pd.NamedAgg(cols[0], "size")).to_pandas()
        g = g.sort_values("count", ascending=ascending)
        return g.set_index(cols)["count"]

    def nunique(self):
        return pd.Series({c: self[c].nunique() for c in self._columns})

    def describe(self):
        from ..engine import api

        pdf_head = api.collect(pn.Limit(self._plan, 1))
        num = [c for c in self._columns
               if pd.api.types.is_numeric_dtype(pdf_head[c].dtype)
               and not pd.api.types.is_bool_dtype(pdf_head[c].dtype)]
        return pd.DataFrame({c: self[c].describe() for c in num})

    def query(self, expr: str, **kwargs) -> "BodoDataFrame":
        """pandas query-string filter: the expression is rewritten to the
        operator form (and->&, or->|, not->~) and evaluated lazily against
        this frame's columns (reference: frame.py query)."""
        import ast as _ast

        tree = _ast.parse(expr, mode="eval")

        class _Rw(_ast.NodeTransformer):
            def visit_BoolOp(self, node):
                self.generic_visit(node)
                op = _ast.BitAnd() if isinstance(node.op, _ast.And) \
                    else _ast.BitOr()
                out = node.values[0]
                for v in node.values[1:]:
                    out = _ast.BinOp(left=out, op=op, right=v)
                return out

            def visit_UnaryOp(self, node):
                self.generic_visit(node)
                if isinstance(node.op, _ast.Not):
                    return _ast.UnaryOp(op=_ast.Invert(), operand=node.operand)
                return node

        new = _ast.fix_missing_locations(_Rw().visit(tree))
        ns = {c: self[c] for c in self._columns if c.isidentifier()}
        ns.update(kwargs.get("local_dict") or {})
        mask = eval(compile(new, "<query>", "eval"), {"__builtins__": {}}, ns)
        return self[mask]

    @property
    def iloc(self):
        return _ILoc(self)

    def corrwith(self, other):
        """Per-column Pearson correlation with a series (distributed
        pairwise co-moments)."""
        import pandas as _pd

        out = {}
        head = self.head(1).to_pandas()
        for c in self._columns:
            if _pd.api.types.is_numeric_dtype(head[c].dtype):
                out[c] = float(self[c].corr(other))
        return _pd.Series(out)

    def rolling(self, window, min_periods=None, center=False, **kwargs):
        if isinstance(window, str):
            return _RollingHost(self, window, min_periods)
        return _RollingFrame(self, int(window), min_periods, center)

    def filter(self, items=None, like=None, regex=None, axis=None):
        """Column-label filtering (reference: frame.py filter)."""
        if items is not None:
            keep = [c for c in self._columns if c in set(items)]
        elif like is not None:
            keep = [c for c in self._columns if like in c]
        elif regex is not None:
            import re as _re

            pat = _re.compile(regex)
            keep = [c for c in self._columns if pat.search(c)]
        else:
            raise TypeError("must pass items, like, or regex")
        return self[keep]

    def get(self, key, default=None):
        if isinstance(key, str):
            return self[key] if key in self._columns else default
        return self[key]

    def set_index(self, keys, drop=True, append=False) -> "BodoDataFrame":
        """Mark columns as the frame's (Multi)Index (reference:
        frame.py set_index).  The index columns stay physical in the
        distributed table; collection re-applies them as the pandas index.
        Lazy ops on the frame see them as regular columns."""
        cols = [keys] if isinstance(keys, str) else list(keys)
        missing = [c for c in cols if c not in self._columns]
        if missing:
            raise KeyError(missing)
        out = BodoDataFrame(self._plan, self._columns)
        prev = list(getattr(self, "_index_cols", []) or []) if append else []
        object.__setattr__(out, "_index_cols", prev + cols)
        object.__setattr__(out, "_result", self._result)
        return out

    def reset_index(self, drop=False, **kwargs):
        """Positional index restored; a set_index marker either rejoins the
        columns (drop=False) or is projected away (drop=True)."""
        idx = list(getattr(self, "_index_cols", []) or [])
        if not idx:
            if drop:
                return self
            return self._fallback("reset_index", drop=drop, **kwargs)
        if drop:
            keep = [c for c in self._columns if c not in idx]
            exprs = tuple(ColRef(c) for c in keep)
            return BodoDataFrame(pn.Projection(self._plan, tuple(keep),
                                               exprs), keep)
        order = idx + [c for c in self._columns if c not in idx]
        exprs = tuple(ColRef(c) for c in order)
        return BodoDataFrame(pn.Projection(self._plan, tuple(order), exprs),
                             order)

    def sort_index(self, ascending=True, **kwargs):
        idx = list(getattr(self, "_index_cols", []) or [])
        if idx:
            out = self.sort_values(idx, ascending=ascending)
            object.__setattr__(out, "_index_cols", idx)
            return out
        return self

    @property
    def index(self):
        idx = list(getattr(self, "_index_cols", []) or [])
        if idx:
            pdf = self[idx].to_pandas()
            if len(idx) == 1:
                return pd.Index(pdf[idx[0]], name=idx[0])
            return pd.MultiIndex.from_frame(pdf[idx])
        return pd.RangeIndex(len(self))

    def explain(self, optimized: bool = True) -> str:
        """Pretty-print this frame's logical plan (reference: plan dumps at
        tracing_level>=2, bodo/pandas/plan.py:1090-1096)."""
        plan = self._plan
        if optimized:
            from ..engine.optimizer import optimize

            plan = optimize(plan)
        return pn.explain(plan)

    def explode(self, column: str) -> "BodoDataFrame":
        """pandas explode over LIST columns (reference: frame explode /
        _lateral.cpp FLATTEN): runs distributed, row-local."""
        if isinstance(column, list):
            if len(column) != 1:
                return self._fallback("explode", column)
            column = column[0]
        return BodoDataFrame(pn.Explode(self._plan, column), self._columns)

    def melt(self, id_vars=None, value_vars=None, var_name="variable",
             value_name="value") -> "BodoDataFrame":
        """Wide-to-long unpivot as a lazy plan: one projection per value
        column (variable = constant) unioned together — stays distributed
        (reference: frame.py melt)."""
        from ..plan.expr import Const

        ids = list(id_vars) if id_vars is not None else []
        if isinstance(id_vars, str):
            ids = [id_vars]
        vals = list(value_vars) if value_vars is not None else \
            [c for c in self._columns if c not in ids]
        if isinstance(value_vars, str):
            vals = [value_vars]
        names = tuple(ids + [var_name, value_name])
        parts = []
        for v in vals:
            exprs = tuple([ColRef(c) for c in ids]
                          + [Const(v), ColRef(v)])
            parts.append(pn.Projection(self._plan, names, exprs))
        plan = parts[0] if len(parts) == 1 else pn.Union(tuple(parts), False)
        return BodoDataFrame(plan, list(names))

    def pivot_table(self, values=None, index=None, columns=None,
                    aggfunc="mean"):
        """Distributed groupby([index, columns]) then a host pivot of the
        (small) aggregated result (reference: frame.py pivot_table)."""
        assert index is not None
        idx = [index] if isinstance(index, str) else list(index)
        if columns is None:
            # no columns: a grouped aggregate indexed by `index`
            vals0 = ([values] if isinstance(values, str) else
                     list(values) if values is not None else
                     [c for c in self._columns if c not in idx])
            small = self.groupby(idx, as_index=False).agg(
                **{v: pd.NamedAgg(v, aggfunc) for v in vals0})
            pdf = small.to_pandas()
            for c in pdf.columns:
                if isinstance(pdf[c].dtype, pd.CategoricalDtype):
                    pdf[c] = pdf[c].astype(object)
            return pdf.set_index(idx).sort_index()
        cols = [columns] if isinstance(columns, str) else list(columns)
        if values is None:
            values = [c for c in self._columns
                      if c not in idx + cols]
        vals = [values] if isinstance(values, str) else list(values)
        agg_kwargs = {v: (v, aggfunc) for v in vals}
        small = self.groupby(idx + cols, as_index=False).agg(
            **{k: pd.NamedAgg(c, f) for k, (c, f) in agg_kwargs.items()})
        pdf = small.to_pandas()
        for c in pdf.columns:
            if isinstance(pdf[c].dtype, pd.CategoricalDtype):
                pdf[c] = pdf[c].astype(object)
        out_vals = values if isinstance(values, str) else vals
        return pdf.pivot_table(values=out_vals, index=idx, columns=cols,
                               aggfunc="first")

    # ------------------------------------------------------------------
    # IO
    # ------------------------------------------------------------------
    def to_parquet(self, path: str, compression="snappy",
                   partition_cols=None, **kwargs):
        from ..engine import api

        plan = pn.ParquetWrite(self._plan, path, compression,
                               tuple(partition_cols or ()))
        api.materialize(plan)

    def to_iceberg(self, path: str, mode: str = "create", **kwargs):
        """Transactional snapshot write to a filesystem Iceberg table
        (reference: frame.py to_iceberg; metadata is committed only after
        every rank's data files land)."""
        from ..engine import api

        plan = pn.IcebergWrite(self._plan, path, mode)
        api.materialize(plan)

    def to_csv(self, path=None, **kwargs):
        pdf = self.to_pandas()
        return pdf.to_csv(path, index=False, **kwargs)

    def to_json(self, path=None, orient="records", lines=True, **kwargs):
        pdf = self.to_pandas()
        return pdf.to_json(path, orient=orient, lines=lines, **kwargs)

    def to_sql(self, name, con, if_exists="fail", **kwargs):
        """Write through a DB-API connection or sqlite path (reference:
        to_sql via the Snowflake writer; gathered then written by rank 0)."""
        import sqlite3

        from ..parallel import comm

        pdf = self.to_pandas()
        if comm.get_rank() != 0:
            return
        close = None
        if isinstance(con, str):
            p = con[len("sqlite://"):] if con.startswith("sqlite://") else con
            con = sqlite3.connect(p)
            close = con
        try:
            pdf.to_sql(name, con, if_exists=if_exists, index=False, **kwargs)
            if close is not None:
                con.commit()
        finally:
            if close is not None:
                close.close()

    def corr(self, numeric_only=True) -> pd.DataFrame:
        """Pairwise Pearson correlation of numeric columns via distributed
        co-moment reductions."""
        from ..engine import api

        head = api.collect(pn.Limit(self._plan, 1))
        num = [c for c in self._columns
               if pd.api.types.is_numeric_dtype(head[c].dtype)
               and not pd.api.types.is_bool_dtype(head[c].dtype)]
        out = pd.DataFrame(np.eye(len(num)), index=num, columns=num)
        for i, a in enumerate(num):
            for b2 in num[i + 1:]:
                v = self[a].corr(self[b2])
                out.loc[a, b2] = v
                out.loc[b2, a] = v
        return out

    def memory_usage(self, index=False, deep=False) -> pd.Series:
        tbl = self.execute()
        return pd.Series({n: c.nbytes() for n, c in
                          zip(tbl.names, tbl.columns)})

    def agg(self, arg):
        """Frame-level eager aggregation: {'col': 'func'} or
        {'col': ['f1', 'f2']}."""
        if isinstance(arg, dict):
            out = {}
            for col, f in arg.items():
                if isinstance(f, (list, tuple)):
                    out[col] = {x: getattr(self[col], x)() for x in f}
                else:
                    out[col] = getattr(self[col], f)()
            if all(not isinstance(v, dict) for v in out.values()):
                return pd.Series(out)
            return pd.DataFrame(out)
        if isinstance(arg, str):
            return self._frame_reduce(arg)
        raise NotImplementedError(f"df.agg({arg!r})")

    def _frame_reduce(self, func):
        from ..engine import api

        head = api.collect(pn.Limit(self._plan, 1))
        num = [c for c in self._columns
               if pd.api.types.is_numeric_dtype(head[c].dtype)]
        plan = pn.Reduce(self._plan, tuple((c, c, func) for c in num))
        row = api.collect(plan)
        return pd.Series({c: row[c].iloc[0] for c in num})

    def sum(self, numeric_only=True):
        return self._frame_reduce("sum")

    def mean(self, numeric_only=True):
        return self._frame_reduce("mean")

    def min(self, numeric_only=True):
        return self._frame_reduce("min")

    def max(self, numeric_only=True):
        return self._frame_reduce("max")

    def std(self, numeric_only=True, ddof=1):
        return self._frame_reduce("std")

    def var(self, numeric_only=True, ddof=1):
        return self._frame_reduce("var")

    # reductions over the whole frame fall back (rare)
    def count(self):
        return self.to_pandas().count()

    def copy(self, deep=True):
        return BodoDataFrame(self._plan, list(self._columns))


def _join_out_columns(lcols, rcols, left_on, right_on, suffixes, how):
    shared_keys = [k for k, rk in zip(left_on, right_on) if k == rk]
    out = []
    rset, lset = set(rcols), set(lcols)
    for c in lcols:
        if c in rset and c not in shared_keys:
            out.append(c + suffixes[0])
        else:
            out.append(c)
    for c in rcols:
        if c in shared_keys:
            continue
        if c in lset:
            out.append(c + suffixes[1])
        else:
            out.append(c)
    return out


def wrap_result(res):
    if isinstance(res, pd.DataFrame):
        if isinstance(res.index, pd.RangeIndex):
            return from_pandas_df(res)
        return res  # keep meaningful indexes (describe(), set_index results)
    return res


def from_pandas_df(df: pd.DataFrame) -> BodoDataFrame:
    key = ex.register_object(df.reset_index(drop=True))
    plan = pn.PandasScan(key, tuple(df.columns), distributed=False)
    return BodoDataFrame(plan, list(df.columns))


class _RollingFrame:
    """df.rolling(w): per-column rolling aggregation (Rolling plan node with
    distributed halo exchange; reference: hiframes rolling).  center=True
    relabels the trailing window onto its middle row (a -(w//2) shift)."""

    _FUNCS = ("sum", "mean", "min", "max", "count", "std", "var", "median")

    def __init__(self, frame: BodoDataFrame, window: int, min_periods,
                 center: bool = False):
        self._frame = frame
        self._window = window
        self._min_periods = min_periods
        self._center = center

    def _agg(self, func):
        from ..engine import api

        head = api.collect(pn.Limit(self._frame._plan, 1))
        cols = [c for c in self._frame._columns
                if pd.api.types.is_numeric_dtype(head[c].dtype)]
        specs = tuple((c, c, func) for c in cols)
        plan = pn.Rolling(self._frame._plan, self._window, self._min_periods,
                          specs)
        if self._center and self._window > 1:
            plan = pn.Shift(plan, -(self._window // 2),
                            tuple((c, c) for c in cols))
        return BodoDataFrame(plan, list(cols))

    def __getattr__(self, name):
        if name in self._FUNCS:
            return lambda: self._agg(name)
        raise AttributeError(name)


class _RollingHost:
    """Offset-string windows ('1D', '2h'): host pandas on the gathered
    frame (time-based windows need the timestamp index; replicated result,
    exact)."""

    _FUNCS = ("sum", "mean", "min", "max", "count", "std", "var", "median")

    def __init__(self, frame: BodoDataFrame, window: str, min_periods):
        self._frame = frame
        self._window = window
        self._min_periods = min_periods

    def _agg(self, func):
        pdf = self._frame.to_pandas()
        r = pdf.rolling(self._window, min_periods=self._min_periods or 1)
        return getattr(r, func)()

    def __getattr__(self, name):
        if name in self._FUNCS:
            return lambda: self._agg(name)
        raise AttributeError(name)


class _ILoc:
    """df.iloc[:n] / df.iloc[:n, :] head-style slicing stays lazy (Limit
    plan); anything else materializes through the pandas fallback."""

    def __init__(self, frame: BodoDataFrame):
        self._frame = frame

    def __getitem__(self, key):
        rows = key[0] if isinstance(key, tuple) else key
        if isinstance(rows, slice) and rows.start in (None, 0) and \
                rows.step in (None, 1) and rows.stop is not None and \
                rows.stop >= 0:
            out = BodoDataFrame(pn.Limit(self._frame._plan, rows.stop),
                                list(self._frame._columns))
            if isinstance(key, tuple) and len(key) == 2:
                cols = key[1]
                if isinstance(cols, slice) and cols == slice(None):
                    return out
                if isinstance(cols, list):
                    names = [self._frame._columns[i] for i in cols] \
                        if all(isinstance(i, int) for i in cols) else cols
                    return out[names]
                if isinstance(cols, int):
                    return out[self._frame._columns[cols]]
            return out
        return self._frame.to_pandas().iloc[key]


class _LocIndexer:
    """df.loc[:, [cols]] and df.loc[boolean_series] support."""

    def __init__(self, frame: BodoDataFrame):
        self._frame = frame

    def __getitem__(self, key):
        from .series import BodoSeries

        if isinstance(key, tuple) and len(key) == 2:
            rows, cols = key
            out = self._frame
            if isinstance(rows, BodoSeries):
                out = out[rows]
            elif not (isinstance(rows, slice) and rows == slice(None)):
                raise NotImplementedError("loc row selection")
            if isinstance(cols, list):
                return out[cols]
            if isinstance(cols, str):
                return out[cols]
            if isinstance(cols, slice) and cols == slice(None):
                return out
            raise NotImplementedError("loc column selection")
        if isinstance(key, BodoSeries):
            return self._frame[key]
        raise NotImplementedError("loc")
