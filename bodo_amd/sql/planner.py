"""SQL AST -> logical plan (reference: BodoSQL plan_conversion.py converting
Calcite RelNodes into bodo.pandas LazyPlan; here the parser AST maps straight
onto our plan nodes)."""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import pandas as pd

from ..core import types as bt
from ..plan import expr as ex
from ..plan import nodes as pn
from . import parser as ast

AGG_FUNCS = {"sum": "sum", "avg": "mean", "count": "count", "min": "min",
             "max": "max", "stddev": "std", "variance": "var",
             "median": "median", "approx_count_distinct": "approx_nunique",
             "array_agg": "array_agg",
             "mode": "mode", "kurtosis": "kurt", "skew": "skew",
             "stddev_samp": "std", "var_samp": "var", "any_value": "first",
             "booland_agg": "all", "boolor_agg": "any"}

CAST_TYPES = {
    "int": bt.int64, "integer": bt.int64, "bigint": bt.int64,
    "smallint": bt.int16, "double": bt.float64, "float": bt.float64,
    "real": bt.float32, "decimal": bt.float64, "numeric": bt.float64,
    "varchar": bt.string, "char": bt.string, "text": bt.string,
    "date": bt.date32, "timestamp": bt.timestamp_ns, "boolean": bt.boolean,
}


class Scope:
    """Maps (alias, column) -> internal plan column name; `parent` links a
    subquery scope to the outer query for correlation detection."""

    def __init__(self, parent: Optional["Scope"] = None):
        self.entries: List[Tuple[str, str, str]] = []  # alias, col, internal
        self.parent = parent

    def add(self, alias: str, col: str, internal: str):
        self.entries.append((alias, col, internal))

    def resolve_local(self, table: Optional[str], name: str) -> Optional[str]:
        cands = []
        for alias, col, internal in self.entries:
            if col.lower() == name.lower() and (table is None
                                                or alias.lower() == table.lower()):
                cands.append(internal)
        if not cands:
            return None
        if len(set(cands)) > 1:
            raise KeyError(f"ambiguous column {name}")
        return cands[0]

    def resolve(self, table: Optional[str], name: str) -> str:
        r = self.resolve_local(table, name)
        if r is not None:
            return r
        if self.parent is not None:
            return self.parent.resolve(table, name)
        raise KeyError(f"unknown column {table + '.' if table else ''}{name}")

    def is_outer(self, table: Optional[str], name: str) -> bool:
        """True when the column resolves only in an enclosing scope."""
        if self.resolve_local(table, name) is not None:
            return False
        return self.parent is not None

    def tables_of(self, e) -> set:
        """Set of internal columns referenced by an AST expression."""
        cols = set()

        def walk(x):
            if isinstance(x, ast.Col):
                cols.add(self.resolve(x.table, x.name))
            for f in getattr(x, "__dataclass_fields__", {}):
                v = getattr(x, f)
                if isinstance(v, (list, tuple)):
                    for i in v:
                        if hasattr(i, "__dataclass_fields__"):
                            walk(i)
                        elif isinstance(i, tuple):
                            for j in i:
                                if hasattr(j, "__dataclass_fields__"):
                                    walk(j)
                elif hasattr(v, "__dataclass_fields__"):
                    walk(v)

        walk(e)
        return cols


class Planner:
    def __init__(self, tables: Dict[str, "object"]):
        self.tables = {k.lower(): v for k, v in tables.items()}
        self._counter = 0
        # WITH-clause views: name -> AST, planned lazily on first reference
        # and cached so multiple references share one plan subtree
        self.cte_asts: Dict[str, object] = {}
        self._cte_cache: Dict[str, tuple] = {}

    def _uniq(self, base: str) -> str:
        self._counter += 1
        return f"{base}__{self._counter}"

    # ------------------------------------------------------------------
    def plan(self, q):
        for name, sub in (getattr(q, "ctes", None) or ()):
            self.cte_asts[name] = sub
        if isinstance(q, ast.SetOpQ):
            return self._plan_setop(q)
        scope = Scope()
        plan = None
        joined_cols: set = set()
        where_conjuncts = _split_conjuncts(q.where) if q.where else []
        used_conjuncts = [False] * len(where_conjuncts)

        refs = ([q.table] if q.table else []) + [j.table for j in q.joins]
        join_kinds = ["base"] + [j.kind for j in q.joins]
        join_ons = [None] + [j.on for j in q.joins]

        for idx, tr in enumerate(refs):
            if getattr(tr, "flatten", None) is not None:
                # , LATERAL FLATTEN(input => expr) f  ->  Explode over the
                # accumulated plan (reference: _lateral.cpp FLATTEN)
                assert plan is not None, "LATERAL FLATTEN needs a base table"
                vcol = self._uniq("__flat_v")
                icol = self._uniq("__flat_i")
                keep = sorted(joined_cols)
                exprs = tuple(ex.ColRef(c) for c in keep) + (
                    self.expr(tr.flatten, scope),)
                plan = pn.Projection(plan, tuple(keep) + (vcol,), exprs)
                plan = pn.Explode(plan, vcol, pos=icol)
                alias = tr.alias or "f"
                scope.add(alias, "value", vcol)
                scope.add(alias, "index", icol)
                joined_cols |= {vcol, icol}
                continue
            sub_plan, sub_cols = self._table_plan(tr, scope)
            if plan is None:
                plan = sub_plan
                joined_cols |= sub_cols
                continue
            kind = join_kinds[idx]
            on = join_ons[idx]
            eq_pairs: List[Tuple[str, str]] = []
            post: List[ast.Bin] = []
            if on is not None:
                for c in _split_conjuncts(on):
                    pair = self._equi_pair(c, scope, joined_cols, sub_cols)
                    if pair:
                        eq_pairs.append(pair)
                        continue
                    refs_c = {scope.resolve(x.table, x.name)
                              for x in _col_refs(c)}
                    if kind in ("left", "inner") and refs_c <= sub_cols:
                        # condition touches only the new (right) table:
                        # pre-filter it (required for LEFT join semantics)
                        sub_plan = pn.Filter(sub_plan, self.expr(c, scope))
                    elif kind == "inner":
                        post.append(c)
                    else:
                        raise NotImplementedError(
                            "outer-join ON condition across both sides")
            if kind == "cross" or not eq_pairs:
                # pull applicable equi conditions from WHERE (FROM a, b style)
                for ci, c in enumerate(where_conjuncts):
                    if used_conjuncts[ci]:
                        continue
                    pair = self._equi_pair(c, scope, joined_cols, sub_cols)
                    if pair:
                        eq_pairs.append(pair)
                        used_conjuncts[ci] = True
                if eq_pairs:
                    kind = "inner" if kind == "cross" else kind
            if eq_pairs:
                lks = tuple(p[0] for p in eq_pairs)
                rks = tuple(p[1] for p in eq_pairs)
                plan = pn.Join(plan, sub_plan, lks, rks,
                               kind if kind != "base" else "inner")
            else:
                plan = pn.Join(plan, sub_plan, (), (), "cross")
            joined_cols |= sub_cols
            for c in post:
                plan = pn.Filter(plan, self.expr(c, scope))
        if plan is None:
            raise ValueError("SELECT without FROM not supported")
        # remaining WHERE: subquery predicates become plan transforms
        rest = [c for ci, c in enumerate(where_conjuncts) if not used_conjuncts[ci]]
        plain = []
        for c in rest:
            handled, plan = self._apply_subquery_pred(c, plan, scope)
            if not handled:
                plain.append(c)
        if plain:
            cond = plain[0]
            for c in plain[1:]:
                cond = ast.Bin("and", cond, c)
            plan = pn.Filter(plan, self.expr(cond, scope))

        # correlated scalar subqueries in the SELECT list decorrelate into
        # LEFT JOINs before projection planning
        plan = self._rewrite_select_subqueries(plan, scope, q)

        if q.group_by == "ALL":  # Snowflake GROUP BY ALL: non-agg items
            q.group_by = [it.expr for it in q.items
                          if not it.star and not _has_agg(it.expr)]
        # GROUP BY ordinal (1-based) / SELECT alias; HAVING may also use
        # aliases (Snowflake scoping)
        alias_map = {it.alias.lower(): it.expr for it in q.items
                     if it.alias and not it.star}
        if isinstance(q.group_by, list) and q.group_by:
            def norm_gb(e):
                if isinstance(e, ast.Lit) and e.kind == "num":
                    return q.items[int(e.value) - 1].expr
                if isinstance(e, ast.Col) and e.table is None \
                        and e.name.lower() in alias_map:
                    try:
                        scope.resolve(None, e.name)
                        return e  # a real column of that name wins
                    except KeyError:
                        return alias_map[e.name.lower()]
                return e

            q.group_by = [norm_gb(e) for e in q.group_by]
        if q.having is not None and alias_map:
            q.having = _sub_aliases(q.having, alias_map, scope)

        # ------------------------------------------------- aggregation
        has_agg = any(_has_agg(it.expr) for it in q.items if not it.star) \
            or (q.having is not None and _has_agg(q.having)) or q.group_by \
            or (getattr(q, "qualify", None) is not None
                and _has_agg(q.qualify))
        out_names: List[str] = []
        out_exprs: List[ex.Expr] = []
        if has_agg and getattr(q, "grouping_sets", None) is not None:
            plan, out_names = self._plan_grouping_sets(plan, scope, q)
        elif has_agg:
            plan, scope2, key_map, agg_map = self._aggregate(
                plan, scope, q)
            if any(_has_window(it.expr) for it in q.items if not it.star) \
                    or getattr(q, "qualify", None) is not None:
                plan = self._plan_windows_post_agg(plan, scope, q, key_map,
                                                   agg_map)
            # build output projection over the agg result
            for it in q.items:
                if it.star:
                    raise ValueError("SELECT * with GROUP BY not supported")
                name = it.alias or _default_name(it.expr)
                out_names.append(name)
                out_exprs.append(self._post_agg_expr(it.expr, key_map, agg_map,
                                                     scope))
            if q.having is not None:
                plan = pn.Filter(plan, self._post_agg_expr(
                    q.having, key_map, agg_map, scope))
            proj = pn.Projection(plan, tuple(out_names), tuple(out_exprs))
            plan = proj
            order_scope = dict(zip(out_names, out_names))
            order_resolver = lambda e: self._post_agg_expr(
                e, key_map, agg_map, scope) if not isinstance(e, ast.Col) \
                or (e.table is None and e.name in out_names and False) else None
        else:
            if any(_has_window(it.expr) for it in q.items if not it.star) \
                    or (getattr(q, "qualify", None) is not None):
                plan = self._plan_windows(plan, scope, q)
            for it in q.items:
                if it.star:
                    want = getattr(it, "star_table", None)
                    excl = set(getattr(it, "exclude", ()) or ())
                    for alias, col, internal in scope.entries:
                        if want is not None and \
                                alias.lower() != want.lower():
                            continue
                        if col.lower() in excl:
                            continue
                        out_names.append(col)
                        out_exprs.append(ex.ColRef(internal))
                    continue
                name = it.alias or _default_name(it.expr)
                out_names.append(name)
                out_exprs.append(self.expr(it.expr, scope))
            plan = pn.Projection(plan, tuple(out_names), tuple(out_exprs))
        if q.distinct:
            plan = pn.Distinct(plan, None)
        if q.order_by == "ALL":  # Snowflake ORDER BY ALL: every output col
            q.order_by = [(ast.Col(None, n), True) for n in out_names]
        if q.order_by:
            keys, asc = [], []
            hidden: List[str] = []
            for e, a in q.order_by:
                if isinstance(e, ast.Lit) and e.kind == "num":
                    keys.append(out_names[int(e.value) - 1])
                elif isinstance(e, ast.Col) and e.table is None \
                        and e.name in out_names:
                    keys.append(e.name)
                elif isinstance(e, ast.Col):
                    # ORDER BY a source column not in the SELECT list:
                    # carry it as a hidden projection column through the
                    # sort and drop it afterwards (standard SQL)
                    internal = None
                    if not has_agg and not q.distinct and \
                            isinstance(plan, pn.Projection):
                        try:
                            internal = scope.resolve(e.table, e.name)
                        except KeyError:
                            internal = None
                    if internal is not None and internal not in hidden:
                        hidden.append(internal)
                        keys.append(internal)
                    elif internal is not None:
                        keys.append(internal)
                    else:
                        keys.append(e.name)  # hope it's an output name
                else:
                    raise ValueError("ORDER BY expressions must be output "
                                     "columns or positions")
                asc.append(a)
            if hidden:
                plan = pn.Projection(
                    plan.child, plan.names + tuple(hidden),
                    plan.exprs + tuple(ex.ColRef(h) for h in hidden))
            plan = pn.Sort(plan, tuple(keys), tuple(asc))
        if q.limit is not None:
            plan = pn.Limit(plan, q.limit,
                            getattr(q, "limit_offset", 0) or 0)
        if q.order_by and hidden:
            plan = pn.Projection(plan, tuple(out_names),
                                 tuple(ex.ColRef(n) for n in out_names))
        return plan, out_names

    # ------------------------------------------------------------------
    def _plan_windows(self, plan, scope: Scope, q: ast.Query):
        """OVER-clause planning (reference: BodoSQL window RexOver lowering
        onto bodo/libs/window/_window_calculator.cpp): each WindowE in the
        select list is replaced by a reference to a Window plan-node output
        column; windows sharing (partition, order) run in one node."""
        pre_names = [e[2] for e in scope.entries]
        pre_exprs = {n: ex.ColRef(n) for n in pre_names}
        need_pre = False

        def as_col(e, base):
            nonlocal need_pre
            if isinstance(e, ast.Col):
                return scope.resolve(e.table, e.name)
            name = self._uniq(base)
            pre_exprs[name] = self.expr(e, scope)
            pre_names.append(name)
            need_pre = True
            return name

        groups: Dict[tuple, list] = {}  # (keys, order, asc) -> specs

        def register(w: ast.WindowE) -> str:
            nonlocal need_pre
            keys = tuple(as_col(pe, "__wk") for pe in w.partition_by)
            if not keys:
                kname = self._uniq("__wk")
                pre_exprs[kname] = ex.Const(1)
                pre_names.append(kname)
                need_pre = True
                keys = (kname,)
            order = tuple(as_col(oe, "__wo") for oe, _ in w.order_by)
            asc = tuple(a for _, a in w.order_by)
            out = self._uniq("__win")
            fn = w.func
            arg_col = ""
            if w.args and not w.star:
                arg_col = as_col(w.args[0], "__wa")
            if fn == "row_number":
                spec = (out, "", "row_number", None)
            elif fn in ("first_value", "last_value"):
                if getattr(w, "ignore_nulls", False):
                    spec = (out, arg_col, fn + "_ig", None)
                else:
                    spec = (out, arg_col, fn, None)
            elif fn == "nth_value":
                k = int(w.args[1].value) if len(w.args) > 1 else 1
                spec = (out, arg_col, "nth_value", k)
            elif fn == "ntile":
                k = int(w.args[0].value) if w.args else 1
                spec = (out, "", "ntile", k)
            elif fn in ("rank", "dense_rank", "percent_rank",
                        "cume_dist"):
                spec = (out, "", fn, None)
            elif fn in ("lag", "lead"):
                n = 1
                if len(w.args) > 1:
                    n = _lit_int(w.args[1])
                k = n if fn == "lag" else -n
                if len(w.args) > 2:  # LAG(x, n, default)
                    dflt = w.args[2].value if isinstance(
                        w.args[2], ast.Lit) else _lit_int(w.args[2])
                    spec = (out, arg_col, "shift", (k, dflt))
                else:
                    spec = (out, arg_col, "shift", k)
            elif fn in AGG_FUNCS or fn == "count":
                frame = getattr(w, "frame_preceding", None)
                if order and frame is not None and frame >= 0:
                    # ROWS BETWEEN n PRECEDING AND CURRENT ROW
                    base = AGG_FUNCS.get(fn, fn)
                    if base not in ("sum", "mean", "min", "max", "count"):
                        raise NotImplementedError(
                            f"{fn.upper()} with a window frame")
                    spec = (out, arg_col, f"rolling_{base}", frame + 1)
                elif order:
                    # default frame (and explicit UNBOUNDED PRECEDING ..
                    # CURRENT ROW): running aggregate along the order
                    if fn == "count" and (w.star or not w.args):
                        spec = (out, "", "row_number", None)
                    else:
                        running = {"sum": "cumsum", "min": "cummin",
                                   "max": "cummax", "avg": "cummean",
                                   "mean": "cummean", "count": "cumcount_v"}
                        base = AGG_FUNCS.get(fn, fn)
                        if base not in running:
                            raise NotImplementedError(
                                f"{fn.upper()} OVER (... ORDER BY) "
                                "not supported")
                        spec = (out, arg_col, running[base], None)
                elif fn == "count" and (w.star or not w.args):
                    spec = (out, "", "transform_size", None)
                elif fn == "count" and w.distinct:
                    spec = (out, arg_col, "transform_nunique", None)
                else:
                    spec = (out, arg_col,
                            f"transform_{AGG_FUNCS.get(fn, fn)}", None)
            else:
                raise NotImplementedError(f"window function {fn}")
            groups.setdefault((keys, order, asc), []).append(spec)
            return out

        def rewrite(e):
            if isinstance(e, ast.WindowE):
                name = register(e)
                scope.add("", name, name)
                return ast.Col(None, name)
            if isinstance(e, (ast.Query, ast.SetOpQ)):
                return e
            for f in getattr(e, "__dataclass_fields__", {}):
                v = getattr(e, f)
                if isinstance(v, list):
                    setattr(e, f, [rewrite(x) if hasattr(
                        x, "__dataclass_fields__") else x for x in v])
                elif hasattr(v, "__dataclass_fields__"):
                    setattr(e, f, rewrite(v))
            return e

        for it in q.items:
            if not it.star:
                it.expr = rewrite(it.expr)
        qual = getattr(q, "qualify", None)
        if qual is not None:
            q.qualify = rewrite(qual)
        if need_pre:
            plan = pn.Projection(
                plan, tuple(pre_names),
                tuple(pre_exprs[n] for n in pre_names))
        for (keys, order, asc), specs in groups.items():
            plan = pn.Window(plan, keys, order, asc, tuple(specs))
        if qual is not None:
            plan = pn.Filter(plan, self.expr(q.qualify, scope))
            q.qualify = None
        return plan

    def _plan_grouping_sets(self, plan, scope: Scope, q: ast.Query):
        """ROLLUP / CUBE / GROUPING SETS: one aggregate per grouping set,
        rolled-up keys project as typed nulls, arms UNION ALL (reference:
        BodoSQL grouping-sets lowering via Calcite Aggregate.groupSets)."""
        import copy

        full_keys = {_ast_key(g) for g in q.group_by}
        gall = None
        arms = []
        out_names: List[str] = []
        for gset in q.grouping_sets:
            q2 = copy.copy(q)
            q2.grouping_sets = None
            q2.group_by = list(gset)
            arm_plan = plan
            if not gset:
                # grand-total arm: constant grouping key, dropped on output
                if gall is None:
                    gall = self._uniq("__gall")
                    scope.add("", gall, gall)
                cols = [e[2] for e in scope.entries if e[2] != gall]
                arm_plan = pn.Projection(
                    plan, tuple(cols + [gall]),
                    tuple([ex.ColRef(c) for c in cols] + [ex.Const(1)]))
                q2.group_by = [ast.Col(None, gall)]
            aplan, scope2, key_map, agg_map = self._aggregate(
                arm_plan, scope, q2)
            if q.having is not None:
                aplan = pn.Filter(aplan, self._post_agg_expr(
                    q.having, key_map, agg_map, scope))
            names, exprs = [], []
            for it in q.items:
                if it.star:
                    raise ValueError("SELECT * with GROUPING SETS")
                name = it.alias or _default_name(it.expr)
                k = _ast_key(it.expr)
                if k in full_keys and k not in key_map:
                    e2 = ex.Const(None, bt.float64)
                else:
                    try:
                        e2 = self._post_agg_expr(it.expr, key_map, agg_map,
                                                 scope)
                    except KeyError:
                        e2 = ex.Const(None, bt.float64)
                names.append(name)
                exprs.append(e2)
            arms.append(pn.Projection(aplan, tuple(names), tuple(exprs)))
            out_names = names
        return pn.Union(tuple(arms), False), out_names

    def _plan_windows_post_agg(self, plan, scope: Scope, q: ast.Query,
                               key_map, agg_map):
        """Window functions over aggregate output (RANK() OVER (ORDER BY
        SUM(v))): _aggregate already computed every aggregate referenced
        inside OVER(...), so partition/order/arg expressions resolve via
        the post-agg maps and the Window node runs on the agg plan."""
        pre_names = list(key_map.values()) + list(agg_map.values())
        pre_exprs = {n: ex.ColRef(n) for n in pre_names}
        need_pre = False
        # Snowflake scope: QUALIFY / OVER may reference SELECT-list aliases
        # (e.g. count(*) AS n ... ORDER BY n) — substitute before resolving
        alias_map = {it.alias.lower(): it.expr for it in q.items
                     if it.alias and not it.star}

        def dealias(e):
            if isinstance(e, ast.Col) and e.table is None \
                    and e.name.lower() in alias_map:
                k = _ast_key(e)
                if k not in key_map and k not in agg_map:
                    return alias_map[e.name.lower()]
            return e

        def conv(e):
            return self._post_agg_expr(dealias(e), key_map, agg_map, scope)

        def as_col(e, base):
            nonlocal need_pre
            e = dealias(e)
            k = _ast_key(e)
            if k in key_map:
                return key_map[k]
            if k in agg_map:
                return agg_map[k]
            name = self._uniq(base)
            pre_exprs[name] = conv(e)
            pre_names.append(name)
            need_pre = True
            return name

        groups: Dict[tuple, list] = {}

        def register(w: ast.WindowE) -> str:
            nonlocal need_pre
            keys = tuple(as_col(pe, "__wk") for pe in w.partition_by)
            if not keys:
                kname = self._uniq("__wk")
                pre_exprs[kname] = ex.Const(1)
                pre_names.append(kname)
                need_pre = True
                keys = (kname,)
            order = tuple(as_col(oe, "__wo") for oe, _ in w.order_by)
            asc = tuple(a for _, a in w.order_by)
            out = self._uniq("__win")
            fn = w.func
            arg_col = ""
            if w.args and not w.star:
                arg_col = as_col(w.args[0], "__wa")
            if fn == "row_number":
                spec = (out, "", "row_number", None)
            elif fn in ("rank", "dense_rank", "percent_rank",
                        "cume_dist"):
                spec = (out, "", fn, None)
            elif fn in ("first_value", "last_value"):
                spec = (out, arg_col, fn, None)
            elif fn == "nth_value":
                k = int(w.args[1].value) if len(w.args) > 1 else 1
                spec = (out, arg_col, "nth_value", k)
            elif fn == "ntile":
                k = int(w.args[0].value) if w.args else 1
                spec = (out, "", "ntile", k)
            elif fn in ("lag", "lead"):
                n = 1
                if len(w.args) > 1:
                    n = _lit_int(w.args[1])
                k = n if fn == "lag" else -n
                if len(w.args) > 2:
                    dflt = w.args[2].value if isinstance(
                        w.args[2], ast.Lit) else _lit_int(w.args[2])
                    spec = (out, arg_col, "shift", (k, dflt))
                else:
                    spec = (out, arg_col, "shift", k)
            elif fn in AGG_FUNCS or fn == "count":
                if order:
                    if fn == "count" and (w.star or not w.args):
                        spec = (out, "", "row_number", None)
                    else:
                        running = {"sum": "cumsum", "min": "cummin",
                                   "max": "cummax", "avg": "cummean",
                                   "mean": "cummean", "count": "cumcount_v"}
                        base = AGG_FUNCS.get(fn, fn)
                        if base not in running:
                            raise NotImplementedError(
                                f"{fn.upper()} OVER ORDER BY over aggregates")
                        spec = (out, arg_col, running[base], None)
                elif fn == "count" and (w.star or not w.args):
                    spec = (out, "", "transform_size", None)
                else:
                    spec = (out, arg_col,
                            f"transform_{AGG_FUNCS.get(fn, fn)}", None)
            else:
                raise NotImplementedError(f"window function {fn}")
            groups.setdefault((keys, order, asc), []).append(spec)
            return out

        def rewrite(e):
            if isinstance(e, ast.WindowE):
                name = register(e)
                col = ast.Col(None, name)
                key_map[_ast_key(col)] = name
                return col
            if isinstance(e, (ast.Query, ast.SetOpQ)):
                return e
            for f in getattr(e, "__dataclass_fields__", {}):
                v = getattr(e, f)
                if isinstance(v, list):
                    setattr(e, f, [rewrite(x) if hasattr(
                        x, "__dataclass_fields__") else x for x in v])
                elif hasattr(v, "__dataclass_fields__"):
                    setattr(e, f, rewrite(v))
            return e

        for it in q.items:
            if not it.star:
                it.expr = rewrite(it.expr)
        qual = getattr(q, "qualify", None)
        if qual is not None:
            q.qualify = rewrite(qual)
        if need_pre:
            plan = pn.Projection(plan, tuple(pre_names),
                                 tuple(pre_exprs[n] for n in pre_names))
        for (keys, order, asc), specs in groups.items():
            plan = pn.Window(plan, keys, order, asc, tuple(specs))
        if qual is not None:
            plan = pn.Filter(plan, conv(q.qualify))
            q.qualify = None
        return plan

    def _plan_setop(self, q: "ast.SetOpQ"):
        """UNION [ALL] / INTERSECT / EXCEPT.  A trailing ORDER BY/LIMIT
        parsed into the right arm applies to the whole set expression
        (standard SQL binding)."""
        order_by, limit = [], None
        rq = q.right
        if isinstance(rq, ast.Query) and (rq.order_by or rq.limit is not None):
            order_by, limit = rq.order_by, rq.limit
            rq.order_by, rq.limit = [], None
        lplan, lnames = self.plan(q.left)
        rplan, rnames = self.plan(rq)
        if len(lnames) != len(rnames):
            raise ValueError("set operation arms have different column counts")
        # positional alignment: right arm takes the left arm's column names
        rplan = pn.Projection(rplan, tuple(lnames),
                              tuple(ex.ColRef(c) for c in rnames))
        if q.op == "union":
            out = pn.Union((lplan, rplan), distinct=not q.all)
        else:
            if q.all:
                raise NotImplementedError(f"{q.op.upper()} ALL")
            pfx = self._uniq("__set")
            renamed = [f"{pfx}_{c}" for c in lnames]
            rproj = pn.Projection(rplan, tuple(renamed),
                                  tuple(ex.ColRef(c) for c in lnames))
            rdis = pn.Distinct(rproj, tuple(renamed))
            out = pn.Join(pn.Distinct(lplan, None), rdis, tuple(lnames),
                          tuple(renamed),
                          "semi" if q.op == "intersect" else "anti")
        if order_by:
            keys, asc = [], []
            for e, a in order_by:
                if isinstance(e, ast.Lit) and e.kind == "num":
                    keys.append(lnames[int(e.value) - 1])
                elif isinstance(e, ast.Col) and e.table is None:
                    keys.append(e.name)
                else:
                    raise ValueError("set-op ORDER BY must be output "
                                     "columns or positions")
                asc.append(a)
            out = pn.Sort(out, tuple(keys), tuple(asc))
        if limit is not None:
            out = pn.Limit(out, limit)
        return out, lnames

    def _table_plan(self, tr: ast.TableRef, scope: Scope):
        alias = (tr.alias or tr.name).lower()
        if tr.subquery is not None:
            sub, cols = self.plan(tr.subquery)
        else:
            name = tr.name.lower()
            if name in self.cte_asts:
                if name not in self._cte_cache:
                    self._cte_cache[name] = self.plan(self.cte_asts[name])
                sub, cols = self._cte_cache[name]
                cols = list(cols)
            elif name not in self.tables:
                raise KeyError(f"unknown table {tr.name}")
            else:
                frame = self.tables[name]
                sub = frame._lazy_plan
                cols = list(frame._columns)
        internal = []
        need_rename = False

        def _taken(name):
            # collision check climbs to enclosing scopes so a subquery's
            # columns get distinct internal names from the outer query's
            # (needed by row-id decorrelation, where both sides of the
            # decorrelating join can come from the SAME base table)
            s = scope
            while s is not None:
                for e in s.entries:
                    if e[1].lower() == name.lower() or \
                            e[2].lower() == name.lower():
                        return True
                s = s.parent
            return False

        for c in cols:
            iname = f"{alias}_{c}" if _taken(c) else c
            if iname != c:
                need_rename = True
            internal.append(iname)
            scope.add(alias, c, iname)
        if need_rename:
            sub = pn.Projection(sub, tuple(internal),
                                tuple(ex.ColRef(c) for c in cols))
        return sub, set(internal)

    def _equi_pair(self, c, scope: Scope, left_cols: set, right_cols: set):
        if isinstance(c, ast.Bin) and c.op == "eq" \
                and isinstance(c.left, ast.Col) and isinstance(c.right, ast.Col):
            try:
                l = scope.resolve(c.left.table, c.left.name)
                r = scope.resolve(c.right.table, c.right.name)
            except KeyError:
                return None
            if l in left_cols and r in right_cols:
                return (l, r)
            if r in left_cols and l in right_cols:
                return (r, l)
        return None

    # ------------------------------------------------------------------
    # subqueries (reference: BodoSQL/Calcite decorrelation rule sets)
    # ------------------------------------------------------------------
    def _apply_subquery_pred(self, c, plan, scope: Scope):
        """IN/EXISTS/correlated-scalar predicates -> semi/anti/agg joins.
        Returns (handled, new_plan)."""
        neg = False
        inner = c
        while isinstance(inner, ast.Un) and inner.op == "not":
            neg = not neg
            inner = inner.operand
        if isinstance(inner, ast.InSubquery):
            return True, self._in_subquery(inner.operand, inner.query,
                                           neg ^ inner.negated, plan, scope)
        if isinstance(inner, ast.ExistsE):
            return True, self._exists_subquery(inner.query,
                                               neg ^ inner.negated, plan,
                                               scope)
        if isinstance(inner, ast.Bin) and inner.op in (
                "lt", "le", "gt", "ge", "eq", "ne"):
            l, r = inner.left, inner.right
            flip = {"lt": "gt", "le": "ge", "gt": "lt", "ge": "le",
                    "eq": "eq", "ne": "ne"}
            if isinstance(l, ast.SubqueryE) and not isinstance(r, ast.SubqueryE):
                l, r = r, l
                inner = ast.Bin(flip[inner.op], l, r)
            if isinstance(inner.right, ast.SubqueryE):
                if neg:
                    return False, plan  # NOT (x < (subq)): keep expr path
                return True, self._scalar_cmp_subquery(
                    inner.op, inner.left, inner.right.query, plan, scope)
        return False, plan

    def _plan_inner(self, q: ast.Query, outer: Scope):
        """Plan a subquery's FROM+WHERE; returns (plan, scope, corr_pairs)
        with correlation equalities removed from the filter."""
        scope = Scope(parent=outer)
        plan = None
        joined = set()
        conjs = _split_conjuncts(q.where) if q.where is not None else []
        used = [False] * len(conjs)
        refs = ([q.table] if q.table else []) + [j.table for j in q.joins]
        kinds = ["base"] + [j.kind for j in q.joins]
        ons = [None] + [j.on for j in q.joins]
        for idx, tr in enumerate(refs):
            if getattr(tr, "flatten", None) is not None:
                # , LATERAL FLATTEN(input => expr) f  ->  Explode over the
                # accumulated plan (reference: _lateral.cpp FLATTEN)
                assert plan is not None, "LATERAL FLATTEN needs a base table"
                vcol = self._uniq("__flat_v")
                icol = self._uniq("__flat_i")
                keep = sorted(joined_cols)
                exprs = tuple(ex.ColRef(c) for c in keep) + (
                    self.expr(tr.flatten, scope),)
                plan = pn.Projection(plan, tuple(keep) + (vcol,), exprs)
                plan = pn.Explode(plan, vcol, pos=icol)
                alias = tr.alias or "f"
                scope.add(alias, "value", vcol)
                scope.add(alias, "index", icol)
                joined_cols |= {vcol, icol}
                continue
            sub_plan, sub_cols = self._table_plan(tr, scope)
            if plan is None:
                plan = sub_plan
                joined |= sub_cols
                continue
            eq = []
            if ons[idx] is not None:
                for cj in _split_conjuncts(ons[idx]):
                    p = self._equi_pair(cj, scope, joined, sub_cols)
                    if p is None:
                        raise NotImplementedError("non-equi subquery join")
                    eq.append(p)
            for ci, cj in enumerate(conjs):
                if used[ci]:
                    continue
                p = self._equi_pair(cj, scope, joined, sub_cols)
                if p:
                    eq.append(p)
                    used[ci] = True
            if eq:
                plan = pn.Join(plan, sub_plan, tuple(x[0] for x in eq),
                               tuple(x[1] for x in eq), "inner")
            else:
                plan = pn.Join(plan, sub_plan, (), (), "cross")
            joined |= sub_cols
        corr_pairs = []
        corr_extra = []  # correlated conjuncts that are NOT equality pairs
        inner_keep = []
        for ci, cj in enumerate(conjs):
            if used[ci]:
                continue
            pair = self._corr_pair(cj, scope)
            if pair is not None:
                corr_pairs.append(pair)
                continue
            if self._refs_outer(cj, scope):
                corr_extra.append(cj)
                continue
            handled, plan = self._apply_subquery_pred(cj, plan, scope)
            if not handled:
                inner_keep.append(cj)
        if inner_keep:
            cond = inner_keep[0]
            for cj in inner_keep[1:]:
                cond = ast.Bin("and", cond, cj)
            plan = pn.Filter(plan, self.expr(cond, scope))
        return plan, scope, corr_pairs, corr_extra

    def _refs_outer(self, node, scope: Scope) -> bool:
        """True when the AST references any enclosing-scope column."""
        if isinstance(node, ast.Col):
            return scope.is_outer(node.table, node.name)
        if isinstance(node, ast.Query):
            # nested subqueries resolve through their own child Scope; their
            # columns are not this scope's correlation concern
            return False
        for f in getattr(node, "__dataclass_fields__", {}):
            v = getattr(node, f)
            items = v if isinstance(v, (tuple, list)) else (v,)
            for x in items:
                if hasattr(x, "__dataclass_fields__") and \
                        self._refs_outer(x, scope):
                    return True
        return False

    def _corr_pair(self, cj, sub_scope: Scope):
        """inner_col = outer_col -> (outer_internal, inner_internal)."""
        if not (isinstance(cj, ast.Bin) and cj.op == "eq"
                and isinstance(cj.left, ast.Col)
                and isinstance(cj.right, ast.Col)):
            return None
        l, r = cj.left, cj.right
        l_outer = sub_scope.is_outer(l.table, l.name)
        r_outer = sub_scope.is_outer(r.table, r.name)
        if l_outer == r_outer:
            return None
        o, i = (l, r) if l_outer else (r, l)
        return (sub_scope.parent.resolve(o.table, o.name),
                sub_scope.resolve_local(i.table, i.name))

    def _try_plan_full(self, q: ast.Query):
        """Full planner for an (assumed) uncorrelated subquery; None when it
        references enclosing columns (KeyError) or uses unsupported shapes."""
        try:
            return self.plan(q)
        except (KeyError, NotImplementedError, ValueError):
            return None

    def _in_subquery(self, operand, q: ast.Query, negated, plan, scope):
        if q.group_by or q.having is not None or any(
                _has_agg(it.expr) for it in q.items if not it.star):
            full = self._try_plan_full(q)
            if full is None:
                raise NotImplementedError("correlated grouped IN subquery")
            sub_plan, out_names = full
            op_expr = self.expr(operand, scope)
            if not isinstance(op_expr, ex.ColRef):
                raise NotImplementedError("IN subquery on computed operand")
            pfx = self._uniq("__sq")
            renamed = (f"{pfx}_{out_names[0]}",)
            proj = pn.Projection(sub_plan, renamed,
                                 (ex.ColRef(out_names[0]),))
            dis = pn.Distinct(proj, renamed)
            return pn.Join(plan, dis, (op_expr.name,), renamed,
                           "anti" if negated else "semi")
        sub_plan, sub_scope, corr, corr_extra = self._plan_inner(q, scope)
        if corr_extra:
            raise NotImplementedError("non-equality correlated IN subquery")
        item = q.items[0]
        val_internal = self._subq_value_col(item, sub_scope, sub_plan)
        sub_plan = val_internal[1]
        val_col = val_internal[0]
        op_expr = self.expr(operand, scope)
        if not isinstance(op_expr, ex.ColRef):
            raise NotImplementedError("IN subquery on computed operand")
        pfx = self._uniq("__sq")
        keep = [val_col] + [i for _, i in corr]
        renamed = [f"{pfx}_{c}" for c in keep]
        proj = pn.Projection(sub_plan, tuple(renamed),
                             tuple(ex.ColRef(c) for c in keep))
        dis = pn.Distinct(proj, tuple(renamed))
        lks = (op_expr.name,) + tuple(o for o, _ in corr)
        rks = tuple(renamed)
        return pn.Join(plan, dis, lks, rks, "anti" if negated else "semi")

    def _exists_subquery(self, q: ast.Query, negated, plan, scope):
        sub_plan, sub_scope, corr, corr_extra = self._plan_inner(q, scope)
        if not corr and not corr_extra:
            raise NotImplementedError("uncorrelated EXISTS")
        if not corr_extra:
            pfx = self._uniq("__sq")
            keep = [i for _, i in corr]
            renamed = [f"{pfx}_{c}" for c in keep]
            proj = pn.Projection(sub_plan, tuple(renamed),
                                 tuple(ex.ColRef(c) for c in keep))
            dis = pn.Distinct(proj, tuple(renamed))
            lks = tuple(o for o, _ in corr)
            return pn.Join(plan, dis, lks, tuple(renamed),
                           "anti" if negated else "semi")
        # row-id decorrelation (TPC-H q21 shape: EXISTS with an extra
        # non-equality correlated predicate, e.g. l2.l_suppkey <> l1.l_suppkey;
        # reference analog: Calcite RelDecorrelator used by BodoSQL):
        # tag the outer rows with a global row id, inner-join outer x inner on
        # the equality correlations, filter by the non-equality predicates
        # (both sides' columns are in scope post-join, with distinct internal
        # names), then semi/anti join the tagged outer plan on the surviving
        # row ids.
        if not corr:
            raise NotImplementedError(
                "EXISTS with only non-equality correlation (cross-product "
                "decorrelation not supported)")
        rid = self._uniq("__rid")
        tagged = pn.RowId(plan, rid)
        lks = tuple(o for o, _ in corr)
        rks = tuple(i for _, i in corr)
        jn = pn.Join(tagged, sub_plan, lks, rks, "inner")
        cond = self.expr(corr_extra[0], sub_scope)
        for cj in corr_extra[1:]:
            cond = ex.BoolOp("and", cond, self.expr(cj, sub_scope))
        flt = pn.Filter(jn, cond)
        proj = pn.Projection(flt, (rid,), (ex.ColRef(rid),))
        dis = pn.Distinct(proj, (rid,))
        out = pn.Join(tagged, dis, (rid,), (rid,), "anti" if negated else "semi")
        # drop the row-id column so downstream output shape is unchanged
        cols = plan.out_columns()
        if cols is None:
            return out
        keep_cols = tuple(cols)
        return pn.Projection(out, keep_cols,
                             tuple(ex.ColRef(c) for c in keep_cols))

    def _rewrite_select_subqueries(self, plan, scope: Scope, q: ast.Query):
        """SELECT-list scalar subqueries.  Correlated equi ones decorrelate
        into a LEFT JOIN against the grouped inner query (NULL when no
        match, per SQL); uncorrelated ones stay as ScalarSubquery exprs
        handled by expr() (reference: Calcite RelDecorrelator as used by
        BodoSQL)."""
        state = {"plan": plan}

        def rewrite(e):
            if isinstance(e, ast.SubqueryE):
                repl = self._decorrelate_select_subq(e.query, state, scope)
                return repl if repl is not None else e
            if isinstance(e, (ast.Query, ast.SetOpQ)):
                return e
            for f in getattr(e, "__dataclass_fields__", {}):
                v = getattr(e, f)
                if isinstance(v, list):
                    setattr(e, f, [rewrite(x) if hasattr(
                        x, "__dataclass_fields__") else x for x in v])
                elif hasattr(v, "__dataclass_fields__"):
                    setattr(e, f, rewrite(v))
            return e

        for it in q.items:
            if not it.star and it.expr is not None:
                it.expr = rewrite(it.expr)
        return state["plan"]

    def _decorrelate_select_subq(self, subq, state, scope: Scope):
        if not isinstance(subq, ast.Query) or len(subq.items) != 1 \
                or subq.items[0].star:
            return None
        if self._try_plan_full(subq) is not None:
            return None  # uncorrelated: expr() evaluates it once
        try:
            sub_plan, sub_scope, corr, corr_extra = self._plan_inner(
                subq, scope)
        except (NotImplementedError, KeyError):
            return None
        if corr_extra or not corr:
            return None
        item = subq.items[0]
        agg_ast, rebuild = _extract_single_agg(item.expr)
        keys = tuple(i for _, i in corr)
        if agg_ast is not None:
            val, aplan = self._subq_agg_plan(agg_ast, sub_scope, sub_plan,
                                             keys)
            if rebuild is not None:
                wrapped = self._uniq("__sw")
                aplan = pn.Projection(
                    aplan, tuple(list(keys) + [wrapped]),
                    tuple([ex.ColRef(k) for k in keys]
                          + [rebuild(ex.ColRef(val), self, sub_scope)]))
                val = wrapped
        else:
            # bare value: SQL requires at most one inner row per outer row
            vcol, vplan = self._subq_value_col(item, sub_scope, sub_plan)
            val = self._uniq("__sv")
            aplan = pn.Aggregate(vplan, keys, ((val, vcol, "first"),))
        pfx = self._uniq("__ss")
        renamed = [f"{pfx}_{c}" for c in keys] + [f"{pfx}_{val}"]
        proj = pn.Projection(aplan, tuple(renamed),
                             tuple(ex.ColRef(c) for c in list(keys) + [val]))
        state["plan"] = pn.Join(state["plan"], proj,
                                tuple(o for o, _ in corr),
                                tuple(renamed[:-1]), "left")
        name = renamed[-1]
        scope.add("", name, name)
        return ast.Col(None, name)

    def _scalar_cmp_subquery(self, op, outer_expr, q: ast.Query, plan, scope):
        full = self._try_plan_full(q)
        if full is not None:
            sub_plan, out_names = full
            return pn.Filter(plan, ex.Cmp(
                op, self.expr(outer_expr, scope),
                ex.ScalarSubquery(sub_plan, out_names[0])))
        sub_plan, sub_scope, corr, corr_extra = self._plan_inner(q, scope)
        if corr_extra:
            raise NotImplementedError(
                "non-equality correlated scalar subquery")
        item = q.items[0]
        agg_ast, rebuild = _extract_single_agg(item.expr)
        if agg_ast is None:
            raise NotImplementedError("scalar subquery must be an aggregate")
        keys = tuple(i for _, i in corr)
        val, aplan = self._subq_agg_plan(agg_ast, sub_scope, sub_plan, keys)
        if rebuild is not None:
            # arithmetic around the aggregate (e.g. 0.2 * avg(x))
            wrapped = self._uniq("__sw")
            aplan = pn.Projection(
                aplan, tuple(list(keys) + [wrapped]),
                tuple([ex.ColRef(k) for k in keys]
                      + [rebuild(ex.ColRef(val), self, sub_scope)]))
            val = wrapped
        if not corr:
            return pn.Filter(plan, ex.Cmp(
                op, self.expr(outer_expr, scope),
                ex.ScalarSubquery(aplan, val)))
        pfx = self._uniq("__sq")
        renamed = [f"{pfx}_{c}" for c in keys] + [f"{pfx}_{val}"]
        proj = pn.Projection(aplan, tuple(renamed),
                             tuple(ex.ColRef(c) for c in list(keys) + [val]))
        joined = pn.Join(plan, proj, tuple(o for o, _ in corr),
                         tuple(renamed[:-1]), "inner")
        return pn.Filter(joined, ex.Cmp(
            op, self.expr(outer_expr, scope), ex.ColRef(renamed[-1])))

    def _subq_agg_plan(self, agg_ast, sub_scope, sub_plan, keys):
        func = AGG_FUNCS[agg_ast.name]
        out = self._uniq("__sv")
        if agg_ast.star or not agg_ast.args:
            return out, pn.Aggregate(sub_plan, tuple(keys),
                                     ((out, "", "size"),)) if keys else                 (out, pn.Reduce(sub_plan, ((out, "", "size"),)))
        arg = agg_ast.args[0]
        if isinstance(arg, ast.Col):
            in_name = sub_scope.resolve(arg.table, arg.name)
            pre = sub_plan
        else:
            in_name = self._uniq("__ain")
            cols = [e[2] for e in sub_scope.entries]
            pre = pn.Projection(
                sub_plan, tuple(cols + [in_name]),
                tuple([ex.ColRef(c) for c in cols]
                      + [self.expr(arg, sub_scope)]))
        if keys:
            return out, pn.Aggregate(pre, tuple(keys), ((out, in_name, func),))
        return out, pn.Reduce(pre, ((out, in_name, func),))

    def _subq_value_col(self, item, sub_scope, sub_plan):
        if isinstance(item.expr, ast.Col):
            return sub_scope.resolve(item.expr.table, item.expr.name), sub_plan
        name = self._uniq("__sqv")
        cols = [e[2] for e in sub_scope.entries]
        proj = pn.Projection(
            sub_plan, tuple(cols + [name]),
            tuple([ex.ColRef(c) for c in cols]
                  + [self.expr(item.expr, sub_scope)]))
        return name, proj

    # ------------------------------------------------------------------
    def _aggregate(self, plan, scope: Scope, q: ast.Query):
        # 1. group keys: plain columns stay; computed keys get a pre-projection
        pre_names = [e[2] for e in scope.entries]
        pre_exprs = {n: ex.ColRef(n) for n in pre_names}
        keys = []
        key_map = {}  # AST repr -> key column name
        for g in q.group_by:
            if isinstance(g, ast.Col):
                internal = scope.resolve(g.table, g.name)
                keys.append(internal)
                key_map[_ast_key(g)] = internal
            else:
                kname = self._uniq("__gk")
                pre_exprs[kname] = self.expr(g, scope)
                pre_names.append(kname)
                keys.append(kname)
                key_map[_ast_key(g)] = kname
        # 2. collect aggregate calls from select/having
        agg_specs = []  # (out_name, in_name or "", func)
        agg_map = {}

        def collect(e):
            if isinstance(e, ast.Func) and (
                    e.name in AGG_FUNCS
                    or e.name in ("listagg", "string_agg",
                                  "percentile_cont", "percentile_disc")):
                k = _ast_key(e)
                if k in agg_map:
                    return
                out = self._uniq("__agg")
                if e.star or not e.args:
                    agg_specs.append((out, "", "size"))
                else:
                    arg = e.args[0]
                    if isinstance(arg, ast.Col):
                        in_name = scope.resolve(arg.table, arg.name)
                    else:
                        in_name = self._uniq("__ain")
                        pre_exprs[in_name] = self.expr(arg, scope)
                        pre_names.append(in_name)
                    if e.name in ("percentile_cont", "percentile_disc"):
                        qv = float(e.args[0].value)
                        interp = ("linear" if e.name == "percentile_cont"
                                  else "nearest")
                        wo = getattr(e, "within_order", None)
                        if wo is not None:
                            # value column is the WITHIN GROUP order expr
                            oe = wo[0]
                            if isinstance(oe, ast.Col):
                                in_name = scope.resolve(oe.table, oe.name)
                            else:
                                in_name = self._uniq("__ain")
                                pre_exprs[in_name] = self.expr(oe, scope)
                                pre_names.append(in_name)
                        func = (lambda s_, _q=qv, _i=interp:
                                s_.quantile(_q, interpolation=_i))
                    elif e.name in ("listagg", "string_agg"):
                        sep = e.args[1].value if len(e.args) > 1 else ""
                        wo = getattr(e, "within_order", None)
                        if wo is not None:
                            if _ast_key(wo[0]) != _ast_key(e.args[0]):
                                raise NotImplementedError(
                                    "LISTAGG WITHIN GROUP ordering by a "
                                    "different column")
                            func = _listagg_func(sep, sort=True,
                                                 ascending=wo[1])
                        else:
                            func = _listagg_func(sep)
                    else:
                        func = AGG_FUNCS[e.name]
                        if e.distinct and e.name == "count":
                            func = "nunique"
                    agg_specs.append((out, in_name, func))
                agg_map[k] = out
                return
            if isinstance(e, (ast.Query, ast.SetOpQ, ast.SubqueryE)):
                return  # a subquery's aggregates are planned by its own plan
            for f in getattr(e, "__dataclass_fields__", {}):
                v = getattr(e, f)
                if isinstance(v, (list, tuple)):
                    for i in v:
                        if hasattr(i, "__dataclass_fields__"):
                            collect(i)
                        elif isinstance(i, tuple):
                            for j in i:
                                if hasattr(j, "__dataclass_fields__"):
                                    collect(j)
                elif hasattr(v, "__dataclass_fields__"):
                    collect(v)

        for it in q.items:
            if not it.star:
                collect(it.expr)
        if q.having is not None:
            collect(q.having)
        pre = pn.Projection(plan, tuple(pre_names),
                            tuple(pre_exprs[n] for n in pre_names))
        if keys:
            agg = pn.Aggregate(pre, tuple(keys), tuple(agg_specs))
        else:
            agg = pn.Reduce(pre, tuple(agg_specs))
        return agg, scope, key_map, agg_map

    def _post_agg_expr(self, e, key_map, agg_map, scope: Scope) -> ex.Expr:
        k = _ast_key(e)
        if k in agg_map:
            return ex.ColRef(agg_map[k])
        if k in key_map:
            return ex.ColRef(key_map[k])
        if isinstance(e, ast.Col):
            internal = scope.resolve(e.table, e.name)
            if internal in key_map.values():
                return ex.ColRef(internal)
            raise ValueError(f"column {e.name} is neither grouped nor aggregated")
        if isinstance(e, ast.Lit):
            return self.expr(e, scope)
        if isinstance(e, ast.Bin):
            return ex.BinOp(e.op, self._post_agg_expr(e.left, key_map, agg_map, scope),
                            self._post_agg_expr(e.right, key_map, agg_map, scope)) \
                if e.op in ("add", "sub", "mul", "div", "mod") else \
                (ex.BoolOp(e.op, self._post_agg_expr(e.left, key_map, agg_map, scope),
                           self._post_agg_expr(e.right, key_map, agg_map, scope))
                 if e.op in ("and", "or") else
                 ex.Cmp(e.op, self._post_agg_expr(e.left, key_map, agg_map, scope),
                        self._post_agg_expr(e.right, key_map, agg_map, scope)))
        if isinstance(e, ast.Func) and e.name == "round":
            return ex.RoundExpr(self._post_agg_expr(e.args[0], key_map, agg_map,
                                                    scope),
                                int(e.args[1].value) if len(e.args) > 1 else 0)
        if isinstance(e, ast.Func) and _FN_ALIASES.get(
                e.name, e.name) == "coalesce":
            # IFNULL/NVL/COALESCE over aggregate outputs
            args = [self._post_agg_expr(a, key_map, agg_map, scope)
                    for a in e.args]
            out = args[-1]
            for a in reversed(args[:-1]):
                out = ex.Case((ex.IsNull(a),), (out,), a)
            return out
        if isinstance(e, ast.CastE):
            return ex.Cast(self._post_agg_expr(e.operand, key_map, agg_map,
                                               scope), CAST_TYPES[e.to],
                           getattr(e, "safe", False))
        if isinstance(e, ast.SubqueryE):
            full = self._try_plan_full(e.query)
            if full is None:
                raise NotImplementedError("correlated subquery in HAVING")
            sub_plan, out_names = full
            return ex.ScalarSubquery(sub_plan, out_names[0])
        raise NotImplementedError(f"post-agg expr {e}")

    # ------------------------------------------------------------------
    def expr(self, e, scope: Scope) -> ex.Expr:
        if isinstance(e, ast.Col):
            return ex.ColRef(scope.resolve(e.table, e.name))
        if isinstance(e, ast.Lit):
            return _lit_expr(e)
        if isinstance(e, ast.Bin):
            l, r = self.expr(e.left, scope), self.expr(e.right, scope)
            if e.op in ("and", "or"):
                return ex.BoolOp(e.op, l, r)
            if e.op in ("eq", "ne", "lt", "le", "gt", "ge"):
                return ex.Cmp(e.op, l, r)
            # date +/- interval
            if e.op in ("add", "sub") and isinstance(e.right, ast.Lit) \
                    and e.right.kind == "interval":
                qty, unit = e.right.value
                delta = _interval_to_timestamp_delta(e.left, qty, unit, e.op)
                if delta is not None:
                    return delta
                # fixed-length units add as nanosecond offsets on columns
                ns_per = {"day": 86400 * 10**9, "hour": 3600 * 10**9,
                          "minute": 60 * 10**9, "second": 10**9,
                          "week": 7 * 86400 * 10**9}
                if unit in ns_per:
                    off = ex.Const(int(qty) * ns_per[unit], bt.int64)
                    return ex.BinOp(e.op, l, off)
                if unit in ("month", "quarter", "year"):
                    mult = {"month": 1, "quarter": 3, "year": 12}[unit]
                    n_m = int(qty) * mult * (1 if e.op == "add" else -1)
                    return ex.DtField(l, f"add_months:{n_m}")
                raise NotImplementedError(
                    f"interval '{unit}' arithmetic on columns")
            return ex.BinOp(e.op, l, r)
        if isinstance(e, ast.Un):
            assert e.op == "not"
            return ex.Not(self.expr(e.operand, scope))
        if isinstance(e, ast.InE):
            vals = tuple(v.value for v in e.values)
            out = ex.IsIn(self.expr(e.operand, scope), vals)
            return ex.Not(out) if e.negated else out
        if isinstance(e, ast.BetweenE):
            lo = ex.Cmp("ge", self.expr(e.operand, scope), self.expr(e.lo, scope))
            hi = ex.Cmp("le", self.expr(e.operand, scope), self.expr(e.hi, scope))
            out = ex.BoolOp("and", lo, hi)
            return ex.Not(out) if e.negated else out
        if isinstance(e, ast.LikeE):
            operand = self.expr(e.operand, scope)
            if getattr(e, "ci", False):  # ILIKE: case-insensitive
                out = _like_expr(ex.StrOp(operand, "lower"),
                                 e.pattern.lower())
            else:
                out = _like_expr(operand, e.pattern)
            return ex.Not(out) if e.negated else out
        if isinstance(e, ast.IsNullE):
            return ex.IsNull(self.expr(e.operand, scope), negate=e.negated)
        if isinstance(e, ast.CastE):
            if e.to not in CAST_TYPES:
                raise NotImplementedError(f"CAST to {e.to}")
            return ex.Cast(self.expr(e.operand, scope), CAST_TYPES[e.to],
                           getattr(e, "safe", False))
        if isinstance(e, ast.ExtractE):
            fld = {"dow": "dayofweek", "doy": "dayofyear",
                   "isodow": "dayofweek"}.get(e.fld, e.fld)
            return ex.DtField(self.expr(e.operand, scope), fld)
        if isinstance(e, ast.CaseE):
            conds = tuple(self.expr(c, scope) for c, _ in e.whens)
            thens = tuple(self.expr(v, scope) for _, v in e.whens)
            els = self.expr(e.els, scope) if e.els is not None else ex.Const(None, bt.float64)
            return ex.Case(conds, thens, els)
        if isinstance(e, ast.Func):
            name = _FN_ALIASES.get(e.name, e.name)
            if name != e.name:
                e = ast.Func(name, e.args, star=getattr(e, "star", False),
                             distinct=getattr(e, "distinct", False))
            if name == "array_construct":
                return ex.ListBuild(tuple(self.expr(a, scope)
                                          for a in e.args))
            if name == "object_construct":
                names = tuple(str(e.args[i].value)
                              for i in range(0, len(e.args), 2))
                items = tuple(self.expr(e.args[i], scope)
                              for i in range(1, len(e.args), 2))
                return ex.StructBuild(names, items)
            if name == "parse_json":
                import json as _json

                def _pj(v):
                    try:
                        return _json.loads(v)
                    except Exception:
                        return None

                return ex.UdfMap(self.expr(e.args[0], scope), _pj, "ignore")
            if name == "regexp_instr":
                import re as _re

                pat = e.args[1].value

                def _ri(v, _p=pat):
                    m = _re.search(_p, str(v))
                    return m.start() + 1 if m else 0

                return ex.UdfMap(self.expr(e.args[0], scope), _ri, "ignore")
            if name == "getbit":
                k = _lit_int(e.args[1])
                shifted = ex.BinOp("rshift", self.expr(e.args[0], scope),
                                   ex.Const(k))
                return ex.BinOp("bitand", shifted, ex.Const(1))
            if name == "atan2":
                import math as _math

                y = self.expr(e.args[0], scope)
                xv = _lit_num(e.args[1])
                return ex.UdfMap(y, lambda v, _x=xv: _math.atan2(v, _x),
                                 "ignore")
            if name == "cot":
                import math as _math

                return ex.UdfMap(self.expr(e.args[0], scope),
                                 lambda v: 1.0 / _math.tan(v), "ignore")
            if name == "haversine":
                import math as _math

                vals = [_lit_num(a) for a in e.args]
                la1, lo1, la2, lo2 = [_math.radians(v) for v in vals]
                h = (_math.sin((la2 - la1) / 2) ** 2
                     + _math.cos(la1) * _math.cos(la2)
                     * _math.sin((lo2 - lo1) / 2) ** 2)
                return ex.Const(2 * 6371.0087714 * _math.asin(_math.sqrt(h)))
            if name == "soundex":
                return ex.UdfMap(self.expr(e.args[0], scope), _soundex,
                                 "ignore")
            if name in ("editdistance", "edit_distance"):
                other = e.args[1].value
                return ex.UdfMap(self.expr(e.args[0], scope),
                                 lambda v, _o=other: _levenshtein(str(v),
                                                                  _o),
                                 "ignore")
            if name == "jarowinkler_similarity":
                other = e.args[1].value
                return ex.UdfMap(
                    self.expr(e.args[0], scope),
                    lambda v, _o=other: int(round(
                        _jaro_winkler(str(v), _o) * 100)), "ignore")
            if name == "div0":
                # a / b, 0 when b = 0 (Snowflake DIV0)
                a = self.expr(e.args[0], scope)
                b2 = self.expr(e.args[1], scope)
                return ex.Case((ex.Cmp("eq", b2, ex.Const(0)),),
                               (ex.Const(0.0),), ex.BinOp("div", a, b2))
            if name == "square":
                a = self.expr(e.args[0], scope)
                return ex.BinOp("mul", a, a)
            if name == "factorial":
                import math as _m

                return ex.UdfMap(self.expr(e.args[0], scope),
                                 lambda v: _m.factorial(int(v)), "ignore")
            if name in ("booland", "boolor"):
                a = self.expr(e.args[0], scope)
                b2 = self.expr(e.args[1], scope)
                op = "and" if name == "booland" else "or"
                return ex.BoolOp(op, ex.Cmp("ne", a, ex.Const(0)),
                                 ex.Cmp("ne", b2, ex.Const(0)))
            if name == "boolnot":
                return ex.Cmp("eq", self.expr(e.args[0], scope),
                              ex.Const(0))
            if name == "strtok":
                # STRTOK(s, delim, part): 1-based like SPLIT_PART but NULL
                # past the end
                delim = e.args[1].value if len(e.args) > 1 else " "
                part = _lit_int(e.args[2]) if len(e.args) > 2 else 1
                return ex.StrOp(self.expr(e.args[0], scope), "split_get",
                                (delim, part - 1))
            if name == "insert" and len(e.args) == 4:
                # INSERT(s, pos, len, repl): splice (1-based)
                pos = _lit_int(e.args[1])
                ln = _lit_int(e.args[2])
                repl = e.args[3].value
                return ex.UdfMap(
                    self.expr(e.args[0], scope),
                    lambda v, _p=pos, _l=ln, _r=repl:
                    str(v)[:_p - 1] + _r + str(v)[_p - 1 + _l:], "ignore")
            if name == "rtrimmed_length":
                return ex.StrOp(ex.StrOp(self.expr(e.args[0], scope),
                                         "rstrip"), "len")
            if name in ("to_char", "to_varchar"):
                return ex.Cast(self.expr(e.args[0], scope), bt.string)
            if name in ("to_number", "to_decimal", "to_numeric"):
                return ex.Cast(self.expr(e.args[0], scope), bt.float64)
            if name == "to_double":
                return ex.Cast(self.expr(e.args[0], scope), bt.float64)
            if name == "random":
                # per-row deterministic stream; each call site gets its own
                # seed so two random() in one query differ
                self._counter += 1
                return ex.RandomExpr(self._counter)
            if name == "uniform" and len(e.args) == 3:
                lo = _lit_num(e.args[0])
                hi = _lit_num(e.args[1])
                gen = self.expr(e.args[2], scope)
                span = hi - lo
                scaled = ex.BinOp("div", gen, ex.Const(float(1 << 63)))
                return ex.BinOp("add", ex.Const(lo),
                                ex.BinOp("mul", scaled, ex.Const(span)))
            if name == "is_null_value":
                return ex.IsNull(self.expr(e.args[0], scope))
            if name in ("year", "month", "day", "hour", "minute", "second",
                        "quarter"):
                return ex.DtField(self.expr(e.args[0], scope), name)
            if name == "upper":
                return ex.StrOp(self.expr(e.args[0], scope), "upper")
            if name == "lower":
                return ex.StrOp(self.expr(e.args[0], scope), "lower")
            if name == "length":
                return ex.StrOp(self.expr(e.args[0], scope), "len")
            if name == "trim":
                return ex.StrOp(self.expr(e.args[0], scope), "strip")
            if name in ("substring", "substr", "mid"):
                arg = e.args[0]
                start = _lit_int(e.args[1])
                ln = (_lit_int(e.args[2])
                      if len(e.args) > 2 and e.args[2] is not None else None)
                # Snowflake: 1-based; 0 behaves like 1; negative counts
                # from the end
                s = start - 1 if start > 0 else (start if start < 0 else 0)
                if ln is None:
                    stop = None
                elif s < 0:
                    stop = s + ln if s + ln < 0 else None
                else:
                    stop = s + ln
                return ex.StrOp(self.expr(arg, scope), "slice", (s, stop, 1))
            if name == "round":
                return ex.RoundExpr(self.expr(e.args[0], scope),
                                    int(e.args[1].value) if len(e.args) > 1 else 0)
            if name == "abs":
                inner = self.expr(e.args[0], scope)
                zero = ex.Const(0)
                return ex.Case((ex.Cmp("lt", inner, zero),),
                               (ex.BinOp("sub", zero, inner),), inner)
            if name in ("iff", "if"):
                return ex.Case((self.expr(e.args[0], scope),),
                               (self.expr(e.args[1], scope),),
                               self.expr(e.args[2], scope))
            if name == "nvl":
                a = self.expr(e.args[0], scope)
                return ex.Case((ex.IsNull(a, negate=True),), (a,),
                               self.expr(e.args[1], scope))
            if name == "nvl2":
                a = self.expr(e.args[0], scope)
                return ex.Case((ex.IsNull(a, negate=True),),
                               (self.expr(e.args[1], scope),),
                               self.expr(e.args[2], scope))
            if name == "zeroifnull":
                a = self.expr(e.args[0], scope)
                return ex.Case((ex.IsNull(a),), (ex.Const(0),), a)
            if name == "coalesce":
                args = [self.expr(a, scope) for a in e.args]
                out = args[-1]
                for a in reversed(args[:-1]):
                    out = ex.Case((ex.IsNull(a, negate=True),), (a,), out)
                return out
            if name == "nullif":
                a = self.expr(e.args[0], scope)
                b = self.expr(e.args[1], scope)
                return ex.Case((ex.Cmp("eq", a, b),),
                               (ex.Const(None, bt.float64),), a)
            if name == "floor":
                return ex.BinOp("floordiv", self.expr(e.args[0], scope),
                                ex.Const(1))
            if name in ("ceil", "ceiling"):
                x = self.expr(e.args[0], scope)
                zero = ex.Const(0)
                return ex.BinOp(
                    "sub", zero,
                    ex.BinOp("floordiv", ex.BinOp("sub", zero, x),
                             ex.Const(1)))
            if name == "mod":
                # python modulo semantics (sign of divisor); SQL MOD takes
                # the dividend's sign — identical for positive operands
                return ex.BinOp("mod", self.expr(e.args[0], scope),
                                self.expr(e.args[1], scope))
            if name in ("power", "pow"):
                return ex.BinOp("pow", self.expr(e.args[0], scope),
                                self.expr(e.args[1], scope))
            if name == "sqrt":
                return ex.BinOp("pow", self.expr(e.args[0], scope),
                                ex.Const(0.5))
            if name in ("exp", "ln", "log", "log10"):
                import math

                f = {"exp": math.exp, "ln": math.log, "log": math.log,
                     "log10": math.log10}[name]
                return ex.UdfMap(self.expr(e.args[0], scope), f, None)
            if name == "sign":
                x = self.expr(e.args[0], scope)
                return ex.Case(
                    (ex.IsNull(x), ex.Cmp("lt", x, ex.Const(0)),
                     ex.Cmp("gt", x, ex.Const(0))),
                    (ex.Const(None, None), ex.Const(-1), ex.Const(1)),
                    ex.Const(0))
            if name in ("greatest", "least"):
                # Snowflake: NULL if ANY argument is NULL (the pairwise
                # CASE chain alone leaked null-row storage values)
                op = "gt" if name == "greatest" else "lt"
                args = [self.expr(a, scope) for a in e.args]
                out = args[0]
                for a in args[1:]:
                    out = ex.Case((ex.Cmp(op, out, a),), (out,), a)
                any_null = ex.IsNull(args[0])
                for a in args[1:]:
                    any_null = ex.BoolOp("or", any_null, ex.IsNull(a))
                return ex.Case((any_null,), (ex.Const(None, None),), out)
            if name == "replace":
                return ex.StrOp(self.expr(e.args[0], scope), "replace",
                                (e.args[1].value, e.args[2].value),
                                (("regex", False),))
            if name in ("ltrim", "rtrim"):
                op = "lstrip" if name == "ltrim" else "rstrip"
                chars = (e.args[1].value,) if len(e.args) > 1 else ()
                return ex.StrOp(self.expr(e.args[0], scope), op, chars)
            if name == "left":
                n = int(e.args[1].value)
                return ex.StrOp(self.expr(e.args[0], scope), "slice",
                                (0, n, 1))
            if name == "right":
                n = int(e.args[1].value)
                return ex.StrOp(self.expr(e.args[0], scope), "slice",
                                (-n, None, 1))
            if name == "initcap":
                return ex.StrOp(self.expr(e.args[0], scope), "title")
            if name in ("char_length", "character_length", "len"):
                return ex.StrOp(self.expr(e.args[0], scope), "len")
            if name in ("strpos", "position", "instr"):
                # STRPOS(str, sub) / two-arg POSITION(sub, str): 1-based,
                # 0 when absent (find returns -1)
                if name == "position":
                    sub, hay = e.args[0], e.args[1]
                else:
                    hay, sub = e.args[0], e.args[1]
                found = ex.StrOp(self.expr(hay, scope), "find",
                                 (sub.value,))
                return ex.BinOp("add", found, ex.Const(1))
            if name == "split_part":
                # Snowflake 1-based part index; out-of-range -> ''
                return ex.StrOp(self.expr(e.args[0], scope), "split_part",
                                (e.args[1].value,
                                 int(e.args[2].value) - 1))
            if name in ("lpad", "rpad"):
                # Snowflake: a string longer than n is TRUNCATED to n
                n = int(e.args[1].value)
                fill = e.args[2].value if len(e.args) > 2 else " "
                op = "rjust" if name == "lpad" else "ljust"
                padded = ex.StrOp(self.expr(e.args[0], scope), op, (n, fill))
                return ex.StrOp(padded, "slice", (0, n))
            if name == "repeat":
                return ex.StrOp(self.expr(e.args[0], scope), "repeat",
                                (int(e.args[1].value),))
            if name == "date_trunc":
                unit = e.args[0].value.lower().rstrip("s")
                fld = {"day": "floor_day", "month": "trunc_month",
                       "year": "trunc_year", "quarter": "trunc_quarter",
                       "week": "trunc_week"}[unit]
                return ex.DtField(self.expr(e.args[1], scope), fld)
            if name in ("current_date", "current_timestamp", "now",
                        "getdate"):
                # SPMD-stable: every rank must see the SAME constant, so the
                # value broadcasts from rank 0 when a process group is up
                import pandas as _pdm

                from ..parallel import comm as _comm

                v = _pdm.Timestamp.now()
                if _comm.initialized() and _comm.get_world_size() > 1:
                    v = _comm.allgather_obj(v)[0]
                if name == "current_date":
                    v = v.normalize()
                return ex.Const(v, bt.timestamp_ns)
            if name in ("dateadd", "timestampadd"):
                unit = (e.args[0].value if isinstance(e.args[0], ast.Lit)
                        else e.args[0].name).lower().rstrip("s")
                qty = self.expr(e.args[1], scope)
                base = self.expr(e.args[2], scope)
                ns_per = {"day": 86400 * 10**9, "hour": 3600 * 10**9,
                          "minute": 60 * 10**9, "second": 10**9,
                          "week": 7 * 86400 * 10**9}
                if unit in ("month", "quarter", "year"):
                    mult = {"month": 1, "quarter": 3, "year": 12}[unit]
                    n = _lit_int(e.args[1]) * mult
                    return ex.DtField(base, f"add_months:{n}")
                if unit not in ns_per:
                    raise NotImplementedError(f"DATEADD unit {unit}")
                return ex.BinOp("add", base,
                                ex.BinOp("mul", qty,
                                         ex.Const(ns_per[unit], bt.int64)))
            if name in ("datediff", "timestampdiff"):
                unit = (e.args[0].value if isinstance(e.args[0], ast.Lit)
                        else e.args[0].name).lower().rstrip("s")
                a = self.expr(e.args[1], scope)
                b2 = self.expr(e.args[2], scope)
                if unit == "year":
                    return ex.BinOp("sub", ex.DtField(b2, "year"),
                                    ex.DtField(a, "year"))
                if unit == "month":
                    ydiff = ex.BinOp("sub", ex.DtField(b2, "year"),
                                     ex.DtField(a, "year"))
                    mdiff = ex.BinOp("sub", ex.DtField(b2, "month"),
                                     ex.DtField(a, "month"))
                    return ex.BinOp("add", ex.BinOp("mul", ydiff,
                                                    ex.Const(12)), mdiff)
                if unit == "quarter":
                    ydiff = ex.BinOp("sub", ex.DtField(b2, "year"),
                                     ex.DtField(a, "year"))
                    qdiff = ex.BinOp("sub", ex.DtField(b2, "quarter"),
                                     ex.DtField(a, "quarter"))
                    return ex.BinOp("add", ex.BinOp("mul", ydiff,
                                                    ex.Const(4)), qdiff)
                ns_per = {"day": 86400 * 10**9, "hour": 3600 * 10**9,
                          "minute": 60 * 10**9, "second": 10**9,
                          "week": 7 * 86400 * 10**9}
                if unit not in ns_per:
                    raise NotImplementedError(f"DATEDIFF unit {unit}")
                da = ex.DtField(a, "floor_day") if unit in ("day", "week")                     else a
                db = ex.DtField(b2, "floor_day") if unit in ("day", "week")                     else b2
                return ex.BinOp("floordiv", ex.BinOp("sub", db, da),
                                ex.Const(ns_per[unit], bt.int64))
            if name == "concat":
                args = [self.expr(a, scope) for a in e.args]
                out = args[0]
                for a in args[1:]:
                    out = ex.BinOp("concat", out, a)
                return out
            if name == "reverse":
                return ex.StrOp(self.expr(e.args[0], scope), "reverse")
            if name == "contains":
                return ex.StrOp(self.expr(e.args[0], scope), "contains",
                                (e.args[1].value,))
            if name == "startswith":
                return ex.StrOp(self.expr(e.args[0], scope), "startswith",
                                (e.args[1].value,))
            if name == "endswith":
                return ex.StrOp(self.expr(e.args[0], scope), "endswith",
                                (e.args[1].value,))
            if name in ("regexp_like", "rlike", "regexp"):
                # Snowflake: the pattern must match the WHOLE subject
                return ex.StrOp(self.expr(e.args[0], scope), "match_full",
                                (e.args[1].value,))
            if name == "regexp_count":
                return ex.StrOp(self.expr(e.args[0], scope), "count_re",
                                (e.args[1].value,))
            if name == "regexp_replace":
                rep = e.args[2].value if len(e.args) > 2 else ""
                return ex.StrOp(self.expr(e.args[0], scope), "replace",
                                (e.args[1].value, rep), (("regex", True),))
            if name == "regexp_substr":
                import re as _re

                pat = e.args[1].value
                rx = _re.compile(pat)

                def _sub(v, rx=rx):
                    if v is None:
                        return None
                    m = rx.search(v)
                    return m.group(0) if m else None

                return ex.UdfMap(self.expr(e.args[0], scope), _sub, "ignore")
            if name == "translate":
                tbl_map = str.maketrans(e.args[1].value, e.args[2].value)
                return ex.UdfMap(self.expr(e.args[0], scope),
                                 lambda v, _t=tbl_map: v.translate(_t),
                                 "ignore")
            if name == "ascii":
                return ex.UdfMap(self.expr(e.args[0], scope),
                                 lambda v: ord(v[0]) if v else 0, "ignore")
            if name in ("chr", "char"):
                return ex.UdfMap(self.expr(e.args[0], scope),
                                 lambda v: chr(int(v)), "ignore")
            if name == "space":
                return ex.UdfMap(self.expr(e.args[0], scope),
                                 lambda v: " " * int(v), "ignore")
            if name == "md5":
                import hashlib

                return ex.UdfMap(
                    self.expr(e.args[0], scope),
                    lambda v: hashlib.md5(str(v).encode()).hexdigest(),
                    "ignore")
            if name == "sha1":
                import hashlib

                return ex.UdfMap(
                    self.expr(e.args[0], scope),
                    lambda v: hashlib.sha1(str(v).encode()).hexdigest(),
                    "ignore")
            if name == "sha2":
                import hashlib

                bits = int(e.args[1].value) if len(e.args) > 1 else 256
                algo = {224: hashlib.sha224, 256: hashlib.sha256,
                        384: hashlib.sha384, 512: hashlib.sha512}[bits]
                return ex.UdfMap(
                    self.expr(e.args[0], scope),
                    lambda v, _a=algo: _a(str(v).encode()).hexdigest(),
                    "ignore")
            if name == "base64_encode":
                import base64

                return ex.UdfMap(
                    self.expr(e.args[0], scope),
                    lambda v: base64.b64encode(str(v).encode()).decode(),
                    "ignore")
            if name == "base64_decode_string":
                import base64

                return ex.UdfMap(
                    self.expr(e.args[0], scope),
                    lambda v: base64.b64decode(v).decode(), "ignore")
            if name == "hex_encode":
                return ex.UdfMap(self.expr(e.args[0], scope),
                                 lambda v: str(v).encode().hex().upper(),
                                 "ignore")
            if name == "hash":
                # deterministic per-value FNV-1a (reference role: HASH())
                def _fnv(v):
                    h = 0xcbf29ce484222325
                    for b in str(v).encode():
                        h = ((h ^ b) * 0x100000001b3) & ((1 << 64) - 1)
                    return h - (1 << 64) if h >= (1 << 63) else h

                return ex.UdfMap(self.expr(e.args[0], scope), _fnv, "ignore")
            if name in ("truncate", "trunc") and len(e.args) <= 2 and not (
                    e.args and isinstance(e.args[-1], ast.Lit)
                    and isinstance(e.args[-1].value, str)):
                import math as _math

                d = int(e.args[1].value) if len(e.args) > 1 else 0
                sc = 10.0 ** d

                def _tr(v, _s=sc):
                    # NaN/inf pass through (math.trunc raises on both)
                    if v != v or v in (float("inf"), float("-inf")):
                        return v
                    return _math.trunc(v * _s) / _s

                return ex.UdfMap(self.expr(e.args[0], scope), _tr, None)
            if name in ("sin", "cos", "tan", "asin", "acos", "atan", "sinh",
                        "cosh", "tanh", "degrees", "radians", "cbrt"):
                import math as _math

                f = {"sin": _math.sin, "cos": _math.cos, "tan": _math.tan,
                     "asin": _math.asin, "acos": _math.acos,
                     "atan": _math.atan, "sinh": _math.sinh,
                     "cosh": _math.cosh, "tanh": _math.tanh,
                     "degrees": _math.degrees, "radians": _math.radians,
                     "cbrt": lambda v: _math.copysign(abs(v) ** (1 / 3), v),
                     }[name]
                return ex.UdfMap(self.expr(e.args[0], scope), f, None)
            if name == "pi":
                import math as _math

                return ex.Const(_math.pi, bt.float64)
            if name in ("bitand", "bitor", "bitxor"):
                return ex.BinOp(name, self.expr(e.args[0], scope),
                                self.expr(e.args[1], scope))
            if name in ("bitshiftleft", "bitshiftright"):
                op = "lshift" if name == "bitshiftleft" else "rshift"
                return ex.BinOp(op, self.expr(e.args[0], scope),
                                self.expr(e.args[1], scope))
            if name == "bitnot":
                return ex.BinOp("bitxor", self.expr(e.args[0], scope),
                                ex.Const(-1, bt.int64))
            if name == "nullifzero":
                a = self.expr(e.args[0], scope)
                return ex.Case((ex.Cmp("eq", a, ex.Const(0)),),
                               (ex.Const(None, bt.float64),), a)
            if name == "equal_null":
                a = self.expr(e.args[0], scope)
                b = self.expr(e.args[1], scope)
                return ex.BoolOp("or", ex.Cmp("eq", a, b),
                                 ex.BoolOp("and", ex.IsNull(a),
                                           ex.IsNull(b)))
            if name in ("dayofweek", "dayofweekiso"):
                # pandas convention: Monday=0 (ISO: Monday=1)
                f = ex.DtField(self.expr(e.args[0], scope), "dayofweek")
                return ex.BinOp("add", f, ex.Const(1)) \
                    if name == "dayofweekiso" else f
            if name == "dayofyear":
                return ex.DtField(self.expr(e.args[0], scope), "dayofyear")
            if name in ("week", "weekofyear", "weekiso"):
                d = ex.DtField(self.expr(e.args[0], scope), "dayofyear")
                return ex.BinOp("add",
                                ex.BinOp("floordiv",
                                         ex.BinOp("sub", d, ex.Const(1)),
                                         ex.Const(7)), ex.Const(1))
            if name == "array_size":
                from ..plan.expr import ListOp as _LO

                return _LO(self.expr(e.args[0], scope), "len")
            if name == "get" and len(e.args) == 2:
                # GET(list, i) for lists, GET(struct, 'field') for structs
                from ..plan.expr import ListOp as _LO

                key = e.args[1].value
                return _LO(self.expr(e.args[0], scope), "get",
                           key if isinstance(key, str) else int(key))
            if name in ("json_extract_path_text", "get_path"):
                import json as _json

                path_expr = e.args[1].value

                def _jx(v, _p=path_expr):
                    try:
                        cur = _json.loads(v)
                    except Exception:
                        return None
                    for part in str(_p).replace("[", ".").replace(
                            "]", "").split("."):
                        if part == "":
                            continue
                        if isinstance(cur, list):
                            try:
                                cur = cur[int(part)]
                            except Exception:
                                return None
                        elif isinstance(cur, dict):
                            cur = cur.get(part)
                        else:
                            return None
                        if cur is None:
                            return None
                    if isinstance(cur, (dict, list)):
                        return _json.dumps(cur)
                    return str(cur)

                return ex.UdfMap(self.expr(e.args[0], scope), _jx, "ignore")
            if name in ("dayname", "monthname"):
                # dedicated dt field: a CASE lowering leaked the ELSE value
                # into NULL rows (string CASE has no validity mask)
                return ex.DtField(self.expr(e.args[0], scope), name)
            if name == "decode":
                # DECODE(e, v1, r1 [, v2, r2 ...] [, default]) -> CASE
                operand = self.expr(e.args[0], scope)
                rest = e.args[1:]
                pairs, default = [], None
                if len(rest) % 2 == 1:
                    default = rest[-1]
                    rest = rest[:-1]
                for i in range(0, len(rest), 2):
                    pairs.append((rest[i], rest[i + 1]))
                conds = tuple(ex.Cmp("eq", operand, self.expr(v, scope))
                              for v, _ in pairs)
                thens = tuple(self.expr(r, scope) for _, r in pairs)
                other = (self.expr(default, scope) if default is not None
                         else ex.Const(None, None))
                return ex.Case(conds, thens, other)
            if name == "width_bucket":
                # WIDTH_BUCKET(x, lo, hi, n): 1..n inside, 0 below, n+1 above
                x = self.expr(e.args[0], scope)
                lo = float(_lit_num(e.args[1]))
                hi = float(_lit_num(e.args[2]))
                nb = _lit_int(e.args[3])
                w = (hi - lo) / nb
                raw = ex.BinOp("floordiv",
                               ex.BinOp("sub", x, ex.Const(lo)),
                               ex.Const(w))
                bucket = ex.BinOp("add", raw, ex.Const(1))
                return ex.Case(
                    (ex.Cmp("lt", x, ex.Const(lo)),
                     ex.Cmp("ge", x, ex.Const(hi))),
                    (ex.Const(0), ex.Const(nb + 1)),
                    bucket)
            if name == "last_day":
                return ex.DtField(self.expr(e.args[0], scope), "last_day")
            if name in ("to_date", "try_to_date", "date"):
                return ex.DtField(self.expr(e.args[0], scope), "floor_day")
            if name in ("to_timestamp", "try_to_timestamp",
                        "to_timestamp_ntz"):
                return ex.Cast(self.expr(e.args[0], scope), bt.timestamp_ns,
                               name.startswith("try"))
            if name in ("epoch_second", "date_part_epoch_second"):
                return ex.DtField(self.expr(e.args[0], scope),
                                  "epoch_second")
            if name == "add_months":
                return ex.DtField(self.expr(e.args[0], scope),
                                  f"add_months:{_lit_int(e.args[1])}")
            raise NotImplementedError(f"SQL function {name}")
        if isinstance(e, ast.SubqueryE):
            # uncorrelated scalar subquery in an expression position:
            # evaluates once (memoized in the expr layer)
            full = self._try_plan_full(e.query)
            if full is not None:
                sub_plan, out_names = full
                return ex.ScalarSubquery(sub_plan, out_names[0])
            raise NotImplementedError(
                "correlated scalar subquery in this context")
        raise NotImplementedError(f"expr {e}")


def _interval_to_timestamp_delta(left_ast, qty, unit, op):
    """Constant date +/- interval folds to a Timestamp constant."""
    if isinstance(left_ast, ast.Lit) and left_ast.kind == "date":
        base = pd.Timestamp(left_ast.value)
        off = pd.DateOffset(**{unit + "s": qty})
        res = base + off if op == "add" else base - off
        return ex.Const(pd.Timestamp(res), bt.timestamp_ns)
    return None


def _lit_int(e) -> int:
    """Constant integer from a literal or a unary-minus literal (the parser
    lowers `-3` to Bin('sub', Lit(0), Lit(3)))."""
    if isinstance(e, ast.Lit):
        return int(e.value)
    if isinstance(e, ast.Bin) and e.op == "sub" \
            and isinstance(e.left, ast.Lit) and e.left.value == 0 \
            and isinstance(e.right, ast.Lit):
        return -int(e.right.value)
    raise NotImplementedError(f"constant integer expected, got {e!r}")


# Snowflake alias names -> canonical function names
_FN_ALIASES = {
    "lcase": "lower", "ucase": "upper", "char_length": "length",
    "character_length": "length", "charindex": "position",
    "ifnull": "coalesce", "nvl": "coalesce",
    "ceiling": "ceil", "mod_": "mod",
    "strtok_to_array": "split", "len": "length",
}


def _sub_aliases(e, alias_map, scope):
    """Replace unqualified Col refs that name a SELECT alias (and are not
    real source columns) with the aliased expression — Snowflake HAVING /
    GROUP BY scoping."""
    if isinstance(e, ast.Col) and e.table is None \
            and e.name.lower() in alias_map:
        try:
            scope.resolve(None, e.name)
            return e
        except KeyError:
            return alias_map[e.name.lower()]
    if isinstance(e, (ast.Query, ast.SetOpQ)):
        return e
    for f in getattr(e, "__dataclass_fields__", {}):
        v = getattr(e, f)
        if isinstance(v, list):
            setattr(e, f, [_sub_aliases(x, alias_map, scope)
                           if hasattr(x, "__dataclass_fields__") else x
                           for x in v])
        elif hasattr(v, "__dataclass_fields__"):
            setattr(e, f, _sub_aliases(v, alias_map, scope))
    return e


def _lit_num(e) -> float:
    """Constant number from a literal or unary-minus literal."""
    if isinstance(e, ast.Lit):
        return float(e.value)
    if isinstance(e, ast.Bin) and e.op == "sub" \
            and isinstance(e.left, ast.Lit) and e.left.value == 0 \
            and isinstance(e.right, ast.Lit):
        return -float(e.right.value)
    raise NotImplementedError(f"constant number expected, got {e!r}")


def _lit_expr(e: ast.Lit) -> ex.Expr:
    if e.kind == "date":
        return ex.Const(pd.Timestamp(e.value), bt.timestamp_ns)
    if e.kind == "null":
        return ex.Const(None, bt.float64)
    return ex.Const(e.value)


def _like_expr(operand: ex.Expr, pattern: str) -> ex.Expr:
    body = pattern.strip("%")
    if "%" not in body and "_" not in body:
        if pattern.startswith("%") and pattern.endswith("%"):
            return ex.StrOp(operand, "contains", (body,))
        if pattern.endswith("%"):
            return ex.StrOp(operand, "startswith", (body,))
        if pattern.startswith("%"):
            return ex.StrOp(operand, "endswith", (body,))
        return ex.Cmp("eq", operand, ex.Const(pattern, bt.string))
    import re as _re

    rx = _re.escape(pattern).replace("%", ".*").replace("_", ".")
    return ex.StrOp(operand, "contains_re", ("^" + rx + "$",))


def _extract_single_agg(e):
    """(agg_func_ast, rebuild) where rebuild(value_expr, planner, scope)
    re-applies any arithmetic wrapper; (None, None) if there is no single
    aggregate call."""
    if isinstance(e, ast.Func) and e.name in AGG_FUNCS:
        return e, None
    if isinstance(e, ast.Bin) and e.op in ("add", "sub", "mul", "div"):
        for side, other, right in ((e.left, e.right, True),
                                   (e.right, e.left, False)):
            if isinstance(side, ast.Func) and side.name in AGG_FUNCS                     and isinstance(other, ast.Lit):
                op = e.op

                def rebuild(val, planner, scope, _op=op, _lit=other,
                            _agg_left=not right):
                    lit = planner.expr(_lit, scope)
                    if _agg_left:
                        return ex.BinOp(_op, lit, val)
                    return ex.BinOp(_op, val, lit)

                return side, rebuild
    return None, None


def _col_refs(e) -> list:
    out = []

    def walk(x):
        if isinstance(x, ast.Col):
            out.append(x)
        for f in getattr(x, "__dataclass_fields__", {}):
            v = getattr(x, f)
            if isinstance(v, (list, tuple)):
                for i in v:
                    if hasattr(i, "__dataclass_fields__"):
                        walk(i)
            elif hasattr(v, "__dataclass_fields__"):
                walk(v)

    walk(e)
    return out


def sub_scope_outer(scope, col, q):
    return scope.is_outer(col.table, col.name)


def _split_conjuncts(e) -> list:
    if isinstance(e, ast.Bin) and e.op == "and":
        return _split_conjuncts(e.left) + _split_conjuncts(e.right)
    return [e]


def _listagg_func(sep: str, sort: bool = False, ascending: bool = True):
    """Callable agg for LISTAGG/STRING_AGG (runs on co-located shards via
    the single-phase path); WITHIN GROUP (ORDER BY <same col>) sorts."""

    def listagg(s):
        vals = [v for v in s if v is not None and v == v]
        if sort:
            vals = sorted(vals, reverse=not ascending)
        return sep.join(str(v) for v in vals)

    return listagg


def _has_agg(e) -> bool:
    if isinstance(e, ast.Func) and (
            e.name in AGG_FUNCS or e.name in (
                "listagg", "string_agg", "percentile_cont",
                "percentile_disc")):
        return True
    if isinstance(e, (ast.Query, ast.SetOpQ)):
        return False  # a subquery's aggregates are its own, not the outer's
    for f in getattr(e, "__dataclass_fields__", {}):
        v = getattr(e, f)
        if isinstance(v, (list, tuple)):
            for i in v:
                if hasattr(i, "__dataclass_fields__") and _has_agg(i):
                    return True
                if isinstance(i, tuple):
                    for j in i:
                        if hasattr(j, "__dataclass_fields__") and _has_agg(j):
                            return True
        elif hasattr(v, "__dataclass_fields__") and _has_agg(v):
            return True
    return False


def _ast_key(e) -> str:
    return repr(e)


def _default_name(e) -> str:
    if isinstance(e, ast.Col):
        return e.name
    if isinstance(e, ast.Func):
        return e.name
    if isinstance(e, ast.ExtractE):
        return e.fld
    return "expr"


def _has_window(e) -> bool:
    if isinstance(e, ast.WindowE):
        return True
    if isinstance(e, (ast.Query,)):
        return False
    for f in getattr(e, "__dataclass_fields__", {}):
        v = getattr(e, f)
        if isinstance(v, (list, tuple)):
            for i in v:
                if hasattr(i, "__dataclass_fields__") and _has_window(i):
                    return True
        elif hasattr(v, "__dataclass_fields__") and _has_window(v):
            return True
    return False


def _soundex(v):
    """American Soundex (Snowflake SOUNDEX)."""
    s = "".join(c for c in str(v).upper() if c.isalpha())
    if not s:
        return ""
    codes = {**{c: "1" for c in "BFPV"}, **{c: "2" for c in "CGJKQSXZ"},
             **{c: "3" for c in "DT"}, "L": "4",
             **{c: "5" for c in "MN"}, "R": "6"}
    out = s[0]
    prev = codes.get(s[0], "")
    for c in s[1:]:
        d = codes.get(c, "")
        if d and d != prev:
            out += d
        if c not in "HW":
            prev = d
    return (out + "000")[:4]


def _levenshtein(a: str, b: str) -> int:
    if len(a) < len(b):
        a, b = b, a
    prev = list(range(len(b) + 1))
    for i, ca in enumerate(a, 1):
        cur = [i]
        for j, cb in enumerate(b, 1):
            cur.append(min(prev[j] + 1, cur[j - 1] + 1,
                           prev[j - 1] + (ca != cb)))
        prev = cur
    return prev[-1]


def _jaro_winkler(a: str, b: str) -> float:
    if a == b:
        return 1.0
    la, lb = len(a), len(b)
    if not la or not lb:
        return 0.0
    win = max(la, lb) // 2 - 1
    ma = [False] * la
    mb = [False] * lb
    m = 0
    for i in range(la):
        lo, hi = max(0, i - win), min(lb, i + win + 1)
        for j in range(lo, hi):
            if not mb[j] and a[i] == b[j]:
                ma[i] = mb[j] = True
                m += 1
                break
    if m == 0:
        return 0.0
    t = 0
    k = 0
    for i in range(la):
        if ma[i]:
            while not mb[k]:
                k += 1
            if a[i] != b[k]:
                t += 1
            k += 1
    t //= 2
    jaro = (m / la + m / lb + (m - t) / m) / 3
    pref = 0
    for x, y in zip(a, b):
        if x != y or pref == 4:
            break
        pref += 1
    return jaro + pref * 0.1 * (1 - jaro)
