"""Logical plan optimizer.

Own rule set instead of vendoring DuckDB (reference vendors ~115 kLoC of
DuckDB optimizer at bodo/pandas/vendor/duckdb; SURVEY §7 step 6 calls for a
native rule set): column pruning into scans, filter pushdown into parquet
scans (row-group stats + row-level via Arrow dataset), projection fusion,
filter fusion.
"""

from __future__ import annotations

from typing import List, Optional, Sequence, Set

from ..plan import nodes as pn
from ..plan.expr import (
    BinOp, BoolOp, ColRef, Cmp, Const, Expr, expr_columns, substitute_cols,
)


def optimize(plan: pn.PlanNode) -> pn.PlanNode:
    plan = fuse_projections(plan)
    plan = push_filters(plan)
    import os as _os

    if _os.environ.get("BODO_AMD_JOIN_REORDER", "") == "1":
        from .join_order import reorder

        plan = reorder(plan)
    plan = push_limits(plan)
    plan = prune_columns(plan, None)
    return plan


def push_limits(node: pn.PlanNode) -> pn.PlanNode:
    """head(n) after a computed projection evaluates the expressions over
    only the surviving rows: Limit(Projection) -> Projection(Limit)."""
    node = node.with_children(*[push_limits(c) for c in node.children()]) \
        if node.children() else node
    if isinstance(node, pn.Limit) and not node.tail and \
            isinstance(node.child, pn.Projection):
        proj = node.child
        return pn.Projection(
            pn.Limit(proj.child, node.n, node.offset, node.tail),
            proj.names, proj.exprs)
    return node


# ---------------------------------------------------------------- fusion

def fuse_projections(node: pn.PlanNode) -> pn.PlanNode:
    node = node.with_children(*[fuse_projections(c) for c in node.children()]) \
        if node.children() else node
    if isinstance(node, pn.Projection) and isinstance(node.child, pn.Projection):
        inner = node.child
        mapping = dict(zip(inner.names, inner.exprs))
        new_exprs = tuple(substitute_cols(e, mapping) for e in node.exprs)
        return pn.Projection(inner.child, node.names, new_exprs)
    return node


# ---------------------------------------------------------------- filters

def push_filters(node: pn.PlanNode) -> pn.PlanNode:
    node = node.with_children(*[push_filters(c) for c in node.children()]) \
        if node.children() else node
    if isinstance(node, pn.Filter):
        child = node.child
        # split conjunctions
        conjuncts = _split_and(node.cond)
        if isinstance(child, pn.ParquetScan):
            pushable = [c for c in conjuncts if _arrow_convertible(c)]
            rest = [c for c in conjuncts if not _arrow_convertible(c)]
            if pushable:
                from ..user_logging import log_message

                log_message("Filter Pushdown",
                            f"pushed {len(pushable)} filter(s) into parquet "
                            f"scan of {child.path}")
                new_scan = pn.ParquetScan(
                    child.path, child.columns,
                    child.filters + tuple(pushable), child.schema_names)
                if rest:
                    return pn.Filter(new_scan, _join_and(rest))
                return new_scan
        if isinstance(child, pn.Filter):
            return pn.Filter(child.child, _join_and(
                _split_and(child.cond) + conjuncts))
        if isinstance(child, pn.Projection):
            # push through projection when every referenced column is a
            # plain passthrough ColRef
            mapping = dict(zip(child.names, child.exprs))
            refs = expr_columns(node.cond)
            if all(isinstance(mapping.get(r), ColRef) for r in refs):
                newcond = substitute_cols(node.cond, mapping)
                return pn.Projection(
                    push_filters(pn.Filter(child.child, newcond)),
                    child.names, child.exprs)
        if isinstance(child, pn.Join):
            anti = _left_isnull_to_anti(child, conjuncts)
            if anti is not None:
                return push_filters(anti)
            pushed = _push_into_join(child, conjuncts)
            if pushed is not None:
                return pushed
    return node


def _left_isnull_to_anti(join: pn.Join, conjuncts: List[Expr]):
    """LEFT JOIN + `right_key IS NULL` filter -> ANTI join (reference
    analog: Calcite's outer-join-to-anti rule; TPC-H q22 shape).  Only a
    JOIN KEY being null proves non-match, so the rule keys on right_on
    columns; surviving rows have every right column null, so they are
    re-added as null constants to keep the output schema."""
    from ..plan.expr import IsNull

    if join.how != "left" or not join.right_on:
        return None
    lcols = join.left.out_columns()
    rcols = join.right.out_columns()
    if lcols is None or rcols is None or (set(lcols) & set(rcols)):
        return None  # suffixed overlap: keep it simple, skip
    hit = None
    for i, cj in enumerate(conjuncts):
        if isinstance(cj, IsNull) and not cj.negate and \
                isinstance(cj.operand, ColRef) and \
                cj.operand.name in join.right_on and \
                cj.operand.name in rcols:
            hit = i
            break
    if hit is None:
        return None
    from ..user_logging import log_message

    log_message("Join Rewrite", "LEFT JOIN + IS NULL(right key) -> ANTI join")
    anti = pn.Join(join.left, join.right, join.left_on, join.right_on,
                   "anti", join.suffixes)
    out_cols = join.out_columns()
    from ..core import types as _bt

    exprs = tuple(ColRef(c) if c in set(lcols) else Const(None, _bt.float64)
                  for c in out_cols)
    proj = pn.Projection(anti, tuple(out_cols), exprs)
    rest = conjuncts[:hit] + conjuncts[hit + 1:]
    if rest:
        return pn.Filter(proj, _join_and(rest))
    return proj


def _push_into_join(join: pn.Join, conjuncts: List[Expr]):
    """Push row-local conjuncts below a join to the side whose columns cover
    them (the single biggest intermediate-size lever: TPC-H q5/q8 join
    chains materialize 37-60M rows before late filters without it;
    reference analog: DuckDB FilterPushdown used via the vendored
    optimizer).  Preserved-side rules: a left join keeps left-side pushes
    only, a right join right-side only; FULL OUTER pushes nothing."""
    from ..plan.expr import ScalarSubquery, SemiJoinIn

    lcols = join.left.out_columns()
    rcols = join.right.out_columns()
    if lcols is None or rcols is None:
        return None
    sl, sr = set(lcols), set(rcols)
    overlap = sl & sr
    left_ok = join.how in ("inner", "left", "semi", "anti", "cross")
    right_ok = join.how in ("inner", "right", "cross")

    def has_subplan(e) -> bool:
        if isinstance(e, (ScalarSubquery, SemiJoinIn)):
            return True
        return any(has_subplan(c) for c in e.children())

    push_l: List[Expr] = []
    push_r: List[Expr] = []
    keep: List[Expr] = []
    for cj in conjuncts:
        try:
            refs = expr_columns(cj)
        except Exception:
            refs = set()
        if refs and not (refs & overlap) and not has_subplan(cj):
            if refs <= sl and left_ok:
                push_l.append(cj)
                continue
            if refs <= sr and right_ok:
                push_r.append(cj)
                continue
        keep.append(cj)
    # predicate inference for OR-of-ANDs (TPC-H q7's nation pairs:
    # (n1=A and n2=B) or (n1=B and n2=A)): each side's implied
    # disjunction pushes as a WEAKER derived filter while the original
    # stays above the join
    for cj in keep:
        if not (isinstance(cj, BoolOp) and cj.op == "or"):
            continue
        disjuncts = _split_or(cj)
        for side_set, ok, bucket in ((sl, left_ok, push_l),
                                     (sr, right_ok, push_r)):
            if not ok:
                continue
            per_disjunct = []
            for d in disjuncts:
                side_conj = [c for c in _split_and(d)
                             if expr_columns(c) and
                             expr_columns(c) <= side_set
                             and not has_subplan(c)]
                if not side_conj:
                    per_disjunct = None
                    break
                per_disjunct.append(_join_and(side_conj))
            if per_disjunct:
                derived = per_disjunct[0]
                for d in per_disjunct[1:]:
                    derived = BoolOp("or", derived, d)
                bucket.append(derived)
    if not push_l and not push_r:
        return None
    from ..user_logging import log_message

    log_message("Filter Pushdown",
                f"pushed {len(push_l)}+{len(push_r)} filter(s) below "
                f"{join.how} join")
    nl = push_filters(pn.Filter(join.left, _join_and(push_l))) \
        if push_l else join.left
    nr = push_filters(pn.Filter(join.right, _join_and(push_r))) \
        if push_r else join.right
    new_join = pn.Join(nl, nr, join.left_on, join.right_on, join.how,
                       join.suffixes)
    if keep:
        return pn.Filter(new_join, _join_and(keep))
    return new_join


def _split_or(e: Expr) -> List[Expr]:
    if isinstance(e, BoolOp) and e.op == "or":
        return _split_or(e.left) + _split_or(e.right)
    return [e]


def _split_and(e: Expr) -> List[Expr]:
    if isinstance(e, BoolOp) and e.op == "and":
        return _split_and(e.left) + _split_and(e.right)
    return [e]


def _join_and(es: Sequence[Expr]) -> Expr:
    out = es[0]
    for e in es[1:]:
        out = BoolOp("and", out, e)
    return out


def _arrow_convertible(e: Expr) -> bool:
    """Filters we can hand to the Arrow dataset scanner exactly."""
    if isinstance(e, Cmp):
        lc = isinstance(e.left, ColRef) and isinstance(e.right, Const)
        rc = isinstance(e.right, ColRef) and isinstance(e.left, Const)
        return lc or rc
    from ..plan.expr import IsIn, IsNull

    if isinstance(e, IsIn):
        return isinstance(e.operand, ColRef)
    if isinstance(e, IsNull):
        return isinstance(e.operand, ColRef)
    if isinstance(e, BoolOp):
        return _arrow_convertible(e.left) and _arrow_convertible(e.right)
    return False


# ---------------------------------------------------------------- pruning

def prune_columns(node: pn.PlanNode, required: Optional[Set[str]]) -> pn.PlanNode:
    """Rewrite the tree so each subtree produces at least `required` columns
    (None = all).  Narrows ParquetScan.columns for IO pruning."""
    if isinstance(node, pn.ParquetScan):
        if required is None:
            return node
        schema = list(node.schema_names) or None
        filt_cols = set()
        for f in node.filters:
            filt_cols |= expr_columns(f)
        want = set(required) | filt_cols
        if schema:
            cols = tuple(c for c in schema if c in want)
        else:
            cols = tuple(sorted(want))
        return pn.ParquetScan(node.path, cols, node.filters, node.schema_names)
    if isinstance(node, pn.CsvScan):
        if required is None:
            return node
        schema = list(node.schema_names) or None
        if schema:
            cols = tuple(c for c in schema if c in required)
        else:
            cols = tuple(sorted(required))
        return pn.CsvScan(node.path, node.options, cols, node.schema_names)
    if isinstance(node, pn.PandasScan):
        if required is None or not node.names:
            return node
        keep = [c for c in node.names if c in required]
        if len(keep) == len(node.names) or not keep:
            return node
        from ..plan.expr import ColRef as _CR

        return pn.Projection(node, tuple(keep), tuple(_CR(c) for c in keep))
    if isinstance(node, pn.Projection):
        if required is not None:
            keep = [(n, e) for n, e in zip(node.names, node.exprs) if n in required]
            if not keep:  # keep at least one column for row count
                keep = [(node.names[0], node.exprs[0])] if node.names else []
        else:
            keep = list(zip(node.names, node.exprs))
        child_req = set()
        for _, e in keep:
            child_req |= expr_columns(e)
        child = prune_columns(node.child, child_req)
        return pn.Projection(child, tuple(n for n, _ in keep),
                             tuple(e for _, e in keep))
    if isinstance(node, pn.Filter):
        child_req = None if required is None else set(required) | expr_columns(node.cond)
        return pn.Filter(prune_columns(node.child, child_req), node.cond)
    if isinstance(node, pn.Aggregate):
        child_req = set(node.keys) | {a[1] for a in node.aggs if a[1]}
        return node.with_children(prune_columns(node.child, child_req))
    if isinstance(node, pn.Reduce):
        child_req = {a[1] for a in node.aggs if a[1]}
        return node.with_children(prune_columns(node.child, child_req))
    if isinstance(node, pn.Window):
        child_req = None
        if required is not None:
            child_req = (set(required) | set(node.keys) | set(node.order_by)
                         | {s[1] for s in node.specs if s[1]})
            child_req -= {s[0] for s in node.specs}
        return node.with_children(prune_columns(node.child, child_req))
    if isinstance(node, (pn.Sort, pn.Limit, pn.Distinct, pn.Sample)):
        req2 = None
        if required is not None:
            req2 = set(required)
            if isinstance(node, pn.Sort):
                req2 |= set(node.keys)
            if isinstance(node, pn.Distinct) and node.subset:
                req2 |= set(node.subset)
        return node.with_children(prune_columns(node.children()[0], req2))
    if isinstance(node, pn.Join):
        lcols = node.left.out_columns()
        rcols = node.right.out_columns()
        if required is None or lcols is None or rcols is None:
            lreq = rreq = None
        else:
            sl, sr = set(lcols), set(rcols)
            sfx_l, sfx_r = node.suffixes
            lreq, rreq = set(node.left_on), set(node.right_on)
            for r in required:
                base_l = r[:-len(sfx_l)] if sfx_l and r.endswith(sfx_l) else None
                base_r = r[:-len(sfx_r)] if sfx_r and r.endswith(sfx_r) else None
                if r in sl:
                    lreq.add(r)
                if r in sr:
                    rreq.add(r)
                # a suffixed requirement needs the base column on BOTH sides
                # so the runtime overlap (and hence the suffix) is preserved
                if base_l and base_l in sl and base_l in sr:
                    lreq.add(base_l)
                    rreq.add(base_l)
                if base_r and base_r in sr and base_r in sl:
                    lreq.add(base_r)
                    rreq.add(base_r)
        return pn.Join(
            prune_columns(node.left, lreq), prune_columns(node.right, rreq),
            node.left_on, node.right_on, node.how, node.suffixes)
    if isinstance(node, pn.Union):
        return node.with_children(*[prune_columns(c, required) for c in node.inputs])
    if isinstance(node, pn.RowId):
        req2 = None if required is None else set(required) - {node.name}
        return node.with_children(prune_columns(node.child, req2))
    # unknown shape: don't prune below
    ch = node.children()
    if not ch:
        return node
    return node.with_children(*[prune_columns(c, None) for c in ch])
