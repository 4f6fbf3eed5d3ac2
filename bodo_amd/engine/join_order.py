"""Greedy join reordering by cardinality estimate (reference analog: the
vendored DuckDB JoinOrderOptimizer; SURVEY §7 step 6 "join ordering by
cardinality estimate").

Scope: clusters of INNER equi-joins whose relations have pairwise-disjoint
column names (the SQL planner's internal naming guarantees this), so
reordering can never change suffix assignment and every downstream
reference is by name.  Estimates are structural: scan row counts from
registry objects / parquet metadata, fixed selectivities for filters and
aggregations.  The greedy order starts from the smallest relation and
repeatedly joins the connected relation with the smallest estimate — the
standard left-deep heuristic that keeps intermediate results near the
small end of the chain.

OPT-IN (BODO_AMD_JOIN_REORDER=1): without NDV statistics the heuristic
can fan out on dimension-dimension edges (e.g. two relations sharing only
a low-cardinality key), so user-written join orders stay authoritative by
default; round 2 adds distinct-count sketches to turn this on.
"""

from __future__ import annotations

from typing import List, Optional, Set, Tuple

from ..plan import nodes as pn
from ..plan.expr import ColRef

_BIG = float(1 << 60)


def _estimate_rows(node: pn.PlanNode) -> float:
    if isinstance(node, pn.PandasScan):
        try:
            from . import executor as ex

            return float(len(ex.get_object(node.data_id)))
        except Exception:
            return _BIG
    if isinstance(node, pn.ParquetScan):
        try:
            import pyarrow.parquet as pq

            return float(pq.ParquetFile(node.path).metadata.num_rows)
        except Exception:
            return _BIG
    if isinstance(node, pn.Filter):
        return 0.3 * _estimate_rows(node.child)
    if isinstance(node, pn.Projection):
        return _estimate_rows(node.child)
    if isinstance(node, pn.Aggregate):
        return 0.1 * _estimate_rows(node.child)
    if isinstance(node, pn.Distinct):
        return 0.5 * _estimate_rows(node.child)
    if isinstance(node, pn.Limit):
        return float(node.n)
    if isinstance(node, pn.Join):
        l = _estimate_rows(node.left)
        r = _estimate_rows(node.right)
        if node.how in ("semi", "anti"):
            return 0.5 * l
        return max(l, r)
    ch = node.children()
    if len(ch) == 1:
        return _estimate_rows(ch[0])
    return _BIG


def _collect_cluster(node: pn.PlanNode, rels: List[pn.PlanNode],
                     edges: List[Tuple[int, int, str, str]]) -> bool:
    """Flatten a tree of inner equi-joins into relations + key edges;
    False when the shape is outside the reorderable subset."""
    if isinstance(node, pn.Join) and node.how == "inner" and node.left_on:
        if not _collect_cluster(node.left, rels, edges):
            return False
        if not _collect_cluster(node.right, rels, edges):
            return False
        # resolve each key pair to its source relation by column name
        for lk, rk in zip(node.left_on, node.right_on):
            li = _rel_of(rels, lk)
            ri = _rel_of(rels, rk)
            if li is None or ri is None:
                return False
            edges.append((li, ri, lk, rk))
        return True
    cols = node.out_columns()
    if cols is None:
        return False
    rels.append(node)
    return True


def _rel_of(rels, col) -> Optional[int]:
    for i, r in enumerate(rels):
        cols = r.out_columns()
        if cols is not None and col in cols:
            return i
    return None


def reorder(node: pn.PlanNode) -> pn.PlanNode:
    node = node.with_children(*[reorder(c) for c in node.children()]) \
        if node.children() else node
    if not (isinstance(node, pn.Join) and node.how == "inner"
            and node.left_on):
        return node
    rels: List[pn.PlanNode] = []
    edges: List[Tuple[int, int, str, str]] = []
    if not _collect_cluster(node, rels, edges) or len(rels) < 3:
        return node
    # disjoint column names required (no suffixes can ever engage)
    seen: Set[str] = set()
    for r in rels:
        cols = r.out_columns()
        if any(c in seen for c in cols):
            return node
        seen.update(cols)
    est = [_estimate_rows(r) for r in rels]
    in_tree = {min(range(len(rels)), key=lambda i: est[i])}
    plan = rels[min(range(len(rels)), key=lambda i: est[i])]
    cur_est = min(est)
    remaining = set(range(len(rels))) - in_tree
    while remaining:
        # candidates connected to the current tree
        cand = [i for i in remaining
                if any((a in in_tree and b == i) or (b in in_tree and a == i)
                       for a, b, _, _ in edges)]
        if not cand:
            return node  # disconnected (cross in the middle): keep original
        nxt = min(cand, key=lambda i: est[i])
        lks, rks = [], []
        for a, b, lk, rk in edges:
            if a in in_tree and b == nxt:
                lks.append(lk)
                rks.append(rk)
            elif b in in_tree and a == nxt:
                lks.append(rk)
                rks.append(lk)
        plan = pn.Join(plan, rels[nxt], tuple(lks), tuple(rks), "inner")
        cur_est = max(cur_est, est[nxt])
        in_tree.add(nxt)
        remaining.discard(nxt)
    # restore the original output column order (user-visible for frames)
    orig = node.out_columns()
    if orig is not None and plan.out_columns() != list(orig):
        plan = pn.Projection(plan, tuple(orig),
                             tuple(ColRef(c) for c in orig))
    return plan
