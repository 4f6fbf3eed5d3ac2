"""Distribution analysis for @bodo_amd.jit (reference:
bodo/transforms/distributed_analysis.py:83 — the REP / 1D / 1D_Var
meet-semilattice fixpoint over the function IR).  Here the IR is the
function AST: parameters seed the lattice from their runtime types,
assignments propagate distributions through expressions, reductions and
len() produce REP scalars, boolean selection produces 1D_Var.  The result
drives argument scatter decisions and the distributed_diagnostics report;
runtime semantics are enforced by the DistArray/BodoDataFrame objects
themselves (no IR lowering — the numpy/pandas protocol layer is the
lowering)."""

from __future__ import annotations

import ast
import inspect
import textwrap
from enum import IntEnum
from typing import Dict, Optional


class Dist(IntEnum):
    REP = 1
    ONED_VAR = 4
    ONED = 5


_REDUCERS = {"sum", "mean", "min", "max", "prod", "std", "var", "dot",
             "nunique", "count", "len"}
_CREATORS = {"arange", "zeros", "ones", "empty", "full", "ranf", "rand",
             "random", "read_parquet", "read_csv", "from_pandas"}


def _meet(a: Optional[Dist], b: Optional[Dist]) -> Optional[Dist]:
    if a is None:
        return b
    if b is None:
        return a
    return Dist(min(a, b))


def _combine(a: Optional[Dist], b: Optional[Dist]) -> Optional[Dist]:
    """Expression combining: REP operands broadcast against arrays (scalar
    + 1D stays 1D); two arrays meet on the lattice."""
    if a is None:
        return b
    if b is None:
        return a
    if a == Dist.REP:
        return b
    if b == Dist.REP:
        return a
    return Dist(min(a, b))


class _Pass(ast.NodeVisitor):
    def __init__(self, arg_dists: Dict[str, Dist]):
        self.dist: Dict[str, Dist] = dict(arg_dists)
        self.changed = False

    def expr_dist(self, e) -> Optional[Dist]:
        if isinstance(e, ast.Name):
            return self.dist.get(e.id)
        if isinstance(e, ast.Constant):
            return Dist.REP
        if isinstance(e, ast.BinOp):
            return _combine(self.expr_dist(e.left), self.expr_dist(e.right))
        if isinstance(e, ast.UnaryOp):
            return self.expr_dist(e.operand)
        if isinstance(e, ast.Compare):
            d = self.expr_dist(e.left)
            for c in e.comparators:
                d = _combine(d, self.expr_dist(c))
            return d
        if isinstance(e, ast.Subscript):
            base = self.expr_dist(e.value)
            if base in (Dist.ONED, Dist.ONED_VAR):
                idx = e.slice
                idx_d = self.expr_dist(idx)
                if idx_d in (Dist.ONED, Dist.ONED_VAR):
                    return Dist.ONED_VAR  # boolean selection
                return Dist.REP  # scalar element
            return base
        if isinstance(e, ast.Call):
            fname = None
            if isinstance(e.func, ast.Attribute):
                fname = e.func.attr
                recv = self.expr_dist(e.func.value)
                if fname in _REDUCERS:
                    return Dist.REP
                if fname in _CREATORS:
                    return Dist.ONED
                if recv in (Dist.ONED, Dist.ONED_VAR):
                    return recv  # method chain stays distributed
            elif isinstance(e.func, ast.Name):
                fname = e.func.id
                if fname == "len":
                    return Dist.REP
                if fname in _CREATORS:
                    return Dist.ONED
            args_d = None
            for a in e.args:
                args_d = _combine(args_d, self.expr_dist(a))
            if fname in _REDUCERS:
                return Dist.REP
            return args_d
        if isinstance(e, (ast.Attribute,)):
            return self.expr_dist(e.value)
        if isinstance(e, ast.IfExp):
            return _combine(self.expr_dist(e.body), self.expr_dist(e.orelse))
        return None

    def visit_Assign(self, node):
        d = self.expr_dist(node.value)
        for t in node.targets:
            if isinstance(t, ast.Name) and d is not None:
                old = self.dist.get(t.id)
                new = _meet(old, d) if old is not None else d
                if new != old:
                    self.dist[t.id] = new
                    self.changed = True
        self.generic_visit(node)

    def visit_AugAssign(self, node):
        if isinstance(node.target, ast.Name):
            d = _meet(self.dist.get(node.target.id),
                      self.expr_dist(node.value))
            if d is not None and d != self.dist.get(node.target.id):
                self.dist[node.target.id] = d
                self.changed = True
        self.generic_visit(node)


def analyze(fn, arg_dists: Dict[str, Dist]):
    """Fixpoint distribution inference; returns ({var: Dist}, report)."""
    try:
        src = textwrap.dedent(inspect.getsource(fn))
        tree = ast.parse(src)
    except (OSError, SyntaxError, TypeError):
        return dict(arg_dists), "(source unavailable: runtime dists only)"
    p = _Pass(arg_dists)
    for _ in range(20):  # fixpoint (monotone lattice, terminates fast)
        p.changed = False
        p.visit(tree)
        if not p.changed:
            break
    lines = [f"Distributed analysis for {fn.__name__}:"]
    for name, d in sorted(p.dist.items()):
        lines.append(f"  {name:24s} {d.name}")
    return p.dist, "\n".join(lines)
