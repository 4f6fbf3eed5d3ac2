"""Parfor-style loop vectorization for @bodo_amd.jit (reference:
BodoSeriesPass + parfor lowering, bodo/transforms/series_pass.py +
distributed_pass.py _run_parfor).

`for i in prange(n): out[i] = f(a[i], b[i], i, scalars)` rewrites into
whole-array expressions over the block-distributed DistArray layer (torch
kernels on the rank shard); `acc += expr` accumulators become global
reductions (the parfor reduction -> dist_reduce lowering).  Any loop that
doesn't match the elementwise pattern is left as-is — `prange` then runs
the rank-local block SPMD-correctly, just without vectorization."""

from __future__ import annotations

import ast
import copy
from typing import Dict, List, Optional, Set


def _is_range_call(node) -> Optional[ast.expr]:
    """Matches range(n) / prange(n) / bodo_amd.prange(n); returns n."""
    if not isinstance(node, ast.Call) or len(node.args) != 1 or node.keywords:
        return None
    f = node.func
    name = None
    if isinstance(f, ast.Name):
        name = f.id
    elif isinstance(f, ast.Attribute):
        name = f.attr
    if name in ("range", "prange"):
        return node.args[0]
    return None


class _ElemRewriter(ast.NodeTransformer):
    """Rewrite a loop-body expression: a[i] -> a, bare i -> __baj_idx;
    records whether anything unsupported appears."""

    def __init__(self, ivar: str):
        self.ivar = ivar
        self.ok = True
        self.used_index = False
        self.used_elem = False

    def visit_Subscript(self, node):
        idx = node.slice
        if isinstance(idx, ast.Name) and idx.id == self.ivar:
            self.used_elem = True
            return self.visit(node.value)
        self.generic_visit(node)
        return node

    def visit_Name(self, node):
        if node.id == self.ivar:
            self.used_index = True
            return ast.copy_location(
                ast.Name(id="__baj_idx", ctx=ast.Load()), node)
        return node

    def visit_Call(self, node):  # math funcs over elements are fine (ufuncs)
        self.generic_visit(node)
        return node

    def visit_For(self, node):  # nested loops: bail
        self.ok = False
        return node

    def visit_While(self, node):
        self.ok = False
        return node


def _vectorize_loop(node: ast.For) -> Optional[List[ast.stmt]]:
    if node.orelse or not isinstance(node.target, ast.Name):
        return None
    n_expr = _is_range_call(node.iter)
    if n_expr is None:
        return None
    ivar = node.target.id
    out: List[ast.stmt] = []
    needs_idx = False
    for stmt in node.body:
        if isinstance(stmt, ast.Assign) and len(stmt.targets) == 1 \
                and isinstance(stmt.targets[0], ast.Subscript) \
                and isinstance(stmt.targets[0].slice, ast.Name) \
                and stmt.targets[0].slice.id == ivar \
                and isinstance(stmt.targets[0].value, ast.Name):
            # out[i] = expr  ->  out = expr_vec
            rw = _ElemRewriter(ivar)
            val = rw.visit(copy.deepcopy(stmt.value))
            if not rw.ok or not (rw.used_elem or rw.used_index):
                return None  # constant/side-effect bodies stay loops
            needs_idx |= rw.used_index
            out.append(ast.Assign(
                targets=[ast.Name(id=stmt.targets[0].value.id,
                                  ctx=ast.Store())],
                value=val))
        elif isinstance(stmt, ast.AugAssign) \
                and isinstance(stmt.target, ast.Name) \
                and isinstance(stmt.op, ast.Add):
            # acc += expr  ->  acc = acc + __baj_gsum(expr_vec)
            rw = _ElemRewriter(ivar)
            val = rw.visit(copy.deepcopy(stmt.value))
            if not rw.ok or not (rw.used_elem or rw.used_index):
                # body doesn't touch the loop index: NOT an elementwise
                # statement (vectorizing would collapse the iteration count)
                return None
            needs_idx |= rw.used_index
            out.append(ast.Assign(
                targets=[ast.Name(id=stmt.target.id, ctx=ast.Store())],
                value=ast.BinOp(
                    left=ast.Name(id=stmt.target.id, ctx=ast.Load()),
                    op=ast.Add(),
                    right=ast.Call(
                        func=ast.Name(id="__baj_gsum", ctx=ast.Load()),
                        args=[val], keywords=[]))))
        else:
            return None
    if needs_idx:
        out.insert(0, ast.Assign(
            targets=[ast.Name(id="__baj_idx", ctx=ast.Store())],
            value=ast.Call(func=ast.Name(id="__baj_arange", ctx=ast.Load()),
                           args=[copy.deepcopy(n_expr)], keywords=[])))
    return out


class _LoopPass(ast.NodeTransformer):
    def __init__(self):
        self.count = 0

    def visit_For(self, node):
        self.generic_visit(node)
        repl = _vectorize_loop(node)
        if repl is None:
            return node
        self.count += 1
        for s in repl:
            ast.copy_location(s, node)
        return repl


def _gsum(x):
    """Global reduction of a vectorized accumulator expression."""
    from .distarray import DistArray

    if isinstance(x, DistArray):
        return x.sum()
    import numpy as np

    return float(np.sum(x))


def _arange_dist(n):
    from ..parallel import comm
    from .distarray import DistArray

    n = int(n)
    if comm.get_world_size() > 1 or n >= 1024:
        return DistArray.arange(n)
    import numpy as np

    return np.arange(n)


def vectorize_fn(fn, new_globals: Dict) -> Optional[object]:
    """Return a compiled clone of fn with elementwise prange/range loops
    vectorized, or None when the source is unavailable or no loop
    matched."""
    import inspect
    import textwrap
    import types

    try:
        src = textwrap.dedent(inspect.getsource(fn))
        tree = ast.parse(src)
    except (OSError, SyntaxError, TypeError):
        return None
    fdef = tree.body[0]
    if not isinstance(fdef, (ast.FunctionDef, ast.AsyncFunctionDef)):
        return None
    fdef.decorator_list = []
    p = _LoopPass()
    p.visit(fdef)
    if p.count == 0:
        return None
    ast.fix_missing_locations(tree)
    g = dict(new_globals)
    g["__baj_gsum"] = _gsum
    g["__baj_arange"] = _arange_dist
    code = compile(tree, filename=f"<bodo_amd.jit {fn.__name__}>",
                   mode="exec")
    ns: Dict = {}
    exec(code, g, ns)
    clone = ns[fdef.name]
    clone.__defaults__ = fn.__defaults__
    clone.__kwdefaults__ = fn.__kwdefaults__
    return clone
