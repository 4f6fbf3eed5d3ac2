"""Expression IR for projections / filters / aggregations.

Role parallel to the reference's plan expressions (bodo/pandas/plan.py:574-1050:
ColRefExpression, ConstantExpression, ArithOpExpression, ComparisonOpExpression,
PythonScalarFuncExpression, ...), redesigned so a whole projection/filter tree
can be fused into one HIP kernel pass on MI355X (the ``exprvm`` path) instead
of per-op Arrow compute calls.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Callable, Optional, Tuple

from ..core.types import DType


class Expr:
    def _binop(self, op, other, reflect=False):
        other = as_expr(other)
        return BinOp(op, other, self) if reflect else BinOp(op, self, other)

    def __add__(self, o):
        return self._binop("add", o)

    def __radd__(self, o):
        return self._binop("add", o, True)

    def __sub__(self, o):
        return self._binop("sub", o)

    def __rsub__(self, o):
        return self._binop("sub", o, True)

    def __mul__(self, o):
        return self._binop("mul", o)

    def __rmul__(self, o):
        return self._binop("mul", o, True)

    def __truediv__(self, o):
        return self._binop("div", o)

    def __rtruediv__(self, o):
        return self._binop("div", o, True)

    def children(self) -> Tuple["Expr", ...]:
        return ()

    def with_children(self, *ch: "Expr") -> "Expr":
        assert not ch
        return self


@dataclass(frozen=True)
class ColRef(Expr):
    name: str

    def __repr__(self):
        return f"col({self.name})"


@dataclass(frozen=True)
class Const(Expr):
    value: Any
    dtype: Optional[DType] = None

    def __repr__(self):
        return f"const({self.value!r})"


@dataclass(frozen=True)
class BinOp(Expr):
    # op in add/sub/mul/div/floordiv/mod/pow
    op: str
    left: Expr
    right: Expr

    def children(self):
        return (self.left, self.right)

    def with_children(self, *ch):
        return BinOp(self.op, *ch)


@dataclass(frozen=True)
class Cmp(Expr):
    # op in lt/le/gt/ge/eq/ne
    op: str
    left: Expr
    right: Expr

    def children(self):
        return (self.left, self.right)

    def with_children(self, *ch):
        return Cmp(self.op, *ch)


@dataclass(frozen=True)
class BoolOp(Expr):
    # op in and/or
    op: str
    left: Expr
    right: Expr

    def children(self):
        return (self.left, self.right)

    def with_children(self, *ch):
        return BoolOp(self.op, *ch)


@dataclass(frozen=True)
class Not(Expr):
    operand: Expr

    def children(self):
        return (self.operand,)

    def with_children(self, *ch):
        return Not(*ch)


@dataclass(frozen=True)
class IsNull(Expr):
    operand: Expr
    negate: bool = False

    def children(self):
        return (self.operand,)

    def with_children(self, *ch):
        return IsNull(ch[0], self.negate)


@dataclass(frozen=True)
class IsIn(Expr):
    operand: Expr
    values: Tuple[Any, ...]

    def children(self):
        return (self.operand,)

    def with_children(self, *ch):
        return IsIn(ch[0], self.values)


@dataclass(frozen=True)
class DtField(Expr):
    # field in year/month/day/hour/minute/second/dayofweek/date/dayofyear/quarter
    operand: Expr
    fld: str

    def children(self):
        return (self.operand,)

    def with_children(self, *ch):
        return DtField(ch[0], self.fld)


@dataclass(frozen=True)
class Cast(Expr):
    operand: Expr
    to: DType
    safe: bool = False  # TRY_CAST: NULL instead of raising on bad values

    def children(self):
        return (self.operand,)

    def with_children(self, *ch):
        return Cast(ch[0], self.to, self.safe)


@dataclass(frozen=True)
class UdfMap(Expr):
    """Element-wise python UDF (Series.map / row-wise apply collapsed to one
    input). Executed via low-cardinality dictionary evaluation when possible,
    else host round-trip; the @jit path lowers these to HIP (hipRTC)."""

    operand: Expr
    func: Callable = field(compare=False)
    na_action: Optional[str] = None

    def children(self):
        return (self.operand,)

    def with_children(self, *ch):
        return UdfMap(ch[0], self.func, self.na_action)


@dataclass(frozen=True)
class ListOp(Expr):
    """LIST-column operation: len / get(i) (reference role: array_item
    kernels + BodoSQL ARRAY_SIZE/GET)."""

    operand: Expr
    op: str
    arg: Any = None

    def children(self):
        return (self.operand,)

    def with_children(self, *ch):
        return ListOp(ch[0], self.op, self.arg)


@dataclass(frozen=True)
class StrOp(Expr):
    """String method (pandas .str surface); kwargs as a tuple of pairs so
    the node stays hashable."""

    operand: Expr
    op: str
    args: Tuple[Any, ...] = ()
    kwargs: Tuple[Tuple[str, Any], ...] = ()

    def children(self):
        return (self.operand,)

    def with_children(self, *ch):
        return StrOp(ch[0], self.op, self.args)


@dataclass(frozen=True)
class Case(Expr):
    """when/then pairs + else, for SQL CASE and np.where-style selects."""

    conds: Tuple[Expr, ...]
    thens: Tuple[Expr, ...]
    otherwise: Expr

    def children(self):
        return (*self.conds, *self.thens, self.otherwise)

    def with_children(self, *ch):
        k = len(self.conds)
        return Case(tuple(ch[:k]), tuple(ch[k:2 * k]), ch[2 * k])


@dataclass(frozen=True)
class SemiJoinIn(Expr):
    """col.isin(other_series): a semi-join filter.  The frontend rewrites
    Filter(SemiJoinIn) / Filter(Not(SemiJoinIn)) into SEMI/ANTI joins; in
    general expression positions it evaluates against the other side's
    distinct values (allgathered)."""

    operand: Expr
    other_plan: object = field(compare=False, default=None)
    other_expr: Expr = None

    def children(self):
        return (self.operand,)

    def with_children(self, *ch):
        return SemiJoinIn(ch[0], self.other_plan, self.other_expr)


@dataclass(frozen=True)
class ScalarSubquery(Expr):
    """Uncorrelated scalar subquery: executed lazily at evaluation time
    (SPMD-consistent), compared via the scalar fast path."""

    plan: object = field(compare=False, default=None)
    col: str = ""

    def children(self):
        return ()


@dataclass(frozen=True)
class ListBuild(Expr):
    """ARRAY_CONSTRUCT(e1, ..): fixed-arity LIST per row."""

    items: Tuple[Expr, ...] = ()

    def children(self):
        return self.items

    def with_children(self, *ch):
        return ListBuild(tuple(ch))


@dataclass(frozen=True)
class StructBuild(Expr):
    """OBJECT_CONSTRUCT('k', v, ..): STRUCT column per row."""

    names: Tuple[str, ...] = ()
    items: Tuple[Expr, ...] = ()

    def children(self):
        return self.items

    def with_children(self, *ch):
        return StructBuild(self.names, tuple(ch))


@dataclass(frozen=True)
class RandomExpr(Expr):
    """Per-row deterministic pseudo-random int64 stream (SQL RANDOM()):
    seeded so plan re-execution reproduces the same values."""

    seed: int = 0

    def children(self):
        return ()

    def with_children(self, *ch):
        return self


@dataclass(frozen=True)
class RoundExpr(Expr):
    operand: Expr
    decimals: int = 0

    def children(self):
        return (self.operand,)

    def with_children(self, *ch):
        return RoundExpr(ch[0], self.decimals)


def as_expr(v) -> Expr:
    if isinstance(v, Expr):
        return v
    return Const(v)


def expr_columns(e: Expr) -> set:
    """Set of column names referenced by an expression."""
    out = set()

    def walk(x):
        if isinstance(x, ColRef):
            out.add(x.name)
        for c in x.children():
            walk(c)

    walk(e)
    return out


def substitute_cols(e: Expr, mapping) -> Expr:
    """Replace ColRefs by expressions from mapping (for projection pushdown)."""
    if isinstance(e, ColRef) and e.name in mapping:
        return mapping[e.name]
    ch = e.children()
    if not ch:
        return e
    return e.with_children(*[substitute_cols(c, mapping) for c in ch])


def _install_cached_hash_expr():
    """Same per-instance hash memoization as plan nodes (expressions are
    hashed inside every containing plan-node hash)."""
    import sys

    mod = sys.modules[__name__]
    for name in dir(mod):
        cls = getattr(mod, name)
        if isinstance(cls, type) and issubclass(cls, Expr) \
                and cls is not Expr and cls.__hash__ is not None \
                and "__dataclass_fields__" in cls.__dict__:
            orig = cls.__hash__

            def make(orig):
                def __hash__(self):
                    v = self.__dict__.get("_hc")
                    if v is None:
                        v = orig(self)
                        object.__setattr__(self, "_hc", v)
                    return v
                return __hash__

            cls.__hash__ = make(orig)

            def __getstate__(self):
                d = dict(self.__dict__)
                d.pop("_hc", None)  # process-local (salted string hashing)
                return d

            cls.__getstate__ = __getstate__


_install_cached_hash_expr()
