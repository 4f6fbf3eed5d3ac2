"""Logical plan nodes (reference: bodo/pandas/plan.py:305-573 LogicalOperator
hierarchy).  A plan is an immutable tree; the optimizer rewrites it; the
executor (bodo_amd/engine/executor.py) interprets the optimized tree against
the rank-local shard with distributed exchanges inserted per operator."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional, Sequence, Tuple

from .expr import Expr


class PlanNode:
    def children(self) -> Tuple["PlanNode", ...]:
        return ()

    def with_children(self, *ch: "PlanNode") -> "PlanNode":
        assert not ch
        return self

    # schema: list of column names this node produces (best-effort; scans
    # fill from file metadata, other nodes derive)
    def out_columns(self) -> Optional[List[str]]:
        return None


# ---------------------------------------------------------------- sources
@dataclass(frozen=True)
class ParquetScan(PlanNode):
    path: str
    columns: Optional[Tuple[str, ...]] = None  # projection pushdown target
    filters: Tuple[Expr, ...] = ()  # filter pushdown (row-group stats + row-level)
    schema_names: Tuple[str, ...] = ()  # discovered at plan build

    def out_columns(self):
        return list(self.columns) if self.columns else list(self.schema_names)


@dataclass(frozen=True)
class CsvScan(PlanNode):
    path: str
    options: Tuple[Tuple[str, Any], ...] = ()
    columns: Optional[Tuple[str, ...]] = None
    schema_names: Tuple[str, ...] = ()

    def out_columns(self):
        return list(self.columns) if self.columns else list(self.schema_names)


@dataclass(frozen=True)
class PandasScan(PlanNode):
    """In-memory source: a host pandas DataFrame (replicated or to scatter) or
    an already-distributed Table shard registered in the result registry."""

    data_id: str  # key into the executor's object registry
    names: Tuple[str, ...] = ()
    distributed: bool = False  # True: registry holds this rank's shard

    def out_columns(self):
        return list(self.names)


# ---------------------------------------------------------------- unary
@dataclass(frozen=True)
class Projection(PlanNode):
    child: PlanNode
    names: Tuple[str, ...] = ()
    exprs: Tuple[Expr, ...] = ()

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return Projection(ch[0], self.names, self.exprs)

    def out_columns(self):
        return list(self.names)


@dataclass(frozen=True)
class Filter(PlanNode):
    child: PlanNode
    cond: Expr = None

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return Filter(ch[0], self.cond)

    def out_columns(self):
        return self.child.out_columns()


@dataclass(frozen=True)
class Aggregate(PlanNode):
    child: PlanNode
    keys: Tuple[str, ...] = ()
    # aggs: (out_name, in_name, func); func in
    # sum/count/mean/min/max/size/nunique/first/last/var/std/median/prod
    aggs: Tuple[Tuple[str, str, str], ...] = ()
    as_index: bool = False
    dropna: bool = True

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return Aggregate(ch[0], self.keys, self.aggs, self.as_index, self.dropna)

    def out_columns(self):
        return list(self.keys) + [a[0] for a in self.aggs]


@dataclass(frozen=True)
class Sort(PlanNode):
    child: PlanNode
    keys: Tuple[str, ...] = ()
    ascending: Tuple[bool, ...] = ()
    na_position: str = "last"

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return Sort(ch[0], self.keys, self.ascending, self.na_position)

    def out_columns(self):
        return self.child.out_columns()


@dataclass(frozen=True)
class Limit(PlanNode):
    child: PlanNode
    n: int = 0
    offset: int = 0
    tail: bool = False

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return Limit(ch[0], self.n, self.offset, self.tail)

    def out_columns(self):
        return self.child.out_columns()


@dataclass(frozen=True)
class Distinct(PlanNode):
    child: PlanNode
    subset: Optional[Tuple[str, ...]] = None
    keep: str = "first"

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return Distinct(ch[0], self.subset, self.keep)

    def out_columns(self):
        return self.child.out_columns()


@dataclass(frozen=True)
class Explode(PlanNode):
    """Each element of a LIST column becomes a row (pandas explode /
    LATERAL FLATTEN; reference: bodo/libs/_lateral.cpp).  pos names an
    optional element-index output column (FLATTEN .index)."""
    child: PlanNode
    column: str
    pos: Optional[str] = None

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return Explode(ch[0], self.column, self.pos)

    def out_columns(self):
        return self.child.out_columns()


@dataclass(frozen=True)
class Sample(PlanNode):
    child: PlanNode
    n: Optional[int] = None
    frac: Optional[float] = None
    seed: Optional[int] = None

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return Sample(ch[0], self.n, self.frac, self.seed)

    def out_columns(self):
        return self.child.out_columns()


@dataclass(frozen=True)
class MapPartitions(PlanNode):
    """Run a python function over each rank's shard as a pandas DataFrame
    (reference: frame.py map_partitions / apply axis=1 fallback)."""

    child: PlanNode
    func: Callable = field(compare=False)
    args: Tuple[Any, ...] = ()
    names: Tuple[str, ...] = ()

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return MapPartitions(ch[0], self.func, self.args, self.names)

    def out_columns(self):
        return list(self.names) if self.names else None


@dataclass(frozen=True)
class RowId(PlanNode):
    """Append a globally-unique int64 row-id column (rank-offset arange).
    Used by the SQL planner's row-id decorrelation of EXISTS subqueries with
    non-equality correlated predicates (reference analog: Calcite's
    RelDecorrelator as used by BodoSQL)."""

    child: PlanNode
    name: str = "__rid"

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return RowId(ch[0], self.name)

    def out_columns(self):
        cols = self.child.out_columns()
        return None if cols is None else list(cols) + [self.name]


@dataclass(frozen=True)
class Rolling(PlanNode):
    """Rolling-window aggregation over the global row order.  Distributed
    execution exchanges a (window-1)-row halo from preceding ranks so shard
    boundaries produce exactly the single-process result (reference:
    hiframes rolling support in @bodo.jit)."""

    child: PlanNode
    window: int = 2
    min_periods: Optional[int] = None
    specs: Tuple[Tuple[str, str, str], ...] = ()  # (out, in, func)

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return Rolling(ch[0], self.window, self.min_periods, self.specs)

    def out_columns(self):
        return [s[0] for s in self.specs]


@dataclass(frozen=True)
class Shift(PlanNode):
    """Global-order shift: boundary rows exchange between neighboring
    ranks (reference: dist shift in bodo/libs/array_kernels.py +
    Sendrecv in _distributed.h)."""

    child: PlanNode
    periods: int = 1
    specs: Tuple[Tuple[str, str], ...] = ()  # (out, in)

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return Shift(ch[0], self.periods, self.specs)

    def out_columns(self):
        # passes the child's columns through plus the shifted outputs so
        # expressions can combine original and shifted values (diff)
        cols = self.child.out_columns()
        outs = [s[0] for s in self.specs]
        return None if cols is None else list(cols) + outs


@dataclass(frozen=True)
class Fill(PlanNode):
    """Forward/backward fill over the global row order with cross-rank
    boundary carry (reference: array_kernels fillna method='ffill')."""

    child: PlanNode
    forward: bool = True
    specs: Tuple[Tuple[str, str], ...] = ()  # (out, in)

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return Fill(ch[0], self.forward, self.specs)

    def out_columns(self):
        cols = self.child.out_columns()
        outs = [s[0] for s in self.specs]
        return None if cols is None else list(cols) + [o for o in outs
                                                       if o not in cols]


@dataclass(frozen=True)
class Cumulative(PlanNode):
    """Global-order cumulative ops (cumsum/cumprod/cummin/cummax) over the
    distributed row order: local scan + an exscan of shard totals
    (reference: dist_exscan in bodo/transforms/distributed_pass.py,
    _distributed.h Exscan)."""

    child: PlanNode
    specs: Tuple[Tuple[str, str, str], ...] = ()  # (out, in, func)

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return Cumulative(ch[0], self.specs)

    def out_columns(self):
        return [s[0] for s in self.specs]


@dataclass(frozen=True)
class ShuffleByKey(PlanNode):
    """Explicit hash-repartition so equal keys co-locate on one rank
    (used by groupby.apply / median paths)."""

    child: PlanNode
    keys: Tuple[str, ...] = ()

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return ShuffleByKey(ch[0], self.keys)

    def out_columns(self):
        return self.child.out_columns()


@dataclass(frozen=True)
class Window(PlanNode):
    """Window/transform functions over partitions (reference:
    bodo/libs/window/_window_calculator.cpp).  specs = (out_name, in_name,
    func, arg) with func in transform_sum/transform_mean/transform_min/
    transform_max/transform_count/row_number/rank/dense_rank/shift/cumsum/
    cumcount."""

    child: PlanNode
    keys: Tuple[str, ...] = ()
    order_by: Tuple[str, ...] = ()
    ascending: Tuple[bool, ...] = ()
    specs: Tuple[Tuple[str, str, str, Any], ...] = ()

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return Window(ch[0], self.keys, self.order_by, self.ascending,
                      self.specs)

    def out_columns(self):
        base = self.child.out_columns()
        if base is None:
            return None
        return list(base) + [s[0] for s in self.specs]


# ---------------------------------------------------------------- binary
@dataclass(frozen=True)
class Join(PlanNode):
    left: PlanNode
    right: PlanNode
    left_on: Tuple[str, ...] = ()
    right_on: Tuple[str, ...] = ()
    how: str = "inner"  # inner/left/right/outer/semi/anti/cross
    suffixes: Tuple[str, str] = ("_x", "_y")

    def children(self):
        return (self.left, self.right)

    def with_children(self, *ch):
        return Join(ch[0], ch[1], self.left_on, self.right_on, self.how, self.suffixes)

    def out_columns(self):
        if self.how in ("semi", "anti"):
            return self.left.out_columns()
        lcols = self.left.out_columns()
        rcols = self.right.out_columns()
        if lcols is None or rcols is None:
            return None
        shared = {k for k, rk in zip(self.left_on, self.right_on) if k == rk}
        lset, rset = set(lcols), set(rcols)
        out = []
        for c in lcols:
            out.append(c + self.suffixes[0]
                       if (c in rset and c not in shared) else c)
        for c in rcols:
            if c in shared:
                continue
            out.append(c + self.suffixes[1] if c in lset else c)
        return out


@dataclass(frozen=True)
class Union(PlanNode):
    inputs: Tuple[PlanNode, ...] = ()
    distinct: bool = False

    def children(self):
        return self.inputs

    def with_children(self, *ch):
        return Union(tuple(ch), self.distinct)

    def out_columns(self):
        return self.inputs[0].out_columns()


# ---------------------------------------------------------------- sinks
@dataclass(frozen=True)
class ParquetWrite(PlanNode):
    child: PlanNode
    path: str = ""
    compression: Optional[str] = "snappy"
    partition_cols: Tuple[str, ...] = ()

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return ParquetWrite(ch[0], self.path, self.compression,
                            self.partition_cols)


@dataclass(frozen=True)
class IcebergWrite(PlanNode):
    """Transactional snapshot write to a filesystem Iceberg table
    (reference: frame.py to_iceberg, bodo/io/iceberg/write.py)."""

    child: PlanNode
    path: str = ""
    mode: str = "create"  # create | replace | append

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return IcebergWrite(ch[0], self.path, self.mode)


@dataclass(frozen=True)
class Reduce(PlanNode):
    """Whole-column reductions producing a scalar row (Series.sum() etc.)."""

    child: PlanNode
    aggs: Tuple[Tuple[str, str, str], ...] = ()  # (out, in, func)

    def children(self):
        return (self.child,)

    def with_children(self, *ch):
        return Reduce(ch[0], self.aggs)

    def out_columns(self):
        return [a[0] for a in self.aggs]


def walk(node: PlanNode):
    yield node
    for c in node.children():
        yield from walk(c)


def explain(node: PlanNode, indent: int = 0) -> str:
    """Pretty-print a plan tree (reference: plan dumps at
    tracing_level>=2, bodo/pandas/plan.py:1090)."""
    pad = "  " * indent
    name = type(node).__name__
    detail = ""
    if isinstance(node, Projection):
        detail = f" cols={list(node.names)[:8]}{'...' if len(node.names) > 8 else ''}"
    elif isinstance(node, Filter):
        detail = f" cond={node.cond!r}"[:120]
    elif isinstance(node, Aggregate):
        detail = f" keys={list(node.keys)} aggs={[a[0] for a in node.aggs]}"
    elif isinstance(node, Join):
        detail = f" how={node.how} on={list(node.left_on)}={list(node.right_on)}"
    elif isinstance(node, Sort):
        detail = f" keys={list(node.keys)}"
    elif isinstance(node, ParquetScan):
        detail = f" path={node.path} cols={node.columns} nfilters={len(node.filters)}"
    elif isinstance(node, PandasScan):
        detail = f" id={node.data_id[:8]} cols={list(node.names)[:6]}"
    elif isinstance(node, Limit):
        detail = f" n={node.n}"
    elif isinstance(node, Window):
        detail = f" keys={list(node.keys)} specs={[s[0] for s in node.specs]}"
    lines = [f"{pad}{name}{detail}"]
    for c in node.children():
        lines.append(explain(c, indent + 1))
    return "\n".join(lines)


def _install_cached_hash():
    """Memoize dataclass __hash__ per instance: the executor's CTE memo and
    optimizer rules hash plan subtrees at every visit, which is O(plan^2)
    per query with the generated recursive hash (round-2 host-overhead
    finding: _exec tottime was dominated by hashing on q5/q22)."""
    import sys

    mod = sys.modules[__name__]
    for name in dir(mod):
        cls = getattr(mod, name)
        if isinstance(cls, type) and issubclass(cls, PlanNode) \
                and cls is not PlanNode and cls.__hash__ is not None \
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


_install_cached_hash()
