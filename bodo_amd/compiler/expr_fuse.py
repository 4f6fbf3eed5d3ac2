"""Fused expression execution: compile a whole projection/filter expression
list into ONE hipRTC kernel reading each input column once (the MI355X
answer to the reference's per-op Arrow compute calls,
bodo/pandas/physical/expression.h — HBM traffic is the bound, so fusing
elementwise work into a single pass is the first-order win).

Scope: fixed-width numeric/bool/temporal inputs without validity masks
(float NaN propagates naturally).  Anything else falls back to the unfused
evaluator — the caller treats None/exception as "use the per-op path".
Numerics: int expressions evaluate in int64 (wraparound identical to torch),
float in double; comparisons involving NaN are false (C semantics = pandas).
"""

from __future__ import annotations

import ctypes
from typing import Dict, List, Optional, Tuple

import torch

from ..core import types as bt
from ..core.column import Column
from ..core.table import Table
from ..core.types import TypeKind
from ..plan.expr import (
    BinOp, BoolOp, Case, ColRef, Cmp, Const, DtField, Expr, IsIn, Not,
    RoundExpr,
)

_CTYPE_OF_KIND = {
    TypeKind.INT8: "signed char", TypeKind.INT16: "short",
    TypeKind.INT32: "int", TypeKind.INT64: "long long",
    TypeKind.FLOAT32: "float", TypeKind.FLOAT64: "double",
    TypeKind.BOOL: "unsigned char", TypeKind.DATE32: "int",
    TypeKind.TIMESTAMP_NS: "long long", TypeKind.DICT: "int",
}

_DT_FIELDS_OK = {"year", "month", "day", "hour", "minute", "second",
                 "dayofweek", "weekday", "quarter", "date", "dayofyear"}

PREAMBLE = r"""
typedef long long i64;
__device__ __forceinline__ i64 fdiv(i64 a, i64 b) {
  i64 q = a / b;
  return (a % b != 0 && ((a < 0) != (b < 0))) ? q - 1 : q;
}
__device__ __forceinline__ void civil(i64 z, int* y, int* m, int* d) {
  z += 719468;
  i64 era = (z >= 0 ? z : z - 146096) / 146097;
  unsigned doe = (unsigned)(z - era * 146097);
  unsigned yoe = (doe - doe / 1460 + doe / 36524 - doe / 146096) / 365;
  i64 y_ = (i64)yoe + era * 400;
  unsigned doy = doe - (365 * yoe + yoe / 4 - yoe / 100);
  unsigned mp = (5 * doy + 2) / 153;
  *d = (int)(doy - (153 * mp + 2) / 5 + 1);
  *m = (int)(mp < 10 ? mp + 3 : mp - 9);
  *y = (int)(y_ + (*m <= 2));
}
#define NSD 86400000000000LL
"""


class _Fuser:
    def __init__(self, tbl: Table):
        self.tbl = tbl
        self.inputs: Dict[str, Tuple[int, Column]] = {}  # name -> (idx, col)
        self.lines: List[str] = []
        self.tmp = 0
        # scalar constants passed as kernel ARGUMENTS (not source literals)
        # so one compiled kernel serves every constant value — without this,
        # data-dependent filter scalars (TPC-H q11/q15/q22 computed
        # thresholds, which differ by 1 ULP run-to-run from atomic-add
        # ordering) forced a ~93 ms hipRTC recompile inside the timed query.
        self.scalars: List[Tuple[str, object]] = []  # (ctype, ctypes value)

    def input_of(self, name: str) -> str:
        col = self.tbl.column(name)
        if col.mask is not None:
            raise _NotFusable
        if col.dtype.kind not in _CTYPE_OF_KIND or \
                col.dtype.kind == TypeKind.DICT:
            raise _NotFusable
        if name not in self.inputs:
            self.inputs[name] = (len(self.inputs), col)
        idx = self.inputs[name][0]
        return f"in{idx}[i]"

    def fresh(self) -> str:
        self.tmp += 1
        return f"t{self.tmp}"

    # returns (c_expr, ctype in {"double","i64","bool"})
    def emit(self, e: Expr) -> Tuple[str, str]:
        if isinstance(e, ColRef):
            col = self.tbl.column(e.name)
            ref = self.input_of(e.name)
            k = col.dtype.kind
            if k in (TypeKind.FLOAT32, TypeKind.FLOAT64):
                return f"((double){ref})", "double"
            if k == TypeKind.BOOL:
                return f"((bool){ref})", "bool"
            return f"((i64){ref})", "i64"
        if isinstance(e, Const):
            from ..ops.evaluate import infer_const_dtype, normalize_const

            dtype = e.dtype or infer_const_dtype(e.value)
            if dtype.kind in (TypeKind.STRING, TypeKind.DICT):
                raise _NotFusable
            v = normalize_const(e.value, dtype)
            if isinstance(v, bool):
                return ("true" if v else "false"), "bool"
            name = f"c{len(self.scalars)}"
            if isinstance(v, int):
                self.scalars.append(("long long", ctypes.c_longlong(v)))
                return name, "i64"
            if isinstance(v, float):
                self.scalars.append(("double", ctypes.c_double(float(v))))
                return name, "double"
            raise _NotFusable
        if isinstance(e, BinOp):
            l, lt = self.emit(e.left)
            r, rt = self.emit(e.right)
            if "bool" in (lt, rt):
                raise _NotFusable
            out_t = "double" if ("double" in (lt, rt) or e.op in ("div", "pow")) \
                else "i64"
            lc = f"(({out_t}){l})" if lt != out_t else l
            rc = f"(({out_t}){r})" if rt != out_t else r
            op = {"add": "+", "sub": "-", "mul": "*", "div": "/"}.get(e.op)
            if op:
                return f"({lc} {op} {rc})", out_t
            if e.op == "mod":
                if out_t == "i64":
                    return f"((({lc} % {rc}) + {rc}) % {rc})", "i64"
                return f"fmod(fmod({lc}, {rc}) + {rc}, {rc})", "double"
            if e.op == "floordiv":
                if out_t == "i64":
                    return f"fdiv({lc}, {rc})", "i64"
                return f"floor({lc} / {rc})", "double"
            if e.op == "pow":
                return f"pow({lc}, {rc})", "double"
            raise _NotFusable
        if isinstance(e, Cmp):
            l, lt = self.emit(e.left)
            r, rt = self.emit(e.right)
            t = "double" if "double" in (lt, rt) else \
                ("i64" if "i64" in (lt, rt) else "bool")
            lc = f"(({t}){l})" if lt != t else l
            rc = f"(({t}){r})" if rt != t else r
            op = {"lt": "<", "le": "<=", "gt": ">", "ge": ">=", "eq": "==",
                  "ne": "!="}[e.op]
            return f"({lc} {op} {rc})", "bool"
        if isinstance(e, BoolOp):
            l, lt = self.emit(e.left)
            r, rt = self.emit(e.right)
            if lt != "bool" or rt != "bool":
                raise _NotFusable
            op = "&&" if e.op == "and" else "||"
            return f"({l} {op} {r})", "bool"
        if isinstance(e, Not):
            l, lt = self.emit(e.operand)
            if lt != "bool":
                raise _NotFusable
            return f"(!{l})", "bool"
        if isinstance(e, IsIn):
            l, lt = self.emit(e.operand)
            if lt == "bool" or len(e.values) > 16:
                raise _NotFusable
            from ..ops.evaluate import normalize_const

            terms = []
            for v in e.values:
                if isinstance(v, str):
                    raise _NotFusable
                nv = normalize_const(
                    v, self._operand_dtype(e.operand))
                lit = f"{nv}LL" if lt == "i64" and isinstance(nv, int) \
                    else repr(float(nv))
                terms.append(f"({l} == {lit})")
            return "(" + " || ".join(terms) + ")", "bool"
        if isinstance(e, Case):
            branches = [self.emit(c) for c in e.conds]
            thens = [self.emit(t) for t in e.thens]
            other = self.emit(e.otherwise)
            if any(bt_ != "bool" for _, bt_ in branches):
                raise _NotFusable
            ts = {t for _, t in thens} | {other[1]}
            out_t = "double" if "double" in ts else \
                ("i64" if "i64" in ts else "bool")

            def cast(x, t):
                return f"(({out_t}){x})" if t != out_t else x

            expr = cast(other[0], other[1])
            for (c, _), (v, vt) in zip(reversed(branches), reversed(thens)):
                expr = f"({c} ? {cast(v, vt)} : {expr})"
            return expr, out_t
        if isinstance(e, DtField):
            if e.fld not in _DT_FIELDS_OK:
                raise _NotFusable
            inner = e.operand
            if not isinstance(inner, ColRef):
                raise _NotFusable
            col = self.tbl.column(inner.name)
            ref = self.input_of(inner.name)
            if col.dtype.kind == TypeKind.DATE32:
                days = f"((i64){ref})"
                nsday = "0LL"
            elif col.dtype.kind == TypeKind.TIMESTAMP_NS:
                days = f"fdiv((i64){ref}, NSD)"
                nsday = f"((i64){ref} - {days} * NSD)"
            else:
                raise _NotFusable
            v = self.fresh()
            if e.fld == "date":
                self.lines.append(f"i64 {v} = {days};")
                return v, "i64"
            if e.fld == "hour":
                self.lines.append(f"i64 {v} = {nsday} / 3600000000000LL;")
                return v, "i64"
            if e.fld == "minute":
                self.lines.append(f"i64 {v} = ({nsday} / 60000000000LL) % 60;")
                return v, "i64"
            if e.fld == "second":
                self.lines.append(f"i64 {v} = ({nsday} / 1000000000LL) % 60;")
                return v, "i64"
            if e.fld in ("dayofweek", "weekday"):
                self.lines.append(
                    f"i64 {v} = (({days}) + 3) % 7; if ({v} < 0) {v} += 7;")
                return v, "i64"
            y, m, d = self.fresh(), self.fresh(), self.fresh()
            self.lines.append(f"int {y}, {m}, {d}; civil({days}, &{y}, &{m}, &{d});")
            if e.fld == "year":
                return f"((i64){y})", "i64"
            if e.fld == "month":
                return f"((i64){m})", "i64"
            if e.fld == "day":
                return f"((i64){d})", "i64"
            if e.fld == "quarter":
                return f"((i64)(({m} - 1) / 3 + 1))", "i64"
            raise _NotFusable
        if isinstance(e, RoundExpr):
            l, lt = self.emit(e.operand)
            if lt != "double":
                return l, lt
            scale = 10.0 ** e.decimals
            return f"(round({l} * {scale}) / {scale})", "double"
        raise _NotFusable

    def _operand_dtype(self, e: Expr):
        if isinstance(e, ColRef):
            return self.tbl.column(e.name).dtype
        return bt.float64


class _NotFusable(Exception):
    pass


_OUT_DTYPE = {"double": bt.float64, "i64": bt.int64, "bool": bt.boolean}
_OUT_CTYPE = {"double": "double", "i64": "long long", "bool": "unsigned char"}
_OUT_TORCH = {"double": torch.float64, "i64": torch.int64, "bool": torch.bool}

_KERNEL_CACHE: dict = {}


def _expr_signature(e: Expr, tbl: Table) -> str:
    if isinstance(e, ColRef):
        return f"col({e.name}:{tbl.column(e.name).dtype.kind})"
    if isinstance(e, Const):
        # keyed by the emitted TYPE only: the value is a kernel argument,
        # so one kernel serves all constant values of this structure
        from ..ops.evaluate import infer_const_dtype, normalize_const

        try:
            v = normalize_const(e.value, e.dtype or infer_const_dtype(e.value))
        except Exception:
            return f"const({e.value!r})"
        if isinstance(v, bool):
            return f"const(bool:{v})"
        return "const(i64)" if isinstance(v, int) else \
            ("const(f64)" if isinstance(v, float) else f"const({v!r})")
    parts = [type(e).__name__]
    for f in getattr(e, "__dataclass_fields__", {}):
        v = getattr(e, f)
        if isinstance(v, Expr):
            parts.append(_expr_signature(v, tbl))
        elif isinstance(v, tuple):
            parts.append(",".join(
                _expr_signature(x, tbl) if isinstance(x, Expr) else repr(x)
                for x in v))
        elif not callable(v):
            parts.append(repr(v))
    return f"{parts[0]}({';'.join(parts[1:])})"


def try_fuse_exprs(tbl: Table, exprs: List[Expr]) -> Optional[List[Column]]:
    """Compile+run all non-trivial exprs in ONE kernel; returns the output
    Columns in expr order (pass-through ColRefs reuse existing columns), or
    None when any expr is outside the fusable subset."""
    if not tbl.device.type == "cuda" or len(tbl) == 0:
        return None
    compute = [(i, e) for i, e in enumerate(exprs) if not isinstance(e, ColRef)]
    if not compute:
        return None
    fuser = _Fuser(tbl)
    try:
        emitted = [(i, fuser.emit(e)) for i, e in compute]
    except (_NotFusable, KeyError):
        return None
    n = len(tbl)
    sig = "|".join(_expr_signature(e, tbl) for _, e in compute)
    entry = _KERNEL_CACHE.get(sig)
    in_items = sorted(fuser.inputs.values(), key=lambda t: t[0])
    if entry is None:
        params = []
        for idx, col in in_items:
            ct = _CTYPE_OF_KIND[col.dtype.kind]
            params.append(f"const {ct}* __restrict__ in{idx}")
        for j, (ct, _) in enumerate(fuser.scalars):
            params.append(f"{ct} c{j}")
        for j, (_, (cexpr, t)) in enumerate(emitted):
            params.append(f"{_OUT_CTYPE[t]}* __restrict__ out{j}")
        body_lines = "\n    ".join(fuser.lines)
        outs = "\n    ".join(
            f"out{j}[i] = ({_OUT_CTYPE[t]})({cexpr});"
            for j, (_, (cexpr, t)) in enumerate(emitted))
        src = PREAMBLE + f"""
extern "C" __global__ void fused(long long n, {', '.join(params)}) {{
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += stride) {{
    {body_lines}
    {outs}
  }}
}}
"""
        from .hip_udf import _runtime

        rt = _runtime()
        import os as _os
        if _os.environ.get("BODO_AMD_FUSE_LOG"):
            import sys as _sys
            import time as _time
            _t0 = _time.perf_counter()
            fn = rt.get_kernel_named(src, b"fused")
            print(f"[fuse-compile] {1e3 * (_time.perf_counter() - _t0):.0f}ms "
                  f"sig={sig[:160]}", file=_sys.stderr, flush=True)
        else:
            fn = rt.get_kernel_named(src, b"fused")
        entry = (fn, [t for _, (_, t) in emitted])
        if len(_KERNEL_CACHE) > 512:
            _KERNEL_CACHE.clear()
        _KERNEL_CACHE[sig] = entry
    fn, out_types = entry
    outs = [torch.empty(n, dtype=_OUT_TORCH[t], device=tbl.device)
            for t in out_types]
    ptrs = [c.data.data_ptr() if c.data.dtype != torch.bool
            else c.data.view(torch.uint8).data_ptr() for _, c in in_items]
    out_ptrs = [(o.view(torch.uint8) if o.dtype == torch.bool else o).data_ptr()
                for o in outs]
    from .hip_udf import _runtime

    _runtime().launch_generic(fn, n,
                              ptrs + [v for _, v in fuser.scalars] + out_ptrs,
                              torch.cuda.current_stream().cuda_stream)
    result: List[Optional[Column]] = [None] * len(exprs)
    for (i, e), o, t in zip(compute, outs, out_types):
        col = Column(_OUT_DTYPE[t], o)
        if isinstance(e, DtField):
            from ..ops.gpu import _DT_RANGE

            col.val_range = _DT_RANGE.get(e.fld)
        result[i] = col
    for i, e in enumerate(exprs):
        if isinstance(e, ColRef):
            result[i] = tbl.column(e.name)
    return result
