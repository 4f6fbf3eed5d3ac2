"""Logical column types for the bodo_amd columnar engine.

Mirrors the role of the reference's ``Bodo_CTypes`` /``bodo_array_type``
(bodo/libs/_bodo_common.h:340,525) but designed for a torch-tensor-backed
HBM-resident layout on MI355X: fixed-width columns are one 1-D torch tensor
plus an optional byte validity mask; strings are Arrow-style offsets+bytes;
dictionary-encoded strings are int32 indices + a (replicated) dictionary.
"""

from __future__ import annotations

import enum
from dataclasses import dataclass

import numpy as np
import torch


class TypeKind(enum.IntEnum):
    # Numeric ids are part of the kernel ABI (csrc/common.h ColumnDesc.dtype);
    # keep in sync.
    INT8 = 0
    INT16 = 1
    INT32 = 2
    INT64 = 3
    FLOAT32 = 4
    FLOAT64 = 5
    BOOL = 6
    DATE32 = 7  # days since epoch, int32 storage
    TIMESTAMP_NS = 8  # ns since epoch, int64 storage
    STRING = 9  # arrow offsets(int64, n+1) + uint8 bytes
    DICT = 10  # int32 indices into a string dictionary
    DECIMAL128 = 11  # two int64 tensors (lo, hi) - limited support
    UINT8 = 12
    UINT16 = 13
    UINT32 = 14
    UINT64 = 15
    LIST = 16  # arrow list<child>: int64 offsets (n+1) + child column
    STRUCT = 17  # arrow struct<fields>: named child columns + validity
    DURATION_NS = 18  # timedelta64[ns], int64 storage


@dataclass(frozen=True)
class DType:
    kind: TypeKind
    # for DECIMAL128
    precision: int = 0
    scale: int = 0
    # for STRUCT: field names (children carry their own dtypes)
    fields: tuple = ()

    @property
    def is_numeric(self) -> bool:
        return self.kind in _NUMERIC_KINDS

    @property
    def is_integer(self) -> bool:
        return self.kind in (
            TypeKind.INT8, TypeKind.INT16, TypeKind.INT32, TypeKind.INT64,
            TypeKind.UINT8, TypeKind.UINT16, TypeKind.UINT32, TypeKind.UINT64,
        )

    @property
    def is_float(self) -> bool:
        return self.kind in (TypeKind.FLOAT32, TypeKind.FLOAT64)

    @property
    def is_temporal(self) -> bool:
        return self.kind in (TypeKind.DATE32, TypeKind.TIMESTAMP_NS)

    @property
    def is_string_like(self) -> bool:
        return self.kind in (TypeKind.STRING, TypeKind.DICT)

    def __repr__(self) -> str:  # pragma: no cover - debug aid
        if self.kind == TypeKind.DECIMAL128:
            return f"decimal128({self.precision},{self.scale})"
        return self.kind.name.lower()


_NUMERIC_KINDS = frozenset(
    {
        TypeKind.INT8, TypeKind.INT16, TypeKind.INT32, TypeKind.INT64,
        TypeKind.UINT8, TypeKind.UINT16, TypeKind.UINT32, TypeKind.UINT64,
        TypeKind.FLOAT32, TypeKind.FLOAT64,
    }
)

int8 = DType(TypeKind.INT8)
int16 = DType(TypeKind.INT16)
int32 = DType(TypeKind.INT32)
int64 = DType(TypeKind.INT64)
uint8 = DType(TypeKind.UINT8)
uint16 = DType(TypeKind.UINT16)
uint32 = DType(TypeKind.UINT32)
uint64 = DType(TypeKind.UINT64)
float32 = DType(TypeKind.FLOAT32)
float64 = DType(TypeKind.FLOAT64)
boolean = DType(TypeKind.BOOL)
date32 = DType(TypeKind.DATE32)
timestamp_ns = DType(TypeKind.TIMESTAMP_NS)
duration_ns = DType(TypeKind.DURATION_NS)
string = DType(TypeKind.STRING)
# binary shares the STRING layout (offsets+bytes); precision=1 flags it so
# arrow round-trips as large_binary (reference: binary_arr_ext.py)
binary = DType(TypeKind.STRING, precision=1)
dictionary = DType(TypeKind.DICT)
list_ = DType(TypeKind.LIST)


def decimal128(precision: int, scale: int) -> DType:
    return DType(TypeKind.DECIMAL128, precision, scale)


def struct_(field_names) -> DType:
    return DType(TypeKind.STRUCT, fields=tuple(field_names))


# torch storage dtype for the primary data tensor of each kind
_TORCH_STORAGE = {
    TypeKind.INT8: torch.int8,
    TypeKind.INT16: torch.int16,
    TypeKind.INT32: torch.int32,
    TypeKind.INT64: torch.int64,
    TypeKind.UINT8: torch.uint8,
    TypeKind.UINT16: torch.int16,  # stored as bit-identical signed
    TypeKind.UINT32: torch.int32,
    TypeKind.UINT64: torch.int64,
    TypeKind.FLOAT32: torch.float32,
    TypeKind.FLOAT64: torch.float64,
    TypeKind.BOOL: torch.bool,
    TypeKind.DATE32: torch.int32,
    TypeKind.TIMESTAMP_NS: torch.int64,
    TypeKind.DURATION_NS: torch.int64,
    TypeKind.DICT: torch.int32,
    TypeKind.DECIMAL128: torch.int64,  # scaled int64 (exact for p <= 18)
}


def torch_storage_dtype(dtype: DType) -> torch.dtype:
    return _TORCH_STORAGE[dtype.kind]


_NUMPY_TO_KIND = {
    np.dtype("int8"): TypeKind.INT8,
    np.dtype("int16"): TypeKind.INT16,
    np.dtype("int32"): TypeKind.INT32,
    np.dtype("int64"): TypeKind.INT64,
    np.dtype("uint8"): TypeKind.UINT8,
    np.dtype("uint16"): TypeKind.UINT16,
    np.dtype("uint32"): TypeKind.UINT32,
    np.dtype("uint64"): TypeKind.UINT64,
    np.dtype("float32"): TypeKind.FLOAT32,
    np.dtype("float64"): TypeKind.FLOAT64,
    np.dtype("bool"): TypeKind.BOOL,
}


def from_numpy_dtype(nd: np.dtype) -> DType:
    if nd.kind == "M":
        # datetime64 - normalize to ns
        return timestamp_ns
    k = _NUMPY_TO_KIND.get(nd)
    if k is None:
        raise TypeError(f"unsupported numpy dtype {nd}")
    return DType(k)


_KIND_TO_NUMPY = {
    TypeKind.INT8: np.dtype("int8"),
    TypeKind.INT16: np.dtype("int16"),
    TypeKind.INT32: np.dtype("int32"),
    TypeKind.INT64: np.dtype("int64"),
    TypeKind.UINT8: np.dtype("uint8"),
    TypeKind.UINT16: np.dtype("uint16"),
    TypeKind.UINT32: np.dtype("uint32"),
    TypeKind.UINT64: np.dtype("uint64"),
    TypeKind.FLOAT32: np.dtype("float32"),
    TypeKind.FLOAT64: np.dtype("float64"),
    TypeKind.BOOL: np.dtype("bool"),
    TypeKind.DATE32: np.dtype("int32"),
    TypeKind.TIMESTAMP_NS: np.dtype("int64"),
    TypeKind.DURATION_NS: np.dtype("int64"),
    TypeKind.DICT: np.dtype("int32"),
    TypeKind.DECIMAL128: np.dtype("int64"),
}


def numpy_storage_dtype(dtype: DType) -> np.dtype:
    return _KIND_TO_NUMPY[dtype.kind]


def common_numeric_type(a: DType, b: DType) -> DType:
    """Type promotion for arithmetic, numpy-style."""
    if a == b:
        return a
    na, nb = numpy_storage_dtype(a), numpy_storage_dtype(b)
    res = np.result_type(na, nb)
    return from_numpy_dtype(res)
