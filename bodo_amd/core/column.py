"""Columnar array: the device-resident analog of the reference's
``array_info`` (bodo/libs/_bodo_common.h:936).

A Column owns torch tensors (CPU or HBM on MI355X) plus an optional byte
validity mask (True = valid).  String columns use the Arrow layout
(int64 offsets of length n+1 + uint8 byte buffer); dictionary-encoded
strings hold int32 indices plus a host-side pyarrow dictionary that is
replicated across ranks (reference: bodo/libs/_dict_builder.cpp).
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import pyarrow as pa
import torch

from . import types as bt
from .types import DType, TypeKind


def _np_from_tensor(t: torch.Tensor) -> np.ndarray:
    return t.numpy() if t.device.type == "cpu" else t.cpu().numpy()


class Column:
    __slots__ = ("dtype", "data", "mask", "offsets", "dictionary", "name",
                 "_length", "val_range", "child", "children")

    def __init__(
        self,
        dtype: DType,
        data: Optional[torch.Tensor] = None,
        mask: Optional[torch.Tensor] = None,
        offsets: Optional[torch.Tensor] = None,
        dictionary: Optional[pa.Array] = None,
        length: Optional[int] = None,
    ):
        self.dtype = dtype
        self.data = data
        self.mask = mask  # torch.bool, True = valid, or None (all valid)
        self.offsets = offsets  # STRING only: int64 (n+1)
        self.dictionary = dictionary  # DICT only: pa.StringArray (host)
        self.child = None  # LIST only: element Column
        self.children = None  # STRUCT only: field Columns (dtype.fields names)
        self.val_range = None  # optional known (lo, hi) for int columns
        if length is not None:
            self._length = length
        elif dtype.kind in (TypeKind.STRING, TypeKind.LIST):
            self._length = int(offsets.shape[0]) - 1
        else:
            self._length = int(data.shape[0])

    def __len__(self) -> int:
        return self._length

    @property
    def device(self) -> torch.device:
        t = self.data if self.data is not None else self.offsets
        if t is None:
            if self.children:  # STRUCT: no buffers of its own
                return self.children[0].device
            return self.mask.device if self.mask is not None \
                else torch.device("cpu")
        return t.device

    @property
    def is_cuda(self) -> bool:
        return self.device.type == "cuda"

    def nbytes(self) -> int:
        n = 0
        for t in (self.data, self.mask, self.offsets):
            if t is not None:
                n += t.numel() * t.element_size()
        if self.child is not None:
            n += self.child.nbytes()
        if self.children is not None:
            n += sum(c.nbytes() for c in self.children)
        return n

    # ------------------------------------------------------------------
    # movement
    # ------------------------------------------------------------------
    def to_device(self, device) -> "Column":
        device = torch.device(device)
        cur = self.device
        if cur == device or (cur.type == device.type and device.index is None):
            return self

        # non_blocking only for H2D: async D2H into pageable host memory can
        # be read by numpy before the copy completes (observed on ROCm)
        nb = device.type == "cuda"

        def mv(t):
            return None if t is None else t.to(device, non_blocking=nb)

        out = Column(
            self.dtype, mv(self.data), mv(self.mask), mv(self.offsets),
            self.dictionary, self._length,
        )
        out.val_range = self.val_range
        if self.child is not None:
            out.child = self.child.to_device(device)
        if self.children is not None:
            out.children = [c.to_device(device) for c in self.children]
        return out

    # ------------------------------------------------------------------
    # arrow interop
    # ------------------------------------------------------------------
    @staticmethod
    def from_arrow(arr: pa.Array, device="cpu") -> "Column":
        if isinstance(arr, pa.ChunkedArray):
            arr = arr.combine_chunks()
        t = arr.type
        if pa.types.is_dictionary(t):
            idx = arr.indices.cast(pa.int32())
            col = Column.from_arrow(idx, device)
            return Column(
                bt.dictionary, col.data, col.mask,
                dictionary=arr.dictionary.cast(pa.large_string())
                if not pa.types.is_large_string(arr.dictionary.type)
                else arr.dictionary,
                length=len(arr),
            )
        if pa.types.is_binary(t) or pa.types.is_large_binary(t):
            arr = arr.cast(pa.large_binary())
            bufs = arr.buffers()
            validity, off_buf, data_buf = bufs[0], bufs[1], bufs[2]
            offsets = np.frombuffer(off_buf, dtype=np.int64,
                                    count=len(arr) + 1 + arr.offset)[arr.offset:]
            if arr.offset != 0 or offsets[0] != 0:
                offsets = offsets - offsets[0]
            n_bytes = int(offsets[-1])
            base = np.frombuffer(off_buf, dtype=np.int64,
                                 count=1 + arr.offset + len(arr))[arr.offset]
            data = np.frombuffer(data_buf, dtype=np.uint8, count=n_bytes,
                                 offset=int(base)) \
                if data_buf is not None and n_bytes else np.zeros(0, np.uint8)
            mask = None
            if validity is not None and arr.null_count:
                mask = _unpack_validity(validity, arr.offset, len(arr))
            return Column(
                bt.binary,
                torch.from_numpy(np.ascontiguousarray(data)).to(device),
                None if mask is None else torch.from_numpy(mask).to(device),
                offsets=torch.from_numpy(
                    np.ascontiguousarray(offsets)).to(device),
                length=len(arr))
        if pa.types.is_string(t) or pa.types.is_large_string(t):
            arr = arr.cast(pa.large_string())
            bufs = arr.buffers()
            validity, off_buf, data_buf = bufs[0], bufs[1], bufs[2]
            offsets = np.frombuffer(off_buf, dtype=np.int64,
                                    count=len(arr) + 1 + arr.offset)[arr.offset:]
            if arr.offset != 0 or offsets[0] != 0:
                offsets = offsets - offsets[0]
            n_bytes = int(offsets[-1])
            base = np.frombuffer(off_buf, dtype=np.int64, count=1 + arr.offset + len(arr))[arr.offset]
            data = np.frombuffer(data_buf, dtype=np.uint8,
                                 count=n_bytes, offset=int(base)) if data_buf is not None and n_bytes else np.zeros(0, np.uint8)
            mask = None
            if validity is not None and arr.null_count:
                mask = _unpack_validity(validity, arr.offset, len(arr))
            return Column(
                bt.string,
                torch.from_numpy(np.ascontiguousarray(data)).to(device),
                None if mask is None else torch.from_numpy(mask).to(device),
                offsets=torch.from_numpy(np.ascontiguousarray(offsets)).to(device),
                length=len(arr),
            )
        if pa.types.is_list(t) or pa.types.is_large_list(t):
            # arrow list<child>: int64 offsets + recursively converted child
            # (reference: array_item_arr_ext.py ArrayItemArray layout)
            arr = arr.cast(pa.large_list(arr.type.value_type)) \
                if pa.types.is_list(t) else arr
            offs = np.frombuffer(arr.buffers()[1], dtype=np.int64,
                                 count=len(arr) + 1 + arr.offset)[arr.offset:]
            base = int(offs[0])
            offs = offs - base
            child = Column.from_arrow(
                arr.values.slice(base, int(offs[-1])), device)
            mask = None
            if arr.buffers()[0] is not None and arr.null_count:
                mask = _unpack_validity(arr.buffers()[0], arr.offset,
                                        len(arr))
            out = Column(
                bt.list_, None,
                None if mask is None else torch.from_numpy(mask).to(device),
                offsets=torch.from_numpy(
                    np.ascontiguousarray(offs)).to(device),
                length=len(arr))
            out.child = child
            return out
        if pa.types.is_map(t):
            # arrow map<k,v> is physically list<struct<key,value>>: reuse
            # both layouts (reference: map_arr_ext.py over array_item+struct)
            st = pa.StructArray.from_arrays(
                [arr.keys, arr.items], ["key", "value"])
            mask = None
            if arr.null_count:
                mask = pa.array([not v for v in arr.is_valid().to_pylist()])
            lst = pa.LargeListArray.from_arrays(
                arr.offsets.cast(pa.int64()), st, mask=mask)
            return Column.from_arrow(lst, device)
        if pa.types.is_struct(t):
            # arrow struct<fields>: recursively converted field columns +
            # struct-level validity (reference: struct_arr_ext.py layout)
            arr = arr.combine_chunks() if isinstance(arr, pa.ChunkedArray) else arr
            flat = arr.flatten()  # offset/validity-adjusted field arrays
            names = [t.field(i).name for i in range(t.num_fields)]
            children = [Column.from_arrow(f, device) for f in flat]
            mask = None
            if arr.buffers()[0] is not None and arr.null_count:
                mask = _unpack_validity(arr.buffers()[0], arr.offset,
                                        len(arr))
            out = Column(
                bt.struct_(names), None,
                None if mask is None else torch.from_numpy(mask).to(device),
                length=len(arr))
            out.children = children
            return out
        if pa.types.is_null(t):
            # typeless all-null column: represent as float64 NaN
            data = torch.full((len(arr),), float("nan"), dtype=torch.float64)
            return Column(bt.float64, data.to(device), length=len(arr))
        if pa.types.is_duration(t):
            arr = arr.cast(pa.duration("ns"))
            dtype = bt.duration_ns
        elif pa.types.is_timestamp(t):
            arr = arr.cast(pa.timestamp("ns"))
            dtype = bt.timestamp_ns
        elif pa.types.is_date32(t):
            dtype = bt.date32
        elif pa.types.is_date64(t):
            arr = arr.cast(pa.date32())
            dtype = bt.date32
        elif pa.types.is_boolean(t):
            dtype = bt.boolean
        elif pa.types.is_decimal(t):
            if t.precision <= 18:
                # exact money semantics: store the unscaled integer in an
                # int64 tensor (values with p<=18 fit the low limb; the high
                # limb is pure sign extension).  Reference role:
                # bodo/libs/_decimal_ext.cpp int128 decimals — redesigned as
                # scaled int64, which is native on CDNA4; p>18 falls back.
                bufs = arr.buffers()
                validity, data_buf = bufs[0], bufs[1]
                pairs = np.frombuffer(data_buf, dtype=np.int64,
                                      count=2 * (len(arr) + arr.offset))
                lo = pairs[2 * arr.offset::2][:len(arr)]
                mask_np = None
                if validity is not None and arr.null_count:
                    mask_np = _unpack_validity(validity, arr.offset, len(arr))
                    lo = lo.copy()
                    lo[~mask_np] = 0
                return Column(
                    bt.decimal128(t.precision, t.scale),
                    torch.from_numpy(np.ascontiguousarray(lo)).to(device),
                    None if mask_np is None
                    else torch.from_numpy(mask_np).to(device),
                    length=len(arr))
            arr = arr.cast(pa.float64())
            dtype = bt.float64
        else:
            dtype = bt.from_numpy_dtype(np.dtype(t.to_pandas_dtype()))
        np_store = bt.numpy_storage_dtype(dtype)
        if pa.types.is_boolean(arr.type):
            vals = arr.to_numpy(zero_copy_only=False)
            if vals.dtype == object:  # has nulls
                valid = np.array([v is not None for v in vals], dtype=bool)
                out = np.zeros(len(arr), dtype=bool)
                out[valid] = vals[valid].astype(bool)
                data_np, mask_np = out, valid
            else:
                data_np, mask_np = vals.astype(bool), None
        else:
            bufs = arr.buffers()
            validity, data_buf = bufs[0], bufs[1]
            itemsize = np_store.itemsize
            data_np = np.frombuffer(
                data_buf, dtype=np_store, count=len(arr), offset=arr.offset * itemsize
            )
            mask_np = None
            if validity is not None and arr.null_count:
                mask_np = _unpack_validity(validity, arr.offset, len(arr))
                # zero-fill nulls so downstream kernels see deterministic data
                data_np = data_np.copy()
                data_np[~mask_np] = 0
        return Column(
            dtype,
            torch.from_numpy(np.ascontiguousarray(data_np)).to(device),
            None if mask_np is None else torch.from_numpy(mask_np).to(device),
            length=len(arr),
        )

    def to_arrow(self) -> pa.Array:
        k = self.dtype.kind
        if k == TypeKind.LIST:
            offsets = _np_from_tensor(self.offsets)
            child = self.child.to_arrow()
            mask = None if self.mask is None else _np_from_tensor(self.mask)
            return pa.LargeListArray.from_arrays(
                pa.array(offsets, type=pa.int64()), child,
                mask=None if mask is None else pa.array(~mask))
        if k == TypeKind.STRUCT:
            fields = [c.to_arrow() for c in self.children]
            mask = None if self.mask is None else ~_np_from_tensor(self.mask)
            return pa.StructArray.from_arrays(
                fields, list(self.dtype.fields),
                mask=None if mask is None else pa.array(mask))
        if k == TypeKind.STRING:
            offsets = _np_from_tensor(self.offsets)
            data = _np_from_tensor(self.data) if self.data is not None else np.zeros(0, np.uint8)
            mask = None if self.mask is None else ~_np_from_tensor(self.mask)
            if self.dtype.precision == 1:  # binary flag
                vbuf = pa.py_buffer(np.packbits(
                    ~mask, bitorder="little").tobytes()) \
                    if mask is not None else None
                return pa.Array.from_buffers(
                    pa.large_binary(), len(self),
                    [vbuf, pa.py_buffer(offsets.tobytes()),
                     pa.py_buffer(data.tobytes())],
                    -1 if mask is not None else 0)
            return pa.LargeStringArray.from_buffers(
                len(self), pa.py_buffer(offsets.tobytes()), pa.py_buffer(data.tobytes()),
                pa.py_buffer(np.packbits(~mask, bitorder="little").tobytes()) if mask is not None else None,
                -1 if mask is not None else 0,
            )
        if k == TypeKind.DICT:
            idx = _np_from_tensor(self.data)
            mask = None if self.mask is None else _np_from_tensor(self.mask)
            idx_arr = pa.array(idx, type=pa.int32(),
                               mask=None if mask is None else ~mask)
            return pa.DictionaryArray.from_arrays(idx_arr, self.dictionary)
        if k == TypeKind.DECIMAL128:
            lo = _np_from_tensor(self.data).astype(np.int64, copy=False)
            n = len(self)
            pairs = np.empty(2 * n, dtype=np.int64)
            pairs[0::2] = lo
            pairs[1::2] = np.where(lo < 0, -1, 0)  # sign-extended high limb
            mask = None if self.mask is None else _np_from_tensor(self.mask)
            vbuf = None
            null_count = 0
            if mask is not None:
                vbuf = pa.py_buffer(
                    np.packbits(mask, bitorder="little").tobytes())
                null_count = int((~mask).sum())
            return pa.Array.from_buffers(
                pa.decimal128(self.dtype.precision, self.dtype.scale), n,
                [vbuf, pa.py_buffer(pairs.tobytes())], null_count)
        np_data = _np_from_tensor(self.data)
        mask = None if self.mask is None else _np_from_tensor(self.mask)
        pa_type = {
            TypeKind.DATE32: pa.date32(),
            TypeKind.TIMESTAMP_NS: pa.timestamp("ns"),
            TypeKind.DURATION_NS: pa.duration("ns"),
            TypeKind.UINT8: pa.uint8(),
            TypeKind.UINT16: pa.uint16(),
            TypeKind.UINT32: pa.uint32(),
            TypeKind.UINT64: pa.uint64(),
        }.get(k)
        if k in (TypeKind.UINT16, TypeKind.UINT32, TypeKind.UINT64):
            np_data = np_data.view(bt.numpy_storage_dtype(self.dtype))
        if pa_type is None:
            return pa.array(np_data, mask=None if mask is None else ~mask)
        return pa.array(np_data, type=pa_type, mask=None if mask is None else ~mask)

    @staticmethod
    def from_numpy(arr: np.ndarray, device="cpu", mask: Optional[np.ndarray] = None) -> "Column":
        if arr.dtype.kind == "M":
            arr = arr.astype("datetime64[ns]").view("int64")
            dtype = bt.timestamp_ns
        elif arr.dtype.kind == "m":
            arr = arr.astype("timedelta64[ns]").view("int64")
            dtype = bt.duration_ns
        elif arr.dtype == object or arr.dtype.kind == "U":
            # from_pandas treats float NaN placeholders in object arrays as
            # nulls (pa.array would infer double from a leading NaN)
            return Column.from_arrow(pa.Array.from_pandas(arr), device)
        else:
            dtype = bt.from_numpy_dtype(arr.dtype)
            if arr.dtype.kind == "u" and arr.dtype.itemsize > 1:
                arr = arr.view(f"int{arr.dtype.itemsize * 8}")
            if arr.dtype.kind == "f" and mask is None:
                pass  # NaN stands for null in float columns
        t = torch.from_numpy(np.ascontiguousarray(arr)).to(device)
        m = None if mask is None else torch.from_numpy(np.ascontiguousarray(mask)).to(device)
        return Column(dtype, t, m)

    @staticmethod
    def full_const(value, dtype: DType, length: int, device="cpu") -> "Column":
        if dtype.kind == TypeKind.STRING:
            s = str(value).encode()
            offs = torch.arange(0, (length + 1) * len(s), len(s), dtype=torch.int64, device=device) \
                if len(s) else torch.zeros(length + 1, dtype=torch.int64, device=device)
            data = torch.frombuffer(bytearray(s * length), dtype=torch.uint8).to(device) \
                if length * len(s) else torch.zeros(0, dtype=torch.uint8, device=device)
            return Column(bt.string, data, None, offsets=offs, length=length)
        t = torch.full((length,), value, dtype=bt.torch_storage_dtype(dtype), device=device)
        return Column(dtype, t)

    # convenience for tests
    def to_pandas(self):
        import pandas as pd

        arr = self.to_arrow()
        return pd.Series(arr.to_pandas())


def _unpack_validity(buf: pa.Buffer, offset: int, length: int) -> np.ndarray:
    bits = np.frombuffer(buf, dtype=np.uint8)
    out = np.unpackbits(bits, bitorder="little")[offset:offset + length]
    return out.astype(bool)
