"""Table: ordered collection of named Columns (reference: ``table_info``,
bodo/libs/_bodo_common.h)."""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import pandas as pd
import pyarrow as pa
import torch

from .column import Column


def _normalize_device(device) -> torch.device:
    """Resolve a bare ``cuda`` device to an explicit index so equality checks
    compare (type, index), not just type.  Under the 1-process-per-GPU model
    the current device is the only visible one."""
    dev = torch.device(device)
    if dev.type == "cuda" and dev.index is None:
        idx = torch.cuda.current_device() if torch.cuda.is_available() else 0
        dev = torch.device("cuda", idx)
    return dev


class Table:
    __slots__ = ("names", "columns", "_length", "_dev_cache")

    def __init__(self, names: Sequence[str], columns: Sequence[Column],
                 length: Optional[int] = None):
        assert len(names) == len(columns)
        self.names: List[str] = list(names)
        self.columns: List[Column] = list(columns)
        if length is None:
            length = len(columns[0]) if columns else 0
        self._length = length
        # device copies cached per target: repeated query runs over the same
        # registered source must not re-upload the table each execution
        self._dev_cache: Dict[str, "Table"] = {}

    def __len__(self) -> int:
        return self._length

    @property
    def num_columns(self) -> int:
        return len(self.columns)

    def column(self, name: str) -> Column:
        return self.columns[self.names.index(name)]

    def has_column(self, name: str) -> bool:
        return name in self.names

    def as_dict(self) -> Dict[str, Column]:
        return dict(zip(self.names, self.columns))

    def select(self, names: Sequence[str]) -> "Table":
        return Table(list(names), [self.column(n) for n in names], self._length)

    def with_column(self, name: str, col: Column) -> "Table":
        names, cols = list(self.names), list(self.columns)
        if name in names:
            cols[names.index(name)] = col
        else:
            names.append(name)
            cols.append(col)
        return Table(names, cols, self._length)

    def rename(self, mapping: Dict[str, str]) -> "Table":
        return Table([mapping.get(n, n) for n in self.names], self.columns, self._length)

    @property
    def device(self) -> torch.device:
        if not self.columns:
            return torch.device("cpu")
        return self.columns[0].device

    def to_device(self, device) -> "Table":
        dev = _normalize_device(device)
        if dev == _normalize_device(self.device):
            return self
        cache = self._dev_cache
        key = str(dev)
        hit = cache.get(key)
        if hit is not None:
            return hit
        out = Table(self.names,
                    [c.to_device(dev) for c in self.columns], self._length)
        cache[key] = out
        return out

    def nbytes(self) -> int:
        return sum(c.nbytes() for c in self.columns)

    # ------------------------------------------------------------------
    @staticmethod
    def from_arrow(tbl: pa.Table, device="cpu") -> "Table":
        cols = []
        for i in range(tbl.num_columns):
            c = tbl.column(i)
            # avoid 32-bit offset overflow combining >2GB string chunks
            if pa.types.is_string(c.type):
                c = c.cast(pa.large_string())
            cols.append(Column.from_arrow(c.combine_chunks(), device))
        return Table(tbl.column_names, cols, tbl.num_rows)

    def to_arrow(self) -> pa.Table:
        return pa.table(
            {n: c.to_arrow() for n, c in zip(self.names, self.columns)}
        ) if self.columns else pa.table({})

    @staticmethod
    def from_pandas(df: pd.DataFrame, device="cpu") -> "Table":
        df = _unwrap_lazy_scalars(df)
        tbl = pa.Table.from_pandas(df, preserve_index=False)
        tbl = dict_encode_strings(tbl)
        return Table.from_arrow(tbl, device)

    def to_pandas(self) -> pd.DataFrame:
        if not self.columns:
            return pd.DataFrame(index=range(self._length))
        out = self.to_arrow().to_pandas()
        # arrow gives large_string -> object; fine for pandas parity
        return out

    @staticmethod
    def empty_like(other: "Table") -> "Table":
        import numpy as np

        from . import types as bt
        from .types import TypeKind

        cols = []
        dev = other.device
        for c in other.columns:
            if c.dtype.kind == TypeKind.STRING:
                cols.append(Column(
                    bt.string,
                    torch.zeros(0, dtype=torch.uint8, device=dev), None,
                    offsets=torch.zeros(1, dtype=torch.int64, device=dev), length=0))
            else:
                cols.append(Column(
                    c.dtype,
                    torch.zeros(0, dtype=bt.torch_storage_dtype(c.dtype), device=dev),
                    None, dictionary=c.dictionary, length=0))
        return Table(list(other.names), cols, 0)

    def __repr__(self) -> str:  # pragma: no cover
        return f"Table({len(self)} rows x {self.num_columns} cols, device={self.device})"


def _unwrap_lazy_scalars(df: pd.DataFrame) -> pd.DataFrame:
    """Materialize BodoScalar values embedded in object columns (user code
    like pd.DataFrame({"r": [s.sum()]}) — drop-in contract)."""
    from ..pandas.scalar import BodoScalar

    out = None
    for c in df.columns:
        col = df[c]
        if col.dtype != object:
            continue
        probe = col if len(col) <= 1024 else col.head(64)
        if any(isinstance(v, BodoScalar) for v in probe):
            if out is None:
                out = df.copy()
            out[c] = pd.Series(
                [v.value if isinstance(v, BodoScalar) else v for v in col],
                index=col.index).infer_objects()
    return out if out is not None else df


def dict_encode_strings(tbl: pa.Table, threshold: float = 0.7,
                        sample: int = 8192, small_table: int = 65536) -> pa.Table:
    """Dictionary-encode low-cardinality string columns so filters/joins/
    groupbys run on int32 codes on device (reference: dict-encoded string
    arrays, bodo/libs/dict_arr_ext.py).

    SPMD-safe: each rank votes on its shard sample and the decision is the
    AND across ranks (a per-rank decision would give different column kinds
    per rank and deadlock the shuffle collective sequence)."""
    import pyarrow.compute as pc

    want = []
    for i, f in enumerate(tbl.schema):
        col = tbl.column(i)
        w = False
        if pa.types.is_string(f.type) or pa.types.is_large_string(f.type):
            if pa.types.is_string(f.type):
                col = col.cast(pa.large_string())
            head = col.slice(0, min(sample, len(col)))
            try:
                nuniq = len(pc.unique(head.combine_chunks()))
            except Exception:
                nuniq = len(head)
            # empty shards vote yes so non-empty shards decide; small tables
            # always encode (tiny dictionaries, and their values are the
            # common equality-filter targets, e.g. nation/region names)
            w = (len(head) == 0 or len(col) <= small_table
                 or nuniq <= max(1, int(len(head) * threshold)))
        want.append(w)
    from ..parallel import comm

    # SPMD-symmetric condition: empty shards infer pa.null for object
    # columns, so null-typed columns must count as potential strings or the
    # vote participation diverges across ranks
    has_string_cols = any(
        pa.types.is_string(f.type) or pa.types.is_large_string(f.type)
        or pa.types.is_null(f.type)
        for f in tbl.schema)
    # the vote is a COLLECTIVE: run it only when string columns exist, so
    # schemaless/numeric from_pandas calls never join a collective (uneven
    # per-rank call counts would deadlock the sequence otherwise)
    if has_string_cols and comm.initialized() and comm.get_world_size() > 1:
        votes = comm.allgather_obj((want, tbl.schema))
        want = [all(v[0][i] for v in votes) for i in range(len(want))]
        # schema consensus: EMPTY shards infer pa.null for object columns
        # (strings/decimals/lists) — adopt a non-empty rank's type so every
        # rank presents identical column kinds to the collective sequence
        # (reference analog: the plan's empty_data schema, plan.py:44)
        local_fields = list(tbl.schema)
        fixed = []
        changed = False
        for i, f in enumerate(local_fields):
            t_use = f.type
            if pa.types.is_null(t_use):
                for _, sch in votes:
                    if not pa.types.is_null(sch.field(i).type):
                        t_use = sch.field(i).type
                        break
            if t_use != f.type:
                changed = True
                fixed.append(tbl.column(i).cast(t_use))
            else:
                fixed.append(tbl.column(i))
        if changed:
            tbl = pa.table(dict(zip(tbl.column_names, fixed)))
    if not any(want):
        return tbl
    new_cols = []
    for i, f in enumerate(tbl.schema):
        col = tbl.column(i)
        if want[i] and (pa.types.is_string(f.type)
                        or pa.types.is_large_string(f.type)):
            if pa.types.is_string(f.type):
                col = col.cast(pa.large_string())
            col = pc.dictionary_encode(col.combine_chunks())
        new_cols.append(col)
    return pa.table(dict(zip(tbl.column_names, new_cols)))
