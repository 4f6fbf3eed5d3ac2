"""Public distributed primitives (reference: bodo/libs/distributed_api.py —
gatherv:713, scatterv, bcast, allgatherv, rebalance, random_shuffle,
get_rank/get_size/barrier, get_gpu_ranks)."""

from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd
import torch

from . import comm


def get_rank() -> int:
    return comm.get_rank()


def get_size() -> int:
    return comm.get_world_size()


def barrier():
    comm.barrier()


def get_gpu_ranks():
    """Ranks pinned to GPUs (one per device; reference:
    distributed_api.py:2883)."""
    n = torch.cuda.device_count() if torch.cuda.is_available() else 0
    return list(range(min(n, get_size()))) if n else []


def _as_frame(data):
    from ..pandas.frame import BodoDataFrame

    return data


def gatherv(data, root: int = 0):
    """Gather a distributed DataFrame/Series/array onto `root` (None
    elsewhere)."""
    from ..pandas.frame import BodoDataFrame
    from ..pandas.series import BodoSeries

    if isinstance(data, (BodoDataFrame,)):
        shard = data.execute()
        full = comm.gather_table(shard, root)
        return None if full is None else full.to_pandas()
    if isinstance(data, BodoSeries):
        ser = data.to_pandas()
        return ser if get_rank() == root else None
    if isinstance(data, (pd.DataFrame, pd.Series, np.ndarray)):
        parts = comm.gather_obj(data, root)
        if parts is None:
            return None
        if isinstance(data, np.ndarray):
            return np.concatenate(parts)
        return pd.concat(parts, ignore_index=True)
    raise TypeError(type(data))


def allgatherv(data):
    """Replicate the concatenation of all shards on every rank."""
    from ..pandas.frame import BodoDataFrame

    if isinstance(data, BodoDataFrame):
        return data.to_pandas()
    parts = comm.allgather_obj(data)
    if isinstance(data, np.ndarray):
        return np.concatenate(parts)
    if isinstance(data, (pd.DataFrame, pd.Series)):
        return pd.concat(parts, ignore_index=True)
    return parts


def scatterv(data, root: int = 0):
    """Scatter a host object from `root` into this rank's block."""
    obj = comm.bcast_obj(data if get_rank() == root else None, root)
    n = len(obj)
    w, r = get_size(), get_rank()
    base, rem = divmod(n, w)
    start = r * base + min(r, rem)
    stop = start + base + (1 if r < rem else 0)
    if isinstance(obj, pd.DataFrame):
        return obj.iloc[start:stop].reset_index(drop=True)
    if isinstance(obj, pd.Series):
        return obj.iloc[start:stop].reset_index(drop=True)
    return obj[start:stop]


def bcast(data, root: int = 0):
    return comm.bcast_obj(data, root)


def rebalance(data):
    """Rebalance a distributed frame to equal block sizes (reference:
    distributed_api.py rebalance)."""
    from ..core.table import Table
    from ..pandas.frame import BodoDataFrame
    from ..plan import nodes as pn
    from ..engine import executor as ex

    if isinstance(data, BodoDataFrame):
        shard = data.execute()
        w = get_size()
        if w == 1:
            return data
        lengths = comm.allgather_obj(len(shard))
        total = sum(lengths)
        prefix = sum(lengths[:get_rank()])
        # target rank for each local row by global position
        base, rem = divmod(total, w)
        bounds = np.cumsum([0] + [base + (1 if i < rem else 0)
                                  for i in range(w)])
        gpos = np.arange(prefix, prefix + len(shard))
        part = np.searchsorted(bounds, gpos, side="right") - 1
        new_shard = comm.shuffle_table(
            shard, torch.from_numpy(part.astype(np.int64)).to(shard.device))
        key = ex.register_object(new_shard)
        return BodoDataFrame(
            pn.PandasScan(key, tuple(new_shard.names), distributed=True),
            list(new_shard.names))
    raise TypeError(type(data))


def random_shuffle(data, seed: Optional[int] = None):
    """Randomly redistribute rows across ranks."""
    from ..pandas.frame import BodoDataFrame
    from ..plan import nodes as pn
    from ..engine import executor as ex

    if isinstance(data, BodoDataFrame):
        shard = data.execute()
        g = torch.Generator()
        g.manual_seed((seed or 0) + get_rank())
        part = torch.randint(0, get_size(), (len(shard),), generator=g)
        new_shard = comm.shuffle_table(shard, part.to(shard.device))
        key = ex.register_object(new_shard)
        return BodoDataFrame(
            pn.PandasScan(key, tuple(new_shard.names), distributed=True),
            list(new_shard.names))
    raise TypeError(type(data))
