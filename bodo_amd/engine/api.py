"""Execution entry points used by the frontend: route a plan either to the
local SPMD executor (torchrun / single process) or to spawn-mode workers."""

from __future__ import annotations

from typing import Tuple, Union

import pandas as pd

from ..parallel import comm, spawn
from . import executor as ex


def materialize(plan):
    """Execute a plan.  Returns ("local", Table shard) in SPMD/single mode or
    ("remote", RemoteResult) in spawn mode."""
    if spawn.active():
        sp = spawn.get_spawner()
        objs = spawn.collect_plan_objects(plan)
        reps = sp.exec_plan(plan, objs)
        r0 = reps[0]
        return "remote", spawn.RemoteResult(r0["res_id"], r0["names"],
                                            r0["length"])
    shard = ex.execute(plan, ex.ExecutionContext())
    return "local", shard


def collect(plan) -> pd.DataFrame:
    """Execute and return the FULL result as pandas in this process."""
    kind, res = materialize(plan)
    if kind == "remote":
        at = spawn.get_spawner().gather(res.res_id)
        return at.to_pandas()
    full = comm.allgather_table(res)
    return full.to_pandas()


def total_len(kind, res) -> int:
    if kind == "remote":
        return res.length
    return int(sum(comm.allgather_obj(len(res))))
