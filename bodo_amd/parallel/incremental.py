"""Incremental shuffle state (reference: IncrementalShuffleState,
bodo/libs/streaming/_shuffle.h:777 — threshold-buffered async MPI
Issend/Improbe with Ibarrier termination).

MI355X redesign: RCCL has no wildcard receive, so the async-p2p protocol
becomes CADENCED COLLECTIVE ROUNDS: every rank appends morsel rows into
per-destination buffers; at a fixed batch cadence (or when the pending
bytes exceed the threshold vote) all ranks enter an exchange round — one
small count all-to-all + the packed-table alltoallv — and a done-flag
all-reduce decides termination, so ranks with uneven morsel counts keep
participating with empty payloads until everyone has drained (the
Ibarrier-consensus analog).  NCCL collectives enqueue asynchronously on
the comm stream, so the data movement of round k overlaps the partition
and aggregation kernels of the following morsels."""

from __future__ import annotations

from typing import Callable, List, Optional

import torch

from ..core.table import Table
from . import comm

DEFAULT_THRESHOLD = 50 << 20  # bytes, reference DEFAULT_SHUFFLE_THRESHOLD
DEFAULT_CADENCE = 8           # batches between consensus rounds


class IncrementalShuffle:
    def __init__(self, keys: List[str], on_receive: Callable[[Table], None],
                 threshold: int = DEFAULT_THRESHOLD,
                 cadence: int = DEFAULT_CADENCE):
        self.keys = keys
        self.on_receive = on_receive
        self.threshold = threshold
        self.cadence = cadence
        self.pending: List[Table] = []
        self.pending_bytes = 0
        self.batches_since_round = 0
        self.rounds = 0

    def append(self, batch: Table) -> None:
        """Buffer a morsel; runs an exchange round at the cadence or when
        any rank's pending bytes exceed the threshold (majority-free vote:
        MAX all-reduce of pending bytes rides the round consensus)."""
        if len(batch):
            self.pending.append(batch)
            self.pending_bytes += batch.nbytes()
        self.batches_since_round += 1
        if self.batches_since_round >= self.cadence:
            self._round(done=False)

    def _round(self, done: bool) -> bool:
        """One consensus + exchange round; returns the global done flag."""
        from .. import ops

        w = comm.get_world_size()
        self.batches_since_round = 0
        if w == 1:
            for t in self.pending:
                self.on_receive(t)
            self.pending.clear()
            self.pending_bytes = 0
            return done
        dev = comm._comm_device()
        import torch.distributed as dist

        flags = torch.tensor([0 if done else 1, self.pending_bytes],
                             dtype=torch.int64, device=dev)
        dist.all_reduce(flags, op=dist.ReduceOp.MAX)
        any_not_done = bool(flags[0].item())
        max_pending = int(flags[1].item())
        if max_pending > 0:
            local = ops.concat_tables(self.pending) if self.pending else None
            self.pending.clear()
            self.pending_bytes = 0
            recv = _exchange(local, self.keys)
            if recv is not None and len(recv):
                self.on_receive(recv)
        self.rounds += 1
        return not any_not_done

    def finish(self) -> None:
        """Drain: keep participating in rounds until every rank is done."""
        while True:
            if self._round(done=True):
                return


def _exchange(local: Optional[Table], keys) -> Optional[Table]:
    """One hash-partition alltoallv; ranks with nothing pending this round
    join the collective sequence with a schema-matched empty shard (the
    schema rides the consensus as a 0-row arrow table)."""
    from .. import config, ops

    w = comm.get_world_size()
    proto = None
    if local is not None:
        proto = ops.slice_table(local, 0, 0).to_device("cpu").to_arrow()
    metas = comm.allgather_obj(proto)
    schema_tbl = next((m for m in metas if m is not None), None)
    if schema_tbl is None:
        return None  # nobody had rows this round
    if local is None:
        local = Table.from_arrow(schema_tbl, config.default_device())
    h = ops.hash_columns([local.column(k) for k in keys]) if len(local) else         torch.zeros(0, dtype=torch.int64, device=local.device)
    part = torch.remainder(h, w)
    part = torch.where(part < 0, part + w, part)
    return comm.shuffle_table(local, part)
