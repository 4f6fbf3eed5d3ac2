"""Spawn mode: a plain python process transparently executes lazy frames on
N worker processes (one per GPU), like the reference's
MPI_Comm_spawn-based spawner (bodo/spawn/spawner.py:134, worker.py:636) —
redesigned with a socket control plane (cloudpickled commands) and a
torch.distributed (RCCL/gloo) data plane among the workers only.

Activated by BODO_NUM_WORKERS=N; the driver process is NOT in the worker
process group, so user code never blocks on collectives.
"""

from __future__ import annotations

import atexit
import os
import pickle
import socket
import struct
import subprocess
import sys
import time
from typing import Any, Dict, List, Optional

try:
    import cloudpickle
except ImportError:  # pragma: no cover
    cloudpickle = pickle

_SPAWNER: Optional["Spawner"] = None


def active() -> bool:
    """True in the user process when spawn mode should route executions."""
    from .. import config
    from . import comm

    if os.environ.get("BODO_AMD_WORKER"):
        return False  # we ARE a worker
    if comm.initialized():
        return False  # already SPMD (torchrun)
    return config.NUM_WORKERS > 0


def get_spawner() -> "Spawner":
    global _SPAWNER
    if _SPAWNER is None:
        from .. import config

        _SPAWNER = Spawner(config.NUM_WORKERS)
    return _SPAWNER


def _send_msg(sock: socket.socket, obj) -> None:
    payload = cloudpickle.dumps(obj)
    sock.sendall(struct.pack("<Q", len(payload)) + payload)


def _recv_msg(sock: socket.socket):
    hdr = _recv_exact(sock, 8)
    (n,) = struct.unpack("<Q", hdr)
    return pickle.loads(_recv_exact(sock, n))


def _recv_exact(sock: socket.socket, n: int) -> bytes:
    buf = bytearray()
    while len(buf) < n:
        chunk = sock.recv(min(n - len(buf), 1 << 20))
        if not chunk:
            raise ConnectionError("worker connection closed")
        buf.extend(chunk)
    return bytes(buf)


class Spawner:
    def __init__(self, n_workers: int):
        self.n = n_workers
        self.procs: List[subprocess.Popen] = []
        self.socks: List[socket.socket] = []
        self._start()
        atexit.register(self.shutdown)

    def _start(self):
        listener = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        listener.bind(("127.0.0.1", 0))
        listener.listen(self.n)
        ctrl_port = listener.getsockname()[1]
        # find a free port for the workers' own process group
        pg = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        pg.bind(("127.0.0.1", 0))
        pg_port = pg.getsockname()[1]
        pg.close()
        for rank in range(self.n):
            env = dict(os.environ)
            env.update({
                "BODO_AMD_WORKER": "1",
                "BODO_AMD_CTRL_PORT": str(ctrl_port),
                "RANK": str(rank),
                "LOCAL_RANK": str(rank),
                "WORLD_SIZE": str(self.n),
                "MASTER_ADDR": "127.0.0.1",
                "MASTER_PORT": str(pg_port),
                "BODO_NUM_WORKERS": "0",
            })
            p = subprocess.Popen(
                [sys.executable, "-m", "bodo_amd.parallel.worker"],
                env=env, cwd=os.getcwd())
            self.procs.append(p)
        conns = {}
        listener.settimeout(300)
        for _ in range(self.n):
            c, _addr = listener.accept()
            hello = _recv_msg(c)
            conns[hello["rank"]] = c
        listener.close()
        self.socks = [conns[r] for r in range(self.n)]

    # ------------------------------------------------------------------
    def command(self, cmd: Dict[str, Any], per_rank: Optional[List[dict]] = None):
        """Broadcast a command (optionally with per-rank extras); collect
        one reply per worker (re-raising remote exceptions)."""
        for r, s in enumerate(self.socks):
            msg = dict(cmd)
            if per_rank is not None:
                msg.update(per_rank[r])
            _send_msg(s, msg)
        replies = []
        for s in self.socks:
            rep = _recv_msg(s)
            if rep.get("error"):
                raise RuntimeError(f"worker error:\n{rep['error']}")
            replies.append(rep)
        return replies

    def exec_plan(self, plan, host_objects: Dict[str, Any]):
        import uuid

        res_id = f"res-{uuid.uuid4().hex}"
        reps = self.command({"cmd": "exec_plan", "plan": plan,
                             "objects": host_objects, "res_id": res_id})
        return reps  # [{res_id, names, length}]

    def gather(self, res_id: str):
        import pyarrow as pa

        reps = self.command({"cmd": "gather", "res_id": res_id})
        tables = [pickle.loads(r["arrow"]) for r in reps]
        return pa.concat_tables(tables, promote_options="permissive")

    def delete(self, res_id: str):
        try:
            self.command({"cmd": "delete", "res_id": res_id})
        except Exception:
            pass

    def exec_func(self, func, args, kwargs):
        import uuid

        reps = self.command({"cmd": "exec_func", "func": func,
                             "args": args, "kwargs": kwargs,
                             "res_id": f"res-{uuid.uuid4().hex}"})
        return reps

    def shutdown(self):
        for s in self.socks:
            try:
                _send_msg(s, {"cmd": "exit"})
            except Exception:
                pass
        deadline = time.time() + 10
        for p in self.procs:
            try:
                p.wait(timeout=max(0.1, deadline - time.time()))
            except Exception:
                p.kill()
        self.socks = []
        self.procs = []
        global _SPAWNER
        _SPAWNER = None


# ----------------------------------------------------------------------
# spawner-side execution entry used by the frontend
# ----------------------------------------------------------------------

def collect_plan_objects(plan) -> Dict[str, Any]:
    """Host objects (from_pandas frames) the workers need for this plan."""
    from ..engine import executor as ex
    from ..plan import nodes as pn

    out = {}
    for node in pn.walk(plan):
        if isinstance(node, pn.PandasScan) and not node.distributed:
            try:
                out[node.data_id] = ex.get_object(node.data_id)
            except KeyError:
                pass  # already shipped; worker has it
    return out


class RemoteResult:
    """Handle to a distributed result living in the workers' registries."""

    def __init__(self, res_id: str, names, length: int):
        self.res_id = res_id
        self.names = list(names)
        self.length = length

    def __del__(self):
        try:
            if _SPAWNER is not None:
                _SPAWNER.delete(self.res_id)
        except Exception:
            pass
