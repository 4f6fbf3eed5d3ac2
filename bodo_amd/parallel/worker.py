"""Spawn-mode worker process (reference: bodo/spawn/worker.py:636
worker_loop): connects to the spawner's control socket, joins the worker
process group, and executes shipped plans/functions against its shard,
keeping results in a registry keyed by result id."""

from __future__ import annotations

import os
import pickle
import socket
import struct
import sys
import traceback
import uuid


def _send(sock, obj):
    payload = pickle.dumps(obj)
    sock.sendall(struct.pack("<Q", len(payload)) + payload)


def _recv(sock):
    buf = bytearray()
    while len(buf) < 8:
        chunk = sock.recv(8 - len(buf))
        if not chunk:
            return None
        buf.extend(chunk)
    (n,) = struct.unpack("<Q", bytes(buf))
    data = bytearray()
    while len(data) < n:
        chunk = sock.recv(min(n - len(data), 1 << 20))
        if not chunk:
            return None
        data.extend(chunk)
    return pickle.loads(bytes(data))


def main():
    import warnings

    warnings.filterwarnings("ignore")
    import torch

    import bodo_amd  # noqa: F401  (auto-inits the worker process group)
    import bodo_amd.config as cfg
    from bodo_amd.engine import executor as ex
    from bodo_amd.parallel import comm

    comm.init_from_env()
    rank = comm.get_rank()
    if torch.cuda.is_available():
        torch.cuda.set_device(rank % torch.cuda.device_count())
        cfg.DEVICE = "cuda"
    ctrl = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    ctrl.connect(("127.0.0.1", int(os.environ["BODO_AMD_CTRL_PORT"])))
    _send(ctrl, {"rank": rank})

    while True:
        msg = _recv(ctrl)
        if msg is None or msg.get("cmd") == "exit":
            break
        try:
            reply = _dispatch(msg, ex, comm, cfg)
        except Exception:
            reply = {"error": traceback.format_exc()}
        _send(ctrl, reply)
    try:
        import torch.distributed as dist

        if dist.is_initialized():
            dist.destroy_process_group()
    except Exception:
        pass


def _dispatch(msg, ex, comm, cfg):
    cmd = msg["cmd"]
    if cmd == "exec_plan":
        for key, obj in (msg.get("objects") or {}).items():
            ex.register_object(obj, key)
        ctx = ex.ExecutionContext()
        shard = ex.execute(msg["plan"], ctx)
        res_id = msg.get("res_id") or f"res-{uuid.uuid4().hex}"
        ex.register_object(shard, res_id)
        total = sum(comm.allgather_obj(len(shard)))
        return {"res_id": res_id, "names": list(shard.names),
                "length": total}
    if cmd == "gather":
        shard = ex.get_object(msg["res_id"])
        at = shard.to_device("cpu").to_arrow()
        return {"arrow": pickle.dumps(at)}
    if cmd == "delete":
        ex.delete_object(msg["res_id"])
        return {}
    if cmd == "exec_func":
        import pandas as pd

        import bodo_amd.pandas as bpd
        from bodo_amd.pandas.frame import BodoDataFrame, from_pandas_df

        func = msg["func"]
        args = [from_pandas_df(a) if isinstance(a, pd.DataFrame) else a
                for a in msg["args"]]
        kwargs = {k: from_pandas_df(v) if isinstance(v, pd.DataFrame) else v
                  for k, v in msg["kwargs"].items()}
        res = func(*args, **kwargs)
        if isinstance(res, BodoDataFrame):
            shard = res.execute()
            res_id = msg.get("res_id") or f"res-{uuid.uuid4().hex}"
            ex.register_object(shard, res_id)
            total = sum(comm.allgather_obj(len(shard)))
            return {"kind": "frame", "res_id": res_id,
                    "names": list(shard.names), "length": total}
        if hasattr(res, "_frame"):  # _IndexedAggResult
            shard = res._frame.execute()
            res_id = msg.get("res_id") or f"res-{uuid.uuid4().hex}"
            ex.register_object(shard, res_id)
            total = sum(comm.allgather_obj(len(shard)))
            return {"kind": "frame", "res_id": res_id,
                    "names": list(shard.names), "length": total}
        from bodo_amd.pandas.scalar import BodoScalar

        if isinstance(res, BodoScalar):
            res = res.value  # materialize worker-side lazy scalars
        return {"kind": "value", "value": res}
    raise ValueError(f"unknown command {cmd}")


if __name__ == "__main__":
    main()
