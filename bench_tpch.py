#!/usr/bin/env python
"""TPC-H benchmark runner: 22 queries on synthetic
dbgen-shaped data at --sf, N GPUs (torchrun SPMD, same contract as bench.py).

    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        bench_tpch.py --sf 100 --queries 1,3,5 --steps 2 --warmup 1

Rank 0 prints one JSON line: total elapsed over the selected queries.
"""

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "benchmarks"))


def load_tables(sf, rank, world, device):
    """Generate the rank shard of all 8 tables and register as device Tables."""
    import pandas as pd

    from bodo_amd.core.table import Table
    from tpch_data import TABLES, gen_table

    out = {}
    for t in TABLES:
        df = gen_table(t, sf, rank, world)
        out[t] = Table.from_pandas(df, device)
    return out


def wrap_frames(tables):
    from bodo_amd.engine import executor as ex
    from bodo_amd.pandas.frame import BodoDataFrame
    from bodo_amd.plan import nodes as pn

    frames = {}
    keys = []
    for name, tbl in tables.items():
        key = ex.register_object(tbl)
        keys.append(key)
        frames[name] = BodoDataFrame(
            pn.PandasScan(key, tuple(tbl.names), distributed=True),
            list(tbl.names))
    return frames, keys


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--sf", type=float, default=10.0)
    p.add_argument("--queries", type=str, default="all")
    p.add_argument("--steps", type=int, default=1)
    p.add_argument("--warmup", type=int, default=1)
    args = p.parse_args()

    import bodo_amd  # noqa: F401
    import bodo_amd.config as cfg
    import bodo_amd.pandas as bpd
    from bodo_amd.parallel import comm
    from tpch_queries import ALL

    on_gpu = torch.cuda.is_available()
    device = "cuda" if on_gpu else "cpu"
    cfg.DEVICE = device
    if on_gpu:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
        import bodo_amd_kernels  # noqa: F401
    rank, world = comm.get_rank(), comm.get_world_size()

    qlist = (list(range(1, 23)) if args.queries == "all"
             else [int(x) for x in args.queries.split(",")])
    tables = load_tables(args.sf, rank, world, device)

    def run_all():
        times = {}
        for qn in qlist:
            frames, keys = wrap_frames(tables)
            t0 = time.perf_counter()
            res = ALL[qn](bpd, frames)
            shard = res.execute() if hasattr(res, "execute") else \
                res._frame.execute()
            if on_gpu:
                torch.cuda.synchronize()
            comm.barrier()
            times[qn] = time.perf_counter() - t0
            from bodo_amd.engine import executor as ex

            for k in keys:
                ex.delete_object(k)
        return times

    for _ in range(args.warmup):
        run_all()
    comm.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    per_q = {}
    for _ in range(args.steps):
        per_q = run_all()
    if on_gpu:
        torch.cuda.synchronize()
    comm.barrier()
    elapsed = (time.perf_counter() - t0) / args.steps
    elapsed = max(comm.allgather_obj(elapsed))
    if rank == 0:
        print(json.dumps({
            "metric": f"tpch_sf{args.sf:g}_total_elapsed_s",
            "value": elapsed,
            "unit": "s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed * 1000.0,
            "higher_is_better": False,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "fp64",
            "data": "synthetic dbgen-shaped (in-HBM, generation untimed)",
            "config": {
                "model": f"tpch_sf{args.sf:g}",
                "queries": qlist,
                "per_query_s": {str(k): round(v, 4) for k, v in per_q.items()},
                "parallelism": f"dp{world}-hash-shuffle-rccl",
            },
        }), flush=True)


if __name__ == "__main__":
    main()
