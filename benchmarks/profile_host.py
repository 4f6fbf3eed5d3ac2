#!/usr/bin/env python
"""Host-overhead profiler for TPC-H queries: runs selected queries at --sf
with data resident on the device and prints per-query wall time plus the
cProfile top functions (host Python cost).  Used to attack the "join-heavy
queries are host-bound" finding (q5 ~1 ms GPU inside a ~145 ms step).

    python benchmarks/profile_host.py --sf 1 --queries q5,q7 --runs 5
"""

import argparse
import cProfile
import io
import os
import pstats
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--sf", type=float, default=1.0)
    p.add_argument("--queries", type=str, default="q5")
    p.add_argument("--runs", type=int, default=5)
    p.add_argument("--top", type=int, default=30)
    p.add_argument("--sort", type=str, default="cumulative")
    args = p.parse_args()

    import torch

    import bodo_amd.config as cfg
    import bodo_amd.pandas as bpd
    from bench_tpch import load_tables, wrap_frames
    import tpch_queries as tq

    device = "cuda" if torch.cuda.is_available() else "cpu"
    cfg.DEVICE = device
    tables = load_tables(args.sf, 0, 1, device)
    frames, keys = wrap_frames(tables)

    def run(qf):
        r = qf(bpd, frames)
        out = r.execute() if hasattr(r, "execute") else r.to_pandas()
        if device == "cuda":
            torch.cuda.synchronize()
        return out

    for qname in args.queries.split(","):
        qf = getattr(tq, qname)
        run(qf)  # warmup
        t0 = time.perf_counter()
        for _ in range(args.runs):
            run(qf)
        dt = (time.perf_counter() - t0) / args.runs * 1000
        pr = cProfile.Profile()
        pr.enable()
        for _ in range(args.runs):
            run(qf)
        pr.disable()
        s = io.StringIO()
        pstats.Stats(pr, stream=s).sort_stats(args.sort).print_stats(args.top)
        print(f"==== {qname}: {dt:.1f} ms/run (sf={args.sf}, {device}) ====")
        print(s.getvalue())


if __name__ == "__main__":
    main()
