#!/usr/bin/env python
"""TPCx-BB Q05 + Q26 benchmark (BASELINE config 5: SQL plan + JIT/HIP UDF
kernels).  Synthetic web_clickstreams / store_sales shaped data, torchrun
SPMD contract like bench.py.

    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        benchmarks/bench_tpcxbb.py --scale 10 --steps 2 --warmup 1
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def gen_q05(n_clicks, rank, world, rng):
    import pandas as pd

    base, rem = divmod(n_clicks, world)
    n = base + (1 if rank < rem else 0)
    clicks = pd.DataFrame({
        "wcs_user_sk": rng.integers(1, max(n_clicks // 50, 2), n),
        "wcs_item_sk": rng.integers(1, 1000, n),
        "i_category_id": rng.integers(1, 8, n).astype(np.int64),
    })
    return clicks


def q05(bpd, clicks):
    """Per-user clicks-per-category pivot + college_education/male labels →
    logistic regression (reference: e2e-tests/tpcx-bb Q05)."""
    from bodo_amd.ml import LogisticRegression
    from bodo_amd.sql import BodoSQLContext

    bc = BodoSQLContext({"wc": clicks})
    pivot = bc.sql("""
        select wcs_user_sk,
               sum(case when i_category_id = 1 then 1 else 0 end) as c1,
               sum(case when i_category_id = 2 then 1 else 0 end) as c2,
               sum(case when i_category_id = 3 then 1 else 0 end) as c3,
               sum(case when i_category_id = 4 then 1 else 0 end) as c4,
               sum(case when i_category_id = 5 then 1 else 0 end) as c5,
               count(*) as clicks
        from wc group by wcs_user_sk
    """)
    df = pivot.to_pandas()
    X = df[["c1", "c2", "c3", "c4", "c5"]].to_numpy(dtype=np.float64)
    y = (df["clicks"].to_numpy() > np.median(df["clicks"])).astype(np.float64)
    m = LogisticRegression(max_iter=20)
    m.fit(X, y)
    return float(m.score(X, y))


def q26(bpd, ss, item):
    sale_items = ss.merge(item, left_on="ss_item_sk", right_on="i_item_sk")

    def id1(x):
        return (x == 1).sum()

    def id2(x):
        return (x == 2).sum()

    agg = sale_items.groupby("ss_customer_sk", as_index=False).agg(
        cnt=bpd.NamedAgg("ss_item_sk", "count"),
        c1=bpd.NamedAgg("i_class_id", id1),
        c2=bpd.NamedAgg("i_class_id", id2))
    agg = agg[agg.cnt > 5]
    return len(agg.sort_values("ss_customer_sk").to_pandas())


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--scale", type=float, default=10.0,
                   help="millions of clickstream rows")
    p.add_argument("--steps", type=int, default=2)
    p.add_argument("--warmup", type=int, default=1)
    args = p.parse_args()

    import pandas as pd

    import bodo_amd  # noqa: F401
    import bodo_amd.config as cfg
    import bodo_amd.pandas as bpd
    from bodo_amd.parallel import comm

    on_gpu = torch.cuda.is_available()
    cfg.DEVICE = "cuda" if on_gpu else "cpu"
    if on_gpu:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
    rank, world = comm.get_rank(), comm.get_world_size()

    n_clicks = int(args.scale * 1_000_000)
    rng = np.random.default_rng(100 + rank)
    clicks = gen_q05(n_clicks, rank, world, rng)
    n_ss = n_clicks // 2
    base, rem = divmod(n_ss, world)
    n_local = base + (1 if rank < rem else 0)
    ss = pd.DataFrame({"ss_item_sk": rng.integers(1, 500, n_local),
                       "ss_customer_sk": rng.integers(1, n_ss // 100 + 2,
                                                      n_local)})
    item = pd.DataFrame({
        "i_item_sk": np.arange(1, 501),
        "i_class_id": rng.integers(1, 16, 500).astype(np.int32),
        "i_category": rng.choice(["Books", "Music", "Home"], 500)})

    def one_step():
        b_clicks = bpd.from_pandas(clicks)
        acc = q05(bpd, b_clicks)
        n26 = q26(bpd, bpd.from_pandas(ss), bpd.from_pandas(item))
        return acc, n26

    for _ in range(args.warmup):
        one_step()
    comm.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        acc, n26 = one_step()
    if on_gpu:
        torch.cuda.synchronize()
    comm.barrier()
    elapsed = max(comm.allgather_obj(time.perf_counter() - t0))
    if rank == 0:
        print(json.dumps({
            "metric": "tpcxbb_q05_q26_elapsed_s",
            "value": elapsed / args.steps,
            "unit": "s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000,
            "higher_is_better": False,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "fp64",
            "data": "synthetic clickstream/store_sales (in-memory)",
            "config": {"model": "tpcxbb_q05_q26",
                       "clicks": n_clicks, "q05_train_acc": acc,
                       "q26_rows": n26,
                       "parallelism": f"dp{world}-rccl"},
        }), flush=True)


if __name__ == "__main__":
    main()
