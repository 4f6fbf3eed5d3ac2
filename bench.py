#!/usr/bin/env python
"""Flagship benchmark: NYC-Taxi Q1 ("Monthly Trips with Precipitation",
BASELINE.md config 2/3) on synthetic FHVHV-shaped data, N MI355X GPUs.

Reference: benchmarks/nyc_taxi/bodo/nyc_taxi_precipitation.py (query shape),
run on 1,036,465,968 rows x the columns the query touches
(benchmarks/nyc_taxi/README.md).  Data is synthetic and resident in HBM
(there is no network for the S3 dataset); generation is untimed; the timed
region is the full query: datetime extraction, inner merge with weather,
UDF time-bucketing, 6-key hash groupby (count + mean), and sort.

Driver contract:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
Rank 0 prints one JSON line with the aggregate rows/sec.
"""

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

TOTAL_ROWS = 1_036_465_968  # FHVHV row count, benchmarks/nyc_taxi/README.md
N_LOCATIONS = 265
YEAR_NS = 365 * 86400 * 10**9
BASE_TS = 1_672_531_200_000_000_000  # 2023-01-01


def make_trips_shard(n, rank, device):
    """Synthetic trips shard generated directly in HBM."""
    import pyarrow as pa

    from bodo_amd.core import types as bt
    from bodo_amd.core.column import Column
    from bodo_amd.core.table import Table

    g = torch.Generator(device=device)
    g.manual_seed(12345 + rank)
    dev = torch.device(device)
    pickup = BASE_TS + torch.randint(0, YEAR_NS, (n,), generator=g,
                                     dtype=torch.int64, device=dev)
    pu = torch.randint(1, N_LOCATIONS + 1, (n,), generator=g,
                       dtype=torch.int64, device=dev)
    do = torch.randint(1, N_LOCATIONS + 1, (n,), generator=g,
                       dtype=torch.int64, device=dev)
    miles = torch.rand(n, generator=g, dtype=torch.float64, device=dev) * 20.0
    lic = torch.randint(0, 4, (n,), generator=g, dtype=torch.int32, device=dev)
    lic_dict = pa.array(["HV0002", "HV0003", "HV0004", "HV0005"],
                        type=pa.large_string())
    pu_col = Column(bt.int64, pu)
    pu_col.val_range = (1, N_LOCATIONS)
    do_col = Column(bt.int64, do)
    do_col.val_range = (1, N_LOCATIONS)
    cols = {
        "hvfhs_license_num": Column(bt.dictionary, lic, dictionary=lic_dict),
        "pickup_datetime": Column(bt.timestamp_ns, pickup),
        "PULocationID": pu_col,
        "DOLocationID": do_col,
        "trip_miles": Column(bt.float64, miles),
    }
    return Table(list(cols), list(cols.values()), n)


def make_weather():
    import pandas as pd

    rng = np.random.default_rng(7)
    dates = pd.date_range("2023-01-01", "2023-12-31")
    return pd.DataFrame({"DATE": dates,
                         "PRCP": rng.exponential(0.05, len(dates)).round(2)})


def get_time_bucket(t):
    if t in (8, 9, 10):
        return "morning"
    if t in (11, 12, 13, 14, 15):
        return "midday"
    if t in (16, 17, 18):
        return "afternoon"
    if t in (19, 20, 21):
        return "evening"
    return "other"


def run_query(bpd, trips_table, weather_df):
    """The exact NYC-Taxi Q1 query shape through the lazy frontend."""
    from bodo_amd.engine import executor as ex
    from bodo_amd.pandas.frame import BodoDataFrame
    from bodo_amd.plan import nodes as pn

    key = ex.register_object(trips_table)
    t = BodoDataFrame(pn.PandasScan(key, tuple(trips_table.names),
                                    distributed=True),
                      list(trips_table.names))
    w = bpd.from_pandas(weather_df)
    w = w.rename(columns={"DATE": "date", "PRCP": "precipitation"})
    w["date"] = w["date"].dt.date
    t["date"] = t["pickup_datetime"].dt.date
    t["month"] = t["pickup_datetime"].dt.month
    t["hour"] = t["pickup_datetime"].dt.hour
    t["weekday"] = t["pickup_datetime"].dt.dayofweek.isin([0, 1, 2, 3, 4])
    m = t.merge(w, on="date", how="inner")
    m["date_with_precipitation"] = m["precipitation"] > 0.1
    m["time_bucket"] = m.hour.map(get_time_bucket)
    g = m.groupby(
        ["PULocationID", "DOLocationID", "month", "weekday",
         "date_with_precipitation", "time_bucket"],
        as_index=False).agg(
        trips=bpd.NamedAgg("hvfhs_license_num", "count"),
        avg_distance=bpd.NamedAgg("trip_miles", "mean"))
    g = g.sort_values(
        by=["PULocationID", "DOLocationID", "month", "weekday",
            "date_with_precipitation", "time_bucket"])
    shard = g.execute()  # materialized device shard
    ex.delete_object(key)
    return shard


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--rows", type=int, default=TOTAL_ROWS,
                   help="total rows (dev override; judged runs use default)")
    p.add_argument("--parquet", action="store_true", default=True,
                   help="read the trips table from snappy parquet each step "
                        "with on-GPU decode (BASELINE config 2; the default)")
    p.add_argument("--no-parquet", dest="parquet", action="store_false",
                   help="skip parquet IO: query in-HBM synthetic shards")
    args = p.parse_args()

    import bodo_amd  # noqa: F401  (inits process group under torchrun)
    import bodo_amd.config as cfg
    from bodo_amd.parallel import comm

    on_gpu = torch.cuda.is_available()
    device = "cuda" if on_gpu else "cpu"
    cfg.DEVICE = device
    if on_gpu:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
    rank, world = comm.get_rank(), comm.get_world_size()
    if on_gpu:
        import bodo_amd_kernels  # noqa: F401 - fail loudly if ext missing
    import bodo_amd.pandas as bpd

    n_total = args.rows
    base, rem = divmod(n_total, world)
    n_local = base + (1 if rank < rem else 0)
    trips = make_trips_shard(n_local, rank, device)
    weather = make_weather()

    def log(msg):
        if rank == 0:
            print(f"[bench +{time.perf_counter() - T0:.1f}s] {msg}",
                  file=sys.stderr, flush=True)

    T0 = time.perf_counter()
    log(f"generated {n_local} rows on {device}")
    pq_path = None
    if args.parquet:
        # write this rank's shard as snappy parquet (untimed; the realistic
        # on-disk layout), then each step re-reads it with the on-GPU
        # decoder before the query
        import pyarrow.parquet as pq

        from bodo_amd import ops as _ops

        os.makedirs("/tmp/bodo_bench_pq", exist_ok=True)
        pq_path = f"/tmp/bodo_bench_pq/trips_rank{rank}.parquet"
        # chunked write keeps the host staging bounded at ~1.3 GB even for
        # the full 1B-row shard
        CHUNK = 1 << 25
        writer = None
        for s in range(0, len(trips), CHUNK):
            at = _ops.slice_table(trips, s, min(s + CHUNK, len(trips))) \
                .to_device("cpu").to_arrow()
            if writer is None:
                writer = pq.ParquetWriter(
                    pq_path, at.schema, compression="SNAPPY",
                    use_dictionary=["hvfhs_license_num"])
            writer.write_table(at, row_group_size=1 << 23)
            del at
        writer.close()
        # the generated device table is only the write source; freeing it
        # returns ~37 GB of HBM to the query working set at 1B rows
        del trips
        trips = None
        if on_gpu:
            torch.cuda.empty_cache()
        log(f"parquet written: {os.path.getsize(pq_path)} bytes")

    def read_trips():
        if pq_path is None:
            return trips
        from bodo_amd.engine.executor import ExecutionContext
        from bodo_amd.io import parquet_gpu

        ctx1 = ExecutionContext(device)
        ctx1.world, ctx1.rank = 1, 0  # rank-private file
        if on_gpu:
            t = parquet_gpu.read_shard_gpu(pq_path, None, ctx1)
            assert t is not None, "GPU parquet decode fell back"
            return t
        from bodo_amd.io import parquet as pio

        return pio.read_shard(pq_path, None, (), ctx1)

    def one_step():
        shard = run_query(bpd, read_trips(), weather)
        return sum(comm.allgather_obj(len(shard)))

    for _ in range(args.warmup):
        one_step()
        log("warmup step done")
    comm.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    out_rows = 0
    for _ in range(args.steps):
        out_rows = one_step()
        log("timed step done")
    if on_gpu:
        torch.cuda.synchronize()
    comm.barrier()
    elapsed = time.perf_counter() - t0
    elapsed = max(comm.allgather_obj(elapsed))
    ms_per_step = elapsed / args.steps * 1000.0
    rows_per_sec = n_total / (elapsed / args.steps)
    if rank == 0:
        print(json.dumps({
            "metric": "nyc_taxi_q1_rows_per_sec",
            "value": rows_per_sec,
            "unit": "rows/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "fp64",
            "data": ("synthetic snappy parquet (on-GPU decode in timed "
                     "region)" if args.parquet else
                     "synthetic (in-HBM, generation untimed; no parquet IO)"),
            "config": {
                "model": "nyc_taxi_q1_monthly_trips_precipitation",
                "rows": n_total,
                "out_groups": out_rows,
                "query": "dt-extract + inner-merge(weather) + udf-bucket + "
                         "6-key groupby(count,mean) + sort",
                "parallelism": f"dp{world}-hash-shuffle-rccl",
            },
        }), flush=True)


if __name__ == "__main__":
    main()
