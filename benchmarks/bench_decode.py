#!/usr/bin/env python
"""Decode-only microbenchmark: writes an NYC-taxi-shaped snappy parquet file
once, then times read_shard_gpu end-to-end (disk -> pinned -> HBM -> decoded
columns), printing the phase breakdown from parquet_gpu.STATS."""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--rows", type=int, default=100_000_000)
    p.add_argument("--reps", type=int, default=2)
    args = p.parse_args()

    from bench import make_trips_shard
    from bodo_amd.engine.executor import ExecutionContext
    from bodo_amd.io import parquet_gpu as g

    device = "cuda" if torch.cuda.is_available() else "cpu"
    trips = make_trips_shard(args.rows, 0, device)
    import pyarrow.parquet as pq

    from bodo_amd import ops as _ops

    path = "/tmp/bench_decode.parquet"
    CHUNK = 1 << 25
    writer = None
    for s in range(0, len(trips), CHUNK):
        at = _ops.slice_table(trips, s, min(s + CHUNK, len(trips))) \
            .to_device("cpu").to_arrow()
        if writer is None:
            writer = pq.ParquetWriter(path, at.schema, compression="SNAPPY",
                                      use_dictionary=["hvfhs_license_num"])
        writer.write_table(at, row_group_size=1 << 23)
    writer.close()
    del trips
    if device == "cuda":
        torch.cuda.empty_cache()
    sz = os.path.getsize(path)

    ctx = ExecutionContext(device)
    ctx.world, ctx.rank = 1, 0
    # warm page cache + jit
    t = g.read_shard_gpu(path, None, ctx)
    if t is None and device == "cpu":
        print(json.dumps({"skipped": "device decode needs a GPU"}))
        return
    assert t is not None and len(t) == args.rows, (t and len(t), args.rows)
    if device == "cuda":
        torch.cuda.synchronize()
    for k in list(g.STATS):
        g.STATS[k] = 0
    t0 = time.perf_counter()
    for _ in range(args.reps):
        t = g.read_shard_gpu(path, None, ctx)
        if device == "cuda":
            torch.cuda.synchronize()
        del t
    dt = (time.perf_counter() - t0) / args.reps
    print(json.dumps({
        "rows": args.rows,
        "file_bytes": sz,
        "sec_per_read": dt,
        "rows_per_sec": args.rows / dt,
        "bytes_per_sec": sz / dt,
        "stats": dict(g.STATS),
    }))


if __name__ == "__main__":
    main()
