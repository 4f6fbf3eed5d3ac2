"""Aggregate per-rank query-profile JSON files into one summary (reference:
bodo/utils/aggregate_query_profiles/ CLI).

    python -m bodo_amd.utils.aggregate_query_profiles <dir> [-o out.json]
"""

from __future__ import annotations

import argparse
import glob
import json
import os
from collections import defaultdict


def aggregate(directory: str) -> dict:
    files = sorted(glob.glob(os.path.join(directory, "query_profile_rank*.json")))
    per_op = defaultdict(lambda: {"count": 0, "total_s": 0.0, "max_s": 0.0,
                                  "rows_out": 0})
    for f in files:
        data = json.load(open(f))
        for r in data.get("records", []):
            key = f"q{r['query']}/{r['operator']}"
            agg = per_op[key]
            agg["count"] += 1
            agg["total_s"] += r["duration_s"]
            agg["max_s"] = max(agg["max_s"], r["duration_s"])
            if r.get("rows_out", -1) >= 0:
                agg["rows_out"] += r["rows_out"]
    return {"n_ranks": len(files), "operators": dict(per_op)}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("directory")
    p.add_argument("-o", "--output", default=None)
    args = p.parse_args()
    out = aggregate(args.directory)
    text = json.dumps(out, indent=1, sort_keys=True)
    if args.output:
        open(args.output, "w").write(text)
    else:
        print(text)


if __name__ == "__main__":
    main()
