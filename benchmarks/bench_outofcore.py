"""Out-of-core benchmark: TPC-H tables resident in HOST memory, streamed
through HBM morsel-by-morsel (InMemorySource slicing + streamed partial
aggregation + per-morsel join probes).  This is the larger-than-HBM
execution mode: device memory use is bounded by the morsel size, not the
table size.

  python benchmarks/bench_outofcore.py --sf 100 --queries 1,6
"""
from __future__ import annotations

import argparse
import json
import time

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--sf", type=float, default=100.0)
    ap.add_argument("--queries", type=str, default="1,6")  # "all" = 1..22
    ap.add_argument("--morsel", type=int, default=1 << 26)
    args = ap.parse_args()

    from benchmarks.tpch import datagen
    from benchmarks.tpch.queries import run_query
    from daft_amd.context import get_context

    get_context().execution_config.stream_morsel_rows = args.morsel
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"

    t0 = time.time()
    tables = datagen.dataframes(args.sf, device="cpu")  # host RAM
    lineitem_rows = tables["lineitem"].count_rows()
    print(f"datagen sf={args.sf} on host: {time.time()-t0:.1f}s, "
          f"lineitem={lineitem_rows:,} rows")

    qs = list(range(1, 23)) if args.queries == "all" else \
        [int(q) for q in args.queries.split(",")]
    results = {}
    for q in qs:
        # warmup not meaningful: each run re-streams from host
        t0 = time.time()
        run_query(q, tables, args.sf).collect()
        if dev != "cpu":
            torch.cuda.synchronize()
        dt = time.time() - t0
        results[f"q{q}"] = round(dt, 3)
        print(f"q{q}: {dt:.3f}s  ({lineitem_rows/dt/1e9:.2f} B rows/s "
              f"through host->HBM)")
    print(json.dumps({
        "metric": "tpch_outofcore_host_streamed_s", "sf": args.sf,
        "morsel_rows": args.morsel, "per_query_s": results,
        "device": dev}))


if __name__ == "__main__":
    import os
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    main()
