"""Scan+aggregate bandwidth benchmark (the "GB/s scan+agg" component of
BASELINE.json's metric): Q6-shaped predicate + sum over HBM-resident
lineitem columns, reporting effective scan bandwidth.

  python benchmarks/bench_scan_agg.py --sf 100
"""
from __future__ import annotations

import argparse
import datetime as dt
import json
import time

import torch

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--sf", type=float, default=None)
    ap.add_argument("--iters", type=int, default=5)
    args = ap.parse_args()

    import daft_amd as daft
    from daft_amd import col
    from benchmarks.tpch import datagen

    on_gpu = torch.cuda.is_available()
    device = "cuda:0" if on_gpu else "cpu"
    sf = args.sf if args.sf is not None else (100.0 if on_gpu else 0.05)

    li = datagen.gen_orders_lineitem(sf, device)[1]
    from daft_amd.io import from_recordbatches
    df = from_recordbatches([li])
    n = len(li)

    def q6():
        return df.where(
            (col("l_shipdate") >= dt.date(1994, 1, 1)) &
            (col("l_shipdate") < dt.date(1995, 1, 1)) &
            (col("l_discount") >= 0.05) & (col("l_discount") <= 0.07) &
            (col("l_quantity") < 24)
        ).agg((col("l_extendedprice") * col("l_discount")).sum()
              .alias("revenue")).to_pydict()

    q6()  # warmup
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        out = q6()
    if on_gpu:
        torch.cuda.synchronize()
    dt_s = (time.perf_counter() - t0) / args.iters

    # bytes the query must touch: shipdate(i32) + discount(f64) +
    # quantity(f64) + extendedprice(f64) for every row (selected rows
    # re-read price/discount for the product)
    sel = 0.015  # ~1.5% selectivity of the Q6 predicate
    bytes_scanned = n * (4 + 8 + 8) + int(n * sel) * 16
    gbps = bytes_scanned / dt_s / 1e9
    print(json.dumps({
        "metric": "scan_agg_GBps", "value": round(gbps, 1), "unit": "GB/s",
        "rows": n, "seconds": round(dt_s, 4), "sf": sf, "device": device,
        "query": "tpch_q6_shape", "revenue": out["revenue"][0],
    }))


if __name__ == "__main__":
    main()
