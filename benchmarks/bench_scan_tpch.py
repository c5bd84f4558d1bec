"""Scan-included TPC-H: tables live as PARQUET FILES on local NVMe and
every query's timed region includes the parquet scan -> decode -> H2D ->
compute pipeline (the reference's published TPC-H numbers read parquet
from S3; BASELINE.md:212).  Complements bench.py's in-HBM numbers.

  python benchmarks/bench_scan_tpch.py --sf 30 --queries all
"""
from __future__ import annotations

import argparse
import json
import os
import time

import torch

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


def materialize_parquet(sf: float, root: str, gen_device: str,
                        shards: int = 0) -> None:
    import pyarrow.parquet as pq
    from benchmarks.tpch import datagen
    if shards <= 0:
        shards = max(4, int(sf) // 3)      # multiple files per table
    os.makedirs(root, exist_ok=True)
    for r in range(shards):
        tables = datagen.generate(sf, gen_device, r, shards)
        for name, rb in tables.items():
            d = os.path.join(root, name)
            os.makedirs(d, exist_ok=True)
            tbl = rb.cpu().to_arrow()
            pq.write_table(tbl, os.path.join(d, f"part-{r:04d}.parquet"),
                           compression="snappy",
                           row_group_size=1 << 21)
        if str(gen_device).startswith("cuda"):
            torch.cuda.empty_cache()


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--sf", type=float, default=None)
    ap.add_argument("--queries", type=str, default="all")
    ap.add_argument("--dir", type=str, default="/tmp/tpch_parquet")
    ap.add_argument("--keep", action="store_true")
    args = ap.parse_args()

    on_gpu = torch.cuda.is_available()
    sf = args.sf if args.sf is not None else (30.0 if on_gpu else 0.05)
    dev = "cuda:0" if on_gpu else "cpu"

    import daft_amd as daft
    from benchmarks.tpch.queries import run_query

    root = args.dir
    marker = os.path.join(root, f".sf{sf}")
    if not os.path.exists(marker):
        t0 = time.time()
        materialize_parquet(sf, root, dev)
        open(marker, "w").write("ok")
        print(f"[scan-bench] wrote sf={sf} parquet in {time.time()-t0:.1f}s",
              flush=True)

    tables = {
        name: daft.read_parquet(os.path.join(root, name, "*.parquet"))
        for name in ("nation", "region", "supplier", "part", "partsupp",
                     "customer", "orders", "lineitem")
    }
    qs = list(range(1, 23)) if args.queries == "all" else \
        [int(q) for q in args.queries.split(",")]

    # warmup (page cache + jit) on the cheapest query
    run_query(6, tables, sf=sf).to_pydict()
    if on_gpu:
        torch.cuda.synchronize()
    per_q = {}
    for qi in qs:
        t0 = time.perf_counter()
        out = run_query(qi, tables, sf=sf)
        for part in out._result:
            pass
        if on_gpu:
            torch.cuda.synchronize()
        per_q[qi] = round(time.perf_counter() - t0, 3)
        print(f"[scan-bench] q{qi}: {per_q[qi]:.3f}s", flush=True)
    total = round(sum(per_q.values()), 3)
    print(json.dumps({
        "metric": "tpch_scan_included_total_s", "value": total, "unit": "s",
        "sf": sf, "source": "parquet files on local NVMe (snappy)",
        "per_query_s": {f"q{k}": v for k, v in per_q.items()},
        "device": dev}), flush=True)


if __name__ == "__main__":
    main()
