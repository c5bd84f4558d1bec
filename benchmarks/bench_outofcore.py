"""Out-of-core benchmark: TPC-H tables resident in HOST memory, streamed
through HBM morsel-by-morsel (InMemorySource slicing + streamed partial
aggregation + per-morsel join probes).  This is the larger-than-HBM
execution mode: device memory use is bounded by the morsel size, not the
table size.

  python benchmarks/bench_outofcore.py --sf 100 --queries 1,6
"""
from __future__ import annotations

import argparse


# per-query column whitelists for --prune-columns (host RAM sizing for
# SF1000; the engine's source-level projection pushdown does the same
# pruning when reading full tables)
_QCOLS = {
    1: {"lineitem": ["l_quantity", "l_extendedprice", "l_discount",
                     "l_tax", "l_returnflag", "l_linestatus",
                     "l_shipdate"]},
    6: {"lineitem": ["l_shipdate", "l_discount", "l_quantity",
                     "l_extendedprice"]},
    9: {"lineitem": ["l_partkey", "l_suppkey", "l_orderkey", "l_quantity",
                     "l_extendedprice", "l_discount"],
        "part": ["p_partkey", "p_name"],
        "supplier": ["s_suppkey", "s_nationkey"],
        "partsupp": ["ps_partkey", "ps_suppkey", "ps_supplycost"],
        "orders": ["o_orderkey", "o_orderdate"],
        "nation": ["n_nationkey", "n_name"]},
}


def _columns_for(qs):
    out = {}
    for q in qs:
        for t, cols in _QCOLS.get(q, {}).items():
            out.setdefault(t, set()).update(cols)
    return {t: sorted(c) for t, c in out.items()} or None
import json
import time

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--sf", type=float, default=100.0)
    ap.add_argument("--queries", type=str, default="1,6")  # "all" = 1..22
    ap.add_argument("--morsel", type=int, default=1 << 26)
    ap.add_argument("--prune-columns", action="store_true",
                    help="stage only the columns the selected queries "
                    "read (fits SF1000 lineitem in host RAM)")
    args = ap.parse_args()

    from benchmarks.tpch import datagen
    from benchmarks.tpch.queries import run_query
    from daft_amd.context import get_context

    get_context().execution_config.stream_morsel_rows = args.morsel
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"

    t0 = time.time()
    qs_probe = list(range(1, 23)) if args.queries == "all" else \
        [int(q) for q in args.queries.split(",")]
    columns = None
    if args.prune_columns:
        columns = _columns_for(qs_probe)
    if torch.cuda.is_available():
        # GPU-sharded generation staged to host: the SF1000 path
        tables = datagen.dataframes_host_staged(args.sf, columns=columns)
    else:
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
