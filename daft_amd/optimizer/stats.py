"""Sampled column statistics for join-cost estimation.

NDV (number of distinct values) per source column, estimated from a
strided 64k-row sample of the cached partitions.  |A ⋈ B on k| ≈
|A|·|B| / max(ndv_A(k), ndv_B(k)) — with only row counts, a
many-to-many key (25 distinct nationkeys joining 15M × 1M rows) looks
identical to a PK-FK join and the reorderer can pick a 600-billion-row
intermediate (observed as a 1.3 TB OOM on SF100 q5 in round 2).

SPMD: under a multi-rank run the estimates are computed once per source
by the distributed runner's stats sync (all ranks combine local sample
NDVs), never lazily — rank-local samples differ and lazily-diverging
estimates desynchronize the collective schedule.

(ref: /root/reference/src/daft-stats/src/column_stats/ — the reference
tracks min/max/null stats; NDV here serves its reorder_joins cost model,
rules/reorder_joins/)
"""
from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

# (cache_key, column) -> estimated distinct count (global under SPMD)
NDV_HINTS: Dict[Tuple[str, str], float] = {}

SAMPLE_ROWS = 1 << 16


def sample_ndv(series, n_rows: Optional[int] = None) -> float:
    """Estimate the distinct count of a Series from a strided sample.

    Two regimes dominate: category-like columns (the sample saturates the
    value set: distinct fraction is tiny -> the sampled unique count IS
    the estimate) and key-like columns (sample is almost all-distinct ->
    scale the fraction up to the full length)."""
    n = len(series)
    if n_rows is None:
        n_rows = n
    if n == 0:
        return 0.0
    k = min(n, SAMPLE_ROWS)
    stride = max(1, n // k)
    idx = torch.arange(0, n, stride, dtype=torch.int64,
                       device=series.device)[:k]
    k = int(idx.numel())
    sampled = series.take(idx, has_neg=False)
    h = sampled.hash()
    u = int(torch.unique(h).numel())
    ratio = u / max(k, 1)
    if ratio < 0.1:
        # saturated: the sample has seen (nearly) every value
        return float(u)
    # invert the expected-distinct curve u = N·(1 − e^(−k/N)) for the
    # domain size N (monotone in N: bisection).  Naive ratio-scaling
    # (u/k × n_rows) overestimates key NDVs by 500×: 65k draws from 1M
    # distinct values collide only ~2000 times, and the q9 reorder then
    # priced a 600M-row join at 1M rows (3× regression).
    import math
    if k - u < 3:
        return float(n_rows)      # too few collisions to bound N
    lo, hi = float(u), float(max(n_rows, u + 1))

    def expected_u(N: float) -> float:
        return N * (1.0 - math.exp(-k / N))
    if expected_u(hi) <= u:
        return hi
    for _ in range(60):
        mid = 0.5 * (lo + hi)
        if expected_u(mid) < u:
            lo = mid
        else:
            hi = mid
    return float(min(max(0.5 * (lo + hi), u), n_rows))


def ndv_for_source(cache_key: str, column: str,
                   est_rows: Optional[float]) -> Optional[float]:
    """NDV for a source column; computes lazily on single-rank runs,
    returns only synced hints under SPMD."""
    hint = NDV_HINTS.get((cache_key, column))
    if hint is not None:
        return min(hint, est_rows) if est_rows is not None else hint
    try:
        from ..distributed import comm
        if comm.is_dist() and comm.world() > 1:
            return None     # must have been synced; do not diverge
    except Exception:
        pass
    try:
        from ..context import get_context
        parts = get_context().cache.get(cache_key)
    except Exception:
        return None
    if not parts:
        return None
    total = sum(len(p) for p in parts)
    # sample the largest partition, scale to the total
    big = max(parts, key=len)
    try:
        s = big.column(column)
    except Exception:
        return None
    if s.pyobjs is not None:
        return None
    nd = sample_ndv(s, n_rows=total)
    NDV_HINTS[(cache_key, column)] = nd
    return min(nd, est_rows) if est_rows is not None else nd


# ---------------------------------------------------------------------------
# exact per-source column ranges for TruthValue filter folding (ref:
# daft-stats/src/column_stats/ ColumnRangeStatistics + TruthValue:
# a predicate over a column whose [min, max] is known evaluates to
# definitely-True / definitely-False / Maybe)
# ---------------------------------------------------------------------------

# (cache_key, column) -> (min, max, has_nulls) — EXACT bounds (not
# sampled: pruning on approximate bounds would be a correctness bug)
RANGE_HINTS: Dict[Tuple[str, str], tuple] = {}


def range_for_source(cache_key: str, column: str):
    """Exact (min, max, has_nulls) of a cached source column, or None.
    Lazily computed on single-rank runs only — under SPMD rank-local
    ranges could fold a filter on one rank and not another, desyncing
    the collective schedule (same invariant as ndv_for_source)."""
    hint = RANGE_HINTS.get((cache_key, column))
    if hint is not None:
        return hint
    try:
        from ..distributed import comm
        if comm.is_dist() and comm.world() > 1:
            return None
    except Exception:
        pass
    try:
        from ..context import get_context
        parts = get_context().cache.get(cache_key)
    except Exception:
        return None
    if not parts:
        return None
    lo = hi = None
    has_nulls = False
    for p in parts:
        try:
            s = p.column(column)
        except Exception:
            return None
        if s.pyobjs is not None or s.data is None or s.offsets is not None \
                or s.children or len(s) == 0:
            return None          # numeric/temporal fixed-width only
        if not (s.dtype.is_numeric() or s.dtype.is_temporal() or
                s.dtype.kind.value in ("date", "bool")):
            return None
        d = s.data
        if s.validity is not None:
            if bool((~s.validity).any()):
                has_nulls = True
            valid = s.validity
            if not bool(valid.any()):
                continue
            d = d[valid]
        if d.dtype == torch.uint64:
            d = d.view(torch.int64)
        pmin = d.min().item()
        pmax = d.max().item()
        lo = pmin if lo is None else min(lo, pmin)
        hi = pmax if hi is None else max(hi, pmax)
    if lo is None:
        return None
    out = (lo, hi, has_nulls)
    RANGE_HINTS[(cache_key, column)] = out
    return out
