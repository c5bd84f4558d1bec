"""Hash groupby-aggregate execution (ref:
/root/reference/src/daft-local-execution/src/sinks/grouped_aggregate.rs and
daft-recordbatch/src/ops/agg.rs).  The hash-table build runs as a HIP kernel
(csrc/groupby.hip) on GPU; per-group reductions are atomic HIP kernels.

Also implements agg-expression decomposition: an agg output expression is an
arbitrary tree over Agg nodes (e.g. ``sum(a*b) / sum(c) + 1``); we extract the
distinct Agg nodes, compute them per group, then evaluate the residual
expression over the per-group columns."""
from __future__ import annotations

import math
from typing import Dict, List, Optional, Sequence, Tuple

import torch

from ..expressions.expressions import (Agg, AggKind, Alias, ColumnRef,
                                       ExprNode)
from ..kernels import rowops
from ..recordbatch import RecordBatch
from ..schema import DataType, Field, Schema, TypeKind
from ..series import Series, full_null


def decompose_agg_exprs(exprs: Sequence[ExprNode]
                        ) -> Tuple[List[Tuple[str, Agg]], List[ExprNode]]:
    """Extract distinct Agg nodes; return (named agg list, residual exprs).

    Residual exprs reference the aggs by generated column names."""
    agg_map: Dict[str, Tuple[str, Agg]] = {}
    residuals: List[ExprNode] = []

    def rewrite(e: ExprNode) -> ExprNode:
        if isinstance(e, Agg):
            key = repr(e)
            if key not in agg_map:
                agg_map[key] = (f"__agg_{len(agg_map)}", e)
            return ColumnRef(agg_map[key][0])
        ch = e.children()
        if not ch:
            return e
        return e.with_children([rewrite(c) for c in ch])

    for e in exprs:
        out_name = e.out_name()
        r = rewrite(e)
        if isinstance(r, ColumnRef) and r.name.startswith("__agg_"):
            r = Alias(r, out_name)
        residuals.append(r)
    return list(agg_map.values()), residuals


def compute_agg(batch: RecordBatch, group_ids: Optional[torch.Tensor],
                num_groups: int, name: str, agg: Agg,
                mask: Optional[torch.Tensor] = None) -> Series:
    """Compute one aggregation over groups; group_ids None = single group.
    `mask` (bool[n]) restricts the aggregation to selected rows — the
    filter-into-aggregate fusion path (no compaction materialized)."""
    n = len(batch)
    dev = batch.device
    if group_ids is None:
        group_ids = torch.zeros(n, dtype=torch.int64, device=dev)
    kind = agg.kind

    if kind == AggKind.COUNT_ALL or (kind == AggKind.COUNT and
                                     agg.child is None):
        ones = _ones_series(n, dev)
        if mask is not None:
            ones = ones.with_validity(mask)
            data, _ = rowops.grouped_agg(group_ids, num_groups, ones,
                                         "count_valid")
        else:
            data, _ = rowops.grouped_agg(group_ids, num_groups, ones,
                                         "count")
        return Series(name, DataType.uint64(), data=data.view(torch.uint64))

    values = agg.child.evaluate(batch)
    if len(values) == 1 and n > 1:
        values = values.broadcast(n)
    if mask is not None:
        v = mask if values.validity is None else (values.validity & mask)
        values = values.with_validity(v)
    if values.dtype.is_decimal() and kind in (
            AggKind.MEAN, AggKind.STDDEV, AggKind.VARIANCE, AggKind.SKEW,
            AggKind.APPROX_PERCENTILE, AggKind.SKETCH):
        # moment-style aggs run in f64; SUM/MIN/MAX stay exact scaled-int
        values = values.cast(DataType.float64())

    if kind == AggKind.COUNT:
        data, _ = rowops.grouped_agg(group_ids, num_groups, values,
                                     "count_valid")
        return Series(name, DataType.uint64(), data=data.view(torch.uint64))

    if kind in (AggKind.SUM, AggKind.MIN, AggKind.MAX):
        if values.dtype.kind in (TypeKind.STRING, TypeKind.BINARY) and \
                kind in (AggKind.MIN, AggKind.MAX):
            return _string_minmax(group_ids, num_groups, values, kind, name)
        if values.dtype.is_decimal() and values.children:
            return _wide_decimal_agg(group_ids, num_groups, values, kind,
                                     name,
                                     agg.to_field(batch.schema).dtype)
        data, cnt = rowops.grouped_agg(group_ids, num_groups, values,
                                       kind)
        out_dt = agg.to_field(batch.schema).dtype
        validity = cnt > 0 if cnt is not None else None
        if kind == AggKind.SUM and values.dtype.is_integer():
            out = data.to(torch.int64)
            return Series(name, out_dt, data=out.view(out_dt.to_torch())
                          if out_dt.to_torch() != torch.int64 else out,
                          validity=validity)
        return Series(name, out_dt, data=data.to(out_dt.to_torch()),
                      validity=validity)

    if kind == AggKind.MEAN:
        s, cnt = rowops.grouped_agg(group_ids, num_groups, values, "sum")
        out = s.to(torch.float64) / cnt.clamp(min=1).to(torch.float64)
        return Series(name, DataType.float64(), data=out, validity=cnt > 0)

    if kind in (AggKind.STDDEV, AggKind.VARIANCE):
        s, cnt = rowops.grouped_agg(group_ids, num_groups, values, "sum")
        s2, _ = rowops.grouped_agg(group_ids, num_groups, values, "sum_sq")
        c = cnt.clamp(min=1).to(torch.float64)
        mean = s.to(torch.float64) / c
        var = (s2.to(torch.float64) / c - mean * mean).clamp(min=0.0)
        out = torch.sqrt(var) if kind == AggKind.STDDEV else var
        return Series(name, DataType.float64(), data=out, validity=cnt > 0)

    if kind == AggKind.SKEW:
        s, cnt = rowops.grouped_agg(group_ids, num_groups, values, "sum")
        s2, _ = rowops.grouped_agg(group_ids, num_groups, values, "sum_sq")
        v3 = Series(values.name, DataType.float64(),
                    data=values.data.to(torch.float64) ** 3,
                    validity=values.validity)
        s3, _ = rowops.grouped_agg(group_ids, num_groups, v3, "sum")
        c = cnt.clamp(min=1).to(torch.float64)
        m = s.to(torch.float64) / c
        m2 = s2.to(torch.float64) / c - m * m
        m3 = s3.to(torch.float64) / c - 3 * m * m2 - m ** 3
        out = m3 / torch.pow(m2.clamp(min=1e-300), 1.5)
        return Series(name, DataType.float64(), data=out, validity=cnt > 0)

    if kind == AggKind.ANY_VALUE:
        first_idx = _first_valid_index(group_ids, num_groups, values)
        return values.take(first_idx).rename(name)

    if kind == AggKind.APPROX_COUNT_DISTINCT and values.is_gpu():
        # HyperLogLog: 2^14 dense registers per group, atomicMax HIP kernel
        # (ref: hyperloglog/src/lib.rs:19-48 + daft-core hll_sketch/merge)
        from ..kernels import native_required
        h = rowops.hash_columns([values])
        vmask = values.validity if values.validity is not None else \
            torch.ones(len(values), dtype=torch.bool, device=dev)
        regs = native_required().hll_update(h, group_ids, vmask, num_groups)
        regs = regs.view(num_groups, 16384)
        m = 16384.0
        alpha = 0.7213 / (1.0 + 1.079 / m)
        rf = regs.to(torch.float64)
        est = alpha * m * m / torch.pow(2.0, -rf).sum(dim=1)
        zeros = (regs == 0).sum(dim=1).to(torch.float64)
        small = (est < 2.5 * m) & (zeros > 0)
        lin = m * torch.log(m / zeros.clamp(min=1.0))
        out = torch.where(small, lin, est).round().to(torch.int64)
        return Series(name, DataType.uint64(), data=out.view(torch.uint64))

    if kind == AggKind.COUNT_DISTINCT and values.is_gpu():
        # one-pass HIP kernel: table insert of (group, value); CAS winners
        # bump the group counter (ref capability: dedup + count pattern)
        from ..kernels import _descs, native_required
        gseries = Series("__gid", DataType.int64(), data=group_ids)
        h = rowops.hash_columns([gseries, values])
        tags, datas, offs, vals = _descs([gseries, values])
        cnt = native_required().count_distinct_pairs(
            h, tags, datas, offs, vals, group_ids, num_groups)
        return Series(name, DataType.uint64(), data=cnt.view(torch.uint64))

    if kind in (AggKind.COUNT_DISTINCT, AggKind.APPROX_COUNT_DISTINCT):
        # exact two-level groupby: distinct (group, value) pairs, then count
        gseries = Series("__gid", DataType.int64(), data=group_ids)
        sub_gids, sub_reps = rowops.groupby([gseries, values])
        outer = group_ids[sub_reps]
        valid = values.validity[sub_reps] if values.validity is not None \
            else None
        ones = torch.ones(len(sub_reps), dtype=torch.int64,
                          device=group_ids.device)
        if valid is not None:
            ones = ones * valid.to(torch.int64)
        out = torch.zeros(num_groups, dtype=torch.int64,
                          device=group_ids.device)
        out.scatter_add_(0, outer, ones)
        return Series(name, DataType.uint64(), data=out.view(torch.uint64))

    if kind in (AggKind.BOOL_AND, AggKind.BOOL_OR):
        iv = Series(values.name, DataType.int64(),
                    data=values.data.to(torch.int64),
                    validity=values.validity)
        op = "min" if kind == AggKind.BOOL_AND else "max"
        data, cnt = rowops.grouped_agg(group_ids, num_groups, iv, op)
        return Series(name, DataType.bool(), data=data != 0,
                      validity=cnt > 0)

    if kind in (AggKind.LIST, AggKind.CONCAT):
        # sort rows by group id (stable) then slice by counts
        perm = rowops._stable_sort_perm_by(group_ids)
        sorted_vals = values.take(perm, has_neg=False)
        counts = torch.bincount(group_ids, minlength=num_groups)
        offs = torch.zeros(num_groups + 1, dtype=torch.int64,
                           device=group_ids.device)
        torch.cumsum(counts, 0, out=offs[1:])
        if kind == AggKind.CONCAT:
            assert values.dtype.is_list(), "agg_concat requires list input"
            # flatten one level: gather child ranges
            inner = sorted_vals
            child = inner.children[0]
            lens = inner.offsets[1:] - inner.offsets[:-1]
            gcounts = torch.zeros(num_groups, dtype=torch.int64,
                                  device=group_ids.device)
            gcounts.scatter_add_(0, group_ids[perm], lens)
            goffs = torch.zeros(num_groups + 1, dtype=torch.int64,
                                device=group_ids.device)
            torch.cumsum(gcounts, 0, out=goffs[1:])
            return Series(name, values.dtype, offsets=goffs,
                          children=[child])
        return Series(name, DataType.list(values.dtype), offsets=offs,
                      children=[sorted_vals.rename("item")])

    if kind == AggKind.SKETCH:
        from . import sketch
        return sketch.grouped_sketch(values.cast(DataType.float64()),
                                     group_ids, num_groups, name)

    if kind == AggKind.SKETCH_FINAL:
        from . import sketch
        return sketch.grouped_sketch_final(values, group_ids, num_groups,
                                           float(agg.param), name)

    if kind == AggKind.PY_UDAF:
        # user-defined aggregation (daft_amd.udaf): sort rows by group,
        # call aggregate() per group slice, then finalize() (ref:
        # /root/reference/daft/udf/udaf.py:16-105 aggregate/combine/
        # finalize pipeline; combine() is exercised when partial states
        # merge across shards)
        inst, ret_dt = agg.param[0], agg.param[1]
        perm = torch.argsort(group_ids, stable=True)
        svals = values.take(perm, has_neg=False).cpu()
        counts = torch.bincount(group_ids, minlength=num_groups)
        offs = torch.zeros(num_groups + 1, dtype=torch.int64)
        torch.cumsum(counts.cpu(), 0, out=offs[1:])
        outs = []
        for g in range(num_groups):
            lo, hi = int(offs[g]), int(offs[g + 1])
            state = inst.aggregate(svals.slice(lo, hi))
            outs.append(inst.finalize(state))
        return Series.from_pylist(name, outs, ret_dt)

    if kind == AggKind.APPROX_PERCENTILE:
        # per-group exact percentile via sort (single-node; the
        # distributed planner splits into SKETCH/SKETCH_FINAL — a
        # DDSketch build + mergeable bucket exchange, physical/sketch.py)
        q = float(agg.param)
        perm = rowops.argsort_multi(
            [Series("g", DataType.int64(), data=group_ids), values],
            [False, False], [False, False])
        svals = values.take(perm, has_neg=False).cast(DataType.float64())
        sg = group_ids[perm]
        counts = torch.bincount(group_ids, minlength=num_groups)
        offs = torch.zeros(num_groups + 1, dtype=torch.int64,
                           device=group_ids.device)
        torch.cumsum(counts, 0, out=offs[1:])
        pos = offs[:-1] + ((counts - 1).to(torch.float64) * q).to(torch.int64)
        pos = torch.minimum(pos, (offs[1:] - 1).clamp(min=0))
        out = svals.take(pos, has_neg=False)
        return out.rename(name).with_validity(counts > 0)

    raise ValueError(f"unsupported aggregation {kind}")


def _ones_series(n: int, dev) -> Series:
    return Series("ones", DataType.int64(),
                  data=torch.ones(n, dtype=torch.int64, device=dev))


def _first_valid_index(group_ids: torch.Tensor, num_groups: int,
                       values: Series) -> torch.Tensor:
    dev = group_ids.device
    n = len(values)
    idx = torch.arange(n, dtype=torch.int64, device=dev)
    if values.validity is not None:
        idx = torch.where(values.validity, idx,
                          torch.full_like(idx, n))
    out = torch.full((num_groups,), n, dtype=torch.int64, device=dev)
    out.scatter_reduce_(0, group_ids, idx, reduce="amin")
    return torch.where(out == n, torch.full_like(out, -1), out)


def _wide_decimal_agg(group_ids: torch.Tensor, num_groups: int,
                      values: Series, kind: "AggKind", name: str,
                      out_dt: DataType) -> Series:
    """Exact SUM/MIN/MAX for wide (p>18) decimals on two int64 limbs.
    SUM accumulates 32-bit halves (kernels/decimal128.sum128); MIN/MAX
    run two passes: extreme of the signed hi limb, then extreme of the
    unsigned lo limb among rows tying on hi."""
    from ..kernels import decimal128 as d128
    lo, hi = d128.limbs(values)
    validity = values.validity
    dev = values.device
    cnt_t = torch.zeros(num_groups, dtype=torch.int64, device=dev)
    ones = torch.ones_like(lo)
    if validity is not None:
        ones = ones * validity.to(torch.int64)
    cnt_t.scatter_add_(0, group_ids, ones)
    out_valid = cnt_t > 0

    if kind == AggKind.SUM:
        slo, shi = lo, hi
        if validity is not None:
            z = torch.zeros_like(lo)
            slo = torch.where(validity, lo, z)
            shi = torch.where(validity, hi, z)

        def seg(v: torch.Tensor) -> torch.Tensor:
            out = torch.zeros(num_groups, dtype=torch.int64, device=dev)
            out.scatter_add_(0, group_ids, v)
            return out

        olo, ohi = d128.sum128(slo, shi, seg)
    else:
        op = "min" if kind == AggKind.MIN else "max"
        hi_ext, _ = rowops.grouped_agg(
            group_ids, num_groups,
            Series("h", DataType.int64(), data=hi, validity=validity), op)
        hi_ext = hi_ext.to(torch.int64)
        cand = hi == hi_ext[group_ids]
        if validity is not None:
            cand = cand & validity
        lo_u = d128.u_order_key(lo)
        lo_ext, _ = rowops.grouped_agg(
            group_ids, num_groups,
            Series("l", DataType.int64(), data=lo_u, validity=cand), op)
        olo = d128.u_order_key(lo_ext.to(torch.int64))   # undo the flip
        ohi = hi_ext
    res = d128.make(name, values.dtype, olo, ohi,
                    None if bool(out_valid.all().item()) else out_valid)
    return res if res.dtype == out_dt else res.cast(out_dt).rename(name)


def _string_minmax(group_ids, num_groups, values: Series, kind, name):
    # argsort once, take first per group (min) or last (max)
    desc = kind == AggKind.MAX
    perm = rowops.argsort_multi([values], [desc], [False])
    # rank of each row in sorted order
    n = len(values)
    dev = values.device
    rank = torch.empty(n, dtype=torch.int64, device=dev)
    rank[perm] = torch.arange(n, dtype=torch.int64, device=dev)
    if values.validity is not None:
        rank = torch.where(values.validity, rank, torch.full_like(rank, n))
    best = torch.full((num_groups,), n, dtype=torch.int64, device=dev)
    best.scatter_reduce_(0, group_ids, rank, reduce="amin")
    inv = torch.full((n + 1,), -1, dtype=torch.int64, device=dev)
    inv[:n] = perm
    sel = inv[best.clamp(max=n)]
    sel = torch.where(best == n, torch.full_like(sel, -1), sel)
    return values.take(sel).rename(name)


_MULTI_AGG_KINDS = {AggKind.SUM, AggKind.MIN, AggKind.MAX, AggKind.MEAN,
                    AggKind.COUNT, AggKind.COUNT_ALL}


def _gids_sorted(gids: torch.Tensor) -> bool:
    if gids.numel() < 2:
        return True
    return bool((gids[1:] >= gids[:-1]).all().item())


def _segmented_multi_agg(gids, num_groups, datas, valids, ops, n, dev):
    """Sorted-gids multi-aggregate via torch.segment_reduce.  gids are
    dense AND sorted, so segment k == group k and boundaries come from
    one vectorized searchsorted.  Output layout matches
    grouped_multi_agg_big: (flat out[n_aggs*G] f64, cnt[n_aggs*G] i64)."""
    G = num_groups
    starts = torch.searchsorted(gids, torch.arange(
        G, dtype=gids.dtype, device=dev))
    bounds = torch.cat([starts, torch.tensor([n], dtype=starts.dtype,
                                             device=dev)])
    lengths = torch.diff(bounds)
    outs = []
    cnts = []
    for d, v, op in zip(datas, valids, ops):
        if v is None:
            cnt = lengths.to(torch.int64)
        else:
            cnt = torch.segment_reduce(v.to(torch.float64), "sum",
                                       lengths=lengths).to(torch.int64)
        if op == 3:       # count only
            outs.append(torch.zeros(G, dtype=torch.float64, device=dev))
            cnts.append(cnt)
            continue
        if op == 0:       # sum
            dd = d if v is None else torch.where(
                v, d, torch.zeros_like(d))
            red = "sum"
        elif op == 1:     # min
            dd = d if v is None else torch.where(
                v, d, torch.full_like(d, float("inf")))
            red = "min"
        else:             # max
            dd = d if v is None else torch.where(
                v, d, torch.full_like(d, float("-inf")))
            red = "max"
        outs.append(torch.segment_reduce(dd, red, lengths=lengths,
                                         initial=0 if red == "sum" else
                                         (float("inf") if red == "min"
                                          else float("-inf"))))
        cnts.append(cnt)
    return torch.cat(outs), torch.cat(cnts)


def _multi_agg(batch, gids, num_groups, named_aggs, mask):
    """Fused one-pass aggregation (csrc grouped_multi_agg): every
    sum/min/max/mean/count computed from a single read of (gids, values)
    with all accumulators in LDS.  Returns None when the shape doesn't
    fit (few aggs, non-float values, too many slots) — per-agg kernels
    handle the rest."""
    from ..kernels import load_native
    n = len(batch)
    dev = batch.device
    if dev.type != "cuda" or len(named_aggs) < 2 or n == 0:
        return None
    big = num_groups * len(named_aggs) > 2048
    # memory guard on the big variant: n_aggs * num_groups * 16B
    if big and num_groups * len(named_aggs) > (1 << 31):
        return None
    if any(a.kind not in _MULTI_AGG_KINDS for _, a in named_aggs):
        return None
    nat = load_native()
    if nat is None:
        return None
    datas, valids, ops, meta = [], [], [], []
    # evaluate all distinct agg inputs in ONE fused kernel (q1: the five
    # value expressions share column reads and launch once)
    value_cache: dict = {}
    uniq = []
    for _cname, a in named_aggs:
        if a.child is not None and a.kind != AggKind.COUNT_ALL:
            key = repr(a.child)
            if key not in value_cache:
                value_cache[key] = None
                uniq.append((key, a.child))
    if uniq:
        from .cse import evaluate_with_cse
        for (key, _e), s in zip(uniq, evaluate_with_cse(
                [e for _k, e in uniq], batch)):
            value_cache[key] = s
    for cname, a in named_aggs:
        if a.kind == AggKind.COUNT_ALL or (a.kind == AggKind.COUNT and
                                           a.child is None):
            datas.append(torch.empty(0, dtype=torch.float64, device=dev))
            valids.append(mask)
            ops.append(3)
            meta.append((cname, a, None))
            continue
        key = repr(a.child)
        values = value_cache.get(key)
        if values is None:
            values = a.child.evaluate(batch)
            if len(values) == 1 and n > 1:
                values = values.broadcast(n)
            value_cache[key] = values
        if values.dtype.kind not in (TypeKind.FLOAT32, TypeKind.FLOAT64):
            return None
        d = values.data.to(torch.float64)
        v = values.validity
        if mask is not None:
            v = mask if v is None else (v & mask)
        if a.kind == AggKind.COUNT:
            datas.append(torch.empty(0, dtype=torch.float64, device=dev))
            valids.append(v)
            ops.append(3)
        else:
            datas.append(d.contiguous())
            valids.append(v.contiguous() if v is not None else None)
            ops.append({AggKind.SUM: 0, AggKind.MIN: 1, AggKind.MAX: 2,
                        AggKind.MEAN: 0}[a.kind])
        meta.append((cname, a, values))
    if big and num_groups > 1 and _gids_sorted(gids):
        # clustered keys (lineitem is orderkey-ordered: the q21/q18
        # per-order aggregations): segmented reduction instead of 1.5B
        # scattered global atomics
        out, cnt = _segmented_multi_agg(gids, num_groups, datas, valids,
                                        ops, n, dev)
    else:
        fn = nat.grouped_multi_agg_big if big else nat.grouped_multi_agg
        out, cnt = fn(gids, num_groups, datas, valids, ops)
    out = out.view(len(named_aggs), num_groups)
    cnt = cnt.view(len(named_aggs), num_groups)
    cols = []
    for i, (cname, a, values) in enumerate(meta):
        k = a.kind
        if k in (AggKind.COUNT, AggKind.COUNT_ALL):
            cols.append(Series(cname, DataType.uint64(),
                               data=cnt[i].contiguous()
                               .view(torch.uint64)))
            continue
        c = cnt[i]
        validity = c > 0
        if bool(validity.all()):
            validity = None
        if k == AggKind.MEAN:
            data = out[i] / c.clamp(min=1).to(torch.float64)
            cols.append(Series(cname, DataType.float64(),
                               data=data.contiguous(), validity=validity))
            continue
        out_dt = a.to_field(batch.schema).dtype
        data = out[i].contiguous().to(out_dt.to_torch())
        cols.append(Series(cname, out_dt, data=data, validity=validity))
    return cols


def run_aggregate(batch: RecordBatch, groupby: List[ExprNode],
                  aggs: List[ExprNode],
                  mask: Optional[torch.Tensor] = None) -> RecordBatch:
    """One-shot (grouped or global) aggregation of a materialized batch.
    With `mask`, rows where mask is False are excluded (fused filter:
    the predicate never materializes a compacted copy of the input)."""
    named_aggs, residuals = decompose_agg_exprs(aggs)
    n = len(batch)
    dev = batch.device

    if groupby:
        key_series = [e.evaluate(batch) for e in groupby]
        key_series = [s.broadcast(n) if len(s) == 1 else s for s in key_series]
        gids, reps = rowops.groupby(key_series)
        num_groups = int(reps.shape[0])
        key_cols = [s.take(reps, has_neg=False) for s in key_series]
    else:
        gids = torch.zeros(n, dtype=torch.int64, device=dev)
        num_groups = 1
        key_cols = []

    agg_cols = _multi_agg(batch, gids, num_groups, named_aggs, mask)
    if agg_cols is None:
        agg_cols = [compute_agg(batch, gids, num_groups, cname, a,
                                mask=mask)
                    for cname, a in named_aggs]
    inter = RecordBatch(key_cols + agg_cols,
                        num_rows=num_groups)
    out_cols = list(key_cols)
    for r in residuals:
        out_cols.append(r.evaluate(inter))
    out = RecordBatch(out_cols, num_rows=num_groups)
    if mask is not None and groupby:
        # drop groups whose rows were all masked out.  NOT a torch
        # scatter_add: 600M atomic adds onto a handful of group slots
        # serialize on L2 (measured ~100 ms on SF100 Q1); the LDS
        # grouped-count kernel does this in one memory-bound pass.
        ones = Series("__m", DataType.bool(),
                      data=torch.ones(n, dtype=torch.bool, device=dev),
                      validity=mask)
        hit, _ = rowops.grouped_agg(gids, num_groups, ones, "count_valid")
        keep = torch.nonzero(hit > 0).reshape(-1)
        if int(keep.numel()) != num_groups:
            out = out.take(keep, has_neg=False)
    return out
