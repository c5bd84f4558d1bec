"""Window function execution (ref:
/root/reference/src/daft-local-execution/src/streaming_sink/
window_partition_*.rs and daft-recordbatch/src/ops/window_states/).

Supported: per-partition aggregates (sum/min/max/mean/count/...), running
aggregates under an order-by, row_number / rank / dense_rank, lag / lead.
"""
from __future__ import annotations

from typing import List, Optional

import torch

from ..expressions.expressions import Agg, AggKind, ExprNode
from ..kernels import rowops
from ..recordbatch import RecordBatch
from ..schema import DataType, Field, Schema
from ..series import Series
from . import agg as agg_mod


class WindowFn(ExprNode):
    """A window function bound to a Window spec.  kind: 'agg' wraps an Agg
    node; 'row_number'/'rank'/'dense_rank' are rank functions; 'lag'/'lead'
    shift within partitions."""

    def __init__(self, kind: str, inner: Optional[ExprNode], spec,
                 offset: int = 1, default=None):
        self.kind = kind
        self.inner = inner
        self.spec = spec
        self.offset = offset
        self.default = default

    def children(self):
        return [self.inner] if self.inner is not None else []

    def with_children(self, ch):
        return WindowFn(self.kind, ch[0] if ch else None, self.spec,
                        self.offset, self.default)

    def out_name(self):
        if self.inner is not None:
            return self.inner.out_name()
        return self.kind

    def to_field(self, schema):
        return window_out_field(self, self.out_name(), schema)

    def evaluate(self, batch):
        raise RuntimeError("window expressions must run under a Window node")

    def __repr__(self):
        return f"{self.kind}({self.inner!r}).over(...)"


def window_out_field(e: ExprNode, name: str, schema: Schema) -> Field:
    if isinstance(e, WindowFn):
        if e.kind == "agg":
            f = e.inner.to_field(schema)
            return Field(name, f.dtype)
        if e.kind in ("row_number", "rank", "dense_rank"):
            return Field(name, DataType.uint64())
        if e.kind in ("first_value", "last_value"):
            return Field(name, e.inner.to_field(schema).dtype)
        if e.kind in ("lag", "lead"):
            f = e.inner.to_field(schema)
            return Field(name, f.dtype)
    return Field(name, e.to_field(schema).dtype)


def run_window(batch: RecordBatch, window_exprs: List[ExprNode],
               partition_by: List[ExprNode], order_by: List[ExprNode],
               descending: List[bool], names: List[str]) -> RecordBatch:
    n = len(batch)
    dev = batch.device
    if n == 0:
        fields = batch.schema.fields()
        from ..series import empty_series
        cols = list(batch.columns)
        for e, nm in zip(window_exprs, names):
            f = window_out_field(e, nm, batch.schema)
            cols.append(empty_series(f.name, f.dtype, dev))
        return RecordBatch(cols, num_rows=0)

    if partition_by:
        pkeys = [e.evaluate(batch) for e in partition_by]
        gids, reps = rowops.groupby(pkeys)
        num_groups = int(reps.shape[0])
    else:
        gids = torch.zeros(n, dtype=torch.int64, device=dev)
        num_groups = 1

    sorted_pos = None
    if order_by:
        okeys = [e.evaluate(batch) for e in order_by]
        gseries = Series("__g", DataType.int64(), data=gids)
        perm = rowops.argsort_multi([gseries] + okeys,
                                    [False] + list(descending),
                                    [False] * (1 + len(okeys)))
        # rank of each row within its sorted partition
        sorted_gids = gids[perm]
        part_start = _partition_starts(sorted_gids, num_groups)
        pos_in_sorted = torch.empty(n, dtype=torch.int64, device=dev)
        pos_in_sorted[perm] = torch.arange(n, dtype=torch.int64, device=dev)
        sorted_pos = (perm, pos_in_sorted, part_start)

    out_cols = list(batch.columns)
    for e, nm in zip(window_exprs, names):
        assert isinstance(e, WindowFn)
        if e.kind == "agg":
            a: Agg = e.inner
            frame = getattr(e.spec, "frame", None) if e.spec else None
            if sorted_pos is not None:
                # With an ORDER BY the frame is the explicit ROWS frame or
                # the SQL RANGE default (unbounded preceding .. current row,
                # peers share) — never the whole partition.
                # (ref: window_partition_and_order_by.rs incremental states)
                if a.kind in _FRAMEABLE:
                    out_cols.append(_framed_agg(batch, a, nm, gids,
                                                sorted_pos, order_by, frame))
                    continue
                raise NotImplementedError(
                    f"window aggregate {a.kind!r} with ORDER BY / frame is "
                    f"not supported (supported: sum/count/mean/min/max/"
                    f"stddev/variance)")
            per_group = agg_mod.compute_agg(batch, gids, num_groups, nm, a)
            out_cols.append(per_group.take(gids).rename(nm))
        elif e.kind in ("row_number", "rank", "dense_rank"):
            assert sorted_pos is not None, f"{e.kind} requires order_by"
            perm, pos, part_start = sorted_pos
            rn = pos - part_start[gids]
            if e.kind == "row_number":
                vals = rn + 1
            else:
                okeys = [k.evaluate(batch) for k in order_by]
                same_as_prev = _same_as_prev(okeys, gids, perm, pos)
                vals = _rank_values(rn, same_as_prev, gids, perm,
                                    dense=e.kind == "dense_rank")
            out_cols.append(Series(nm, DataType.uint64(),
                                   data=vals.view(torch.uint64)))
        elif e.kind in ("first_value", "last_value"):
            assert sorted_pos is not None, f"{e.kind} requires order_by"
            perm, pos, part_start = sorted_pos
            vals = e.inner.evaluate(batch)
            if e.kind == "first_value":
                # row at each partition's first sorted position
                src = perm[part_start[gids]]
            else:
                # running last value = the current row itself under the
                # SQL default frame (unbounded preceding .. current row)
                src = torch.arange(n, dtype=torch.int64, device=dev)
            out_cols.append(vals.take(src, has_neg=False).rename(nm))
        elif e.kind in ("lag", "lead"):
            assert sorted_pos is not None, f"{e.kind} requires order_by"
            perm, pos, part_start = sorted_pos
            vals = e.inner.evaluate(batch)
            off = e.offset if e.kind == "lag" else -e.offset
            src_sorted_pos = pos - off
            # stay within partition bounds
            inv = torch.empty(n, dtype=torch.int64, device=dev)
            inv = perm  # inv[sorted_pos] = original row
            src_pos_clamped = src_sorted_pos.clamp(min=0, max=n - 1)
            src_row = inv[src_pos_clamped]
            same_part = gids[src_row] == gids
            in_bounds = (src_sorted_pos >= 0) & (src_sorted_pos < n) & same_part
            src_idx = torch.where(in_bounds, src_row,
                                  torch.full_like(src_row, -1))
            shifted = vals.take(src_idx).rename(nm)
            if e.default is not None:
                # SQL applies the default only when the offset falls outside
                # the partition — genuinely-NULL source values stay NULL.
                fill = Series.from_pylist(nm, [e.default],
                                          device=dev).broadcast(n)
                mask = Series("__m", DataType.bool(), data=in_bounds)
                shifted = mask.if_else(shifted, fill).rename(nm)
            out_cols.append(shifted)
        else:
            raise ValueError(f"unknown window fn {e.kind}")
    return RecordBatch(out_cols, num_rows=n)


_FRAMEABLE = {AggKind.SUM, AggKind.COUNT, AggKind.COUNT_ALL, AggKind.MEAN,
              AggKind.MIN, AggKind.MAX, AggKind.STDDEV, AggKind.VARIANCE}


def _framed_agg(batch: RecordBatch, a: Agg, name: str, gids: torch.Tensor,
                sorted_pos, order_by, frame) -> Series:
    """Windowed aggregate over an explicit ROWS frame, or the SQL
    RANGE-default frame (unbounded preceding .. current row, peers share)
    when `frame` is None.  sum/count/mean/stddev/variance run on per-
    partition prefix sums; min/max on a sparse-table range query
    (ref: window_partition_and_dynamic_frame.rs / window_states/minmax.rs).
    All positions are in sorted space; results scatter back through perm."""
    perm, pos, part_start = sorted_pos
    n = gids.shape[0]
    dev = gids.device
    g_sorted = gids[perm]
    ps = part_start[g_sorted]                     # partition start (sorted)
    counts = torch.bincount(g_sorted, minlength=int(part_start.shape[0]))
    pe = ps + counts[g_sorted] - 1                # partition end (inclusive)
    idx = torch.arange(n, dtype=torch.int64, device=dev)

    if frame is not None:
        from ..window import Window as W

        def bound(spec):
            if spec == W.unbounded_preceding:
                return ps
            if spec == W.unbounded_following:
                return pe
            if spec == W.current_row:
                return idx
            return idx + int(spec)
        start, end = frame
        lo = torch.maximum(bound(start), ps)
        hi = torch.minimum(bound(end), pe)
    else:
        # RANGE default: frame end is the LAST peer (equal order keys)
        okeys = [k.evaluate(batch) for k in order_by]
        sap = _same_as_prev(okeys, gids, perm, pos)[perm]
        next_new = torch.ones(n, dtype=torch.bool, device=dev)
        next_new[:-1] = ~sap[1:]
        run_end = torch.where(next_new, idx, torch.full_like(idx, n))
        run_end = torch.flip(
            torch.cummin(torch.flip(run_end, [0]), 0).values, [0])
        lo, hi = ps, run_end
    empty = lo > hi
    lo_c = lo.clamp(0, max(n - 1, 0))
    hi_c = hi.clamp(0, max(n - 1, 0))

    # values in sorted order
    if a.kind == AggKind.COUNT_ALL or a.child is None:
        vdata = torch.ones(n, dtype=torch.float64, device=dev)
        vvalid = None
    else:
        values = a.child.evaluate(batch)
        if len(values) == 1 and n > 1:
            values = values.broadcast(n)
        if values.is_dict():
            values = values.dict_decode()
        vdata = values.data
        vvalid = values.validity
    valid_sorted = vvalid[perm] if vvalid is not None else None

    def prefix(t):
        return torch.cumsum(t, 0)

    def rsum(pre):
        upper = pre[hi_c]
        lower = torch.where(lo_c > 0, pre[(lo_c - 1).clamp(min=0)],
                            torch.zeros_like(upper))
        out = upper - lower
        return torch.where(empty, torch.zeros_like(out), out)

    ones = torch.ones(n, dtype=torch.float64, device=dev)
    if valid_sorted is not None:
        ones = torch.where(valid_sorted, ones, torch.zeros_like(ones))
    rcnt = rsum(prefix(ones))

    if a.kind in (AggKind.COUNT, AggKind.COUNT_ALL):
        out_sorted = rcnt
        validity_sorted = None           # COUNT of an empty frame is 0
    elif a.kind in (AggKind.MIN, AggKind.MAX):
        if vdata is None or vdata.dtype in (torch.uint8,):
            raise NotImplementedError(
                f"windowed min/max over dtype is not supported")
        out_sorted = _range_minmax(vdata[perm], valid_sorted, lo_c, hi_c,
                                   is_max=a.kind == AggKind.MAX)
        validity_sorted = (rcnt > 0) & ~empty
        out = torch.empty(n, dtype=out_sorted.dtype, device=dev)
        out[perm] = out_sorted
        v = torch.empty(n, dtype=torch.bool, device=dev)
        v[perm] = validity_sorted
        out_dt = a.to_field(batch.schema).dtype
        if out.dtype != out_dt.to_torch():
            out = out.view(out_dt.to_torch()) if                 out.element_size() == out_dt.to_torch().itemsize and                 not out.dtype.is_floating_point else out.to(out_dt.to_torch())
        return Series(name, out_dt, data=out,
                      validity=None if bool(v.all().item()) else v)
    else:
        vs = vdata.to(torch.float64)[perm]
        if valid_sorted is not None:
            vs = torch.where(valid_sorted, vs, torch.zeros_like(vs))
        s = rsum(prefix(vs))
        if a.kind == AggKind.SUM:
            out_sorted = s
            validity_sorted = (rcnt > 0) & ~empty
        elif a.kind == AggKind.MEAN:
            out_sorted = s / rcnt.clamp(min=1.0)
            validity_sorted = (rcnt > 0) & ~empty
        else:  # STDDEV / VARIANCE (population, matching physical/agg.py)
            s2 = rsum(prefix(vs * vs))
            c = rcnt.clamp(min=1.0)
            mean = s / c
            var = (s2 / c - mean * mean).clamp(min=0.0)
            out_sorted = torch.sqrt(var) if a.kind == AggKind.STDDEV else var
            validity_sorted = (rcnt > 0) & ~empty

    out = torch.empty(n, dtype=torch.float64, device=dev)
    out[perm] = out_sorted
    validity = None
    if validity_sorted is not None and not bool(validity_sorted.all().item()):
        v = torch.empty(n, dtype=torch.bool, device=dev)
        v[perm] = validity_sorted
        validity = v
    if a.kind in (AggKind.COUNT, AggKind.COUNT_ALL):
        return Series(name, DataType.uint64(),
                      data=out.to(torch.int64).view(torch.uint64),
                      validity=validity)
    out_dt = a.to_field(batch.schema).dtype
    if a.kind in (AggKind.MEAN, AggKind.STDDEV, AggKind.VARIANCE):
        out_dt = DataType.float64()
    return Series(name, out_dt, data=out.to(out_dt.to_torch()),
                  validity=validity)


def _range_minmax(v_sorted: torch.Tensor, valid_sorted, lo: torch.Tensor,
                  hi: torch.Tensor, is_max: bool) -> torch.Tensor:
    """Range min/max over [lo, hi] (inclusive, sorted space) via an
    O(n log n) sparse table; nulls excluded through sentinels."""
    n = v_sorted.shape[0]
    dt = v_sorted.dtype
    if dt == torch.bool:
        v_sorted = v_sorted.to(torch.int8)
        dt = torch.int8
    unsigned_view = None
    if dt in (torch.uint16, torch.uint32, torch.uint64):
        # compare through the signed view (exact for values < 2^(w-1))
        unsigned_view = dt
        signed = {torch.uint16: torch.int16, torch.uint32: torch.int32,
                  torch.uint64: torch.int64}[dt]
        v_sorted = v_sorted.view(signed)
        dt = signed
    if dt.is_floating_point:
        sent = float("-inf") if is_max else float("inf")
    else:
        ii = torch.iinfo(dt)
        sent = ii.min if is_max else ii.max
    v = v_sorted
    if valid_sorted is not None:
        v = torch.where(valid_sorted, v, torch.full_like(v, sent))
    op = torch.maximum if is_max else torch.minimum
    levels = [v]
    j = 1
    while (1 << j) <= n:
        prev = levels[-1]
        half = 1 << (j - 1)
        m = n - (1 << j) + 1
        cur = torch.full_like(v, sent)
        cur[:m] = op(prev[:m], prev[half:half + m])
        levels.append(cur)
        j += 1
    st = torch.stack(levels)                       # (J, n)
    length = (hi - lo + 1).clamp(min=1)
    jj = torch.floor(torch.log2(length.to(torch.float64))).to(torch.int64)
    jj = jj.clamp(0, len(levels) - 1)
    pow2 = torch.bitwise_left_shift(torch.ones_like(jj), jj)
    res = op(st[jj, lo], st[jj, (hi - pow2 + 1).clamp(min=0)])
    if unsigned_view is not None:
        res = res.view(unsigned_view)
    return res


def _partition_starts(sorted_gids: torch.Tensor,
                      num_groups: int) -> torch.Tensor:
    n = sorted_gids.shape[0]
    dev = sorted_gids.device
    idx = torch.arange(n, dtype=torch.int64, device=dev)
    starts = torch.full((num_groups,), n, dtype=torch.int64, device=dev)
    starts.scatter_reduce_(0, sorted_gids, idx, reduce="amin")
    return starts


def _same_as_prev(okeys: List[Series], gids: torch.Tensor,
                  perm: torch.Tensor, pos: torch.Tensor) -> torch.Tensor:
    """For each row: do its order keys equal the previous row's (sorted order,
    same partition)?  Returned in original row order."""
    n = perm.shape[0]
    dev = perm.device
    prev_sorted = (pos - 1).clamp(min=0)
    prev_row = perm[prev_sorted]
    same = torch.ones(n, dtype=torch.bool, device=dev)
    for k in okeys:
        prev_k = k.take(prev_row)
        eq = k.compare(prev_k, "eq")
        e = eq.data.clone()
        if eq.validity is not None:
            # only null-vs-null counts as a peer: both sides must be null
            cur_null = ~k.validity if k.validity is not None else \
                torch.zeros(n, dtype=torch.bool, device=dev)
            prev_null = ~prev_k.validity if prev_k.validity is not None else \
                torch.zeros(n, dtype=torch.bool, device=dev)
            e = torch.where(eq.validity, e, cur_null & prev_null)
        same &= e
    same &= gids[prev_row] == gids
    same &= pos > 0
    return same


def _rank_values(rn: torch.Tensor, same_as_prev: torch.Tensor,
                 gids: torch.Tensor, perm: torch.Tensor,
                 dense: bool) -> torch.Tensor:
    """rank: 1 + count of strictly-smaller rows; dense_rank: 1 + distinct
    smaller keys.  Computed via a cumulative pass in sorted order (host loop
    avoided: cumsum of "new key" indicator per partition)."""
    n = rn.shape[0]
    dev = rn.device
    sap = same_as_prev[perm]       # sorted-order "same key as previous row"
    newkey = (~sap).to(torch.int64)  # 1 at every key start (incl. partition starts)
    g_sorted = gids[perm]
    starts = _partition_starts(g_sorted, int(g_sorted.max().item()) + 1
                               if n else 1)
    if dense:
        # dense rank = per-partition cumsum of newkey
        csum = torch.cumsum(newkey, 0)
        base = csum[starts[g_sorted]] - 1  # csum at partition start is its 1
        vals_sorted = csum - base
    else:
        # rank = (position of this key's first row within partition) + 1.
        # Global key-start positions are strictly increasing, so a global
        # cummax of (pos+1 at key starts, else 0) never crosses partitions
        # (newkey is always 1 at partition starts).
        pos_sorted = torch.arange(n, dtype=torch.int64, device=dev)
        key_start = torch.where(newkey.bool(), pos_sorted + 1,
                                torch.zeros_like(pos_sorted))
        key_start_pos = torch.cummax(key_start, 0).values - 1
        vals_sorted = key_start_pos - starts[g_sorted] + 1
    out = torch.empty(n, dtype=torch.int64, device=dev)
    out[perm] = vals_sorted
    return out
