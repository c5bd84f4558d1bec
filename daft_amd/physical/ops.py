"""Physical operators — the Swordfish-analog operator set (ref:
/root/reference/src/daft-local-execution/src/{sources,intermediate_ops,
streaming_sink,sinks,join}/ and SURVEY.md §2.3), executed as generator
pipelines over device-resident RecordBatches.

Operator taxonomy mirrors the reference:
  SourceOp            -> sources/ (InMemory, ScanTask)
  streaming (map) ops -> intermediate_ops/ (project, filter, explode, ...)
  streaming sinks     -> streaming_sink/ (limit, sample, monotonic id)
  blocking sinks      -> sinks/ (aggregate, sort, topn, join build, write,
                                  distinct, pivot, window)
"""
from __future__ import annotations

import math
from typing import Dict, Iterator, List, Optional, Sequence

import torch

from ..expressions.expressions import (Agg, AggKind, Alias, ColumnRef,
                                       ExprNode)
from ..kernels import rowops
from ..recordbatch import RecordBatch
from ..schema import DataType, Field, Schema, TypeKind
from ..series import Series, full_null
from . import agg as agg_mod

BatchIter = Iterator[RecordBatch]


class ExecContext:
    def __init__(self, ctx, device, query_id: str = ""):
        self.ctx = ctx
        self.device = device
        self.query_id = query_id
        # per-operator runtime stats (ref: daft-local-execution
        # src/runtime_stats/): node id -> [rows_out, batches, seconds]
        self.stats: Dict[int, list] = {}
        self.op_names: Dict[int, str] = {}
        from ..execution.memory import MemoryManager
        self.memory = MemoryManager(ctx)


class PhysicalOp:
    """Base physical operator; children execute lazily as generators."""

    def __init__(self, children: List["PhysicalOp"], schema: Schema,
                 name: str = ""):
        self.children = children
        self.schema = schema
        self.op_name = name or type(self).__name__

    def execute(self, ectx: ExecContext) -> BatchIter:
        raise NotImplementedError(type(self))

    def execute_tracked(self, ectx: ExecContext) -> BatchIter:
        """execute() wrapped with per-operator runtime stats and roctx
        ranges (ref: runtime_stats/ + OTLP spans in common/tracing; GPU
        ranges show up in rocprof timelines via roctx)."""
        import time as _time
        key = id(self)
        st = ectx.stats.setdefault(key, [0, 0, 0.0])
        ectx.op_names[key] = self.op_name
        use_roctx = str(ectx.device).startswith("cuda")
        gen = self.execute(ectx)
        while True:
            t0 = _time.perf_counter()
            if use_roctx:
                import torch as _torch
                _torch.cuda.nvtx.range_push(self.op_name)
            try:
                batch = next(gen)
            except StopIteration:
                st[2] += _time.perf_counter() - t0
                return
            finally:
                if use_roctx:
                    import torch as _torch
                    _torch.cuda.nvtx.range_pop()
            st[0] += len(batch)
            st[1] += 1
            st[2] += _time.perf_counter() - t0
            yield batch

    def _child_iters(self, ectx) -> List[BatchIter]:
        return [c.execute_tracked(ectx) for c in self.children]

    def _materialize_child(self, ectx, i: int = 0) -> RecordBatch:
        batches = list(self.children[i].execute_tracked(ectx))
        if not batches:
            return RecordBatch.empty(self.children[i].schema,
                                     device=ectx.device)
        if len(batches) == 1:
            return batches[0]
        # blocking materialization: admit the concat copy under the HBM
        # watermark (spills cached partition sets if needed)
        ectx.memory.admit(sum(b.size_bytes() for b in batches), ectx.device)
        return RecordBatch.concat(batches)

    def explain_lines(self, indent=0) -> List[str]:
        lines = ["  " * indent + f"* {self.op_name}"]
        for c in self.children:
            lines.extend(c.explain_lines(indent + 1))
        return lines


def stream_host_batch(part, device, morsel: int):
    """Out-of-core H2D: slice a host partition into morsels, stage each in
    pinned memory, and DMA it on a dedicated copy stream that overlaps the
    previous morsel's compute (double buffering).  The compute stream
    waits on a per-morsel event, never on the host.  (ref concurrency
    shape: daft-local-execution channel.rs bounded pipelining; H2D overlap
    is the MI355X-native replacement for its disk prefetch.)"""
    import torch as _t
    if not str(device).startswith("cuda") or not _t.cuda.is_available():
        for lo in range(0, max(len(part), 1), morsel):
            yield part.slice(lo, lo + morsel).to(device)
        return
    copy_stream = _stream_pool(device)
    compute = _t.cuda.current_stream(device)
    staged = None        # (device_batch, event, pinned_keepalive)

    already_pinned = part.is_pinned()

    def stage(lo: int):
        sl = part.slice(lo, lo + morsel)
        # slices of pinned storage stay pinned: skip the staging copy
        pin = sl if already_pinned else sl.pinned()
        with _t.cuda.stream(copy_stream):
            dev = pin.to(device, non_blocking=True)
            ev = _t.cuda.Event()
            ev.record(copy_stream)
        return dev, ev, pin

    n = len(part)
    offs = list(range(0, max(n, 1), morsel))
    for i, lo in enumerate(offs):
        nxt = stage(lo)
        if staged is not None:
            dev, ev, _pin = staged
            compute.wait_event(ev)
            yield dev
        staged = nxt
    if staged is not None:
        dev, ev, _pin = staged
        compute.wait_event(ev)
        yield dev


_COPY_STREAMS: dict = {}


def _stream_pool(device):
    import torch as _t
    key = str(device)
    s = _COPY_STREAMS.get(key)
    if s is None:
        s = _t.cuda.Stream(device)
        _COPY_STREAMS[key] = s
    return s


class InMemorySourceOp(PhysicalOp):
    def __init__(self, schema: Schema, cache_key: str, columns=None):
        super().__init__([], schema, "InMemorySource")
        self.cache_key = cache_key
        self.columns = columns

    def execute(self, ectx) -> BatchIter:
        morsel = getattr(ectx.ctx.execution_config, "stream_morsel_rows",
                         1 << 26)
        for part in ectx.ctx.cache.get(self.cache_key):
            if self.columns is not None and \
                    len(self.columns) < len(part.columns):
                part = part.select_columns(self.columns)
            if part.device != ectx.device:
                if str(part.device) == "cpu" and \
                        str(ectx.device).startswith("cuda") and \
                        len(part) > morsel:
                    # out-of-core: host partition larger than a morsel —
                    # stream pinned slices through HBM on the copy stream
                    # (overlapped with the previous morsel's compute)
                    yield from stream_host_batch(part, ectx.device, morsel)
                    continue
                part = part.to(ectx.device)
            yield part


class ScanOp(PhysicalOp):
    """File scan: host decode (pyarrow) -> H2D transfer -> device batches
    (ref: sources/scan_task.rs; GPU page decode is a later-round upgrade)."""

    def __init__(self, schema: Schema, paths: List[str], file_format: str,
                 storage_options: dict, read_options: dict,
                 columns: Optional[List[str]],
                 predicate: Optional[ExprNode], limit: Optional[int]):
        super().__init__([], schema, f"Scan({file_format})")
        self.paths = paths
        self.file_format = file_format
        self.storage_options = storage_options
        self.read_options = read_options
        self.columns = columns
        self.predicate = predicate
        self.limit = limit

    def execute(self, ectx) -> BatchIter:
        from ..io import readers
        remaining = self.limit
        if remaining is None and len(self.paths) > 1:
            # no limit to push down: overlap host decode across files
            yield from readers.read_files_prefetch(
                self.paths, self.file_format, self.columns,
                self.storage_options, self.read_options, ectx.device,
                predicate=self.predicate)
            return
        for path in self.paths:
            if remaining is not None and remaining <= 0:
                return
            for rb in readers.read_file(
                    path, self.file_format, self.columns, self.predicate,
                    remaining, self.storage_options, self.read_options,
                    ectx.device):
                if remaining is not None:
                    rb = rb.head(remaining)
                    remaining -= len(rb)
                yield rb
                if remaining is not None and remaining <= 0:
                    return


class ProjectOp(PhysicalOp):
    def __init__(self, child: PhysicalOp, exprs: List[ExprNode],
                 schema: Schema):
        super().__init__([child], schema, "Project")
        self.exprs = exprs

    def execute(self, ectx) -> BatchIter:
        from .cse import evaluate_with_cse
        for rb in self.children[0].execute_tracked(ectx):
            cols = []
            n = len(rb)
            outs = evaluate_with_cse(self.exprs, rb)
            for e, s in zip(self.exprs, outs):
                if len(s) == 1 and n != 1:
                    s = s.broadcast(n)
                cols.append(s.rename(e.to_field(rb.schema).name))
            yield RecordBatch(cols, num_rows=n)


class FilterOp(PhysicalOp):
    def __init__(self, child: PhysicalOp, predicate: ExprNode):
        super().__init__([child], child.schema, "Filter")
        self.predicate = predicate

    def execute(self, ectx) -> BatchIter:
        from .cse import evaluate_with_cse
        for rb in self.children[0].execute_tracked(ectx):
            mask = evaluate_with_cse([self.predicate], rb)[0]
            if len(mask) == 1 and len(rb) != 1:
                mask = mask.broadcast(len(rb))
            out = rb.filter(mask)
            if len(out):
                yield out


class LimitOp(PhysicalOp):
    """Streaming limit with offset (ref: streaming_sink/limit.rs:20-24)."""

    def __init__(self, child: PhysicalOp, limit: int, offset: int = 0):
        super().__init__([child], child.schema, f"Limit({limit})")
        self.limit = limit
        self.offset = offset

    def execute(self, ectx) -> BatchIter:
        to_skip = self.offset
        to_take = self.limit
        for rb in self.children[0].execute_tracked(ectx):
            if to_take <= 0:
                return
            if to_skip:
                if len(rb) <= to_skip:
                    to_skip -= len(rb)
                    continue
                rb = rb.slice(to_skip, len(rb))
                to_skip = 0
            if len(rb) > to_take:
                rb = rb.head(to_take)
            to_take -= len(rb)
            if len(rb):
                yield rb


class ExplodeOp(PhysicalOp):
    def __init__(self, child: PhysicalOp, exprs: List[ExprNode],
                 schema: Schema):
        super().__init__([child], schema, "Explode")
        self.exprs = exprs

    def execute(self, ectx) -> BatchIter:
        explode_names = [e.to_field(self.children[0].schema).name
                         for e in self.exprs]
        for rb in self.children[0].execute_tracked(ectx):
            n = len(rb)
            cols = {c.name: c for c in rb.columns}
            for e, nm in zip(self.exprs, explode_names):
                cols[nm] = e.evaluate(rb)
            first = cols[explode_names[0]]
            if first.dtype.kind == TypeKind.LIST:
                lens = first.offsets[1:] - first.offsets[:-1]
            else:
                lens = torch.full((n,), first.dtype.size, dtype=torch.int64,
                                  device=first.device)
            # empty/null lists explode to one null row (daft semantics)
            eff = lens.clamp(min=1)
            row_idx = torch.repeat_interleave(
                torch.arange(n, device=first.device, dtype=torch.int64), eff)
            out_cols = []
            for name, c in cols.items():
                if name in explode_names:
                    offs = torch.zeros(n + 1, dtype=torch.int64,
                                       device=first.device)
                    torch.cumsum(eff, 0, out=offs[1:])
                    total = int(offs[-1].item())
                    j = torch.arange(total, dtype=torch.int64,
                                     device=first.device)
                    within = j - offs[row_idx]
                    if c.dtype.kind == TypeKind.LIST:
                        starts = c.offsets[:-1]
                        child_idx = starts[row_idx] + within
                        child_idx = torch.where(
                            within < lens[row_idx], child_idx,
                            torch.full_like(child_idx, -1))
                        out_cols.append(
                            c.children[0].take(child_idx).rename(name))
                    else:
                        sz = c.dtype.size
                        child_idx = row_idx * sz + within
                        child_idx = torch.where(
                            within < lens[row_idx], child_idx,
                            torch.full_like(child_idx, -1))
                        out_cols.append(
                            c.children[0].take(child_idx).rename(name))
                else:
                    out_cols.append(c.take(row_idx))
            yield RecordBatch(out_cols, num_rows=int(row_idx.shape[0]))


class UnpivotOp(PhysicalOp):
    def __init__(self, child: PhysicalOp, ids: List[ExprNode],
                 values: List[ExprNode], variable_name: str, value_name: str,
                 schema: Schema):
        super().__init__([child], schema, "Unpivot")
        self.ids = ids
        self.values = values
        self.variable_name = variable_name
        self.value_name = value_name

    def execute(self, ectx) -> BatchIter:
        for rb in self.children[0].execute_tracked(ectx):
            n = len(rb)
            k = len(self.values)
            dev = rb.device
            id_series = [e.evaluate(rb) for e in self.ids]
            val_series = [e.evaluate(rb) for e in self.values]
            vdt = self.schema[self.value_name].dtype
            val_series = [s.cast(vdt) for s in val_series]
            # output rows: for each source row, k rows (one per value col)
            row_idx = torch.repeat_interleave(
                torch.arange(n, dtype=torch.int64, device=dev), k)
            out_cols = [s.take(row_idx) for s in id_series]
            names = [s.name for s in val_series]
            var = Series.from_pylist(self.variable_name, names * 1,
                                     DataType.string(), device=dev)
            var_idx = torch.arange(n * k, dtype=torch.int64,
                                   device=dev) % k
            out_cols.append(var.take(var_idx).rename(self.variable_name))
            stacked = Series.concat(val_series)
            gather = var_idx * n + row_idx
            out_cols.append(stacked.take(gather).rename(self.value_name))
            yield RecordBatch(out_cols, num_rows=n * k)


class SampleOp(PhysicalOp):
    def __init__(self, child: PhysicalOp, fraction: float,
                 with_replacement: bool, seed: Optional[int]):
        super().__init__([child], child.schema, "Sample")
        self.fraction = fraction
        self.with_replacement = with_replacement
        self.seed = seed

    def execute(self, ectx) -> BatchIter:
        gen = torch.Generator(device="cpu")
        if self.seed is not None:
            gen.manual_seed(self.seed)
        for rb in self.children[0].execute_tracked(ectx):
            n = len(rb)
            if self.with_replacement:
                k = int(round(n * self.fraction))
                idx = torch.randint(0, max(n, 1), (k,), generator=gen,
                                    dtype=torch.int64).to(rb.device)
                yield rb.take(idx)
            else:
                mask = (torch.rand(n, generator=gen) < self.fraction) \
                    .to(rb.device)
                out = rb.filter(Series("m", DataType.bool(), data=mask))
                if len(out):
                    yield out


class MonotonicIdOp(PhysicalOp):
    """partition_id << 36 | row_number (ref: make_monotonically_increasing_id)."""

    def __init__(self, child: PhysicalOp, column_name: str, schema: Schema,
                 partition_id: int = 0):
        super().__init__([child], schema, "MonotonicallyIncreasingId")
        self.column_name = column_name
        self.partition_id = partition_id

    def execute(self, ectx) -> BatchIter:
        base = self.partition_id << 36
        count = 0
        for rb in self.children[0].execute_tracked(ectx):
            n = len(rb)
            ids = torch.arange(base + count, base + count + n,
                               dtype=torch.int64, device=rb.device)
            count += n
            s = Series(self.column_name, DataType.uint64(),
                       data=ids.view(torch.uint64))
            yield RecordBatch([s] + list(rb.columns), num_rows=n)


class IntoBatchesOp(PhysicalOp):
    def __init__(self, child: PhysicalOp, batch_size: int):
        super().__init__([child], child.schema, f"IntoBatches({batch_size})")
        self.batch_size = batch_size

    def execute(self, ectx) -> BatchIter:
        pending: List[RecordBatch] = []
        pending_rows = 0
        for rb in self.children[0].execute_tracked(ectx):
            pending.append(rb)
            pending_rows += len(rb)
            while pending_rows >= self.batch_size:
                merged = RecordBatch.concat(pending) if len(pending) > 1 \
                    else pending[0]
                out = merged.head(self.batch_size)
                rest = merged.slice(self.batch_size, len(merged))
                yield out
                pending = [rest] if len(rest) else []
                pending_rows = len(rest)
        if pending_rows:
            merged = RecordBatch.concat(pending) if len(pending) > 1 \
                else pending[0]
            yield merged


class AggregateOp(PhysicalOp):
    """Blocking hash groupby-aggregate (ref: sinks/grouped_aggregate.rs)."""

    def __init__(self, child: PhysicalOp, groupby: List[ExprNode],
                 aggs: List[ExprNode], schema: Schema):
        super().__init__([child], schema, "HashAggregate" if groupby
                         else "Aggregate")
        self.groupby = groupby
        self.aggs = aggs

    _FUSABLE_AGGS = {
        AggKind.SUM, AggKind.COUNT, AggKind.COUNT_ALL, AggKind.MIN,
        AggKind.MAX, AggKind.MEAN, AggKind.STDDEV, AggKind.VARIANCE,
        AggKind.SKEW, AggKind.ANY_VALUE, AggKind.BOOL_AND,
        AggKind.BOOL_OR, AggKind.COUNT_DISTINCT,
        AggKind.APPROX_COUNT_DISTINCT, AggKind.SKETCH,
    }

    def _fusion_pred(self):
        """Filter-into-aggregate fusion: when the child is a Filter and
        every aggregation tolerates a row mask, evaluate the predicate
        as a mask instead of materializing the compacted input (high-
        selectivity scans stop paying a full-table gather)."""
        child = self.children[0]
        if not isinstance(child, FilterOp):
            return None, child
        named, _ = agg_mod.decompose_agg_exprs(self.aggs)
        if not named or any(a.kind not in self._FUSABLE_AGGS
                            for _, a in named):
            return None, child
        return child.predicate, child.children[0]

    @staticmethod
    def _prep(b, pred):
        """(batch, mask) for one input batch; low selectivity compacts
        (cheaper to gather few rows than to stream-mask every column)."""
        if pred is None or len(b) == 0:
            return b, None
        m = pred.evaluate(b)
        mask = m.data
        if len(m) == 1 and len(b) != 1:
            mask = mask.expand(len(b))
        if m.validity is not None:
            v = m.validity
            if v.numel() == 1 and len(b) != 1:
                v = v.expand(len(b))
            mask = mask & v
        nsel = int(mask.sum().item())
        if nsel * 2 < len(b):
            from ..kernels import compact_indices
            ms = Series("__m", DataType.bool(), data=mask.contiguous())
            return b.take(compact_indices(ms), has_neg=False), None
        return b, mask.contiguous()

    def execute(self, ectx) -> BatchIter:
        from itertools import chain as _chain
        pred, child = self._fusion_pred()
        it = child.execute_tracked(ectx)
        first = next(it, None)
        second = next(it, None) if first is not None else None
        if second is None:
            batch = first if first is not None else RecordBatch.empty(
                child.schema, device=ectx.device)
            batch, mask = self._prep(batch, pred)
            yield agg_mod.run_aggregate(batch, self.groupby, self.aggs,
                                        mask=mask)
            return
        # multiple input batches: fold each into per-group partial states
        # (same partial/final decomposition the distributed planner uses)
        # instead of concatenating the whole input — this is what lets
        # larger-than-HBM inputs stream through (out-of-core aggregation)
        from .agg_partial import split_partial_final
        split = split_partial_final(self.aggs)
        if split is None:
            batches = []
            for b in _chain([first, second], it):
                b, mask = self._prep(b, pred)
                if mask is not None:
                    idx = torch.nonzero(mask).reshape(-1)
                    b = b.take(idx, has_neg=False)
                batches.append(b)
            ectx.memory.admit(sum(b.size_bytes() for b in batches),
                              ectx.device)
            yield agg_mod.run_aggregate(RecordBatch.concat(batches),
                                        self.groupby, self.aggs)
            return
        partials, final_named, residuals = split
        from ..expressions.expressions import ColumnRef
        cschema = child.schema
        gnames = [e.to_field(cschema).name for e in self.groupby]
        parts = []
        for b in _chain([first, second], it):
            b, mask = self._prep(b, pred)
            parts.append(agg_mod.run_aggregate(b, self.groupby, partials,
                                               mask=mask))
        merged = RecordBatch.concat(parts)
        final = agg_mod.run_aggregate(
            merged, [ColumnRef(n) for n in gnames], final_named)
        cols = [final.column(n) for n in gnames]
        cols += [r.evaluate(final) for r in residuals]
        yield RecordBatch(cols, num_rows=len(final))


class DistinctOp(PhysicalOp):
    def __init__(self, child: PhysicalOp, subset: Optional[List[ExprNode]]):
        super().__init__([child], child.schema, "Dedup")
        self.subset = subset

    def execute(self, ectx) -> BatchIter:
        batch = self._materialize_child(ectx)
        if len(batch) == 0:
            yield batch
            return
        if self.subset:
            keys = [e.evaluate(batch) for e in self.subset]
        else:
            keys = list(batch.columns)
        _gids, reps = rowops.groupby(keys)
        yield batch.take(reps, has_neg=False)


class SortOp(PhysicalOp):
    """Blocking sort: materialize -> multi-key radix argsort -> gather
    (ref: sinks/sort.rs:63-141)."""

    def __init__(self, child: PhysicalOp, by: List[ExprNode],
                 descending: List[bool], nulls_first: List[bool]):
        super().__init__([child], child.schema, "Sort")
        self.by = by
        self.descending = descending
        self.nulls_first = nulls_first

    def _budget(self, ectx):
        cfg = getattr(ectx.ctx, "execution_config", None)
        lim = getattr(cfg, "memory_limit_bytes", None) if cfg else None
        if lim is not None:
            return lim
        if str(ectx.device).startswith("cuda"):
            import torch as _t
            if _t.cuda.is_available():
                free, _tot = _t.cuda.mem_get_info(ectx.device)
                return free // 3
        return None

    def execute(self, ectx) -> BatchIter:
        budget = self._budget(ectx)
        runs: List[RecordBatch] = []
        spilled: List[RecordBatch] = []
        acc = 0
        for rb in self.children[0].execute_tracked(ectx):
            if len(rb) == 0:
                continue
            runs.append(rb)
            acc += rb.size_bytes()
            if budget is not None and acc > budget:
                spilled.extend(r.cpu() for r in runs)
                runs, acc = [], 0
        if not spilled:
            if not runs:
                yield RecordBatch.empty(self.schema, device=ectx.device)
                return
            batch = RecordBatch.concat(runs) if len(runs) > 1 else runs[0]
            keys = [e.evaluate(batch) for e in self.by]
            perm = rowops.argsort_multi(keys, self.descending,
                                        self.nulls_first)
            yield batch.take(perm, has_neg=False)
            return
        spilled.extend(r.cpu() for r in runs)
        yield from self._external_sort(spilled, ectx, budget)

    def _external_sort(self, spilled: List[RecordBatch], ectx,
                       budget) -> BatchIter:
        """Out-of-core sort: inputs larger than the HBM budget spill to
        host, get range-partitioned by sampled boundaries into host
        buckets (one device pass), then each bucket sorts in HBM and is
        emitted in boundary order (ref: sinks/sort.rs buffering + the
        distributed sample-sort, pipeline_node/sort.rs:84-130 — here the
        'ranks' are HBM-sized buckets on one GPU)."""
        from ..recordbatch import _range_partition_ids
        total = sum(r.size_bytes() for r in spilled)
        nb = max(2, -(-int(total) // max(int(budget) // 4, 1)))
        names = [f"__k{i}" for i in range(len(self.by))]

        def keys_of(rb):
            cols = [e.evaluate(rb).rename(names[i])
                    for i, e in enumerate(self.by)]
            return RecordBatch(cols, num_rows=len(rb))

        # sample boundaries across all runs
        samples = []
        for run in spilled:
            k = min(len(run), 1024)
            if k == 0:
                continue
            step = max(1, len(run) // k)
            idx = torch.arange(0, len(run), step, dtype=torch.int64)[:k]
            samples.append(keys_of(run.take(idx)))
        allsamp = RecordBatch.concat(samples)
        m = len(allsamp)
        sorted_samp = allsamp.sort(names, self.descending, self.nulls_first)
        bidx = torch.tensor([min(m - 1, ((i + 1) * m) // nb)
                             for i in range(nb - 1)], dtype=torch.int64)
        boundaries = sorted_samp.take(bidx).to(ectx.device)

        morsel = getattr(ectx.ctx.execution_config, "stream_morsel_rows",
                         1 << 26)
        buckets: List[List[RecordBatch]] = [[] for _ in range(nb)]
        for run in spilled:
            for mo in stream_host_batch(run, ectx.device, morsel):
                pid = _range_partition_ids(keys_of(mo), names, boundaries,
                                           self.descending,
                                           self.nulls_first)
                perm, counts = rowops.partition_by_value(pid, nb)
                reordered = mo.take(perm)
                off = 0
                for b, c in enumerate(counts.tolist()):
                    if c:
                        buckets[b].append(
                            reordered.slice(off, off + c).cpu())
                    off += c
        for b in range(nb):
            parts = buckets[b]
            if not parts:
                continue
            dev_parts = [p.to(ectx.device) for p in parts]
            batch = RecordBatch.concat(dev_parts) if len(dev_parts) > 1 \
                else dev_parts[0]
            keys = [e.evaluate(batch) for e in self.by]
            perm = rowops.argsort_multi(keys, self.descending,
                                        self.nulls_first)
            yield batch.take(perm, has_neg=False)
            buckets[b] = []


class TopNOp(PhysicalOp):
    def __init__(self, child: PhysicalOp, by: List[ExprNode],
                 descending: List[bool], nulls_first: List[bool], limit: int,
                 offset: int = 0):
        super().__init__([child], child.schema, f"TopN({limit})")
        self.by = by
        self.descending = descending
        self.nulls_first = nulls_first
        self.limit = limit
        self.offset = offset

    def execute(self, ectx) -> BatchIter:
        # per-batch prune to limit+offset, folded into a bounded running
        # top-k accumulator (memory stays O(k), not O(batches*k))
        k = self.limit + self.offset

        def prune(rb: RecordBatch) -> RecordBatch:
            if len(rb) <= k:
                return rb
            keys = [e.evaluate(rb) for e in self.by]
            perm = rowops.argsort_multi(keys, self.descending,
                                        self.nulls_first)
            return rb.take(perm[:k])

        acc: Optional[RecordBatch] = None
        for rb in self.children[0].execute_tracked(ectx):
            if len(rb) == 0:
                continue
            rb = prune(rb)
            acc = rb if acc is None else prune(
                RecordBatch.concat([acc, rb]))
        if acc is None:
            yield RecordBatch.empty(self.schema, device=ectx.device)
            return
        keys = [e.evaluate(acc) for e in self.by]
        perm = rowops.argsort_multi(keys, self.descending, self.nulls_first)
        yield acc.take(perm[self.offset:k])


class JoinOp(PhysicalOp):
    """Hash join: build side = right child (ref: join/join_operator.rs;
    GPU build/probe via csrc/join.hip bucket-chained table)."""

    def __init__(self, left: PhysicalOp, right: PhysicalOp,
                 left_on: List[ExprNode], right_on: List[ExprNode],
                 how: str, schema: Schema,
                 right_cols: List):
        super().__init__([left, right], schema, f"HashJoin({how})")
        self.left_on = left_on
        self.right_on = right_on
        self.how = how
        self.right_cols = right_cols  # (src_name, out_name)

    def execute(self, ectx) -> BatchIter:
        right = self._materialize_child(ectx, 1)
        rkeys = [e.evaluate(right) for e in self.right_on]

        if self.how == "cross":
            left = self._materialize_child(ectx, 0)
            nl, nr = len(left), len(right)
            dev = left.device
            lidx = torch.repeat_interleave(
                torch.arange(nl, dtype=torch.int64, device=dev), nr)
            ridx = torch.arange(nr, dtype=torch.int64, device=dev).repeat(nl)
            yield self._emit(left, right, lidx, ridx)
            return

        if self.how in ("inner", "left", "semi", "anti"):
            # streamed probe: the build side is resident, each probe-side
            # batch joins and emits independently (out-of-core probes)
            for left in self.children[0].execute_tracked(ectx):
                if len(left) == 0:
                    continue
                lkeys = [e.evaluate(left) for e in self.left_on]
                lidx, ridx = rowops.join(lkeys, rkeys, self.how)
                if self.how in ("semi", "anti"):
                    yield left.take(lidx)
                else:
                    yield self._emit(left, right, lidx, ridx)
            return
        # right/outer joins track unmatched build rows across the whole
        # probe side: materialize
        left = self._materialize_child(ectx, 0)
        lkeys = [e.evaluate(left) for e in self.left_on]
        lidx, ridx = rowops.join(lkeys, rkeys, self.how)
        yield self._emit(left, right, lidx, ridx)

    def _emit(self, left: RecordBatch, right: RecordBatch,
              lidx: torch.Tensor, ridx: torch.Tensor) -> RecordBatch:
        # index signs are statically known per join type: passing the
        # has_neg hint skips a (idx < 0).any().item() device sync (and the
        # clamp kernel) per gathered column
        l_neg = self.how in ("right", "outer")
        r_neg = self.how in ("left", "outer")
        cols = [c.take(lidx, has_neg=l_neg) for c in left.columns]
        fan_out = len(ridx) >= 4 * max(len(right), 1)
        for src, out in self.right_cols:
            c = right.column(src)
            if fan_out and not c.is_dict() and \
                    c.dtype.kind in (TypeKind.STRING, TypeKind.BINARY) and \
                    len(c) <= 8_000_000:
                # high-fan-out payload strings: dict-encode the (small)
                # build side once so the probe-scale gather moves int32
                # codes, not bytes — and downstream groupbys take the
                # dict fast path
                c = _dict_encode(c)
            cols.append(c.take(ridx, has_neg=r_neg).rename(out))
        # outer/right joins: fill left-side join keys from the right keys
        if self.how in ("right", "outer") and len(self.left_on):
            has_null_left = bool((lidx < 0).any().item()) if lidx.numel() else False
            if has_null_left:
                lschema = left.schema
                for le, re in zip(self.left_on, self.right_on):
                    lname = le.to_field(lschema).name
                    rcol = re.evaluate(right).take(ridx)
                    i = left.schema.index_of(lname)
                    filled = cols[i].fill_null(rcol)
                    cols[i] = filled.rename(lname)
        return RecordBatch(cols, num_rows=int(lidx.shape[0]))


def _dict_encode(c: Series) -> Series:
    """Dedup a string column into a dictionary Series (distinct vocab,
    int32 codes) — groupby dedup, so safe even with duplicate values."""
    gids, reps = rowops.groupby([c])
    vocab = c.take(reps, has_neg=False)
    vocab = Series(vocab.name, vocab.dtype, data=vocab.data,
                   offsets=vocab.offsets, children=vocab.children)
    return Series.make_dict(c.name, vocab, gids.to(torch.int32), c.validity)


class AsofJoinOp(PhysicalOp):
    def __init__(self, left: PhysicalOp, right: PhysicalOp, left_on, right_on,
                 left_by, right_by, strategy, schema, right_cols):
        super().__init__([left, right], schema, f"AsofJoin({strategy})")
        self.left_on = left_on
        self.right_on = right_on
        self.left_by = left_by
        self.right_by = right_by
        self.strategy = strategy
        self.right_cols = right_cols

    def execute(self, ectx) -> BatchIter:
        from .asof import run_asof_join
        left = self._materialize_child(ectx, 0)
        right = self._materialize_child(ectx, 1)
        yield run_asof_join(left, right, self.left_on, self.right_on,
                            self.left_by, self.right_by, self.strategy,
                            self.right_cols)


class ConcatOp(PhysicalOp):
    def __init__(self, children: List[PhysicalOp], schema: Schema):
        super().__init__(children, schema, "Concat")

    def execute(self, ectx) -> BatchIter:
        for c in self.children:
            yield from c.execute_tracked(ectx)


class RepartitionOp(PhysicalOp):
    """Native-runner repartition: re-slice into N local parts (the
    distributed layer replaces this with an RCCL all-to-all exchange)."""

    def __init__(self, child: PhysicalOp, scheme: str,
                 num_partitions: Optional[int], by: List[ExprNode]):
        super().__init__([child], child.schema, f"Repartition({scheme})")
        self.scheme = scheme
        self.num_partitions = num_partitions
        self.by = by

    def execute(self, ectx) -> BatchIter:
        batch = self._materialize_child(ectx)
        n_parts = self.num_partitions or 1
        if len(batch) == 0 or n_parts <= 1:
            yield batch
            return
        if self.scheme == "hash":
            keys = [e.evaluate(batch) for e in self.by]
            perm, counts = rowops.partition_by_hash(keys, n_parts)
            reordered = batch.take(perm, has_neg=False)
            start = 0
            for c in counts.tolist():
                if c:
                    yield reordered.slice(start, start + c)
                start += c
        elif self.scheme in ("into", "random"):
            rows_per = math.ceil(len(batch) / n_parts)
            for i in range(n_parts):
                part = batch.slice(i * rows_per, (i + 1) * rows_per)
                if len(part):
                    yield part
        else:
            raise ValueError(f"unknown repartition scheme {self.scheme}")


class PivotOp(PhysicalOp):
    def __init__(self, child: PhysicalOp, groupby: List[ExprNode],
                 pivot_col: ExprNode, value_col: ExprNode, agg_kind: str,
                 names: List[str], schema: Schema):
        super().__init__([child], schema, "Pivot")
        self.groupby = groupby
        self.pivot_col = pivot_col
        self.value_col = value_col
        self.agg_kind = agg_kind
        self.names = names

    def execute(self, ectx) -> BatchIter:
        batch = self._materialize_child(ectx)
        # group by (groupby + pivot), aggregate value, then scatter to columns
        gb = self.groupby + [self.pivot_col]
        agg_expr = Alias(Agg(self.agg_kind, self.value_col), "__pv")
        inter = agg_mod.run_aggregate(batch, gb, [agg_expr])
        key_names = [e.to_field(batch.schema).name for e in self.groupby]
        pname = self.pivot_col.to_field(batch.schema).name
        keys = [inter.column(n) for n in key_names]
        gids, reps = rowops.groupby(keys) if keys else (
            torch.zeros(len(inter), dtype=torch.int64, device=inter.device),
            torch.zeros(1 if len(inter) else 0, dtype=torch.int64,
                        device=inter.device))
        num_groups = int(reps.shape[0])
        pv = inter.column(pname).cpu().to_pylist()
        vals = inter.column("__pv")
        out_cols = [k.take(reps, has_neg=False) for k in keys]
        gid_cpu = gids.cpu().tolist()
        for name in self.names:
            sel = torch.full((num_groups,), -1, dtype=torch.int64,
                             device=inter.device)
            sel_cpu = sel.cpu()
            for row, (p, g) in enumerate(zip(pv, gid_cpu)):
                if p is not None and str(p) == name:
                    sel_cpu[g] = row
            out_cols.append(vals.take(sel_cpu.to(inter.device)).rename(name))
        yield RecordBatch(out_cols, num_rows=num_groups)


class WindowOp(PhysicalOp):
    def __init__(self, child: PhysicalOp, window_exprs, partition_by,
                 order_by, descending, names, schema: Schema):
        super().__init__([child], schema, "Window")
        self.window_exprs = window_exprs
        self.partition_by = partition_by
        self.order_by = order_by
        self.descending = descending
        self.names = names

    def execute(self, ectx) -> BatchIter:
        from .window import run_window
        batch = self._materialize_child(ectx)
        yield run_window(batch, self.window_exprs, self.partition_by,
                         self.order_by, self.descending, self.names)


class UDFProjectOp(PhysicalOp):
    def __init__(self, child: PhysicalOp, udf_expr: ExprNode,
                 passthrough: List[ExprNode], schema: Schema):
        super().__init__([child], schema, "UDFProject")
        self.udf_expr = udf_expr
        self.passthrough = passthrough

    def execute(self, ectx) -> BatchIter:
        for rb in self.children[0].execute_tracked(ectx):
            cols = [e.evaluate(rb).rename(e.to_field(rb.schema).name)
                    for e in self.passthrough]
            s = self.udf_expr.evaluate(rb)
            if len(s) == 1 and len(rb) != 1:
                s = s.broadcast(len(rb))
            cols.append(s.rename(self.udf_expr.to_field(rb.schema).name))
            yield RecordBatch(cols, num_rows=len(rb))


class WriteOp(PhysicalOp):
    """Physical write sink -> manifest of written paths (ref: sinks/write.rs
    + daft-writers)."""

    def __init__(self, child: PhysicalOp, file_format: str, root_dir: str,
                 write_mode: str, partition_cols: List[ExprNode],
                 options: dict, schema: Schema):
        super().__init__([child], schema, f"Write({file_format})")
        self.file_format = file_format
        self.root_dir = root_dir
        self.write_mode = write_mode
        self.partition_cols = partition_cols
        self.options = options

    def execute(self, ectx) -> BatchIter:
        from ..io import writers
        paths = writers.write_batches(
            self.children[0].execute_tracked(ectx), self.file_format,
            self.root_dir,
            self.write_mode, self.partition_cols, self.options, ectx)
        yield RecordBatch([Series.from_pylist("path", paths,
                                              DataType.string())],
                          num_rows=len(paths))
