"""Physical exchange operators (RCCL over xGMI; gloo on the CPU test tier)."""
from __future__ import annotations

from typing import List

import torch

from ..expressions.expressions import ExprNode
from ..kernels import rowops
from ..physical.ops import BatchIter, PhysicalOp
from ..recordbatch import RecordBatch, _range_partition_ids
from ..schema import DataType, Schema, TypeKind
from ..series import Series
from . import comm


def _exchange_budget(ectx) -> "int | None":
    """HBM budget for one exchange (send+recv working set); over it, the
    exchange chunks rows and stages receives to host (spill tier)."""
    cfg = getattr(ectx.ctx, "execution_config", None)
    explicit = getattr(cfg, "exchange_hbm_budget_bytes", None) if cfg else None
    if explicit is not None:
        return explicit
    if str(ectx.device).startswith("cuda") and torch.cuda.is_available():
        free, _total = torch.cuda.mem_get_info(ectx.device)
        return free // 2
    return None


def _yield_exchanged(out: RecordBatch, ectx):
    """Yield an exchange result; a spilled (host-staged) result streams back
    to the device in morsel slices so downstream operators fold partial
    states instead of re-materializing the whole partition in HBM."""
    if str(out.device) == "cpu" and str(ectx.device).startswith("cuda"):
        from ..physical.ops import stream_host_batch
        morsel = getattr(ectx.ctx.execution_config, "stream_morsel_rows",
                         1 << 26)
        yield from stream_host_batch(out, ectx.device, morsel)
        return
    yield out


def _normalize_key(s: Series) -> Series:
    """Cast keys to width-stable dtypes so both sides of a co-partition hash
    identically (int->int64, float->float64, bool->int64)."""
    dt = s.dtype
    if dt.is_integer() or dt.is_boolean() or dt.is_temporal():
        return s.cast(DataType.int64()) if s.data is not None and \
            s.data.dtype != torch.int64 else s
    if dt.kind == TypeKind.FLOAT32:
        return s.cast(DataType.float64())
    if dt.is_decimal():
        # p <= 18 decimals are scaled int64: hash the raw ints (exact);
        # wide decimals stored f64 normalize like floats
        if s.data is not None and s.data.dtype == torch.int64:
            return s
        return s.cast(DataType.float64())
    if s.is_dict():
        # vocab order may differ across ranks/tables: hash real bytes
        return s.dict_decode()
    return s


class ExchangeByKeyOp(PhysicalOp):
    def __init__(self, child: PhysicalOp, keys: List[ExprNode]):
        super().__init__([child], child.schema, "ExchangeByKey[RCCL a2a]")
        self.keys = keys

    def execute(self, ectx) -> BatchIter:
        batch = self._materialize_child(ectx)
        w = comm.world()
        if w == 1:
            yield batch
            return
        key_series = [_normalize_key(e.evaluate(batch)) for e in self.keys]
        if len(batch) == 0:
            parts = [batch] * w
        else:
            perm, counts = rowops.partition_by_hash(key_series, w)
            reordered = batch.take(perm)
            parts, start = [], 0
            for c in counts.tolist():
                parts.append(reordered.slice(start, start + c))
                start += c
        out = comm.exchange_batches(parts,
                                    hbm_budget=_exchange_budget(ectx))
        yield from _yield_exchanged(out, ectx)


class GatherToRank0Op(PhysicalOp):
    def __init__(self, child: PhysicalOp):
        super().__init__([child], child.schema, "GatherToRank0")

    def execute(self, ectx) -> BatchIter:
        batch = self._materialize_child(ectx)
        w = comm.world()
        if w == 1:
            yield batch
            return
        empty = batch.slice(0, 0)
        parts = [batch if p == 0 else empty for p in range(w)]
        yield comm.exchange_batches(parts)


class ReplicateAllOp(PhysicalOp):
    def __init__(self, child: PhysicalOp):
        super().__init__([child], child.schema, "ReplicateAll[allgather]")

    def execute(self, ectx) -> BatchIter:
        batch = self._materialize_child(ectx)
        yield comm.allgather_batch(batch, ectx.device)


class RangeExchangeOp(PhysicalOp):
    """Sample keys -> allgather -> boundaries -> range partition -> a2a
    (ref: daft-distributed pipeline_node/sort.rs:84-130 sample phase)."""

    SAMPLES_PER_RANK = 256

    def __init__(self, child: PhysicalOp, by: List[ExprNode],
                 descending: List[bool], nulls_first: List[bool]):
        super().__init__([child], child.schema, "RangeExchange")
        self.by = by
        self.descending = descending
        self.nulls_first = nulls_first

    def execute(self, ectx) -> BatchIter:
        batch = self._materialize_child(ectx)
        w = comm.world()
        if w == 1:
            yield batch
            return
        n = len(batch)
        key_cols = [e.evaluate(batch).rename(f"__k{i}")
                    for i, e in enumerate(self.by)]
        keys_rb = RecordBatch(key_cols, num_rows=n)
        # sample
        k = min(n, self.SAMPLES_PER_RANK)
        if k > 0:
            step = max(1, n // k)
            idx = torch.arange(0, n, step, dtype=torch.int64,
                               device=batch.device)[:k]
            sample = keys_rb.take(idx)
        else:
            sample = keys_rb.slice(0, 0)
        all_samples = comm.allgather_batch(sample, ectx.device)
        m = len(all_samples)
        if m == 0:
            yield comm.exchange_batches(
                [batch if p == comm.rank() else batch.slice(0, 0)
                 for p in range(w)])
            return
        names = [c.name for c in key_cols]
        sorted_samples = all_samples.sort(names, self.descending,
                                          self.nulls_first)
        bidx = torch.tensor([min(m - 1, ((i + 1) * m) // w)
                             for i in range(w - 1)], dtype=torch.int64,
                            device=batch.device)
        boundaries = sorted_samples.take(bidx)
        part = _range_partition_ids(keys_rb, names, boundaries,
                                    self.descending)
        perm, counts = rowops.partition_by_value(part, w)
        reordered = batch.take(perm)
        parts, start = [], 0
        for c in counts.tolist():
            parts.append(reordered.slice(start, start + c))
            start += c
        out = comm.exchange_batches(parts,
                                    hbm_budget=_exchange_budget(ectx))
        yield from _yield_exchanged(out, ectx)


class Rank0OnlyOp(PhysicalOp):
    def __init__(self, child: PhysicalOp):
        super().__init__([child], child.schema, "Rank0Only")

    def execute(self, ectx) -> BatchIter:
        batch = self._materialize_child(ectx)
        yield batch if comm.rank() == 0 else batch.slice(0, 0)
