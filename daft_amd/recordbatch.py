"""RecordBatch: a batch of equal-length Series (ref:
/root/reference/src/daft-recordbatch/src/lib.rs:68-72) and MicroPartition,
the unit of data movement between operators (ref:
/root/reference/src/daft-micropartition/src/micropartition.rs:35-53).

On GPU every column buffer lives in HBM3E; the vectorized ops here
(filter/take/sort/agg/join/partition) dispatch to the HIP kernel layer.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple, Union

import torch

from .schema import DataType, Field, Schema, TypeKind
from .series import Series, empty_series, full_null
from . import kernels
from .kernels import rowops


class RecordBatch:
    __slots__ = ("schema", "columns", "_num_rows")

    def __init__(self, columns: List[Series], num_rows: Optional[int] = None):
        self.columns = columns
        self.schema = Schema([c.field() for c in columns])
        if num_rows is None:
            num_rows = len(columns[0]) if columns else 0
        for c in columns:
            assert len(c) == num_rows, \
                f"column {c.name} length {len(c)} != {num_rows}"
        self._num_rows = num_rows

    # ------------------------------------------------------------------
    @staticmethod
    def from_pydict(data: Dict[str, list], device="cpu",
                    schema: Optional[Schema] = None) -> "RecordBatch":
        cols = []
        for name, vals in data.items():
            dt = schema[name].dtype if schema is not None and name in schema else None
            if isinstance(vals, Series):
                v = vals.rename(name)
                cols.append(v.to(device) if str(v.device) != str(device)
                            else v)
            elif isinstance(vals, torch.Tensor):
                cols.append(Series.from_torch(name, vals.to(device)))
            else:
                cols.append(Series.from_pylist(name, vals, dt, device=device))
        return RecordBatch(cols)

    @staticmethod
    def empty(schema: Schema, device="cpu") -> "RecordBatch":
        return RecordBatch([empty_series(f.name, f.dtype, device)
                            for f in schema], num_rows=0)

    @staticmethod
    def from_arrow(table, device="cpu") -> "RecordBatch":
        import pyarrow as pa
        if isinstance(table, pa.RecordBatch):
            table = pa.Table.from_batches([table])
        cols = [Series.from_arrow(name, table.column(name))
                for name in table.column_names]
        rb = RecordBatch(cols, num_rows=table.num_rows)
        return rb.to(device) if str(device) != "cpu" else rb

    # ------------------------------------------------------------------
    def __len__(self) -> int:
        return self._num_rows

    @property
    def num_rows(self) -> int:
        return self._num_rows

    @property
    def device(self) -> torch.device:
        return self.columns[0].device if self.columns else torch.device("cpu")

    def column(self, name: str) -> Series:
        return self.columns[self.schema.index_of(name)]

    def column_names(self) -> List[str]:
        return self.schema.names()

    def to(self, device, non_blocking: bool = False) -> "RecordBatch":
        return RecordBatch([c.to(device, non_blocking)
                            for c in self.columns], self._num_rows)

    def pinned(self) -> "RecordBatch":
        return RecordBatch([c.pinned() for c in self.columns],
                           self._num_rows)

    def cpu_pinned(self) -> "RecordBatch":
        return RecordBatch([c.cpu_pinned() for c in self.columns],
                           self._num_rows)

    def is_pinned(self) -> bool:
        return all(c.is_pinned() for c in self.columns)

    def cpu(self) -> "RecordBatch":
        return self.to("cpu")

    def size_bytes(self) -> int:
        total = 0
        for c in self.columns:
            for t in (c.data, c.validity, c.offsets):
                if t is not None:
                    total += t.numel() * t.element_size()
            for ch in c.children:
                total += RecordBatch([ch]).size_bytes()
        return total

    # ------------------------------------------------------------------
    def select_columns(self, names: Sequence[str]) -> "RecordBatch":
        return RecordBatch([self.column(n) for n in names], self._num_rows)

    def with_columns(self, series: List[Series]) -> "RecordBatch":
        cols = list(self.columns)
        idx = {c.name: i for i, c in enumerate(cols)}
        for s in series:
            if s.name in idx:
                cols[idx[s.name]] = s
            else:
                idx[s.name] = len(cols)
                cols.append(s)
        return RecordBatch(cols, self._num_rows)

    def rename(self, mapping: Dict[str, str]) -> "RecordBatch":
        return RecordBatch([c.rename(mapping.get(c.name, c.name))
                            for c in self.columns], self._num_rows)

    # ------------------------------------------------------------------
    def take(self, indices: torch.Tensor,
             has_neg: "bool | None" = None) -> "RecordBatch":
        # hoist the negative-index host sync across columns
        if has_neg is None:
            has_neg = bool((indices < 0).any().item()) if indices.numel() \
                else False
        return RecordBatch([c.take(indices, has_neg=has_neg)
                            for c in self.columns],
                           int(indices.shape[0]))

    def filter(self, mask: Series) -> "RecordBatch":
        idx = kernels.compact_indices(mask)
        return self.take(idx)

    def slice(self, start: int, end: int) -> "RecordBatch":
        end = min(end, self._num_rows)
        start = min(start, end)
        return RecordBatch([c.slice(start, end) for c in self.columns],
                           end - start)

    def head(self, n: int) -> "RecordBatch":
        return self.slice(0, n)

    @staticmethod
    def concat(batches: List["RecordBatch"]) -> "RecordBatch":
        assert batches
        if len(batches) == 1:
            return batches[0]
        names = batches[0].column_names()
        cols = [Series.concat([b.column(n) for b in batches]) for n in names]
        return RecordBatch(cols, sum(len(b) for b in batches))

    # ------------------------------------------------------------------
    def argsort(self, by: Sequence[str], descending: Sequence[bool],
                nulls_first: Sequence[bool]) -> torch.Tensor:
        keys = [self.column(n) for n in by]
        return rowops.argsort_multi(keys, descending, nulls_first)

    def sort(self, by: Sequence[str], descending: Sequence[bool],
             nulls_first: Sequence[bool]) -> "RecordBatch":
        return self.take(self.argsort(by, descending, nulls_first))

    def hash_rows(self, columns: Optional[Sequence[str]] = None,
                  seed: int = 0) -> torch.Tensor:
        cols = [self.column(n) for n in (columns or self.column_names())]
        return rowops.hash_columns(cols, seed)

    def partition_by_hash(self, columns: Sequence[str],
                          num_partitions: int) -> List["RecordBatch"]:
        keys = [self.column(n) for n in columns]
        perm, counts = rowops.partition_by_hash(keys, num_partitions)
        reordered = self.take(perm)
        out = []
        start = 0
        counts_l = counts.tolist()
        for c in counts_l:
            out.append(reordered.slice(start, start + c))
            start += c
        return out

    def partition_by_range(self, keys: Sequence[str],
                           boundaries: "RecordBatch",
                           descending: Sequence[bool]) -> List["RecordBatch"]:
        """Range partition: boundaries has num_partitions-1 sorted rows."""
        part = _range_partition_ids(self, keys, boundaries, descending)
        perm, counts = rowops.partition_by_value(part, len(boundaries) + 1)
        reordered = self.take(perm)
        out, start = [], 0
        for c in counts.tolist():
            out.append(reordered.slice(start, start + c))
            start += c
        return out

    # ------------------------------------------------------------------
    def to_pydict(self) -> Dict[str, list]:
        return {c.name: c.to_pylist() for c in self.columns}

    def to_arrow(self):
        import pyarrow as pa
        arrays = [c.to_arrow() for c in self.columns]
        return pa.table(dict(zip(self.column_names(), arrays)))

    def to_pandas(self):
        cols = [c.dict_decode() if c.is_dict() else c for c in self.columns]
        return RecordBatch(cols, self._num_rows).to_arrow().to_pandas()

    def __repr__(self) -> str:
        return (f"RecordBatch(rows={self._num_rows}, dev={self.device}, "
                f"schema={self.schema!r})")


def _range_partition_ids(rb: RecordBatch, keys: Sequence[str],
                         boundaries: RecordBatch,
                         descending: Sequence[bool],
                         nulls_first: Optional[Sequence[bool]] = None
                         ) -> torch.Tensor:
    """For each row, count how many boundary rows sort strictly before it.
    Null keys order per nulls_first (default: the argsort convention,
    nulls first exactly when descending)."""
    n = len(rb)
    dev = rb.device
    if nulls_first is None:
        nulls_first = list(descending)
    part = torch.zeros(n, dtype=torch.int64, device=dev)
    for b in range(len(boundaries)):
        # row > boundary_b (lexicographically, honoring per-key direction)
        gt = None  # strictly greater so far
        eq = None  # equal so far
        for ki, kname in enumerate(keys):
            col = rb.column(kname)
            bcol = boundaries.column(kname).slice(b, b + 1)
            bval = bcol.broadcast(n)
            op_gt = "lt" if descending[ki] else "gt"
            g = col.compare(bval, op_gt).data.clone()
            e = col.compare(bval, "eq").data.clone()
            row_null = (~col.validity) if col.validity is not None else None
            b_null = bcol.validity is not None and \
                not bool(bcol.validity.item())
            if row_null is not None:
                if b_null:
                    # null vs null: equal
                    g = torch.where(row_null, torch.zeros_like(g), g)
                    e = torch.where(row_null, torch.ones_like(e), e)
                else:
                    # null row vs value: after iff nulls last
                    val = not nulls_first[ki]
                    g = torch.where(row_null,
                                    torch.full_like(g, val), g)
                    e = torch.where(row_null, torch.zeros_like(e), e)
                if not b_null:
                    pass
            if b_null and row_null is None:
                # value vs null boundary: after iff nulls first
                g = torch.full_like(g, nulls_first[ki])
                e = torch.zeros_like(e)
            elif b_null and row_null is not None:
                nonnull = ~row_null
                g = torch.where(nonnull,
                                torch.full_like(g, nulls_first[ki]), g)
                e = torch.where(nonnull, torch.zeros_like(e), e)
            if gt is None:
                gt, eq = g, e
            else:
                gt = gt | (eq & g)
                eq = eq & e
        part += gt.to(torch.int64)
    return part


# reference-name alias: the reference's MicroPartition is its batch-of-
# record-batches unit; here RecordBatch fills both roles
MicroPartition = RecordBatch


def read_parquet_into_pyarrow(path, columns=None, **kwargs):
    """Thin helper mirroring daft.recordbatch.read_parquet_into_pyarrow."""
    import pyarrow.parquet as pq
    return pq.read_table(path, columns=columns)


def read_parquet_into_pyarrow_bulk(paths, columns=None, **kwargs):
    return [read_parquet_into_pyarrow(p, columns=columns) for p in paths]
