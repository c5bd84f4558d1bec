"""LogicalPlanBuilder (ref: /root/reference/src/daft-logical-plan/src/builder/
and daft/logical/builder.py:55) — the fluent layer DataFrame drives."""
from __future__ import annotations

from typing import List, Optional, Sequence, Union

from ..expressions.expressions import (Agg, AggKind, Alias, ColumnRef,
                                       ExprNode, Expression, resolve_exprs)
from ..schema import Schema
from . import plan as lp


class LogicalPlanBuilder:
    def __init__(self, node: lp.LogicalPlan):
        self.plan = node

    # ------------------------------------------------------------------
    @staticmethod
    def from_in_memory(schema: Schema, cache_key: str, num_rows: int,
                       size_bytes: int = 0,
                       partitioning=None) -> "LogicalPlanBuilder":
        return LogicalPlanBuilder(
            lp.Source(schema, cache_key, num_rows, size_bytes,
                      partitioning))

    @staticmethod
    def from_scan(schema: Schema, paths: List[str], file_format: str,
                  storage_options=None, read_options=None) -> "LogicalPlanBuilder":
        return LogicalPlanBuilder(
            lp.ScanSource(schema, paths, file_format, storage_options,
                          read_options=read_options))

    @property
    def schema(self) -> Schema:
        return self.plan.schema

    def _wrap(self, node: lp.LogicalPlan) -> "LogicalPlanBuilder":
        return LogicalPlanBuilder(node)

    # ------------------------------------------------------------------
    def select(self, exprs: Sequence) -> "LogicalPlanBuilder":
        return self._wrap(lp.Project(self.plan, resolve_exprs(exprs)))

    def with_columns(self, exprs: Sequence) -> "LogicalPlanBuilder":
        nodes = resolve_exprs(exprs)
        cschema = self.schema
        new_names = {n.to_field(cschema).name for n in nodes}
        keep = [ColumnRef(f.name) for f in cschema
                if f.name not in new_names]
        return self._wrap(lp.Project(self.plan, keep + nodes))

    def exclude(self, names: Sequence[str]) -> "LogicalPlanBuilder":
        keep = [ColumnRef(f.name) for f in self.schema
                if f.name not in set(names)]
        return self._wrap(lp.Project(self.plan, keep))

    def rename(self, mapping: dict) -> "LogicalPlanBuilder":
        exprs = []
        for f in self.schema:
            if f.name in mapping:
                exprs.append(Alias(ColumnRef(f.name), mapping[f.name]))
            else:
                exprs.append(ColumnRef(f.name))
        return self._wrap(lp.Project(self.plan, exprs))

    def filter(self, predicate) -> "LogicalPlanBuilder":
        [node] = resolve_exprs([predicate])
        return self._wrap(lp.Filter(self.plan, node))

    def limit(self, n: int, offset: int = 0) -> "LogicalPlanBuilder":
        return self._wrap(lp.Limit(self.plan, n, offset))

    def explode(self, exprs: Sequence) -> "LogicalPlanBuilder":
        return self._wrap(lp.Explode(self.plan, resolve_exprs(exprs)))

    def unpivot(self, ids: Sequence, values: Sequence, variable_name: str,
                value_name: str) -> "LogicalPlanBuilder":
        return self._wrap(lp.Unpivot(self.plan, resolve_exprs(ids),
                                     resolve_exprs(values), variable_name,
                                     value_name))

    def sort(self, by: Sequence, descending, nulls_first) -> "LogicalPlanBuilder":
        nodes = resolve_exprs(by)
        k = len(nodes)
        desc = _normalize_flags(descending, k, False)
        nf = _normalize_flags(nulls_first, k, None)
        nf = [d if f is None else f for f, d in zip(nf, desc)]
        return self._wrap(lp.Sort(self.plan, nodes, desc, nf))

    def distinct(self, subset: Optional[Sequence] = None) -> "LogicalPlanBuilder":
        sub = resolve_exprs(subset) if subset else None
        return self._wrap(lp.Distinct(self.plan, sub))

    def aggregate(self, aggs: Sequence, groupby: Sequence) -> "LogicalPlanBuilder":
        return self._wrap(lp.Aggregate(self.plan, resolve_exprs(groupby),
                                       resolve_exprs(aggs)))

    def pivot(self, groupby: Sequence, pivot_col, value_col, agg_kind: str,
              names: List[str]) -> "LogicalPlanBuilder":
        [p] = resolve_exprs([pivot_col])
        [v] = resolve_exprs([value_col])
        return self._wrap(lp.Pivot(self.plan, resolve_exprs(groupby), p, v,
                                   agg_kind, names))

    def concat(self, other: "LogicalPlanBuilder") -> "LogicalPlanBuilder":
        return self._wrap(lp.Concat(self.plan, other.plan))

    def join(self, right: "LogicalPlanBuilder", left_on: Sequence,
             right_on: Sequence, how: str = "inner",
             suffix: str = "_right", prefix=None) -> "LogicalPlanBuilder":
        return self._wrap(lp.Join(self.plan, right.plan,
                                  resolve_exprs(left_on),
                                  resolve_exprs(right_on), how, suffix,
                                  prefix))

    def cross_join(self, right: "LogicalPlanBuilder",
                   suffix: str = "_right") -> "LogicalPlanBuilder":
        return self._wrap(lp.Join(self.plan, right.plan, [], [], "cross",
                                  suffix))

    def repartition(self, num_partitions: Optional[int], scheme: str = "hash",
                    by: Optional[Sequence] = None) -> "LogicalPlanBuilder":
        nodes = resolve_exprs(by) if by else []
        return self._wrap(lp.Repartition(self.plan, scheme, num_partitions,
                                         nodes))

    def into_partitions(self, n: int) -> "LogicalPlanBuilder":
        return self._wrap(lp.Repartition(self.plan, "into", n))

    def into_batches(self, batch_size: int) -> "LogicalPlanBuilder":
        return self._wrap(lp.IntoBatches(self.plan, batch_size))

    def sample(self, fraction: float, with_replacement=False,
               seed=None) -> "LogicalPlanBuilder":
        return self._wrap(lp.Sample(self.plan, fraction, with_replacement,
                                    seed))

    def add_monotonically_increasing_id(self, name: str) -> "LogicalPlanBuilder":
        return self._wrap(lp.MonotonicallyIncreasingId(self.plan, name))

    def window(self, window_exprs, partition_by, order_by, descending,
               names) -> "LogicalPlanBuilder":
        return self._wrap(lp.Window(
            self.plan, resolve_exprs(window_exprs),
            resolve_exprs(partition_by), resolve_exprs(order_by),
            list(descending), names))

    def write(self, file_format: str, root_dir: str, write_mode="overwrite",
              partition_cols=None, options=None) -> "LogicalPlanBuilder":
        pc = resolve_exprs(partition_cols) if partition_cols else None
        return self._wrap(lp.Sink(self.plan, file_format, root_dir,
                                  write_mode, pc, options))

    # ------------------------------------------------------------------
    def optimize(self) -> "LogicalPlanBuilder":
        from ..optimizer import optimize
        return self._wrap(optimize(self.plan))

    def explain(self) -> str:
        return "\n".join(self.plan.explain_lines())


def _normalize_flags(flags, k: int, default):
    if flags is None:
        return [default] * k
    if isinstance(flags, bool):
        return [flags] * k
    flags = list(flags)
    assert len(flags) == k
    return flags
