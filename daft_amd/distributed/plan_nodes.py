"""Distributed logical plan nodes: exchanges over RCCL (ref capability:
daft-distributed pipeline_node/shuffles + gather + sort.rs sample-based
range partitioning)."""
from __future__ import annotations

from typing import List

from ..expressions.expressions import ExprNode
from ..logical.plan import LogicalPlan


class ExchangeByKey(LogicalPlan):
    """Hash co-partition rows across ranks by key exprs (RCCL all-to-all)."""

    def __init__(self, child: LogicalPlan, keys: List[ExprNode]):
        super().__init__([child])
        self.keys = keys

    def _compute_schema(self):
        return self.children[0].schema

    def with_children(self, ch):
        return ExchangeByKey(ch[0], self.keys)

    def describe(self):
        return f"ExchangeByKey({self.keys!r})"


class GatherToRank0(LogicalPlan):
    """All rows to rank 0 (other ranks emit empty)."""

    def __init__(self, child: LogicalPlan):
        super().__init__([child])

    def _compute_schema(self):
        return self.children[0].schema

    def with_children(self, ch):
        return GatherToRank0(ch[0])


class ReplicateAll(LogicalPlan):
    """Allgather: every rank receives the full concatenation (broadcast-join
    build side replication)."""

    def __init__(self, child: LogicalPlan):
        super().__init__([child])

    def _compute_schema(self):
        return self.children[0].schema

    def with_children(self, ch):
        return ReplicateAll(ch[0])


class RangeExchange(LogicalPlan):
    """Sample-based range partition across ranks (pre-sort exchange)."""

    def __init__(self, child: LogicalPlan, by: List[ExprNode],
                 descending: List[bool], nulls_first: List[bool]):
        super().__init__([child])
        self.by = by
        self.descending = descending
        self.nulls_first = nulls_first

    def _compute_schema(self):
        return self.children[0].schema

    def with_children(self, ch):
        return RangeExchange(ch[0], self.by, self.descending,
                             self.nulls_first)


class Rank0Only(LogicalPlan):
    """Emit child output on rank 0 only (empty elsewhere).  The child still
    executes on every rank (it contains collectives)."""

    def __init__(self, child: LogicalPlan):
        super().__init__([child])

    def _compute_schema(self):
        return self.children[0].schema

    def with_children(self, ch):
        return Rank0Only(ch[0])
