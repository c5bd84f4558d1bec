"""Window specification (ref: /root/reference/daft/window.py and
daft-dsl/src/expr/window.rs)."""
from __future__ import annotations

from typing import List, Optional, Sequence, Union


class Window:
    """Partition/order/frame window spec.

    Round-1 frames: whole-partition aggregation, running (order-by)
    aggregation, and rank/row-number functions."""

    def __init__(self):
        self.partition_by_exprs: list = []
        self.order_by_exprs: list = []
        self.descending: List[bool] = []
        self.frame: Optional[tuple] = None  # (start, end) row offsets

    def partition_by(self, *cols) -> "Window":
        from .expressions.expressions import resolve_exprs
        w = self._copy()
        w.partition_by_exprs = w.partition_by_exprs + resolve_exprs(list(cols))
        return w

    def order_by(self, *cols, desc: Union[bool, Sequence[bool]] = False
                 ) -> "Window":
        from .expressions.expressions import resolve_exprs
        w = self._copy()
        nodes = resolve_exprs(list(cols))
        if isinstance(desc, bool):
            d = [desc] * len(nodes)
        else:
            d = list(desc)
        w.order_by_exprs = w.order_by_exprs + nodes
        w.descending = w.descending + d
        return w

    def rows_between(self, start, end) -> "Window":
        w = self._copy()
        w.frame = (start, end)
        return w

    unbounded_preceding = "unbounded_preceding"
    unbounded_following = "unbounded_following"
    current_row = "current_row"

    def _copy(self) -> "Window":
        w = Window()
        w.partition_by_exprs = list(self.partition_by_exprs)
        w.order_by_exprs = list(self.order_by_exprs)
        w.descending = list(self.descending)
        w.frame = self.frame
        return w
