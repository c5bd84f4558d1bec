"""Common-subexpression elimination for per-batch expression evaluation
(capability of the reference's daft-algebra expression rewrites).

Pure-core subtrees (arithmetic / comparisons / conditionals over columns
and literals) that appear more than once across a projection's
expressions are evaluated ONCE per batch and substituted; ScalarFn /
UDF / Agg / Window subtrees are never shared (their reprs don't uniquely
identify the function)."""
from __future__ import annotations

from typing import Dict, List

from ..expressions.expressions import (Alias, Between, BinaryOp, Cast,
                                       Coalesce, ColumnRef, ExprNode,
                                       FillNull, IfElse, IsIn, IsNull,
                                       Literal, Not)

_PURE = (Alias, Between, BinaryOp, Cast, Coalesce, ColumnRef, FillNull,
         IfElse, IsIn, IsNull, Literal, Not)


class _Precomputed(ExprNode):
    __slots__ = ("series",)

    def __init__(self, series):
        self.series = series

    def evaluate(self, batch):
        return self.series

    def __repr__(self):
        return f"<pre {self.series.name}>"


def _pure(e: ExprNode) -> bool:
    if not isinstance(e, _PURE):
        return False
    return all(_pure(c) for c in e.children())


def shared_subtrees(exprs: List[ExprNode]) -> set:
    """repr keys of pure subtrees (with children) occurring >= 2 times."""
    counts: Dict[str, int] = {}

    def walk(e: ExprNode, in_pure: bool):
        pure_here = in_pure or _pure(e)
        if e.children():
            if pure_here and isinstance(e, _PURE):
                key = repr(e)
                counts[key] = counts.get(key, 0) + 1
            for c in e.children():
                walk(c, False)
    for e in exprs:
        walk(e, False)
    return {k for k, c in counts.items() if c >= 2}


def evaluate_with_cse(exprs: List[ExprNode], batch):
    """Evaluate exprs over the batch, computing shared pure subtrees
    once.  Returns the list of result Series (unnamed).

    On device batches the whole list first tries the fused interpreter
    kernel (kernels/fused.py): everything numeric compiles into ONE
    launch; only unfusable expressions fall back here."""
    if batch.device.type == "cuda":
        from ..kernels.fused import try_fuse
        fused = try_fuse(exprs, batch)
        if fused is not None:
            return fused
    shared = shared_subtrees(exprs)
    if not shared:
        return [e.evaluate(batch) for e in exprs]
    cache: Dict[str, object] = {}

    def rewrite(e: ExprNode) -> ExprNode:
        if e.children() and isinstance(e, _PURE):
            key = repr(e)
            if key in shared and _pure(e):
                if key not in cache:
                    sub = e.with_children(
                        [rewrite(c) for c in e.children()])
                    cache[key] = sub.evaluate(batch)
                return _Precomputed(cache[key])
        if not e.children():
            return e
        return e.with_children([rewrite(c) for c in e.children()])

    return [rewrite(e).evaluate(batch) for e in exprs]
