"""Two-phase (partial/final) aggregation splitting, used by the distributed
planner (ref: the reference's partial-agg strategy in
daft-local-execution/src/sinks/grouped_aggregate.rs AggStrategy).

split_partial_final(aggs, schema) -> (partial_exprs, final_exprs) or None
if some aggregation cannot be decomposed (caller falls back to a row
exchange / gather)."""
from __future__ import annotations

from typing import List, Optional, Tuple

from ..expressions.expressions import (Agg, AggKind, Alias, BinaryOp,
                                       ColumnRef, ExprNode, Literal, ScalarFn)
from ..schema import Schema
from .agg import decompose_agg_exprs

_DECOMPOSABLE = {
    AggKind.SUM, AggKind.COUNT, AggKind.COUNT_ALL, AggKind.MIN, AggKind.MAX,
    AggKind.MEAN, AggKind.ANY_VALUE, AggKind.BOOL_AND, AggKind.BOOL_OR,
    AggKind.STDDEV, AggKind.VARIANCE, AggKind.APPROX_PERCENTILE,
}


def split_partial_final(aggs: List[ExprNode]
                        ) -> Optional[Tuple[List[ExprNode], List[ExprNode]]]:
    named, residuals = decompose_agg_exprs(aggs)
    if any(a.kind not in _DECOMPOSABLE for _, a in named):
        return None
    partials: List[ExprNode] = []
    final_named: List[ExprNode] = []  # produce columns named like `named`
    for name, a in named:
        k = a.kind
        if k in (AggKind.SUM, AggKind.MIN, AggKind.MAX, AggKind.ANY_VALUE,
                 AggKind.BOOL_AND, AggKind.BOOL_OR):
            partials.append(Alias(Agg(k, a.child), name))
            merge = {AggKind.SUM: AggKind.SUM, AggKind.MIN: AggKind.MIN,
                     AggKind.MAX: AggKind.MAX,
                     AggKind.ANY_VALUE: AggKind.ANY_VALUE,
                     AggKind.BOOL_AND: AggKind.BOOL_AND,
                     AggKind.BOOL_OR: AggKind.BOOL_OR}[k]
            final_named.append(Alias(Agg(merge, ColumnRef(name)), name))
        elif k in (AggKind.COUNT, AggKind.COUNT_ALL):
            partials.append(Alias(Agg(k, a.child), name))
            final_named.append(Alias(Agg(AggKind.SUM, ColumnRef(name)), name))
        elif k == AggKind.MEAN:
            partials.append(Alias(Agg(AggKind.SUM, a.child), f"{name}__s"))
            partials.append(Alias(Agg(AggKind.COUNT, a.child), f"{name}__c"))
            final_named.append(Alias(
                BinaryOp("div", Agg(AggKind.SUM, ColumnRef(f"{name}__s")),
                         Agg(AggKind.SUM, ColumnRef(f"{name}__c"))), name))
        elif k in (AggKind.STDDEV, AggKind.VARIANCE):
            sq = BinaryOp("mul", a.child, a.child)
            partials.append(Alias(Agg(AggKind.SUM, a.child), f"{name}__s"))
            partials.append(Alias(Agg(AggKind.SUM, sq), f"{name}__s2"))
            partials.append(Alias(Agg(AggKind.COUNT, a.child), f"{name}__c"))
            s = Agg(AggKind.SUM, ColumnRef(f"{name}__s"))
            s2 = Agg(AggKind.SUM, ColumnRef(f"{name}__s2"))
            c = Agg(AggKind.SUM, ColumnRef(f"{name}__c"))
            mean = BinaryOp("div", s, c)
            var = BinaryOp("sub", BinaryOp("div", s2, c),
                           BinaryOp("mul", mean, mean))
            var = ScalarFn("clip0", _clip0, [var],
                           _float64_dt())
            if k == AggKind.STDDEV:
                final_named.append(Alias(ScalarFn(
                    "sqrt", _sqrt_series, [var], _float64_dt()), name))
            else:
                final_named.append(Alias(var, name))
        elif k == AggKind.APPROX_PERCENTILE:
            # DDSketch build per shard; merged buckets + quantile extract
            # after the exchange (physical/sketch.py)
            partials.append(Alias(Agg(AggKind.SKETCH, a.child),
                                  f"{name}__sk"))
            final_named.append(Alias(
                Agg(AggKind.SKETCH_FINAL, ColumnRef(f"{name}__sk"),
                    a.param), name))
        else:  # pragma: no cover
            return None
    finals = final_named + [r for r in residuals]
    # residuals reference the named agg columns; the final Aggregate's
    # decompose pass maps Agg(sum, col(name)) etc. and then evaluates
    # residuals over them.  But residuals as-is reference `name` as a plain
    # column, which after the final aggregate exists only if some final
    # produced it — it does (final_named has every `name`).  We therefore
    # return final aggregate exprs = final_named, plus post-projection
    # residuals handled by the caller.
    return partials, final_named, residuals


def _float64_dt():
    from ..schema import DataType
    return DataType.float64()


def _clip0(s):
    import torch
    from ..series import Series
    from ..schema import DataType
    return Series(s.name, DataType.float64(),
                  data=s.data.to(torch.float64).clamp(min=0.0),
                  validity=s.validity)


def _sqrt_series(s):
    import torch
    from ..series import Series
    from ..schema import DataType
    return Series(s.name, DataType.float64(),
                  data=torch.sqrt(s.data.to(torch.float64)),
                  validity=s.validity)
