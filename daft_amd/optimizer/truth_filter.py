"""TruthValue filter folding over exact source column ranges.

The reference propagates per-column [min, max] statistics through the
plan and evaluates filter predicates to a three-valued TruthValue
(definitely-true / definitely-false / maybe), pruning scans and
dropping redundant filters (ref: /root/reference/src/daft-stats/src/
column_stats/ + daft-logical-plan/src/stats.rs).  This rule is the plan-
level equivalent here:

  * predicate definitely TRUE for every row (and the referenced columns
    are null-free — a filter also drops null rows!)  ->  drop the Filter
  * predicate definitely FALSE  ->  replace with Limit 0

Bounds are EXACT per-source min/max (optimizer/stats.range_for_source),
never sampled, and the verdicts stay valid below intermediate Filters /
Limits / Samples (a subset of the source range can only keep the verdict
true).  Under SPMD the ranges come back None unless synced, so plans
never diverge across ranks.
"""
from __future__ import annotations

import datetime as _dt
from typing import Dict, Optional, Tuple

from ..expressions.expressions import (Between, BinaryOp, ColumnRef, IsNull,
                                       Literal, Not)
from ..logical import plan as lp
from .stats import range_for_source

_EPOCH = _dt.date(1970, 1, 1)

# verdict: True / False / None (maybe); null_free: bool


def _lit_value(e) -> Optional[float]:
    if not isinstance(e, Literal):
        return None
    v = e.value
    if isinstance(v, bool):
        return float(v)
    if isinstance(v, (int, float)):
        return float(v)
    if isinstance(v, _dt.date) and not isinstance(v, _dt.datetime):
        return float((v - _EPOCH).days)
    return None


def _col_range(e, ranges):
    if isinstance(e, ColumnRef):
        return ranges.get(e.name)
    return None


def _cmp_verdict(op: str, lo: float, hi: float, v: float):
    """TruthValue of `col <op> v` given col in [lo, hi]."""
    if op == "lt":
        return True if hi < v else (False if lo >= v else None)
    if op == "le":
        return True if hi <= v else (False if lo > v else None)
    if op == "gt":
        return True if lo > v else (False if hi <= v else None)
    if op == "ge":
        return True if lo >= v else (False if hi < v else None)
    if op == "eq":
        return True if lo == hi == v else \
            (False if v < lo or v > hi else None)
    if op == "ne":
        return False if lo == hi == v else \
            (True if v < lo or v > hi else None)
    return None


_FLIP = {"lt": "gt", "le": "ge", "gt": "lt", "ge": "le", "eq": "eq",
         "ne": "ne"}


def truth_value(e, ranges: Dict[str, tuple]
                ) -> Tuple[Optional[bool], bool]:
    """(verdict, null_free) of predicate `e` over the column ranges."""
    if isinstance(e, BinaryOp):
        if e.op == "and":
            lv, ln = truth_value(e.left, ranges)
            rv, rn = truth_value(e.right, ranges)
            nf = ln and rn
            if lv is False or rv is False:
                return False, nf
            if lv is True and rv is True:
                return True, nf
            return None, nf
        if e.op == "or":
            lv, ln = truth_value(e.left, ranges)
            rv, rn = truth_value(e.right, ranges)
            nf = ln and rn
            if lv is True and ln:
                return True, nf or ln
            if rv is True and rn:
                return True, nf or rn
            if lv is False and rv is False:
                return False, nf
            return None, nf
        if e.op in _FLIP:
            cr, v = _col_range(e.left, ranges), _lit_value(e.right)
            op = e.op
            if cr is None or v is None:
                cr, v = _col_range(e.right, ranges), _lit_value(e.left)
                op = _FLIP.get(e.op)
            if cr is None or v is None or op is None:
                return None, False
            lo, hi, has_nulls = cr
            return _cmp_verdict(op, lo, hi, v), not has_nulls
    if isinstance(e, Between):
        cr = _col_range(e.child, ranges)
        vlo, vhi = _lit_value(e.lo), _lit_value(e.hi)
        if cr is None or vlo is None or vhi is None:
            return None, False
        lo, hi, has_nulls = cr
        if lo >= vlo and hi <= vhi:
            return True, not has_nulls
        if hi < vlo or lo > vhi:
            return False, not has_nulls
        return None, not has_nulls
    if isinstance(e, Not):
        v, nf = truth_value(e.child, ranges)
        if v is None or not nf:
            # NOT over a nullable column keeps nulls null: no safe fold
            return None, nf
        return (not v), nf
    if isinstance(e, IsNull):
        cr = _col_range(e.child, ranges)
        if cr is None:
            return None, False
        has_nulls = cr[2]
        if not has_nulls:
            # is_null -> definitely False; not_null (negate) -> True
            return (True, True) if getattr(e, "negate", False) \
                else (False, True)
        return None, False
    return None, False


def _source_of(node):
    """Resolve through verdict-preserving ops to the backing Source."""
    while isinstance(node, (lp.Filter, lp.Limit, lp.Sample)) or \
            (isinstance(node, lp.LogicalPlan) and
             type(node).__name__ == "Shard"):
        node = node.children[0]
    return node if isinstance(node, lp.Source) else None


def fold_filter_by_stats(plan: lp.LogicalPlan) -> Optional[lp.LogicalPlan]:
    if not isinstance(plan, lp.Filter):
        return None
    src = _source_of(plan.children[0])
    if src is None:
        return None
    ranges = {}
    for name in set(plan.predicate.column_refs()):
        r = range_for_source(src.cache_key, name)
        if r is not None:
            ranges[name] = r
    if not ranges:
        return None
    verdict, null_free = truth_value(plan.predicate, ranges)
    if verdict is True and null_free:
        return plan.children[0]
    if verdict is False:
        return lp.Limit(plan.children[0], 0)
    return None
