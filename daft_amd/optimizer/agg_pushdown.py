"""PushDownAggregation: an Aggregate over a Join whose measures come only
from the join's right side pre-aggregates that side by its join keys, then
merges partials above the (now much smaller) join.

This turns the spec-shaped TPC-H q13

    Aggregate(by=c_custkey, count(o_orderkey),
              Join(customer, orders, left, c_custkey=o_custkey))

into the orders-pre-aggregated plan the hand-tuned benchmark used in
round 1 — the build side drops from |orders| to |distinct custkey| and the
join takes the direct-address PK path.

(ref: /root/reference/src/daft-logical-plan/src/optimization/rules/
push_down_aggregation.rs)
"""
from __future__ import annotations

from typing import List, Optional

from ..expressions.expressions import (Agg, AggKind, Alias, Cast, ColumnRef,
                                       ExprNode, FillNull, Literal)
from ..logical import plan as lp
from ..schema import DataType

# kinds whose partial-per-key states merge losslessly above the join
_MERGEABLE = {AggKind.SUM, AggKind.COUNT, AggKind.COUNT_ALL,
              AggKind.MIN, AggKind.MAX}


def _collect_aggs(e: ExprNode, out: List[Agg]):
    if isinstance(e, Agg):
        out.append(e)
        return
    for c in e.children():
        _collect_aggs(c, out)


def _refs_outside_aggs(e: ExprNode) -> set:
    if isinstance(e, Agg):
        return set()
    s = set()
    if isinstance(e, ColumnRef):
        s.add(e.name)
    for c in e.children():
        s |= _refs_outside_aggs(c)
    return s


def _replace_aggs(e: ExprNode, repl: dict) -> ExprNode:
    if id(e) in repl:
        return repl[id(e)]
    ch = e.children()
    if not ch:
        return e
    new_ch = [_replace_aggs(c, repl) for c in ch]
    if all(n is o for n, o in zip(new_ch, ch)):
        return e
    return e.with_children(new_ch)


def _subst_cols(e: ExprNode, mapping: dict) -> ExprNode:
    if isinstance(e, ColumnRef) and e.name in mapping:
        return ColumnRef(mapping[e.name])
    ch = e.children()
    if not ch:
        return e
    return e.with_children([_subst_cols(c, mapping) for c in ch])


def push_down_aggregation(plan: lp.LogicalPlan) -> Optional[lp.LogicalPlan]:
    if not isinstance(plan, lp.Aggregate) or not plan.groupby:
        return None
    join = plan.children[0]
    if not isinstance(join, lp.Join) or join.how not in ("inner", "left"):
        return None
    right = join.children[1]
    ls = join.children[0].schema
    lnames = set(ls.names())
    rout = dict(join.right_passthrough())       # right src -> join out
    out2src = {v: k for k, v in rout.items()}
    right_out = set(rout.values())

    # right join keys must be plain columns (we re-join on the pre-agg keys)
    rk_names = []
    for e in join.right_on:
        base = e.child if isinstance(e, Alias) else e
        if not isinstance(base, ColumnRef):
            return None
        rk_names.append(base.name)
    # already pre-aggregated by exactly these keys: nothing to gain (also
    # the structural guard that terminates the rewrite)
    if isinstance(right, lp.Aggregate):
        gb_names = set()
        for g in right.groupby:
            base = g.child if isinstance(g, Alias) else g
            if isinstance(base, ColumnRef):
                gb_names.add(base.name)
        if gb_names == set(rk_names):
            return None

    # group keys reference left columns only
    for g in plan.groupby:
        if not set(g.column_refs()) <= lnames:
            return None

    aggs_found: List[Agg] = []
    for a in plan.aggs:
        _collect_aggs(a, aggs_found)
        if not _refs_outside_aggs(a) <= lnames:
            return None
    if not aggs_found:
        return None
    uses_right = False
    for a in aggs_found:
        if a.kind not in _MERGEABLE:
            return None
        if a.child is not None:
            refs = set(a.child.column_refs())
            if not refs:
                continue
            if not refs <= right_out:
                return None
            uses_right = True
    # measures purely from the left side gain nothing from pre-aggregating
    # the right (except pure count(*), which still shrinks the join)
    if not uses_right and not any(a.kind == AggKind.COUNT_ALL
                                  for a in aggs_found):
        return None

    # only worthwhile when grouping by the join key actually SHRINKS the
    # right side: pre-aggregating 150M orders by their unique o_orderkey
    # (q12) is a full extra groupby for zero reduction
    r_est = right.approx_num_rows()
    if r_est is not None and r_est > 10_000 and len(rk_names) == 1:
        from .join_reorder import _ndv_of
        nd = _ndv_of(right, rk_names[0], r_est)
        if nd is not None and nd > 0.5 * r_est:
            return None

    u64 = DataType.uint64()
    pre_aggs: List[ExprNode] = []
    repl: dict = {}
    for i, a in enumerate(aggs_found):
        pname = f"__pa{i}"
        if a.kind == AggKind.COUNT_ALL or a.child is None:
            pre_aggs.append(Alias(Agg(AggKind.COUNT_ALL, None), pname))
            fill = 1 if join.how == "left" else 0
            repl[id(a)] = Cast(
                Agg(AggKind.SUM, FillNull(ColumnRef(pname),
                                          Literal(fill, u64))), u64)
            continue
        src_child = _subst_cols(a.child, out2src)
        if a.kind == AggKind.COUNT:
            pre_aggs.append(Alias(Agg(AggKind.COUNT, src_child), pname))
            repl[id(a)] = Cast(
                Agg(AggKind.SUM, FillNull(ColumnRef(pname),
                                          Literal(0, u64))), u64)
        elif a.kind == AggKind.SUM:
            pre_aggs.append(Alias(Agg(AggKind.SUM, src_child), pname))
            repl[id(a)] = Agg(AggKind.SUM, ColumnRef(pname))
        else:  # MIN / MAX
            pre_aggs.append(Alias(Agg(a.kind, src_child), pname))
            repl[id(a)] = Agg(a.kind, ColumnRef(pname))

    new_right = lp.Aggregate(right, [ColumnRef(k) for k in rk_names],
                             pre_aggs)
    new_join = lp.Join(join.children[0], new_right, join.left_on,
                       [ColumnRef(k) for k in rk_names], join.how,
                       join.suffix, join.prefix)
    old_schema = plan.children[0].schema
    new_aggs = []
    for a in plan.aggs:
        na = _replace_aggs(a, repl)
        want = a.to_field(old_schema).name
        if na.to_field(new_join.schema).name != want:
            na = Alias(na, want)    # keep the output schema identical
        new_aggs.append(na)
    return lp.Aggregate(new_join, plan.groupby, new_aggs)
