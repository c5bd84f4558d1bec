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


def push_semi_into_agg(plan: lp.LogicalPlan) -> Optional[lp.LogicalPlan]:
    """When an Aggregate's output inner-joins on its groupby key against a
    much smaller relation, semi-filter the aggregate's INPUT by that
    relation first: q17 aggregates per-partkey means over all of lineitem
    and then joins 0.1% of parts — the filter belongs below the groupby.
    (ref: the reference's KeyFilteringJoinNode + push_down_aggregation
    interplay, key_filtering_join.rs:38)"""
    if not (isinstance(plan, lp.Join) and plan.how == "inner"):
        return None
    for side in (0, 1):
        node = plan.children[side]
        other = plan.children[1 - side]
        keys = list(plan.left_on if side == 0 else plan.right_on)
        other_keys = list(plan.right_on if side == 0 else plan.left_on)
        o_est = other.approx_num_rows()
        n_est = node.approx_num_rows()
        if o_est is None or n_est is None or o_est > 0.3 * n_est:
            continue
        # resolve key names through Project renames down to an Aggregate
        key_names = []
        ok = True
        for e in keys:
            base = e.child if isinstance(e, Alias) else e
            if not isinstance(base, ColumnRef):
                ok = False
                break
            key_names.append(base.name)
        if not ok:
            continue
        cur = node
        while isinstance(cur, lp.Project):
            mapping = {}
            for pe in cur.exprs:
                out = pe.to_field(cur.children[0].schema).name
                pb = pe.child if isinstance(pe, Alias) else pe
                if isinstance(pb, ColumnRef):
                    mapping[out] = pb.name
            if not all(k in mapping for k in key_names):
                ok = False
                break
            key_names = [mapping[k] for k in key_names]
            cur = cur.children[0]
        if not ok or not isinstance(cur, lp.Aggregate) or not cur.groupby:
            continue
        gb_src = {}
        for g in cur.groupby:
            out = g.to_field(cur.children[0].schema).name
            gb = g.child if isinstance(g, Alias) else g
            if isinstance(gb, ColumnRef):
                gb_src[out] = gb.name
        if not all(k in gb_src for k in key_names):
            continue
        src_names = [gb_src[k] for k in key_names]
        inner = cur.children[0]
        # already filtered by this relation: stop the fixed-point loop
        if isinstance(inner, lp.Join) and inner.how == "semi" and \
                inner.children[1].semantic_id() == other.semantic_id():
            continue
        semi = lp.Join(inner, other,
                       [ColumnRef(s) for s in src_names], other_keys,
                       "semi")
        new_agg = lp.Aggregate(semi, cur.groupby, cur.aggs)
        # rebuild the Project chain above the aggregate
        rebuilt = new_agg
        chain = []
        c = node
        while isinstance(c, lp.Project):
            chain.append(c)
            c = c.children[0]
        for pr in reversed(chain):
            rebuilt = lp.Project(rebuilt, pr.exprs)
        new_children = [rebuilt, other] if side == 0 else [other, rebuilt]
        return plan.with_children(new_children)
    return None
