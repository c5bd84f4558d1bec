"""Logical optimizer: ordered rule batches run to fixed point (ref:
/root/reference/src/daft-logical-plan/src/optimization/optimizer.rs:111-300
and rules/ — this implements the load-bearing subset for a GPU engine:
SimplifyExpressions, PushDownFilter, PushDownProjection, PushDownLimit,
DropRepartition, TopN rewrite, join-side sizing)."""
from __future__ import annotations

from typing import Callable, List, Optional, Set

from ..expressions.expressions import (Agg, Alias, Between, BinaryOp, Cast,
                                       ColumnRef, Coalesce, ExprNode, FillNull,
                                       IfElse, IsIn, IsNull, Literal, Not,
                                       PyUDF, ScalarFn)
from ..logical import plan as lp

Rule = Callable[[lp.LogicalPlan], Optional[lp.LogicalPlan]]


def optimize(plan: lp.LogicalPlan) -> lp.LogicalPlan:
    import os
    from .agg_pushdown import push_down_aggregation
    from .agg_pushdown import push_semi_into_agg as _push_semi_into_agg
    from .join_reorder import reorder_joins
    disabled = set((os.environ.get("DAFT_AMD_DISABLE_RULES") or "")
                   .split(","))
    once = _apply_bottom_up(plan, derive_or_implications_rule)
    if once is not None:
        plan = once
    from .truth_filter import fold_filter_by_stats
    batches: List[List[Rule]] = [
        [simplify_expressions],
        [push_down_filter, drop_repartition] +
        ([] if "statsfold" in disabled else [fold_filter_by_stats]) +
        ([] if "semipush" in disabled else [push_down_anti_semi_join]),
        [] if "aggpush" in disabled else [push_down_aggregation,
                                          _push_semi_into_agg],
        [push_down_projection],
        [push_down_limit, rewrite_topn],
        [simplify_expressions],
    ]
    if "reorder" in disabled:
        reorder_joins = lambda p: p  # noqa: E731
    reorder_after = 1  # join reordering once filters sit at the sources
    for bi, rules in enumerate(batches):
        changed = False
        for _round in range(8):  # fixed-point cap
            changed = False
            for rule in rules:
                new = _apply_bottom_up(plan, rule)
                if new is not None:
                    plan = new
                    changed = True
            if not changed:
                break
        if changed:
            # convergence telemetry: a batch still rewriting at the cap
            # means deep plans may be silently under-optimized
            import warnings
            names = [getattr(r, "__name__", "?") for r in rules]
            warnings.warn(
                f"optimizer batch {bi} ({names}) hit the fixed-point cap "
                f"(8 rounds) while still rewriting", RuntimeWarning)
        if bi == reorder_after:
            plan = reorder_joins(plan)
            if "reorder" not in disabled:
                from .join_reorder import swap_join_builds
                plan = swap_join_builds(plan)
    return plan


def _apply_bottom_up(plan: lp.LogicalPlan, rule: Rule) -> Optional[lp.LogicalPlan]:
    changed = False
    new_children = []
    for c in plan.children:
        nc = _apply_bottom_up(c, rule)
        if nc is not None:
            changed = True
            new_children.append(nc)
        else:
            new_children.append(c)
    if changed:
        plan = plan.with_children(new_children)
    out = rule(plan)
    if out is not None:
        return out
    return plan if changed else None


# ---------------------------------------------------------------------------
# expression rewrites
# ---------------------------------------------------------------------------

def _map_expr(e: ExprNode, fn: Callable[[ExprNode], Optional[ExprNode]]):
    ch = e.children()
    new_ch = []
    changed = False
    for c in ch:
        nc = _map_expr(c, fn)
        if nc is not None:
            changed = True
            new_ch.append(nc)
        else:
            new_ch.append(c)
    if changed:
        e = e.with_children(new_ch)
    out = fn(e)
    if out is not None:
        return out
    return e if changed else None


def _simplify_node(e: ExprNode) -> Optional[ExprNode]:
    if isinstance(e, BinaryOp):
        l, r = e.left, e.right
        if e.op == "and":
            if isinstance(l, Literal) and l.value is True:
                return r
            if isinstance(r, Literal) and r.value is True:
                return l
            if isinstance(l, Literal) and l.value is False:
                return l
            if isinstance(r, Literal) and r.value is False:
                return r
        if e.op == "or":
            if isinstance(l, Literal) and l.value is False:
                return r
            if isinstance(r, Literal) and r.value is False:
                return l
            if isinstance(l, Literal) and l.value is True:
                return l
            if isinstance(r, Literal) and r.value is True:
                return r
        # constant fold pure-literal arithmetic
        if isinstance(l, Literal) and isinstance(r, Literal) and \
                e.op in ("add", "sub", "mul", "div") and \
                isinstance(l.value, (int, float)) and \
                isinstance(r.value, (int, float)):
            try:
                v = {"add": lambda: l.value + r.value,
                     "sub": lambda: l.value - r.value,
                     "mul": lambda: l.value * r.value,
                     "div": lambda: l.value / r.value}[e.op]()
                return Literal(v)
            except ZeroDivisionError:
                return None
    if isinstance(e, Not) and isinstance(e.child, Not):
        return e.child.child
    if isinstance(e, Alias) and isinstance(e.child, Alias):
        return Alias(e.child.child, e.name)
    return None


def simplify_expressions(plan: lp.LogicalPlan) -> Optional[lp.LogicalPlan]:
    def rewrite_list(exprs):
        out, changed = [], False
        for e in exprs:
            ne = _map_expr(e, _simplify_node)
            if ne is not None:
                changed = True
                out.append(ne)
            else:
                out.append(e)
        return out, changed

    if isinstance(plan, lp.Project):
        exprs, ch = rewrite_list(plan.exprs)
        if ch:
            return lp.Project(plan.children[0], exprs)
    if isinstance(plan, lp.Filter):
        ne = _map_expr(plan.predicate, _simplify_node)
        if ne is not None:
            if isinstance(ne, Literal) and ne.value is True:
                return plan.children[0]
            return lp.Filter(plan.children[0], ne)
    if isinstance(plan, lp.Aggregate):
        g, ch1 = rewrite_list(plan.groupby)
        a, ch2 = rewrite_list(plan.aggs)
        if ch1 or ch2:
            return lp.Aggregate(plan.children[0], g, a)
    return None


# ---------------------------------------------------------------------------
# filter pushdown
# ---------------------------------------------------------------------------

def _split_conjunctions(e: ExprNode) -> List[ExprNode]:
    if isinstance(e, BinaryOp) and e.op == "and":
        return _split_conjunctions(e.left) + _split_conjunctions(e.right)
    return [e]


def _split_disjunctions(e: ExprNode) -> List[ExprNode]:
    if isinstance(e, BinaryOp) and e.op == "or":
        return _split_disjunctions(e.left) + _split_disjunctions(e.right)
    return [e]


def _derive_or_implications(preds: List[ExprNode]) -> List[ExprNode]:
    """For `(a=x and ...) or (a=y and ...)`, every branch pins column `a`,
    so `a IN (x, y)` is implied — derive it as an extra conjunct that CAN be
    pushed below joins the OR itself cannot cross (the TPC-H Q7/Q19 shape).
    """
    derived: List[ExprNode] = []
    for p in preds:
        branches = _split_disjunctions(p)
        if len(branches) < 2:
            continue
        # column -> set of pinned literal values per branch
        per_branch = []
        for b in branches:
            pins = {}
            for cj in _split_conjunctions(b):
                if isinstance(cj, BinaryOp) and cj.op == "eq":
                    cr, litv = None, None
                    if isinstance(cj.left, ColumnRef) and                             isinstance(cj.right, Literal):
                        cr, litv = cj.left, cj.right
                    elif isinstance(cj.right, ColumnRef) and                             isinstance(cj.left, Literal):
                        cr, litv = cj.right, cj.left
                    if cr is not None and isinstance(
                            litv.value, (int, float, str, bool)):
                        pins.setdefault(cr.name, set()).add(litv.value)
                elif isinstance(cj, IsIn):
                    base = cj.child
                    if isinstance(base, ColumnRef) and all(
                            isinstance(v, (int, float, str, bool))
                            for v in cj.values):
                        pins.setdefault(base.name, set()).update(cj.values)
            per_branch.append(pins)
        common = set(per_branch[0])
        for pb in per_branch[1:]:
            common &= set(pb)
        for name in common:
            vals = set()
            for pb in per_branch:
                vals |= pb[name]
            derived.append(IsIn(ColumnRef(name), sorted(vals, key=repr)))
    return derived


def _conjoin(es: List[ExprNode]) -> ExprNode:
    out = es[0]
    for e in es[1:]:
        out = BinaryOp("and", out, e)
    return out


def _substitute_cols(e: ExprNode, mapping: dict) -> ExprNode:
    def fn(n: ExprNode):
        if isinstance(n, ColumnRef) and n.name in mapping:
            return mapping[n.name]
        return None
    out = _map_expr(e, fn)
    return out if out is not None else e


def _has_udf(e: ExprNode) -> bool:
    if isinstance(e, PyUDF):
        return True
    return any(_has_udf(c) for c in e.children())


def push_down_filter(plan: lp.LogicalPlan) -> Optional[lp.LogicalPlan]:
    if not isinstance(plan, lp.Filter):
        return None
    child = plan.children[0]
    preds = _split_conjunctions(plan.predicate)

    if isinstance(child, lp.Filter):
        # merge adjacent filters
        return lp.Filter(child.children[0],
                         _conjoin(preds + _split_conjunctions(child.predicate)))

    if isinstance(child, lp.Project):
        # map output names -> defining exprs; only push predicates whose
        # referenced columns resolve to pushable (non-UDF) exprs
        cschema = child.children[0].schema
        mapping = {}
        for e in child.exprs:
            name = e.to_field(child.children[0].schema).name
            base = e.child if isinstance(e, Alias) else e
            mapping[name] = base
        pushable, kept = [], []
        for p in preds:
            refs = p.column_refs()
            ok = all(r in mapping for r in refs)
            if ok:
                sub = _substitute_cols(p, mapping)
                if not _has_udf(sub) and not sub.is_aggregation():
                    pushable.append(sub)
                    continue
            kept.append(p)
        if pushable:
            new_child = lp.Project(
                lp.Filter(child.children[0], _conjoin(pushable)), child.exprs)
            if kept:
                return lp.Filter(new_child, _conjoin(kept))
            return new_child
        return None

    if isinstance(child, lp.Sort):
        return lp.Sort(lp.Filter(child.children[0], plan.predicate),
                       child.by, child.descending, child.nulls_first)

    if isinstance(child, lp.Repartition):
        return child.with_children(
            [lp.Filter(child.children[0], plan.predicate)])

    if isinstance(child, lp.Concat):
        return lp.Concat(lp.Filter(child.children[0], plan.predicate),
                         lp.Filter(child.children[1], plan.predicate))

    if isinstance(child, lp.Join):
        ls = child.children[0].schema
        rs = child.children[1].schema
        lnames: Set[str] = set(ls.names())
        rout = dict(child.right_passthrough())  # src -> out
        out2rsrc = {v: k for k, v in rout.items()}
        lp_preds, rp_preds, kept = [], [], []
        for p in preds:
            refs = set(p.column_refs())
            if refs <= lnames and child.how in ("inner", "left", "semi",
                                                "anti"):
                lp_preds.append(p)
            elif child.how in ("inner", "right") and \
                    all(r in out2rsrc or r in set(rs.names()) and
                        r not in lnames for r in refs):
                sub = _substitute_cols(
                    p, {o: ColumnRef(s) for o, s in out2rsrc.items()})
                rp_preds.append(sub)
            else:
                kept.append(p)
        if lp_preds or rp_preds:
            lc = child.children[0]
            rc = child.children[1]
            if lp_preds:
                lc = lp.Filter(lc, _conjoin(lp_preds))
            if rp_preds:
                rc = lp.Filter(rc, _conjoin(rp_preds))
            new_join = child.with_children([lc, rc])
            if kept:
                return lp.Filter(new_join, _conjoin(kept))
            return new_join
        return None

    if isinstance(child, lp.ScanSource) and child.pushdown_filter is None:
        # push the full residual predicate into the scan (still re-applied
        # by the scan operator; used for row-group pruning)
        if not _has_udf(plan.predicate):
            new_scan = lp.ScanSource(
                child._full_schema, child.paths, child.file_format,
                child.storage_options, child.pushdown_columns,
                plan.predicate, child.pushdown_limit, child.read_options)
            return lp.Filter(new_scan, plan.predicate)
        return None
    return None


def derive_or_implications_rule(plan):
    """Single-pass rule (NOT in a fixed-point batch: the derived conjuncts
    get pushed away from this node and would be re-derived forever):
    `(a=x and ..) or (a=y and ..)` implies `a IN (x, y)`, which CAN cross
    joins the OR itself cannot (q7/q19)."""
    if not isinstance(plan, lp.Filter):
        return None
    preds = _split_conjunctions(plan.predicate)
    have = {repr(p) for p in preds}
    extra = [e for e in _derive_or_implications(preds)
             if repr(e) not in have]
    if extra:
        return lp.Filter(plan.children[0], _conjoin(preds + extra))
    return None


def push_down_anti_semi_join(plan: lp.LogicalPlan) -> Optional[lp.LogicalPlan]:
    """Push a semi/anti join below an inner join (or filter) on its probe
    side, so key-existence filtering happens before the fan-out join —
    gives q18 the orders-first reduction from the spec subquery.
    (ref: rules/push_down_anti_semi_join.rs)  Also drops a redundant
    Distinct on the semi/anti build side (semi joins are set-semantics
    already)."""
    if not (isinstance(plan, lp.Join) and plan.how in ("semi", "anti")):
        return None
    left, q = plan.children
    if isinstance(q, lp.Distinct) and q.subset is None:
        return plan.with_children([left, q.children[0]])
    key_refs = set()
    for e in plan.left_on:
        key_refs.update(e.column_refs())
    if isinstance(left, lp.Filter):
        # only hop over the filter when an inner join underneath can host
        # the semi — otherwise this rule and filter pushdown swap the two
        # nodes forever (observed: batch-1 fixed-point cap on q21)
        below = left.children[0]
        while isinstance(below, lp.Filter):
            below = below.children[0]
        rout_vals = set(dict(below.right_passthrough()).values()) \
            if isinstance(below, lp.Join) else set()
        if isinstance(below, lp.Join) and below.how == "inner" and \
                key_refs and (
                key_refs <= set(below.children[0].schema.names()) or
                key_refs <= rout_vals):
            return lp.Filter(
                plan.with_children([left.children[0], q]), left.predicate)
        return None
    if isinstance(left, lp.Join) and left.how == "inner":
        a, b = left.children
        if key_refs and key_refs <= set(a.schema.names()):
            new_a = lp.Join(a, q, plan.left_on, plan.right_on, plan.how,
                            plan.suffix, plan.prefix)
            return left.with_children([new_a, b])
        rout = dict(left.right_passthrough())      # src -> out
        out2src = {v: k for k, v in rout.items()}
        if key_refs and all(r in out2src for r in key_refs):
            new_on = [_substitute_cols(
                e, {o: ColumnRef(s) for o, s in out2src.items()})
                for e in plan.left_on]
            new_b = lp.Join(b, q, new_on, plan.right_on, plan.how,
                            plan.suffix, plan.prefix)
            return left.with_children([a, new_b])
    return None


def drop_repartition(plan: lp.LogicalPlan) -> Optional[lp.LogicalPlan]:
    if isinstance(plan, lp.Repartition) and \
            isinstance(plan.children[0], lp.Repartition):
        return plan.with_children([plan.children[0].children[0]])
    return None


# ---------------------------------------------------------------------------
# projection pushdown (column pruning)
# ---------------------------------------------------------------------------

def _required_columns(plan: lp.LogicalPlan) -> List[Optional[Set[str]]]:
    """Columns each child must provide (None = all)."""
    if isinstance(plan, lp.Project):
        req: Set[str] = set()
        for e in plan.exprs:
            req.update(e.column_refs())
        return [req]
    if isinstance(plan, lp.UDFProject):
        req = set(plan.udf_expr.column_refs())
        for e in plan.passthrough:
            req.update(e.column_refs())
        return [req]
    if isinstance(plan, lp.Filter):
        return [None]  # filter passes everything through
    if isinstance(plan, lp.Aggregate):
        req = set()
        for e in plan.groupby + plan.aggs:
            req.update(e.column_refs())
        return [req]
    if isinstance(plan, lp.Join):
        lreq = {e.to_field(plan.children[0].schema).name
                for e in plan.left_on if True}
        lreq = set()
        for e in plan.left_on:
            lreq.update(e.column_refs())
        rreq = set()
        for e in plan.right_on:
            rreq.update(e.column_refs())
        return [None, None]  # refined by the Project-over-Join rule below
    return [None for _ in plan.children]


def push_down_projection(plan: lp.LogicalPlan) -> Optional[lp.LogicalPlan]:
    """Prune columns at ScanSource/Source boundaries and through joins when a
    Project sits above."""
    if isinstance(plan, (lp.Project, lp.Aggregate)):
        needed: Set[str] = set()
        exprs = plan.exprs if isinstance(plan, lp.Project) else \
            plan.groupby + plan.aggs
        for e in exprs:
            needed.update(e.column_refs())
        child = plan.children[0]
        pruned = _prune(child, needed)
        if pruned is not None:
            return plan.with_children([pruned])
    if isinstance(plan, lp.Filter):
        # filters don't prune (they pass through), but combined
        # Filter(Project) handled by filter pushdown
        return None
    return None


def _prune(node: lp.LogicalPlan, needed: Set[str]) -> Optional[lp.LogicalPlan]:
    """Try to narrow `node`'s output to `needed` columns; returns a new node
    or None if nothing changed."""
    out_names = node.schema.names()
    if set(out_names) <= needed:
        return None
    keep = [n for n in out_names if n in needed]
    if not keep:
        keep = out_names[:1]

    if isinstance(node, lp.ScanSource):
        if node.pushdown_columns is not None and \
                set(node.pushdown_columns) <= set(keep):
            return None
        # keep filter-referenced columns in the scan output
        fkeep = list(keep)
        if node.pushdown_filter is not None:
            for r in node.pushdown_filter.column_refs():
                if r not in fkeep:
                    fkeep.append(r)
        return lp.ScanSource(node._full_schema, node.paths, node.file_format,
                             node.storage_options, fkeep,
                             node.pushdown_filter, node.pushdown_limit,
                             node.read_options)
    if isinstance(node, lp.Source):
        # narrow INSIDE the source: columns are dropped before partitions
        # move to the device (matters for host-resident / streamed tables)
        cols = [f.name for f in node._full_schema if f.name in keep]
        if node.columns is not None and list(node.columns) == cols:
            return None
        return lp.Source(node._full_schema, node.cache_key, node.num_rows,
                         node.size_bytes, node.partitioning, cols)
    if isinstance(node, lp.Project):
        new_exprs = [e for e in node.exprs
                     if e.to_field(node.children[0].schema).name in keep]
        if len(new_exprs) == len(node.exprs):
            return None
        inner_needed = set()
        for e in new_exprs:
            inner_needed.update(e.column_refs())
        inner = _prune(node.children[0], inner_needed)
        return lp.Project(inner if inner is not None else node.children[0],
                          new_exprs)
    if isinstance(node, lp.Filter):
        inner_needed = set(needed)
        inner_needed.update(node.predicate.column_refs())
        inner = _prune(node.children[0], inner_needed)
        if inner is not None:
            return lp.Filter(inner, node.predicate)
        return None
    if isinstance(node, lp.Join):
        ls = node.children[0].schema
        rs_map = dict(node.right_passthrough())  # src -> out
        lneeded = {n for n in needed if n in set(ls.names())}
        for e in node.left_on:
            lneeded.update(e.column_refs())
        rneeded = {src for src, out in rs_map.items() if out in needed}
        for e in node.right_on:
            rneeded.update(e.column_refs())
        lc = _prune(node.children[0], lneeded)
        rc = _prune(node.children[1], rneeded)
        if lc is None and rc is None:
            return None
        return node.with_children([lc or node.children[0],
                                   rc or node.children[1]])
    if isinstance(node, (lp.Sort, lp.Limit, lp.Repartition, lp.TopN,
                         lp.Sample, lp.IntoBatches)):
        inner_needed = set(needed)
        for e in getattr(node, "by", []):
            inner_needed.update(e.column_refs())
        inner = _prune(node.children[0], inner_needed)
        if inner is not None:
            return node.with_children([inner])
        return None
    return None


# ---------------------------------------------------------------------------
# limit pushdown + TopN
# ---------------------------------------------------------------------------

def push_down_limit(plan: lp.LogicalPlan) -> Optional[lp.LogicalPlan]:
    if not isinstance(plan, lp.Limit):
        return None
    child = plan.children[0]
    total = plan.limit + plan.offset
    if isinstance(child, lp.Limit):
        lo = min(child.limit, plan.limit)
        return lp.Limit(child.children[0], lo, plan.offset + child.offset)
    if isinstance(child, lp.Project):
        return lp.Project(lp.Limit(child.children[0], plan.limit,
                                   plan.offset), child.exprs)
    if isinstance(child, lp.ScanSource):
        if child.pushdown_limit is None or child.pushdown_limit > total:
            if child.pushdown_filter is None:
                new_scan = lp.ScanSource(
                    child._full_schema, child.paths, child.file_format,
                    child.storage_options, child.pushdown_columns,
                    child.pushdown_filter, total, child.read_options)
                return lp.Limit(new_scan, plan.limit, plan.offset)
    return None


def rewrite_topn(plan: lp.LogicalPlan) -> Optional[lp.LogicalPlan]:
    if isinstance(plan, lp.Limit) and isinstance(plan.children[0], lp.Sort):
        s = plan.children[0]
        return lp.TopN(s.children[0], s.by, s.descending, s.nulls_first,
                       plan.limit, plan.offset)
    return None
