"""Distributed planning: rewrite an optimized logical plan for SPMD
execution across N ranks (one per GPU), inserting RCCL exchange nodes.

This is the MI355X-native replacement for the reference's Flotilla
distributed planner (/root/reference/src/daft-distributed/src/pipeline_node/
translate.rs:130-688): instead of a task scheduler shipping plan fragments to
per-node workers, every rank executes the SAME plan over its shard, with
hash / range / gather exchanges over xGMI at the partition boundaries.
Partitioning states: sharded | hash(keys) | range | rank0.
"""
from __future__ import annotations

from typing import List, Optional, Set, Tuple

from ..expressions.expressions import (Agg, Alias, Cast, ColumnRef, ExprNode)
from ..logical import plan as lp
from ..physical.agg_partial import split_partial_final
from ..schema import supertype
from . import plan_nodes as dn

SHARDED = ("sharded",)
RANK0 = ("rank0",)
_RCCL = "__rccl_hash"  # token of planner-inserted hash exchanges


def _hash_state(names: List[str], token: str = _RCCL):
    return ("hash", tuple(sorted(names)), token) if names else SHARDED


def _key_names(exprs: List[ExprNode], schema) -> Optional[List[str]]:
    names = []
    for e in exprs:
        base = e.child if isinstance(e, Alias) else e
        if isinstance(base, ColumnRef):
            names.append(base.name)
        else:
            return None
    return names


def distribute(plan: lp.LogicalPlan, world: int,
               rank: int) -> lp.LogicalPlan:
    node, _state = _rewrite(plan, world, rank)
    return node


def _rewrite(node: lp.LogicalPlan, world: int,
             rank: int) -> Tuple[lp.LogicalPlan, tuple]:
    if isinstance(node, lp.Source):
        if node.partitioning:
            token, keys = node.partitioning
            return node, _hash_state(list(keys), token)
        return node, SHARDED
    if isinstance(node, lp.ScanSource):
        # split files across ranks
        paths = node.paths[rank::world]
        shard = lp.ScanSource(node._full_schema, paths, node.file_format,
                              node.storage_options, node.pushdown_columns,
                              node.pushdown_filter, node.pushdown_limit,
                              node.read_options)
        return shard, SHARDED

    ch = [_rewrite(c, world, rank) for c in node.children]
    kids = [c[0] for c in ch]
    states = [c[1] for c in ch]
    node = node.with_children(kids)

    # elementwise / streaming ops preserve partitioning
    if isinstance(node, (lp.Filter, lp.Explode, lp.Unpivot, lp.Sample,
                         lp.IntoBatches, lp.UDFProject)):
        return node, states[0]
    if isinstance(node, (lp.Project,)):
        st = states[0]
        if st[0] == "hash":
            out_names = set(node.schema.names())
            if not set(st[1]) <= out_names:
                st = SHARDED
        return node, st
    if isinstance(node, lp.MonotonicallyIncreasingId):
        return node, states[0]
    if isinstance(node, lp.Concat):
        return node, SHARDED

    if isinstance(node, lp.Aggregate):
        return _rewrite_aggregate(node, states[0])
    if isinstance(node, lp.Distinct):
        child = node.children[0]
        keys = node.subset if node.subset else \
            [ColumnRef(n) for n in child.schema.names()]
        knames = _key_names(keys, child.schema)
        st = states[0]
        if st[0] == "hash" and knames and set(st[1]) <= set(knames):
            return node, st
        ex = dn.ExchangeByKey(child, keys)
        return lp.Distinct(ex, node.subset), _hash_state(knames or [])

    if isinstance(node, lp.Join):
        return _rewrite_join(node, states)
    if isinstance(node, lp.AsofJoin):
        left, right = node.children
        if node.left_by:
            lc = dn.ExchangeByKey(left, [ColumnRef(c)
                                         for c in node.left_by])
            rc = dn.ExchangeByKey(right, [ColumnRef(c)
                                          for c in node.right_by])
            return node.with_children([lc, rc]), \
                _hash_state(list(node.left_by))
        rep = dn.ReplicateAll(right)
        return node.with_children([left, rep]), states[0]

    if isinstance(node, lp.Sort):
        ex = dn.RangeExchange(node.children[0], node.by, node.descending,
                              node.nulls_first)
        return lp.Sort(ex, node.by, node.descending, node.nulls_first), \
            ("range",)
    if isinstance(node, lp.TopN):
        local = lp.TopN(node.children[0], node.by, node.descending,
                        node.nulls_first, node.limit + node.offset, 0)
        g = dn.GatherToRank0(local)
        return lp.TopN(g, node.by, node.descending, node.nulls_first,
                       node.limit, node.offset), RANK0
    if isinstance(node, lp.Limit):
        if states[0] == RANK0:
            return node, RANK0
        local = lp.Limit(node.children[0], node.limit + node.offset, 0)
        g = dn.GatherToRank0(local)
        return lp.Limit(g, node.limit, node.offset), RANK0

    if isinstance(node, lp.Window):
        child = node.children[0]
        if node.partition_by:
            knames = _key_names(node.partition_by, child.schema)
            st = states[0]
            if not (st[0] == "hash" and knames and
                    set(st[1]) <= set(knames)):
                child = dn.ExchangeByKey(child, node.partition_by)
            return node.with_children([child]), \
                _hash_state(knames or [])
        g = dn.GatherToRank0(child)
        return node.with_children([g]), RANK0

    if isinstance(node, lp.Pivot):
        child = node.children[0]
        if node.groupby:
            child = dn.ExchangeByKey(child, node.groupby)
        else:
            child = dn.GatherToRank0(child)
        return node.with_children([child]), SHARDED

    if isinstance(node, lp.Repartition):
        if node.scheme == "hash" and node.by:
            ex = dn.ExchangeByKey(node.children[0], node.by)
            knames = _key_names(node.by, node.children[0].schema)
            return ex, _hash_state(knames or [])
        return node, states[0]

    if isinstance(node, lp.Sink):
        return node, states[0]

    # distributed nodes reached via recursion of our own wrappers
    return node, states[0] if states else SHARDED


def _rewrite_aggregate(node: lp.Aggregate, child_state: tuple):
    child = node.children[0]
    cschema = child.schema
    if node.groupby:
        knames = _key_names(node.groupby, cschema)
        already = child_state[0] == "hash" and knames and \
            set(child_state[1]) <= set(knames)
        split = split_partial_final(node.aggs)
        if split is not None and not already:
            partials, final_named, residuals = split
            partial = lp.Aggregate(child, node.groupby, partials)
            gnames = [e.to_field(cschema).name for e in node.groupby]
            ex = dn.ExchangeByKey(partial, [ColumnRef(n) for n in gnames])
            final = lp.Aggregate(ex, [ColumnRef(n) for n in gnames],
                                 final_named)
            proj = lp.Project(final,
                              [ColumnRef(n) for n in gnames] + residuals)
            return proj, _hash_state(gnames)
        if not already:
            ex = dn.ExchangeByKey(child, node.groupby)
            return node.with_children([ex]), _hash_state(knames or [])
        return node, child_state
    # ungrouped
    split = split_partial_final(node.aggs)
    if split is not None:
        partials, final_named, residuals = split
        partial = lp.Aggregate(child, [], partials)
        g = dn.GatherToRank0(partial)
        final = lp.Aggregate(g, [], final_named)
        proj = lp.Project(final, list(residuals))
        return dn.Rank0Only(proj), RANK0
    g = dn.GatherToRank0(child)
    return dn.Rank0Only(node.with_children([g])), RANK0


_BROADCAST_ROWS = 4_000_000  # replicate build side below this estimate


def _rewrite_join(node: lp.Join, states):
    left, right = node.children
    if node.how == "cross":
        rep = dn.ReplicateAll(right)
        return node.with_children([left, rep]), states[0]

    lschema, rschema = left.schema, right.schema
    lnames = _key_names(node.left_on, lschema)
    rnames = _key_names(node.right_on, rschema)

    # align key dtypes so both sides hash identically
    lkeys, rkeys = [], []
    for le, re_ in zip(node.left_on, node.right_on):
        lt = le.to_field(lschema).dtype
        rt = re_.to_field(rschema).dtype
        st = supertype(lt, rt)
        lkeys.append(le if lt == st else Cast(le, st))
        rkeys.append(re_ if rt == st else Cast(re_, st))

    est = right.approx_num_rows()
    if est is not None and est <= _BROADCAST_ROWS and \
            node.how in ("inner", "left", "semi", "anti"):
        # broadcast join: replicate small build side, keep probe sharded
        rep = dn.ReplicateAll(right)
        return node.with_children([left, rep]), states[0]

    lst, rst = states
    l_ok = lst[0] == "hash" and lnames and set(lst[1]) == set(lnames)
    r_ok = rst[0] == "hash" and rnames and set(rst[1]) == set(rnames)
    same_dist = l_ok and r_ok and lst[2] == rst[2]
    lc, rc = left, right
    if not same_dist:
        # both sides must land under the SAME distribution; reuse one side's
        # planner hash partitioning when present
        if l_ok and lst[2] == _RCCL:
            rc = dn.ExchangeByKey(right, rkeys)
        elif r_ok and rst[2] == _RCCL:
            lc = dn.ExchangeByKey(left, lkeys)
        else:
            lc = dn.ExchangeByKey(left, lkeys)
            rc = dn.ExchangeByKey(right, rkeys)
    out_state = ("hash", tuple(sorted(lnames)), lst[2] if same_dist
                 else _RCCL) if lnames else SHARDED
    return node.with_children([lc, rc]), out_state
