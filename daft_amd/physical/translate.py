"""Logical -> physical translation (ref:
/root/reference/src/daft-local-plan/src/translate.rs:21)."""
from __future__ import annotations

from ..logical import plan as lp
from . import ops


def translate(plan: lp.LogicalPlan) -> ops.PhysicalOp:
    ch = [translate(c) for c in plan.children]

    from ..distributed import plan_nodes as dn
    from ..distributed import ops as dops
    if isinstance(plan, dn.ExchangeByKey):
        return dops.ExchangeByKeyOp(ch[0], plan.keys)
    if isinstance(plan, dn.GatherToRank0):
        return dops.GatherToRank0Op(ch[0])
    if isinstance(plan, dn.ReplicateAll):
        return dops.ReplicateAllOp(ch[0])
    if isinstance(plan, dn.Rank0Only):
        return dops.Rank0OnlyOp(ch[0])
    if isinstance(plan, dn.RangeExchange):
        return dops.RangeExchangeOp(ch[0], plan.by, plan.descending,
                                    plan.nulls_first)

    if isinstance(plan, lp.Source):
        return ops.InMemorySourceOp(plan.schema, plan.cache_key,
                                     plan.columns)
    if isinstance(plan, lp.ScanSource):
        return ops.ScanOp(plan.schema, plan.paths, plan.file_format,
                          plan.storage_options, plan.read_options,
                          plan.pushdown_columns, plan.pushdown_filter,
                          plan.pushdown_limit)
    if isinstance(plan, lp.Project):
        return ops.ProjectOp(ch[0], plan.exprs, plan.schema)
    if isinstance(plan, lp.UDFProject):
        return ops.UDFProjectOp(ch[0], plan.udf_expr, plan.passthrough,
                                plan.schema)
    if isinstance(plan, lp.Filter):
        return ops.FilterOp(ch[0], plan.predicate)
    if isinstance(plan, lp.Limit):
        return ops.LimitOp(ch[0], plan.limit, plan.offset)
    if isinstance(plan, lp.Explode):
        return ops.ExplodeOp(ch[0], plan.exprs, plan.schema)
    if isinstance(plan, lp.Unpivot):
        return ops.UnpivotOp(ch[0], plan.ids, plan.values,
                             plan.variable_name, plan.value_name,
                             plan.schema)
    if isinstance(plan, lp.Sort):
        return ops.SortOp(ch[0], plan.by, plan.descending, plan.nulls_first)
    if isinstance(plan, lp.TopN):
        return ops.TopNOp(ch[0], plan.by, plan.descending, plan.nulls_first,
                          plan.limit, plan.offset)
    if isinstance(plan, lp.Repartition):
        return ops.RepartitionOp(ch[0], plan.scheme, plan.num_partitions,
                                 plan.by)
    if isinstance(plan, lp.Distinct):
        return ops.DistinctOp(ch[0], plan.subset)
    if isinstance(plan, lp.Aggregate):
        return ops.AggregateOp(ch[0], plan.groupby, plan.aggs, plan.schema)
    if isinstance(plan, lp.Pivot):
        return ops.PivotOp(ch[0], plan.groupby, plan.pivot_col,
                           plan.value_col, plan.agg_kind, plan.names,
                           plan.schema)
    if isinstance(plan, lp.Concat):
        return ops.ConcatOp(ch, plan.schema)
    if isinstance(plan, lp.AsofJoin):
        return ops.AsofJoinOp(ch[0], ch[1], plan.left_on, plan.right_on,
                              plan.left_by, plan.right_by, plan.strategy,
                              plan.schema, plan.right_passthrough())
    if isinstance(plan, lp.Join):
        return ops.JoinOp(ch[0], ch[1], plan.left_on, plan.right_on,
                          plan.how, plan.schema, plan.right_passthrough()
                          if plan.how not in ("semi", "anti") else [])
    if isinstance(plan, lp.Sample):
        return ops.SampleOp(ch[0], plan.fraction, plan.with_replacement,
                            plan.seed)
    if isinstance(plan, lp.MonotonicallyIncreasingId):
        from ..distributed import comm
        return ops.MonotonicIdOp(ch[0], plan.column_name, plan.schema,
                                 partition_id=comm.rank())
    if isinstance(plan, lp.IntoBatches):
        return ops.IntoBatchesOp(ch[0], plan.batch_size)
    if isinstance(plan, lp.Window):
        return ops.WindowOp(ch[0], plan.window_exprs, plan.partition_by,
                            plan.order_by, plan.descending, plan.names,
                            plan.schema)
    if isinstance(plan, lp.Sink):
        return ops.WriteOp(ch[0], plan.file_format, plan.root_dir,
                           plan.write_mode, plan.partition_cols,
                           plan.options, plan.schema)
    raise NotImplementedError(f"no physical translation for {plan.name()}")
