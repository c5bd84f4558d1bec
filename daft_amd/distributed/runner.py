"""Distributed SPMD runner: every rank (one per GPU, RCCL over xGMI)
executes the distributed-rewritten plan over its shard.

Replaces the reference's RayRunner/Flotilla control plane
(/root/reference/daft/runners/ray_runner.py, flotilla.py): the scheduler is
torch.distributed's process group; exchanges are collectives, not object
stores."""
from __future__ import annotations

import time
import uuid
from typing import Iterator, List

from ..context import Context, Heartbeat
from ..logical.builder import LogicalPlanBuilder
from ..physical import translate
from ..physical.ops import ExecContext
from ..recordbatch import RecordBatch
from . import comm
from .planner import distribute


def _sync_source_stats(plan) -> None:
    """Agree on GLOBAL per-source row counts before optimization.

    The optimizer's cost decisions (join reordering, build-side choice)
    must be identical on every rank or the collective schedules diverge
    and the job deadlocks; rank-local shard sizes differ, so each query
    starts with one tiny all_gather that sums every in-memory source's
    rows across ranks into logical.plan.GLOBAL_ROW_HINTS."""
    from ..logical import plan as lp

    # cache keys are rank-local UUIDs: sources must be matched by POSITION
    # in the (identical-by-construction) plan tree, in DFS order
    sources: List = []

    def walk(p):
        if isinstance(p, lp.Source):
            sources.append(p)
        for c in p.children:
            walk(c)
    walk(plan)
    if not sources:
        return
    counts = [s.num_rows for s in sources]
    # sample-NDV per column for join-cost estimation (lazy computation is
    # forbidden under SPMD: local samples differ across ranks).  Cached by
    # cache_key so each table is sampled once per process.
    from ..optimizer.stats import NDV_HINTS, SAMPLE_ROWS, sample_ndv
    from ..context import get_context
    ndvs: List[dict] = []
    for s in sources:
        if s.cache_key in getattr(_sync_source_stats, "_seen", set()):
            ndvs.append({})
            continue
        per_col: dict = {}
        try:
            parts = get_context().cache.get(s.cache_key)
            big = max(parts, key=len) if parts else None
            if big is not None and len(big):
                for c in big.columns:
                    if c.pyobjs is None:
                        per_col[c.name] = sample_ndv(c, n_rows=len(big))
        except Exception:
            per_col = {}
        ndvs.append(per_col)
    import torch.distributed as dist
    gathered: List[list] = [None] * comm.world()  # type: ignore
    dist.all_gather_object(gathered, [counts, ndvs])
    if any(len(g[0]) != len(counts) for g in gathered):
        raise RuntimeError(
            "SPMD plans disagree on source count: "
            f"{[len(g[0]) for g in gathered]}")
    seen = getattr(_sync_source_stats, "_seen", set())
    for i, s in enumerate(sources):
        lp.GLOBAL_ROW_HINTS[s.cache_key] = sum(g[0][i] for g in gathered)
        if s.cache_key in seen:
            continue
        seen.add(s.cache_key)
        cols = set()
        for g in gathered:
            cols.update(g[1][i].keys())
        for cname in cols:
            locs = [g[1][i].get(cname, 0.0) for g in gathered]
            mx = max(locs)
            if mx <= SAMPLE_ROWS * 0.1:
                # category-like everywhere: shards repeat the same values
                comb = mx
            else:
                comb = min(float(lp.GLOBAL_ROW_HINTS[s.cache_key]),
                           float(sum(locs)))
            NDV_HINTS[(s.cache_key, cname)] = comb
    _sync_source_stats._seen = seen


class DistributedRunner:
    name = "distributed"

    def __init__(self, ctx: Context, backend=None):
        self.ctx = ctx

    def run_iter(self, builder: LogicalPlanBuilder,
                 device=None) -> Iterator[RecordBatch]:
        ctx = self.ctx
        query_id = uuid.uuid4().hex[:12]
        ctx.notify("on_query_start", query_id, builder.explain())
        t0 = time.perf_counter()
        err = None
        try:
            if comm.world() > 1:
                _sync_source_stats(builder.plan)
            optimized = builder.optimize()
            plan = distribute(optimized.plan, comm.world(), comm.rank())
            phys = translate(plan)
            device = device or ctx.device()
            ectx = ExecContext(ctx, device, query_id)
            yield from phys.execute(ectx)
        except Exception as e:
            err = str(e)
            raise
        finally:
            ctx.notify("on_query_end", query_id,
                       time.perf_counter() - t0, err)

    def run(self, builder: LogicalPlanBuilder,
            device=None) -> List[RecordBatch]:
        return list(self.run_iter(builder, device=device))
