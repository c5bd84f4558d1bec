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
    import torch.distributed as dist
    gathered: List[list] = [None] * comm.world()  # type: ignore
    dist.all_gather_object(gathered, counts)
    if any(len(g) != len(counts) for g in gathered):
        raise RuntimeError(
            "SPMD plans disagree on source count: "
            f"{[len(g) for g in gathered]}")
    for i, s in enumerate(sources):
        lp.GLOBAL_ROW_HINTS[s.cache_key] = sum(g[i] for g in gathered)


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
