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
