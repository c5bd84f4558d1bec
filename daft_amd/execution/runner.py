"""Native (single-process, one-GPU) runner (ref:
/root/reference/daft/runners/native_runner.py:70-214 driving Swordfish
run.rs:332-407).  Optimize -> translate -> generator pipeline over
device-resident batches; events dispatched to subscribers."""
from __future__ import annotations

import time
import uuid
from typing import Iterator, List, Optional

from ..context import Context, Heartbeat
from ..logical.builder import LogicalPlanBuilder
from ..physical import translate
from ..physical.ops import ExecContext
from ..recordbatch import RecordBatch


class NativeRunner:
    name = "native"

    def __init__(self, ctx: Context):
        self.ctx = ctx

    def run_iter(self, builder: LogicalPlanBuilder,
                 device=None) -> Iterator[RecordBatch]:
        ctx = self.ctx
        query_id = uuid.uuid4().hex[:12]
        ctx.notify("on_query_start", query_id, builder.explain())
        t0 = time.perf_counter()
        err: Optional[str] = None
        try:
            with Heartbeat(ctx, query_id):
                topt = time.perf_counter()
                ctx.notify("on_optimization_start", query_id)
                optimized = builder.optimize()
                ctx.notify("on_optimization_end", query_id,
                           time.perf_counter() - topt)
                phys = translate(optimized.plan)
                ctx.notify("on_exec_start", query_id,
                           [l.strip("* ") for l in phys.explain_lines()])
                device = device or ctx.device()
                ectx = ExecContext(ctx, device, query_id)
                yield from phys.execute_tracked(ectx)
                for key, (rows, batches, secs) in ectx.stats.items():
                    ctx.notify("on_operator_end", query_id, key,
                               ectx.op_names.get(key, "?"), -1, rows, secs)
                ctx.notify("on_exec_end", query_id)
        except Exception as e:
            err = str(e)
            raise
        finally:
            ctx.notify("on_query_end", query_id,
                       time.perf_counter() - t0, err)

    def run(self, builder: LogicalPlanBuilder,
            device=None) -> List[RecordBatch]:
        return list(self.run_iter(builder, device=device))
