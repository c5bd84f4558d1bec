"""HBM memory management: admission watermark + LRU spill of cached
partition sets to host memory (ref: the reference's MemoryManager permits,
daft-local-execution/src/resource_manager.rs:19-50 + DAFT_MEMORY_LIMIT, and
its flight_shuffle_dirs disk spill tier).

Round-1 scope: cached partition sets (collect() results, loaded tables)
spill from HBM to host when a blocking operator's admission would exceed
the limit; they transparently reload on next access (InMemorySourceOp moves
partitions back to the execution device).  Operator-internal spill is a
later-round item — with 288 GB of HBM per GPU, cached data dominates
footprint at the benchmarked scales.
"""
from __future__ import annotations

import threading
from typing import Optional

import torch


class MemoryManager:
    def __init__(self, ctx):
        self.ctx = ctx
        self._lock = threading.Lock()

    def limit_bytes(self, device) -> Optional[int]:
        cfg = self.ctx.execution_config.memory_limit_bytes
        if cfg:
            return cfg
        if str(device).startswith("cuda") and torch.cuda.is_available():
            total = torch.cuda.get_device_properties(device).total_memory
            return int(total * 0.92)
        return None

    def used_bytes(self, device) -> int:
        if str(device).startswith("cuda") and torch.cuda.is_available():
            return torch.cuda.memory_allocated(device)
        # host fallback: cached partition footprint
        return self.ctx.cache.total_bytes()

    def admit(self, nbytes: int, device) -> None:
        """Ensure `nbytes` more can be allocated; spill cached partition
        sets (LRU) off the device until it fits or nothing is left."""
        if not str(device).startswith("cuda"):
            return  # host memory is the spill target, not a spill source
        limit = self.limit_bytes(device)
        if limit is None:
            return
        with self._lock:
            while self.used_bytes(device) + nbytes > limit:
                freed = self.ctx.cache.spill_lru(device)
                if freed == 0:
                    break
            if str(device).startswith("cuda") and torch.cuda.is_available():
                torch.cuda.empty_cache()
