"""Runtime: the single thread that owns the GPU and drains task pools.

Parity target: reference ``hivemind/moe/server/runtime.py:22-199``: select the
non-empty pool whose earliest task is oldest, assemble a batch, run the pool's
process_func on the device, return outputs. The reference multiplexes pools
with ``selectors`` over pipes from forked pool processes; here pools are
in-process queues sharing one condition variable -- batches go straight from
the handler coroutines' tensors to the GPU.

A ``StatsReporter`` thread logs batches/s and examples/s per pool
(reference runtime.py:158-199).
"""

from __future__ import annotations

import threading
import time
from collections import defaultdict
from queue import SimpleQueue
from typing import Dict, Optional, Sequence

import torch

from ...utils.logging import get_logger
from .module_backend import ModuleBackend
from .task_pool import TaskPool

logger = get_logger(__name__)


class Runtime(threading.Thread):
    def __init__(
        self,
        module_backends: Dict[str, ModuleBackend],
        device: Optional[torch.device] = None,
        stats_report_interval: Optional[float] = None,
    ):
        super().__init__(name="moe-runtime", daemon=True)
        self.module_backends = module_backends
        self.pools: Sequence[TaskPool] = [pool for backend in module_backends.values() for pool in backend.get_pools()]
        self.device = device
        self.shutdown_requested = threading.Event()
        self.ready = threading.Event()
        self.work_condition = threading.Condition()
        for pool in self.pools:
            pool.attach_condition(self.work_condition)
        self.stats_report_interval = stats_report_interval
        self._stats_reporter: Optional[StatsReporter] = None

    def run(self):
        # move experts to the device once, then serve
        if self.device is not None:
            for backend in self.module_backends.values():
                backend.module.to(self.device)
        if self.stats_report_interval is not None:
            self._stats_reporter = StatsReporter(self.pools, self.stats_report_interval, self.shutdown_requested)
            self._stats_reporter.start()
        self.ready.set()
        logger.debug(f"runtime started with {len(self.pools)} pools on {self.device}")
        while not self.shutdown_requested.is_set():
            with self.work_condition:
                pool = self._choose_pool()
                if pool is None:
                    self.work_condition.wait(timeout=0.1)
                    continue
                batch, batched_inputs = pool.load_batch()
            if not batch:
                continue
            try:
                if self.device is not None:
                    batched_inputs = [t.to(self.device, non_blocking=True) for t in batched_inputs]
                outputs = pool.process_func(*batched_inputs)
                pool.send_outputs(batch, [t.detach() for t in outputs])
            except Exception as e:
                logger.warning(f"runtime: pool {pool.name} batch failed: {e!r}")
                pool.send_exception(batch, e)

    def _choose_pool(self) -> Optional[TaskPool]:
        """Earliest-undispatched-task priority (reference runtime.py:133-156)."""
        best, best_priority = None, float("inf")
        for pool in self.pools:
            if pool.tasks and pool.tasks[0].timestamp < best_priority:
                best, best_priority = pool, pool.tasks[0].timestamp
        return best

    def shutdown(self):
        self.shutdown_requested.set()
        with self.work_condition:
            self.work_condition.notify_all()
        self.join(timeout=5)


class StatsReporter(threading.Thread):
    def __init__(self, pools: Sequence[TaskPool], report_interval: float, stop_event: threading.Event):
        super().__init__(name="moe-stats", daemon=True)
        self.pools, self.report_interval, self.stop_event = pools, report_interval, stop_event
        self._last_processed: Dict[str, int] = defaultdict(int)

    def run(self):
        while not self.stop_event.wait(self.report_interval):
            for pool in self.pools:
                processed = pool.total_processed
                delta = processed - self._last_processed[pool.name]
                self._last_processed[pool.name] = processed
                if delta:
                    logger.info(
                        f"{pool.name}: {delta / self.report_interval:.1f} examples/s, pending={len(pool.tasks)}"
                    )
