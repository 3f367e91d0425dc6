"""TaskPool: batches independent requests for one expert call type.

Parity target: reference ``hivemind/moe/server/task_pool.py:59-256``. The
reference runs each pool as a forked process moving tensors through
shared-memory pipes because its server is a process tree; this framework's
server is one process per GPU, so a pool is a thread-safe queue feeding the
Runtime thread directly -- same batching semantics (min/max batch size,
priority = timestamp of the earliest undispatched task), zero IPC copies.
"""

from __future__ import annotations

import concurrent.futures
import threading
import time
from collections import deque
from typing import Any, Callable, List, NamedTuple, Sequence, Tuple

import torch

from ...utils.logging import get_logger

logger = get_logger(__name__)


class Task(NamedTuple):
    args: Tuple[torch.Tensor, ...]
    future: concurrent.futures.Future
    timestamp: float


class TaskPoolBase:
    pass


class TaskPool(TaskPoolBase):
    def __init__(
        self,
        process_func: Callable[..., Sequence[torch.Tensor]],
        name: str,
        max_batch_size: int,
        min_batch_size: int = 1,
        daemon: bool = True,
    ):
        self.process_func = process_func
        self.name = name
        self.max_batch_size, self.min_batch_size = max_batch_size, min_batch_size
        self.tasks: deque = deque()
        self.work_available: threading.Condition = threading.Condition()
        self.total_submitted = 0
        self.total_processed = 0

    def attach_condition(self, condition: threading.Condition):
        """The Runtime shares one condition across pools to sleep on."""
        self.work_available = condition

    def submit_task(self, *args: torch.Tensor) -> concurrent.futures.Future:
        """Thread-safe: enqueue one task, return a future for its outputs."""
        future: concurrent.futures.Future = concurrent.futures.Future()
        task = Task(tuple(args), future, time.monotonic())
        with self.work_available:
            self.tasks.append(task)
            self.total_submitted += 1
            self.work_available.notify_all()
        return future

    @property
    def priority(self) -> float:
        """Timestamp of the earliest undispatched task (lower = more urgent)."""
        with self.work_available:
            return self.tasks[0].timestamp if self.tasks else float("inf")

    def __len__(self) -> int:
        return len(self.tasks)

    def load_batch(self) -> Tuple[List[Task], List[torch.Tensor]]:
        """Pop up to max_batch_size tasks and concatenate their inputs along dim 0.
        Caller must hold the pool lock via the Runtime condition."""
        batch: List[Task] = []
        total_size = 0
        while self.tasks and total_size < self.max_batch_size:
            next_size = self.tasks[0].args[0].shape[0] if self.tasks[0].args else 1
            if batch and total_size + next_size > self.max_batch_size:
                break
            task = self.tasks.popleft()
            batch.append(task)
            total_size += next_size
        if not batch:
            return [], []
        num_args = len(batch[0].args)
        batched = [torch.cat([task.args[i] for task in batch], dim=0) for i in range(num_args)]
        return batch, batched

    def send_outputs(self, batch: List[Task], outputs: Sequence[torch.Tensor]):
        """Split batched outputs back to per-task futures."""
        offset = 0
        for task in batch:
            size = task.args[0].shape[0] if task.args else 1
            task_outputs = tuple(out[offset : offset + size] for out in outputs)
            offset += size
            if not task.future.cancelled():
                task.future.set_result(task_outputs)
        self.total_processed += len(batch)

    def send_exception(self, batch: List[Task], exception: BaseException):
        for task in batch:
            if not task.future.cancelled():
                task.future.set_exception(exception)

    def __repr__(self):
        return f"TaskPool({self.name}, pending={len(self.tasks)}, processed={self.total_processed})"
