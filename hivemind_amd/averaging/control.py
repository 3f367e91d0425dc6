"""StepControl: a cross-thread handle over one averaging step.

Parity target: reference ``hivemind/averaging/control.py:14-165``. The
reference backs this with a shared-memory buffer because its averager lives in
another process; here the averager runs on an event-loop thread of the same
process, so plain atomics + concurrent futures suffice (SURVEY.md §7 "hard
parts": keep ``step(wait=False) -> StepControl`` semantics without MPFuture).
"""

from __future__ import annotations

import concurrent.futures
import threading
from enum import Enum
from typing import Any, Optional


class AveragingStage(Enum):
    IDLE = 0  # still waiting for the trigger
    LOOKING_FOR_GROUP = 1  # running matchmaking
    AWAITING_TRIGGER = 2  # waiting for the user to allow all-reduce
    RUNNING_ALLREDUCE = 3
    FINISHED = 4


class StepControl:
    """Returned by ``DecentralizedAverager.step(wait=False)``; lets the caller
    adjust weight/scheduled time before the all-reduce begins, trigger it,
    cancel it, or block on the result."""

    def __init__(self, scheduled_time: float, deadline: float, allow_retries: bool, weight: float, gather_binary: bytes):
        self._lock = threading.Lock()
        self._scheduled_time = scheduled_time
        self._deadline = deadline
        self.allow_retries = allow_retries
        self._weight = weight
        self._gather_binary = gather_binary
        self._stage = AveragingStage.IDLE
        self._began_allreduce = False
        self._trigger = concurrent.futures.Future()  # set by allow_allreduce()
        self._result: concurrent.futures.Future = concurrent.futures.Future()
        # launch-order ticket for the RCCL data plane (rccl.CollectiveSequencer);
        # issued by the optimizer at scheduling time, released when the round ends
        self.rccl_ticket = None

    # ------------------------------------------------------------ attributes

    @property
    def scheduled_time(self) -> float:
        with self._lock:
            return self._scheduled_time

    @scheduled_time.setter
    def scheduled_time(self, value: float):
        with self._lock:
            if self._began_allreduce:
                raise RuntimeError("can't change scheduled time after all-reduce has begun")
            self._scheduled_time = value

    @property
    def weight(self) -> float:
        with self._lock:
            return self._weight

    @weight.setter
    def weight(self, value: float):
        assert value >= 0, "averaging weight must be non-negative"
        with self._lock:
            if self._began_allreduce:
                raise RuntimeError("can't change weight after all-reduce has begun")
            self._weight = value

    @property
    def deadline(self) -> float:
        return self._deadline

    @property
    def gather_binary(self) -> bytes:
        return self._gather_binary

    @property
    def stage(self) -> AveragingStage:
        return self._stage

    @stage.setter
    def stage(self, value: AveragingStage):
        self._stage = value
        if value == AveragingStage.RUNNING_ALLREDUCE:
            with self._lock:
                self._began_allreduce = True

    @property
    def began_allreduce(self) -> bool:
        return self._began_allreduce

    # --------------------------------------------------------------- trigger

    def allow_allreduce(self):
        """Let the averager proceed into all-reduce once a group is assembled."""
        if not self._trigger.done():
            self._trigger.set_result(None)

    trigger = allow_allreduce  # alias

    @property
    def triggered(self) -> bool:
        return self._trigger.done() and not self._trigger.cancelled()

    async def wait_for_trigger(self):
        import asyncio

        await asyncio.wrap_future(self._trigger)

    # ---------------------------------------------------------------- result

    def result(self, timeout: Optional[float] = None) -> Any:
        return self._result.result(timeout)

    def done(self) -> bool:
        return self._result.done()

    def set_result(self, value: Any):
        if not self._result.done():
            self._stage = AveragingStage.FINISHED
            self._result.set_result(value)

    def set_exception(self, exc: BaseException):
        if not self._result.done():
            self._stage = AveragingStage.FINISHED
            self._result.set_exception(exc)

    def cancel(self) -> bool:
        self._trigger.cancel()
        if not self._result.done():
            self._result.cancel()
            self._stage = AveragingStage.FINISHED
            return True
        return False

    def cancelled(self) -> bool:
        return self._result.cancelled()

    def exception(self, timeout: Optional[float] = None):
        return self._result.exception(timeout)

    def add_done_callback(self, callback):
        self._result.add_done_callback(callback)

    def attach(self, result_future: concurrent.futures.Future):
        self._result = result_future

    def __await__(self):
        import asyncio

        return asyncio.wrap_future(self._result).__await__()

    def __repr__(self):
        return f"StepControl(stage={self._stage.name}, weight={self._weight}, triggered={self.triggered})"
