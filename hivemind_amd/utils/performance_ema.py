"""Bias-corrected exponential moving average of throughput.

Parity target: reference ``hivemind/utils/performance_ema.py:7-70``.
"""

from __future__ import annotations

import time
from contextlib import contextmanager
from threading import Lock


class PerformanceEMA:
    """Estimate samples-per-second as a bias-corrected EMA over reported batches."""

    def __init__(self, alpha: float = 0.1, paused: bool = False):
        self.alpha = alpha
        self.interval = 0.0
        self.ema_seconds_per_sample = 0.0
        self.num_updates = 0
        self.samples_per_second = 1e-9
        self.timestamp = time.perf_counter()
        self.paused = paused
        self.lock = Lock()

    def update(self, task_size: float, interval: float | None = None) -> float:
        assert task_size > 0, "task size must be positive"
        if interval is None:
            now = time.perf_counter()
            interval = 0.0 if self.paused else now - self.timestamp
            self.timestamp = now
        self.interval += interval
        if self.interval > 0:
            seconds_per_sample = self.interval / task_size
            self.ema_seconds_per_sample = (
                self.alpha * seconds_per_sample + (1 - self.alpha) * self.ema_seconds_per_sample
            )
            self.num_updates += 1
            adjusted = self.ema_seconds_per_sample / (1 - (1 - self.alpha) ** self.num_updates)
            self.samples_per_second = 1 / max(adjusted, 1e-20)
        self.interval = 0.0
        return self.samples_per_second

    def reset_timer(self):
        self.timestamp = time.perf_counter()

    @contextmanager
    def pause(self):
        """Ignore time while paused (e.g. while waiting for the network)."""
        self.paused, was_paused = True, self.paused
        try:
            yield
        finally:
            self.paused = was_paused
            self.reset_timer()

    @contextmanager
    def update_threadsafe(self, task_size: float):
        with self.lock:
            yield
            self.update(task_size, interval=None)

    def __repr__(self):
        return f"{self.__class__.__name__}({self.samples_per_second:.3f} samples/s, {self.num_updates} updates)"
