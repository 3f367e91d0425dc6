"""AMP GradScaler aware of delayed/offloaded optimizer steps.

Parity target: reference ``hivemind/optim/grad_scaler.py:25-127``: unscale
only happens inside the global (swarm-synchronized) step; ``update`` runs once
per global step; inf/nan gradients make ``step`` a no-op that reports failure
so the caller can discard the round.
"""

from __future__ import annotations

import contextlib
import threading
from typing import Optional

import torch

from ..utils.logging import get_logger

logger = get_logger(__name__)


class GradScaler(torch.amp.GradScaler):
    def __init__(self, *args, **kwargs):
        kwargs.setdefault("device", "cuda" if torch.cuda.is_available() else "cpu")
        super().__init__(*args, **kwargs)
        self._is_running_global_step = False
        self._is_ready_to_update = False
        self._inner_optimizer_states = {}
        self._lock = threading.RLock()

    @contextlib.contextmanager
    def running_global_step(self):
        """Mark the region where the swarm-wide optimizer step happens."""
        with self._lock:
            previous, self._is_running_global_step = self._is_running_global_step, True
            try:
                yield
            finally:
                self._is_running_global_step = previous

    def unscale_(self, optimizer: torch.optim.Optimizer) -> bool:
        with self._lock:
            if not self._is_running_global_step:
                # accumulation-only steps never unscale (reference grad_scaler.py:59-66)
                return False
            super().unscale_(optimizer)
            return True

    def step(self, optimizer: torch.optim.Optimizer, *args, **kwargs) -> bool:
        with self._lock:
            if not self._is_running_global_step:
                # ordinary (accumulation) steps don't run the inner optimizer at all
                return False
            if self.are_grads_finite(optimizer, use_cached=False):
                state = self._per_optimizer_states.get(id(optimizer))
                if state is not None and not state["found_inf_per_device"]:
                    pass
                super().step(optimizer, *args, **kwargs)
                self._is_ready_to_update = True
                return True
            logger.warning("skipping global step: gradients contain inf/nan")
            self._is_ready_to_update = True
            return False

    def update(self, new_scale: Optional[float] = None) -> bool:
        with self._lock:
            if self._is_ready_to_update:
                super().update(new_scale)
                self._is_ready_to_update = False
                return True
            return False

    def are_grads_finite(self, optimizer: torch.optim.Optimizer, use_cached: bool = False) -> bool:
        state = self._per_optimizer_states[id(optimizer)]
        if use_cached and state["found_inf_per_device"]:
            found = state["found_inf_per_device"]
        else:
            found = self._check_inf_per_device(optimizer)
        return not sum(v.item() for v in found.values())
