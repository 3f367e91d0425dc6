"""TrainingAverager (legacy API): averages params/grads/optimizer stats in step().

Parity target: reference ``hivemind/optim/training_averager.py:18-252`` -- the
older all-in-one averager kept for API compatibility; new code should use
TrainingStateAverager + GradientAverager. Applies the "old tensor" delta
correction: local updates made while the (possibly slow) averaging round ran
are preserved (new = old_local + (averaged - snapshot)).
"""

from __future__ import annotations

import threading
from typing import Optional, Sequence

import torch

from ..averaging import DecentralizedAverager
from ..dht import DHT
from ..utils.logging import get_logger

logger = get_logger(__name__)


class TrainingAverager(DecentralizedAverager):
    def __init__(
        self,
        opt: torch.optim.Optimizer,
        *,
        average_parameters: bool,
        average_gradients: bool,
        average_opt_statistics: Sequence[str] = (),
        extra_tensors: Sequence[torch.Tensor] = (),
        initialize_optimizer: bool = True,
        **kwargs,
    ):
        self.opt = opt
        self.opt_statistics = tuple(average_opt_statistics)
        self.average_parameters, self.average_gradients = average_parameters, average_gradients
        self.lock_averager_step = threading.Lock()
        self.local_step = 0
        if initialize_optimizer:
            from .state_averager import initialize_optimizer_state_

            initialize_optimizer_state_(opt)
        averaged_tensors = [tensor.detach().clone().float() for tensor in self.local_tensors()]
        super().__init__(averaged_tensors=averaged_tensors, **kwargs)

    def local_tensors(self) -> list:
        """Tensors to average, in schema order (reference training_averager.py:94-113)."""
        local = []
        if self.average_parameters:
            for group in self.opt.param_groups:
                local.extend(p for p in group["params"])
        if self.average_gradients:
            for group in self.opt.param_groups:
                for p in group["params"]:
                    if p.grad is None:
                        p.grad = torch.zeros_like(p)
                    local.append(p.grad)
        for stat in self.opt_statistics:
            for group in self.opt.param_groups:
                for p in group["params"]:
                    local.append(self.opt.state[p][stat])
        return local

    @torch.no_grad()
    def step(self, data_lock: Optional[threading.Lock] = None, wait: bool = True, **kwargs):
        """Average with peers, preserving concurrent local updates
        (reference training_averager.py:115-180)."""
        if not wait:
            return threading.Thread(target=self.step, kwargs=dict(data_lock=data_lock, **kwargs), daemon=True).start()
        if data_lock is None:
            data_lock = _NULL_LOCK
        with self.lock_averager_step:
            with data_lock, self.get_tensors() as averaged_tensors:
                local_tensors = self.local_tensors()
                assert len(local_tensors) == len(averaged_tensors)
                for av, loc in zip(averaged_tensors, local_tensors):
                    av.copy_(loc.to(av.dtype), non_blocking=True)
                old_local = [loc.detach().clone() for loc in local_tensors]
            gathered = super().step(**kwargs)
            if gathered is not None:
                with data_lock, self.get_tensors() as averaged_tensors:
                    for av, loc, old in zip(averaged_tensors, self.local_tensors(), old_local):
                        # delta correction: keep updates made during the round
                        loc.add_((av.to(loc.dtype) - old.to(loc.dtype)))
            self.local_step += 1
            return gathered


class _NullLock:
    def __enter__(self):
        return self

    def __exit__(self, *args):
        return False


_NULL_LOCK = _NullLock()
