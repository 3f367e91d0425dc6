"""Optimizer wrappers for expert training (reference hivemind/moe/server/layers/optim.py)."""

from __future__ import annotations

import torch


class OptimizerWrapper(torch.optim.Optimizer):
    """Delegating base wrapper around an inner torch optimizer."""

    def __init__(self, optim: torch.optim.Optimizer):
        object.__setattr__(self, "optim", optim)

    @property
    def defaults(self):
        return self.optim.defaults

    @property
    def state(self):
        return self.optim.state

    @property
    def param_groups(self):
        return self.optim.param_groups

    def add_param_group(self, param_group: dict) -> None:
        return self.optim.add_param_group(param_group)

    def load_state_dict(self, state_dict: dict) -> None:
        return self.optim.load_state_dict(state_dict)

    def state_dict(self) -> dict:
        return self.optim.state_dict()

    def step(self, *args, **kwargs):
        return self.optim.step(*args, **kwargs)

    def zero_grad(self, *args, **kwargs):
        return self.optim.zero_grad(*args, **kwargs)

    def __repr__(self):
        return f"{self.__class__.__name__}({self.optim})"


class ClippingWrapper(OptimizerWrapper):
    """Clips gradient norm before every step (reference optim.py:47)."""

    def __init__(self, optim: torch.optim.Optimizer, clip_grad_norm: float):
        super().__init__(optim)
        object.__setattr__(self, "clip_grad_norm", clip_grad_norm)

    def step(self, *args, **kwargs):
        parameters = tuple(param for group in self.param_groups for param in group["params"])
        torch.nn.utils.clip_grad_norm_(parameters, self.clip_grad_norm)
        return super().step(*args, **kwargs)

    @classmethod
    def create(cls, optim_cls, *args, clip_grad_norm: float, **kwargs):
        return lambda params: cls(optim_cls(params, *args, **kwargs), clip_grad_norm)
