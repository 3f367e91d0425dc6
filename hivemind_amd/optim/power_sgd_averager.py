"""PowerSGD gradient averager: rank-r compression with error feedback.

Parity target: reference ``hivemind/optim/power_sgd_averager.py:28-222``.
Algorithm per round (Vogels et al., PowerSGD):

    m       += grad                      (error feedback accumulation)
    P        = M @ Q;     all-reduce P   (phase 1)
    P        = orthogonalize(P)
    Q        = M^T @ P;   all-reduce Q + uncompressed small tensors (phase 2)
    new_m    = P @ Q^T
    m       -= new_m                     (residual error kept locally)
    grad     = new_m

Tensors with < min_compression_ratio benefit (1-D biases etc.) ride
uncompressed in phase 2. The two chained rounds reuse one matched group with
distinct group-id suffixes (reference power_sgd_averager.py:23-26 phases).
On MI355X both phases go over RCCL when the group is the local world; the
P/Q GEMMs and Gram-Schmidt run on-GPU (torch -> rocBLAS).
"""

from __future__ import annotations

import asyncio
import contextlib
from enum import Enum
from typing import Any, Iterable, Optional, Sequence

import torch

from ..averaging.averager import GroupMetadata
from ..averaging.group_info import GroupInfo
from ..dht import DHT
from ..utils.asyncio_utils import enter_asynchronously
from ..utils.logging import get_logger
from ..utils.math_utils import get_flatten_greedy_dims, orthogonalize_
from .grad_averager import GradientAverager

logger = get_logger(__name__)


class AllReducePhases(Enum):
    PHASE_P = 1
    PHASE_Q = 2


class PowerSGDGradientAverager(GradientAverager):
    def __init__(
        self,
        parameters: Iterable[torch.nn.Parameter],
        averager_rank: int,
        *,
        dht: DHT,
        prefix: str,
        min_compression_ratio: float = 0.5,
        **kwargs,
    ):
        self.rank = averager_rank
        self.parameters = tuple(parameters)
        self._uncompressed_gradients_indexes = set(
            i
            for i, grad in enumerate(self._grads_from_parameters())
            if grad.ndim <= 1
            or (1 - self.rank * sum(get_flatten_greedy_dims(grad)) / grad.numel()) < min_compression_ratio
        )
        self._ms = [
            torch.zeros_like(grad, device="cpu" if not grad.is_cuda else grad.device)
            for idx, grad in enumerate(self._grads_from_parameters())
            if idx not in self._uncompressed_gradients_indexes
        ]
        self._qs = [
            torch.rand((get_flatten_greedy_dims(grad)[1], self.rank), device=grad.device)
            for idx, grad in enumerate(self._grads_from_parameters())
            if idx not in self._uncompressed_gradients_indexes
        ]
        super().__init__(self.parameters, dht=dht, prefix=prefix, **kwargs)

    @contextlib.contextmanager
    def _register_allreduce_group_ctx(self, group_info: GroupInfo, suffix: bytes):
        yield group_info.group_id + suffix

    async def _aggregate_tensors_with_group(self, group_info: GroupInfo, weight: float, meta: GroupMetadata):
        """Two chained rounds: P matrices, then Q matrices + uncompressed tensors
        (reference power_sgd_averager.py:132-188)."""
        async with enter_asynchronously(self.lock_averaged_tensors):
            averaged_grads = list(self._averaged_tensors)
            compressed_grads = [
                grad for idx, grad in enumerate(averaged_grads) if idx not in self._uncompressed_gradients_indexes
            ]
            uncompressed_grads = [
                grad for idx, grad in enumerate(averaged_grads) if idx in self._uncompressed_gradients_indexes
            ]

            # error feedback: m += grad
            for m, grad in zip(self._ms, compressed_grads):
                m.add_(grad.to(m.device, m.dtype))

            # phase P: P_i = M_i @ Q_i, all-reduce P
            ps = [
                torch.zeros((get_flatten_greedy_dims(grad)[0], self.rank), device=grad.device, dtype=torch.float32)
                for grad in compressed_grads
            ]
            for p, q, m in zip(ps, self._qs, self._ms):
                torch.matmul(m.reshape(-1, q.size(0)).to(torch.float32), q.to(torch.float32), out=p)
            await self._average_tensors_with_group(
                ps, group_info, weight, meta, group_id=group_info.group_id + b"::P", take_lock=False
            )
            for p in ps:
                orthogonalize_(p)

            # phase Q: Q_i = M_i^T @ P_i, all-reduce Q + uncompressed tensors
            for p, q, m in zip(ps, self._qs, self._ms):
                torch.matmul(m.reshape(-1, q.size(0)).t().to(torch.float32), p, out=q)
            phase_q_tensors = self._qs + [g for g in uncompressed_grads]
            await self._average_tensors_with_group(
                phase_q_tensors, group_info, weight, meta, group_id=group_info.group_id + b"::Q", take_lock=False
            )

            # reconstruct: new_m = P @ Q^T; error feedback m -= new_m; grad <- new_m
            for p, q, m, grad in zip(ps, self._qs, self._ms, compressed_grads):
                new_m = torch.matmul(p, q.t()).reshape(m.shape)
                m.sub_(new_m.to(m.device, m.dtype))
                grad.copy_(new_m.to(grad.device, grad.dtype))