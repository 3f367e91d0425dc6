"""Switch-Transformer style routing: one expert per sample.

Parity target: reference ``hivemind/moe/client/switch_moe.py:17-225``:
``k_best=1`` routing with multiplicative jitter noise on the inputs to the
gate, per-dimension grid dropout, an EMA of expert utilization, and the
load-balancing auxiliary loss (mean utilization x mean routing probability).
"""

from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import torch
import torch.nn as nn

from ...utils.logging import get_logger
from ...utils.nested import nested_flatten, nested_pack
from ..expert_uid import UID_DELIMITER
from .expert import DUMMY, RemoteExpert, create_remote_experts
from .moe import RemoteMixtureOfExperts, _RemoteCallMany

logger = get_logger(__name__)


class RemoteSwitchMixtureOfExperts(RemoteMixtureOfExperts):
    def __init__(
        self,
        *,
        grid_size: Sequence[int],
        utilization_alpha: float = 0.9,
        grid_dropout: float = 1.0,
        jitter_eps: float = 1e-2,
        k_best: int = 1,
        k_min: int = 0,
        backward_k_min: int = 0,
        allow_zero_outputs: bool = True,
        **kwargs,
    ):
        super().__init__(
            grid_size=grid_size,
            k_best=k_best,
            k_min=k_min,
            backward_k_min=backward_k_min,
            allow_zero_outputs=allow_zero_outputs,
            **kwargs,
        )
        self.utilization_alpha = utilization_alpha
        self.grid_dropout = grid_dropout
        self.jitter_eps = jitter_eps
        initial_utilization = torch.cat([torch.full((dim,), 1.0 / dim) for dim in grid_size])
        self.register_buffer("grid_utilization", initial_utilization)

    def forward(self, input: torch.Tensor, *args, **kwargs):
        """Route each sample to its top-1 expert (reference switch_moe.py:74-139)."""
        if input.ndim != 2:
            input_for_gating = input.mean(dim=tuple(range(1, input.ndim - 1)))
        else:
            input_for_gating = input

        # multiplicative jitter (training only)
        if self.training and self.jitter_eps:
            input_for_gating = input_for_gating * torch.empty_like(input_for_gating).uniform_(
                1 - self.jitter_eps, 1 + self.jitter_eps
            )

        grid_scores = self.proj(input_for_gating).split_with_sizes(list(self.beam_search.grid_size), dim=-1)
        grid_softmax = [torch.softmax(dim_scores, dim=-1) for dim_scores in grid_scores]
        grid_dropout_masks = (
            [
                torch.rand(size=(dim_size,), device=input.device) < self.grid_dropout
                for dim_size in self.beam_search.grid_size
            ]
            if self.training
            else [torch.ones(d, dtype=torch.bool, device=input.device) for d in self.beam_search.grid_size]
        )
        grid_scores_dropout = [
            torch.where(mask, scores, torch.full((1,), float("-inf"), device=scores.device, dtype=scores.dtype))
            for scores, mask in zip(grid_scores, grid_dropout_masks)
        ]

        chosen_experts: List[List[RemoteExpert]] = [
            [e for e in create_remote_experts(sample_infos, self.dht) if e is not None]
            for sample_infos in self.beam_search.batch_find_best_experts(
                [[ds[i].detach().cpu().tolist() for ds in grid_scores_dropout] for i in range(len(input))],
                self.k_best,
            )
        ]
        if self._expert_info is None:
            for experts in chosen_experts:
                for expert in experts:
                    try:
                        self._expert_info = expert.info
                        break
                    except Exception:
                        continue
                if self._expert_info is not None:
                    break
        if self._expert_info is None:
            raise RuntimeError("no alive experts found to infer the I/O schema from")

        expert_mask, *expert_outputs = _RemoteCallMany.apply(
            DUMMY,
            chosen_experts,
            self.k_min,
            self.backward_k_min,
            self.timeout_after_k_min,
            self.forward_timeout,
            self.backward_timeout,
            self.detect_anomalies,
            self.allow_zero_outputs,
            self.info,
            self.dht.loop,
            *nested_flatten(((input, *args), kwargs)),
        )

        expert_probs = self.compute_expert_scores(grid_softmax, chosen_experts)
        masked_probs = torch.zeros((1,), device=expert_probs.device, dtype=expert_probs.dtype)
        expert_probs = torch.where(expert_mask, expert_probs, masked_probs)
        averaged_outputs_flat = [
            (expert_probs[..., None] * tensor.flatten(start_dim=2)).view(tensor.shape).sum(dim=1)
            for tensor in expert_outputs
        ]
        packed_outputs = nested_pack(averaged_outputs_flat, self.info["outputs_schema"])

        if self.training:
            self._update_utilization(grid_softmax)
        balancing_loss = self._compute_balancing_loss(grid_softmax, grid_dropout_masks)
        return (packed_outputs, balancing_loss) if isinstance(packed_outputs, torch.Tensor) else (packed_outputs, balancing_loss)

    def compute_expert_scores(
        self, grid_probs: Sequence[torch.Tensor], batch_experts: List[List[RemoteExpert]]
    ) -> torch.Tensor:
        """Product of per-dimension routing probabilities (reference switch_moe.py:141-176)."""
        batch_size = len(batch_experts)
        max_k = max((len(experts) for experts in batch_experts), default=1)
        scores = torch.zeros((batch_size, max_k), device=grid_probs[0].device, dtype=grid_probs[0].dtype)
        prefix_len = len(self.beam_search.uid_prefix)
        for i, experts in enumerate(batch_experts):
            for j, expert in enumerate(experts):
                coords = [int(x) for x in expert.uid[prefix_len:].strip(UID_DELIMITER).split(UID_DELIMITER)]
                prob = torch.ones((), device=scores.device, dtype=scores.dtype)
                for d, coord in enumerate(coords):
                    prob = prob * grid_probs[d][i, coord]
                scores[i, j] = prob
        return scores

    @torch.no_grad()
    def _update_utilization(self, grid_softmax: Sequence[torch.Tensor]):
        batch_utilization = torch.cat([probs.mean(0).detach() for probs in grid_softmax])
        self.grid_utilization.mul_(self.utilization_alpha).add_(
            batch_utilization.to(self.grid_utilization.device), alpha=1 - self.utilization_alpha
        )

    def _compute_balancing_loss(self, grid_softmax, masks) -> torch.Tensor:
        """num_experts * sum(mean_probability * utilization) per dim (Switch eq. 4)."""
        loss = torch.zeros((), device=grid_softmax[0].device, dtype=grid_softmax[0].dtype)
        offset = 0
        for probs, dim_size in zip(grid_softmax, self.beam_search.grid_size):
            util = self.grid_utilization[offset : offset + dim_size].to(probs.device, probs.dtype)
            loss = loss + dim_size * torch.sum(probs.mean(0) * util)
            offset += dim_size
        return loss
