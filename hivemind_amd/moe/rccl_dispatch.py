"""Intra-node expert parallelism over RCCL all-to-all (xGMI data plane).

The RPC MoE stack (moe/client + moe/server) is the hivemind-parity path:
experts discovered via DHT, called over the fault-tolerant transport -- right
for WAN swarms, but on one 8xMI355X node it routes activations through
loopback sockets. This module is the MI355X-native expression of the
reference's expert dispatch ("MoE rpc_forward/backward ...
RCCL equivalent: ncclSend/Recv grouped alltoallv across the 8 GPUs"):

* each rank (GPU) hosts ``num_local_experts`` experts of a global pool of
  ``world * num_local_experts``;
* tokens pick top-k experts via a linear gate; (token, k) pairs are grouped
  by destination rank and exchanged with ONE variable-split all-to-all per
  direction -- RCCL implements it as grouped send/recv loading all 7 xGMI
  links at once;
* collectives go through ``torch.distributed.nn.functional`` so autograd
  runs the reverse all-to-all in backward; expert gradients stay local to
  the owning rank (true expert parallelism, no gradient sync for experts).

Works on gloo for CPU tests; single-rank worlds degrade to a local top-k MoE
(the all-to-all becomes a copy), which is also the GPU unit-test mode.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F
import torch.distributed as dist

from ..utils.logging import get_logger

logger = get_logger(__name__)


class _LocalFFNExpert(nn.Module):
    """Default expert: 4x GELU FFN (reference moe/server/layers/common.py:18-30)."""

    def __init__(self, hidden_dim: int, dtype: torch.dtype = torch.float32):
        super().__init__()
        self.up = nn.Linear(hidden_dim, 4 * hidden_dim, dtype=dtype)
        self.down = nn.Linear(4 * hidden_dim, hidden_dim, dtype=dtype)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.down(F.gelu(self.up(x), approximate="tanh"))


class RcclMixtureOfExperts(nn.Module):
    """Top-k mixture of experts with all-to-all dispatch across one
    torch.distributed world (one process per GPU over RCCL/xGMI).

    Drop-in for a transformer FFN block: ``forward(x[N, hidden])`` returns
    ``[N, hidden]``. Every rank must call forward the same number of times
    (collective); uneven token counts per rank are fine (variable splits).
    """

    def __init__(
        self,
        hidden_dim: int,
        num_local_experts: int = 1,
        k: int = 2,
        expert_factory=None,
        process_group: Optional["dist.ProcessGroup"] = None,
        dtype: torch.dtype = torch.float32,
    ):
        super().__init__()
        self.hidden_dim = hidden_dim
        self.num_local_experts = num_local_experts
        self.process_group = process_group
        if dist.is_available() and dist.is_initialized():
            self.world_size = dist.get_world_size(process_group)
            self.rank = dist.get_rank(process_group)
        else:
            self.world_size, self.rank = 1, 0
        self.num_experts = self.world_size * num_local_experts
        self.k = min(k, self.num_experts)
        factory = expert_factory or (lambda: _LocalFFNExpert(hidden_dim, dtype=dtype))
        self.experts = nn.ModuleList([factory() for _ in range(num_local_experts)])
        self.gate = nn.Linear(hidden_dim, self.num_experts, bias=False, dtype=dtype)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        assert x.dim() == 2 and x.shape[1] == self.hidden_dim, "expected [tokens, hidden]"
        n_tokens = x.shape[0]
        scores = self.gate(x)  # [N, E]
        top_scores, top_experts = scores.topk(self.k, dim=-1)
        gate_weights = torch.softmax(top_scores.float(), dim=-1).to(x.dtype)  # [N, k]

        flat_expert = top_experts.reshape(-1)  # [N*k]
        flat_token = torch.arange(n_tokens, device=x.device).repeat_interleave(self.k)
        flat_weight = gate_weights.reshape(-1)

        # group (token, expert) pairs by destination rank, then by expert --
        # sorting by global expert id achieves both (experts are rank-major)
        order = torch.argsort(flat_expert, stable=True)
        send_experts = flat_expert[order]
        send_tokens = flat_token[order]
        send_x = x.index_select(0, send_tokens)  # [N*k, hidden], grouped

        counts_per_expert = torch.bincount(send_experts, minlength=self.num_experts)
        send_splits = counts_per_expert.reshape(self.world_size, self.num_local_experts).sum(1)

        if self.world_size > 1:
            import torch.distributed.nn.functional as dF

            recv_splits = torch.empty_like(send_splits)
            dist.all_to_all_single(recv_splits, send_splits.contiguous(), group=self.process_group)
            send_list = send_splits.tolist()
            recv_list = recv_splits.tolist()
            n_recv = sum(recv_list)
            recv_x = x.new_empty(n_recv, self.hidden_dim)
            recv_x = dF.all_to_all_single(
                recv_x, send_x.contiguous(), output_split_sizes=recv_list,
                input_split_sizes=send_list, group=self.process_group,
            )
            # expert ids ride a plain (non-differentiable) all-to-all
            recv_experts = send_experts.new_empty(n_recv)
            dist.all_to_all_single(
                recv_experts, send_experts.contiguous(), output_split_sizes=recv_list,
                input_split_sizes=send_list, group=self.process_group,
            )
        else:
            recv_x, recv_experts = send_x, send_experts

        # run local experts on their token groups
        local_ids = recv_experts - self.rank * self.num_local_experts
        expert_out = torch.zeros_like(recv_x)
        for local_id, expert in enumerate(self.experts):
            mask = local_ids == local_id
            if bool(mask.any()):
                idx = mask.nonzero(as_tuple=True)[0]
                expert_out = expert_out.index_copy(0, idx, expert(recv_x.index_select(0, idx)))

        if self.world_size > 1:
            import torch.distributed.nn.functional as dF

            back_x = x.new_empty(send_x.shape[0], self.hidden_dim)
            back_x = dF.all_to_all_single(
                back_x, expert_out.contiguous(), output_split_sizes=send_list,
                input_split_sizes=recv_list, group=self.process_group,
            )
        else:
            back_x = expert_out

        # combine: weighted sum of each token's k expert outputs
        out = x.new_zeros(n_tokens, self.hidden_dim)
        out = out.index_add(0, send_tokens, back_x * flat_weight[order].unsqueeze(1))
        return out
