"""RemoteMixtureOfExperts: gate -> beam search -> fault-tolerant fan-out.

Parity target: reference ``hivemind/moe/client/moe.py:25-442``:
a linear gate projects inputs to per-dimension grid scores; beam search picks
``k_best`` experts per sample; ``_RemoteCallMany`` fans out per-sample RPC
calls tolerating failures (each sample needs >= ``k_min`` responses; after
that, stragglers get ``timeout_after_k_min``); dead experts are masked with
-inf before the softmax that mixes expert outputs; backward is similarly
fault-tolerant with ``backward_k_min``.
"""

from __future__ import annotations

import asyncio
from typing import Any, Dict, List, Optional, Sequence, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...dht import DHT
from ...utils.logging import get_logger
from ...utils.nested import nested_flatten, nested_pack
from ..expert_uid import UID_DELIMITER
from .beam_search import MoEBeamSearcher
from .expert import DUMMY, RemoteExpert, create_remote_experts, expert_backward, expert_forward
from .remote_expert_worker import RemoteExpertWorker

logger = get_logger(__name__)


class RemoteMixtureOfExperts(nn.Module):
    """A torch module mixing outputs of experts distributed across the swarm."""

    def __init__(
        self,
        *,
        in_features: int,
        grid_size: Sequence[int],
        dht: DHT,
        uid_prefix: str,
        k_best: int,
        k_min: int = 1,
        forward_timeout: Optional[float] = None,
        timeout_after_k_min: Optional[float] = None,
        backward_k_min: int = 1,
        backward_timeout: Optional[float] = None,
        detect_anomalies: bool = False,
        allow_zero_outputs: bool = False,
        **dht_kwargs,
    ):
        super().__init__()
        self.dht = dht
        self.beam_search = MoEBeamSearcher(dht, uid_prefix, grid_size, **dht_kwargs)
        self.k_best, self.k_min, self.backward_k_min = k_best, k_min, backward_k_min
        self.forward_timeout, self.backward_timeout = forward_timeout or 30.0, backward_timeout or 30.0
        self.timeout_after_k_min = timeout_after_k_min if timeout_after_k_min is not None else 1.0
        self.detect_anomalies = detect_anomalies
        self.allow_zero_outputs = allow_zero_outputs
        self.proj = nn.Linear(in_features, sum(grid_size))
        self._expert_info: Optional[Dict[str, Any]] = None

    def forward(self, input: torch.Tensor, *args: torch.Tensor, **kwargs: torch.Tensor):
        """Average outputs of the best experts (reference moe.py:77-139)."""
        if input.ndim != 2:
            input_for_gating = input.mean(dim=tuple(range(1, input.ndim - 1)))
        else:
            input_for_gating = input
        grid_scores = self.proj(input_for_gating).split_with_sizes(list(self.beam_search.grid_size), dim=-1)

        chosen_experts: List[List[RemoteExpert]] = [
            [e for e in create_remote_experts(sample_infos, self.dht) if e is not None]
            for sample_infos in self.beam_search.batch_find_best_experts(
                [[dim_scores[i].detach().cpu().tolist() for dim_scores in grid_scores] for i in range(len(input))],
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
        expert_logits = self.compute_expert_scores(grid_scores, chosen_experts)
        masked_logits = torch.full((1,), float("-inf"), device=expert_logits.device, dtype=expert_logits.dtype)
        expert_logits = torch.where(expert_mask, expert_logits, masked_logits)
        expert_weights = torch.softmax(expert_logits, dim=1)
        averaged_outputs_flat = [
            (expert_weights[..., None] * tensor.flatten(start_dim=2)).view(tensor.shape).sum(dim=1)
            for tensor in expert_outputs
        ]
        return nested_pack(averaged_outputs_flat, self.info["outputs_schema"])

    def compute_expert_scores(
        self, grid_scores: Sequence[torch.Tensor], batch_experts: List[List[RemoteExpert]]
    ) -> torch.Tensor:
        """Sum of per-dimension gate scores for each chosen expert (reference moe.py:141-178)."""
        batch_size = len(batch_experts)
        max_k = max((len(experts) for experts in batch_experts), default=1)
        scores = torch.full((batch_size, max_k), float("-inf"), device=grid_scores[0].device, dtype=grid_scores[0].dtype)
        prefix_len = len(self.beam_search.uid_prefix)
        for i, experts in enumerate(batch_experts):
            for j, expert in enumerate(experts):
                coords = [int(x) for x in expert.uid[prefix_len:].strip(UID_DELIMITER).split(UID_DELIMITER)]
                scores[i, j] = sum(grid_scores[d][i, coord] for d, coord in enumerate(coords))
        return scores

    @property
    def info(self) -> Dict[str, Any]:
        assert self._expert_info is not None
        return self._expert_info


class _RemoteCallMany(torch.autograd.Function):
    """Per-sample fan-out with failure tolerance (reference moe.py:192-428)."""

    @staticmethod
    def forward(
        ctx,
        dummy: torch.Tensor,
        experts_per_sample: List[List[RemoteExpert]],
        k_min: int,
        backward_k_min: int,
        timeout_after_k_min: float,
        forward_timeout: float,
        backward_timeout: float,
        detect_anomalies: bool,
        allow_zero_outputs: bool,
        info: Dict[str, Any],
        loop,
        *flat_inputs: torch.Tensor,
    ):
        num_samples, max_experts = len(experts_per_sample), max(len(e) for e in experts_per_sample)

        if detect_anomalies:
            for t in flat_inputs:
                if not torch.isfinite(t).all():
                    raise ValueError("One of the inputs has nan/inf values")

        async def _forward_all():
            tasks = {}
            for i, experts in enumerate(experts_per_sample):
                sample_inputs = [t[i : i + 1] for t in flat_inputs]
                for j, expert in enumerate(experts):
                    coro = expert_forward(expert.uid, expert.stub, sample_inputs)
                    tasks[asyncio.ensure_future(coro)] = (i, j)
            return await _collect_responses(tasks, num_samples, k_min, timeout_after_k_min, forward_timeout)

        results = RemoteExpertWorker.run_coroutine(_forward_all(), loop=loop)

        if detect_anomalies:
            # treat experts that returned non-finite outputs as dead
            bad = [pair for pair, outs in results.items()
                   if any(not torch.isfinite(o).all() for o in outs)]
            for pair in bad:
                logger.warning(f"dropping expert response {experts_per_sample[pair[0]][pair[1]].uid}: nonfinite output")
                del results[pair]

        alive_counts = [0] * num_samples
        for (i, _j) in results:
            alive_counts[i] += 1
        if not allow_zero_outputs and min(alive_counts) < k_min:
            raise TimeoutError(
                f"forward: some samples got fewer than k_min={k_min} expert responses ({alive_counts})"
            )

        # assemble [num_samples, max_experts, ...] padded outputs + mask
        outputs_schema = info["outputs_schema"]
        flat_out_schemas = list(nested_flatten(outputs_schema))
        mask = torch.zeros(num_samples, max_experts, dtype=torch.bool)
        stacked_outputs = []
        for out_idx, schema in enumerate(flat_out_schemas):
            example = None
            for (i, j), outs in results.items():
                example = outs[out_idx]
                break
            if example is None:
                shape_tail = tuple(int(s) for s in schema.shape[1:])
                example = torch.zeros(1, *shape_tail)
            stacked = torch.zeros(num_samples, max_experts, *example.shape[1:], dtype=example.dtype)
            for (i, j), outs in results.items():
                stacked[i, j] = outs[out_idx][0]
            stacked_outputs.append(stacked)
        for (i, j) in results:
            mask[i, j] = True

        ctx.save_for_backward(*flat_inputs)
        ctx._saved = dict(
            experts_per_sample=experts_per_sample,
            alive_pairs=set(results.keys()),
            backward_k_min=backward_k_min,
            backward_timeout=backward_timeout,
            timeout_after_k_min=timeout_after_k_min,
            info=info,
            loop=loop,
            num_samples=num_samples,
            detect_anomalies=detect_anomalies,
        )
        return (mask, *(t.requires_grad_(True) for t in stacked_outputs))

    @staticmethod
    @torch.autograd.function.once_differentiable
    def backward(ctx, _grad_mask, *grad_outputs_stacked):
        flat_inputs = ctx.saved_tensors
        saved = ctx._saved
        experts_per_sample = saved["experts_per_sample"]
        alive_pairs = saved["alive_pairs"]
        loop = saved["loop"]
        num_samples = saved["num_samples"]

        async def _backward_all():
            tasks = {}
            for (i, j) in alive_pairs:
                expert = experts_per_sample[i][j]
                sample_inputs = [t[i : i + 1].detach() for t in flat_inputs]
                sample_grads = [g[i, j].unsqueeze(0).contiguous() for g in grad_outputs_stacked]
                coro = expert_backward(expert.uid, expert.stub, [*sample_inputs, *sample_grads])
                tasks[asyncio.ensure_future(coro)] = (i, j)
            return await _collect_responses(
                tasks, num_samples, saved["backward_k_min"], saved["timeout_after_k_min"], saved["backward_timeout"]
            )

        if saved["detect_anomalies"]:
            for g in grad_outputs_stacked:
                if not torch.isfinite(g).all():
                    raise ValueError("One of the gradients has nan/inf values")

        results = RemoteExpertWorker.run_coroutine(_backward_all(), loop=loop)
        grad_inputs = [torch.zeros_like(t) for t in flat_inputs]
        for (i, _j), grads in results.items():
            if saved["detect_anomalies"] and any(not torch.isfinite(g).all() for g in grads):
                continue  # dead reducer: exclude nonfinite expert gradients
            for gi, g in zip(grad_inputs, grads[: len(grad_inputs)]):
                gi[i : i + 1] += g.to(gi.dtype)
        return (DUMMY, None, None, None, None, None, None, None, None, None, None, *grad_inputs)


async def _collect_responses(
    tasks: Dict[asyncio.Future, Tuple[int, int]],
    num_samples: int,
    k_min: int,
    timeout_after_k_min: float,
    timeout_total: float,
) -> Dict[Tuple[int, int], Any]:
    """Wait until every sample has >= k_min responses, then give stragglers
    timeout_after_k_min more seconds (reference moe.py:371-428)."""
    loop = asyncio.get_event_loop()
    t_start = loop.time()
    t_reached_k_min: Optional[float] = None
    responded: Dict[Tuple[int, int], Any] = {}
    per_sample = [0] * num_samples
    pending = set(tasks.keys())
    while pending:
        now = loop.time()
        remaining_total = timeout_total - (now - t_start)
        if remaining_total <= 0:
            break
        if t_reached_k_min is not None:
            remaining = min(remaining_total, timeout_after_k_min - (now - t_reached_k_min))
            if remaining <= 0:
                break
        else:
            remaining = remaining_total
        done, pending = await asyncio.wait(pending, timeout=remaining, return_when=asyncio.FIRST_COMPLETED)
        if not done:
            break
        for task in done:
            i, j = tasks[task]
            try:
                responded[(i, j)] = task.result()
                per_sample[i] += 1
            except Exception as e:
                logger.debug(f"expert call ({i},{j}) failed: {e!r}")
        if t_reached_k_min is None and all(c >= k_min for c in per_sample):
            t_reached_k_min = loop.time()
    for task in pending:
        task.cancel()
    return responded
