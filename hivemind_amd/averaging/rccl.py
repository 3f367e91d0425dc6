"""RCCL-over-xGMI data plane for averaging groups that live inside one
``torch.distributed`` world.

This is the MI355X-native replacement for the reference's per-peer TCP
butterfly (``hivemind/averaging/allreduce.py``) when the matched group is
exactly the local 8-GPU node: one process per GPU, ``torch.distributed`` with
the nccl backend (RCCL on ROCm). A weighted average is computed as

    x_i  <-  sum_j (w_j / W) * x_j      (W = sum of weights)

by pre-scaling the local flat buckets with ``w_i / W`` and running a bucketed
``all_reduce(SUM)`` -- RCCL's ring reduce-scatter+all-gather saturates the 7
xGMI links per GPU, which is the same communication pattern the reference's
butterfly emulates over TCP.

Buckets default to 64 MiB: large enough to amortize RCCL launch overhead,
small enough to pipeline scale/cast work with communication. On GPU, the
collectives run on a dedicated side stream so averaging overlaps compute.
Optionally casts buckets to bf16/fp16 on the wire (halves xGMI bytes).

Fault semantics: a RCCL collective is all-or-nothing (SURVEY.md §7 "hard
parts"), so this path is only chosen when every group member is a rank of the
same healthy world; WAN/elastic peers take the RPC butterfly instead.
"""

from __future__ import annotations

import threading
from typing import List, Optional, Sequence

import torch
import torch.distributed as dist

from ..utils.logging import get_logger

logger = get_logger(__name__)

DEFAULT_BUCKET_BYTES = 64 * 1024 * 1024

# collectives from different averagers in one process must not interleave
_COLLECTIVE_LOCK = threading.Lock()


def distributed_world_info() -> Optional[tuple]:
    """(world_size, rank) if torch.distributed is initialized, else None."""
    if dist.is_available() and dist.is_initialized():
        return (dist.get_world_size(), dist.get_rank())
    return None


def group_matches_world(group_dist_info: Sequence[Optional[tuple]]) -> bool:
    """True if the assembled group is exactly the ranks 0..W-1 of our world."""
    me = distributed_world_info()
    if me is None:
        return False
    world_size = me[0]
    if len(group_dist_info) != world_size:
        return False
    ranks = set()
    for info in group_dist_info:
        if info is None or len(info) != 2 or info[0] != world_size:
            return False
        ranks.add(info[1])
    return ranks == set(range(world_size))


class DistributedAllReduceRunner:
    """Weighted in-place average of a tensor list over torch.distributed.

    Works on the gloo backend for CPU tests and on nccl (=RCCL) for the
    8xMI355X data plane; same semantics either way.
    """

    def __init__(
        self,
        tensors: Sequence[torch.Tensor],
        weight: float,
        *,
        process_group: Optional[dist.ProcessGroup] = None,
        bucket_size_bytes: int = DEFAULT_BUCKET_BYTES,
        wire_dtype: Optional[torch.dtype] = None,
        averaging_alpha: float = 1.0,
    ):
        self.tensors = list(tensors)
        self.weight = weight
        self.process_group = process_group
        self.bucket_size_bytes = bucket_size_bytes
        self.wire_dtype = wire_dtype
        self.averaging_alpha = averaging_alpha

    def _gather_total_weight(self, device: torch.device) -> float:
        """All ranks must agree on the weight denominator: all-gather the live
        per-rank weights (they may have changed after matchmaking)."""
        backend_device = device if (device.type == "cuda") else torch.device("cpu")
        w = torch.tensor([self.weight], dtype=torch.float64, device=backend_device)
        world = dist.get_world_size(self.process_group)
        gathered = [torch.zeros_like(w) for _ in range(world)]
        dist.all_gather(gathered, w, group=self.process_group)
        return float(sum(g.item() for g in gathered))

    def run(self) -> None:
        """Execute the bucketed weighted all-reduce, in place (blocking).

        The whole round (weight gather + buckets) runs under the process-wide
        collective lock: collectives from two averagers must never interleave
        on this rank. NOTE (cross-rank ordering): rounds on different process
        groups must be INITIATED in the same order on every rank -- the
        Optimizer guarantees this by awaiting each averaging round before
        starting the next; background-overlapped rounds (DPU) rely on the
        pre-scheduled matchmaking making every rank trigger rounds in epoch
        order."""
        device = self.tensors[0].device if self.tensors else torch.device("cpu")
        use_side_stream = device.type == "cuda"
        with _COLLECTIVE_LOCK:
            total_weight = self._gather_total_weight(device)
            scale = self.weight / total_weight if total_weight > 0 else 0.0
            if use_side_stream:
                stream = _get_side_stream(device)
                stream.wait_stream(torch.cuda.current_stream(device))
                ctx = torch.cuda.stream(stream)
            else:
                ctx = _NullCtx()
            # gloo lacks reduced-precision collectives; wire casting is an
            # xGMI-bandwidth optimization for the nccl(=RCCL) backend only
            try:
                backend = dist.get_backend(self.process_group)
            except Exception:
                backend = None
            wire_dtype = self.wire_dtype if str(backend) == "nccl" else None
            with ctx:
                for bucket_tensors in self._iter_buckets():
                    flat = torch.cat([t.detach().reshape(-1) for t in bucket_tensors])
                    work_dtype = wire_dtype or flat.dtype
                    flat_scaled = flat.to(work_dtype)
                    flat_scaled.mul_(scale)
                    dist.all_reduce(flat_scaled, op=dist.ReduceOp.SUM, group=self.process_group)
                    averaged = flat_scaled.to(flat.dtype)
                    offset = 0
                    for tensor in bucket_tensors:
                        n = tensor.numel()
                        avg_part = averaged[offset : offset + n].view_as(tensor)
                        # .data: with reuse_tensors the target IS a live model
                        # parameter, and a backward pass may be in flight on
                        # another thread (DPU) -- bypass the version counter
                        if self.averaging_alpha == 1.0:
                            tensor.data.copy_(avg_part)
                        else:
                            tensor.data.add_(avg_part - tensor.data, alpha=self.averaging_alpha)
                        offset += n
            if use_side_stream:
                torch.cuda.current_stream(device).wait_stream(stream)

    def _iter_buckets(self):
        bucket: List[torch.Tensor] = []
        bucket_bytes = 0
        for tensor in self.tensors:
            t_bytes = tensor.numel() * tensor.element_size()
            if bucket and bucket_bytes + t_bytes > self.bucket_size_bytes:
                yield bucket
                bucket, bucket_bytes = [], 0
            bucket.append(tensor)
            bucket_bytes += t_bytes
        if bucket:
            yield bucket


class _NullCtx:
    def __enter__(self):
        return None

    def __exit__(self, *args):
        return False


_side_streams = {}


def _get_side_stream(device: torch.device) -> "torch.cuda.Stream":
    key = device.index
    if key not in _side_streams:
        _side_streams[key] = torch.cuda.Stream(device)
    return _side_streams[key]
