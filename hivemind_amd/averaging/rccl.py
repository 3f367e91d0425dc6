"""RCCL-over-xGMI data plane for averaging groups whose members are ranks of
one ``torch.distributed`` world.

This is the MI355X-native replacement for the reference's per-peer TCP
butterfly (``hivemind/averaging/allreduce.py``) when the matched group lives
inside the local node: one process per GPU, ``torch.distributed`` with the
nccl backend (RCCL on ROCm). A weighted average is computed as

    x_i  <-  sum_j (w_j / W) * x_j      (W = sum of weights)

by pre-scaling the local tensors with ``w_i / W`` and summing across the
group. Three wire formats:

* full precision -- bucketed ``all_reduce(SUM)``; RCCL's ring
  reduce-scatter+all-gather is per-xGMI-link bound (~153 GB/s);
* ``wire_dtype=bf16/fp16`` -- the buckets are cast before the collective
  (halves xGMI bytes);
* ``codec="blockwise_int8"`` -- a direct-send butterfly composed from
  all-to-all + all-gather: quantize (per-4096-block absmax int8, HIP kernel) ->
  all-to-all codes -> dequant-accumulate own slice -> requantize ->
  all-gather. RCCL cannot sum int8 codes, so the reduction runs as a HIP
  dequant+add on each slice owner -- exactly the reference's
  compress->send->reduce->send-back butterfly (SURVEY.md §2.4 C1,
  reference partition.py:104-112 + quantization.py:128-201), with the wire
  cost cut to ~2 bytes/element/round vs 8 for fp32 ring all-reduce.

Sub-world groups (Moshpit subgroups of the node) get a cached
``dist.new_group(ranks, use_local_synchronization=True)`` -- only the matched
members participate in communicator creation, which is what dynamic
matchmaking requires.

Cross-rank collective ordering: rounds from different averagers (grad/state,
possibly DPU-overlapped) MUST be launched in the same order on every rank or
two blocking collectives deadlock each other. ``CollectiveSequencer`` enforces
a process-local FIFO of *tickets* issued at round-scheduling time -- a point
that executes in the same program order on every rank (the optimizer's epoch
logic) -- so collectives launch in ticket order everywhere.

Fault semantics: a RCCL collective is all-or-nothing (SURVEY.md §7 "hard
parts"), so this plane is only chosen when every group member is a rank of
the same healthy world; WAN/elastic peers take the RPC butterfly instead.
"""

from __future__ import annotations

import math
import threading
from typing import Dict, List, Optional, Sequence, Tuple

import torch
import torch.distributed as dist

from ..utils.logging import get_logger

logger = get_logger(__name__)

DEFAULT_BUCKET_BYTES = 64 * 1024 * 1024
QUANT_BLOCK = 4096  # matches the HIP blockwise-int8 kernel and the wire codec

# collectives from different averagers in one process must not interleave
_COLLECTIVE_LOCK = threading.Lock()


class CollectiveSequencer:
    """Process-local FIFO that makes every rank launch RCCL rounds in the same
    order (DPU-overlapped grad/state rounds could be
    initiated in opposite orders on two ranks and deadlock RCCL).

    ``issue()`` is called where cross-rank program order is deterministic (the
    optimizer's foreground epoch logic); ``wait_turn(ticket)`` blocks the data
    plane until every earlier-issued round finished; ``release(ticket)`` is
    idempotent and must run when the round completes or is abandoned."""

    def __init__(self, stall_timeout: float = 300.0):
        self._cond = threading.Condition()
        self._queue: List[int] = []
        self._counter = 0
        self.stall_timeout = stall_timeout

    def issue(self) -> int:
        with self._cond:
            ticket = self._counter
            self._counter += 1
            self._queue.append(ticket)
            return ticket

    def wait_turn(self, ticket: Optional[int]) -> bool:
        """Block until `ticket` is at the head of the queue. Returns False if
        the wait timed out (the caller proceeds anyway -- a possible launch
        reorder beats a guaranteed hang; collective timeouts backstop)."""
        if ticket is None:
            return True
        deadline = None
        with self._cond:
            while self._queue and ticket in self._queue and self._queue[0] != ticket:
                import time

                if deadline is None:
                    deadline = time.monotonic() + self.stall_timeout
                remaining = deadline - time.monotonic()
                if remaining <= 0:
                    logger.warning(
                        f"collective ticket {ticket} waited {self.stall_timeout:.0f}s behind "
                        f"{self._queue[: self._queue.index(ticket)]}; proceeding out of order"
                    )
                    return False
                self._cond.wait(timeout=min(remaining, 5.0))
            return True

    def release(self, ticket: Optional[int]) -> None:
        if ticket is None:
            return
        with self._cond:
            try:
                self._queue.remove(ticket)
            except ValueError:
                pass
            self._cond.notify_all()


COLLECTIVE_SEQUENCER = CollectiveSequencer()


def distributed_world_info() -> Optional[tuple]:
    """(world_size, rank) if torch.distributed is initialized, else None."""
    if dist.is_available() and dist.is_initialized():
        return (dist.get_world_size(), dist.get_rank())
    return None


def issue_collective_ticket() -> Optional[int]:
    """Reserve a launch slot for an upcoming RCCL round. Call at a point whose
    execution order is identical on every rank (optimizer epoch logic)."""
    if distributed_world_info() is None:
        return None
    return COLLECTIVE_SEQUENCER.issue()


def release_collective_ticket(ticket: Optional[int]) -> None:
    COLLECTIVE_SEQUENCER.release(ticket)


def group_world_ranks(group_dist_info: Sequence[Optional[tuple]]) -> Optional[List[int]]:
    """If every group member is a distinct rank of OUR torch.distributed
    world, return their world ranks in group order; else None (RPC plane).

    Round 1 only matched group == whole world; Moshpit subgroups of the node
    now qualify too."""
    me = distributed_world_info()
    if me is None:
        return None
    world_size = me[0]
    ranks = []
    for info in group_dist_info:
        if info is None or len(info) != 2 or info[0] != world_size:
            return None
        ranks.append(int(info[1]))
    if len(set(ranks)) != len(ranks) or not all(0 <= r < world_size for r in ranks):
        return None
    return ranks


def group_matches_world(group_dist_info: Sequence[Optional[tuple]]) -> bool:
    """True if the assembled group is exactly the ranks 0..W-1 of our world."""
    me = distributed_world_info()
    ranks = group_world_ranks(group_dist_info)
    return me is not None and ranks is not None and set(ranks) == set(range(me[0]))


_subgroup_cache: Dict[tuple, "dist.ProcessGroup"] = {}
_subgroup_lock = threading.Lock()


def get_process_group_for_ranks(ranks: Sequence[int]) -> "dist.ProcessGroup":
    """Cached communicator for a subset of world ranks. Uses
    ``use_local_synchronization=True`` so only the members rendezvous --
    required for dynamically-matched Moshpit subgroups. All members must call
    this (they do: each runs its own data-plane round), and the
    CollectiveSequencer makes creation order consistent."""
    key = tuple(sorted(int(r) for r in ranks))
    with _subgroup_lock:
        pg = _subgroup_cache.get(key)
        if pg is None:
            logger.debug(f"creating process subgroup for world ranks {key}")
            pg = dist.new_group(list(key), use_local_synchronization=True)
            _subgroup_cache[key] = pg
        return pg


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
        codec: Optional[str] = None,
        averaging_alpha: float = 1.0,
        ticket: Optional[int] = None,
    ):
        assert codec in (None, "blockwise_int8"), f"unknown RCCL codec {codec}"
        self.tensors = list(tensors)
        self.weight = weight
        self.process_group = process_group
        self.bucket_size_bytes = bucket_size_bytes
        self.wire_dtype = wire_dtype
        self.codec = codec
        self.averaging_alpha = averaging_alpha
        self.ticket = ticket

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
        """Execute the weighted all-reduce, in place (blocking).

        Launch order across averagers is enforced by the CollectiveSequencer
        ticket (issued in deterministic cross-rank program order); within the
        process the collective lock keeps two rounds from interleaving."""
        device = self.tensors[0].device if self.tensors else torch.device("cpu")
        use_side_stream = device.type == "cuda"
        COLLECTIVE_SEQUENCER.wait_turn(self.ticket)
        with _COLLECTIVE_LOCK:
            total_weight = self._gather_total_weight(device)
            scale = self.weight / total_weight if total_weight > 0 else 0.0
            if use_side_stream:
                stream = _get_side_stream(device)
                stream.wait_stream(torch.cuda.current_stream(device))
                ctx = torch.cuda.stream(stream)
            else:
                ctx = _NullCtx()
            with ctx:
                if self.codec == "blockwise_int8" and dist.get_world_size(self.process_group) > 1:
                    self._run_blockwise_int8(scale, device)
                else:
                    self._run_bucketed(scale, device)
            if use_side_stream:
                torch.cuda.current_stream(device).wait_stream(stream)

    # ------------------------------------------------- full-precision buckets

    def _run_bucketed(self, scale: float, device: torch.device) -> None:
        # gloo lacks reduced-precision collectives; wire casting is an
        # xGMI-bandwidth optimization for the nccl(=RCCL) backend only
        try:
            backend = dist.get_backend(self.process_group)
        except Exception:
            backend = None
        wire_dtype = self.wire_dtype if str(backend) == "nccl" else None
        for bucket_tensors in self._iter_buckets():
            flat = torch.cat([t.detach().reshape(-1) for t in bucket_tensors])
            work_dtype = wire_dtype or flat.dtype
            flat_scaled = flat.to(work_dtype)
            flat_scaled.mul_(scale)
            dist.all_reduce(flat_scaled, op=dist.ReduceOp.SUM, group=self.process_group)
            averaged = flat_scaled.to(flat.dtype)
            self._copy_back(bucket_tensors, averaged)

    # ------------------------------------------------ int8 blockwise butterfly

    def _run_blockwise_int8(self, scale: float, device: torch.device) -> None:
        """quantize -> all-to-all -> dequant-accumulate -> requant ->
        all-gather. Each group rank owns 1/W of the flat vector and receives
        the other W-1 slices simultaneously -- on the 8-GPU node this loads
        all 7 outgoing xGMI links at once, like the reference butterfly's
        "every sender streams to every owner" shape (allreduce.py:163-166)."""
        from ..ops import dequantize_blockwise, quantize_blockwise

        pg = self.process_group
        world = dist.get_world_size(pg)
        flat = torch.cat([t.detach().reshape(-1).float() for t in self.tensors]).mul_(scale)
        n = flat.numel()
        chunk = -(-n // world)  # ceil
        chunk = -(-chunk // QUANT_BLOCK) * QUANT_BLOCK  # block-align slices
        total = chunk * world
        if total > n:
            flat = torch.cat([flat, torch.zeros(total - n, device=device, dtype=torch.float32)])
        blocks_per_chunk = chunk // QUANT_BLOCK

        q, absmax = quantize_blockwise(flat)
        q_recv = torch.empty_like(q)
        amax_recv = torch.empty_like(absmax)
        dist.all_to_all_single(q_recv, q, group=pg)
        dist.all_to_all_single(amax_recv, absmax, group=pg)

        # reduce my slice: sum of every member's (pre-scaled) contribution
        acc = torch.zeros(chunk, device=device, dtype=torch.float32)
        for i in range(world):
            acc += dequantize_blockwise(
                q_recv[i * chunk : (i + 1) * chunk],
                amax_recv[i * blocks_per_chunk : (i + 1) * blocks_per_chunk],
            ).to(device)

        q_avg, amax_avg = quantize_blockwise(acc)
        q_full = [torch.empty_like(q_avg) for _ in range(world)]
        amax_full = [torch.empty_like(amax_avg) for _ in range(world)]
        dist.all_gather(q_full, q_avg, group=pg)
        dist.all_gather(amax_full, amax_avg, group=pg)
        averaged = dequantize_blockwise(torch.cat(q_full), torch.cat(amax_full)).to(device)[:n]
        self._copy_back(self.tensors, averaged)

    # ------------------------------------------------------------------ utils

    def _copy_back(self, tensors: Sequence[torch.Tensor], averaged_flat: torch.Tensor) -> None:
        offset = 0
        for tensor in tensors:
            n = tensor.numel()
            avg_part = averaged_flat[offset : offset + n].view(tensor.shape).to(tensor.dtype)
            # .data: with reuse_tensors the target IS a live model parameter,
            # and a backward pass may be in flight on another thread (DPU) --
            # bypass the version counter
            if self.averaging_alpha == 1.0:
                tensor.data.copy_(avg_part)
            else:
                tensor.data.add_(avg_part - tensor.data, alpha=self.averaging_alpha)
            offset += n

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
