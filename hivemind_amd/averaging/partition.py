"""Slice ownership: split a tensor list into per-peer parts; reduce owned parts.

Parity target: reference ``hivemind/averaging/partition.py:21-285``
(``TensorPartContainer`` assigns contiguous slices of the flat vector to peers
proportionally to their fractions and pipelines compression in a background
executor; ``TensorPartReducer`` accumulates ``acc += w*x`` per part and
divides by the total weight on completion, with failed-sender fallback).

This module is the *shared* slice-ownership abstraction: the RPC data plane
streams these parts over TCP; the RCCL data plane uses the same fraction
arithmetic to build reduce-scatter splits (averaging/rccl.py).
"""

from __future__ import annotations

import asyncio
from collections import deque
from typing import AsyncIterable, AsyncIterator, List, Optional, Sequence, Tuple

import torch

from ..compression import CompressionBase, CompressionInfo, NoCompression, WireTensor, deserialize_torch_tensor
from ..utils.asyncio_utils import amap_in_executor
from ..utils.logging import get_logger

logger = get_logger(__name__)

DEFAULT_PART_SIZE_BYTES = 2**19  # 512 KiB, reference partition.py:17


class TensorPartContainer:
    """Splits tensors into parts assigned to peers; compresses own outputs and
    re-assembles averaged results from incoming parts."""

    def __init__(
        self,
        tensors: Sequence[torch.Tensor],
        peer_fractions: Sequence[float],
        compression: CompressionBase = NoCompression(),
        part_size_bytes: int = DEFAULT_PART_SIZE_BYTES,
        tensor_infos: Optional[Sequence[CompressionInfo]] = None,
        return_deltas: bool = True,
        prefetch: int = 5,
    ):
        if tensor_infos is None:
            tensor_infos = tuple(CompressionInfo.from_tensor(x, key=i) for i, x in enumerate(tensors))
        assert len(tensor_infos) == len(tensors), "compression types do not match the number of tensors"
        self.local_tensors, self.peer_fractions = tensors, peer_fractions
        self.compression, self.part_size_bytes, self.tensor_infos = compression, part_size_bytes, tensor_infos
        self.total_size = sum(tensor.numel() for tensor in tensors)
        self.return_deltas = return_deltas
        self.prefetch = prefetch
        self.failed_size = 0
        self.num_parts_by_tensor = []
        self._input_parts_by_peer: List[deque] = [deque() for _ in range(len(peer_fractions))]
        self._output_parts_by_peer: List[deque] = [deque() for _ in range(len(peer_fractions))]
        self._inputs_consumed_by_peer = [False for _ in range(len(peer_fractions))]
        self._output_part_available = [asyncio.Event() for _ in range(len(peer_fractions))]
        self._outputs_registered_by_peer = [0 for _ in range(len(peer_fractions))]
        self._outputs_consumed = False
        self.finished = asyncio.Event()

        # split tensors into parts: contiguous slices of the flat vector per peer
        # (reference partition.py:64-93)
        peer_pieces = self._split_numel_by_fractions(self.total_size, peer_fractions)
        self.num_parts_by_peer = []
        current_length = 0
        current_peer_index = 0
        pivots = list(self._cumsum(peer_pieces))
        flat_parts_per_peer: List[List[Tuple[int, torch.Tensor, CompressionInfo]]] = [
            [] for _ in range(len(peer_fractions))
        ]
        for tensor_idx, (tensor, info) in enumerate(zip(tensors, tensor_infos)):
            part_size_values = max(1, self.part_size_bytes // max(tensor.element_size(), 1))
            flat = tensor.detach().reshape(-1)
            num_parts = 0
            offset = 0
            while offset < flat.numel():
                # a part may not cross a peer boundary
                peer_end = pivots[current_peer_index] - current_length
                length = min(part_size_values, flat.numel() - offset, peer_end)
                if length <= 0:
                    current_peer_index += 1
                    continue
                part = flat[offset : offset + length]
                part_info = info.get_part(num_parts, part_size_values)
                flat_parts_per_peer[current_peer_index].append((tensor_idx, part, part_info))
                offset += length
                current_length += length
                num_parts += 1
                if current_length >= pivots[current_peer_index] and current_peer_index + 1 < len(peer_fractions):
                    current_peer_index += 1
            self.num_parts_by_tensor.append(num_parts)
        for peer_idx, parts in enumerate(flat_parts_per_peer):
            for entry in parts:
                self._input_parts_by_peer[peer_idx].append(entry)
            self.num_parts_by_peer.append(len(parts))

    @staticmethod
    def _split_numel_by_fractions(total: int, fractions: Sequence[float]) -> List[int]:
        """Integer split of `total` elements proportional to fractions (sums exactly)."""
        total_fraction = sum(fractions)
        if total_fraction <= 0:
            return [0] * len(fractions)
        raw = [total * f / total_fraction for f in fractions]
        floors = [int(x) for x in raw]
        remainder = total - sum(floors)
        # largest-remainder apportionment
        order = sorted(range(len(raw)), key=lambda i: raw[i] - floors[i], reverse=True)
        for i in order[:remainder]:
            floors[i] += 1
        return floors

    @staticmethod
    def _cumsum(values: Sequence[int]):
        total = 0
        for v in values:
            total += v
            yield total

    @property
    def local_parts(self) -> List[torch.Tensor]:
        """Raw (uncompressed) parts owned by each peer, flattened in order."""
        return [part for peer_parts in self._input_parts_by_peer for (_, part, _) in peer_parts]

    def get_raw_input_parts(self, peer_index: int) -> Tuple[torch.Tensor, ...]:
        """All input parts assigned to one peer (tensors only)."""
        self._inputs_consumed_by_peer[peer_index] = True
        parts = tuple(part for (_, part, _) in self._input_parts_by_peer[peer_index])
        return parts

    async def iterate_input_parts_for(self, peer_index: int) -> AsyncIterator[WireTensor]:
        """Yield compressed input parts for the given peer (background compression
        with prefetch, reference partition.py:104-112)."""
        self._inputs_consumed_by_peer[peer_index] = True

        async def _aiter_parts():
            for (_, part, info) in self._input_parts_by_peer[peer_index]:
                yield part, info

        async for wire in amap_in_executor(
            lambda pair: self.compression.compress(pair[0], pair[1], allow_inplace=False),
            _aiter_parts(),
            max_prefetch=self.prefetch,
        ):
            yield wire

    def register_processed_part(self, peer_index: int, part_index: int, part: torch.Tensor):
        """Accept the next averaged part (or delta) from a peer, in order."""
        if part_index != self._outputs_registered_by_peer[peer_index]:
            raise ValueError(
                f"expected part {self._outputs_registered_by_peer[peer_index]} from peer {peer_index}, got {part_index}"
            )
        self._output_parts_by_peer[peer_index].append(part)
        self._outputs_registered_by_peer[peer_index] += 1
        self._output_part_available[peer_index].set()

    def register_failed_reducer(self, peer_index: int):
        """Fill the peer's remaining parts with zero deltas (local values survive)
        -- reference partition.py:128-136."""
        for part_index in range(self._outputs_registered_by_peer[peer_index], self.num_parts_by_peer[peer_index]):
            part = self._input_parts_by_peer[peer_index][part_index][1]
            self.failed_size += part.numel()
            self.register_processed_part(peer_index, part_index, torch.zeros_like(part))

    async def iterate_output_tensors(self) -> AsyncIterator[torch.Tensor]:
        """Yield averaged tensors (deltas applied by the caller), in input order."""
        assert not self._outputs_consumed, "output tensors are already being iterated"
        self._outputs_consumed = True
        peer_index = num_parts_processed = 0
        part_in_peer_index = [0 for _ in range(len(self.peer_fractions))]
        for tensor_index in range(len(self.local_tensors)):
            tensor_parts = []
            for _ in range(self.num_parts_by_tensor[tensor_index]):
                # advance to the peer owning the next part
                while part_in_peer_index[peer_index] >= self.num_parts_by_peer[peer_index]:
                    peer_index += 1
                while len(self._output_parts_by_peer[peer_index]) == 0:
                    self._output_part_available[peer_index].clear()
                    await self._output_part_available[peer_index].wait()
                    if self.finished.is_set():
                        raise AllreduceException("output iterator was interrupted")
                tensor_parts.append(self._output_parts_by_peer[peer_index].popleft())
                part_in_peer_index[peer_index] += 1
                num_parts_processed += 1
            tensor = torch.cat(tensor_parts) if len(tensor_parts) != 1 else tensor_parts[0]
            yield tensor.reshape(self.local_tensors[tensor_index].shape)

    def __del__(self):
        self.finalize()

    def finalize(self):
        if not self.finished.is_set():
            for peer_index in range(len(self.peer_fractions)):
                self._inputs_consumed_by_peer[peer_index] = True
                self._input_parts_by_peer[peer_index].clear()
                self._output_parts_by_peer[peer_index].clear()
                self._output_part_available[peer_index].set()
            self.finished.set()


class AllreduceException(Exception):
    """Internal error in an all-reduce round."""


class TensorPartReducer:
    """Owns a set of parts: accumulates weighted contributions from all senders,
    emits the average when each part is complete (reference partition.py:179-261)."""

    def __init__(self, part_shapes: Sequence[torch.Size], num_senders: int, weights: Optional[Sequence[float]] = None):
        self.part_shapes, self.num_senders, self.num_parts = part_shapes, num_senders, len(part_shapes)
        self.weights = tuple(weights or (1.0 for _ in range(num_senders)))
        assert len(self.weights) == self.num_senders
        self.current_part_index = -1
        self.current_part_accumulated_from = 0
        self.accumulator: Optional[torch.Tensor] = None
        self.denominator = 0.0
        self.current_part_future: Optional[asyncio.Future] = None
        self.finished = asyncio.Event()
        self.num_parts_received = [0 for _ in range(self.num_senders)]
        self.sender_failed_after = [float("inf") for _ in range(self.num_senders)]
        self.num_current_senders = self.num_senders
        self.reset_accumulators()

    def reset_accumulators(self):
        """Prepare for the next part (reference partition.py:202-216)."""
        assert self.current_part_accumulated_from == self.num_current_senders or self.current_part_index == -1
        if self.current_part_index >= self.num_parts - 1:
            self.finalize()
            return
        self.current_part_index += 1
        self.current_part_accumulated_from = 0
        self.current_part_future = asyncio.Future()
        self.num_current_senders = sum(
            self.current_part_index < failed_after for failed_after in self.sender_failed_after
        )
        self.accumulator = torch.zeros(self.part_shapes[self.current_part_index])
        self.denominator = 0.0

    async def accumulate_part(self, sender_index: int, part_index: int, tensor_part: torch.Tensor, weight: float = 1.0) -> torch.Tensor:
        """Add a part from a sender; return the averaged part when all contributions arrive."""
        assert 0 <= sender_index < self.num_senders, "invalid sender index"
        assert 0 <= part_index < self.num_parts, "invalid part index"
        self.num_parts_received[sender_index] += 1

        while part_index > self.current_part_index:
            await asyncio.wait(
                [asyncio.ensure_future(self.current_part_future), asyncio.ensure_future(self.finished.wait())],
                return_when=asyncio.FIRST_COMPLETED,
            )
            if self.finished.is_set():
                raise AllreduceException(f"reducer finished early; can't accumulate part {part_index}")
        if self.finished.is_set():
            raise AllreduceException("reducer is finished")
        assert part_index == self.current_part_index, "parts must arrive in order"

        current_part_future = self.current_part_future
        effective_weight = weight * self.weights[sender_index]
        # the fused weighted-accumulate HIP kernel on GPU parts
        # (bf16/f32 into the fp32 accumulator in one pass); torch add_ on CPU
        from ..ops import weighted_accumulate_

        weighted_accumulate_(self.accumulator, tensor_part, effective_weight)
        self.denominator += effective_weight
        self.current_part_accumulated_from += 1

        assert self.current_part_accumulated_from <= self.num_current_senders
        if self.current_part_accumulated_from == self.num_current_senders:
            if self.denominator > 0:
                self.accumulator.div_(self.denominator)
            current_part_future.set_result(self.accumulator.clone())
            self.reset_accumulators()
        return await current_part_future

    def on_sender_failed(self, sender_index: int):
        """Exclude a sender from all future parts (reference partition.py:248-261)."""
        self.sender_failed_after[sender_index] = self.num_parts_received[sender_index]
        if self.finished.is_set():
            return
        if self.current_part_index == self.num_parts_received[sender_index]:
            self.num_current_senders -= 1
            if self.current_part_accumulated_from == self.num_current_senders and self.num_current_senders > 0:
                if self.denominator > 0:
                    self.accumulator.div_(self.denominator)
                self.current_part_future.set_result(self.accumulator.clone())
                self.reset_accumulators()

    def finalize(self):
        if not self.finished.is_set():
            if self.current_part_future is not None and not self.current_part_future.done():
                self.current_part_future.cancel()
            self.finished.set()

            if self.num_parts > 0 and self.num_senders > 0:
                parts_expected = self.num_parts * self.num_senders
                parts_received = sum(self.num_parts_received)
                if parts_expected != parts_received:
                    logger.info(f"reducer: received {parts_received / max(parts_expected,1) * 100:.1f}% of input parts")

    def __del__(self):
        self.finalize()
