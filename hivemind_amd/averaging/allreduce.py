"""Butterfly all-reduce over RPC streams (the WAN / CPU data plane).

Parity target: reference ``hivemind/averaging/allreduce.py:32-377``: each of N
peers owns a fraction of the flat vector; every sender streams its slice of
each owner's fraction to that owner (``rpc_aggregate_part``); the owner
accumulates the weighted sum and streams back *deltas* (avg - input) so
senders can apply them in place. Client-mode peers own zero fraction; aux
peers reduce but send nothing. Failed senders are banned from the rest of the
round; failed reducers are replaced by zero deltas (local values survive).

On a single MI355X node the averager prefers the RCCL data plane
(averaging/rccl.py) and only uses this path for cross-node/WAN peers or CPU
tests -- same group semantics either way.
"""

from __future__ import annotations

import asyncio
import enum
from dataclasses import dataclass, field
from typing import Any, AsyncIterator, Dict, Optional, Sequence, Tuple

import torch

from ..compression import CompressionBase, NoCompression, WireTensor, deserialize_torch_tensor
from ..p2p import P2P, PeerID, RpcContext, RpcMessage, ServicerBase
from ..utils.asyncio_utils import achain, aiter_with_timeout, amap_in_executor, anext_impl, as_aiter, attach_event_on_finished
from ..utils.logging import get_logger
from .partition import AllreduceException, TensorPartContainer, TensorPartReducer

logger = get_logger(__name__)


class AveragingMode(enum.Enum):
    NODE = 0  # full participant: sends, reduces, owns a fraction
    CLIENT = 1  # sends its tensors but owns no fraction (can't accept connections)
    AUX = 2  # owns a fraction and reduces, but has no data of its own (weight 0)


# stream message codes
class DataCode:
    PART_FOR_AVERAGING = 0
    AVERAGED_PART = 1
    ERROR = 2


@dataclass
class AveragingData(RpcMessage):
    code: int = DataCode.PART_FOR_AVERAGING
    group_id: bytes = b""
    peer_id: bytes = b""
    tensor: Optional[WireTensor] = None
    weight: float = 1.0


class AllReduceRunner:
    """One all-reduce round within an assembled group.

    The hosting averager routes incoming ``rpc_aggregate_part`` streams for
    ``group_id`` into this runner (reference averager.py:581-598 group registry).
    """

    def __init__(
        self,
        *,
        p2p: P2P,
        servicer_type,
        group_id: bytes,
        tensors: Sequence[torch.Tensor],
        ordered_peer_ids: Sequence[PeerID],
        peer_fractions: Tuple[float, ...],
        weight: float = 1.0,
        modes: Optional[Sequence[AveragingMode]] = None,
        compression: CompressionBase = NoCompression(),
        part_size_bytes: int = 2**19,
        gathered: Optional[Dict[PeerID, Any]] = None,
        sender_timeout: float = 10.0,
        reducer_timeout: float = 15.0,
        namespace: Optional[str] = None,
    ):
        self._p2p = p2p
        self.servicer_type = servicer_type
        self.namespace = namespace
        self.group_id = group_id
        self.ordered_peer_ids = list(ordered_peer_ids)
        self.peer_id = p2p.peer_id
        assert self.peer_id in self.ordered_peer_ids, "peer is not a part of the group"
        self.modes = list(modes) if modes is not None else [AveragingMode.NODE] * len(self.ordered_peer_ids)
        self.peer_fractions = peer_fractions
        self.weight = weight
        self.gathered = gathered
        self.sender_timeout, self.reducer_timeout = sender_timeout, reducer_timeout

        for peer_id, frac, mode in zip(self.ordered_peer_ids, peer_fractions, self.modes):
            assert mode != AveragingMode.CLIENT or frac == 0, "client-mode peers should not own fractions"
            assert mode != AveragingMode.AUX or True

        self.sender_peer_ids = [
            pid for pid, mode in zip(self.ordered_peer_ids, self.modes) if mode != AveragingMode.AUX
        ]
        self.sender_timeout_events: Dict[PeerID, asyncio.Event] = {}
        self.all_senders_started = asyncio.Event()
        self.banned_senders: set = set()
        self.banlock = asyncio.Lock()
        self.active_senders: set = set()

        self.tensor_part_container = TensorPartContainer(
            tensors, peer_fractions, compression=compression, part_size_bytes=part_size_bytes
        )
        my_index = self.ordered_peer_ids.index(self.peer_id)
        self.parts_for_local_averaging = self.tensor_part_container.get_raw_input_parts(my_index)
        self.tensor_part_reducer = TensorPartReducer(
            tuple(part.shape for part in self.parts_for_local_averaging),
            len(self.sender_peer_ids),
        )
        self._future: asyncio.Future = asyncio.Future()
        self.finished = asyncio.Event()

    def __aiter__(self):
        return self.run()

    @property
    def group_size(self):
        return len(self.ordered_peer_ids)

    def _get_peer_stub(self, peer: PeerID):
        return self.servicer_type.get_stub(self._p2p, peer, namespace=self.namespace)

    async def run(self) -> AsyncIterator[torch.Tensor]:
        """Run all-reduce; yield averaged tensor deltas in order (reference allreduce.py:151-199)."""
        pending_tasks = set()
        try:
            if len(self.sender_peer_ids) == 0:
                logger.debug(f"{self} finished: no senders")
                self.tensor_part_container.finalize()
                return
            my_mode = self.modes[self.ordered_peer_ids.index(self.peer_id)]
            my_fraction = self.peer_fractions[self.ordered_peer_ids.index(self.peer_id)]
            if my_fraction > 0:
                # reducer side: a sender that dies after matchmaking but BEFORE
                # opening its stream would stall our parts forever -- ban every
                # sender that has not shown up within sender_timeout (reference
                # allreduce.py:192-199)
                pending_tasks.add(asyncio.create_task(self._ban_missing_senders()))
            if my_mode != AveragingMode.AUX:
                for peer_index, (peer_id, fraction) in enumerate(zip(self.ordered_peer_ids, self.peer_fractions)):
                    if fraction > 0:
                        pending_tasks.add(asyncio.create_task(self._communicate_with_peer(peer_id)))
                async for averaged_tensor_delta in self.tensor_part_container.iterate_output_tensors():
                    yield averaged_tensor_delta
            else:
                # aux peers only reduce: wait for the reducer to run dry
                await self.tensor_part_reducer.finished.wait()
            self.finalize()
            for task in pending_tasks:
                await task
        except BaseException as e:
            self.finalize(exception=e)
            for task in pending_tasks:
                task.cancel()
            raise

    async def _communicate_with_peer(self, peer_id: PeerID):
        """Send our slice of peer_id's fraction; apply returned deltas
        (reference allreduce.py:201-257)."""
        peer_index = self.ordered_peer_ids.index(peer_id)
        sender_index = self.sender_peer_ids.index(self.peer_id)
        if peer_id == self.peer_id:
            for part_index, tensor_part in enumerate(self.parts_for_local_averaging):
                averaged_part = await self.tensor_part_reducer.accumulate_part(
                    sender_index, part_index, tensor_part, weight=self.weight
                )
                self.tensor_part_container.register_processed_part(
                    peer_index, part_index, averaged_part - tensor_part
                )
            return

        try:
            done_sending = asyncio.Event()
            inputs_aiter = attach_event_on_finished(self._generate_input_for_peer(peer_index), done_sending)
            stream = self._get_peer_stub(peer_id).rpc_aggregate_part(inputs_aiter)

            part_index = 0

            def _try_deserialize(msg: AveragingData):
                if msg.code != DataCode.AVERAGED_PART:
                    raise AllreduceException(f"peer {peer_id} returned code {msg.code}")
                return deserialize_torch_tensor(msg.tensor)

            async for delta in amap_in_executor(
                _try_deserialize,
                aiter_with_timeout(_typed(stream, AveragingData), self.reducer_timeout),
                max_prefetch=5,
            ):
                self.tensor_part_container.register_processed_part(peer_index, part_index, delta)
                part_index += 1

            if part_index < self.tensor_part_container.num_parts_by_peer[peer_index]:
                logger.warning(
                    f"{self}: peer {peer_id} returned only {part_index} of "
                    f"{self.tensor_part_container.num_parts_by_peer[peer_index]} parts; using local values for the rest"
                )
                self.tensor_part_container.register_failed_reducer(peer_index)
        except BaseException as e:
            if isinstance(e, asyncio.CancelledError):
                raise
            logger.debug(f"{self}: communication with reducer {peer_id} failed: {e!r}")
            self.tensor_part_container.register_failed_reducer(peer_index)

    async def _generate_input_for_peer(self, peer_index: int) -> AsyncIterator[AveragingData]:
        first = True
        async for wire_part in self.tensor_part_container.iterate_input_parts_for(peer_index):
            yield AveragingData(
                code=DataCode.PART_FOR_AVERAGING,
                group_id=self.group_id,
                peer_id=self.peer_id.to_bytes(),
                tensor=wire_part,
                weight=self.weight,
            )
            first = False

    # -------------------------------------------------------------- reducer side

    async def rpc_aggregate_part(
        self, stream: AsyncIterator[AveragingData], context: RpcContext
    ) -> AsyncIterator[AveragingData]:
        """Accumulate incoming parts from one sender; stream back deltas
        (reference allreduce.py:259-333)."""
        sender_peer_id = None
        sender_index = None
        try:
            first_message = await asyncio.wait_for(anext_impl(stream.__aiter__() if hasattr(stream, "__aiter__") else stream), self.sender_timeout)
            # identify the sender from the AUTHENTICATED connection, not the
            # self-declared message field (reference allreduce.py:263 uses
            # context.remote_id; ADVICE round 1: the message field would let a
            # group member impersonate another sender or double-contribute)
            sender_peer_id = context.remote_id
            if first_message.peer_id and PeerID(first_message.peer_id) != sender_peer_id:
                logger.debug(
                    f"{self}: sender {sender_peer_id} declared a different peer_id in-band; ignoring the in-band value"
                )
            if sender_peer_id not in self.sender_peer_ids:
                yield AveragingData(code=DataCode.ERROR, group_id=self.group_id)
                return
            sender_index = self.sender_peer_ids.index(sender_peer_id)
            self.active_senders.add(sender_peer_id)

            async def full_stream():
                yield first_message
                async for msg in aiter_with_timeout(stream, self.sender_timeout):
                    yield msg

            part_index = 0
            async for message in full_stream():
                if message.code != DataCode.PART_FOR_AVERAGING or message.tensor is None:
                    raise AllreduceException(f"sender {sender_peer_id} sent code {message.code}")
                tensor_part = deserialize_torch_tensor(message.tensor)
                averaged_part = await self.tensor_part_reducer.accumulate_part(
                    sender_index, part_index, tensor_part, weight=message.weight
                )
                delta = averaged_part.to(tensor_part.dtype) - tensor_part
                serialized = _recompress_like(delta, message.tensor)
                yield AveragingData(code=DataCode.AVERAGED_PART, group_id=self.group_id, tensor=serialized)
                part_index += 1
        except BaseException as e:
            if isinstance(e, asyncio.CancelledError):
                raise
            logger.debug(f"{self}: rpc_aggregate_part from {sender_peer_id} failed: {e!r}")
            if sender_index is not None:
                await self._ban_sender(sender_peer_id, sender_index)
            yield AveragingData(code=DataCode.ERROR, group_id=self.group_id)

    async def _ban_missing_senders(self):
        """Ban senders that never opened their rpc_aggregate_part stream."""
        try:
            # if every part reduces before the deadline, nobody is missing
            await asyncio.wait_for(self.tensor_part_reducer.finished.wait(), timeout=self.sender_timeout)
            return
        except asyncio.TimeoutError:
            pass
        for sender_index, sender_peer_id in enumerate(self.sender_peer_ids):
            if sender_peer_id == self.peer_id:
                continue  # local parts accumulate synchronously
            if sender_peer_id not in self.active_senders:
                logger.debug(f"{self}: sender {sender_peer_id} never showed up; banning")
                await self._ban_sender(sender_peer_id, sender_index)

    async def _ban_sender(self, sender_peer_id: PeerID, sender_index: int):
        """Exclude a failed sender from the rest of this round (reference allreduce.py:317-321)."""
        async with self.banlock:
            if sender_peer_id not in self.banned_senders:
                self.banned_senders.add(sender_peer_id)
                self.tensor_part_reducer.on_sender_failed(sender_index)

    def finalize(self, exception: Optional[BaseException] = None):
        if not self.finished.is_set():
            self.finished.set()
            self.tensor_part_reducer.finalize()
            if exception is not None:
                self.tensor_part_container.finalize()

    def __repr__(self):
        return f"AllReduceRunner(group={self.group_id.hex()[:8]}, size={self.group_size})"


async def _typed(stream: AsyncIterator[bytes], message_type) -> AsyncIterator:
    async for payload in stream:
        yield message_type.loads(payload) if isinstance(payload, (bytes, bytearray)) else payload


def _recompress_like(delta: torch.Tensor, original: WireTensor) -> WireTensor:
    """Compress the delta with the same codec the sender used."""
    from ..compression import BASE_COMPRESSION_TYPES, CompressionType

    codec = BASE_COMPRESSION_TYPES[CompressionType(original.compression)]
    return codec.compress(delta)
