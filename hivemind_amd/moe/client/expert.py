"""RemoteExpert: an nn.Module proxy that calls an expert hosted elsewhere.

Parity target: reference ``hivemind/moe/client/expert.py:32-233``: an
autograd-compatible ``_RemoteModuleCall`` calls ``rpc_forward``/``rpc_backward``
(automatically switching to the streaming variant above 2 MB payloads), the
DUMMY empty tensor triggers autograd, and ``rpc_info`` caches I/O schemas.
"""

from __future__ import annotations

import asyncio
from typing import Any, Dict, List, Optional, Sequence, Tuple

import torch
import torch.nn as nn

from ...compression import WireTensor, deserialize_torch_tensor, serialize_torch_tensor
from ...compression.base import CompressionType
from ...compression.serialization import combine_from_streaming, split_for_streaming
from ...dht import DHT
from ...p2p import P2P, PeerID
from ...utils.logging import get_logger
from ...utils.nested import nested_compare, nested_flatten, nested_pack
from ...utils.serializer import MSGPackSerializer
from ..expert_uid import ExpertInfo
from ..server.connection_handler import (
    MAX_UNARY_PAYLOAD_SIZE,
    ConnectionHandler,
    ExpertRequest,
    ExpertResponse,
    ExpertUIDRequest,
    ExpertInfoResponse,
)
from .remote_expert_worker import RemoteExpertWorker

logger = get_logger(__name__)

DUMMY = torch.empty(0, requires_grad=True)  # triggers autograd in RemoteExpert


def _get_expert_stub(p2p: P2P, peer: PeerID):
    return ConnectionHandler.get_stub(p2p, peer)


class RemoteExpert(nn.Module):
    """Calls an expert on a remote peer during forward/backward passes."""

    def __init__(self, expert_info: ExpertInfo, p2p: P2P, loop: Optional[asyncio.AbstractEventLoop] = None):
        super().__init__()
        self._info, self.p2p = expert_info, p2p
        self._loop = loop
        self._rpc_info: Optional[Dict[str, Any]] = None
        if expert_info.endpoint:
            p2p.learn_endpoint(expert_info.peer_id, expert_info.endpoint)

    @property
    def uid(self):
        return self._info.uid

    @property
    def peer_id(self) -> PeerID:
        return self._info.peer_id

    @property
    def stub(self):
        return _get_expert_stub(self.p2p, self.peer_id)

    def forward(self, *args, **kwargs):
        assert len(kwargs) == len(self.info["keyword_names"]), f"expected {self.info['keyword_names']} kwargs"
        kwargs = {key: kwargs[key] for key in self.info["keyword_names"]}
        forward_inputs = (args, kwargs)
        if not nested_compare(forward_inputs, self.info["forward_schema"]):
            raise TypeError("inputs do not match expert's input schema")
        flat_inputs = list(nested_flatten(forward_inputs))
        forward_task_size = flat_inputs[0].shape[0]
        flat_outputs = _RemoteModuleCall.apply(
            DUMMY, self.uid, self.stub, self._loop, self.info, forward_task_size, *flat_inputs
        )
        return nested_pack(flat_outputs, structure=self.info["outputs_schema"])

    @property
    def info(self) -> Dict[str, Any]:
        if self._rpc_info is None:
            raw = RemoteExpertWorker.run_coroutine(
                self.stub.rpc_info(ExpertUIDRequest(uid=self.uid), timeout=10), loop=self._loop
            )
            info = MSGPackSerializer.loads(ExpertInfoResponse.loads(raw).serialized_info)
            args_schema, kwargs_schema = info["forward_schema"]
            info["forward_schema"] = (tuple(args_schema), dict(kwargs_schema))
            info["keyword_names"] = tuple(kwargs_schema.keys())
            self._rpc_info = info
        return self._rpc_info

    def extra_repr(self):
        return f"uid={self.uid}, peer_id={self.peer_id}"


def create_remote_experts(
    infos: Sequence[Optional[ExpertInfo]], dht: DHT
) -> List[Optional[RemoteExpert]]:
    p2p = dht.replicate_p2p()
    RemoteExpertWorker.set_default_loop(dht.loop)
    return [RemoteExpert(info, p2p, loop=dht.loop) if info is not None else None for info in infos]


def _tensors_to_parts(uid: str, tensors: Sequence[torch.Tensor]) -> Tuple[List[WireTensor], int]:
    parts = [serialize_torch_tensor(t.detach().cpu(), CompressionType.NONE) for t in tensors]
    total = sum(len(p.buffer) for p in parts)
    return parts, total


RAW_PATH_MAX_BYTES = 128 * 1024 * 1024  # single-frame fast path bound


async def expert_forward(uid: str, stub, tensors: Sequence[torch.Tensor], timeout: Optional[float] = None) -> List[torch.Tensor]:
    """Raw single-frame fast path (moe/wire.py) below 128 MB, chunk-streamed
    above (reference's unary/stream split at expert.py:149-191)."""
    from ..wire import pack_tensors, unpack_tensors

    total = sum(t.numel() * t.element_size() for t in tensors)
    if total <= RAW_PATH_MAX_BYTES:
        payload = pack_tensors(uid, tensors)
        raw = await asyncio.wait_for(
            stub._p2p.call_unary(stub._peer, "expert::fwd_raw", payload, timeout=timeout), timeout
        )
        return unpack_tensors(raw)[1]
    parts, _ = _tensors_to_parts(uid, tensors)
    return await _expert_stream_call(stub.rpc_forward_stream, uid, parts, timeout)


async def expert_backward(uid: str, stub, tensors: Sequence[torch.Tensor], timeout: Optional[float] = None) -> List[torch.Tensor]:
    from ..wire import pack_tensors, unpack_tensors

    total = sum(t.numel() * t.element_size() for t in tensors)
    if total <= RAW_PATH_MAX_BYTES:
        payload = pack_tensors(uid, tensors)
        raw = await asyncio.wait_for(
            stub._p2p.call_unary(stub._peer, "expert::bwd_raw", payload, timeout=timeout), timeout
        )
        return unpack_tensors(raw)[1]
    parts, _ = _tensors_to_parts(uid, tensors)
    return await _expert_stream_call(stub.rpc_backward_stream, uid, parts, timeout)


async def _expert_stream_call(method, uid: str, parts: List[WireTensor], timeout: Optional[float]) -> List[torch.Tensor]:
    async def request_iter():
        first = True
        for serialized in parts:
            for chunk in split_for_streaming(serialized):
                yield ExpertRequest(uid=uid if first else "", tensors=[chunk])
                first = False

    async def _call():
        chunks: List[WireTensor] = []
        tensors: List[torch.Tensor] = []
        async for raw in method(request_iter()):
            response = ExpertResponse.loads(raw)
            for part in response.tensors:
                if part.chunks and chunks:
                    tensors.append(deserialize_torch_tensor(combine_from_streaming(chunks)))
                    chunks = []
                chunks.append(part)
        if chunks:
            tensors.append(deserialize_torch_tensor(combine_from_streaming(chunks)))
        return tensors

    return await asyncio.wait_for(_call(), timeout)


class _RemoteModuleCall(torch.autograd.Function):
    """RPC-backed forward/backward (reference expert.py:194-233)."""

    @staticmethod
    def forward(ctx, dummy: torch.Tensor, uid: str, stub, loop, info: Dict[str, Any], forward_task_size: int, *inputs: torch.Tensor):
        inputs = tuple(map(torch.Tensor.detach, inputs))
        ctx.uid, ctx.stub, ctx.loop, ctx.info = uid, stub, loop, info
        ctx.save_for_backward(*inputs)
        outputs = RemoteExpertWorker.run_coroutine(expert_forward(uid, stub, inputs), loop=loop)
        return tuple(outputs)

    @staticmethod
    @torch.autograd.function.once_differentiable
    def backward(ctx, *grad_outputs):
        inputs = ctx.saved_tensors
        backward_inputs = tuple(inputs) + tuple(g.contiguous() for g in grad_outputs)
        grad_inputs = RemoteExpertWorker.run_coroutine(
            expert_backward(ctx.uid, ctx.stub, backward_inputs), loop=ctx.loop
        )
        return (DUMMY, None, None, None, None, None, *grad_inputs)
