"""ConnectionHandler: the RPC surface of an expert server.

Parity target: reference ``hivemind/moe/server/connection_handler.py:22-177``:
``rpc_info`` (schemas), ``rpc_forward``/``rpc_backward`` (unary, payloads
<= 2 MB) and their ``_stream`` variants (chunked tensors). The reference runs
N forked handler processes load-balanced by the daemon; here handlers are
asyncio coroutines on the server's event loop -- concurrency comes from the
loop, and tensor bytes go straight into the Runtime's task pools.
"""

from __future__ import annotations

import asyncio
from dataclasses import dataclass, field
from typing import AsyncIterator, Dict, List

import torch

from ...compression import WireTensor, combine_from_streaming, deserialize_torch_tensor, serialize_torch_tensor, split_for_streaming
from ...compression.base import CompressionType
from ...p2p import P2P, RpcContext, RpcMessage, ServicerBase
from ...utils.logging import get_logger
from ...utils.nested import nested_flatten
from ...utils.serializer import MSGPackSerializer
from .module_backend import ModuleBackend

logger = get_logger(__name__)

# The reference caps unary payloads at 2 MB because its p2pd control channel
# requires it (control.py:36-39); this transport has no such constraint, and on
# a node-local/xGMI control network larger unary messages skip per-chunk
# envelope overhead entirely.
MAX_UNARY_PAYLOAD_SIZE = 32 * 1024 * 1024


@dataclass
class ExpertUIDRequest(RpcMessage):
    uid: str = ""


@dataclass
class ExpertInfoResponse(RpcMessage):
    serialized_info: bytes = b""


@dataclass
class ExpertRequest(RpcMessage):
    uid: str = ""
    tensors: List[WireTensor] = field(default_factory=list)


@dataclass
class ExpertResponse(RpcMessage):
    tensors: List[WireTensor] = field(default_factory=list)


class ConnectionHandler(ServicerBase):
    def __init__(self, module_backends: Dict[str, ModuleBackend]):
        self.module_backends = module_backends

    async def add_handlers(self, p2p: P2P):
        await self.add_p2p_handlers(p2p)
        # fast path: tensors ride the raw frame payload (moe/wire.py), one copy
        # per direction instead of the generic msgpack-wrapped route
        p2p.add_unary_handler("expert::fwd_raw", self._raw_forward)
        p2p.add_unary_handler("expert::bwd_raw", self._raw_backward)

    async def _raw_forward(self, payload: bytes, context: RpcContext) -> bytes:
        from ..wire import pack_tensors, unpack_tensors

        uid, inputs = await self._in_executor(unpack_tensors, payload)
        outputs = await self._process(uid, inputs, backward=False)
        return await self._in_executor(pack_tensors, uid, outputs)

    async def _raw_backward(self, payload: bytes, context: RpcContext) -> bytes:
        from ..wire import pack_tensors, unpack_tensors

        uid, inputs = await self._in_executor(unpack_tensors, payload)
        outputs = await self._process(uid, inputs, backward=True)
        return await self._in_executor(pack_tensors, uid, outputs)

    def _backend(self, uid: str) -> ModuleBackend:
        if uid not in self.module_backends:
            raise KeyError(f"unknown expert uid: {uid}")
        return self.module_backends[uid]

    async def rpc_info(self, request: ExpertUIDRequest, context: RpcContext) -> ExpertInfoResponse:
        info = self._backend(request.uid).get_info()
        return ExpertInfoResponse(serialized_info=MSGPackSerializer.dumps(info))

    async def _process(self, uid: str, tensors: List[torch.Tensor], backward: bool) -> List[torch.Tensor]:
        backend = self._backend(uid)
        pool = backend.backward_pool if backward else backend.forward_pool
        future = pool.submit_task(*tensors)
        outputs = await asyncio.wrap_future(future)
        return list(outputs)

    @staticmethod
    async def _in_executor(func, *args):
        """numpy (de)serialization releases the GIL -- run it off the event loop
        so concurrent requests pipeline instead of serializing on the loop."""
        return await asyncio.get_event_loop().run_in_executor(None, func, *args)

    async def rpc_forward(self, request: ExpertRequest, context: RpcContext) -> ExpertResponse:
        inputs = await self._in_executor(lambda: [deserialize_torch_tensor(t) for t in request.tensors])
        outputs = await self._process(request.uid, inputs, backward=False)
        return ExpertResponse(
            tensors=await self._in_executor(lambda: [serialize_torch_tensor(t.cpu()) for t in outputs])
        )

    async def rpc_backward(self, request: ExpertRequest, context: RpcContext) -> ExpertResponse:
        inputs = await self._in_executor(lambda: [deserialize_torch_tensor(t) for t in request.tensors])
        outputs = await self._process(request.uid, inputs, backward=True)
        return ExpertResponse(
            tensors=await self._in_executor(lambda: [serialize_torch_tensor(t.cpu()) for t in outputs])
        )

    async def rpc_forward_stream(
        self, requests: AsyncIterator[ExpertRequest], context: RpcContext
    ) -> AsyncIterator[ExpertResponse]:
        uid, inputs = await self._gather_stream(requests)
        outputs = await self._process(uid, inputs, backward=False)
        async for response in self._stream_outputs(outputs):
            yield response

    async def rpc_backward_stream(
        self, requests: AsyncIterator[ExpertRequest], context: RpcContext
    ) -> AsyncIterator[ExpertResponse]:
        uid, inputs = await self._gather_stream(requests)
        outputs = await self._process(uid, inputs, backward=True)
        async for response in self._stream_outputs(outputs):
            yield response

    @staticmethod
    async def _gather_stream(requests: AsyncIterator[ExpertRequest]):
        uid = None
        parts: List[WireTensor] = []
        async for request in requests:
            if request.uid:
                uid = request.uid
            parts.extend(request.tensors)
        assert uid is not None, "stream carried no expert uid"
        tensors = []
        chunk_buf: List[WireTensor] = []
        for part in parts:
            if part.chunks and chunk_buf:
                tensors.append(deserialize_torch_tensor(combine_from_streaming(chunk_buf)))
                chunk_buf = []
            chunk_buf.append(part)
        if chunk_buf:
            tensors.append(deserialize_torch_tensor(combine_from_streaming(chunk_buf)))
        return uid, tensors

    @staticmethod
    async def _stream_outputs(outputs: List[torch.Tensor]) -> AsyncIterator[ExpertResponse]:
        for tensor in outputs:
            serialized = serialize_torch_tensor(tensor.cpu(), CompressionType.NONE)
            for part in split_for_streaming(serialized):
                yield ExpertResponse(tensors=[part])
