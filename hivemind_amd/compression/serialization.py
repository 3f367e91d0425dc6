"""(De)serialization entry points + streaming split/combine.

Parity target: reference ``hivemind/compression/serialization.py:30-77``
(serialize_torch_tensor / deserialize_torch_tensor / deserialize_tensor_stream)
and ``hivemind/utils/streaming.py:17-47`` (split_for_streaming /
combine_from_streaming). Stream chunks default to 256 KiB -- tuned for
localhost/xGMI-node control links rather than the reference's 64 KiB WAN
chunks.
"""

from __future__ import annotations

import dataclasses
from typing import AsyncIterator, Iterable, Iterator, List, Optional

import torch

from .base import CompressionBase, CompressionInfo, CompressionType, NoCompression, WireTensor
from .floating import Float16Compression, ScaledFloat16Compression
from .quantization import BlockwiseQuantization, Quantile8BitQuantization, Uniform8BitQuantization

STREAMING_CHUNK_SIZE_BYTES = 2**22

BASE_COMPRESSION_TYPES = {
    CompressionType.NONE: NoCompression(),
    CompressionType.FLOAT16: Float16Compression(),
    CompressionType.MEANSTD_16BIT: ScaledFloat16Compression(),
    CompressionType.UNIFORM_8BIT: Uniform8BitQuantization(),
    CompressionType.QUANTILE_8BIT: Quantile8BitQuantization(),
    CompressionType.BLOCKWISE_8BIT: BlockwiseQuantization(),
}


def serialize_torch_tensor(
    tensor: torch.Tensor,
    compression_type: CompressionType = CompressionType.NONE,
    info: Optional[CompressionInfo] = None,
    allow_inplace: bool = False,
    **kwargs,
) -> WireTensor:
    assert tensor.device.type in ("cpu", "cuda"), "tensor must be on cpu or gpu"
    if isinstance(compression_type, CompressionBase):
        compression = compression_type
    else:
        compression = BASE_COMPRESSION_TYPES[CompressionType(compression_type)]
    info = info or CompressionInfo.from_tensor(tensor, **kwargs)
    return compression.compress(tensor, info, allow_inplace)


def deserialize_torch_tensor(serialized_tensor: WireTensor) -> torch.Tensor:
    codec = BASE_COMPRESSION_TYPES[CompressionType(serialized_tensor.compression)]
    return codec.extract(serialized_tensor).requires_grad_(serialized_tensor.requires_grad)


def split_for_streaming(serialized_tensor: WireTensor, chunk_size_bytes: int = STREAMING_CHUNK_SIZE_BYTES) -> Iterator[WireTensor]:
    """Split one serialized tensor into metadata-bearing first chunk + raw chunks."""
    buffer = serialized_tensor.buffer
    num_chunks = max(1, (len(buffer) - 1) // chunk_size_bytes + 1)
    first = dataclasses.replace(serialized_tensor, buffer=buffer[:chunk_size_bytes], chunks=num_chunks)
    yield first
    for i in range(1, num_chunks):
        yield WireTensor(buffer=buffer[i * chunk_size_bytes : (i + 1) * chunk_size_bytes])


def combine_from_streaming(stream: Iterable[WireTensor]) -> WireTensor:
    stream = iter(stream)
    first = next(stream)
    chunks = [first.buffer]
    for _ in range(first.chunks - 1):
        chunks.append(next(stream).buffer)
    return dataclasses.replace(first, buffer=b"".join(chunks), chunks=0)


async def deserialize_tensor_stream(stream: AsyncIterator[List[WireTensor]]) -> List[torch.Tensor]:
    """Async stream of WireTensor lists -> tensors (reference serialization.py:50-77)."""
    tensors = []
    tensor_parts: List[WireTensor] = []
    async for parts in stream:
        for part in parts:
            if part.chunks and tensor_parts:
                tensors.append(deserialize_torch_tensor(combine_from_streaming(tensor_parts)))
                tensor_parts = []
            tensor_parts.append(part)
    if tensor_parts:
        tensors.append(deserialize_torch_tensor(combine_from_streaming(tensor_parts)))
    return tensors
