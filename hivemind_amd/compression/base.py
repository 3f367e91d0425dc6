"""Tensor codec framework.

Parity target: reference ``hivemind/compression/base.py:17-111``
(``CompressionBase.compress/extract/estimate_compression_ratio``,
``CompressionInfo`` with key/descriptor/role, ``NoCompression``) and the
``CompressionType`` enum from ``proto/runtime.proto:31-39``.

The wire format here is a msgpack dataclass (``WireTensor``) instead of a
protobuf; the GPU path runs the same codecs as HIP kernels via
``hivemind_amd.ops`` with identical semantics (tested against these
torch implementations).
"""

from __future__ import annotations

import dataclasses
from enum import Enum, IntEnum
from typing import Any, List, Optional

import torch

from ..p2p import RpcMessage
from ..utils.tensor_descr import TensorDescriptor


class CompressionType(IntEnum):
    NONE = 0
    MEANSTD_16BIT = 1
    FLOAT16 = 2
    QUANTILE_8BIT = 3
    UNIFORM_8BIT = 4
    BLOCKWISE_8BIT = 5


class TensorRole(Enum):
    ACTIVATION = "activation"
    PARAMETER = "parameter"
    GRADIENT = "gradient"
    OPTIMIZER = "optimizer"
    UNSPECIFIED = "unspecified"


@dataclasses.dataclass(frozen=True)
class CompressionInfo:
    """Everything a codec needs to know about the tensor it compresses."""

    key: Any = None
    descriptor: Optional[TensorDescriptor] = None
    role: TensorRole = TensorRole.UNSPECIFIED
    part_index: int = 0
    part_size: Optional[int] = None

    @classmethod
    def from_tensor(cls, tensor: torch.Tensor, key: Any = None, role: TensorRole = TensorRole.UNSPECIFIED, **kwargs):
        return cls(key=key, descriptor=TensorDescriptor.from_tensor(tensor), role=role, **kwargs)

    def get_part(self, part_index: int, part_size: int) -> "CompressionInfo":
        return dataclasses.replace(self, part_index=part_index, part_size=part_size)


@dataclasses.dataclass
class WireTensor(RpcMessage):
    """Serialized tensor envelope (reference proto/runtime.proto:23-40 Tensor)."""

    buffer: bytes = b""
    size: List[int] = dataclasses.field(default_factory=list)
    dtype: str = ""
    compression: int = 0
    requires_grad: bool = False
    chunks: int = 0  # >0 on the first part of a streamed tensor


_DTYPE_TO_STR = {}
_STR_TO_DTYPE = {}
for _name in ("float32", "float64", "float16", "bfloat16", "int8", "uint8", "int16", "int32", "int64", "bool"):
    _dt = getattr(torch, _name)
    _DTYPE_TO_STR[_dt] = _name
    _STR_TO_DTYPE[_name] = _dt


def dtype_to_str(dtype: torch.dtype) -> str:
    return _DTYPE_TO_STR[dtype]


def str_to_dtype(name: str) -> torch.dtype:
    return _STR_TO_DTYPE[name]


def tensor_to_bytes(tensor: torch.Tensor) -> bytes:
    tensor = tensor.detach().contiguous()
    if tensor.device.type != "cpu":
        tensor = tensor.cpu()
    if tensor.dtype == torch.bfloat16:
        # numpy has no bfloat16: ship raw bits as int16
        return tensor.view(torch.int16).numpy().tobytes()
    return tensor.numpy().tobytes()


def bytes_to_tensor(buffer: bytes, dtype: torch.dtype, size: List[int]) -> torch.Tensor:
    import numpy as np

    if dtype == torch.bfloat16:
        arr = np.frombuffer(buffer, dtype=np.int16).copy()
        return torch.from_numpy(arr).view(torch.bfloat16).reshape(size)
    np_dtype = torch.empty(0, dtype=dtype).numpy().dtype
    arr = np.frombuffer(buffer, dtype=np_dtype).copy()
    return torch.from_numpy(arr).reshape(size)


class CompressionBase:
    """Codec interface: compress a tensor to a WireTensor and extract it back."""

    compression_type: CompressionType

    def compress(self, tensor: torch.Tensor, info: CompressionInfo = CompressionInfo(), allow_inplace: bool = False) -> WireTensor:
        raise NotImplementedError

    def extract(self, serialized: WireTensor) -> torch.Tensor:
        raise NotImplementedError

    def estimate_compression_ratio(self, info: CompressionInfo) -> float:
        """Compressed size / original size."""
        raise NotImplementedError

    def __repr__(self):
        return f"{self.__class__.__name__}()"


class NoCompression(CompressionBase):
    compression_type = CompressionType.NONE

    def compress(self, tensor: torch.Tensor, info: CompressionInfo = CompressionInfo(), allow_inplace: bool = False) -> WireTensor:
        return WireTensor(
            buffer=tensor_to_bytes(tensor),
            size=list(tensor.shape),
            dtype=dtype_to_str(tensor.dtype),
            compression=int(self.compression_type),
            requires_grad=tensor.requires_grad,
        )

    def extract(self, serialized: WireTensor) -> torch.Tensor:
        return bytes_to_tensor(serialized.buffer, str_to_dtype(serialized.dtype), serialized.size)

    def estimate_compression_ratio(self, info: CompressionInfo) -> float:
        return 1.0
