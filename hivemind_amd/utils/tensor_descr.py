"""Tensor schemas: the descriptor language for expert I/O and compression.

Parity target: reference ``hivemind/utils/tensor_descr.py:26-135``
(``TensorDescriptor``, ``BatchTensorDescriptor`` with a None 0-th dim,
msgpack ext-serializable). CompressionType lives in
``hivemind_amd.compression``; stored here as an int to avoid a cycle.
"""

from __future__ import annotations

from dataclasses import dataclass, asdict
from typing import Optional, Tuple

import torch

from .serializer import MSGPackSerializer

DUMMY_BATCH_SIZE = 3  # batch dimension placeholder used when materializing batch schemas


@dataclass(frozen=True)
class DescriptorBase:
    pass


def _safe_check_pinned(tensor: torch.Tensor) -> bool:
    try:
        return torch.cuda.is_available() and tensor.is_pinned()
    except RuntimeError:
        return False


@dataclass(frozen=True)
class TensorDescriptor(DescriptorBase):
    size: Tuple[int, ...]
    dtype: torch.dtype = torch.float32
    layout: torch.layout = torch.strided
    device: Optional[torch.device] = None
    requires_grad: bool = False
    pin_memory: bool = False
    compression: int = 0  # CompressionType value; int to avoid import cycle

    @property
    def shape(self) -> Tuple[int, ...]:
        return self.size

    def numel(self) -> int:
        n = 1
        for s in self.size:
            n *= s
        return n

    @classmethod
    def from_tensor(cls, tensor: torch.Tensor) -> "TensorDescriptor":
        return cls(
            tuple(tensor.shape),
            tensor.dtype,
            tensor.layout,
            tensor.device,
            tensor.requires_grad,
            _safe_check_pinned(tensor),
        )

    def make_zeros(self, **kwargs) -> torch.Tensor:
        properties = asdict(self)
        properties.pop("compression")
        properties.update(kwargs)
        size = properties.pop("size")
        return torch.zeros(size, **properties)


@MSGPackSerializer.ext_serializable(0x51)
@dataclass(repr=True, frozen=True)
class BatchTensorDescriptor(TensorDescriptor):
    """TensorDescriptor whose 0-th (batch) dimension is variable (None)."""

    def __init__(self, *instance_size: int, **kwargs):
        if len(instance_size) == 1 and isinstance(instance_size[0], (list, tuple)):
            instance_size = tuple(instance_size[0])
        super().__init__((None, *instance_size), **kwargs)  # type: ignore[arg-type]

    @classmethod
    def from_tensor(cls, tensor: torch.Tensor, compression: int = 0) -> "BatchTensorDescriptor":
        return cls(
            *tensor.shape[1:],
            dtype=tensor.dtype,
            layout=tensor.layout,
            device=tensor.device,
            requires_grad=tensor.requires_grad,
            pin_memory=_safe_check_pinned(tensor),
            compression=compression if tensor.is_floating_point() else 0,
        )

    def make_zeros(self, *batch_size: int, **kwargs) -> torch.Tensor:
        assert self.shape[0] is None, "expected batch dimension to be None"
        return super().make_zeros(size=(*batch_size, *self.shape[1:]), **kwargs)

    def packb(self) -> bytes:
        obj = asdict(self)
        obj["size"] = list(self.size[1:])
        obj["dtype"] = str(self.dtype).replace("torch.", "")
        obj["layout"] = str(self.layout)
        obj["device"] = str(self.device) if self.device is not None else None
        return MSGPackSerializer.dumps(obj)

    @classmethod
    def unpackb(cls, data: bytes) -> "BatchTensorDescriptor":
        obj = MSGPackSerializer.loads(data)
        size = obj.pop("size")
        dtype = getattr(torch, obj.pop("dtype"))
        obj.pop("layout", None)
        device = obj.pop("device", None)
        return cls(
            *size,
            dtype=dtype,
            device=torch.device(device) if device is not None else None,
            requires_grad=obj.get("requires_grad", False),
            pin_memory=obj.get("pin_memory", False),
            compression=obj.get("compression", 0),
        )
