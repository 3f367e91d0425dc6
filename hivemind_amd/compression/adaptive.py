"""Adaptive codec selection by size / role / tensor key.

Parity target: reference ``hivemind/compression/adaptive.py:25-66``.
"""

from __future__ import annotations

from typing import Mapping

import torch

from .base import CompressionBase, CompressionInfo, TensorRole, WireTensor


class AdaptiveCompressionBase(CompressionBase):
    def choose_compression(self, info: CompressionInfo) -> CompressionBase:
        raise NotImplementedError

    @property
    def compression_type(self):
        raise AttributeError("adaptive codec: type depends on the tensor")

    def estimate_compression_ratio(self, info: CompressionInfo) -> float:
        return self.choose_compression(info).estimate_compression_ratio(info)

    def compress(self, tensor: torch.Tensor, info: CompressionInfo = CompressionInfo(), allow_inplace: bool = False) -> WireTensor:
        return self.choose_compression(info).compress(tensor, info=info, allow_inplace=allow_inplace)

    def extract(self, serialized: WireTensor) -> torch.Tensor:
        from .serialization import deserialize_torch_tensor

        return deserialize_torch_tensor(serialized)


class SizeAdaptiveCompression(AdaptiveCompressionBase):
    """Use `large` codec for tensors with >= threshold elements, else `small`."""

    def __init__(self, threshold: int, less: CompressionBase, greater_equal: CompressionBase):
        self.threshold, self.less, self.greater_equal = threshold, less, greater_equal

    def choose_compression(self, info: CompressionInfo) -> CompressionBase:
        numel = info.descriptor.numel() if info.descriptor is not None else 0
        return self.greater_equal if numel >= self.threshold else self.less


class RoleAdaptiveCompression(AdaptiveCompressionBase):
    """Pick codec by tensor role (activation / parameter / gradient / optimizer)."""

    def __init__(
        self,
        *,
        activation: CompressionBase = None,
        parameter: CompressionBase = None,
        gradient: CompressionBase = None,
        optimizer: CompressionBase = None,
        default: CompressionBase,
    ):
        self.role_compressions = {
            TensorRole.ACTIVATION: activation or default,
            TensorRole.PARAMETER: parameter or default,
            TensorRole.GRADIENT: gradient or default,
            TensorRole.OPTIMIZER: optimizer or default,
            TensorRole.UNSPECIFIED: default,
        }

    def choose_compression(self, info: CompressionInfo) -> CompressionBase:
        return self.role_compressions[info.role]


class PerTensorCompression(AdaptiveCompressionBase):
    """Explicit codec per tensor key."""

    def __init__(self, tensor_compressions: Mapping):
        self.tensor_compressions = tensor_compressions

    def choose_compression(self, info: CompressionInfo) -> CompressionBase:
        return self.tensor_compressions[info.key]
