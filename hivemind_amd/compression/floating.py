"""16-bit float codecs.

Parity target: reference ``hivemind/compression/floating.py:10-103``
(``Float16Compression`` clamps to the fp16 range before casting;
``ScaledFloat16Compression`` normalizes per-last-axis mean/std and appends the
fp32 statistics to the payload).
"""

from __future__ import annotations

import math

import torch

from .base import CompressionBase, CompressionInfo, CompressionType, WireTensor, dtype_to_str, tensor_to_bytes, bytes_to_tensor

FP16_MAX = 65504.0


class Float16Compression(CompressionBase):
    compression_type = CompressionType.FLOAT16

    def compress(self, tensor: torch.Tensor, info: CompressionInfo = CompressionInfo(), allow_inplace: bool = False) -> WireTensor:
        if not tensor.is_floating_point():
            raise ValueError("FLOAT16 compression requires a floating-point tensor")
        dtype_name = dtype_to_str(tensor.dtype)
        tensor = tensor.detach()
        clipped = tensor.to(torch.float32).clamp_(-FP16_MAX, FP16_MAX).to(torch.float16)
        return WireTensor(
            buffer=tensor_to_bytes(clipped),
            size=list(tensor.shape),
            dtype=dtype_name,
            compression=int(self.compression_type),
            requires_grad=tensor.requires_grad,
        )

    def extract(self, serialized: WireTensor) -> torch.Tensor:
        from .base import str_to_dtype

        original_dtype = str_to_dtype(serialized.dtype)
        half = bytes_to_tensor(serialized.buffer, torch.float16, serialized.size)
        return half.to(original_dtype)

    def estimate_compression_ratio(self, info: CompressionInfo) -> float:
        return 16.0 / (torch.finfo(info.descriptor.dtype).bits if info.descriptor else 32)


class ScaledFloat16Compression(Float16Compression):
    """MEANSTD_16BIT: per-last-axis standardization, then fp16 (reference floating.py:43-95)."""

    compression_type = CompressionType.MEANSTD_16BIT
    FP32_EPS = 1e-8

    def compress(self, tensor: torch.Tensor, info: CompressionInfo = CompressionInfo(), allow_inplace: bool = False) -> WireTensor:
        if not tensor.is_floating_point():
            raise ValueError("MEANSTD_16BIT compression requires a floating-point tensor")
        dtype_name = dtype_to_str(tensor.dtype)
        tensor = tensor.detach().to(torch.float32)
        means = torch.mean(tensor, dim=-1, keepdim=True)
        stds = torch.std(tensor, dim=-1, keepdim=True) + self.FP32_EPS
        normalized = (tensor - means) / stds
        half = normalized.clamp_(-FP16_MAX, FP16_MAX).to(torch.float16)
        payload = tensor_to_bytes(half) + tensor_to_bytes(means) + tensor_to_bytes(stds)
        return WireTensor(
            buffer=payload,
            size=list(tensor.shape),
            dtype=dtype_name,
            compression=int(self.compression_type),
            requires_grad=tensor.requires_grad,
        )

    def extract(self, serialized: WireTensor) -> WireTensor:
        from .base import str_to_dtype

        original_dtype = str_to_dtype(serialized.dtype)
        size = list(serialized.size)
        numel = math.prod(size) if size else 1
        stats_size = list(size)
        if stats_size:
            stats_size[-1] = 1
        stats_numel = math.prod(stats_size) if stats_size else 1
        half_bytes = numel * 2
        stats_bytes = stats_numel * 4
        half = bytes_to_tensor(serialized.buffer[:half_bytes], torch.float16, size)
        means = bytes_to_tensor(serialized.buffer[half_bytes : half_bytes + stats_bytes], torch.float32, stats_size)
        stds = bytes_to_tensor(serialized.buffer[half_bytes + stats_bytes :], torch.float32, stats_size)
        return (half.to(torch.float32) * stds + means).to(original_dtype)
