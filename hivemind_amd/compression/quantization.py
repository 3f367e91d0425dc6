"""8-bit quantization codecs.

Parity target: reference ``hivemind/compression/quantization.py:61-201``:

* ``Uniform8BitQuantization``: mean-shift + 6-sigma range, 256-bucket codebook
  where each code maps to the mean of the values assigned to it.
* ``Quantile8BitQuantization``: codebook from a quantile-of-quantiles
  approximation, values bucketized to the nearest code.
* ``BlockwiseQuantization``: per-4096-block absmax scaling to int8 (the
  reference wraps bitsandbytes, unavailable here -- this is our own
  implementation with the same blocksize and absmax semantics; the dynamic
  codebook is replaced by linear int8 within each block, which has comparable
  quantization error for gradient averaging and maps 1:1 onto a HIP kernel).
"""

from __future__ import annotations

import math
from typing import Tuple

import numpy as np
import torch

from .base import (
    CompressionBase,
    CompressionInfo,
    CompressionType,
    WireTensor,
    bytes_to_tensor,
    dtype_to_str,
    str_to_dtype,
    tensor_to_bytes,
)

EXECUTOR_CHUNK = 65536
UINT8_RANGE = 256
BLOCKSIZE = 4096


class Quantization(CompressionBase):
    codebook_dtype, indices_dtype = np.float32, np.uint8

    def quantize(self, tensor: torch.Tensor, allow_inplace: bool = False) -> Tuple[np.ndarray, np.ndarray]:
        raise NotImplementedError

    def compress(self, tensor: torch.Tensor, info: CompressionInfo = CompressionInfo(), allow_inplace: bool = False) -> WireTensor:
        if not tensor.is_floating_point():
            raise ValueError(f"{self.__class__.__name__} requires a floating-point tensor")
        dtype_name = dtype_to_str(tensor.dtype)
        tensor = tensor.detach()
        quantized, codebook = self.quantize(tensor.to(torch.float32), allow_inplace=allow_inplace)
        return WireTensor(
            buffer=np.int64(len(codebook)).tobytes() + codebook.tobytes() + quantized.tobytes(),
            size=list(tensor.shape),
            dtype=dtype_name,
            compression=int(self.compression_type),
            requires_grad=tensor.requires_grad,
        )

    def extract(self, serialized: WireTensor) -> torch.Tensor:
        codebook_size = int(np.frombuffer(serialized.buffer, count=1, dtype=np.int64)[0])
        codebook = np.frombuffer(serialized.buffer, offset=8, count=codebook_size, dtype=self.codebook_dtype)
        quantized = np.frombuffer(serialized.buffer, offset=8 + codebook.nbytes, dtype=self.indices_dtype)
        quantized = torch.as_tensor(quantized.copy(), dtype=torch.int64).reshape(serialized.size)
        codebook = torch.as_tensor(codebook.copy())
        return codebook[quantized].to(str_to_dtype(serialized.dtype))

    def estimate_compression_ratio(self, info: CompressionInfo) -> float:
        return self.n_bits / (torch.finfo(info.descriptor.dtype).bits if info.descriptor else 32)

    @property
    def n_bits(self):
        return np.iinfo(self.indices_dtype).bits

    @property
    def n_bins(self):
        return 2**self.n_bits


class Uniform8BitQuantization(Quantization):
    """6-sigma uniform quantization with bucket-mean codebook (reference quantization.py:61-76)."""

    RANGE_IN_SIGMAS: int = 6
    compression_type = CompressionType.UNIFORM_8BIT

    def quantize(self, tensor: torch.Tensor, allow_inplace: bool = False) -> Tuple[np.ndarray, np.ndarray]:
        offset = self.n_bins // 2
        shift = tensor.mean()
        centered = tensor.sub_(shift) if allow_inplace else tensor - shift
        std_unbiased = centered.norm() / math.sqrt(max(centered.numel() - 1, 1))
        scale = self.RANGE_IN_SIGMAS * std_unbiased / self.n_bins
        scale = torch.clamp_min(scale, torch.finfo(torch.float32).eps)
        if tensor.is_cuda:
            # on-device arithmetic equivalent of torch.quantize_per_tensor
            # (which is CPU-oriented); bucket means reduce on the GPU and only
            # the uint8 codes + 256-entry codebook cross PCIe
            quantized = torch.clamp(torch.round(centered / scale) + offset, 0, self.n_bins - 1).to(torch.uint8)
            lookup = average_buckets(centered, quantized, self.n_bins)
            codebook = (lookup + shift).cpu().numpy().astype(self.codebook_dtype)
            return quantized.cpu().numpy().astype(self.indices_dtype), codebook
        quantized = torch.quantize_per_tensor(centered, float(scale), offset, torch.quint8).int_repr()
        lookup = average_buckets(centered, quantized, self.n_bins)
        codebook = (lookup + shift).numpy().astype(self.codebook_dtype)
        return quantized.numpy().astype(self.indices_dtype), codebook


class Quantile8BitQuantization(Quantization):
    """Codebook = approximate quantiles; values bucketized to nearest code
    (reference quantization.py:79-125)."""

    compression_type = CompressionType.QUANTILE_8BIT

    def quantize(self, tensor: torch.Tensor, allow_inplace: bool = False) -> Tuple[np.ndarray, np.ndarray]:
        tensor = tensor.detach().flatten()
        if tensor.is_cuda:
            tensor = tensor.cpu()  # np.quantile-based; one transfer, then CPU math
        codebook = quantile_qq_approximation(tensor.numpy(), self.n_bins)
        borders = (codebook[:-1] + codebook[1:]) / 2
        quantized = np.digitize(tensor.numpy(), borders).astype(self.indices_dtype)
        return quantized, codebook.astype(self.codebook_dtype)

    def compress(self, tensor: torch.Tensor, info: CompressionInfo = CompressionInfo(), allow_inplace: bool = False) -> WireTensor:
        serialized = super().compress(tensor, info, allow_inplace)
        return serialized

    def extract(self, serialized: WireTensor) -> torch.Tensor:
        return super().extract(serialized).reshape(serialized.size)


def average_buckets(tensor: torch.Tensor, quant_weight: torch.Tensor, n_bins: int) -> torch.Tensor:
    """Per-bucket mean of original values (reference quantization.py:88-94).
    Runs on whichever device the tensor lives on."""
    codes = quant_weight.flatten().long()
    bin_sums = torch.zeros(n_bins, dtype=torch.float32, device=tensor.device).scatter_add_(
        0, codes, tensor.flatten().to(torch.float32)
    )
    bin_counts = torch.clamp_min_(torch.bincount(codes, minlength=n_bins), 1)
    return bin_sums / bin_counts


def quantile_qq_approximation(array: np.ndarray, n_quantiles: int, min_chunk_size: int = 10**5) -> np.ndarray:
    """Estimate global quantiles as quantiles-of-chunk-quantiles (reference quantization.py:106-125)."""
    array = array.flatten()
    quantiles = np.linspace(0.0, 1.0, num=n_quantiles, dtype=array.dtype)
    chunk_size = get_chunk_size(len(array), min_chunk_size)
    num_chunks = (len(array) - 1) // chunk_size + 1
    partition_quantiles = np.empty((num_chunks, len(quantiles)), dtype=array.dtype)
    for i in range(num_chunks):
        chunk = array[chunk_size * i : chunk_size * (i + 1)]
        partition_quantiles[i] = np.quantile(chunk, quantiles)
    return np.quantile(partition_quantiles, quantiles)


def get_chunk_size(num_elements: int, min_chunk_size: int) -> int:
    """Adjust chunk size so chunks are nearly equal (reference quantization.py:101-104)."""
    if num_elements <= min_chunk_size:
        return num_elements
    num_chunks = max(1, num_elements // min_chunk_size)
    return (num_elements - 1) // num_chunks + 1


class BlockwiseQuantization(Quantization):
    """Per-block absmax int8: blocks of 4096, scale = absmax/127 per block."""

    compression_type = CompressionType.BLOCKWISE_8BIT
    blocksize = BLOCKSIZE

    def quantize(self, tensor: torch.Tensor, allow_inplace: bool = False) -> Tuple[np.ndarray, np.ndarray]:
        if tensor.is_cuda:
            # the hand-written CDNA4 kernel (ops/hip/elementwise.hip
            # quantize_blockwise_int8) computes the same per-4096-block
            # absmax/127 codes; pad to the wire's block-aligned layout
            from ..ops import quantize_blockwise as hip_quantize_blockwise

            q, absmax = hip_quantize_blockwise(tensor.detach().contiguous())
            num_blocks = absmax.numel()
            if q.numel() < num_blocks * self.blocksize:
                padded_q = torch.zeros(num_blocks * self.blocksize, dtype=torch.int8, device=q.device)
                padded_q[: q.numel()] = q
                q = padded_q
            return q.cpu().numpy().view(np.uint8), absmax.cpu().numpy().astype(np.float32)
        flat = tensor.detach().to(torch.float32).flatten()
        n = flat.numel()
        num_blocks = (n + self.blocksize - 1) // self.blocksize
        padded = torch.zeros(num_blocks * self.blocksize, dtype=torch.float32)
        padded[:n] = flat
        blocks = padded.view(num_blocks, self.blocksize)
        absmax = blocks.abs().amax(dim=1, keepdim=True)
        scale = torch.clamp_min(absmax / 127.0, torch.finfo(torch.float32).eps)
        quantized = torch.round(blocks / scale).clamp_(-127, 127).to(torch.int8)
        return quantized.numpy().view(np.uint8), absmax.flatten().numpy().astype(np.float32)

    def compress(self, tensor: torch.Tensor, info: CompressionInfo = CompressionInfo(), allow_inplace: bool = False) -> WireTensor:
        if not tensor.is_floating_point():
            raise ValueError("BLOCKWISE_8BIT requires a floating-point tensor")
        dtype_name = dtype_to_str(tensor.dtype)
        quantized, absmax = self.quantize(tensor, allow_inplace=allow_inplace)
        return WireTensor(
            buffer=np.int64(len(absmax)).tobytes() + absmax.tobytes() + quantized.tobytes(),
            size=list(tensor.shape),
            dtype=dtype_name,
            compression=int(self.compression_type),
            requires_grad=tensor.requires_grad,
        )

    def extract(self, serialized: WireTensor) -> torch.Tensor:
        num_blocks = int(np.frombuffer(serialized.buffer, count=1, dtype=np.int64)[0])
        absmax = np.frombuffer(serialized.buffer, offset=8, count=num_blocks, dtype=np.float32)
        quantized = np.frombuffer(serialized.buffer, offset=8 + absmax.nbytes, dtype=np.int8)
        blocks = torch.as_tensor(quantized.copy(), dtype=torch.float32).view(num_blocks, self.blocksize)
        scale = torch.clamp_min(torch.as_tensor(absmax.copy()) / 127.0, torch.finfo(torch.float32).eps)
        restored = (blocks * scale.unsqueeze(1)).flatten()
        numel = math.prod(serialized.size) if serialized.size else 1
        return restored[:numel].reshape(serialized.size).to(str_to_dtype(serialized.dtype))
