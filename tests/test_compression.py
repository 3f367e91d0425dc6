"""Codec round-trip tests with error tolerances (reference tests/test_compression.py:33-65)."""

import pytest
import torch

from hivemind_amd.compression import (
    CompressionType,
    combine_from_streaming,
    deserialize_torch_tensor,
    serialize_torch_tensor,
    split_for_streaming,
)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_tensor_compression_roundtrip(dtype):
    torch.manual_seed(0)
    X = torch.randn(40, 512, dtype=torch.float32).to(dtype)
    error_bounds = {
        CompressionType.NONE: 0,
        CompressionType.FLOAT16: 1e-3,
        CompressionType.MEANSTD_16BIT: 1.1,
        CompressionType.UNIFORM_8BIT: 0.1,
        CompressionType.QUANTILE_8BIT: 0.2,
        CompressionType.BLOCKWISE_8BIT: 0.05,
    }
    for compression_type, bound in error_bounds.items():
        restored = deserialize_torch_tensor(serialize_torch_tensor(X, compression_type))
        assert restored.shape == X.shape
        assert restored.dtype == X.dtype
        err = (restored.float() - X.float()).abs().mean().item()
        assert err <= max(bound, 1e-2 if dtype == torch.bfloat16 and bound == 0 else bound), (
            compression_type,
            err,
        )


def test_nan_and_inf_fp16():
    X = torch.tensor([1e9, -1e9, 0.0, 3.14])
    restored = deserialize_torch_tensor(serialize_torch_tensor(X, CompressionType.FLOAT16))
    assert torch.isfinite(restored).all()  # clamped, not inf


def test_int_tensor_passthrough():
    X = torch.arange(100, dtype=torch.int64)
    restored = deserialize_torch_tensor(serialize_torch_tensor(X, CompressionType.NONE))
    assert torch.equal(restored, X)
    with pytest.raises(ValueError):
        serialize_torch_tensor(X, CompressionType.FLOAT16)


def test_streaming_split_combine():
    X = torch.randn(1000, 100)
    serialized = serialize_torch_tensor(X, CompressionType.FLOAT16)
    parts = list(split_for_streaming(serialized, chunk_size_bytes=16384))
    assert len(parts) > 1
    assert parts[0].chunks == len(parts)
    combined = combine_from_streaming(parts)
    restored = deserialize_torch_tensor(combined)
    assert torch.allclose(restored, X, rtol=1e-2, atol=1e-2)


def test_requires_grad_preserved():
    X = torch.randn(10, requires_grad=True)
    restored = deserialize_torch_tensor(serialize_torch_tensor(X, CompressionType.NONE))
    assert restored.requires_grad


@pytest.mark.parametrize(
    "compression_type",
    [CompressionType.UNIFORM_8BIT, CompressionType.BLOCKWISE_8BIT, CompressionType.QUANTILE_8BIT],
)
def test_quantization_preserves_average(compression_type):
    """Averaging across quantized peers must stay near the true average."""
    torch.manual_seed(1)
    tensors = [torch.randn(8192) for _ in range(4)]
    restored = [
        deserialize_torch_tensor(serialize_torch_tensor(t, compression_type)) for t in tensors
    ]
    true_avg = torch.stack(tensors).mean(0)
    approx_avg = torch.stack(restored).mean(0)
    assert (true_avg - approx_avg).abs().mean() < 0.05


def test_adaptive_compression_selection_and_roundtrip():
    """Size / role / per-key adaptive codecs pick the right child codec and
    round-trip through the wire format (reference test: adaptive.py:25-66)."""
    from hivemind_amd.compression import (
        Float16Compression,
        NoCompression,
        Uniform8BitQuantization,
        deserialize_torch_tensor,
    )
    from hivemind_amd.compression.adaptive import (
        PerTensorCompression,
        RoleAdaptiveCompression,
        SizeAdaptiveCompression,
    )
    from hivemind_amd.compression.base import CompressionInfo, CompressionType, TensorRole

    small = torch.randn(10)
    large = torch.randn(10_000)

    size_adaptive = SizeAdaptiveCompression(
        threshold=2**10, less=NoCompression(), greater_equal=Float16Compression()
    )
    w_small = size_adaptive.compress(small, CompressionInfo.from_tensor(small))
    w_large = size_adaptive.compress(large, CompressionInfo.from_tensor(large))
    assert w_small.compression == CompressionType.NONE
    assert w_large.compression == CompressionType.FLOAT16
    assert torch.allclose(deserialize_torch_tensor(w_small), small)
    assert torch.allclose(deserialize_torch_tensor(w_large), large, atol=1e-2)

    role_adaptive = RoleAdaptiveCompression(
        gradient=Uniform8BitQuantization(), parameter=Float16Compression(), default=NoCompression()
    )
    g = role_adaptive.compress(large, CompressionInfo.from_tensor(large, role=TensorRole.GRADIENT))
    p = role_adaptive.compress(large, CompressionInfo.from_tensor(large, role=TensorRole.PARAMETER))
    u = role_adaptive.compress(large, CompressionInfo.from_tensor(large, role=TensorRole.UNSPECIFIED))
    assert g.compression == CompressionType.UNIFORM_8BIT
    assert p.compression == CompressionType.FLOAT16
    assert u.compression == CompressionType.NONE
    assert (deserialize_torch_tensor(g) - large).abs().mean() < 0.05  # int8 quantization error

    per_tensor = PerTensorCompression({"a": NoCompression(), "b": Float16Compression()})
    wa = per_tensor.compress(small, CompressionInfo.from_tensor(small, key="a"))
    wb = per_tensor.compress(small, CompressionInfo.from_tensor(small, key="b"))
    assert wa.compression == CompressionType.NONE and wb.compression == CompressionType.FLOAT16
