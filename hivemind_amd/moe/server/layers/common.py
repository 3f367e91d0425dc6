"""Built-in expert layers (reference hivemind/moe/server/layers/common.py).

The ``ffn`` expert is the benchmark workhorse (BASELINE.md MoE throughput):
Linear(h, 4h) -> gelu -> Linear(4h, h) -> LayerNorm(x + ffn). On GPU its
GELU and LayerNorm run through the fused CDNA4 kernels (hivemind_amd.ops);
the GEMMs use rocBLAS/hipBLASLt via nn.Linear.
"""

from __future__ import annotations

import torch
import torch.nn as nn

import os

from ....ops import fused_bias_gelu, fused_layernorm, fused_linear_gelu
from .custom_experts import register_expert_class

# hand-written MFMA path for the expert up-projection (fused GEMM+bias+GELU);
# set HIVEMIND_AMD_MFMA_EXPERT=0 to force the hipBLASLt + fused-epilogue path
_USE_MFMA_EXPERT = os.environ.get("HIVEMIND_AMD_MFMA_EXPERT", "1") != "0"


def sample_ffn_input(batch_size: int, hidden_dim: int) -> torch.Tensor:
    return torch.empty((batch_size, hidden_dim))


@register_expert_class("ffn", sample_ffn_input)
class FeedforwardBlock(nn.Module):
    def __init__(self, hid_dim: int):
        super().__init__()
        self.ffn_up_weight = nn.Parameter(torch.empty(4 * hid_dim, hid_dim))
        self.ffn_up_bias = nn.Parameter(torch.zeros(4 * hid_dim))
        self.ffn_down = nn.Linear(4 * hid_dim, hid_dim)
        self.layer_norm_weight = nn.Parameter(torch.ones(hid_dim, dtype=torch.float32))
        self.layer_norm_bias = nn.Parameter(torch.zeros(hid_dim, dtype=torch.float32))
        nn.init.normal_(self.ffn_up_weight, std=0.02)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        weight = self.ffn_up_weight.to(x.dtype)
        bias = self.ffn_up_bias.to(x.dtype)
        if _USE_MFMA_EXPERT and x.is_cuda and x.dtype == torch.bfloat16:
            act = fused_linear_gelu(x, weight, bias)  # one MFMA kernel, fused epilogue
        else:
            act = fused_bias_gelu(torch.nn.functional.linear(x, weight), bias)
        down = self.ffn_down.to(x.dtype)(act) if self.ffn_down.weight.dtype != x.dtype else self.ffn_down(act)
        return fused_layernorm(down, self.layer_norm_weight, self.layer_norm_bias, residual=x)


def sample_transformer_input(batch_size: int, hidden_dim: int) -> torch.Tensor:
    return torch.empty((batch_size, 128, hidden_dim))


@register_expert_class("transformer", sample_transformer_input)
class TransformerEncoderLayer(nn.Module):
    """A single transformer encoder layer expert (reference common.py:33-80)."""

    def __init__(self, hid_dim: int, num_heads: int = 8):
        super().__init__()
        self.self_attn = nn.MultiheadAttention(hid_dim, num_heads, batch_first=True)
        self.linear1 = nn.Linear(hid_dim, 4 * hid_dim)
        self.linear2 = nn.Linear(4 * hid_dim, hid_dim)
        self.norm1 = nn.LayerNorm(hid_dim)
        self.norm2 = nn.LayerNorm(hid_dim)
        self.activation = nn.GELU()

    def forward(self, src: torch.Tensor) -> torch.Tensor:
        attn_out, _ = self.self_attn(src, src, src, need_weights=False)
        src = self.norm1(src + attn_out)
        ff = self.linear2(self.activation(self.linear1(src)))
        return self.norm2(src + ff)


def sample_nop_input(batch_size: int, hidden_dim: int) -> torch.Tensor:
    return torch.empty((batch_size, hidden_dim))


@register_expert_class("nop", sample_nop_input)
class NopExpert(nn.Module):
    """Identity expert for communication benchmarks (reference common.py)."""

    def __init__(self, hid_dim: int):
        super().__init__()
        self.scale = nn.Parameter(torch.ones(1))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return x * self.scale


class DeterministicDropout(nn.Module):
    """Dropout driven by an explicit mask input (reference layers/dropout.py)."""

    def __init__(self, drop_prob: float):
        super().__init__()
        self.keep_prob = 1.0 - drop_prob

    def forward(self, x: torch.Tensor, mask: torch.Tensor) -> torch.Tensor:
        if self.training:
            return x * mask.to(x.dtype) / self.keep_prob
        return x


def sample_det_dropout_input(batch_size: int, hidden_dim: int):
    return torch.empty((batch_size, hidden_dim)), torch.randint(0, 2, (batch_size, hidden_dim))


@register_expert_class("det_dropout", sample_det_dropout_input)
class DeterministicDropoutNetwork(nn.Module):
    def __init__(self, hid_dim: int, dropout_prob: float = 0.2):
        super().__init__()
        self.linear_in = nn.Linear(hid_dim, 2 * hid_dim)
        self.activation = nn.ReLU()
        self.dropout = DeterministicDropout(dropout_prob)
        self.linear_out = nn.Linear(2 * hid_dim, hid_dim)

    def forward(self, x: torch.Tensor, mask: torch.Tensor) -> torch.Tensor:
        x = self.linear_in(x * mask.to(x.dtype))
        return self.linear_out(self.activation(x))
