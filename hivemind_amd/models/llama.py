"""Llama-3-style decoder, MI355X-first.

BASELINE.json names "Llama-3-8B hivemind.Optimizer + delayed param update,
bf16, 8 peers, 288 GB HBM sizing" as a benchmark config. Memory budget per
GPU at 8B params: bf16 weights 16 GB + fp32 masters 32 GB + Adam moments
64 GB + gradients 16 GB ~= 128 GB, comfortably inside 288 GB HBM3E -- the
whole DPU pipeline stays on-device (no host offload).

Kernel mapping (hivemind_amd.ops):
* RMSNorm fwd/bwd        -- fused HIP kernel, block-per-row;
* SwiGLU                 -- fused silu(gate)*up elementwise kernel;
* RoPE                   -- fused rotate-half kernel, host-precomputed tables;
* attention              -- torch scaled_dot_product_attention (GQA expanded);
* GEMMs                  -- rocBLAS/hipBLASLt via F.linear (library GEMMs).
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import build_rope_tables, fused_cross_entropy, fused_rmsnorm, fused_rope, fused_swiglu


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    hidden_size: int = 4096
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    intermediate_size: int = 14336
    max_position_embeddings: int = 8192
    rms_norm_eps: float = 1e-5
    rope_base: float = 500000.0
    dtype: torch.dtype = torch.bfloat16

    @classmethod
    def llama_3_8b(cls) -> "LlamaConfig":
        return cls()

    @classmethod
    def llama_1b(cls) -> "LlamaConfig":
        return cls(hidden_size=2048, num_hidden_layers=16, num_attention_heads=32,
                   num_key_value_heads=8, intermediate_size=8192)

    @classmethod
    def tiny(cls) -> "LlamaConfig":
        return cls(vocab_size=512, hidden_size=64, num_hidden_layers=2, num_attention_heads=4,
                   num_key_value_heads=2, intermediate_size=128, max_position_embeddings=128)


class RMSNorm(nn.Module):
    def __init__(self, hidden: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden, dtype=torch.float32))
        self.eps = eps

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return fused_rmsnorm(x, self.weight, self.eps)


class LlamaDecoderLayer(nn.Module):
    def __init__(self, config: LlamaConfig):
        super().__init__()
        h, heads, kv_heads = config.hidden_size, config.num_attention_heads, config.num_key_value_heads
        self.num_heads, self.num_kv_heads = heads, kv_heads
        self.head_dim = h // heads
        dt = config.dtype
        self.q_proj = nn.Linear(h, heads * self.head_dim, bias=False, dtype=dt)
        self.k_proj = nn.Linear(h, kv_heads * self.head_dim, bias=False, dtype=dt)
        self.v_proj = nn.Linear(h, kv_heads * self.head_dim, bias=False, dtype=dt)
        self.o_proj = nn.Linear(heads * self.head_dim, h, bias=False, dtype=dt)
        self.gate_proj = nn.Linear(h, config.intermediate_size, bias=False, dtype=dt)
        self.up_proj = nn.Linear(h, config.intermediate_size, bias=False, dtype=dt)
        self.down_proj = nn.Linear(config.intermediate_size, h, bias=False, dtype=dt)
        self.input_norm = RMSNorm(h, config.rms_norm_eps)
        self.post_attn_norm = RMSNorm(h, config.rms_norm_eps)

    def forward(self, x: torch.Tensor, cos_table: torch.Tensor, sin_table: torch.Tensor) -> torch.Tensor:
        B, S, H = x.shape
        residual = x
        h = self.input_norm(x)
        q = self.q_proj(h).view(B, S, self.num_heads, self.head_dim)
        k = self.k_proj(h).view(B, S, self.num_kv_heads, self.head_dim)
        v = self.v_proj(h).view(B, S, self.num_kv_heads, self.head_dim)
        q = fused_rope(q, cos_table, sin_table)
        k = fused_rope(k, cos_table, sin_table)
        q, k, v = (t.transpose(1, 2) for t in (q, k, v))  # [B, heads, S, hd]
        from ..ops import flash_attention, flash_attention_usable

        if flash_attention_usable(q, k):
            # CDNA4 flash kernels; backward needs H == Hkv, so expand grouped
            # kv heads (autograd sum-reduces dk/dv back through the repeat)
            groups = self.num_heads // self.num_kv_heads
            if groups > 1:
                k = k.repeat_interleave(groups, dim=1)
                v = v.repeat_interleave(groups, dim=1)
            attn = flash_attention(q, k, v, causal=True)
        else:
            attn = F.scaled_dot_product_attention(q, k, v, is_causal=True, enable_gqa=True)
        attn = attn.transpose(1, 2).reshape(B, S, H)
        x = residual + self.o_proj(attn)
        residual = x
        h = self.post_attn_norm(x)
        mlp = self.down_proj(fused_swiglu(self.gate_proj(h), self.up_proj(h)))
        return residual + mlp


class LlamaForCausalLM(nn.Module):
    def __init__(self, config: LlamaConfig):
        super().__init__()
        self.config = config
        dt = config.dtype
        self.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size, dtype=dt)
        self.layers = nn.ModuleList(LlamaDecoderLayer(config) for _ in range(config.num_hidden_layers))
        self.norm = RMSNorm(config.hidden_size, config.rms_norm_eps)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False, dtype=dt)
        self._rope_cache: Optional[tuple] = None
        self.apply(self._init_weights)

    @staticmethod
    def _init_weights(module):
        if isinstance(module, nn.Linear):
            nn.init.normal_(module.weight, std=0.02)
        elif isinstance(module, nn.Embedding):
            nn.init.normal_(module.weight, std=0.02)

    def _rope_tables(self, seq_len: int, device):
        if self._rope_cache is None or self._rope_cache[0] < seq_len or self._rope_cache[1].device != device:
            cos, sin = build_rope_tables(
                max(seq_len, 512), self.config.hidden_size // self.config.num_attention_heads,
                base=self.config.rope_base, device=device,
            )
            self._rope_cache = (cos.shape[0], cos, sin)
        _, cos, sin = self._rope_cache
        return cos[:seq_len].contiguous(), sin[:seq_len].contiguous()

    def forward(self, input_ids: torch.Tensor, labels: Optional[torch.Tensor] = None):
        B, S = input_ids.shape
        cos, sin = self._rope_tables(S, input_ids.device)
        hidden = self.embed_tokens(input_ids)
        for layer in self.layers:
            hidden = layer(hidden, cos, sin)
        hidden = self.norm(hidden)
        logits = self.lm_head(hidden)
        if labels is not None:
            # fused streaming cross-entropy: saves only the per-row logsumexp
            # instead of a [B*S, 128k-vocab] log-softmax (4.2 GB at batch 8 x 2048)
            loss = fused_cross_entropy(
                logits[:, :-1].reshape(-1, self.config.vocab_size).contiguous(), labels[:, 1:].reshape(-1).contiguous()
            )
            return loss, logits
        return logits

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
