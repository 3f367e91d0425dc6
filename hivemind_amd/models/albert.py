"""ALBERT-style encoder, MI355X-first.

This is the flagship benchmark model (collaborative ALBERT
training with hivemind.Optimizer). Architecture follows ALBERT: factorized
embeddings (vocab -> 128 -> hidden) and one transformer layer whose weights
are shared across all 12 depths (reference example: examples/albert/).

MI355X mapping:
* weights and activations in bf16 -- plain GEMMs go through rocBLAS/hipBLASLt
  via torch.nn.functional.linear (library GEMMs, per the build rules);
* everything between the GEMMs is fused HIP kernels from hivemind_amd.ops:
  residual+LayerNorm (one wave per row) and bias+GELU, eliminating 5
  memory-bound elementwise passes per layer;
* attention uses torch scaled_dot_product_attention (MIOpen/AOTriton-backed
  on ROCm); a hand-written CDNA4 flash kernel is a planned upgrade;
* the optimizer path keeps fp32 masters + Adam state in HBM and updates them
  with the fused AdamW kernel (ops.FusedAdamW).
"""

from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import flash_attention, flash_attention_usable, fused_bias_gelu, fused_cross_entropy, fused_layernorm


@dataclass
class AlbertConfig:
    vocab_size: int = 30000
    embedding_size: int = 128
    hidden_size: int = 768
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    intermediate_size: int = 3072
    max_position_embeddings: int = 512
    type_vocab_size: int = 2
    layer_norm_eps: float = 1e-12
    dtype: torch.dtype = torch.bfloat16

    @classmethod
    def base(cls) -> "AlbertConfig":
        return cls()

    @classmethod
    def large(cls) -> "AlbertConfig":
        return cls(hidden_size=1024, num_hidden_layers=24, num_attention_heads=16, intermediate_size=4096)

    @classmethod
    def tiny(cls) -> "AlbertConfig":
        """For CPU tests."""
        return cls(vocab_size=512, embedding_size=32, hidden_size=64, num_hidden_layers=2,
                   num_attention_heads=4, intermediate_size=128, max_position_embeddings=64)


class FusedLayerNorm(nn.Module):
    """LayerNorm with fp32 affine params, backed by the HIP kernel on GPU."""

    def __init__(self, hidden: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden, dtype=torch.float32))
        self.bias = nn.Parameter(torch.zeros(hidden, dtype=torch.float32))
        self.eps = eps

    def forward(self, x: torch.Tensor, residual: Optional[torch.Tensor] = None) -> torch.Tensor:
        return fused_layernorm(x, self.weight, self.bias, residual=residual, eps=self.eps)


class AlbertLayer(nn.Module):
    """One shared transformer layer: MHA + FFN with fused epilogues."""

    def __init__(self, config: AlbertConfig):
        super().__init__()
        h, heads = config.hidden_size, config.num_attention_heads
        assert h % heads == 0
        self.num_heads, self.head_dim = heads, h // heads
        dt = config.dtype
        self.qkv = nn.Linear(h, 3 * h, dtype=dt)
        self.attn_out = nn.Linear(h, h, dtype=dt)
        self.attn_norm = FusedLayerNorm(h, config.layer_norm_eps)
        # FFN: up-projection bias is fused into the GELU kernel
        self.ffn_up_weight = nn.Parameter(torch.empty(config.intermediate_size, h, dtype=dt))
        self.ffn_up_bias = nn.Parameter(torch.zeros(config.intermediate_size, dtype=dt))
        self.ffn_down = nn.Linear(config.intermediate_size, h, dtype=dt)
        self.ffn_norm = FusedLayerNorm(h, config.layer_norm_eps)
        nn.init.normal_(self.ffn_up_weight, std=0.02)

    def forward(self, x: torch.Tensor, attn_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        B, S, H = x.shape
        qkv = self.qkv(x).view(B, S, 3, self.num_heads, self.head_dim)
        q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))  # [B, heads, S, hd]
        if attn_mask is None and flash_attention_usable(q, k):
            attn = flash_attention(q, k, v)  # hand-written CDNA4 kernels
        else:
            attn = F.scaled_dot_product_attention(q, k, v, attn_mask=attn_mask)
        attn = attn.transpose(1, 2).reshape(B, S, H)
        x = self.attn_norm(self.attn_out(attn), residual=x)
        up = F.linear(x, self.ffn_up_weight)  # rocBLAS GEMM, no bias
        act = fused_bias_gelu(up, self.ffn_up_bias)  # fused bias+gelu HIP kernel
        x = self.ffn_norm(self.ffn_down(act), residual=x)
        return x


class AlbertModel(nn.Module):
    def __init__(self, config: AlbertConfig):
        super().__init__()
        self.config = config
        dt = config.dtype
        self.word_embeddings = nn.Embedding(config.vocab_size, config.embedding_size, dtype=dt)
        self.position_embeddings = nn.Embedding(config.max_position_embeddings, config.embedding_size, dtype=dt)
        self.embedding_norm = FusedLayerNorm(config.embedding_size, config.layer_norm_eps)
        self.embedding_projection = nn.Linear(config.embedding_size, config.hidden_size, dtype=dt)
        self.layer = AlbertLayer(config)  # weights shared across depth (ALBERT)
        self.apply(self._init_weights)

    @staticmethod
    def _init_weights(module):
        if isinstance(module, nn.Linear):
            nn.init.normal_(module.weight, std=0.02)
            if module.bias is not None:
                nn.init.zeros_(module.bias)
        elif isinstance(module, nn.Embedding):
            nn.init.normal_(module.weight, std=0.02)

    def forward(self, input_ids: torch.Tensor, attn_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        B, S = input_ids.shape
        positions = torch.arange(S, device=input_ids.device).unsqueeze(0)
        emb = self.word_embeddings(input_ids) + self.position_embeddings(positions)
        emb = self.embedding_norm(emb)
        hidden = self.embedding_projection(emb)
        for _ in range(self.config.num_hidden_layers):
            hidden = self.layer(hidden, attn_mask)
        return hidden


class AlbertForMaskedLM(nn.Module):
    """ALBERT with an MLM head (embedding-tied decoder), the benchmark model."""

    def __init__(self, config: AlbertConfig):
        super().__init__()
        self.config = config
        dt = config.dtype
        self.albert = AlbertModel(config)
        self.mlm_dense = nn.Linear(config.hidden_size, config.embedding_size, dtype=dt)
        self.mlm_norm = FusedLayerNorm(config.embedding_size, config.layer_norm_eps)
        self.mlm_bias = nn.Parameter(torch.zeros(config.vocab_size, dtype=dt))

    def forward(self, input_ids: torch.Tensor, labels: Optional[torch.Tensor] = None):
        hidden = self.albert(input_ids)
        x = self.mlm_dense(hidden)
        x = F.gelu(x, approximate="tanh")
        x = self.mlm_norm(x)
        # logits stay in bf16: fused_cross_entropy streams them twice and saves
        # only the per-row logsumexp; torch's F.cross_entropy would keep a
        # [B*S, vocab] log-softmax (~3.9 GB at batch 128) alive for backward
        logits = F.linear(x, self.albert.word_embeddings.weight, self.mlm_bias)
        if labels is not None:
            loss = fused_cross_entropy(logits.view(-1, self.config.vocab_size), labels.view(-1))
            return loss, logits
        return logits

    def num_parameters(self) -> int:
        return sum(p.numel() for p in self.parameters())
