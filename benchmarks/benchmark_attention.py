#!/usr/bin/env python3
"""Flash-attention A/B: hand-written CDNA4 kernels vs torch sdpa (AOTriton).

Round-1 profile showed AOTriton's sdpa backward at 18.6% of the ALBERT step
(profiles/albert_kernels.md) -- this measures the replacement on the same
shapes. Prints one JSON line per (shape, impl, direction).
"""

import json
import sys
import time

import torch
import torch.nn.functional as F

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from hivemind_amd.ops import flash_attention  # noqa: E402


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    assert torch.cuda.is_available()
    shapes = [
        ("albert-base-b128", 128, 12, 512, 64, False),
        ("albert-large-b64", 64, 16, 512, 64, False),
        ("llama-b8", 8, 32, 512, 128, True),
    ]
    for name, B, H, S, D, causal in shapes:
        torch.manual_seed(0)
        q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
        k = torch.randn_like(q, requires_grad=True)
        v = torch.randn_like(q, requires_grad=True)
        d_out = torch.randn_like(q)
        flops_fwd = 4 * B * H * S * S * D * (0.5 if causal else 1.0)

        for impl, attn in (("cdna4_flash", flash_attention),
                           ("torch_sdpa", lambda q, k, v, causal: F.scaled_dot_product_attention(q, k, v, is_causal=causal))):
            out = attn(q, k, v, causal)

            def fwd():
                with torch.no_grad():
                    attn(q, k, v, causal)

            def fwdbwd():
                for t in (q, k, v):
                    t.grad = None
                attn(q, k, v, causal).backward(d_out)

            ms_f = bench(fwd)
            ms_fb = bench(fwdbwd)
            print(json.dumps({
                "bench": "attention", "shape": name, "impl": impl, "causal": causal,
                "B": B, "H": H, "S": S, "D": D,
                "fwd_ms": round(ms_f, 3), "fwdbwd_ms": round(ms_fb, 3),
                "fwd_tflops": round(flops_fwd / ms_f / 1e9, 1),
                "fwdbwd_tflops": round(3.5 * flops_fwd / ms_fb / 1e9, 1),
            }), flush=True)


if __name__ == "__main__":
    main()
