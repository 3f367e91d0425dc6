"""Minimal fwd-only flash kernel driver for PMC profiling."""
import sys, torch
sys.path.insert(0, __file__.rsplit("/", 2)[0])
from hivemind_amd.ops import flash_attention

q = torch.randn(128, 12, 512, 64, device="cuda", dtype=torch.bfloat16)
k, v = torch.randn_like(q), torch.randn_like(q)
with torch.no_grad():
    for _ in range(10):
        flash_attention(q, k, v)
torch.cuda.synchronize()
print("done")
