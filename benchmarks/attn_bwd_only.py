"""dkdv/dq-only driver for PMC profiling."""
import sys, torch
sys.path.insert(0, __file__.rsplit("/", 2)[0])
from hivemind_amd.ops import flash_attention

q = torch.randn(128, 12, 512, 64, device="cuda", dtype=torch.bfloat16, requires_grad=True)
k, v = torch.randn_like(q, requires_grad=True), torch.randn_like(q, requires_grad=True)
d_out = torch.randn_like(q)
for _ in range(10):
    for t in (q, k, v):
        t.grad = None
    flash_attention(q, k, v).backward(d_out)
torch.cuda.synchronize()
print("done")
