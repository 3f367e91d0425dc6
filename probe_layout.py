import sys, torch
sys.path.insert(0, "/root/repo")
from hivemind_amd import _hip_ops
torch.manual_seed(0)
M = N = 128; K = 32
x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
(out,) = _hip_ops.mfma_linear_bf16(x, w, None, False, False)
ref = x.float() @ w.float().t()
diff = (out.float() - ref).abs()
print("max err", diff.max().item(), "mean", diff.mean().item())
print("err vs ref.T:", (out.float() - ref.t()).abs().max().item())
# per 16x16 block error map
blocks = diff.view(8, 16, 8, 16).amax(dim=(1, 3))
torch.set_printoptions(precision=2, linewidth=200)
print(blocks)
# check a couple of specific entries
print("out[0,:4]", out[0,:4].float().tolist())
print("ref[0,:4]", ref[0,:4].tolist())
print("out[0:4,0]", out[0:4,0].float().tolist())
print("ref[0:4,0]", ref[0:4,0].tolist())
# is out some permutation? check row 0 of ref found in out rows
sims = (out.float() @ ref.t()) / (out.float().norm(dim=1, keepdim=True) * ref.norm(dim=1))
print("best-match rows for ref rows 0..7:", sims.argmax(0)[:8].tolist())
