import sys, time, torch
sys.path.insert(0, "/root/repo")
from hivemind_amd.ops import mfma_matmul
torch.manual_seed(0)
for M, N, K in [(4096, 4096, 4096), (8192, 4096, 1024)]:
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    for fn, name in [(lambda: mfma_matmul(x, w), "mfma"), (lambda: x @ w.t(), "hipblaslt")]:
        for _ in range(3): fn()
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(20): fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 20
        tf = 2 * M * N * K / dt / 1e12
        print(f"{name} M{M} N{N} K{K}: {dt*1e6:.0f}us  {tf:.0f} TF/s", flush=True)
