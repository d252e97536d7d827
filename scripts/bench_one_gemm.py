import sys, torch, time
sys.path.insert(0, "/root/repo")
from kukeon_amd import _C
M, N, K = 64, 6144, 4096
x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.3
w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.05
out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
ws = torch.empty(16 * 64 * N, dtype=torch.float32, device="cuda")
for _ in range(20):
    _C.skinny_gemm(out, x, w, ws)
torch.cuda.synchronize()
