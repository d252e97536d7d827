"""Standalone down-proj loop (cold-LLC rotation) for PMC attribution:
skinny5 then hipBLASLt, 30 reps each."""
import sys, torch
sys.path.insert(0, "/root/repo")
from kukeon_amd import _C
import torch.nn.functional as F

M, N, K = 64, 4096, 14336
torch.manual_seed(3)
x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.3
ws_l = [torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.05
        for _ in range(10)]
out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
wrk = torch.empty(32 * 64 * N, dtype=torch.float32, device="cuda")
for i in range(30):
    _C.skinny_gemm5(out, x, ws_l[i % 10], wrk)
for i in range(30):
    F.linear(x, ws_l[i % 10])
torch.cuda.synchronize()
print("done")
