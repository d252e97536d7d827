import sys, os
import torch, torch.nn.functional as F
sys.path.insert(0, ".")
which = sys.argv[1]
torch.manual_seed(13)
M, N, K = 64, 28672, 4096
x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.5
ws_list = [torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.05 for _ in range(3)]
out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
if which == "blas":
    for i in range(30):
        F.linear(x, ws_list[i % 3], out=out)
else:
    from kukeon_amd import _C
    wrk = torch.empty(16 * 64 * N, dtype=torch.float32, device="cuda")
    fn = {"v2": _C.skinny_gemm2, "v5": _C.skinny_gemm5}[which]
    for i in range(30):
        fn(out, x, ws_list[i % 3], wrk)
torch.cuda.synchronize()
print("done")
