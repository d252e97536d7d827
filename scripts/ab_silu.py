"""A/B: fused silu-down vs silu_mul + fused-norm down, cold-LLC rotation."""
import sys, time, torch
sys.path.insert(0, ".")
from kukeon_amd import _C

M, N, K = 64, 4096, 14336
nw_copies = 10
torch.manual_seed(3)
gus = [torch.randn(M, 2 * K, dtype=torch.bfloat16, device="cuda") * 0.4
       for _ in range(4)]
ws_l = [torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.05
        for _ in range(nw_copies)]
resid = torch.randn(M, N, dtype=torch.bfloat16, device="cuda")
nw = torch.rand(N, dtype=torch.bfloat16, device="cuda") + 0.5
wrk = torch.empty(32 * 64 * N, dtype=torch.float32, device="cuda")
normed = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
act = torch.empty(M, K, dtype=torch.bfloat16, device="cuda")

def t(fn, n=30):
    for i in range(5):
        fn(i % nw_copies)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(n):
        fn(i % nw_copies)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6

def fused(i):
    _C.skinny_gemm5_silu_fused_norm(normed, gus[i % 4], ws_l[i], wrk,
                                    resid, nw, 1e-5)

def unfused(i):
    _C.silu_mul(act, gus[i % 4])
    _C.skinny_gemm5_fused_norm(normed, act, ws_l[i], wrk, resid, nw, 1e-5)

print(f"fused={t(fused):.1f}us  unfused(silu+gemm)={t(unfused):.1f}us",
      flush=True)
