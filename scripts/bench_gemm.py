import sys, torch, time
sys.path.insert(0, ".")
from kukeon_amd import _C
import torch.nn.functional as F

def t(fn, n=50):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6

for (M, N, K, tag) in [(64, 6144, 4096, "qkv"), (64, 4096, 4096, "o"),
                       (64, 28672, 4096, "gate_up"), (64, 4096, 14336, "down"),
                       (64, 128256, 4096, "lm_head")]:
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.3
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.05
    out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    ws = torch.empty(16 * 64 * N, dtype=torch.float32, device="cuda")
    us_sk = t(lambda: _C.skinny_gemm(out, x, w, ws))
    us_bl = t(lambda: F.linear(x, w))
    floor = N * K * 2 / 6.3e12 * 1e6
    print(f"{tag:>8}: skinny {us_sk:7.1f}us  hipblaslt {us_bl:7.1f}us  floor {floor:6.1f}us", flush=True)

# paged attn + rmsnorm A/B sanity (new kernels)
import kukeon_amd.ops as ops
x = torch.randn(64, 4096, dtype=torch.bfloat16, device="cuda")
wgt = torch.randn(4096, dtype=torch.bfloat16, device="cuda")
o = torch.empty_like(x)
print(f"rmsnorm64: {t(lambda: ops.rmsnorm(o, x, wgt, 1e-5)):.1f}us")
