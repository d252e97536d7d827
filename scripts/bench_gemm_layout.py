import sys, torch, time
sys.path.insert(0, "/root/repo")
import torch.nn.functional as F

def t(fn, n=50):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1e6

for (M, N, K, tag) in [(64, 6144, 4096, "qkv"), (64, 4096, 4096, "o"),
                       (64, 28672, 4096, "gate_up"), (64, 4096, 14336, "down"),
                       (64, 128256, 4096, "lm_head")]:
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")*0.3
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")*0.05
    wt = w.t().contiguous()          # [K, N] row-major
    a = t(lambda: F.linear(x, w))            # TN
    b = t(lambda: torch.matmul(x, wt))       # NN
    floor = N*K*2/6.3e12*1e6
    print(f"{tag:>8}: TN {a:7.1f}us  NN {b:7.1f}us  floor {floor:6.1f}us", flush=True)
