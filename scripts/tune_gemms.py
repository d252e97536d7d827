import sys, os, time, torch
sys.path.insert(0, "/root/repo")
import torch.nn.functional as F
import torch.cuda.tunable as tunable

shapes = [(64, 6144, 4096), (64, 4096, 4096), (64, 28672, 4096),
          (64, 4096, 14336), (64, 128256, 4096),
          (16448, 6144, 4096), (16448, 28672, 4096), (16448, 4096, 14336)]

def timeit(fn, n=30):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1e6

tens = {}
for (M,N,K) in shapes:
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")*0.3
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")*0.05
    tens[(M,N,K)] = (x, w)

base = {s: timeit(lambda s=s: F.linear(*tens[s])) for s in shapes}
tunable.enable(True)
tunable.tuning_enable(True)
tunable.set_max_tuning_duration(1000)
for s in shapes:
    F.linear(*tens[s])  # triggers tuning
torch.cuda.synchronize()
tunable.tuning_enable(False)
tuned = {s: timeit(lambda s=s: F.linear(*tens[s])) for s in shapes}
for s in shapes:
    print(f"M{s[0]} N{s[1]} K{s[2]}: base {base[s]:7.1f}us tuned {tuned[s]:7.1f}us ({base[s]/tuned[s]:.2f}x)")
tunable.write_file("/root/repo/gpurun_out/tunableop_gfx950.csv")
