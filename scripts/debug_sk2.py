"""Per-config isolation for skinny_gemm2: run one (M,N,K,splitk) and print
max-abs-diff + first-mismatch pattern, then exit."""
import os, sys
import torch
sys.path.insert(0, ".")
M, N, K, sk = (int(a) for a in sys.argv[1:5])
if sk:
    os.environ["KUKEON_SK2_SPLITK"] = str(sk)
from kukeon_amd import _C
torch.manual_seed(13)
x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.5
w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.05
out = torch.full((M, N), float("nan"), dtype=torch.bfloat16, device="cuda")
ws = torch.empty(16 * 64 * N, dtype=torch.float32, device="cuda")
_C.skinny_gemm2(out, x, w, ws)
torch.cuda.synchronize()
ref = x.float() @ w.float().T
d = (out.float() - ref).abs()
print(f"M{M} N{N} K{K} sk{sk}: maxdiff {d.max().item():.4f} "
      f"mean {d.mean().item():.5f} nan={torch.isnan(out.float()).sum().item()}")
if d.max().item() > 0.05:
    bad = (d > 0.05)
    idx = bad.nonzero()[:8]
    print("first bad:", idx.tolist())
    rows = bad.any(1).nonzero().flatten().tolist()[:10]
    cols = bad.any(0).nonzero().flatten()
    print("bad rows:", rows, "ncols bad:", cols.numel(),
          "col range:", (cols.min().item(), cols.max().item()) if cols.numel() else None)
