# ablation timing: mask 2=full(prod), 34=no-consume, 42=no-consume+no-glds, 10=no-glds(consume garbage)
for m in 2 34 42 10; do
  KUKEON_SK2_SERIAL=$m timeout 120 python - "$m" <<'PYEOF' 2>&1 | grep -E "^mask"
import os, sys, time
sys.path.insert(0, ".")
import torch
from kukeon_amd import _C
m = sys.argv[1]
torch.manual_seed(13)
M, N, K = 64, 28672, 4096
x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.5
nw = 3
ws_list = [torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.05 for _ in range(nw)]
out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
wrk = torch.empty(16 * 64 * N, dtype=torch.float32, device="cuda")
for i in range(5): _C.skinny_gemm2(out, x, ws_list[i % nw], wrk)
torch.cuda.synchronize()
t0 = time.perf_counter()
for i in range(30): _C.skinny_gemm2(out, x, ws_list[i % nw], wrk)
torch.cuda.synchronize()
print(f"mask{m}: {(time.perf_counter()-t0)/30*1e6:.1f}us")
PYEOF
done
