"""Profile harness: run one skinny GEMM shape repeatedly (cold-LLC weight
rotation) for rocprofv3 kernel-trace/PMC attribution."""
import os, sys
import torch
sys.path.insert(0, ".")
M, N, K, sk, ver, iters = (int(a) for a in (sys.argv[1:7] + ["30"])[:6])
if sk:
    os.environ["KUKEON_SK2_SPLITK" if ver == 2 else "KUKEON_SKINNY_SPLITK"] = str(sk)
from kukeon_amd import _C
torch.manual_seed(13)
x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.5
wbytes = N * K * 2
nw = min(40, max(2, (2 * 256 * (1 << 20) + wbytes - 1) // wbytes))
ws_list = [torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.05
           for _ in range(nw)]
out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
wrk = torch.empty(16 * 64 * N, dtype=torch.float32, device="cuda")
fn = _C.skinny_gemm2 if ver == 2 else _C.skinny_gemm
for i in range(iters):
    fn(out, x, ws_list[i % nw], wrk)
torch.cuda.synchronize()
print("done", flush=True)
