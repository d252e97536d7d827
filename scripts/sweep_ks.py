"""A/B: v5 TILES=2 (256 blocks) vs TILES=1 N-split (512 blocks, same
slab traffic, 2 blocks/CU) on the decode shapes, cold-LLC rotation."""
import os, sys, time, torch
sys.path.insert(0, ".")
from kukeon_amd import _C
import torch.nn.functional as F

LLC = 256 * (1 << 20)

def t_rot(fn, nw, n=30):
    for i in range(5):
        fn(i % nw)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(n):
        fn(i % nw)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6

SHAPES = [(64, 4096, 14336, "down"), (16, 4096, 14336, "down-m16"),
          (64, 28672, 4096, "gate_up"), (64, 6144, 4096, "qkv"),
          (16, 8192, 28672, "70b-down"), (64, 128256, 4096, "lm_head")]

for (M, N, K, tag) in SHAPES:
    wbytes = N * K * 2
    nw = min(max(2, (2 * LLC + wbytes - 1) // wbytes), 40)
    torch.manual_seed(13)
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.3
    ws_l = [torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.05
            for _ in range(nw)]
    out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    wrk = torch.empty(64 * 64 * N, dtype=torch.float32, device="cuda")
    ref = (x.float() @ ws_l[0].float().T)
    us_bl = t_rot(lambda i: F.linear(x, ws_l[i]), nw)
    line = f"{tag:>9}: blas {us_bl:6.1f}"
    for ks in (64, 128):
        os.environ["KUKEON_SK5_KS"] = str(ks)
        for sk in (0, 4, 8, 16):
            if sk:
                if sk > (K // ks):
                    continue
                os.environ["KUKEON_SK5_SPLITK"] = str(sk)
            else:
                os.environ.pop("KUKEON_SK5_SPLITK", None)
            _C.skinny_gemm5(out, x, ws_l[0], wrk)
            torch.cuda.synchronize()
            md = (out.float() - ref).abs().max().item()
            us = t_rot(lambda i: _C.skinny_gemm5(out, x, ws_l[i], wrk), nw)
            bad = "!" if md > 0.15 else ""
            line += f"  ks{ks}k{sk or 'A'}={us:6.1f}({md:.2f}){bad}"
        os.environ.pop("KUKEON_SK5_SPLITK", None)
    os.environ.pop("KUKEON_SK5_KS", None)
    print(line, flush=True)
