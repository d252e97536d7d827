"""v6 (barrier-free register-x) vs v5 vs hipBLASLt, cold-LLC rotation.

Also checks numerics of every v6 config against the fp32 reference before
timing it — a wrong-fast kernel is worthless.
"""
import os
import sys
import time

import torch

sys.path.insert(0, ".")
from kukeon_amd import _C  # noqa: E402
import torch.nn.functional as F  # noqa: E402

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


SHAPES = [
    (64, 4096, 14336, "down"),
    (16, 4096, 14336, "down-m16"),
    (64, 28672, 4096, "gate_up"),
    (64, 6144, 4096, "qkv"),
    (64, 4096, 4096, "o"),
    (16, 8192, 28672, "70b-down"),
    (64, 128256, 4096, "lm_head"),
]

for (M, N, K, tag) in SHAPES:
    wbytes = N * K * 2
    nw = max(2, (2 * LLC + wbytes - 1) // wbytes)
    nw = min(nw, 40)
    torch.manual_seed(13)
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.3
    ws_list = [torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.05
               for _ in range(nw)]
    out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    nslices = (K + 127) // 128
    wrk = torch.empty(min(48, 2 * nslices) * M * N, dtype=torch.float32,
                      device="cuda")
    ref = (x.float() @ ws_list[0].float().T).bfloat16()
    floor = wbytes / 7.3e12 * 1e6
    us_bl = t_rot(lambda i: F.linear(x, ws_list[i]), nw)
    line = f"{tag:>9} ({nw}w): blas {us_bl:6.1f} floor {floor:5.1f}"
    # v5 best-known config for reference
    us5 = t_rot(lambda i: _C.skinny_gemm5(out, x, ws_list[i], wrk), nw)
    line += f"  v5={us5:6.1f}"
    for ks in (128, 256):
        os.environ["KUKEON_SK6_KS"] = str(ks)
        for sk in (0, 2, 4, 8, 16):
            if sk:
                if sk > (K // ks):
                    continue
                os.environ["KUKEON_SK6_SPLITK"] = str(sk)
            else:
                os.environ.pop("KUKEON_SK6_SPLITK", None)
            _C.skinny_gemm6(out, x, ws_list[0], wrk)
            torch.cuda.synchronize()
            md = (out.float() - ref.float()).abs().max().item()
            us = t_rot(lambda i: _C.skinny_gemm6(out, x, ws_list[i], wrk),
                       nw)
            bad = "!" if md > 0.15 else ""
            line += f"  v6-{ks}k{sk or 'A'}={us:6.1f}({md:.3f}){bad}"
        os.environ.pop("KUKEON_SK6_SPLITK", None)
    os.environ.pop("KUKEON_SK6_KS", None)
    print(line, flush=True)
