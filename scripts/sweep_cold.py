"""Cold-LLC GEMM comparison: skinny vs hipBLASLt on the decode shapes.

Standalone loops rerun ONE weight, so anything <256 MB sits in the LLC and
the numbers are inflated (gate_up measured 36.9us warm vs 50.3us in situ).
Rotating over enough weight copies to overflow the LLC gives the number
that actually predicts in-engine performance."""
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


SHAPES = [(64, 6144, 4096, "qkv"), (64, 4096, 4096, "o"),
          (64, 28672, 4096, "gate_up"), (64, 4096, 14336, "down"),
          (64, 128256, 4096, "lm_head")]

for (M, N, K, tag) in SHAPES:
    wbytes = N * K * 2
    nw = max(2, (2 * LLC + wbytes - 1) // wbytes)
    nw = min(nw, 40)
    torch.manual_seed(13)
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.3
    ws_list = [torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.05
               for _ in range(nw)]
    out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    nslices = (K + 255) // 256
    wrk = torch.empty(min(48, 2 * nslices) * M * N, dtype=torch.float32,
                      device="cuda")
    floor = wbytes / 6.3e12 * 1e6
    us_bl = t_rot(lambda i: F.linear(x, ws_list[i]), nw)
    line = (f"{tag:>8} ({nw} copies): blas {us_bl:6.1f}  floor {floor:6.1f}")
    for sk in (0, 1, 2, 4, 8):
        if sk:
            os.environ["KUKEON_SKINNY_SPLITK"] = str(sk)
        else:
            os.environ.pop("KUKEON_SKINNY_SPLITK", None)
        us = t_rot(lambda i: _C.skinny_gemm(out, x, ws_list[i], wrk), nw)
        line += f"  sk{sk or 'A'}={us:6.1f}"
    os.environ.pop("KUKEON_SKINNY_SPLITK", None)
    for sk in (0, 1, 2, 4, 5, 8, 16):
        if sk:
            os.environ["KUKEON_SK2_SPLITK"] = str(sk)
        else:
            os.environ.pop("KUKEON_SK2_SPLITK", None)
        us = t_rot(lambda i: _C.skinny_gemm2(out, x, ws_list[i], wrk), nw)
        line += f"  v2k{sk or 'A'}={us:6.1f}"
    os.environ.pop("KUKEON_SK2_SPLITK", None)
    for ks in (128, 256):
        os.environ["KUKEON_SK5_KS"] = str(ks)
        for sk in (0, 2, 4, 8, 16, 32):
            if sk:
                os.environ["KUKEON_SK5_SPLITK"] = str(sk)
            else:
                os.environ.pop("KUKEON_SK5_SPLITK", None)
            if sk and sk > (K // ks):
                continue
            us = t_rot(lambda i: _C.skinny_gemm5(out, x, ws_list[i], wrk),
                       nw)
            line += f"  v5-{ks}k{sk or 'A'}={us:6.1f}"
        os.environ.pop("KUKEON_SK5_SPLITK", None)
    os.environ.pop("KUKEON_SK5_KS", None)
    if N % 256 == 0:
        for sk in (0, 2, 4, 8, 16):
            if sk:
                os.environ["KUKEON_SK4_SPLITK"] = str(sk)
            else:
                os.environ.pop("KUKEON_SK4_SPLITK", None)
            us = t_rot(lambda i: _C.skinny_gemm4(out, x, ws_list[i], wrk), nw)
            line += f"  v4k{sk or 'A'}={us:6.1f}"
        os.environ.pop("KUKEON_SK4_SPLITK", None)
    print(line, flush=True)
