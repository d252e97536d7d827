"""Sweep split-K for the skinny GEMM on the 8B decode shapes.

The CDNA4 guide's decomposition rule for this regime: per-block latency is
~constant in N, so wall time tracks max(one block's K-depth, blocks/CUs) —
target ~0.5-1x the 256 CUs in total blocks, not a fixed tile count."""
import os
import sys
import time

import torch

sys.path.insert(0, ".")
from kukeon_amd import _C  # noqa: E402
import torch.nn.functional as F  # noqa: E402


def t(fn, n=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6


SHAPES = [(64, 6144, 4096, "qkv"), (64, 4096, 4096, "o"),
          (64, 28672, 4096, "gate_up"), (64, 4096, 14336, "down"),
          (64, 128256, 4096, "lm_head"),
          (32, 6144, 4096, "qkv32"), (16, 4096, 4096, "o16")]

for (M, N, K, tag) in SHAPES:
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.3
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.05
    out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    ref = (x.float() @ w.float().T)
    nslices = (K + 255) // 256
    ws = torch.empty(min(16, nslices) * M * N, dtype=torch.float32, device="cuda")
    us_bl = t(lambda: F.linear(x, w))
    floor = N * K * 2 / 6.3e12 * 1e6
    row = []
    for sk in (1, 2, 3, 4, 6, 8, 12, 16):
        if sk > nslices and row and row[-1][0] == nslices:
            continue
        os.environ["KUKEON_SKINNY_SPLITK"] = str(sk)
        _C.skinny_gemm(out, x, w, ws)
        ok = bool(torch.isclose(out.float(), ref, rtol=3e-2,
                                atol=3e-2).all().item())
        us = t(lambda: _C.skinny_gemm(out, x, w, ws))
        row.append((min(sk, nslices), us, ok))
    del os.environ["KUKEON_SKINNY_SPLITK"]
    best = min(row, key=lambda r: r[1])
    print(f"{tag:>8} (ntiles {N//64:4d}, nslices {nslices:2d}): "
          f"blas {us_bl:6.1f}  floor {floor:6.1f}  best sk={best[0]} "
          f"{best[1]:6.1f}us", flush=True)
    for sk, us, ok in row:
        blocks = (N // 64) * sk
        print(f"      sk={sk:2d} ({blocks:5d} blk): {us:7.1f}us"
              f"{'' if ok else '  WRONG'}", flush=True)
