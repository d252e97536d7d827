"""Micro-bench the decode paged-attention kernel at bench-like shapes and
report achieved KV bandwidth vs the ~6.3 TB/s practical HBM3E roofline."""
import sys
import time

import torch

sys.path.insert(0, ".")
import kukeon_amd.ops as ops  # noqa: E402


def t(fn, n=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6


DEV = "cuda:0"
B, Hq, Hk, D, BS = 64, 32, 8, 128, 16
torch.manual_seed(0)
for ctx_lo, ctx_hi in [(512, 3584), (1500, 1500), (3584, 3584)]:
    ctxs = torch.randint(ctx_lo, ctx_hi + 1, (B,)).tolist() \
        if ctx_lo != ctx_hi else [ctx_lo] * B
    nb = [(c + BS - 1) // BS for c in ctxs]
    NB = sum(nb) + 8
    kc = torch.randn(NB, Hk, BS, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.randn(NB, Hk, BS, D, dtype=torch.bfloat16, device=DEV)
    maxb = max(nb)
    bt = torch.zeros(B, maxb, dtype=torch.int32, device=DEV)
    nxt = 0
    for b in range(B):
        bt[b, : nb[b]] = torch.arange(nxt, nxt + nb[b], dtype=torch.int32)
        nxt += nb[b]
    seq_lens = torch.tensor(ctxs, dtype=torch.int32, device=DEV)
    q = torch.randn(B, Hq * D, dtype=torch.bfloat16, device=DEV)
    out = torch.empty(B, Hq * D, dtype=torch.bfloat16, device=DEV)
    scale = D ** -0.5
    kv_bytes = sum(ctxs) * Hk * D * 2 * 2
    print(f"ctx {ctx_lo}-{ctx_hi} (KV {kv_bytes/1e6:.0f} MB/call):")
    import os as _os
    for splits in (1, 2, 4, 8, 16):
        tmp_out = torch.zeros(B, Hq, splits, D, dtype=torch.float32,
                              device=DEV)
        tmp_ml = torch.zeros(B, Hq, splits, 2, dtype=torch.float32,
                             device=DEV)
        line = f"  splits={splits:2d}:"
        for mode in ("0", "1", "32"):
            _os.environ["KUKEON_ATTN_MFMA"] = mode
            us = t(lambda: ops.paged_attention(out, q, kc, vc, bt, seq_lens,
                                               0, splits, scale, tmp_out,
                                               tmp_ml))
            bw = kv_bytes / (us * 1e-6) / 1e12
            tag = {"0": "dot2", "1": "m16", "32": "m32"}[mode]
            line += (f"  {tag} {us:7.1f}us {bw:5.2f}TB/s"
                     f" ({bw/6.3*100:4.1f}%)")
        _os.environ.pop("KUKEON_ATTN_MFMA", None)
        print(line, flush=True)
