"""GPU numerics tests: every gfx950 HIP kernel vs the plain-PyTorch fp32
reference of the same op (tests/conftest registers the `gpu` marker)."""
import pytest
import torch

import kukeon_amd.ops as ops
from kukeon_amd.ops import reference

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

DEV = "cuda:0"


def test_native_loaded():
    # on a GPU box the HIP extension must be present — no silent fallback
    assert ops.native_available()


def test_mfma_probe_layout():
    from kukeon_amd import _C
    torch.manual_seed(0)
    # asymmetric inputs so operand/output transposes can't pass (guide G9)
    a = (torch.randn(32, 16) * torch.linspace(0.2, 2.0, 16)).bfloat16().to(DEV)
    b = (torch.randn(16, 32) * torch.linspace(-1.5, 1.5, 32)).bfloat16().to(DEV)
    out = torch.empty(32, 32, dtype=torch.float32, device=DEV)
    _C.mfma_probe(out, a, b)
    ref = a.float() @ b.float()
    torch.testing.assert_close(out.cpu(), ref.cpu(), rtol=2e-2, atol=2e-2)


@pytest.mark.parametrize("T,H", [(1, 4096), (64, 4096), (7, 8192), (3, 5120)])
def test_rmsnorm(T, H):
    torch.manual_seed(1)
    x = torch.randn(T, H, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(H, dtype=torch.bfloat16, device=DEV)
    out = torch.empty_like(x)
    ops.rmsnorm(out, x, w, 1e-5)
    ref = torch.empty_like(x.cpu())
    reference.rmsnorm(ref, x.cpu(), w.cpu(), 1e-5)
    torch.testing.assert_close(out.cpu().float(), ref.float(), rtol=2e-2,
                               atol=2e-2)


def test_fused_add_rmsnorm():
    torch.manual_seed(2)
    T, H = 33, 4096
    x = torch.randn(T, H, dtype=torch.bfloat16, device=DEV)
    res = torch.randn(T, H, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(H, dtype=torch.bfloat16, device=DEV)
    x_ref, res_ref = x.cpu().clone(), res.cpu().clone()
    ops.fused_add_rmsnorm(x, res, w, 1e-5)
    reference.fused_add_rmsnorm(x_ref, res_ref, w.cpu(), 1e-5)
    torch.testing.assert_close(res.cpu().float(), res_ref.float(), rtol=2e-2,
                               atol=2e-2)
    torch.testing.assert_close(x.cpu().float(), x_ref.float(), rtol=2e-2,
                               atol=3e-2)


def test_silu_mul():
    torch.manual_seed(3)
    T, I = 17, 14336
    gu = torch.randn(T, 2 * I, dtype=torch.bfloat16, device=DEV)
    out = torch.empty(T, I, dtype=torch.bfloat16, device=DEV)
    ops.silu_mul(out, gu)
    ref = torch.empty(T, I, dtype=torch.bfloat16)
    reference.silu_mul(ref, gu.cpu())
    torch.testing.assert_close(out.cpu().float(), ref.float(), rtol=2e-2,
                               atol=2e-2)


def _make_cache(nb, hk, bs, d, fp8=False):
    k = torch.randn(nb, hk, bs, d, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(nb, hk, bs, d, dtype=torch.bfloat16, device=DEV)
    if fp8:
        k = k.float().to(torch.float8_e4m3fn).view(torch.uint8)
        v = v.float().to(torch.float8_e4m3fn).view(torch.uint8)
    return k, v


def test_rope_kv_append():
    torch.manual_seed(4)
    T, Hq, Hk, D, BS, NB = 9, 8, 2, 128, 16, 8
    qkv = torch.randn(T, (Hq + 2 * Hk) * D, dtype=torch.bfloat16, device=DEV)
    kc, vc = _make_cache(NB, Hk, BS, D)
    maxpos = 64
    inv = 1.0 / (500000.0 ** (torch.arange(0, D, 2).float() / D))
    t = torch.arange(maxpos).float()
    fr = torch.outer(t, inv)
    cos_sin = torch.cat([fr.cos(), fr.sin()], dim=1).to(DEV)
    pos = torch.randint(0, maxpos, (T,), dtype=torch.int32, device=DEV)
    slots = torch.randperm(NB * BS)[:T].to(torch.int32).to(DEV)

    qkv_ref = qkv.cpu().clone()
    kc_ref, vc_ref = kc.cpu().clone(), vc.cpu().clone()
    ops.rope_kv_append(qkv, kc, vc, cos_sin, pos, slots, Hq, Hk, D)
    reference.rope_kv_append(qkv_ref, kc_ref, vc_ref, cos_sin.cpu(), pos.cpu(),
                             slots.cpu(), Hq, Hk, D)
    torch.testing.assert_close(qkv.cpu().float(), qkv_ref.float(), rtol=2e-2,
                               atol=2e-2)
    torch.testing.assert_close(kc.cpu().float(), kc_ref.float(), rtol=2e-2,
                               atol=2e-2)
    torch.testing.assert_close(vc.cpu().float(), vc_ref.float(), rtol=0, atol=0)


@pytest.mark.parametrize("G,splits,fp8", [(4, 1, False), (4, 4, False),
                                          (8, 1, False), (8, 3, False),
                                          (4, 1, True), (4, 3, True)])
def test_paged_attention(G, splits, fp8):
    torch.manual_seed(5)
    B, Hk, D, BS = 5, 2, 128, 16
    Hq = G * Hk
    ctxs = [1, 16, 17, 100, 250]
    NB = sum((c + BS - 1) // BS for c in ctxs) + 4
    kc, vc = _make_cache(NB, Hk, BS, D, fp8=fp8)
    tol = 7e-2 if fp8 else 2e-2
    maxb = max((c + BS - 1) // BS for c in ctxs)
    bt = torch.zeros(B, maxb, dtype=torch.int32, device=DEV)
    nxt = 0
    for b, c in enumerate(ctxs):
        n = (c + BS - 1) // BS
        bt[b, :n] = torch.arange(nxt, nxt + n, dtype=torch.int32)
        nxt += n
    seq_lens = torch.tensor(ctxs, dtype=torch.int32, device=DEV)
    q = torch.randn(B, Hq * D, dtype=torch.bfloat16, device=DEV)
    out = torch.empty(B, Hq * D, dtype=torch.bfloat16, device=DEV)
    scale = D ** -0.5
    tmp_out = torch.zeros(B, Hq, splits, D, dtype=torch.float32, device=DEV)
    tmp_ml = torch.zeros(B, Hq, splits, 2, dtype=torch.float32, device=DEV)
    ops.paged_attention(out, q, kc, vc, bt, seq_lens, 0, splits, scale,
                        tmp_out, tmp_ml)
    ref = torch.empty(B, Hq * D, dtype=torch.bfloat16)
    reference.paged_attention(ref, q.cpu(), kc.cpu(), vc.cpu(), bt.cpu(),
                              seq_lens.cpu(), 0, 1, scale)
    torch.testing.assert_close(out.cpu().float(), ref.float(), rtol=tol,
                               atol=tol)


def test_paged_attention_q_offset():
    """q passed as a row of a fused qkv tensor with stride + offset."""
    torch.manual_seed(6)
    B, Hq, Hk, D, BS = 3, 8, 2, 128, 16
    ctxs = [40, 8, 64]
    NB = 12
    kc, vc = _make_cache(NB, Hk, BS, D)
    bt = torch.arange(NB, dtype=torch.int32, device=DEV).reshape(3, 4)
    seq_lens = torch.tensor(ctxs, dtype=torch.int32, device=DEV)
    qkv = torch.randn(B, (Hq + 2 * Hk) * D, dtype=torch.bfloat16, device=DEV)
    out = torch.empty(B, Hq * D, dtype=torch.bfloat16, device=DEV)
    t1 = torch.zeros(1, dtype=torch.float32, device=DEV)
    ops.paged_attention(out, qkv, kc, vc, bt, seq_lens, 0, 1, D ** -0.5, t1, t1)
    ref = torch.empty(B, Hq * D, dtype=torch.bfloat16)
    reference.paged_attention(ref, qkv.cpu(), kc.cpu(), vc.cpu(), bt.cpu(),
                              seq_lens.cpu(), 0, 1, D ** -0.5)
    torch.testing.assert_close(out.cpu().float(), ref.float(), rtol=2e-2,
                               atol=2e-2)


def test_sample_greedy_exact():
    torch.manual_seed(7)
    B, V = 8, 128256
    logits = torch.randn(B, V, dtype=torch.bfloat16, device=DEV)
    tokens = torch.zeros(B, dtype=torch.int32, device=DEV)
    temps = torch.zeros(B, dtype=torch.float32, device=DEV)
    tk = torch.zeros(B, dtype=torch.int32, device=DEV)
    tp = torch.ones(B, dtype=torch.float32, device=DEV)
    seed = torch.zeros(1, dtype=torch.int64, device=DEV)
    ws = torch.zeros(B, 528, dtype=torch.float32, device=DEV)
    ops.sample(tokens, logits, temps, tk, tp, seed, ws)
    ref = logits.float().argmax(dim=-1).to(torch.int32)
    assert torch.equal(tokens.cpu(), ref.cpu())


def test_sample_topk_membership():
    torch.manual_seed(8)
    B, V, K = 4, 50000, 20
    logits = torch.randn(B, V, dtype=torch.bfloat16, device=DEV) * 3
    temps = torch.full((B,), 0.8, dtype=torch.float32, device=DEV)
    tk = torch.full((B,), K, dtype=torch.int32, device=DEV)
    tp = torch.ones(B, dtype=torch.float32, device=DEV)
    seed = torch.zeros(1, dtype=torch.int64, device=DEV)
    ws = torch.zeros(B, 528, dtype=torch.float32, device=DEV)
    tokens = torch.zeros(B, dtype=torch.int32, device=DEV)
    # top-k sets computed on bf16-rounded logits (ties at 24-bit granularity
    # can admit a couple extra members — allow K + small slack)
    topsets = [set(torch.topk(logits[b].float(), K + 4).indices.tolist())
               for b in range(B)]
    for _ in range(50):
        ops.sample(tokens, logits, temps, tk, tp, seed, ws)
        for b in range(B):
            assert int(tokens[b]) in topsets[b]


def test_sample_topp_membership():
    torch.manual_seed(9)
    B, V = 4, 50000
    logits = (torch.randn(B, V) * 4).bfloat16().to(DEV)
    temps = torch.ones(B, dtype=torch.float32, device=DEV)
    tk = torch.zeros(B, dtype=torch.int32, device=DEV)
    tp = torch.full((B,), 0.7, dtype=torch.float32, device=DEV)
    seed = torch.zeros(1, dtype=torch.int64, device=DEV)
    ws = torch.zeros(B, 528, dtype=torch.float32, device=DEV)
    tokens = torch.zeros(B, dtype=torch.int32, device=DEV)
    allowed = []
    for b in range(B):
        probs = torch.softmax(logits[b].float(), -1)
        srt, idx = torch.sort(probs, descending=True)
        cum = torch.cumsum(srt, 0)
        cut = int(torch.searchsorted(cum, 0.7).clamp(max=V - 1))
        allowed.append(set(idx[: cut + 8].tolist()))  # small slack for ties
    for _ in range(50):
        ops.sample(tokens, logits, temps, tk, tp, seed, ws)
        for b in range(B):
            assert int(tokens[b]) in allowed[b]


def test_moe_gather_scatter():
    torch.manual_seed(10)
    T, H, K = 13, 4096, 2
    x = torch.randn(T, H, dtype=torch.bfloat16, device=DEV)
    row_map = torch.randint(0, T, (T * K,), dtype=torch.int32, device=DEV)
    out = torch.empty(T * K, H, dtype=torch.bfloat16, device=DEV)
    ops.moe_gather_tokens(out, x, row_map)
    assert torch.equal(out.cpu(), x.cpu()[row_map.cpu().long()])

    e = torch.randn(T * K, H, dtype=torch.bfloat16, device=DEV)
    inv = torch.randperm(T * K, device=DEV).to(torch.int32).reshape(T, K)
    w = torch.rand(T, K, dtype=torch.float32, device=DEV)
    out2 = torch.empty(T, H, dtype=torch.bfloat16, device=DEV)
    ops.moe_scatter_tokens(out2, e, inv, w, K)
    ref = torch.empty(T, H, dtype=torch.bfloat16)
    reference.moe_scatter_tokens(ref, e.cpu(), inv.cpu(), w.cpu(), K)
    torch.testing.assert_close(out2.cpu().float(), ref.float(), rtol=2e-2,
                               atol=2e-2)


def _prefill_case(ctxs_starts, Hq, Hk, seed=11, fp8=False):
    """ctxs_starts: list of (total_ctx, q_start). Builds caches+q and runs
    both impls."""
    torch.manual_seed(seed)
    D, BS = 128, 16
    nseq = len(ctxs_starts)
    NB = sum((c + BS - 1) // BS for c, _ in ctxs_starts) + 2
    kc, vc = _make_cache(NB, Hk, BS, D, fp8=fp8)
    tol = 7e-2 if fp8 else 2.5e-2
    maxb = max((c + BS - 1) // BS for c, _ in ctxs_starts)
    bt = torch.zeros(nseq, maxb, dtype=torch.int32, device=DEV)
    nxt = 0
    for s, (c, _) in enumerate(ctxs_starts):
        n = (c + BS - 1) // BS
        bt[s, :n] = torch.arange(nxt, nxt + n, dtype=torch.int32)
        nxt += n
    seq_lens = torch.tensor([c for c, _ in ctxs_starts], dtype=torch.int32,
                            device=DEV)
    rows = []
    qs_pairs, qb_seq, qb_start = [], [], []
    row = 0
    for s, (c, st) in enumerate(ctxs_starts):
        qlen = c - st
        qs_pairs.append((st, row))
        for qb in range(0, qlen, 32):
            qb_seq.append(s)
            qb_start.append(qb)
        row += qlen
    T = row
    q = torch.randn(T, Hq * D, dtype=torch.bfloat16, device=DEV)
    q_starts = torch.tensor(qs_pairs, dtype=torch.int32, device=DEV)
    t_qb_seq = torch.tensor(qb_seq, dtype=torch.int32, device=DEV)
    t_qb_start = torch.tensor(qb_start, dtype=torch.int32, device=DEV)
    out = torch.empty(T, Hq * D, dtype=torch.bfloat16, device=DEV)
    scale = D ** -0.5
    ops.prefill_attention(out, q, kc, vc, bt, seq_lens, q_starts, t_qb_seq,
                          t_qb_start, 0, scale)
    ref = torch.empty(T, Hq * D, dtype=torch.bfloat16)
    reference.prefill_attention(ref, q.cpu(), kc.cpu(), vc.cpu(), bt.cpu(),
                                seq_lens.cpu(), q_starts.cpu(), None, None,
                                0, scale)
    torch.testing.assert_close(out.cpu().float(), ref.float(), rtol=tol,
                               atol=tol)


def test_prefill_single_seq_aligned():
    _prefill_case([(128, 0)], Hq=8, Hk=2)


def test_prefill_ragged_tails():
    _prefill_case([(33, 0), (100, 0), (17, 0), (160, 0)], Hq=8, Hk=2)


def test_prefill_continuation():
    # multi-turn: queries are the tail of an existing context
    _prefill_case([(200, 150), (90, 64), (70, 69)], Hq=8, Hk=2)


def test_prefill_gqa8():
    _prefill_case([(77, 0), (130, 40)], Hq=8, Hk=1)


def test_prefill_fp8_kv():
    _prefill_case([(100, 0), (90, 64)], Hq=8, Hk=2, fp8=True)


def test_rope_kv_append_fp8():
    torch.manual_seed(14)
    T, Hq, Hk, D, BS, NB = 6, 4, 2, 128, 16, 4
    qkv = torch.randn(T, (Hq + 2 * Hk) * D, dtype=torch.bfloat16, device=DEV)
    kc, vc = _make_cache(NB, Hk, BS, D, fp8=True)
    inv = 1.0 / (500000.0 ** (torch.arange(0, D, 2).float() / D))
    fr = torch.outer(torch.arange(64).float(), inv)
    cos_sin = torch.cat([fr.cos(), fr.sin()], dim=1).to(DEV)
    pos = torch.randint(0, 64, (T,), dtype=torch.int32, device=DEV)
    slots = torch.randperm(NB * BS)[:T].to(torch.int32).to(DEV)
    kc_ref, vc_ref = kc.cpu().clone(), vc.cpu().clone()
    qkv_ref = qkv.cpu().clone()
    ops.rope_kv_append(qkv, kc, vc, cos_sin, pos, slots, Hq, Hk, D)
    reference.rope_kv_append(qkv_ref, kc_ref, vc_ref, cos_sin.cpu(),
                             pos.cpu(), slots.cpu(), Hq, Hk, D)
    # the HW converter and torch's cast may round ties differently:
    # allow one fp8 ulp (2^-3 relative) on a handful of boundary values
    k_gpu = kc.cpu().view(torch.float8_e4m3fn).float()
    k_ref = kc_ref.view(torch.float8_e4m3fn).float()
    torch.testing.assert_close(k_gpu, k_ref, rtol=0.15, atol=8e-2)
    v_gpu = vc.cpu().view(torch.float8_e4m3fn).float()
    v_ref = vc_ref.view(torch.float8_e4m3fn).float()
    torch.testing.assert_close(v_gpu, v_ref, rtol=0.15, atol=8e-2)


def test_prefill_single_token_turns():
    _prefill_case([(40, 39), (16, 15)], Hq=4, Hk=1)


def test_mfma_probe16_layout():
    from kukeon_amd import _C
    torch.manual_seed(12)
    a = (torch.randn(16, 32) * torch.linspace(0.3, 1.7, 32)).bfloat16().to(DEV)
    b = (torch.randn(32, 16) * torch.linspace(-1.2, 1.2, 16)).bfloat16().to(DEV)
    out = torch.empty(16, 16, dtype=torch.float32, device=DEV)
    _C.mfma_probe16(out, a, b)
    ref = a.float() @ b.float()
    torch.testing.assert_close(out.cpu(), ref.cpu(), rtol=2e-2, atol=2e-2)


def test_glds_probe_roundtrip():
    """asm global_load_lds staging: swizzled-source/lane-linear-dest LDS
    image must read back exactly through the xswz addresses."""
    from kukeon_amd import _C
    torch.manual_seed(7)
    src = torch.randn(64, 256, dtype=torch.bfloat16, device=DEV)
    out = torch.zeros_like(src)
    _C.glds_probe(out, src)
    assert torch.equal(out, src)


@pytest.mark.parametrize("M,N,K", [(1, 4096, 4096), (8, 6144, 4096),
                                   (33, 4096, 14336), (64, 28672, 4096),
                                   (64, 128256, 4096), (64, 4096, 4096)])
def test_skinny_gemm(M, N, K):
    from kukeon_amd import _C
    torch.manual_seed(13)
    x = (torch.randn(M, K, dtype=torch.bfloat16, device=DEV) * 0.5)
    w = (torch.randn(N, K, dtype=torch.bfloat16, device=DEV) * 0.05)
    out = torch.empty(M, N, dtype=torch.bfloat16, device=DEV)
    ws = torch.empty(16 * 64 * N, dtype=torch.float32, device=DEV)
    _C.skinny_gemm(out, x, w, ws)
    ref = (x.float() @ w.float().T)
    torch.testing.assert_close(out.float().cpu(), ref.cpu(), rtol=3e-2,
                               atol=3e-2)


@pytest.mark.parametrize("M,N,K,splitk", [
    (1, 4096, 4096, 0), (8, 6144, 4096, 0), (33, 4096, 14336, 0),
    (64, 28672, 4096, 0), (64, 128256, 4096, 0), (64, 4096, 4096, 0),
    (64, 4096, 14336, 8), (64, 6144, 4096, 5), (64, 28672, 4096, 2),
    (17, 6144, 4096, 3)])
def test_skinny_gemm2(M, N, K, splitk, monkeypatch):
    """v2 G-walk hand-counted kernel vs fp32 reference, incl. strided
    split-K walks with uneven slice counts per block."""
    from kukeon_amd import _C
    if splitk:
        monkeypatch.setenv("KUKEON_SK2_SPLITK", str(splitk))
    torch.manual_seed(13)
    x = (torch.randn(M, K, dtype=torch.bfloat16, device=DEV) * 0.5)
    w = (torch.randn(N, K, dtype=torch.bfloat16, device=DEV) * 0.05)
    out = torch.empty(M, N, dtype=torch.bfloat16, device=DEV)
    ws = torch.empty(16 * 64 * N, dtype=torch.float32, device=DEV)
    _C.skinny_gemm2(out, x, w, ws)
    ref = (x.float() @ w.float().T)
    torch.testing.assert_close(out.float().cpu(), ref.cpu(), rtol=3e-2,
                               atol=3e-2)


@pytest.mark.parametrize("M,N,K,ks,sk", [
    (64, 4096, 14336, 128, 0), (64, 28672, 4096, 128, 0),
    (64, 6144, 4096, 128, 4), (17, 6144, 4096, 128, 3),
    (64, 4096, 14336, 256, 8), (64, 128256, 4096, 128, 0),
    (16, 8192, 28672, 128, 0), (64, 8192, 28672, 128, 0)])
def test_skinny_gemm5(M, N, K, ks, sk, monkeypatch):
    """v5 full-line never-drain pipeline (pure HIP, visible loads) vs
    fp32 reference, both LDS geometries."""
    from kukeon_amd import _C
    monkeypatch.setenv("KUKEON_SK5_KS", str(ks))
    if sk:
        monkeypatch.setenv("KUKEON_SK5_SPLITK", str(sk))
    torch.manual_seed(13)
    x = (torch.randn(M, K, dtype=torch.bfloat16, device=DEV) * 0.5)
    w = (torch.randn(N, K, dtype=torch.bfloat16, device=DEV) * 0.05)
    out = torch.empty(M, N, dtype=torch.bfloat16, device=DEV)
    ws = torch.empty(48 * 64 * N, dtype=torch.float32, device=DEV)
    _C.skinny_gemm5(out, x, w, ws)
    ref = (x.float() @ w.float().T)
    torch.testing.assert_close(out.float().cpu(), ref.cpu(), rtol=3e-2,
                               atol=6e-2)


@pytest.mark.parametrize("M,N,K,sk", [
    (64, 4096, 14336, 0), (16, 4096, 4096, 4), (33, 6144, 4096, 0)])
def test_skinny_gemm6(M, N, K, sk, monkeypatch):
    """v6 barrier-free register-x pipeline vs fp32 reference (kept
    in-tree as the documented negative result — slower than v5, see
    profiles/r02_progress.md — but it must stay correct)."""
    from kukeon_amd import _C
    if sk:
        monkeypatch.setenv("KUKEON_SK6_SPLITK", str(sk))
    torch.manual_seed(17)
    x = (torch.randn(M, K, dtype=torch.bfloat16, device=DEV) * 0.5)
    w = (torch.randn(N, K, dtype=torch.bfloat16, device=DEV) * 0.05)
    out = torch.empty(M, N, dtype=torch.bfloat16, device=DEV)
    ws = torch.empty(48 * 64 * N, dtype=torch.float32, device=DEV)
    _C.skinny_gemm6(out, x, w, ws)
    ref = (x.float() @ w.float().T)
    torch.testing.assert_close(out.float().cpu(), ref.cpu(), rtol=3e-2,
                               atol=6e-2)


def test_skinny_gemm5_fused_norm():
    """skinny5 + fused split-K reduce + residual add + RMSNorm (the
    decode down epilogue) vs the unfused reference chain."""
    from kukeon_amd import _C
    torch.manual_seed(7)
    M, N, K = 64, 4096, 14336
    x = torch.randn(M, K, dtype=torch.bfloat16, device=DEV) * 0.3
    w = torch.randn(N, K, dtype=torch.bfloat16, device=DEV) * 0.05
    resid = torch.randn(M, N, dtype=torch.bfloat16, device=DEV)
    nw = torch.rand(N, dtype=torch.bfloat16, device=DEV) + 0.5
    eps = 1e-5
    resid_ref = resid.clone()
    ws = torch.empty(32 * 64 * N, dtype=torch.float32, device=DEV)
    normed = torch.empty(M, N, dtype=torch.bfloat16, device=DEV)
    _C.skinny_gemm5_fused_norm(normed, x, w, ws, resid, nw, eps)
    # reference: fp32 gemm -> bf16 add -> rmsnorm
    y = (x.float() @ w.float().T)
    r2 = (resid_ref.float() + y).to(torch.bfloat16).float()
    ref = (r2 * torch.rsqrt(r2.pow(2).mean(-1, keepdim=True) + eps) *
           nw.float())
    torch.testing.assert_close(normed.float().cpu(), ref.cpu(), rtol=4e-2,
                               atol=6e-2)
    torch.testing.assert_close(resid.float().cpu(), r2.cpu(), rtol=3e-2,
                               atol=5e-2)


@pytest.mark.parametrize("M", [1, 16, 64])
def test_skinny_gemm5_silu_fused_norm(M):
    """Whole decode MLP tail fused — silu(gate)*up folded into the down
    GEMM's x staging + reduce/add/RMSNorm epilogue — vs the unfused
    fp32 reference chain."""
    from kukeon_amd import _C
    torch.manual_seed(11)
    N, K = 4096, 14336
    gu = torch.randn(M, 2 * K, dtype=torch.bfloat16, device=DEV) * 0.4
    w = torch.randn(N, K, dtype=torch.bfloat16, device=DEV) * 0.05
    resid = torch.randn(M, N, dtype=torch.bfloat16, device=DEV)
    nw = torch.rand(N, dtype=torch.bfloat16, device=DEV) + 0.5
    eps = 1e-5
    resid_ref = resid.clone()
    ws = torch.empty(32 * 64 * N, dtype=torch.float32, device=DEV)
    normed = torch.empty(M, N, dtype=torch.bfloat16, device=DEV)
    _C.skinny_gemm5_silu_fused_norm(normed, gu, w, ws, resid, nw, eps)
    g = gu[:, :K].float()
    u = gu[:, K:].float()
    act = (torch.nn.functional.silu(g) * u).to(torch.bfloat16)
    y = act.float() @ w.float().T
    r2 = (resid_ref.float() + y).to(torch.bfloat16).float()
    ref = (r2 * torch.rsqrt(r2.pow(2).mean(-1, keepdim=True) + eps) *
           nw.float())
    torch.testing.assert_close(normed.float().cpu(), ref.cpu(), rtol=4e-2,
                               atol=6e-2)
    torch.testing.assert_close(resid.float().cpu(), r2.cpu(), rtol=3e-2,
                               atol=5e-2)


@pytest.mark.parametrize("T,E,K", [(32, 8, 2), (1, 8, 2), (7, 16, 4)])
def test_moe_router_weights(T, E, K):
    """Fused softmax/top-K/renorm/scatter router vs the eager chain."""
    from kukeon_amd import _C
    torch.manual_seed(23)
    logits = torch.randn(T, E, dtype=torch.float32, device=DEV) * 2
    w = torch.empty(T, E, dtype=torch.float32, device=DEV)
    _C.moe_router_weights(w, logits, K)
    # bf16 input path must agree with the fp32 path on bf16-rounded data
    wb = torch.empty(T, E, dtype=torch.float32, device=DEV)
    lb = logits.to(torch.bfloat16)
    _C.moe_router_weights(wb, lb, K)
    _C.moe_router_weights(w, lb.float(), K)
    torch.testing.assert_close(wb.cpu(), w.cpu(), rtol=1e-4, atol=1e-5)
    _C.moe_router_weights(w, logits, K)
    probs = torch.softmax(logits, dim=-1)
    topv, topi = probs.topk(K, dim=-1)
    topv = topv / topv.sum(dim=-1, keepdim=True)
    ref = torch.zeros(T, E, dtype=torch.float32, device=DEV)
    ref.scatter_(1, topi, topv)
    torch.testing.assert_close(w.cpu(), ref.cpu(), rtol=1e-4, atol=1e-5)
    assert (w != 0).sum().item() == T * K


def test_moe_dense_combine():
    """out = sum_e w[t,e] * y[e,t,:] vs the eager fp32 chain."""
    from kukeon_amd import _C
    torch.manual_seed(29)
    E, T, H = 8, 32, 4096
    y = torch.randn(E, T, H, dtype=torch.bfloat16, device=DEV)
    w = torch.zeros(T, E, dtype=torch.float32, device=DEV)
    idx = torch.stack([torch.randperm(E, device=DEV)[:2]
                       for _ in range(T)])
    vals = torch.rand(T, 2, device=DEV) + 0.1
    vals = vals / vals.sum(-1, keepdim=True)
    w.scatter_(1, idx, vals)
    out = torch.empty(T, H, dtype=torch.bfloat16, device=DEV)
    _C.moe_dense_combine(out, y, w)
    ref = (y.float() * w.t().unsqueeze(-1)).sum(dim=0).to(torch.bfloat16)
    torch.testing.assert_close(out.float().cpu(), ref.float().cpu(),
                               rtol=2e-2, atol=2e-2)


@pytest.mark.parametrize("kv_dtype", ["bf16", "fp8"])
def test_decode_graphs_match_eager(kv_dtype):
    """Self-advancing graph decode must produce the same tokens as the
    eager path (same seeds, greedy), for both KV cache dtypes."""
    from kukeon_amd.engine.config import EngineConfig, SamplingParams, tiny_llama
    from kukeon_amd.engine.engine import LLMEngine
    from kukeon_amd.engine.kv_cache import SequenceKV
    from kukeon_amd.models.llama import LlamaModel

    cfg = tiny_llama()
    prompt = [7, 3, 99, 140, 11, 42, 17, 23, 5, 81]
    outs = {}
    for use_graphs in (False, True):
        torch.manual_seed(0)
        # graph_buckets pinned so bucket == nrows == 1: the graphed and
        # eager paths then run the exact same kernel sequence (a padded
        # bucket changes GEMM reduction order, which legitimately flips
        # argmax ties on a random-init tiny model).
        ecfg = EngineConfig(max_model_len=256, max_sessions=4,
                            num_kv_blocks=128, use_graphs=use_graphs,
                            decode_microbatch=4, graph_buckets=(1, 2, 4),
                            kv_dtype=kv_dtype)
        model = LlamaModel(cfg, device=DEV)
        engine = LLMEngine(model, cfg, ecfg, device=DEV)
        kv = SequenceKV(ecfg.block_size)
        engine.add_request(kv, prompt,
                           SamplingParams(temperature=0.0, max_new_tokens=9))
        toks = []
        while engine.has_work():
            for o in engine.step():
                toks.extend(o.new_tokens)
        outs[use_graphs] = toks
    assert outs[False] == outs[True], outs
    assert len(outs[True]) == 9


def test_mixtral_gpu_engine_and_moe_paths():
    """Tiny-Mixtral on the GPU: the dense-routed (graph-capturable) decode
    MoE and the permute-kernel sparse path must agree, and the engine must
    generate end to end on the HIP kernels."""
    from kukeon_amd.engine.config import (EngineConfig, SamplingParams,
                                          tiny_mixtral)
    from kukeon_amd.engine.engine import LLMEngine
    from kukeon_amd.engine.kv_cache import SequenceKV
    from kukeon_amd.models.mixtral import MixtralModel

    torch.manual_seed(1)
    cfg = tiny_mixtral()
    model = MixtralModel(cfg, device=DEV)
    moe = model.layers[0].moe
    x = torch.randn(11, cfg.hidden_size, dtype=torch.bfloat16, device=DEV)
    sparse = moe._forward_sparse(x.clone())
    dense = moe._forward_dense(x.clone())
    torch.testing.assert_close(dense.float(), sparse.float(), rtol=3e-2,
                               atol=3e-2)

    ecfg = EngineConfig(max_model_len=256, max_sessions=4, num_kv_blocks=128,
                        use_graphs=False)
    engine = LLMEngine(model, cfg, ecfg, device=DEV)
    kv = SequenceKV(ecfg.block_size)
    engine.add_request(kv, [1, 2, 3, 4, 5],
                       SamplingParams(temperature=0.0, max_new_tokens=6))
    toks = []
    while engine.has_work():
        for o in engine.step():
            toks.extend(o.new_tokens)
    assert len(toks) == 6
    assert all(0 <= t < cfg.vocab_size for t in toks)


def test_engine_gpu_stress_churn():
    """Session churn under KV pressure on the HIP path: random-length
    requests with mixed sampling params, a KV pool small enough to force
    preemption-by-recompute, and graphs on. Every request must finish
    with exactly its requested token count."""
    import random

    from kukeon_amd.engine.config import (EngineConfig, SamplingParams,
                                          tiny_llama)
    from kukeon_amd.engine.engine import LLMEngine
    from kukeon_amd.engine.kv_cache import SequenceKV
    from kukeon_amd.models.llama import LlamaModel

    rng = random.Random(0)
    torch.manual_seed(0)
    cfg = tiny_llama()
    ecfg = EngineConfig(max_model_len=192, max_sessions=8, num_kv_blocks=48,
                        use_graphs=True, decode_microbatch=8,
                        graph_buckets=(2, 4, 8))
    model = LlamaModel(cfg, device=DEV)
    engine = LLMEngine(model, cfg, ecfg, device=DEV)

    want = {}
    got = {}
    sessions = {}
    for i in range(24):
        kv = SequenceKV(ecfg.block_size)
        sessions[i] = kv
        n = rng.randint(2, 12)
        rid = engine.add_request(
            kv, [rng.randrange(cfg.vocab_size) for _ in range(
                rng.randint(3, 40))],
            SamplingParams(temperature=rng.choice([0.0, 0.7, 1.0]),
                           top_k=rng.choice([0, 5, 50]),
                           top_p=rng.choice([1.0, 0.9]),
                           max_new_tokens=n))
        want[rid] = n
        got[rid] = 0
    steps = 0
    while engine.has_work():
        for o in engine.step():
            got[o.req_id] += len(o.new_tokens)
        steps += 1
        assert steps < 2000
    assert got == want
    for kv in sessions.values():
        assert all(0 <= t < cfg.vocab_size for t in kv.history)
        engine.free_sequence(kv)
    assert engine.kv.allocator.num_free == ecfg.num_kv_blocks


def test_engine_gpu_preemption_recompute():
    """Mid-decode KV exhaustion on the HIP path: a growing multi-turn
    session forces preempt-by-recompute; every turn must still deliver
    its full token count and end with a consistent context length."""
    from kukeon_amd.engine.config import (EngineConfig, SamplingParams,
                                          tiny_llama)
    from kukeon_amd.engine.engine import LLMEngine
    from kukeon_amd.engine.kv_cache import SequenceKV
    from kukeon_amd.models.llama import LlamaModel

    torch.manual_seed(3)
    cfg = tiny_llama()
    ecfg = EngineConfig(max_model_len=512, max_sessions=2, num_kv_blocks=14,
                        use_graphs=True, decode_microbatch=8,
                        graph_buckets=(2, 4))
    model = LlamaModel(cfg, device=DEV)
    engine = LLMEngine(model, cfg, ecfg, device=DEV)
    kvs = [SequenceKV(ecfg.block_size) for _ in range(2)]
    expect = [0, 0]
    for turn in range(4):
        rids = {}
        for i, kv in enumerate(kvs):
            rid = engine.add_request(
                kv, [11 + turn, 7, 5], SamplingParams(temperature=0.0,
                                                      max_new_tokens=24))
            rids[rid] = i
        got = {rid: 0 for rid in rids}
        steps = 0
        while engine.has_work():
            for o in engine.step():
                got[o.req_id] += len(o.new_tokens)
            steps += 1
            assert steps < 3000
        assert all(v == 24 for v in got.values()), got
        for i in range(2):
            expect[i] += 3 + 24
    for i, kv in enumerate(kvs):
        # context length: all tokens of all turns, last one pending
        assert len(kv.history) + kv.num_tokens >= expect[i] - 1


def test_engine_gpu_stop_token_rollback():
    """Stop-token finish mid-microbatch on the graphed HIP decode path:
    the device ring already advanced past the stop — the host rollback
    must leave the context ending at the stop token, and the session
    must keep serving."""
    from kukeon_amd.engine.config import (EngineConfig, SamplingParams,
                                          tiny_llama)
    from kukeon_amd.engine.engine import LLMEngine
    from kukeon_amd.engine.kv_cache import SequenceKV
    from kukeon_amd.models.llama import LlamaModel

    torch.manual_seed(0)
    cfg = tiny_llama()
    ecfg = EngineConfig(max_model_len=512, max_sessions=2,
                        num_kv_blocks=256, use_graphs=True,
                        decode_microbatch=8, graph_buckets=(2,))
    model = LlamaModel(cfg, device=DEV)
    engine = LLMEngine(model, cfg, ecfg, device=DEV)
    prompt = [7, 3, 99, 140, 11, 42]
    kv = SequenceKV(ecfg.block_size)
    engine.add_request(kv, prompt,
                       SamplingParams(temperature=0.0, max_new_tokens=16))
    ref = []
    while engine.has_work():
        for o in engine.step():
            ref.extend(o.new_tokens)
    assert len(ref) == 16
    stop = ref[2]
    cut = ref.index(stop)  # first occurrence ends the stopped run

    torch.manual_seed(0)
    model2 = LlamaModel(cfg, device=DEV)
    engine2 = LLMEngine(model2, cfg, ecfg, device=DEV)
    kv2 = SequenceKV(ecfg.block_size)
    engine2.add_request(kv2, prompt,
                        SamplingParams(temperature=0.0, max_new_tokens=16,
                                       stop_token_ids=(stop,)))
    got = []
    while engine2.has_work():
        for o in engine2.step():
            got.extend(o.new_tokens)
    assert got == ref[:cut + 1]
    assert kv2.pending_token == stop
    assert kv2.num_tokens == len(prompt) + len(got) - 1
    assert len(kv2.history) == kv2.num_tokens
    # session continues cleanly after the rollback
    engine2.add_request(kv2, [5],
                        SamplingParams(temperature=0.0, max_new_tokens=4))
    more = []
    while engine2.has_work():
        for o in engine2.step():
            more.extend(o.new_tokens)
    assert len(more) == 4


def test_mfma_probe16k_layout():
    from kukeon_amd import _C
    torch.manual_seed(0)
    a = (torch.randn(16, 16) * torch.linspace(0.2, 2.0, 16)).bfloat16().to(DEV)
    b = (torch.randn(16, 16) * torch.linspace(-1.5, 1.5, 16)).bfloat16().to(DEV)
    out = torch.empty(16, 16, dtype=torch.float32, device=DEV)
    _C.mfma_probe16k(out, a, b)
    ref = a.float() @ b.float()
    torch.testing.assert_close(out.cpu(), ref.cpu(), rtol=2e-2, atol=2e-2)


@pytest.mark.parametrize("mode,splits,fp8", [
    ("1", 4, False), ("1", 8, False), ("1", 4, True),
    ("32", 4, False), ("32", 8, False), ("32", 4, True)])
def test_paged_attention_mfma_path(mode, splits, fp8, monkeypatch):
    """The opt-in MFMA decode-attention kernels (KUKEON_ATTN_MFMA=1 is the
    16-token-tile 16x16 variant, =32 the 32-token 32x32 one) must match
    the fp32 reference like the default v_dot2 kernel does."""
    monkeypatch.setenv("KUKEON_ATTN_MFMA", mode)
    torch.manual_seed(5)
    B, Hk, D, BS = 5, 2, 128, 16
    Hq = 4 * Hk
    ctxs = [1, 16, 17, 100, 250]
    NB = sum((c + BS - 1) // BS for c in ctxs) + 4
    kc, vc = _make_cache(NB, Hk, BS, D, fp8=fp8)
    tol = 7e-2 if fp8 else 2e-2
    maxb = max((c + BS - 1) // BS for c in ctxs)
    bt = torch.zeros(B, maxb, dtype=torch.int32, device=DEV)
    nxt = 0
    for b, c in enumerate(ctxs):
        n = (c + BS - 1) // BS
        bt[b, :n] = torch.arange(nxt, nxt + n, dtype=torch.int32)
        nxt += n
    seq_lens = torch.tensor(ctxs, dtype=torch.int32, device=DEV)
    q = torch.randn(B, Hq * D, dtype=torch.bfloat16, device=DEV)
    out = torch.empty(B, Hq * D, dtype=torch.bfloat16, device=DEV)
    scale = D ** -0.5
    tmp_out = torch.zeros(B, Hq, splits, D, dtype=torch.float32, device=DEV)
    tmp_ml = torch.zeros(B, Hq, splits, 2, dtype=torch.float32, device=DEV)
    ops.paged_attention(out, q, kc, vc, bt, seq_lens, 0, splits, scale,
                        tmp_out, tmp_ml)
    ref = torch.empty(B, Hq * D, dtype=torch.bfloat16)
    reference.paged_attention(ref, q.cpu(), kc.cpu(), vc.cpu(), bt.cpu(),
                              seq_lens.cpu(), 0, 1, scale)
    torch.testing.assert_close(out.cpu().float(), ref.float(), rtol=tol,
                               atol=tol)


@pytest.mark.parametrize("M", [1, 17, 64])
def test_linear_add_rmsnorm_fused(M):
    """The fused o-proj epilogue (split-K reduce + residual add + RMSNorm
    in one kernel) must match linear() + fused_add_rmsnorm()."""
    torch.manual_seed(11)
    N = K = 4096
    x = torch.randn(M, K, dtype=torch.bfloat16, device=DEV) * 0.3
    w = torch.randn(N, K, dtype=torch.bfloat16, device=DEV) * 0.02
    resid = torch.randn(M, N, dtype=torch.bfloat16, device=DEV)
    nw = torch.randn(N, dtype=torch.bfloat16, device=DEV)
    resid_ref = resid.cpu().clone()
    h_ref = (x.cpu().float() @ w.cpu().float().T).to(torch.bfloat16)
    reference.fused_add_rmsnorm(h_ref, resid_ref, nw.cpu(), 1e-5)
    out = ops.linear_add_rmsnorm(x, w, resid, nw, 1e-5)
    torch.testing.assert_close(resid.cpu().float(), resid_ref.float(),
                               rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(out.cpu().float(), h_ref.float(), rtol=2e-2,
                               atol=3e-2)


def test_modelhub_server_gpu_streaming():
    """The serving stack end to end on the GPU: engine-loop thread with
    hipGraphs, unix-socket protocol, concurrent sessions, streaming."""
    import threading
    import uuid

    from kukeon_amd.engine.config import EngineConfig, tiny_llama
    from kukeon_amd.models.llama import LlamaModel
    from kukeon_amd.serve.server import ModelhubClient, ModelhubServer

    torch.manual_seed(0)
    cfg = tiny_llama()
    ecfg = EngineConfig(max_model_len=256, max_sessions=8, num_kv_blocks=256,
                        use_graphs=True, decode_microbatch=4,
                        graph_buckets=(2, 4, 8))
    sock = f"/tmp/mhg-{uuid.uuid4().hex[:8]}.sock"
    hub = ModelhubServer(LlamaModel(cfg, device=DEV), cfg, ecfg, sock,
                         device=DEV)
    hub.start()
    try:
        results = {}

        def worker(name):
            c = ModelhubClient(sock, timeout=120)
            deltas, final = [], None
            for frame in c.generate_stream(name, [3, 1, 4, 1, 5],
                                           max_new_tokens=8,
                                           temperature=0.0):
                if "delta" in frame:
                    deltas.extend(frame["delta"])
                else:
                    final = frame
            r2 = c.generate(name, [9, 2], max_new_tokens=4, temperature=0.7)
            results[name] = (deltas, final, r2)
            c.close()

        ts = [threading.Thread(target=worker, args=(f"s{i}",))
              for i in range(4)]
        for t in ts:
            t.start()
        for t in ts:
            t.join(timeout=120)
        assert len(results) == 4
        for deltas, final, r2 in results.values():
            assert final is not None and len(final["tokens"]) == 8
            assert deltas == final["tokens"][: len(deltas)]
            assert len(r2["tokens"]) == 4
            assert r2["context_len"] > 8  # persistent multi-turn KV
    finally:
        hub.stop()
