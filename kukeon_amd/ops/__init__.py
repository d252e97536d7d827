"""Device-op dispatch for kukeon_amd.

On a CUDA (ROCm) device every op is the hand-written gfx950 HIP kernel from
``kukeon_amd._C`` — if the extension is missing there we raise instead of
falling back, so a GPU run can never silently use an eager path.  On CPU the
fp32 reference implementations run (tests, GPU-less control-plane hosts).
"""
from __future__ import annotations

import torch

from . import reference

_C = None
_IMPORT_ERROR: Exception | None = None
try:
    from kukeon_amd import _C  # type: ignore
except Exception as e:  # pragma: no cover - exercised only on broken installs
    _IMPORT_ERROR = e


def native_available() -> bool:
    return _C is not None


def _native():
    if _C is None:
        raise RuntimeError(
            "kukeon_amd._C (gfx950 HIP extension) is not built but an op was "
            "called on a CUDA device. Run `python -m kukeon_amd.ops.build` "
            f"(import error: {_IMPORT_ERROR})")
    return _C


def _impl(t: torch.Tensor):
    return _native() if t.is_cuda else reference


def rmsnorm(out, input, weight, eps: float) -> None:
    _impl(input).rmsnorm(out, input, weight, eps)


def fused_add_rmsnorm(input, residual, weight, eps: float) -> None:
    _impl(input).fused_add_rmsnorm(input, residual, weight, eps)


def silu_mul(out, gate_up) -> None:
    _impl(gate_up).silu_mul(out, gate_up)


def rope_kv_append(qkv, k_cache, v_cache, cos_sin, positions, slot_mapping,
                   num_q_heads: int, num_kv_heads: int, head_dim: int,
                   block_table=None) -> None:
    if block_table is None:
        block_table = torch.empty(0, dtype=torch.int32, device=qkv.device)
    _impl(qkv).rope_kv_append(qkv, k_cache, v_cache, cos_sin, positions,
                              slot_mapping, num_q_heads, num_kv_heads,
                              head_dim, block_table)


def decode_advance(ids, pos, seq_lens, tokens, ring, counter) -> None:
    _impl(ids).decode_advance(ids, pos, seq_lens, tokens, ring, counter)


def paged_attention(out, q, k_cache, v_cache, block_table, seq_lens,
                    q_offset: int, num_splits: int, scale: float,
                    tmp_out, tmp_ml) -> None:
    _impl(q).paged_attention(out, q, k_cache, v_cache, block_table, seq_lens,
                             q_offset, num_splits, scale, tmp_out, tmp_ml)


def prefill_attention(out, q, k_cache, v_cache, block_table, seq_lens,
                      q_starts, qb_seq, qb_start, q_offset: int,
                      scale: float) -> None:
    _impl(q).prefill_attention(out, q, k_cache, v_cache, block_table,
                               seq_lens, q_starts, qb_seq, qb_start, q_offset,
                               scale)


def sample(tokens, logits, temps, top_k, top_p, seed, workspace) -> None:
    _impl(logits).sample(tokens, logits, temps, top_k, top_p, seed, workspace)


_SKINNY_WS = {}
# replaced (outgrown) workspaces are pinned here forever: a hipGraph
# captured earlier may still hold the old tensor's device pointer, and
# letting it free would put a use-after-free inside every later replay
_SKINNY_WS_RETIRED = []
# Per-shape dispatch, from the measured sweep (profiles/r01_progress.md,
# scripts/sweep_splitk.py on MI355X): the glds-staged skinny kernel beats
# hipBLASLt on square o-projection shapes (N==K: 16.6us vs 19.3 at M=64,
# 11.9 vs 19.2 at M=16 on 4096x4096) and trails it on qkv/gate_up/down/
# lm_head, where hipBLASLt is 55-100% of the HBM floor. KUKEON_SKINNY_GEMM=1
# forces the skinny kernel for every eligible shape (benchmarking).
_USE_SKINNY = __import__("os").environ.get("KUKEON_SKINNY_GEMM", "0")
# silu-into-down-staging fusion: measured SLOWER in situ (54.2us fused vs
# 42.9 for silu_mul + gemm cold; bench 51.7 vs 53.6 turns/s same box) —
# the doubled x staging traffic outweighs the saved launch. Off by
# default; kernel + numerics test stay as the documented negative result.
_FUSE_SILU = __import__("os").environ.get("KUKEON_FUSE_SILU", "0") == "1"


def _skinny_wins(rows: int, N: int, K: int) -> bool:
    if _USE_SKINNY == "off":   # A/B: force the library everywhere
        return False
    return N == K == 4096


# v5 (full-line never-drain pipeline, KS=128/2 blocks-per-CU): beats
# hipBLASLt cold on the down-projection (32.2us vs 39.6 on 4096x14336 at
# M=64, floor 18.6 — profiles/r02_progress.md); qkv ties, gate_up and
# lm_head stay on the library. KUKEON_SKINNY_GEMM=5 forces it everywhere
# for benchmarking.
def _skinny5_wins(rows: int, N: int, K: int) -> bool:
    if _USE_SKINNY == "off":   # A/B: force the library everywhere
        return False
    # llama-3-8b down (32.2us vs blas 39.6 cold) and llama-3-70b down
    # (79.3 vs 83.1 at M=16, 92.6 vs 139.3 at M=64) — measured on
    # MI355X, profiles/r02_progress.md
    return (N, K) in ((4096, 14336), (8192, 28672))


def linear(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """F.linear with the weight-streaming skinny-GEMM kernel on the decode
    shapes where it measured faster than hipBLASLt; hipBLASLt otherwise."""
    rows = x.shape[0]
    if (x.is_cuda and x.dim() == 2 and rows <= 64
            and w.shape[0] % 128 == 0 and w.shape[1] % 128 == 0
            and x.dtype == torch.bfloat16
            and (_USE_SKINNY == "5"
                 or _skinny5_wins(rows, w.shape[0], w.shape[1]))):
        N = w.shape[0]
        K = w.shape[1]
        out = torch.empty(rows, N, dtype=x.dtype, device=x.device)
        key = (x.device.index or 0)
        ws = _SKINNY_WS.get(key)
        ngroups = N // 128
        nslices = -(-K // 128)
        splitk = 1 if ngroups >= 256 else min(nslices, -(-256 // ngroups))
        need = max(1, splitk) * 64 * N
        if ws is None or ws.numel() < need:
            if ws is not None:
                _SKINNY_WS_RETIRED.append(ws)
            ws = torch.empty(need, dtype=torch.float32, device=x.device)
            _SKINNY_WS[key] = ws
        _native().skinny_gemm5(out, x, w, ws)
        return out
    if (x.is_cuda and x.dim() == 2 and rows <= 64
            and w.shape[0] % 64 == 0 and w.shape[1] % 32 == 0
            and x.dtype == torch.bfloat16
            and (_USE_SKINNY == "1"
                 or _skinny_wins(rows, w.shape[0], w.shape[1]))):
        N = w.shape[0]
        K = w.shape[1]
        out = torch.empty(rows, N, dtype=x.dtype, device=x.device)
        key = (x.device.index or 0)
        ws = _SKINNY_WS.get(key)
        ntiles = N // 64
        nslices = -(-K // 256)
        splitk = 1 if ntiles >= 512 else min(nslices, -(-256 // ntiles))
        need = max(1, splitk) * 64 * N
        if ws is None or ws.numel() < need:
            # grown only outside graph capture (engine warmup runs eager)
            if ws is not None:
                _SKINNY_WS_RETIRED.append(ws)
            ws = torch.empty(need, dtype=torch.float32, device=x.device)
            _SKINNY_WS[key] = ws
        _native().skinny_gemm(out, x, w, ws)
        return out
    return torch.nn.functional.linear(x, w)


def linear_add_rmsnorm(x: torch.Tensor, w: torch.Tensor,
                       residual: torch.Tensor, norm_weight: torch.Tensor,
                       eps: float) -> torch.Tensor:
    """out-projection + TP all-reduce + residual add + RMSNorm as one
    fused path. On the skinny decode shapes (TP=1) the split-K reduce,
    residual add and norm run in a single kernel — the separate epilogue
    pair cost two launches at the ~4.5us in-graph dispatch floor for
    <1us of work. Mutates `residual` (+= x@w.T) and returns the normed
    activations, exactly like linear() followed by fused_add_rmsnorm().
    """
    from kukeon_amd import parallel
    rows = x.shape[0]
    N, K = w.shape
    if (x.is_cuda and parallel.tp_size() == 1 and x.dim() == 2
            and rows <= 64 and N % 2048 == 0 and N <= 8192 and K % 128 == 0
            and x.dtype == torch.bfloat16
            and (_USE_SKINNY == "5" or _skinny5_wins(rows, N, K))):
        key = (x.device.index or 0)
        ws = _SKINNY_WS.get(key)
        ngroups = N // 128
        nslices = -(-K // 128)
        splitk = min(nslices, -(-256 // ngroups))
        need = max(1, splitk) * 64 * N
        if ws is None or ws.numel() < need:
            if ws is not None:
                _SKINNY_WS_RETIRED.append(ws)
            ws = torch.empty(need, dtype=torch.float32, device=x.device)
            _SKINNY_WS[key] = ws
        normed = torch.empty(rows, N, dtype=x.dtype, device=x.device)
        _native().skinny_gemm5_fused_norm(normed, x, w, ws, residual,
                                          norm_weight, eps)
        return normed
    if (x.is_cuda and parallel.tp_size() == 1 and x.dim() == 2
            and rows <= 64 and N % 2048 == 0 and N <= 8192 and K % 32 == 0
            and x.dtype == torch.bfloat16
            and (_USE_SKINNY == "1" or _skinny_wins(rows, N, K))):
        key = (x.device.index or 0)
        ws = _SKINNY_WS.get(key)
        ntiles = N // 64
        nslices = -(-K // 256)
        splitk = min(nslices, -(-256 // ntiles))
        need = max(1, splitk) * 64 * N
        if ws is None or ws.numel() < need:
            if ws is not None:
                _SKINNY_WS_RETIRED.append(ws)
            ws = torch.empty(need, dtype=torch.float32, device=x.device)
            _SKINNY_WS[key] = ws
        normed = torch.empty(rows, N, dtype=x.dtype, device=x.device)
        _native().skinny_gemm_fused_norm(normed, x, w, ws, residual,
                                         norm_weight, eps)
        return normed
    h = parallel.tp_all_reduce(linear(x, w))
    fused_add_rmsnorm(h, residual, norm_weight, eps)
    return h



def mlp_down_fused(gu: torch.Tensor, w: torch.Tensor,
                   residual: torch.Tensor, norm_weight: torch.Tensor,
                   eps: float) -> torch.Tensor:
    """The decode MLP tail — silu(gate)*up -> down-projection ->
    TP all-reduce -> residual add -> RMSNorm — with the activation fused
    into the down GEMM's x staging on the skinny decode shapes (the
    standalone silu_mul launch and its act-tensor HBM round trip
    disappear; profiles/r02_progress.md). `gu` is the fused gate_up GEMM
    output [rows, 2K]. Equivalent to silu_mul + linear_add_rmsnorm.
    """
    from kukeon_amd import parallel
    rows = gu.shape[0]
    N, K = w.shape
    if (_FUSE_SILU and gu.is_cuda and parallel.tp_size() == 1
            and gu.dim() == 2
            and rows <= 64 and gu.shape[1] == 2 * K
            and N % 2048 == 0 and N <= 8192 and K % 128 == 0
            and gu.dtype == torch.bfloat16
            and (_USE_SKINNY == "5" or _skinny5_wins(rows, N, K))):
        key = (gu.device.index or 0)
        ws = _SKINNY_WS.get(key)
        ngroups = N // 128
        nslices = -(-K // 128)
        splitk = min(nslices, -(-256 // ngroups))
        need = max(1, splitk) * 64 * N
        if ws is None or ws.numel() < need:
            if ws is not None:
                _SKINNY_WS_RETIRED.append(ws)
            ws = torch.empty(need, dtype=torch.float32, device=gu.device)
            _SKINNY_WS[key] = ws
        normed = torch.empty(rows, N, dtype=gu.dtype, device=gu.device)
        _native().skinny_gemm5_silu_fused_norm(normed, gu, w, ws,
                                               residual, norm_weight, eps)
        return normed
    act = torch.empty(rows, K, dtype=gu.dtype, device=gu.device)
    silu_mul(act, gu)
    return linear_add_rmsnorm(act, w, residual, norm_weight, eps)


def moe_router_weights(wdense, logits, k: int) -> None:
    _impl(logits).moe_router_weights(wdense, logits, k)


def moe_dense_combine(out, y, wdense) -> None:
    _impl(y).moe_dense_combine(out, y, wdense)


def moe_gather_tokens(out, input, row_map) -> None:
    _impl(input).moe_gather_tokens(out, input, row_map)


def moe_scatter_tokens(out, input, inv_map, weights, top_k: int) -> None:
    _impl(input).moe_scatter_tokens(out, input, inv_map, weights, top_k)
