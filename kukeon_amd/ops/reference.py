"""Plain-PyTorch fp32 reference implementations of every HIP op.

These are the numerics oracles for the GPU test suite (each HIP kernel is
compared against the fp32 reference of the same op, per the repo test
contract) and the CPU execution path for control-plane / scheduler tests on
GPU-less hosts.  They are never used on a CUDA device — `kukeon_amd.ops`
fails loudly there if the native extension is missing.
"""
from __future__ import annotations

import torch


def rmsnorm(out: torch.Tensor, input: torch.Tensor, weight: torch.Tensor,
            eps: float) -> None:
    x = input.float()
    rs = torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + eps)
    out.copy_((x * rs * weight.float()).to(out.dtype))


def fused_add_rmsnorm(input: torch.Tensor, residual: torch.Tensor,
                      weight: torch.Tensor, eps: float) -> None:
    r = (residual.float() + input.float())
    residual.copy_(r.to(residual.dtype))
    # match the kernel: it normalizes the bf16-rounded residual it wrote
    r = residual.float()
    rs = torch.rsqrt(r.pow(2).mean(-1, keepdim=True) + eps)
    input.copy_((r * rs * weight.float()).to(input.dtype))


def silu_mul(out: torch.Tensor, gate_up: torch.Tensor) -> None:
    I = out.shape[-1]
    g = gate_up[..., :I].float()
    u = gate_up[..., I:].float()
    out.copy_((torch.nn.functional.silu(g) * u).to(out.dtype))


def _rotate(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    # NEOX half-rotation; x [T, H, D], cos/sin [T, D/2]
    half = x.shape[-1] // 2
    x1, x2 = x[..., :half], x[..., half:]
    c = cos.unsqueeze(1)
    s = sin.unsqueeze(1)
    return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)


def rope_kv_append(qkv: torch.Tensor, k_cache: torch.Tensor,
                   v_cache: torch.Tensor, cos_sin: torch.Tensor,
                   positions: torch.Tensor, slot_mapping: torch.Tensor,
                   num_q_heads: int, num_kv_heads: int, head_dim: int,
                   block_table=None) -> None:
    T = qkv.shape[0]
    D = head_dim
    half = D // 2
    q = qkv[:, : num_q_heads * D].view(T, num_q_heads, D).float()
    k = qkv[:, num_q_heads * D: (num_q_heads + num_kv_heads) * D].view(
        T, num_kv_heads, D).float()
    v = qkv[:, (num_q_heads + num_kv_heads) * D:].view(T, num_kv_heads, D)
    cs = cos_sin[positions.long().clamp_min(0)]
    cos, sin = cs[:, :half], cs[:, half:]
    qr = _rotate(q, cos, sin).to(qkv.dtype)
    kr = _rotate(k, cos, sin).to(qkv.dtype)
    # inactive decode rows (pos < 0) are skipped entirely, like the kernel
    act = (positions >= 0).view(T, 1)
    qkv[:, : num_q_heads * D] = torch.where(
        act, qr.reshape(T, -1), qkv[:, : num_q_heads * D])
    qkv[:, num_q_heads * D: (num_q_heads + num_kv_heads) * D] = torch.where(
        act, kr.reshape(T, -1),
        qkv[:, num_q_heads * D: (num_q_heads + num_kv_heads) * D])
    BS = k_cache.shape[2]
    fp8 = k_cache.dtype == torch.uint8
    if fp8:
        kr = kr.float().to(torch.float8_e4m3fn).view(torch.uint8)
        v = v.float().to(torch.float8_e4m3fn).view(torch.uint8)
    table_mode = block_table is not None and block_table.dim() == 2
    for t in range(T):
        if table_mode:
            pos_t = int(positions[t])
            if pos_t < 0:
                continue
            slot = int(block_table[t, pos_t // BS]) * BS + pos_t % BS
        else:
            slot = int(slot_mapping[t])
            if slot < 0:
                continue
        blk, off = slot // BS, slot % BS
        k_cache[blk, :, off] = kr[t]
        v_cache[blk, :, off] = v[t]


def _gather_kv(cache: torch.Tensor, block_table: torch.Tensor, ctx: int,
               b: int) -> torch.Tensor:
    """-> [ctx, Hk, D] from paged cache [NB, Hk, BS, D] (bf16 or fp8)."""
    BS = cache.shape[2]
    nb = (ctx + BS - 1) // BS
    blocks = block_table[b, :nb].long()
    flat = cache[blocks]                      # [nb, Hk, BS, D]
    if flat.dtype == torch.uint8:
        flat = flat.view(torch.float8_e4m3fn).float()
    flat = flat.permute(0, 2, 1, 3).reshape(nb * BS, cache.shape[1], -1)
    return flat[:ctx]


def paged_attention(out: torch.Tensor, q: torch.Tensor, k_cache: torch.Tensor,
                    v_cache: torch.Tensor, block_table: torch.Tensor,
                    seq_lens: torch.Tensor, q_offset: int, num_splits: int,
                    scale: float, tmp_out=None, tmp_ml=None) -> None:
    B = q.shape[0]
    Hk, BS, D = k_cache.shape[1], k_cache.shape[2], k_cache.shape[3]
    Hq = out.shape[1] // D
    G = Hq // Hk
    qv = q.view(B, -1)[:, q_offset: q_offset + Hq * D].view(B, Hq, D).float()
    for b in range(B):
        ctx = int(seq_lens[b])
        if ctx <= 0:
            # inactive decode row: write zeros so downstream logits stay
            # finite (the row's sample is discarded by the advance guard,
            # but NaNs from uninitialized memory would crash the CPU
            # sampler's multinomial)
            out.view(B, -1)[b] = 0
            continue
        k = _gather_kv(k_cache, block_table, ctx, b).float()  # [ctx,Hk,D]
        v = _gather_kv(v_cache, block_table, ctx, b).float()
        for h in range(Hq):
            hk = h // G
            s = (k[:, hk] @ qv[b, h]) * scale                  # [ctx]
            p = torch.softmax(s, dim=-1)
            o = p @ v[:, hk]                                   # [D]
            out.view(B, Hq, D)[b, h] = o.to(out.dtype)


def prefill_attention(out: torch.Tensor, q: torch.Tensor,
                      k_cache: torch.Tensor, v_cache: torch.Tensor,
                      block_table: torch.Tensor, seq_lens: torch.Tensor,
                      q_starts: torch.Tensor, qb_seq=None, qb_start=None,
                      q_offset: int = 0, scale: float = 1.0) -> None:
    """Varlen causal attention where queries are the tail of each sequence.

    q: [T_total, q_stride] packed new tokens over all sequences (row i of
    sequence s is absolute position q_starts[s] + i); KV (including the new
    tokens, already appended) lives in the paged cache; seq_lens[s] is the
    total context after append. out: [T_total, Hq*D].
    """
    Hk, BS, D = k_cache.shape[1], k_cache.shape[2], k_cache.shape[3]
    Hq = out.shape[1] // D
    G = Hq // Hk
    nseq = seq_lens.shape[0]
    for s in range(nseq):
        ctx = int(seq_lens[s])
        start = int(q_starts[s, 0])
        row = int(q_starts[s, 1])
        qlen = ctx - start
        if qlen <= 0:
            continue
        k = _gather_kv(k_cache, block_table, ctx, s).float()
        v = _gather_kv(v_cache, block_table, ctx, s).float()
        qs = q[row: row + qlen, q_offset: q_offset + Hq * D].view(
            qlen, Hq, D).float()
        pos = torch.arange(start, ctx)
        kvpos = torch.arange(ctx)
        mask = kvpos[None, :] <= pos[:, None]                  # [qlen, ctx]
        for h in range(Hq):
            hk = h // G
            sc = (qs[:, h] @ k[:, hk].T) * scale               # [qlen, ctx]
            sc = sc.masked_fill(~mask, float("-inf"))
            p = torch.softmax(sc, dim=-1)
            o = p @ v[:, hk]
            out[row: row + qlen].view(qlen, Hq, D)[:, h] = o.to(out.dtype)
        pass


def sample(tokens: torch.Tensor, logits: torch.Tensor, temps: torch.Tensor,
           top_k: torch.Tensor, top_p: torch.Tensor, seed: torch.Tensor,
           workspace=None) -> None:
    """Reference sampler. Greedy rows match the kernel exactly; stochastic
    rows draw from the same masked distribution (the GPU test checks the
    distribution, not the draw)."""
    B, V = logits.shape
    lg = logits.float()
    for b in range(B):
        T = float(temps[b])
        row = lg[b]
        if T <= 0:
            tokens[b] = int(torch.argmax(row))
            continue
        k = int(top_k[b])
        p = float(top_p[b])
        probs = torch.softmax((row - row.max()) / T, dim=-1)
        keep = torch.ones(V, dtype=torch.bool)
        if 0 < k < V:
            th = torch.topk(row, k).values.min()
            keep &= row >= th
        if 0 < p < 1:
            srt, idx = torch.sort(probs, descending=True)
            cum = torch.cumsum(srt, 0)
            cut = int(torch.searchsorted(cum, p).clamp(max=V - 1))
            keep2 = torch.zeros(V, dtype=torch.bool)
            keep2[idx[: cut + 1]] = True
            keep &= keep2
        probs = probs * keep
        probs = probs / probs.sum()
        tokens[b] = int(torch.multinomial(probs, 1))
    seed += 1


def linear_add_rmsnorm(x, w, residual, norm_weight, eps: float):
    h = torch.nn.functional.linear(x, w)
    fused_add_rmsnorm(h, residual, norm_weight, eps)
    return h


def decode_advance(ids, pos, seq_lens, tokens, ring, counter) -> None:
    step = int(counter[0])
    B = ids.shape[0]
    for b in range(B):
        if int(seq_lens[b]) > 0:
            ids[b] = tokens[b]
            pos[b] += 1
            seq_lens[b] += 1
            ring[step * B + b] = tokens[b]
    counter[0] = step + 1


def moe_gather_tokens(out: torch.Tensor, input: torch.Tensor,
                      row_map: torch.Tensor) -> None:
    out.copy_(input[row_map.long()])


def moe_scatter_tokens(out: torch.Tensor, input: torch.Tensor,
                       inv_map: torch.Tensor, weights: torch.Tensor,
                       top_k: int) -> None:
    T = out.shape[0]
    acc = (input[inv_map.long()].float() *
           weights.unsqueeze(-1)).sum(dim=1)  # [T, H]
    out.copy_(acc.to(out.dtype))

def moe_router_weights(wdense: torch.Tensor, logits: torch.Tensor,
                       K: int) -> None:
    # accepts bf16 or fp32 logits, like the HIP kernel
    probs = torch.softmax(logits.float(), dim=-1)
    topv, topi = probs.topk(K, dim=-1)
    topv = topv / topv.sum(dim=-1, keepdim=True)
    wdense.zero_()
    wdense.scatter_(1, topi, topv)


def moe_dense_combine(out: torch.Tensor, y: torch.Tensor,
                      wdense: torch.Tensor) -> None:
    acc = (y.float() * wdense.t().unsqueeze(-1)).sum(dim=0)
    out.copy_(acc.to(out.dtype))
