"""Llama-family decoder (Llama-3-8B / 70B shapes) built on the kukeon_amd
gfx950 op set: fused qkv/gate_up GEMMs (hipBLASLt via F.linear), fused
RoPE+KV-append, paged GQA attention, fused add+RMSNorm and SwiGLU kernels.

Tensor parallelism: q/k/v and gate/up are column-sharded (head-aligned),
o_proj and down_proj row-sharded with one RCCL all-reduce each over xGMI;
embed/lm_head are replicated so every rank samples identically with no
collective in the sampling path.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from kukeon_amd import ops
from kukeon_amd import parallel
from kukeon_amd.engine.config import ModelConfig


@dataclass
class AttnMeta:
    """Per-forward attention metadata (all tensors on the model device)."""
    mode: str                       # "prefill" | "decode"
    positions: torch.Tensor         # [T] int32
    slot_mapping: torch.Tensor      # [T] int32
    block_table: torch.Tensor       # [nseq, max_blocks] int32
    seq_lens: torch.Tensor          # [nseq] int32
    q_starts: Optional[torch.Tensor] = None   # [nseq] int32 (prefill)
    qb_seq: Optional[torch.Tensor] = None     # [nqb] int32 (prefill grid)
    qb_start: Optional[torch.Tensor] = None   # [nqb] int32
    num_splits: int = 1
    tmp_out: Optional[torch.Tensor] = None    # decode split-KV workspaces
    tmp_ml: Optional[torch.Tensor] = None


def split_prefill_meta(meta: AttnMeta, parts: int = 2):
    """Split a packed prefill batch into sequence groups (token-balanced)
    for the pipelined TP path. Returns [(row0, row1, AttnMeta), ...]."""
    qs = meta.q_starts  # [nseq, 2] (ctx_start, packed row offset)
    nseq = int(qs.shape[0])
    T = int(meta.positions.shape[0])
    row_offs = [int(qs[i, 1]) for i in range(nseq)] + [T]
    # choose the sequence boundary closest to half the tokens
    target = T // 2
    split = max(1, min(nseq - 1,
                       min(range(1, nseq),
                           key=lambda i: abs(row_offs[i] - target))))
    groups = []
    for (s0, s1) in ((0, split), (split, nseq)):
        r0, r1 = row_offs[s0], row_offs[s1]
        g_qs = qs[s0:s1].clone()
        g_qs[:, 1] -= r0
        qb_seq, qb_start = [], []
        for s in range(s0, s1):
            chunk = row_offs[s + 1] - row_offs[s]
            for qb in range(0, chunk, 32):
                qb_seq.append(s - s0)
                qb_start.append(qb)
        dev = meta.positions.device
        gm = AttnMeta(
            mode="prefill",
            positions=meta.positions[r0:r1],
            slot_mapping=meta.slot_mapping[r0:r1],
            block_table=meta.block_table[s0:s1],
            seq_lens=meta.seq_lens[s0:s1],
            q_starts=g_qs,
            qb_seq=torch.tensor(qb_seq, dtype=torch.int32, device=dev),
            qb_start=torch.tensor(qb_start, dtype=torch.int32, device=dev))
        groups.append((r0, r1, gm))
    return groups


def _init_weight(shape, device, std=0.02, seed=None):
    w = torch.empty(shape, dtype=torch.bfloat16, device=device)
    w.normal_(0.0, std)
    return nn.Parameter(w, requires_grad=False)


class LlamaAttention(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int, device):
        super().__init__()
        tp = parallel.tp_size()
        assert cfg.num_q_heads % tp == 0 and cfg.num_kv_heads % tp == 0, (
            "TP degree must divide q and kv head counts")
        self.layer_idx = layer_idx
        self.hq = cfg.num_q_heads // tp
        self.hk = cfg.num_kv_heads // tp
        self.d = cfg.head_dim
        self.scale = cfg.head_dim ** -0.5
        qkv_out = (self.hq + 2 * self.hk) * self.d
        self.qkv_w = _init_weight((qkv_out, cfg.hidden_size), device)
        self.o_w = _init_weight((cfg.hidden_size, self.hq * self.d), device)

    def forward(self, x: torch.Tensor, cos_sin: torch.Tensor,
                kc: torch.Tensor, vc: torch.Tensor, meta: AttnMeta):
        qkv = ops.linear(x, self.qkv_w)
        ops.rope_kv_append(qkv, kc, vc, cos_sin, meta.positions,
                           meta.slot_mapping, self.hq, self.hk, self.d,
                           block_table=(meta.block_table
                                        if meta.mode == "decode" else None))
        out = torch.empty(x.shape[0], self.hq * self.d, dtype=x.dtype,
                          device=x.device)
        if meta.mode == "decode":
            ops.paged_attention(out, qkv, kc, vc, meta.block_table,
                                meta.seq_lens, 0, meta.num_splits, self.scale,
                                meta.tmp_out, meta.tmp_ml)
        else:
            ops.prefill_attention(out, qkv, kc, vc, meta.block_table,
                                  meta.seq_lens, meta.q_starts, meta.qb_seq,
                                  meta.qb_start, 0, self.scale)
        o = ops.linear(out, self.o_w)
        return parallel.tp_all_reduce(o)

    def forward_pre_o(self, x: torch.Tensor, cos_sin: torch.Tensor,
                      kc: torch.Tensor, vc: torch.Tensor, meta: AttnMeta):
        """Attention WITHOUT the output projection: the layer fuses the
        o-proj with the post-attention add+RMSNorm (one epilogue kernel
        instead of reduce + norm)."""
        qkv = ops.linear(x, self.qkv_w)
        ops.rope_kv_append(qkv, kc, vc, cos_sin, meta.positions,
                           meta.slot_mapping, self.hq, self.hk, self.d,
                           block_table=(meta.block_table
                                        if meta.mode == "decode" else None))
        out = torch.empty(x.shape[0], self.hq * self.d, dtype=x.dtype,
                          device=x.device)
        if meta.mode == "decode":
            ops.paged_attention(out, qkv, kc, vc, meta.block_table,
                                meta.seq_lens, 0, meta.num_splits, self.scale,
                                meta.tmp_out, meta.tmp_ml)
        else:
            ops.prefill_attention(out, qkv, kc, vc, meta.block_table,
                                  meta.seq_lens, meta.q_starts, meta.qb_seq,
                                  meta.qb_start, 0, self.scale)
        return out


class LlamaMLP(nn.Module):
    def __init__(self, cfg: ModelConfig, device):
        super().__init__()
        tp = parallel.tp_size()
        assert cfg.intermediate_size % tp == 0
        self.inter = cfg.intermediate_size // tp
        self.gate_up_w = _init_weight((2 * self.inter, cfg.hidden_size), device)
        self.down_w = _init_weight((cfg.hidden_size, self.inter), device)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        gu = ops.linear(x, self.gate_up_w)
        act = torch.empty(x.shape[0], self.inter, dtype=x.dtype,
                          device=x.device)
        ops.silu_mul(act, gu)
        return parallel.tp_all_reduce(ops.linear(act, self.down_w))


class LlamaLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int, device):
        super().__init__()
        self.input_norm = nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=torch.bfloat16, device=device),
            requires_grad=False)
        self.post_norm = nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=torch.bfloat16, device=device),
            requires_grad=False)
        self.attn = LlamaAttention(cfg, layer_idx, device)
        self.mlp = LlamaMLP(cfg, device)
        self.eps = cfg.rms_eps

    def forward(self, x, residual, cos_sin, kc, vc, meta):
        if residual is None:
            residual = x
            h = torch.empty_like(x)
            ops.rmsnorm(h, x, self.input_norm, self.eps)
        else:
            ops.fused_add_rmsnorm(x, residual, self.input_norm, self.eps)
            h = x
        attn_out = self.attn.forward_pre_o(h, cos_sin, kc, vc, meta)
        h = ops.linear_add_rmsnorm(attn_out, self.attn.o_w, residual,
                                   self.post_norm, self.eps)
        h = self.mlp.forward(h)
        return h, residual

    def forward_fused(self, h, residual, cos_sin, kc, vc, meta,
                      next_norm_w, eps):
        """Layer body with the down-projection's split-K reduce fused
        into the NEXT layer's residual-add + input-RMSNorm (one kernel
        instead of reduce + norm — two ~4.5us in-graph launches per
        layer). Takes the already-input-normed h; returns the next
        layer's normed input and the updated residual."""
        attn_out = self.attn.forward_pre_o(h, cos_sin, kc, vc, meta)
        h2 = ops.linear_add_rmsnorm(attn_out, self.attn.o_w, residual,
                                    self.post_norm, self.eps)
        gu = ops.linear(h2, self.mlp.gate_up_w)
        h_next = ops.mlp_down_fused(gu, self.mlp.down_w, residual,
                                    next_norm_w, eps)
        return h_next, residual

    # -- pipelined TP prefill phases (comm/GEMM overlap): the o-proj and
    # down-proj partials are returned UN-reduced; the caller overlaps
    # their all-reduce (comm stream) with the other sequence group's
    # compute (parallel.tp_all_reduce_async) --
    def part_attn(self, x, residual, cos_sin, kc, vc, meta):
        if residual is None:
            residual = x
            h = torch.empty_like(x)
            ops.rmsnorm(h, x, self.input_norm, self.eps)
        else:
            ops.fused_add_rmsnorm(x, residual, self.input_norm, self.eps)
            h = x
        attn_out = self.attn.forward_pre_o(h, cos_sin, kc, vc, meta)
        return ops.linear(attn_out, self.attn.o_w), residual

    def part_mlp(self, o_reduced, residual):
        ops.fused_add_rmsnorm(o_reduced, residual, self.post_norm, self.eps)
        h = o_reduced
        gu = ops.linear(h, self.mlp.gate_up_w)
        act = torch.empty(h.shape[0], self.mlp.inter, dtype=h.dtype,
                          device=h.device)
        ops.silu_mul(act, gu)
        return ops.linear(act, self.mlp.down_w)


class LlamaModel(nn.Module):
    """Random-init Llama decoder for the serving engine (bench contract:
    synthetic data / random weights — no checkpoints in this environment)."""

    def __init__(self, cfg: ModelConfig, device="cuda"):
        super().__init__()
        self.cfg = cfg
        self.device_ = torch.device(device)
        torch.manual_seed(42)
        self.embed = _init_weight((cfg.vocab_size, cfg.hidden_size),
                                  self.device_)
        self.layers = nn.ModuleList(
            [LlamaLayer(cfg, i, self.device_) for i in range(cfg.num_layers)])
        self.final_norm = nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=torch.bfloat16,
                       device=self.device_), requires_grad=False)
        self.lm_head = _init_weight((cfg.vocab_size, cfg.hidden_size),
                                    self.device_)
        self.cos_sin = self._build_rope_table().to(self.device_)

    def _build_rope_table(self) -> torch.Tensor:
        cfg = self.cfg
        half = cfg.head_dim // 2
        inv = 1.0 / (cfg.rope_theta ** (
            torch.arange(0, cfg.head_dim, 2, dtype=torch.float64) /
            cfg.head_dim))
        t = torch.arange(cfg.max_position, dtype=torch.float64)
        fr = torch.outer(t, inv)
        return torch.cat([fr.cos(), fr.sin()], dim=1).float()

    def forward(self, input_ids: torch.Tensor, kv_k: torch.Tensor,
                kv_v: torch.Tensor, meta: AttnMeta) -> torch.Tensor:
        import os as _os
        if (parallel.tp_size() > 1 and meta.mode == "prefill"
                and meta.q_starts is not None
                and int(meta.q_starts.shape[0]) >= 2
                and _os.environ.get("KUKEON_TP_OVERLAP", "1") != "0"):
            return self._forward_prefill_overlap(input_ids, kv_k, kv_v,
                                                 meta)
        x = F.embedding(input_ids.long(), self.embed)
        # fused layer chain: each layer's down-reduce lands in the NEXT
        # layer's add+norm (the final layer folds into final_norm), so
        # the returned hidden is already final-normed
        residual = x
        h = torch.empty_like(x)
        ops.rmsnorm(h, x, self.layers[0].input_norm, self.cfg.rms_eps)
        n = len(self.layers)
        for i, layer in enumerate(self.layers):
            next_w = (self.layers[i + 1].input_norm if i + 1 < n
                      else self.final_norm)
            h, residual = layer.forward_fused(h, residual, self.cos_sin,
                                              kv_k[i], kv_v[i], meta,
                                              next_w, self.cfg.rms_eps)
        return h

    _overlap_runs = 0  # test hook: counts pipelined-path invocations

    def _forward_prefill_overlap(self, input_ids, kv_k, kv_v,
                                 meta: AttnMeta) -> torch.Tensor:
        """TP prefill with comm/GEMM overlap: sequences split into two
        groups whose layer phases interleave, so each group's o-proj and
        down-proj all-reduce (on the comm stream, RCCL over xGMI) flies
        under the OTHER group's attention / MLP GEMMs. Numerically
        identical to the plain path (group split is by whole sequences;
        reduction order within each all-reduce is unchanged) — the
        gloo world-2 equivalence test pins that."""
        LlamaModel._overlap_runs += 1
        groups = split_prefill_meta(meta)
        xs, res, ar = [], [], []
        for (r0, r1, gm) in groups:
            xs.append(F.embedding(input_ids[r0:r1].long(), self.embed))
            res.append(None)
            ar.append(None)
        os_ = [None] * len(groups)
        for i, layer in enumerate(self.layers):
            for g, (r0, r1, gm) in enumerate(groups):
                parallel.wait_comm(ar[g])  # down-proj AR of layer i-1
                os_[g], res[g] = layer.part_attn(xs[g], res[g],
                                                 self.cos_sin, kv_k[i],
                                                 kv_v[i], gm)
                ar[g] = parallel.tp_all_reduce_async(os_[g])
            for g, (r0, r1, gm) in enumerate(groups):
                parallel.wait_comm(ar[g])  # o-proj AR
                xs[g] = layer.part_mlp(os_[g], res[g])
                ar[g] = parallel.tp_all_reduce_async(xs[g])
        out = torch.empty(input_ids.shape[0], self.cfg.hidden_size,
                          dtype=xs[0].dtype, device=xs[0].device)
        for g, (r0, r1, gm) in enumerate(groups):
            parallel.wait_comm(ar[g])
            ops.fused_add_rmsnorm(xs[g], res[g], self.final_norm,
                                  self.cfg.rms_eps)
            out[r0:r1] = xs[g]
        return out

    def compute_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        return ops.linear(hidden, self.lm_head)
