"""Mixtral MoE path on CPU reference ops: routing/permute/combine vs a
naive dense loop over experts."""
import torch
import torch.nn.functional as F

from kukeon_amd.engine.config import (EngineConfig, SamplingParams,
                                      tiny_mixtral)
from kukeon_amd.engine.engine import LLMEngine
from kukeon_amd.engine.kv_cache import SequenceKV
from kukeon_amd.models.mixtral import MixtralMoE, MixtralModel


def test_moe_matches_naive():
    torch.manual_seed(0)
    cfg = tiny_mixtral()
    moe = MixtralMoE(cfg, "cpu")
    x = torch.randn(9, cfg.hidden_size, dtype=torch.bfloat16)
    out = moe.forward(x.clone())

    # naive: every token through its top-k experts, weighted sum
    logits = F.linear(x, moe.router_w).float()
    probs = torch.softmax(logits, -1)
    topv, topi = probs.topk(cfg.top_k_experts, -1)
    topv = topv / topv.sum(-1, keepdim=True)
    ref = torch.zeros(x.shape, dtype=torch.float32)
    for t in range(x.shape[0]):
        for k in range(cfg.top_k_experts):
            e = int(topi[t, k])
            gu = F.linear(x[t:t + 1], moe.gate_up_w[e]).float()
            g, u = gu[:, :moe.inter], gu[:, moe.inter:]
            act = (F.silu(g) * u).to(torch.bfloat16)
            y = F.linear(act, moe.down_w[e]).float()
            ref[t] += float(topv[t, k]) * y[0]
    torch.testing.assert_close(out.float(), ref.to(torch.bfloat16).float(),
                               rtol=3e-2, atol=3e-2)


def test_mixtral_engine_generates():
    torch.manual_seed(0)
    cfg = tiny_mixtral()
    ecfg = EngineConfig(max_model_len=256, max_sessions=4, num_kv_blocks=128,
                        use_graphs=False)
    model = MixtralModel(cfg, device="cpu")
    engine = LLMEngine(model, cfg, ecfg, device="cpu")
    kv = SequenceKV(ecfg.block_size)
    engine.add_request(kv, [1, 2, 3, 4, 5],
                       SamplingParams(temperature=0.0, max_new_tokens=4))
    toks = []
    while engine.has_work():
        for o in engine.step():
            toks.extend(o.new_tokens)
    assert len(toks) == 4
    assert all(0 <= t < cfg.vocab_size for t in toks)


def test_dense_and_sparse_moe_paths_agree():
    torch.manual_seed(1)
    cfg = tiny_mixtral()
    moe = __import__("kukeon_amd.models.mixtral",
                     fromlist=["MixtralMoE"]).MixtralMoE(cfg, "cpu")
    x = torch.randn(11, cfg.hidden_size, dtype=torch.bfloat16)
    sparse = moe._forward_sparse(x.clone())
    dense = moe._forward_dense(x.clone())
    torch.testing.assert_close(dense.float(), sparse.float(), rtol=3e-2,
                               atol=3e-2)
