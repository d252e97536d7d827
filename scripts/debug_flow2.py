import sys, torch
sys.path.insert(0, ".")
from kukeon_amd.engine.config import EngineConfig, MODEL_PRESETS, SamplingParams
from kukeon_amd.engine.engine import LLMEngine
from kukeon_amd.engine.kv_cache import SequenceKV
from kukeon_amd.models.llama import LlamaModel

cfg = MODEL_PRESETS["llama-3-8b"]()
ecfg = EngineConfig(max_model_len=4096, max_sessions=64, use_graphs=True)
model = LlamaModel(cfg, device="cuda:0")
engine = LLMEngine(model, cfg, ecfg, device="cuda:0")
sp = SamplingParams(temperature=0.7, top_k=50, top_p=0.9, max_new_tokens=8)
kvs = [SequenceKV(16) for _ in range(64)]
for kv in kvs:
    engine.add_request(kv, list(range(100, 180)), sp)
steps = 0
while engine.has_work() and steps < 6:
    outs = engine.step()
    torch.cuda.synchronize()
    toks = [t for o in outs for t in o.new_tokens]
    bad = [t for t in toks if t < 0 or t >= cfg.vocab_size]
    ws = engine.d_ws[:64].cpu()
    print(f"step {steps}: ntoks={len(toks)} min={min(toks)} max={max(toks)} bad={bad[:5]}",
          flush=True)
    print("  ws[0][:10]:", [round(float(x),3) for x in ws[0][:10]], flush=True)
    print("  seq_lens[:4]:", engine.d_seq_lens[:4].cpu().tolist(),
          "slots[:4]:", engine.d_slots[:4].cpu().tolist(),
          "ids[:4]:", engine.d_ids[:4].cpu().tolist(), flush=True)
    steps += 1
print("done", flush=True)
