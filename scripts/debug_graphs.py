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
print("capturing buckets one by one", flush=True)
engine.d_slots.fill_(-1); engine.d_seq_lens.zero_(); engine.d_ids.zero_(); engine.d_pos.zero_()
for b in sorted(ecfg.graph_buckets, reverse=True):
    print("bucket", b, flush=True)
    engine._capture(b)
    torch.cuda.synchronize()
    print("  captured + synced", flush=True)
    engine.graphs[b].replay()
    torch.cuda.synchronize()
    print("  replayed", flush=True)
print("all buckets ok", flush=True)
# now a real session flow
sp = SamplingParams(temperature=0.7, top_k=50, top_p=0.9, max_new_tokens=8)
kvs = [SequenceKV(16) for _ in range(64)]
for kv in kvs:
    engine.add_request(kv, list(range(100, 180)), sp)
steps = 0
while engine.has_work():
    outs = engine.step()
    steps += 1
torch.cuda.synchronize()
print("session flow ok, steps", steps, flush=True)
