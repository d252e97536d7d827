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
while engine.has_work():
    print("step", steps, "waiting", len(engine.waiting), "running", engine.num_running, flush=True)
    outs = engine.step()
    torch.cuda.synchronize()
    print("  ok", flush=True)
    steps += 1
print("flow ok", steps, flush=True)
