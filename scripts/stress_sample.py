import torch
import kukeon_amd.ops as ops
DEV = "cuda:0"
torch.manual_seed(0)
B, V = 64, 128256
logits = torch.randn(B, V, dtype=torch.bfloat16, device=DEV) * 3
temps = torch.full((B,), 0.7, device=DEV)
tk = torch.full((B,), 50, dtype=torch.int32, device=DEV)
tp = torch.full((B,), 0.9, device=DEV)
seed = torch.zeros(1, dtype=torch.int64, device=DEV)
ws = torch.zeros(B, 528, dtype=torch.float32, device=DEV)
tokens = torch.zeros(B, dtype=torch.int32, device=DEV)
for i in range(100):
    ops.sample(tokens, logits, temps, tk, tp, seed, ws)
torch.cuda.synchronize()
print("eager ok", tokens[:5].tolist())
# in-graph
g = torch.cuda.CUDAGraph()
ops.sample(tokens, logits, temps, tk, tp, seed, ws)
torch.cuda.synchronize()
with torch.cuda.graph(g):
    ops.sample(tokens, logits, temps, tk, tp, seed, ws)
for i in range(100):
    g.replay()
torch.cuda.synchronize()
print("graph ok", tokens[:5].tolist())
# mixed batch sizes
for b in (1, 3, 8, 17, 64):
    ops.sample(tokens[:b], logits[:b], temps[:b], tk[:b], tp[:b], seed, ws[:b])
torch.cuda.synchronize()
print("sizes ok")
