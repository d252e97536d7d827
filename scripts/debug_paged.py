import torch
import kukeon_amd.ops as ops
from kukeon_amd.ops import reference

DEV = "cuda:0"
torch.manual_seed(0)
D, BS = 128, 16

def case(ctx, Hq, Hk, tag):
    NB = (ctx + BS - 1) // BS + 1
    kc = torch.randn(NB, Hk, BS, D, dtype=torch.bfloat16, device=DEV)
    vc = torch.randn(NB, Hk, BS, D, dtype=torch.bfloat16, device=DEV)
    bt = torch.arange(NB, dtype=torch.int32, device=DEV).unsqueeze(0)
    sl = torch.tensor([ctx], dtype=torch.int32, device=DEV)
    q = torch.randn(1, Hq * D, dtype=torch.bfloat16, device=DEV)
    out = torch.empty(1, Hq * D, dtype=torch.bfloat16, device=DEV)
    t1 = torch.zeros(1, dtype=torch.float32, device=DEV)
    ops.paged_attention(out, q, kc, vc, bt, sl, 0, 1, 1.0, t1, t1)
    ref = torch.empty(1, Hq * D, dtype=torch.bfloat16)
    reference.paged_attention(ref, q.cpu(), kc.cpu(), vc.cpu(), bt.cpu(), sl.cpu(), 0, 1, 1.0)
    o = out.cpu().float().view(Hq, D)
    r = ref.float().view(Hq, D)
    print(f"== {tag} ctx={ctx} Hq={Hq} Hk={Hk}")
    for h in range(Hq):
        d = (o[h] - r[h]).abs().max().item()
        print(f"  head {h}: maxdiff {d:.4f} out[0:4]={o[h,:4].tolist()} ref={r[h,:4].tolist()}")
    if ctx == 1:
        v0 = vc[0, 0, 0].cpu().float()
        print("  v[0][0:4] =", v0[:4].tolist())

case(1, 1, 1, "single")
case(1, 4, 1, "gqa4-ctx1")
case(16, 1, 1, "1blk")
case(17, 1, 1, "partial2")
case(40, 4, 1, "gqa4")
