import math, sys, torch
sys.path.insert(0, ".")
from sutro_amd import ops
from sutro_amd.ops import torch_ref as R

def bf(t): return t.bfloat16().cuda()
torch.manual_seed(3)
D, bs = 64, 32
kc = bf(torch.randn(2, 1, bs, D)); vc = bf(torch.randn(2, 1, bs, D))
bt = torch.tensor([[1]], dtype=torch.int32)
q = bf(torch.randn(1, 1, D))
sl = torch.tensor([1], dtype=torch.int32); ql = torch.tensor([0,1], dtype=torch.int32)
got = ops.paged_attention(q, kc, vc, bt.cuda(), sl.cuda(), ql.cuda(),
                          1.0/math.sqrt(D), num_decodes_tail=1, prefill_token_count=0)
g = got[0,0].float().cpu()
V = vc[1,0].float().cpu()   # [32 pos, 64 d]
for d in [0, 16, 17, 30, 40, 63]:
    tgt = g[d].item()
    hits = (V - tgt).abs() < 1e-3
    idx = hits.nonzero().tolist()
    print(f"out[{d}]={tgt:.4f} V[0][{d}]={V[0,d].item():.4f} matches V at {idx[:4]}")
