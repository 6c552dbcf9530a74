import math, sys, torch
sys.path.insert(0, ".")
from sutro_amd import _C

torch.manual_seed(3)
D = 64
K = (torch.randn(32, D)).bfloat16().cuda()
V = (torch.randn(32, D)).bfloat16().cuda()
q = (torch.randn(1, D)).bfloat16().cuda()
vt, p, out = _C.hd64_stage_probe(q, K, V, 1, 1.0/math.sqrt(D))
vt = vt.cpu().view(64, 40); p = p.cpu().view(16, 40)
Vc = V.float().cpu()
# vt[d][pos] should equal V[pos][d]
bad = []
for d in range(64):
    for pos in range(32):
        if abs(vt[d, pos].item() - Vc[pos, d].item()) > 1e-6:
            bad.append((d, pos))
print("vt bad cells:", len(bad), bad[:10])
print("p[0][:4] (want [1,0,0,0]):", p[0, :4].tolist())
print("p rows 1-3 max:", p[1:4, :32].max().item())
g = out[0].float().cpu()
d0 = (g - Vc[0]).abs()
print("out vs V[0]: max diff", d0.max().item(),
      "bad dims", (d0 > 3e-2).nonzero().flatten().tolist()[:12])
