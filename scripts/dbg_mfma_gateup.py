import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from opsagent_amd import ops

def dquant(t):
    amax = t.float().abs().amax(dim=-1, keepdim=True).clamp_min(1e-8)
    sc = amax / 448.0
    return ((t.float() / sc).clamp(-448, 448).to(torch.float8_e4m3fn).float() * sc)

M, K, I = 8, 4096, 992
torch.manual_seed(M + 77)
x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.3
w = torch.randn(2 * I, K, dtype=torch.bfloat16, device="cuda") * 0.3
nw = torch.randn(K, dtype=torch.bfloat16, device="cuda") * 0.1 + 1.0
q, s = ops.quant_fp8(w)
got = ops.gateup_silu_fp8(x, q, s, I, norm_w=nw, eps=1e-5).float().cpu()
xf = x.float().cpu()
xf = xf * torch.rsqrt((xf * xf).mean(-1, keepdim=True) + 1e-5) * nw.float().cpu()
gu = dquant(xf) @ dquant(w.cpu()).T
g, u = gu[:, :I], gu[:, I:]
ref = g * torch.sigmoid(g) * u
err = (got - ref).abs() / ref.abs().clamp_min(1.0)
print("max err", err.max().item(), "n>0.08:", (err > 0.08).sum().item(), "of", err.numel())
idx = err.argmax().item()
r, c = idx // I, idx % I
print("worst at", r, c, "got", got[r, c].item(), "ref", ref[r, c].item(),
      "g", g[r, c].item(), "u", u[r, c].item())
# per-row max err
print("per-row max:", [round(err[m].max().item(), 3) for m in range(M)])
