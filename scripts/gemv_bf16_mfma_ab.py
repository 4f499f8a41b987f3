"""bf16 batched-decode GEMV A/B: MFMA kernel vs hipBLASLt (the old M>2
fallback) vs the VALU kernel, at concurrent-decode shapes. Correctness vs
fp32 torch."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from opsagent_amd import ops


def t(fn, it=30, wu=3):
    for _ in range(wu): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(it): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / it


shapes = [
    ("8B qkv", 4, 6144, 4096, "plain"), ("8B qkv", 8, 6144, 4096, "plain"),
    ("8B qkv", 16, 6144, 4096, "plain"),
    ("8B o", 8, 4096, 4096, "plain"),
    ("8B down", 8, 4096, 14336, "plain"),
    ("8B gateup", 8, 14336, 4096, "gateup"),
    ("8B gateup", 16, 14336, 4096, "gateup"),
    ("70B qkv", 8, 10240, 8192, "plain"),
]
print(f"{'shape':<14} {'M':<3} {'path':<6} {'maxrel':<8} {'TB/s':<6} time")
for tag, M, N, K, kind in shapes:
    torch.manual_seed(M + N)
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.5
    rows = 2 * N if kind == "gateup" else N
    w = torch.randn(rows, K, dtype=torch.bfloat16, device="cuda") * 0.3
    if kind == "gateup":
        gu = x.float().cpu() @ w.float().cpu().T
        g, u = gu[:, :N], gu[:, N:]
        ref = g * torch.sigmoid(g) * u
        run = lambda: ops.gateup_silu(x, w, N)
    else:
        ref = x.float().cpu() @ w.float().cpu().T
        run = lambda: ops.linear(x, w)
    by = rows * K * 2
    for path, env in (("blaslt", "0"), ("mfma", "m2")):
        os.environ["OPSAGENT_BF16_GEMV_MFMA"] = env
        out = run()
        err = ((out.float().cpu() - ref).abs().max() /
               ref.abs().mean().clamp_min(1e-3)).item()
        tt = t(run)
        ok = "" if err < 0.3 else " <-- FAIL"
        print(f"{tag:<14} {M:<3} {path:<6} {err:<8.4f} {by/1e12/tt:<6.2f} {tt*1e3:.3f} ms{ok}")
os.environ.pop("OPSAGENT_BF16_GEMV_MFMA", None)
