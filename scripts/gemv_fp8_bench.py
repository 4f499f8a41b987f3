"""fp8 GEMV path A/B: VALU dequant stream vs MFMA (OPSAGENT_FP8_GEMV_MFMA)
at the 70B/8B decode shapes. Correctness vs a double-quantized torch ref
(fp8 W AND fp8 x — the MFMA path's numerics), plus streamed TB/s.

Run under gpurun; commit the summary to profiles/README.md.
"""

from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from opsagent_amd import ops


def quant_ref(t: torch.Tensor):
    amax = t.float().abs().amax(dim=-1, keepdim=True).clamp_min(1e-8)
    sc = amax / 448.0
    q = (t.float() / sc).clamp(-448, 448)
    q8 = q.to(torch.float8_e4m3fn).float() * sc
    return q8


def timeit(fn, iters, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=30)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    dev = "cuda"
    torch.manual_seed(7)

    # (tag, M, N, K, kind)
    shapes = [
        ("70B qkv", 1, 10240, 8192, "plain"),
        ("70B o", 1, 8192, 8192, "plain"),
        ("70B gateup", 1, 28672, 8192, "gateup"),
        ("70B down", 1, 8192, 28672, "plain"),
        ("8B qkv", 1, 6144, 4096, "plain"),
        ("8B gateup", 1, 14336, 4096, "gateup"),
        ("70B qkv M4", 4, 10240, 8192, "plain"),
        ("70B gateup M4", 4, 28672, 8192, "gateup"),
    ]
    print(f"{'shape':<18} {'path':<5} {'maxrel':<9} {'TB/s':<7} time")
    for tag, M, N, K, kind in shapes:
        x = torch.randn(M, K, dtype=torch.bfloat16, device=dev) * 0.5
        rows = 2 * N if kind == "gateup" else N
        w = torch.randn(rows, K, dtype=torch.bfloat16, device=dev) * 0.3
        w8, wsc = ops.quant_fp8(w)
        wd = quant_ref(w)
        xd = quant_ref(x)
        if kind == "gateup":
            gu = xd @ wd.t()
            g, u = gu[:, :N], gu[:, N:]
            ref = (g * torch.sigmoid(g) * u).to(torch.bfloat16)
            run = lambda: ops.gateup_silu_fp8(x, w8, wsc, N)
        else:
            ref = (xd @ wd.t()).to(torch.bfloat16)
            run = lambda: ops.linear_fp8(x, w8, wsc)
        bytes_w = rows * K  # fp8 weight stream
        for path, env in (("valu", "0"), ("mfma", "1")):
            os.environ["OPSAGENT_FP8_GEMV_MFMA"] = env
            out = run()
            scale = ref.float().abs().mean().clamp_min(1e-3)
            err = ((out.float() - ref.float()).abs().max() / scale).item()
            t = timeit(run, args.iters)
            ok = "" if err < 0.35 else "  <-- FAIL"
            print(f"{tag:<18} {path:<5} {err:<9.4f} {bytes_w/1e12/t:<7.2f} "
                  f"{t*1e3:.3f} ms{ok}")
    os.environ.pop("OPSAGENT_FP8_GEMV_MFMA", None)


if __name__ == "__main__":
    main()
