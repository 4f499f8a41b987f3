"""Per-kernel microbenchmarks on MI355X — prints a table of achieved
bandwidth/TFLOPs vs the chip ceilings (HBM ~6.3 TB/s achievable, bf16 MFMA
~2.5 PF dense). Run via gpurun; results land in gpurun_out/ and the summary
is committed under profiles/.

Usage: python scripts/kernel_bench.py [--iters 50]
"""

from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from opsagent_amd import ops


def timeit(fn, iters=50, warmup=10):
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
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    dev = "cuda"
    rows = []

    # ---- RMSNorm (8B shape): memory-bound, bytes = 2*rows*dim*2 (r+w) + w
    for r, d in [(1024, 4096), (8192, 4096), (1024, 8192)]:
        x = torch.randn(r, d, dtype=torch.bfloat16, device=dev)
        w = torch.ones(d, dtype=torch.bfloat16, device=dev)
        t = timeit(lambda: ops.rms_norm(x, w), args.iters)
        gb = (2 * r * d * 2) / 1e9
        rows.append(("rmsnorm", f"{r}x{d}", f"{gb/t:.2f} TB/s", f"{t*1e6:.1f} us"))

    # fused add variant: 4 tensors touched (x, res r+w, out) = 4*r*d*2 bytes
    r, d = 8192, 4096
    x = torch.randn(r, d, dtype=torch.bfloat16, device=dev)
    res = torch.randn(r, d, dtype=torch.bfloat16, device=dev)
    w = torch.ones(d, dtype=torch.bfloat16, device=dev)
    t = timeit(lambda: ops.fused_add_rms_norm(x, res, w), args.iters)
    rows.append(("fused_add_rmsnorm", f"{r}x{d}", f"{4*r*d*2/1e9/t:.2f} TB/s", f"{t*1e6:.1f} us"))

    # ---- SiLU-mul: 3*n*2 bytes
    n = 8192 * 14336
    g = torch.randn(n, dtype=torch.bfloat16, device=dev)
    u = torch.randn(n, dtype=torch.bfloat16, device=dev)
    t = timeit(lambda: ops.silu_mul(g, u), args.iters)
    rows.append(("silu_mul", f"{n/1e6:.0f}M", f"{3*n*2/1e9/t:.2f} TB/s", f"{t*1e6:.1f} us"))

    # ---- RoPE (8B prefill shape)
    T, Hq, Hk, D = 8192, 32, 8, 128
    cos, sin = ops.rope_cos_sin(T, D, 500000.0)
    cos, sin = cos.to(dev), sin.to(dev)
    q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(T, Hk, D, dtype=torch.bfloat16, device=dev)
    pos = torch.arange(T, dtype=torch.int32, device=dev)
    t = timeit(lambda: ops.rope_apply_(q, k, cos, sin, pos), args.iters)
    byts = 2 * (T * Hq * D + T * Hk * D) * 2  # r+w bf16
    rows.append(("rope", f"T{T} H{Hq}/{Hk}", f"{byts/1e9/t:.2f} TB/s", f"{t*1e6:.1f} us"))

    # ---- Prefill attention (8B shape, causal): flops = 2*2*Hq*Sq*Skv/2*D
    for B, Hq_, Hk_, S in [(1, 32, 8, 2048), (1, 32, 8, 8192), (16, 32, 8, 1024)]:
        q = torch.randn(B, S, Hq_, 128, dtype=torch.bfloat16, device=dev) * 0.3
        kk = torch.randn(B, S, Hk_, 128, dtype=torch.bfloat16, device=dev) * 0.3
        v = torch.randn(B, S, Hk_, 128, dtype=torch.bfloat16, device=dev) * 0.3
        t = timeit(lambda: ops.attention_prefill(q, kk, v), max(5, args.iters // 5))
        fl = 4 * B * Hq_ * S * S / 2 * 128  # causal half
        rows.append(("attn_prefill", f"B{B} H{Hq_} S{S}", f"{fl/1e12/t:.1f} TF/s", f"{t*1e3:.2f} ms"))

    # ---- Decode attention: bytes ≈ B*len*2*Hk*D*2
    bs = 32
    for B, Hq_, Hk_, length in [(1, 32, 8, 2048), (1, 32, 8, 8192), (16, 32, 8, 2048), (64, 32, 8, 1024)]:
        maxb = (length + bs - 1) // bs
        nb = B * maxb + 1
        kc = torch.randn(nb, bs, Hk_, 128, dtype=torch.bfloat16, device=dev) * 0.3
        vc = torch.randn_like(kc)
        q = torch.randn(B, Hq_, 128, dtype=torch.bfloat16, device=dev) * 0.3
        bt = torch.arange(B * maxb, dtype=torch.int32, device=dev).reshape(B, maxb)
        lens = torch.full((B,), length, dtype=torch.int32, device=dev)
        ns = ops.decode_nsplit(B, Hk_, length)
        t = timeit(lambda: ops.attention_decode_paged(q, kc, vc, bt, lens, nsplit=ns), args.iters)
        byts = B * length * 2 * Hk_ * 128 * 2
        rows.append(
            ("attn_decode", f"B{B} len{length} ns{ns}", f"{byts/1e9/t:.2f} TB/s", f"{t*1e6:.1f} us")
        )

    # ---- Sampling: read B*V*2 bytes
    B, V = 16, 128256
    lg = torch.randn(B, V, dtype=torch.bfloat16, device=dev)
    mask = torch.full((B, (V + 31) // 32), -1, dtype=torch.int32, device=dev)
    t = timeit(lambda: ops.greedy_sample_masked(lg, mask), args.iters)
    rows.append(("masked_argmax", f"B{B} V{V}", f"{B*V*2/1e9/t:.2f} TB/s", f"{t*1e6:.1f} us"))

    # ---- custom HIP gemv vs hipBLASLt at decode shapes
    for m, k_, n_ in [(1, 4096, 6144), (1, 4096, 28672), (1, 14336, 4096),
                      (1, 4096, 128256), (4, 4096, 28672), (8, 4096, 6144)]:
        a = torch.randn(m, k_, dtype=torch.bfloat16, device=dev)
        w = torch.randn(n_, k_, dtype=torch.bfloat16, device=dev)
        t = timeit(lambda: ops.linear(a, w), args.iters)
        bw = (k_ * n_) * 2
        rows.append(("gemv(hip)", f"{m}x{k_}x{n_}", f"{bw/1e9/t:.2f} TB/s(W)", f"{t*1e6:.1f} us"))

    # ---- reference GEMMs (hipBLASLt via torch) at 8B decode/prefill shapes
    for m, k_, n_ in [(1, 4096, 6144), (1, 4096, 28672), (1024, 4096, 14336), (8192, 4096, 4096)]:
        a = torch.randn(m, k_, dtype=torch.bfloat16, device=dev)
        w = torch.randn(n_, k_, dtype=torch.bfloat16, device=dev)
        t = timeit(lambda: torch.nn.functional.linear(a, w), args.iters)
        fl = 2 * m * k_ * n_
        bw = (m * k_ + k_ * n_ + m * n_) * 2
        rows.append(
            ("linear(blaslt)", f"{m}x{k_}x{n_}", f"{fl/1e12/t:.1f} TF/s | {bw/1e9/t:.2f} TB/s", f"{t*1e6:.1f} us")
        )

    print(f"{'kernel':<20} {'shape':<20} {'rate':<28} {'time'}")
    for r in rows:
        print(f"{r[0]:<20} {r[1]:<20} {r[2]:<28} {r[3]}")


if __name__ == "__main__":
    main()
