"""Prefill-attention variant bench: correctness vs torch fp32 + TF/s per
variant (OPSAGENT_PREFILL_VARIANT 1..4) at agent-relevant shapes.

Run under gpurun; summary goes to stdout (commit to profiles/README.md).
"""

from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from opsagent_amd import ops


def ref_attention(q, k, v, scale):
    """Causal fp32 reference, [B,S,H,D] layout, GQA."""
    B, Sq, Hq, D = q.shape
    Skv, Hk = k.shape[1], k.shape[2]
    g = Hq // Hk
    qf = q.float().permute(0, 2, 1, 3)                     # B,Hq,Sq,D
    kf = k.float().permute(0, 2, 1, 3)                     # B,Hk,Skv,D
    vf = v.float().permute(0, 2, 1, 3)
    kf = kf.repeat_interleave(g, dim=1)
    vf = vf.repeat_interleave(g, dim=1)
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    off = Skv - Sq
    qpos = torch.arange(Sq, device=q.device)[:, None] + off
    kpos = torch.arange(Skv, device=q.device)[None, :]
    s = s.masked_fill(kpos > qpos, float("-inf"))
    o = torch.matmul(torch.softmax(s, dim=-1), vf)
    return o.permute(0, 2, 1, 3).to(q.dtype)


def timeit(fn, iters, warmup=5):
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
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--variants", default="1,2,3,4")
    ap.add_argument("--shapes", default="", help="comma list of shape indices")
    args = ap.parse_args()
    assert torch.cuda.is_available()
    dev = "cuda"
    torch.manual_seed(5)
    variants = [int(x) for x in args.variants.split(",")]

    shapes = [
        # (B, Hq, Hk, Sq, Skv) — bench chunk, long-context, odd tail, chunked
        (1, 32, 8, 1024, 1024),
        (1, 32, 8, 2048, 2048),
        (1, 32, 8, 8192, 8192),
        (16, 32, 8, 1024, 1024),
        (1, 32, 8, 736, 1760),   # chunked prefill w/ past + ragged tail
        (1, 8, 1, 512, 512),     # 70B tp8 head geometry
        (1, 64, 8, 4096, 4096),  # 70B tp1 head geometry (GQ8 merge)
    ]
    if args.shapes:
        idx = [int(i) for i in args.shapes.split(",")]
        shapes = [shapes[i] for i in idx]
    print(f"{'shape':<26} {'var':<4} {'maxerr':<10} {'TF/s':<8} time")
    for (B, Hq, Hk, Sq, Skv) in shapes:
        q = (torch.randn(B, Sq, Hq, 128, dtype=torch.bfloat16, device=dev) * 0.5)
        k = (torch.randn(B, Skv, Hk, 128, dtype=torch.bfloat16, device=dev) * 0.5)
        v = (torch.randn(B, Skv, Hk, 128, dtype=torch.bfloat16, device=dev) * 0.5)
        scale = 128 ** -0.5
        ref = ref_attention(q, k, v, scale)
        off = Skv - Sq
        # causal flops: sum over rows of (off + i + 1) keys
        keys_total = Sq * off + Sq * (Sq + 1) // 2
        fl = 4 * B * Hq * keys_total * 128
        for var in variants:
            os.environ["OPSAGENT_PREFILL_VARIANT"] = str(var)
            out = ops.attention_prefill(q, k, v, scale=scale)
            err = (out.float() - ref.float()).abs().max().item()
            t = timeit(lambda: ops.attention_prefill(q, k, v, scale=scale),
                       args.iters)
            tag = f"B{B} H{Hq}/{Hk} S{Sq}/{Skv}"
            ok = "" if err < 0.1 else "  <-- FAIL"
            print(f"{tag:<26} {var:<4} {err:<10.4f} {fl/1e12/t:<8.1f} "
                  f"{t*1e3:.3f} ms{ok}")
    os.environ.pop("OPSAGENT_PREFILL_VARIANT", None)


if __name__ == "__main__":
    main()
