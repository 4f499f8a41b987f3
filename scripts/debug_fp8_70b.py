"""Probe: fp8 vs bf16 GEMV at 70B shapes + quantize check on the engine."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from opsagent_amd import ops


def t(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


for (N, K) in [(10240, 8192), (8192, 8192), (57344, 8192), (8192, 28672)]:
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.1
    q, s = ops.quant_fp8(w)
    x = torch.randn(1, K, dtype=torch.bfloat16, device="cuda")
    us_bf = t(lambda: ops.linear(x, w))
    us_f8 = t(lambda: ops.linear_fp8(x, q, s))
    bw_bf = N * K * 2 / us_bf / 1e3
    bw_f8 = N * K * 1 / us_f8 / 1e3
    print(f"N={N:6d} K={K:6d}  bf16 {us_bf:7.1f}us ({bw_bf:4.1f} TB/s)  "
          f"fp8 {us_f8:7.1f}us ({bw_f8:4.1f} TB/s)")
    del w, q
    torch.cuda.empty_cache()

# engine-level check on the micro model: does quantize halve step time?
from opsagent_amd.engine.engine import LLMEngine, SamplingParams

for qz in (None, "fp8"):
    cfg = {"model": "llama3-micro", "max_seq_len": 1024, "kv_block_size": 32,
           "kv_cache_gb": 2, "use_hipgraph": True, "seed": 5}
    if qz:
        cfg["quantize"] = qz
    eng = LLMEngine(cfg)
    at = eng.model.layers[0].attn
    print(f"quantize={qz}: attn.fp8={at.fp8}")
    del eng
    torch.cuda.empty_cache()
