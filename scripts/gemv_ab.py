"""Within-probe interleaved A/B of gemv variants (guide rule 24): RW=2 vs
RW=4 per wave, on REAL-SIZE weights that do NOT fit L3 (many distinct W
tensors cycled so every pass streams from HBM like a decode step does)."""

import ctypes
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from opsagent_amd.ops import hip_lib

lib = hip_lib.get_lib()
lib.oa_gemv_rw4.argtypes = [ctypes.c_void_p] * 4 + [ctypes.c_int] * 3
lib.oa_gemv_rw4.restype = ctypes.c_int
dev = "cuda"

SHAPES = [(1, 4096, 6144), (1, 4096, 4096), (1, 4096, 28672), (1, 14336, 4096)]
NW = 12  # distinct weight copies per shape -> > L3, forces HBM streaming
ROUNDS = 10


def run(variant, x, ws, out, M, N, K):
    fn = lib.oa_gemv if variant == 2 else lib.oa_gemv_rw4
    for w in ws:
        rc = fn(hip_lib.current_stream_ptr(), x.data_ptr(), w.data_ptr(),
                out.data_ptr(), M, N, K)
        assert rc == 0


for M, K, N in SHAPES:
    torch.manual_seed(0)
    x = (torch.randn(M, K, device=dev) * 0.3).to(torch.bfloat16)
    ws = [(torch.randn(N, K, device=dev) * 0.3).to(torch.bfloat16) for _ in range(NW)]
    out = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
    # correctness check once
    ref = torch.nn.functional.linear(x.float(), ws[0].float())
    for variant in (2, 4):
        run(variant, x, ws[:1], out, M, N, K)
        torch.cuda.synchronize()
        err = (out.float() - ref).abs().max() / ref.abs().max()
        assert err < 0.03, f"variant {variant} wrong: {err}"
    # warmup
    run(2, x, ws, out, M, N, K)
    run(4, x, ws, out, M, N, K)
    torch.cuda.synchronize()
    times = {2: [], 4: []}
    for _ in range(ROUNDS):
        for variant in (2, 4):
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            run(variant, x, ws, out, M, N, K)
            torch.cuda.synchronize()
            times[variant].append((time.perf_counter() - t0) / NW)
    gb = N * K * 2 / 1e9
    r2 = statistics.median(times[2])
    r4 = statistics.median(times[4])
    print(f"M{M} K{K} N{N}: RW2 {gb/r2:6.2f} TB/s ({r2*1e6:6.1f}us)  "
          f"RW4 {gb/r4:6.2f} TB/s ({r4*1e6:6.1f}us)  ratio {r2/r4:.3f}", flush=True)
