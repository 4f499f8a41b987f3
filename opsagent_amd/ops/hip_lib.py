"""ctypes loader for the in-tree HIP kernel library.

The kernels are plain hipcc-built (no torch headers, no hipify — native HIP
per the MI355X-first design). Launch functions take raw device pointers plus
the current torch HIP stream, so they compose with torch allocations and
hipGraph capture (torch.cuda.graphs).

Policy (per driver contract): on a GPU box the HIP library is REQUIRED — ops
fail loudly if it is missing rather than silently falling back to eager
PyTorch. CPU runs use opsagent_amd.ops.torch_ref.
"""

from __future__ import annotations

import ctypes
import os
from typing import Optional

import torch

_LIB: Optional[ctypes.CDLL] = None
_LIB_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)), "libopsagent_kernels.so")


class HipKernelsMissing(RuntimeError):
    pass


def _declare(lib: ctypes.CDLL) -> None:
    p = ctypes.c_void_p
    i = ctypes.c_int
    i64 = ctypes.c_int64
    f = ctypes.c_float
    lib.oa_rmsnorm.argtypes = [p, p, p, p, i, i, f]
    lib.oa_rmsnorm.restype = i
    lib.oa_fused_add_rmsnorm.argtypes = [p, p, p, p, p, i, i, f]
    lib.oa_fused_add_rmsnorm.restype = i
    lib.oa_rope.argtypes = [p, p, p, p, p, p, i, i, i, i]
    lib.oa_rope.restype = i
    lib.oa_silu_mul.argtypes = [p, p, p, p, i64]
    lib.oa_silu_mul.restype = i
    lib.oa_kv_write.argtypes = [p, p, p, p, p, p, i, i, i]
    lib.oa_kv_write.restype = i
    lib.oa_rope_kv.argtypes = [p, p, p, p, p, p, p, p, p, p, i, i, i, i, i, i, i]
    lib.oa_rope_kv.restype = i
    lib.oa_gemv_gateup.argtypes = [p, p, p, p, i, i, i]
    lib.oa_gemv_gateup.restype = i
    lib.oa_gemv.argtypes = [p, p, p, p, i, i, i]
    lib.oa_gemv.restype = i
    lib.oa_gemv_ex.argtypes = [p, p, p, p, p, p, i, i, i, f, i]
    lib.oa_gemv_ex.restype = i
    lib.oa_gemv_gateup_ex.argtypes = [p, p, p, p, p, i, i, i, f, i]
    lib.oa_gemv_gateup_ex.restype = i
    lib.oa_masked_argmax.argtypes = [p, p, p, p, p, i, i]
    lib.oa_masked_argmax.restype = i
    lib.oa_attention_prefill.argtypes = [p, p, p, p, p, i, i, i, i, i, i, f, i, i, i]
    lib.oa_attention_prefill.restype = i
    lib.oa_attention_decode.argtypes = [p, p, p, p, p, p, p, p, p, p, i, i, i, i, i, i, i, f, i, i]
    lib.oa_attention_decode.restype = i
    lib.oa_attention_decode_rope.argtypes = [
        p, p, p, p, p, p, p, p, p, p, p, p, p, p,
        i, i, i, i, i, i, i, f, i, i, i,
    ]
    lib.oa_attention_decode_rope.restype = i
    lib.oa_quant_fp8.argtypes = [p, p, p, p, i, i]
    lib.oa_quant_fp8.restype = i
    lib.oa_gemv_fp8.argtypes = [p, p, p, p, p, i, i, i]
    lib.oa_gemv_fp8.restype = i
    lib.oa_gemv_fp8_ex.argtypes = [p, p, p, p, p, p, p, i, i, i, f, i]
    lib.oa_gemv_fp8_ex.restype = i
    lib.oa_gemv_gateup_fp8.argtypes = [p, p, p, p, p, p, i, i, i, f, i]
    lib.oa_gemv_gateup_fp8.restype = i
    lib.oa_gemm_fp8.argtypes = [p, p, p, p, p, p, i, i, i]
    lib.oa_gemm_fp8.restype = i
    lib.oa_moe_gateup.argtypes = [p, p, p, p, p, p, p, i, i, i, i]
    lib.oa_moe_gateup.restype = i
    lib.oa_moe_down.argtypes = [p, p, p, p, p, p, p, i, i, i, i]
    lib.oa_moe_down.restype = i
    lib.oa_moe_gateup_emaj.argtypes = [p, p, p, p, p, p, p, i, i, i, i, i]
    lib.oa_moe_gateup_emaj.restype = i
    lib.oa_moe_down_emaj.argtypes = [p, p, p, p, p, p, p, i, i, i, i, i]
    lib.oa_moe_down_emaj.restype = i


def get_lib() -> ctypes.CDLL:
    global _LIB
    if _LIB is None:
        if not os.path.exists(_LIB_PATH):
            # try an in-tree build (hipcc cross-compiles without a GPU)
            try:
                from opsagent_amd.ops.build import build

                build()
            except Exception as e:
                raise HipKernelsMissing(
                    f"HIP kernel library missing and build failed: {e}. "
                    "Run `python -m opsagent_amd.ops.build`."
                ) from e
        if not os.path.exists(_LIB_PATH):
            raise HipKernelsMissing(f"HIP kernel library not found at {_LIB_PATH}")
        _LIB = ctypes.CDLL(_LIB_PATH)
        _declare(_LIB)
    return _LIB


def available() -> bool:
    try:
        get_lib()
        return True
    except HipKernelsMissing:
        return False


def current_stream_ptr() -> int:
    return torch.cuda.current_stream().cuda_stream


def check(rc: int, name: str) -> None:
    if rc != 0:
        raise RuntimeError(f"HIP kernel {name} failed with code {rc}")
