"""The in-tree HIP library's extern-C surface, checked on CPU: ctypes-load
the built .so and assert every entry point the Python dispatch layer (and
the GPU tests) rely on is exported. Catches symbol renames/removals before
a GPU box ever sees them (the driver's build gate compiles but does not
link-probe)."""

import ctypes
import os

import pytest

LIB = os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    "opsagent_amd", "ops", "libopsagent_kernels.so",
)

ENTRY_POINTS = [
    "oa_attention_decode", "oa_attention_decode_rope",
    "oa_attention_prefill", "oa_attention_prefill_variant",
    "oa_fp8_gemv_scratch_init", "oa_fused_add_rmsnorm", "oa_gemm_fp8",
    "oa_gemv", "oa_gemv_bf16_mfma", "oa_gemv_ex", "oa_gemv_fp8",
    "oa_gemv_fp8_ex", "oa_gemv_fp8_mfma", "oa_gemv_gateup",
    "oa_gemv_gateup_ex", "oa_gemv_gateup_fp8", "oa_gemv_rw4", "oa_kv_write",
    "oa_masked_argmax", "oa_moe_down", "oa_moe_down_emaj", "oa_moe_gateup",
    "oa_moe_gateup_emaj", "oa_quant_fp8", "oa_rmsnorm", "oa_rope",
    "oa_rope_kv", "oa_silu_mul",
]


@pytest.mark.skipif(not os.path.exists(LIB), reason="kernel .so not built")
def test_kernel_library_exports():
    lib = ctypes.CDLL(LIB)
    missing = [s for s in ENTRY_POINTS if not hasattr(lib, s)]
    assert not missing, f"missing entry points: {missing}"


GRAMMAR_LIB = LIB.replace("libopsagent_kernels", "libopsagent_grammar")


@pytest.mark.skipif(not os.path.exists(GRAMMAR_LIB), reason="grammar .so not built")
def test_grammar_library_exports():
    lib = ctypes.CDLL(GRAMMAR_LIB)
    for s in ["oa_grammar_create", "oa_grammar_destroy",
              "oa_grammar_accept_token", "oa_grammar_fill_mask",
              "oa_grammar_forced_bytes", "oa_grammar_forced_run",
              "oa_grammar_check_tokens", "oa_grammar_masks_along",
              "oa_grammar_completion", "oa_grammar_is_complete",
              "oa_grammar_reset", "oa_vocab_create", "oa_vocab_destroy"]:
        assert hasattr(lib, s), f"missing {s}"
