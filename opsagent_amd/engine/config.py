"""Model architecture registry.

Llama-3 family dims (8B / 70B per BASELINE.json configs #2-#4), a tiny config
for CPU tests, and DeepSeek-V3-style MoE configs (BASELINE config #5). Random
initialization is the benchmark mode (no network for checkpoints); safetensors
loading is supported for real weights.
"""

from __future__ import annotations

import dataclasses
from typing import Optional


@dataclasses.dataclass
class ModelSpec:
    name: str
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    max_seq_len: int = 8192
    tie_embeddings: bool = False
    # MoE (None => dense MLP)
    moe_num_experts: Optional[int] = None
    moe_top_k: int = 2
    moe_intermediate_size: Optional[int] = None
    moe_shared_experts: int = 0
    moe_dtype: str = "bf16"  # "fp8" enables the CDNA4 fp8 expert GEMM path

    @property
    def is_moe(self) -> bool:
        return self.moe_num_experts is not None


MODEL_REGISTRY = {
    # Llama-3 8B (BASELINE configs #2/#3)
    "llama3-8b": ModelSpec(
        name="llama3-8b",
        hidden_size=4096,
        intermediate_size=14336,
        num_layers=32,
        num_heads=32,
        num_kv_heads=8,
    ),
    # Llama-3 70B (BASELINE config #4, TP=8)
    "llama3-70b": ModelSpec(
        name="llama3-70b",
        hidden_size=8192,
        intermediate_size=28672,
        num_layers=80,
        num_heads=64,
        num_kv_heads=8,
    ),
    # tiny model for CPU tests (vocab covers the byte tokenizer's ~260 ids)
    "llama3-tiny": ModelSpec(
        name="llama3-tiny",
        vocab_size=512,
        hidden_size=128,
        intermediate_size=256,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=32,
        max_seq_len=512,
    ),
    # tiny model whose head/vocab geometry divides by 4 (CPU world=4 TP tests)
    "llama3-tiny-w4": ModelSpec(
        name="llama3-tiny-w4",
        vocab_size=512,
        hidden_size=128,
        intermediate_size=256,
        num_layers=2,
        num_heads=8,
        num_kv_heads=4,
        head_dim=16,
        max_seq_len=512,
    ),
    # GPU-smoke-sized model (real head_dim for the HIP kernels)
    "llama3-micro": ModelSpec(
        name="llama3-micro",
        vocab_size=2048,
        hidden_size=1024,
        intermediate_size=2816,
        num_layers=4,
        num_heads=8,
        num_kv_heads=2,
        head_dim=128,
        max_seq_len=4096,
    ),
    # 8B-class projection shapes at 2 layers: exercises the K=4096 MFMA
    # GEMV paths (fp8 and bf16) end-to-end without an 8B weight build
    "llama3-2l4k": ModelSpec(
        name="llama3-2l4k",
        vocab_size=2048,
        hidden_size=4096,
        intermediate_size=14336,
        num_layers=2,
        num_heads=32,
        num_kv_heads=8,
        head_dim=128,
        max_seq_len=4096,
    ),
    # tiny MoE for CPU tests
    "moe-tiny": ModelSpec(
        name="moe-tiny",
        vocab_size=512,
        hidden_size=128,
        intermediate_size=256,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=32,
        max_seq_len=512,
        moe_num_experts=8,
        moe_top_k=2,
        moe_intermediate_size=64,
        moe_shared_experts=1,
    ),
    # GPU-smoke-sized MoE (real head_dim, fp8-capable expert shapes)
    "moe-micro": ModelSpec(
        name="moe-micro",
        vocab_size=2048,
        hidden_size=1024,
        intermediate_size=2816,
        num_layers=4,
        num_heads=8,
        num_kv_heads=2,
        head_dim=128,
        max_seq_len=4096,
        moe_num_experts=16,
        moe_top_k=4,
        moe_intermediate_size=512,
        moe_shared_experts=1,
    ),
    # DeepSeek-V3-style MoE, scaled down to fit one MI355X comfortably
    # (BASELINE config #5: fp8 experts on CDNA4 MFMA)
    "deepseek-moe-small": ModelSpec(
        name="deepseek-moe-small",
        vocab_size=128256,
        hidden_size=2048,
        intermediate_size=8192,       # dense MLP for the first layer(s)
        num_layers=12,
        num_heads=16,
        num_kv_heads=4,
        moe_num_experts=64,
        moe_top_k=6,
        moe_intermediate_size=1408,
        moe_shared_experts=2,
    ),
    "deepseek-moe": ModelSpec(
        name="deepseek-moe",
        vocab_size=128256,
        hidden_size=4096,
        intermediate_size=14336,
        num_layers=28,
        num_heads=32,
        num_kv_heads=8,
        moe_num_experts=128,
        moe_top_k=8,
        moe_intermediate_size=2048,
        moe_shared_experts=1,
    ),
}


def get_model_spec(name: str) -> ModelSpec:
    if name not in MODEL_REGISTRY:
        raise KeyError(f"unknown model {name!r}; known: {sorted(MODEL_REGISTRY)}")
    return MODEL_REGISTRY[name]
