"""Weight checkpoint load/save (safetensors).

The benchmark mode is random-init (no network for checkpoints — BASELINE),
but the engine supports real weights: a canonical full-weight safetensors
layout (written by `save_weights` at TP=1) and HuggingFace Llama naming
("model.layers.N.self_attn.q_proj.weight", ...). Loading is TP-aware: every
rank reads the full tensor and slices its shard along the same dims used at
init (column-parallel: dim 0 for q/k/v/gate/up; row-parallel: dim 1 for
o/down), so a checkpoint written at any TP=1 runs at TP=1..8 unchanged.
"""

from __future__ import annotations

from typing import Dict

import torch

from opsagent_amd.engine.model import LlamaForCausalLM, _shard
from opsagent_amd.parallel import get_tp_rank, get_tp_size
from opsagent_amd.utils.logging import get_logger

log = get_logger("loader")


def _canonical_from_hf(name: str) -> str:
    """Map HF Llama parameter names to canonical names."""
    n = name
    n = n.replace("model.embed_tokens.weight", "embed")
    n = n.replace("model.norm.weight", "final_norm")
    n = n.replace("lm_head.weight", "lm_head")
    n = n.replace("model.layers.", "layers.")
    n = n.replace(".self_attn.q_proj.weight", ".q")
    n = n.replace(".self_attn.k_proj.weight", ".k")
    n = n.replace(".self_attn.v_proj.weight", ".v")
    n = n.replace(".self_attn.o_proj.weight", ".o")
    n = n.replace(".mlp.gate_proj.weight", ".gate")
    n = n.replace(".mlp.up_proj.weight", ".up")
    n = n.replace(".mlp.down_proj.weight", ".down")
    # MoE (DeepSeek naming): router, per-expert projections, shared experts
    n = n.replace(".mlp.gate.weight", ".moe.router")
    n = n.replace(".mlp.shared_experts.gate_proj.weight", ".moe.shared.gate")
    n = n.replace(".mlp.shared_experts.up_proj.weight", ".moe.shared.up")
    n = n.replace(".mlp.shared_experts.down_proj.weight", ".moe.shared.down")
    n = n.replace(".gate_proj.weight", ".gate")  # .mlp.experts.K.gate_proj
    n = n.replace(".up_proj.weight", ".up")
    n = n.replace(".down_proj.weight", ".down")
    n = n.replace(".mlp.experts.", ".moe.experts.")
    n = n.replace(".input_layernorm.weight", ".input_norm")
    n = n.replace(".post_attention_layernorm.weight", ".post_norm")
    return n


def save_weights(model: LlamaForCausalLM, path: str) -> None:
    """Write the canonical full-weight checkpoint. Requires TP=1."""
    assert get_tp_size() == 1, "save_weights requires TP=1 (full weights)"
    from safetensors.torch import save_file

    spec = model.spec
    hd = spec.head_dim
    out: Dict[str, torch.Tensor] = {
        "embed": model.embed.data,
        "final_norm": model.final_norm_w.data,
        "lm_head": model.lm_head.data,
    }
    for i, layer in enumerate(model.layers):
        p = f"layers.{i}"
        out[f"{p}.input_norm"] = layer.input_norm_w.data
        out[f"{p}.post_norm"] = layer.post_norm_w.data
        qkv = layer.attn.qkv_w.data
        nq = spec.num_heads * hd
        nk = spec.num_kv_heads * hd
        out[f"{p}.q"] = qkv[:nq]
        out[f"{p}.k"] = qkv[nq : nq + nk]
        out[f"{p}.v"] = qkv[nq + nk :]
        out[f"{p}.o"] = layer.attn.o_w.data
        if hasattr(layer.mlp, "gate_up_w"):
            gu = layer.mlp.gate_up_w.data
            inter = gu.shape[0] // 2
            out[f"{p}.gate"] = gu[:inter]
            out[f"{p}.up"] = gu[inter:]
            out[f"{p}.down"] = layer.mlp.down_w.data
        elif hasattr(layer.mlp, "router_w"):
            moe = layer.mlp
            out[f"{p}.moe.router"] = moe.router_w.data
            if moe.w13 is not None:  # bf16 experts (fp8 is a derived format)
                out[f"{p}.moe.w13"] = moe.w13.data
                out[f"{p}.moe.w2"] = moe.w2.data
            if moe.shared is not None:
                sgu = moe.shared.gate_up_w.data
                si = sgu.shape[0] // 2
                out[f"{p}.moe.shared.gate"] = sgu[:si]
                out[f"{p}.moe.shared.up"] = sgu[si:]
                out[f"{p}.moe.shared.down"] = moe.shared.down_w.data
    save_file({k: v.contiguous().cpu() for k, v in out.items()}, path)


def _load_checkpoint(path: str) -> dict:
    """Read a safetensors checkpoint: a single .safetensors file, an HF
    shard index (model.safetensors.index.json), or a directory containing
    either."""
    import json as _json
    import os

    from safetensors.torch import load_file

    if os.path.isdir(path):
        idx = os.path.join(path, "model.safetensors.index.json")
        if os.path.exists(idx):
            path = idx
        else:
            single = os.path.join(path, "model.safetensors")
            if not os.path.exists(single):
                raise FileNotFoundError(f"no safetensors checkpoint in {path}")
            path = single
    if path.endswith(".index.json"):
        with open(path) as f:
            index = _json.load(f)
        base = os.path.dirname(path)
        raw: dict = {}
        for shard in sorted(set(index["weight_map"].values())):
            raw.update(load_file(os.path.join(base, shard)))
        return raw
    return load_file(path)


def load_weights(model: LlamaForCausalLM, path: str) -> int:
    """Load a canonical or HF-named safetensors checkpoint (single file,
    sharded HF index, or directory) into the model, slicing TP shards.
    Returns the number of parameters loaded."""
    raw = _load_checkpoint(path)
    tensors = {_canonical_from_hf(k): v for k, v in raw.items()}
    spec = model.spec
    tp, rank = get_tp_size(), get_tp_rank()
    hd = spec.head_dim
    dev = model.embed.device
    dtype = model.dtype
    n_loaded = 0

    def put(param: torch.nn.Parameter, t: torch.Tensor):
        nonlocal n_loaded
        assert param.data.shape == t.shape, f"shape {tuple(t.shape)} vs {tuple(param.shape)}"
        param.data.copy_(t.to(device=dev, dtype=param.dtype))
        n_loaded += 1

    if "embed" in tensors:
        put(model.embed, tensors["embed"])
    if "final_norm" in tensors:
        put(model.final_norm_w, tensors["final_norm"])
    if "lm_head" in tensors and not spec.tie_embeddings:
        # vocab-parallel head: each rank keeps its V/tp rows
        put(model.lm_head, _shard(tensors["lm_head"], 0, rank, tp))
    elif spec.tie_embeddings:
        pass  # tied: lm_head is a (vocab-sharded) view of embed — loaded above

    for i, layer in enumerate(model.layers):
        p = f"layers.{i}"
        if f"{p}.input_norm" in tensors:
            put(layer.input_norm_w, tensors[f"{p}.input_norm"])
        if f"{p}.post_norm" in tensors:
            put(layer.post_norm_w, tensors[f"{p}.post_norm"])
        if f"{p}.q" in tensors:
            q = _shard(tensors[f"{p}.q"], 0, rank, tp)
            k = _shard(tensors[f"{p}.k"], 0, rank, tp)
            v = _shard(tensors[f"{p}.v"], 0, rank, tp)
            put(layer.attn.qkv_w, torch.cat([q, k, v], dim=0))
            put(layer.attn.o_w, _shard(tensors[f"{p}.o"], 1, rank, tp))
        if f"{p}.gate" in tensors and hasattr(layer.mlp, "gate_up_w"):
            g = _shard(tensors[f"{p}.gate"], 0, rank, tp)
            u = _shard(tensors[f"{p}.up"], 0, rank, tp)
            put(layer.mlp.gate_up_w, torch.cat([g, u], dim=0))
            put(layer.mlp.down_w, _shard(tensors[f"{p}.down"], 1, rank, tp))
        elif hasattr(layer.mlp, "router_w"):
            _load_moe(tensors, p, layer.mlp, rank, tp, put)
    log.info("loaded %d tensors from %s", n_loaded, path)
    return n_loaded


def _load_moe(tensors: dict, p: str, moe, rank: int, tp: int, put) -> None:
    """Populate an MoE layer from canonical stacked tensors
    (layers.N.moe.{router,w13,w2}) or per-expert DeepSeek-style keys
    (layers.N.moe.experts.K.{gate,up,down}); fp8 expert layers re-quantize
    the loaded bf16 weights per expert."""
    if f"{p}.moe.router" in tensors:
        put(moe.router_w, tensors[f"{p}.moe.router"])
    w13 = w2 = None
    if f"{p}.moe.w13" in tensors:
        w13 = tensors[f"{p}.moe.w13"]
        w2 = tensors[f"{p}.moe.w2"]
    elif f"{p}.moe.experts.0.gate" in tensors:
        E = moe.num_experts
        w13 = torch.stack(
            [
                torch.cat(
                    [tensors[f"{p}.moe.experts.{k}.gate"],
                     tensors[f"{p}.moe.experts.{k}.up"]], dim=0
                )
                for k in range(E)
            ]
        )
        w2 = torch.stack([tensors[f"{p}.moe.experts.{k}.down"] for k in range(E)])
    if w13 is not None:
        inter = w13.shape[1] // 2
        if tp > 1:
            w13 = torch.cat(
                [_shard(w13[:, :inter], 1, rank, tp),
                 _shard(w13[:, inter:], 1, rank, tp)], dim=1
            )
            w2 = _shard(w2, 2, rank, tp)
        if moe.fp8:
            from opsagent_amd.ops import torch_ref as _tr

            for ei in range(moe.num_experts):
                q13, s13 = _tr.quant_fp8(w13[ei].to(torch.float32))
                q2, s2 = _tr.quant_fp8(w2[ei].to(torch.float32))
                moe.w13_q.data[ei].copy_(q13)
                moe.w13_s.data[ei].copy_(s13)
                moe.w2_q.data[ei].copy_(q2)
                moe.w2_s.data[ei].copy_(s2)
        else:
            put(moe.w13, w13)
            put(moe.w2, w2)
    if moe.shared is not None and f"{p}.moe.shared.gate" in tensors:
        g = _shard(tensors[f"{p}.moe.shared.gate"], 0, rank, tp)
        u = _shard(tensors[f"{p}.moe.shared.up"], 0, rank, tp)
        put(moe.shared.gate_up_w, torch.cat([g, u], dim=0))
        put(moe.shared.down_w, _shard(tensors[f"{p}.moe.shared.down"], 1, rank, tp))
