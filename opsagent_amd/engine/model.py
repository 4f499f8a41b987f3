"""Llama-3-family model with tensor parallelism, built on the ops dispatch
layer (HIP kernels on GPU, torch_ref on CPU).

Replaces the reference's remote model layer. Weight layout is
fused and token-major throughout — activations are [T, hidden]; attention
tensors [T, H, D] (no [B, H, S, D] transposes; the HIP kernels take the
token-major layout directly).

TP sharding (SURVEY.md §2b "TP sharding (row/col-parallel linear,
head-parallel attention)"): QKV and gate/up are column-parallel, o and down
are row-parallel with one RCCL all-reduce each per layer; heads are divided
across ranks (Hq % tp == 0 and Hk % tp == 0). Random init draws the FULL
weight with a fixed seed on every rank and slices the local shard, so TP=k is
numerically the same model as TP=1 (tested via gloo on CPU).
"""

from __future__ import annotations

import dataclasses
import math
from typing import List, Optional, Tuple

import torch
import torch.nn.functional as F
from torch import nn

from opsagent_amd import ops
from opsagent_amd.engine.config import ModelSpec
from opsagent_amd.parallel import (
    get_tp_rank,
    get_tp_size,
    tp_all_gather,
    tp_all_reduce,
)

_ROPE_ATTN_FUSED: Optional[bool] = None


def _rope_attn_fused() -> bool:
    """Fold RoPE + KV scatter into the decode attention launch (saves the
    rope_kv kernel per layer). OPSAGENT_ROPE_ATTN_FUSED=0 restores the
    separate-kernel path (A/B reference)."""
    global _ROPE_ATTN_FUSED
    if _ROPE_ATTN_FUSED is None:
        import os

        _ROPE_ATTN_FUSED = os.environ.get("OPSAGENT_ROPE_ATTN_FUSED", "1") != "0"
    return _ROPE_ATTN_FUSED


@dataclasses.dataclass
class ForwardBatch:
    """Metadata for one engine step.

    prefill: one sequence of T new tokens (with optional cached past);
    decode: B sequences, one new token each.
    """

    kind: str  # "prefill" | "decode"
    input_ids: torch.Tensor      # [T] int64
    positions: torch.Tensor      # [T] int32
    slot_mapping: torch.Tensor   # [T] int32 — flat cache slots for new KV
    # prefill only:
    prefill_slot_gather: Optional[torch.Tensor] = None  # [Skv] int64 — all slots incl. past
    prefill_past_len: int = 0
    # decode only:
    block_table: Optional[torch.Tensor] = None  # [B, maxb] int32
    seq_lens: Optional[torch.Tensor] = None     # [B] int32
    decode_workspace: Optional[Tuple[torch.Tensor, torch.Tensor]] = None
    nsplit: Optional[int] = None


def _shard(t: torch.Tensor, dim: int, rank: int, size: int) -> torch.Tensor:
    if size == 1:
        return t
    n = t.shape[dim]
    assert n % size == 0, f"cannot shard dim {dim} of {tuple(t.shape)} by {size}"
    return t.narrow(dim, rank * (n // size), n // size).contiguous()


def _init_linear(gen: torch.Generator, out_f: int, in_f: int, dtype) -> torch.Tensor:
    std = 1.0 / math.sqrt(in_f)
    dev = gen.device if isinstance(gen, torch.Generator) else "cpu"
    w = torch.randn(out_f, in_f, generator=gen, dtype=torch.float32, device=dev)
    return (w * std).to(dtype)


class Attention(nn.Module):
    def __init__(self, spec: ModelSpec, dtype: torch.dtype, gen: torch.Generator):
        super().__init__()
        tp, rank = get_tp_size(), get_tp_rank()
        assert spec.num_heads % tp == 0 and spec.num_kv_heads % tp == 0, (
            f"heads ({spec.num_heads}/{spec.num_kv_heads}) not divisible by tp={tp}"
        )
        self.hq = spec.num_heads // tp
        self.hk = spec.num_kv_heads // tp
        self.hd = spec.head_dim
        self.scale = self.hd ** -0.5
        h = spec.hidden_size
        # full-weight init then shard (TP-equivalent to TP=1 by construction)
        q_w = _init_linear(gen, spec.num_heads * self.hd, h, dtype)
        k_w = _init_linear(gen, spec.num_kv_heads * self.hd, h, dtype)
        v_w = _init_linear(gen, spec.num_kv_heads * self.hd, h, dtype)
        o_w = _init_linear(gen, h, spec.num_heads * self.hd, dtype)
        self.qkv_w = nn.Parameter(
            torch.cat(
                [
                    _shard(q_w, 0, rank, tp),
                    _shard(k_w, 0, rank, tp),
                    _shard(v_w, 0, rank, tp),
                ],
                dim=0,
            ),
            requires_grad=False,
        )
        self.o_w = nn.Parameter(_shard(o_w, 1, rank, tp), requires_grad=False)
        self.fp8 = False  # set by LlamaForCausalLM.quantize_fp8_

    def _qkv(self, x: torch.Tensor) -> torch.Tensor:
        if self.fp8:
            return ops.linear_fp8(x, self.qkv_q, self.qkv_s)
        return ops.linear(x, self.qkv_w)

    def _o(self, x: torch.Tensor) -> torch.Tensor:
        if self.fp8:
            return ops.linear_fp8(x, self.o_q, self.o_s)
        return ops.linear(x, self.o_w)

    def forward(
        self,
        x: torch.Tensor,  # [T, hidden]
        cos: torch.Tensor,
        sin: torch.Tensor,
        k_cache: torch.Tensor,
        v_cache: torch.Tensor,
        fb: ForwardBatch,
    ) -> torch.Tensor:
        T = x.shape[0]
        qkv = self._qkv(x)  # hipBLASLt / HIP gemv / fp8 at decode
        q, k, v = qkv.split(
            [self.hq * self.hd, self.hk * self.hd, self.hk * self.hd], dim=-1
        )
        # head-slice VIEWS of the fused qkv buffer — the HIP kernels take the
        # shared token stride, so no .contiguous() copies here (GPU path)
        q = q.view(T, self.hq, self.hd)
        k = k.view(T, self.hk, self.hd)
        v = v.view(T, self.hk, self.hd)
        if not x.is_cuda:
            q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        if fb.kind == "decode" and x.is_cuda and _rope_attn_fused():
            # fully-fused decode attention (RoPE + KV scatter in-kernel)
            out = ops.attention_decode_rope(
                q, k, v, k_cache, v_cache, fb.block_table, fb.seq_lens,
                cos, sin, fb.slot_mapping, scale=self.scale,
                workspace=fb.decode_workspace, nsplit=fb.nsplit,
            ).view(T, self.hq * self.hd)
            out = self._o(out)
            return tp_all_reduce(out)
        q, k, v = ops.rope_kv_fused(
            q, k, v, k_cache, v_cache, cos, sin, fb.positions, fb.slot_mapping
        )

        if fb.kind == "prefill":
            skv = fb.prefill_past_len + T
            nb, bs, hk, hd = k_cache.shape
            if fb.prefill_past_len > 0:
                flat_k = k_cache.view(nb * bs, hk, hd)
                flat_v = v_cache.view(nb * bs, hk, hd)
                k_full = flat_k[fb.prefill_slot_gather].view(1, skv, hk, hd)
                v_full = flat_v[fb.prefill_slot_gather].view(1, skv, hk, hd)
            else:
                k_full = k.unsqueeze(0)
                v_full = v.unsqueeze(0)
            out = ops.attention_prefill(
                q.unsqueeze(0), k_full, v_full, scale=self.scale
            )
            out = out.view(T, self.hq * self.hd)
        else:
            out = ops.attention_decode_paged(
                q,
                k_cache,
                v_cache,
                fb.block_table,
                fb.seq_lens,
                scale=self.scale,
                workspace=fb.decode_workspace,
                nsplit=fb.nsplit,
            )
            out = out.view(T, self.hq * self.hd)
        out = self._o(out)
        return tp_all_reduce(out)


class DenseMLP(nn.Module):
    def __init__(self, spec: ModelSpec, dtype: torch.dtype, gen: torch.Generator):
        super().__init__()
        tp, rank = get_tp_size(), get_tp_rank()
        h, inter = spec.hidden_size, spec.intermediate_size
        assert inter % tp == 0
        gate = _init_linear(gen, inter, h, dtype)
        up = _init_linear(gen, inter, h, dtype)
        down = _init_linear(gen, h, inter, dtype)
        self.i_local = inter // tp
        self.gate_up_w = nn.Parameter(
            torch.cat([_shard(gate, 0, rank, tp), _shard(up, 0, rank, tp)], dim=0),
            requires_grad=False,
        )
        self.down_w = nn.Parameter(_shard(down, 1, rank, tp), requires_grad=False)
        self.fp8 = False  # set by LlamaForCausalLM.quantize_fp8_

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.fp8:
            act = ops.gateup_silu_fp8(
                x, self.gate_up_q, self.gate_up_s, self.i_local
            )
            return tp_all_reduce(ops.linear_fp8(act, self.down_q, self.down_s))
        act = ops.gateup_silu(x, self.gate_up_w, self.i_local)
        return tp_all_reduce(ops.linear(act, self.down_w))


class DecoderLayer(nn.Module):
    def __init__(self, spec: ModelSpec, dtype: torch.dtype, gen: torch.Generator):
        super().__init__()
        self.input_norm_w = nn.Parameter(
            torch.ones(spec.hidden_size, dtype=dtype), requires_grad=False
        )
        self.post_norm_w = nn.Parameter(
            torch.ones(spec.hidden_size, dtype=dtype), requires_grad=False
        )
        self.eps = spec.rms_eps
        self.attn = Attention(spec, dtype, gen)
        self.is_moe_layer = spec.is_moe
        if spec.is_moe:
            from opsagent_amd.engine.moe import MoEMLP

            self.mlp: nn.Module = MoEMLP(spec, dtype, gen)
        else:
            self.mlp = DenseMLP(spec, dtype, gen)

    def forward(self, x, residual, cos, sin, k_cache, v_cache, fb):
        if residual is None:
            residual = x
            h = ops.rms_norm(x, self.input_norm_w, self.eps)
        else:
            h, residual = ops.fused_add_rms_norm(x, residual, self.input_norm_w, self.eps)
        a = self.attn(h, cos, sin, k_cache, v_cache, fb)
        h, residual = ops.fused_add_rms_norm(a, residual, self.post_norm_w, self.eps)
        if self.is_moe_layer:
            m = self.mlp(h, decode=fb.kind == "decode")
        else:
            m = self.mlp(h)
        return m, residual

    def fused_decode(self, residual, cos, sin, k_cache, v_cache, fb):
        """Decode with norms fused into GEMV prologues and residual adds into
        GEMV epilogues (see LlamaForCausalLM._use_fused_decode). Takes and
        returns the residual stream."""
        at = self.attn
        T = residual.shape[0]
        if at.fp8:
            qkv = ops.linear_norm_fp8(
                residual, self.input_norm_w, self.eps, at.qkv_q, at.qkv_s
            )
        else:
            qkv = ops.linear_norm(residual, self.input_norm_w, self.eps, at.qkv_w)
        q, k, v = qkv.split([at.hq * at.hd, at.hk * at.hd, at.hk * at.hd], dim=-1)
        q = q.view(T, at.hq, at.hd)
        k = k.view(T, at.hk, at.hd)
        v = v.view(T, at.hk, at.hd)
        if _rope_attn_fused():
            ctx = ops.attention_decode_rope(
                q, k, v, k_cache, v_cache, fb.block_table, fb.seq_lens,
                cos, sin, fb.slot_mapping, scale=at.scale,
                workspace=fb.decode_workspace, nsplit=fb.nsplit,
            ).view(T, at.hq * at.hd)
        else:
            q, k, v = ops.rope_kv_fused(
                q, k, v, k_cache, v_cache, cos, sin, fb.positions, fb.slot_mapping
            )
            ctx = ops.attention_decode_paged(
                q, k_cache, v_cache, fb.block_table, fb.seq_lens, scale=at.scale,
                workspace=fb.decode_workspace, nsplit=fb.nsplit,
            ).view(T, at.hq * at.hd)
        if at.fp8:
            residual = ops.linear_addres_fp8(ctx, at.o_q, at.o_s, residual)
            act = ops.gateup_silu_fp8(
                residual, self.mlp.gate_up_q, self.mlp.gate_up_s,
                self.mlp.i_local, norm_w=self.post_norm_w, eps=self.eps,
            )
            return ops.linear_addres_fp8(
                act, self.mlp.down_q, self.mlp.down_s, residual
            )
        residual = ops.linear_addres(ctx, at.o_w, residual)
        act = ops.gateup_silu_norm(
            residual, self.post_norm_w, self.eps, self.mlp.gate_up_w, self.mlp.i_local
        )
        return ops.linear_addres(act, self.mlp.down_w, residual)


class LlamaForCausalLM(nn.Module):
    """Llama-3 architecture (also hosts DeepSeek-style MoE layers via spec)."""

    def __init__(
        self,
        spec: ModelSpec,
        dtype: torch.dtype = torch.bfloat16,
        device: str = "cpu",
        seed: int = 1234,
    ):
        super().__init__()
        self.spec = spec
        self.dtype = dtype
        # draw weights on the TARGET device (randn of 8B params on CPU takes
        # ~a minute; on HBM it is instant). Same seed + full-draw-then-shard
        # keeps TP=k identical to TP=1 for a given device type.
        gen = torch.Generator(device=device).manual_seed(seed)
        std = 1.0 / math.sqrt(spec.hidden_size)
        self.embed = nn.Parameter(
            (torch.randn(spec.vocab_size, spec.hidden_size, generator=gen,
                         device=device) * std).to(dtype),
            requires_grad=False,
        )
        self.layers = nn.ModuleList(
            [DecoderLayer(spec, dtype, gen) for _ in range(spec.num_layers)]
        )
        self.final_norm_w = nn.Parameter(
            torch.ones(spec.hidden_size, dtype=dtype), requires_grad=False
        )
        # lm_head is VOCAB-PARALLEL at TP>1: each rank holds
        # V/tp rows and computes [T, V/tp] local logits; compute_logits
        # all-gathers along vocab. At 70B TP=8 this cuts the per-token head
        # stream from 1.05 GB/rank (replicated) to 131 MB/rank, and the
        # gather payload at decode is only B*V*2 bytes (~256 KB).
        tp, rank = get_tp_size(), get_tp_rank()
        assert spec.vocab_size % tp == 0, (
            f"vocab {spec.vocab_size} not divisible by tp={tp}"
        )
        self.vocab_local = spec.vocab_size // tp
        if spec.tie_embeddings:
            if tp == 1:
                self.lm_head = self.embed
            else:
                # narrow VIEW of the tied embedding — no extra memory
                self.lm_head = self.embed.data.narrow(
                    0, rank * self.vocab_local, self.vocab_local
                )
        else:
            self.lm_head = nn.Parameter(
                _shard(
                    _init_linear(gen, spec.vocab_size, spec.hidden_size, dtype),
                    0, rank, tp,
                ),
                requires_grad=False,
            )
        cos, sin = ops.rope_cos_sin(spec.max_seq_len, spec.head_dim, spec.rope_theta)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        self.to(device)

    def _use_fused_decode(self, fb: ForwardBatch, x: torch.Tensor) -> bool:
        """Fused-norm decode path: per layer, rmsnorm folds into the qkv /
        gate-up GEMV prologues and the residual add into the o / down GEMV
        epilogues — 7 kernels per layer instead of 9. TP=1 only (the residual
        add must happen after the all-reduce) and dense models only."""
        import os

        return (
            x.is_cuda
            and fb.kind == "decode"
            # fused-norm wins at B=1 (3.73 -> 3.61 ms/step) but its rstd
            # prologue scales with batch and measured a net loss at c=8
            and x.shape[0] <= int(os.environ.get("OPSAGENT_FUSED_DECODE_MAX_B", "1"))
            and get_tp_size() == 1
            and not self.spec.is_moe
            and self.spec.hidden_size % 8 == 0
        )

    @torch.no_grad()
    def quantize_fp8_(self) -> None:
        """Quantize the dense projection weights to OCP e4m3 fp8 with
        per-row scales (engine `quantize: fp8`): decode streams HALF the
        weight bytes (8B: 16.1 -> ~8.2 GB). Embed, lm_head, norms,
        attention math and the KV cache stay bf16. Layer-by-layer so the
        peak is one extra layer, not a second copy of the model."""
        self.quantized_fp8 = True
        for layer in self.layers:
            at = layer.attn
            for name in ("qkv_w", "o_w"):
                w = getattr(at, name).data
                q, s = ops.quant_fp8(w)
                base = name[:-2]
                delattr(at, name)
                setattr(at, base + "_q", nn.Parameter(q, requires_grad=False))
                setattr(at, base + "_s", nn.Parameter(s, requires_grad=False))
            at.fp8 = True
            mlp = layer.mlp
            if hasattr(mlp, "gate_up_w"):
                for name in ("gate_up_w", "down_w"):
                    w = getattr(mlp, name).data
                    q, s = ops.quant_fp8(w)
                    base = name[:-2]
                    delattr(mlp, name)
                    setattr(mlp, base + "_q", nn.Parameter(q, requires_grad=False))
                    setattr(mlp, base + "_s", nn.Parameter(s, requires_grad=False))
                mlp.fp8 = True
        if self.embed.is_cuda:
            torch.cuda.empty_cache()

    def forward(self, fb: ForwardBatch, kv_caches: List[Tuple[torch.Tensor, torch.Tensor]]):
        x = F.embedding(fb.input_ids, self.embed)
        if self._use_fused_decode(fb, x):
            residual = x.contiguous()
            for i, layer in enumerate(self.layers):
                residual = layer.fused_decode(
                    residual, self.rope_cos, self.rope_sin,
                    kv_caches[i][0], kv_caches[i][1], fb,
                )
            return ops.rms_norm(residual, self.final_norm_w, self.spec.rms_eps)
        residual = None
        for i, layer in enumerate(self.layers):
            x, residual = layer(
                x, residual, self.rope_cos, self.rope_sin,
                kv_caches[i][0], kv_caches[i][1], fb,
            )
        x, _ = ops.fused_add_rms_norm(x, residual, self.final_norm_w, self.spec.rms_eps)
        return x  # [T, hidden]

    def compute_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        local = ops.linear(hidden, self.lm_head)  # [T, vocab/tp]
        # vocab-parallel gather: rank order IS vocab order (graph-capturable
        # RCCL all-gather; no-op at TP=1)
        return tp_all_gather(local, dim=-1)
