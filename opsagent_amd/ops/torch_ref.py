"""Pure-PyTorch reference implementations of the engine's hot ops.

These are the numerics oracle for the HIP/CDNA4 kernels (tests compare the
HIP kernel against these in fp32 — SURVEY.md §4 test strategy (c)) and the
CPU execution path. They are intentionally simple and readable; the GPU path
never runs these (opsagent_amd.ops dispatch raises if the HIP extension is
missing on a GPU box).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    dtype = x.dtype
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps) * weight.float()
    return out.to(dtype)


def fused_add_rms_norm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (rmsnorm(x + residual), x + residual)."""
    s = (x.float() + residual.float())
    out = rms_norm(s, weight, eps)
    return out.to(x.dtype), s.to(x.dtype)


def rope_cos_sin(
    max_seq: int, head_dim: int, theta: float = 500000.0, device="cpu", dtype=torch.float32
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Precomputed RoPE tables [max_seq, head_dim/2] (host-side per CDNA guide §B)."""
    inv_freq = 1.0 / (
        theta ** (torch.arange(0, head_dim, 2, device=device, dtype=torch.float32) / head_dim)
    )
    t = torch.arange(max_seq, device=device, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)
    return freqs.cos().to(dtype), freqs.sin().to(dtype)


def rope_apply(
    q: torch.Tensor,
    k: torch.Tensor,
    cos: torch.Tensor,
    sin: torch.Tensor,
    positions: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Neox-style (rotate-half) RoPE.

    q: [T, Hq, D], k: [T, Hk, D]; cos/sin: [max_seq, D/2]; positions: [T].
    """
    def _rot(x: torch.Tensor) -> torch.Tensor:
        d = x.shape[-1]
        c = cos[positions].to(torch.float32).unsqueeze(1)  # [T,1,D/2]
        s = sin[positions].to(torch.float32).unsqueeze(1)
        x1 = x[..., : d // 2].float()
        x2 = x[..., d // 2 :].float()
        return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1).to(x.dtype)

    return _rot(q), _rot(k)


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    return (torch.nn.functional.silu(gate.float()) * up.float()).to(gate.dtype)


def attention_prefill(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    scale: Optional[float] = None,
    causal: bool = True,
) -> torch.Tensor:
    """Causal attention over contiguous tensors.

    q: [B, Hq, Sq, D]; k, v: [B, Hk, Skv, D] with Skv >= Sq (the causal
    diagonal is offset so that query i attends keys [0 .. Skv-Sq+i]).
    GQA: Hq % Hk == 0 (kv heads broadcast over query-head groups).
    """
    B, Hq, Sq, D = q.shape
    Hk, Skv = k.shape[1], k.shape[2]
    scale = scale if scale is not None else D ** -0.5
    group = Hq // Hk
    kf = k.float().repeat_interleave(group, dim=1)
    vf = v.float().repeat_interleave(group, dim=1)
    s = torch.einsum("bhqd,bhkd->bhqk", q.float(), kf) * scale
    if causal:
        offset = Skv - Sq
        qi = torch.arange(Sq, device=q.device).unsqueeze(1)
        ki = torch.arange(Skv, device=q.device).unsqueeze(0)
        mask = ki > (qi + offset)
        s = s.masked_fill(mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    out = torch.einsum("bhqk,bhkd->bhqd", p, vf)
    return out.to(q.dtype)


def attention_decode_paged(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_table: torch.Tensor,
    seq_lens: torch.Tensor,
    scale: Optional[float] = None,
) -> torch.Tensor:
    """Single-token decode attention against a paged KV cache.

    q:           [B, Hq, D]         (one new token per sequence)
    k/v_cache:   [num_blocks, block_size, Hk, D]
    block_table: [B, max_blocks] int32 (block ids per sequence, -1 padded)
    seq_lens:    [B] int32 — total tokens in cache per sequence (incl. current)
    """
    B, Hq, D = q.shape
    block_size = k_cache.shape[1]
    Hk = k_cache.shape[2]
    group = Hq // Hk
    scale = scale if scale is not None else D ** -0.5
    outs = []
    for b in range(B):
        n = int(seq_lens[b].item())
        nblocks = (n + block_size - 1) // block_size
        blocks = block_table[b, :nblocks].long()
        k = k_cache[blocks].reshape(-1, Hk, D)[:n].float()  # [n, Hk, D]
        v = v_cache[blocks].reshape(-1, Hk, D)[:n].float()
        kf = k.repeat_interleave(group, dim=1)  # [n, Hq, D]
        vf = v.repeat_interleave(group, dim=1)
        s = torch.einsum("hd,nhd->hn", q[b].float(), kf) * scale
        p = torch.softmax(s, dim=-1)
        outs.append(torch.einsum("hn,nhd->hd", p, vf))
    return torch.stack(outs).to(q.dtype)


def kv_cache_write(
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    slot_mapping: torch.Tensor,
) -> None:
    """Scatter new K/V rows into the paged cache.

    k/v: [T, Hk, D]; slot_mapping: [T] int64 flat slot = block_id*block_size + offset.
    """
    nb, bs, hk, d = k_cache.shape
    k_cache.view(nb * bs, hk, d)[slot_mapping] = k
    v_cache.view(nb * bs, hk, d)[slot_mapping] = v


def quant_fp8(x: torch.Tensor):
    """Per-row OCP e4m3 quantization reference (torch.float8_e4m3fn)."""
    shape = x.shape
    xf = x.reshape(-1, shape[-1]).float()
    amax = xf.abs().amax(dim=-1, keepdim=True).clamp_min(1e-8)
    scale = (amax / 448.0).squeeze(-1)
    q = (xf / scale.unsqueeze(-1)).to(torch.float8_e4m3fn)
    return q.view(torch.uint8).reshape(shape), scale


def dequant_fp8(q: torch.Tensor, scale: torch.Tensor) -> torch.Tensor:
    f = q.view(torch.float8_e4m3fn).float()
    return f * scale.unsqueeze(-1)


def linear_fp8(x: torch.Tensor, w8: torch.Tensor, w_scale: torch.Tensor) -> torch.Tensor:
    w = dequant_fp8(w8, w_scale)
    return torch.nn.functional.linear(x.float(), w).to(x.dtype)


def greedy_sample_masked(logits: torch.Tensor, mask: Optional[torch.Tensor]) -> torch.Tensor:
    """argmax over allowed tokens. logits [B, V]; mask [B, V] bool (True = allowed)."""
    lf = logits.float()
    if mask is not None:
        lf = lf.masked_fill(~mask, float("-inf"))
    return lf.argmax(dim=-1)
