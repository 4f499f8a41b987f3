"""DeepSeek-V3-style Mixture-of-Experts MLP (BASELINE config #5).

Routing: softmax top-k over a router linear, normalized weights, plus
`moe_shared_experts` always-on shared experts (DeepSeek style). Experts are
TP-sharded on the intermediate dim (column-parallel w1/w3, row-parallel w2,
one all-reduce with the shared-expert output folded in).

Compute path v1: dense grouped GEMMs via torch (hipBLASLt batched) with
token gather/scatter. The CDNA4 fp8-MFMA expert GEMM kernel (spec.moe_dtype
== "fp8") is the planned upgrade — see ops/csrc/ notes.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import nn

from opsagent_amd import ops
from opsagent_amd.engine.config import ModelSpec
from opsagent_amd.engine.model import _init_linear, _shard  # noqa: F401
from opsagent_amd.parallel import get_tp_rank, get_tp_size, tp_all_reduce


class MoEMLP(nn.Module):
    def __init__(self, spec: ModelSpec, dtype: torch.dtype, gen: torch.Generator):
        super().__init__()
        tp, rank = get_tp_size(), get_tp_rank()
        h = spec.hidden_size
        e = spec.moe_num_experts
        i_moe = spec.moe_intermediate_size
        assert i_moe % tp == 0
        self.top_k = spec.moe_top_k
        self.num_experts = e
        self.i_local = i_moe // tp

        self.router_w = nn.Parameter(_init_linear(gen, e, h, torch.float32), requires_grad=False)
        w13 = torch.stack(
            [
                torch.cat([_init_linear(gen, i_moe, h, dtype), _init_linear(gen, i_moe, h, dtype)], 0)
                for _ in range(e)
            ]
        )  # [E, 2*I, H]
        w2 = torch.stack([_init_linear(gen, h, i_moe, dtype) for _ in range(e)])  # [E, H, I]
        # shard: w13 rows (both gate and up halves), w2 cols
        if tp > 1:
            gate = w13[:, :i_moe, :]
            up = w13[:, i_moe:, :]
            w13 = torch.cat(
                [_shard(gate, 1, rank, tp), _shard(up, 1, rank, tp)], dim=1
            )
            w2 = _shard(w2, 2, rank, tp)
        self.fp8 = spec.moe_dtype == "fp8"
        if self.fp8:
            # CDNA4 OCP-e4m3 expert weights (per-row scales): halves the
            # expert weight stream at decode and enables the fp8 MFMA GEMM
            # at prefill (ops/csrc/fp8_moe.hip)
            from opsagent_amd.ops import torch_ref as _tr

            q13 = torch.empty(*w13.shape, dtype=torch.uint8)
            s13 = torch.empty(e, w13.shape[1], dtype=torch.float32)
            q2 = torch.empty(*w2.shape, dtype=torch.uint8)
            s2 = torch.empty(e, w2.shape[1], dtype=torch.float32)
            for ei in range(e):
                q13[ei], s13[ei] = _tr.quant_fp8(w13[ei])
                q2[ei], s2[ei] = _tr.quant_fp8(w2[ei])
            self.w13_q = nn.Parameter(q13, requires_grad=False)
            self.w13_s = nn.Parameter(s13, requires_grad=False)
            self.w2_q = nn.Parameter(q2, requires_grad=False)
            self.w2_s = nn.Parameter(s2, requires_grad=False)
            self.w13 = self.w2 = None
        else:
            self.w13 = nn.Parameter(w13.contiguous(), requires_grad=False)
            self.w2 = nn.Parameter(w2.contiguous(), requires_grad=False)

        self._force_loop = False  # tests compare grouped vs loop paths
        self.n_shared = spec.moe_shared_experts
        if self.n_shared > 0:
            from opsagent_amd.engine.model import DenseMLP
            import copy

            shared_spec = copy.copy(spec)
            shared_spec.intermediate_size = i_moe * self.n_shared
            shared_spec.moe_num_experts = None
            self.shared = DenseMLP(shared_spec, dtype, gen)
        else:
            self.shared = None

    def forward(self, x: torch.Tensor, decode: bool = False) -> torch.Tensor:
        T, H = x.shape
        router_logits = F.linear(x.float(), self.router_w)  # [T, E]
        probs = torch.softmax(router_logits, dim=-1)
        topw, topi = probs.topk(self.top_k, dim=-1)           # [T, K]
        topw = (topw / topw.sum(dim=-1, keepdim=True)).to(x.dtype)

        # grouped path: pair-major kernels for tiny T; for DECODE batches
        # past 64 the expert-major kernels keep the stage in one
        # static-shaped launch pair (graph-capturable to batch 256).
        # Prefill-sized T stays on the sorted per-expert hipBLASLt loop —
        # extending the GEMV-style grouped path to 65..341-token prefill
        # chunks measured -18% on the moe-small c=4 turn.
        grouped_ok = T <= 64 or (decode and T * self.top_k <= 2048)
        if x.is_cuda and grouped_ok and not self._force_loop:
            # grouped-kernel path: one launch per stage for all pairs, no
            # host sync, shape-static -> hipGraph-capturable decode
            flat_e = topi.reshape(-1).to(torch.int32).contiguous()
            # NOTE: for T=1 expand().reshape(-1) is a stride-0 VIEW whose
            # data_ptr covers one element — must materialize contiguously
            flat_t = (
                torch.arange(T, device=x.device, dtype=torch.int32)
                .repeat_interleave(self.top_k)
                .contiguous()
            )
            flat_w = topw.reshape(-1).float().contiguous()
            y = ops.moe_grouped_mlp(
                x.contiguous(),
                self.w13_q if self.fp8 else self.w13,
                self.w13_s if self.fp8 else None,
                self.w2_q if self.fp8 else self.w2,
                self.w2_s if self.fp8 else None,
                flat_e, flat_t, flat_w, self.i_local, self.fp8,
                num_experts=self.num_experts,
            )
            out = torch.zeros_like(x)
            out.index_add_(0, flat_t.long(), y)
            if self.shared is not None:
                return tp_all_reduce(out) + self.shared(x)
            return tp_all_reduce(out)

        out = torch.zeros_like(x)
        # token gather per expert (sorted dispatch)
        flat_e = topi.reshape(-1)                              # [T*K]
        flat_t = (
            torch.arange(T, device=x.device).unsqueeze(1).expand(T, self.top_k).reshape(-1)
        )
        flat_w = topw.reshape(-1)
        order = torch.argsort(flat_e)
        flat_e, flat_t, flat_w = flat_e[order], flat_t[order], flat_w[order]
        # ONE host sync for the whole dispatch (a per-expert .item() would be
        # num_experts GPU->CPU round trips per layer per token)
        counts = torch.bincount(flat_e, minlength=self.num_experts).cpu().tolist()
        start = 0
        for e in range(self.num_experts):
            c = counts[e]
            if c == 0:
                continue
            toks = flat_t[start : start + c]
            wts = flat_w[start : start + c].unsqueeze(1)
            start += c
            xe = x[toks].contiguous()
            if self.fp8:
                gu = ops.linear_fp8(xe, self.w13_q[e], self.w13_s[e])
                gate, up = gu.split([self.i_local, self.i_local], dim=-1)
                act = ops.silu_mul(gate.contiguous(), up.contiguous())
                ye = ops.linear_fp8(act.contiguous(), self.w2_q[e], self.w2_s[e])
            else:
                gu = ops.linear(xe, self.w13[e])
                gate, up = gu.split([self.i_local, self.i_local], dim=-1)
                act = ops.silu_mul(gate.contiguous(), up.contiguous())
                ye = ops.linear(act, self.w2[e])
            out.index_add_(0, toks, ye * wts)
        if self.shared is not None:
            # shared expert runs its own all-reduce; fold by adding after
            out = tp_all_reduce(out) + self.shared(x)
            return out
        return tp_all_reduce(out)

    def forward_loop(self, x: torch.Tensor) -> torch.Tensor:
        """Force the per-expert loop path (test oracle for the grouped kernels)."""
        self._force_loop = True
        try:
            return self.forward(x)
        finally:
            self._force_loop = False
