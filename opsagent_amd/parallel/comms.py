"""Collective helpers tuned for xGMI point-to-point topology.

MI355X xGMI is 7 p2p links x ~153 GB/s per GPU (no switch). A ring
all-reduce moves 2*(n-1)/n of the data over ONE link per step, so for
decode's small hidden-size messages (8-16 KB at batch 1) it is pure latency:
2*(n-1) dependent hops. The one-shot form — every rank broadcasts its shard
to all peers simultaneously over its 7 links, then reduces locally — costs
one hop of latency and n-1 link-parallel transfers, the right trade below
~256 KB (SURVEY.md §5 design note).

`latency_all_reduce` implements the one-shot form portably on
torch.distributed (all_gather into a preallocated buffer + local sum —
all-to-all traffic, exactly the one-shot wire pattern; RCCL executes the
gather over the p2p links). `smart_all_reduce` picks it for small tensors
and ring all-reduce for large ones. Correctness is covered by gloo tests;
both paths are graph-capturable on RCCL.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from opsagent_amd.parallel.state import get_tp_group, get_tp_size

# below this many BYTES the one-shot (all-gather + local reduce) path wins:
# ring latency 2*(n-1) hops vs 1 hop; crossover measured on NVSwitch-less
# p2p fabrics around a few hundred KB
ONE_SHOT_MAX_BYTES = 256 * 1024

_gather_buf: dict = {}


def latency_all_reduce(t: torch.Tensor) -> torch.Tensor:
    """One-shot all-reduce: all-gather the full tensor from every rank, sum
    locally. Latency-optimal for small decode messages over p2p xGMI."""
    world = get_tp_size()
    if world <= 1:
        return t
    key = (t.shape, t.dtype, t.device)
    buf = _gather_buf.get(key)
    if buf is None or buf.shape[0] != world:
        buf = torch.empty(world, *t.shape, dtype=t.dtype, device=t.device)
        _gather_buf[key] = buf
    dist.all_gather_into_tensor(buf.view(-1), t.reshape(-1).contiguous(),
                                group=get_tp_group())
    torch.sum(buf.view(world, -1), dim=0, out=t.view(-1))
    return t


def smart_all_reduce(t: torch.Tensor) -> torch.Tensor:
    """Ring all-reduce for big (prefill) tensors, one-shot for small (decode)."""
    world = get_tp_size()
    if world <= 1:
        return t
    nbytes = t.numel() * t.element_size()
    if nbytes <= ONE_SHOT_MAX_BYTES:
        return latency_all_reduce(t)
    dist.all_reduce(t, op=dist.ReduceOp.SUM, group=get_tp_group())
    return t
