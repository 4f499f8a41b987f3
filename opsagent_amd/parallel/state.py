"""Distributed/tensor-parallel process state — RCCL over xGMI.

One process per GPU (torch.distributed backend "nccl" IS RCCL on ROCm); the
whole single-node job is one TP group. Design notes: xGMI is
point-to-point (7 links x ~153 GB/s per MI355X), so decode's small per-layer
all-reduces care about latency, not ring bandwidth — they stay inside the
hipGraph-captured decode step (RCCL collectives are capturable), and prefill's
large all-reduces use the default ring algorithms.

CPU tests use the gloo backend with world_size 2 (same code path).
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist

_TP_RANK = 0
_TP_SIZE = 1
_TP_GROUP: Optional[object] = None


def init_distributed(backend: Optional[str] = None) -> None:
    """Initialize from torchrun env (RANK/WORLD_SIZE/MASTER_*); no-op if
    WORLD_SIZE <= 1 or already initialized."""
    global _TP_RANK, _TP_SIZE, _TP_GROUP
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        _TP_RANK, _TP_SIZE, _TP_GROUP = 0, 1, None
        return
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        if backend == "nccl":
            # modulo lets a world-N job run on fewer GPUs (e.g. tp=2 RCCL
            # validation on a 1-GPU box: both ranks share device 0)
            torch.cuda.set_device(
                int(os.environ.get("LOCAL_RANK", "0")) %
                max(torch.cuda.device_count(), 1))
        dist.init_process_group(backend=backend)
    _TP_RANK = dist.get_rank()
    _TP_SIZE = dist.get_world_size()
    _TP_GROUP = dist.group.WORLD


def set_tp_state(rank: int, size: int, group=None) -> None:
    """For tests that manage process groups directly."""
    global _TP_RANK, _TP_SIZE, _TP_GROUP
    _TP_RANK, _TP_SIZE, _TP_GROUP = rank, size, group


def get_tp_rank() -> int:
    return _TP_RANK


def get_tp_size() -> int:
    return _TP_SIZE


def get_tp_group():
    return _TP_GROUP


def tp_all_reduce(t: torch.Tensor) -> torch.Tensor:
    """Sum all-reduce across the TP group (no-op at TP=1).

    Small (decode-sized) messages take the one-shot latency path; large
    (prefill) messages use the bandwidth-optimal ring (parallel.comms)."""
    if _TP_SIZE > 1:
        from opsagent_amd.parallel.comms import smart_all_reduce

        smart_all_reduce(t)
    return t


def tp_all_gather(t: torch.Tensor, dim: int = -1) -> torch.Tensor:
    if _TP_SIZE <= 1:
        return t
    parts = [torch.empty_like(t) for _ in range(_TP_SIZE)]
    dist.all_gather(parts, t.contiguous(), group=_TP_GROUP)
    return torch.cat(parts, dim=dim)


def barrier() -> None:
    if _TP_SIZE > 1:
        dist.barrier(group=_TP_GROUP)
