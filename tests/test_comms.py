"""Collective-helper tests over gloo: the one-shot (all-gather + local sum)
all-reduce must equal the ring all-reduce for small and large tensors."""

import multiprocessing as mp
import os
import socket

import pytest
import torch


def _worker(rank, world, port, q):
    os.environ.update(
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port), RANK=str(rank), WORLD_SIZE=str(world)
    )
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    from opsagent_amd.parallel import state
    from opsagent_amd.parallel.comms import latency_all_reduce, smart_all_reduce

    state.set_tp_state(rank, world, dist.group.WORLD)
    torch.manual_seed(rank)
    results = {}
    for name, n in [("small", 1024), ("large", 300_000)]:
        t = torch.randn(n)
        ref = t.clone()
        dist.all_reduce(ref, op=dist.ReduceOp.SUM)
        got_one = latency_all_reduce(t.clone())
        got_smart = smart_all_reduce(t.clone())
        results[name] = (
            torch.allclose(got_one, ref, atol=1e-5),
            torch.allclose(got_smart, ref, atol=1e-5),
        )
    # 2D tensor reuse of the gather buffer
    t2 = torch.randn(4, 64)
    ref2 = t2.clone()
    dist.all_reduce(ref2, op=dist.ReduceOp.SUM)
    ok2 = torch.allclose(latency_all_reduce(t2.clone()), ref2, atol=1e-5)
    if rank == 0:
        q.put((results, ok2))
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_one_shot_all_reduce_matches_ring():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results, ok2 = q.get(timeout=100)
    for p in procs:
        p.join(timeout=30)
    for name, (a, b) in results.items():
        assert a and b, f"{name}: one-shot={a} smart={b}"
    assert ok2
