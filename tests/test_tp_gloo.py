"""Tensor-parallel equivalence tests over gloo (CPU, world_size 2).

SURVEY.md §4 (e): TP sharding equivalence TP=1 vs TP=2 on the same random
weights. Because weights are drawn full-size from a fixed seed and sliced per
rank, the TP=2 model must produce (numerically close) identical logits and
identical greedy tokens to the TP=1 model.
"""

import multiprocessing as mp
import os

import pytest
import torch


def _run_tp_worker(rank: int, world: int, port: int, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    import torch.distributed as dist

    dist.init_process_group(backend="gloo", rank=rank, world_size=world)
    from opsagent_amd.parallel import state

    state.set_tp_state(rank, world, dist.group.WORLD)

    from opsagent_amd.engine.engine import LLMEngine, SamplingParams

    eng = LLMEngine(
        {
            "model": "llama3-tiny",
            "max_seq_len": 128,
            "kv_block_size": 16,
            "use_hipgraph": False,
            "seed": 7,
        }
    )
    ids = eng.tokenizer.encode("tensor parallel check", add_bos=True)
    out, _ = eng.generate(ids, SamplingParams(max_new_tokens=10))
    if rank == 0:
        q.put(out)
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_tp2_matches_tp1():
    # TP=1 baseline in-process
    from opsagent_amd.parallel import state

    state.set_tp_state(0, 1, None)
    from opsagent_amd.engine.engine import LLMEngine, SamplingParams

    eng = LLMEngine(
        {
            "model": "llama3-tiny",
            "max_seq_len": 128,
            "kv_block_size": 16,
            "use_hipgraph": False,
            "seed": 7,
        }
    )
    ids = eng.tokenizer.encode("tensor parallel check", add_bos=True)
    ref, _ = eng.generate(ids, SamplingParams(max_new_tokens=10))

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    procs = [
        ctx.Process(target=_run_tp_worker, args=(r, 2, port, q)) for r in range(2)
    ]
    for p in procs:
        p.start()
    tp_out = q.get(timeout=150)
    for p in procs:
        p.join(timeout=60)
    assert tp_out == ref, f"TP=2 tokens {tp_out} != TP=1 tokens {ref}"


def test_shard_helper():
    from opsagent_amd.engine.model import _shard

    t = torch.arange(12).reshape(4, 3)
    s0 = _shard(t, 0, 0, 2)
    s1 = _shard(t, 0, 1, 2)
    assert torch.equal(torch.cat([s0, s1], 0), t)
    with pytest.raises(AssertionError):
        _shard(t, 1, 0, 2)  # 3 not divisible by 2


def _run_tp_features_worker(rank: int, world: int, port: int, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    import torch.distributed as dist

    dist.init_process_group(backend="gloo", rank=rank, world_size=world)
    from opsagent_amd.parallel import state

    state.set_tp_state(rank, world, dist.group.WORLD)

    from opsagent_amd.engine.engine import LLMEngine, SamplingParams
    from opsagent_amd.engine.grammar import GrammarMode

    cfg = {
        "model": "llama3-tiny",
        "max_seq_len": 256,
        "kv_block_size": 16,
        "use_hipgraph": False,
        "seed": 7,
        "kv_num_blocks": 24,  # tight: exercises preemption under TP
        "spec_decode": True,
    }
    eng = LLMEngine(cfg)
    tok = eng.tokenizer
    results = []
    # jump-ahead (grammar), speculation (repetitive), preemption (2 at once)
    r1 = eng.add_request(
        tok.encode("grammar tp", add_bos=True),
        SamplingParams(max_new_tokens=48, grammar=GrammarMode.TOOLPROMPT),
    )
    r2 = eng.add_request(
        tok.encode("repeat repeat repeat", add_bos=True),
        SamplingParams(max_new_tokens=32),
    )
    for _ in range(2000):
        if eng.requests[r1].finished and eng.requests[r2].finished:
            break
        eng.step()
    results.append(eng.requests.pop(r1).output_ids)
    results.append(eng.requests.pop(r2).output_ids)
    if rank == 0:
        q.put(results)
    dist.destroy_process_group()


@pytest.mark.timeout(240)
def test_tp2_features_match_tp1():
    """Jump-ahead, speculative decoding, and KV-pressure preemption stay
    rank-synchronized (deterministic) and match single-rank outputs."""
    from opsagent_amd.parallel import state

    state.set_tp_state(0, 1, None)
    from opsagent_amd.engine.engine import LLMEngine, SamplingParams
    from opsagent_amd.engine.grammar import GrammarMode

    cfg = {
        "model": "llama3-tiny",
        "max_seq_len": 256,
        "kv_block_size": 16,
        "use_hipgraph": False,
        "seed": 7,
        "kv_num_blocks": 24,
        "spec_decode": True,
    }
    eng = LLMEngine(cfg)
    tok = eng.tokenizer
    r1 = eng.add_request(
        tok.encode("grammar tp", add_bos=True),
        SamplingParams(max_new_tokens=48, grammar=GrammarMode.TOOLPROMPT),
    )
    r2 = eng.add_request(
        tok.encode("repeat repeat repeat", add_bos=True),
        SamplingParams(max_new_tokens=32),
    )
    for _ in range(2000):
        if eng.requests[r1].finished and eng.requests[r2].finished:
            break
        eng.step()
    ref = [eng.requests.pop(r1).output_ids, eng.requests.pop(r2).output_ids]

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29531
    ps = [
        ctx.Process(target=_run_tp_features_worker, args=(r, 2, port, q))
        for r in range(2)
    ]
    for p in ps:
        p.start()
    got = q.get(timeout=220)
    for p in ps:
        p.join(timeout=60)
    assert got == ref
