"""Concurrent serving tests: the engine loop batches concurrent submissions
and produces the same tokens as sequential generation."""

import threading

import pytest

from opsagent_amd.engine.engine import LLMEngine, SamplingParams
from opsagent_amd.engine.serving import EngineLoop

CFG = {
    "model": "llama3-tiny",
    "max_seq_len": 256,
    "kv_block_size": 16,
    "max_batch_size": 8,
    "use_hipgraph": False,
    "seed": 21,
}


@pytest.fixture(scope="module")
def loop():
    lp = EngineLoop(LLMEngine(dict(CFG)))
    yield lp
    lp.shutdown()


def test_concurrent_submissions_match_sequential(loop):
    tok = loop.engine.tokenizer
    prompts = [tok.encode(f"prompt number {i}", add_bos=True) for i in range(6)]
    futures = [loop.submit(p, SamplingParams(max_new_tokens=8)) for p in prompts]
    results = [f.result(timeout=120) for f in futures]
    assert all(len(out) > 0 for out, _ in results)

    # sequential reference on a fresh engine with the same seed
    eng2 = LLMEngine(dict(CFG))
    for p, (out, _) in zip(prompts, results):
        ref, _ = eng2.generate(p, SamplingParams(max_new_tokens=8))
        assert out == ref


def test_submit_from_many_threads(loop):
    tok = loop.engine.tokenizer
    results = {}
    errs = []

    def worker(i):
        try:
            out, reason = loop.generate(
                tok.encode(f"thread {i}", add_bos=True), SamplingParams(max_new_tokens=5)
            )
            results[i] = out
        except Exception as e:  # noqa: BLE001
            errs.append(e)

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(10)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=120)
    assert not errs
    assert len(results) == 10
    assert all(len(v) > 0 for v in results.values())


def test_oversubscription_queues(loop):
    """More requests than max_batch_size: all still complete."""
    tok = loop.engine.tokenizer
    futures = [
        loop.submit(tok.encode(f"r{i}", add_bos=True), SamplingParams(max_new_tokens=3))
        for i in range(20)
    ]
    for f in futures:
        out, _ = f.result(timeout=120)
        assert len(out) > 0


def test_engine_fault_fails_futures_and_recovers(loop):
    """A step exception fails in-flight requests; the loop keeps serving."""
    eng = loop.engine
    orig_step = eng.step
    state = {"raised": False}

    def bomb():
        if not state["raised"]:
            state["raised"] = True
            raise RuntimeError("injected fault")
        return orig_step()

    eng.step = bomb
    try:
        fut = loop.submit(eng.tokenizer.encode("boom", add_bos=True),
                          SamplingParams(max_new_tokens=3))
        import pytest as _pytest

        with _pytest.raises(RuntimeError, match="injected fault"):
            fut.result(timeout=60)
    finally:
        eng.step = orig_step
    # loop recovers: the next request completes
    out, _ = loop.generate(eng.tokenizer.encode("after fault", add_bos=True),
                           SamplingParams(max_new_tokens=3))
    assert len(out) > 0


def test_decode_interleaves_with_long_prefill():
    """A long prompt must not starve a running decode (chunked-prefill
    interleaving): the short request finishes long before the big prefill."""
    eng = LLMEngine(dict(CFG, max_prefill_chunk=8, prefill_policy="interactive"))
    tok = eng.tokenizer
    short_id = eng.add_request(tok.encode("short", add_bos=True),
                               SamplingParams(max_new_tokens=4))
    # get the short request fully prefilled first
    while eng.requests[short_id].prefill_done < len(eng.requests[short_id].prompt_ids):
        eng.step()
    long_id = eng.add_request(tok.encode("x" * 200, add_bos=True),
                              SamplingParams(max_new_tokens=2))
    steps_until_short_done = 0
    while not eng.requests[short_id].finished and steps_until_short_done < 50:
        eng.step()
        steps_until_short_done += 1
    assert eng.requests[short_id].finished, "short decode starved by long prefill"
    # the long request was NOT finished prefilling when short completed
    long_req = eng.requests[long_id]
    for _ in range(300):
        if long_req.finished:
            break
        eng.step()
    assert long_req.finished
    eng.requests.pop(short_id)
    eng.requests.pop(long_id)


def test_stream_with_grammar_fastforward(loop):
    """Streamed token sequence must equal the final output even when the
    engine appends fast-forwarded (forced) tokens in chunks."""
    from opsagent_amd.engine.grammar import GrammarMode

    tok = loop.engine.tokenizer
    it, fut = loop.submit_stream(
        tok.encode("stream a tool prompt", add_bos=True),
        SamplingParams(max_new_tokens=64, grammar=GrammarMode.TOOLPROMPT),
    )
    streamed = list(it)
    out, reason = fut.result(timeout=120)
    assert streamed == out
    import json as _json

    _json.loads(tok.decode_text(out))


def test_abandoned_stream_does_not_wedge_engine(loop):
    """A client that stops consuming an SSE stream mid-generation must not
    wedge the engine loop: the request still completes (its future resolves)
    and subsequent requests are served."""
    tok = loop.engine.tokenizer
    it, fut = loop.submit_stream(
        tok.encode("abandoned stream", add_bos=True), SamplingParams(max_new_tokens=12)
    )
    next(it)  # consume ONE token, then abandon the iterator
    del it
    out, _ = fut.result(timeout=120)
    assert len(out) > 0
    # engine still serves
    out2, _ = loop.generate(
        tok.encode("after abandonment", add_bos=True), SamplingParams(max_new_tokens=4)
    )
    assert len(out2) > 0


def test_fault_releases_kv_blocks():
    """A step fault must FREE the in-flight requests' KV blocks — recurring
    faults previously leaked the cache dry."""
    eng = LLMEngine(dict(CFG, kv_num_blocks=16))
    lp = EngineLoop(eng)
    try:
        free0 = eng.kv.num_free()
        for round_ in range(3):
            orig_step = eng.step
            state = {"n": 0}

            def bomb():
                state["n"] += 1
                if state["n"] >= 2:  # let admission+prefill start, then fail
                    raise RuntimeError("boom")
                return orig_step()

            eng.step = bomb
            fut = lp.submit(eng.tokenizer.encode("x" * 40, add_bos=True),
                            SamplingParams(max_new_tokens=4))
            with pytest.raises(RuntimeError):
                fut.result(timeout=60)
            eng.step = orig_step
        assert eng.kv.num_free() == free0, "KV blocks leaked across faults"
        out, _ = lp.generate(eng.tokenizer.encode("after", add_bos=True),
                             SamplingParams(max_new_tokens=3))
        assert len(out) > 0
    finally:
        lp.shutdown()
