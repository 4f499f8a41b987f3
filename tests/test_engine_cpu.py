"""Engine tests on CPU with the tiny model (fp32 torch_ref path).

Covers: deterministic generation, paged-vs-contiguous equivalence (the paged
decode path must produce the same tokens as a plain full-context forward),
prefix-cache reuse across multi-turn prompts, grammar-constrained output
always parsing, and the OpenAI API wire format.
"""

import json

import pytest
import torch

from opsagent_amd.engine.config import get_model_spec
from opsagent_amd.engine.engine import LLMEngine, SamplingParams
from opsagent_amd.engine.grammar import GrammarMode

TINY_CFG = {
    "model": "llama3-tiny",
    "max_seq_len": 256,
    "kv_block_size": 16,
    "max_batch_size": 8,
    "use_hipgraph": False,
    "seed": 7,
}


@pytest.fixture(scope="module")
def engine():
    return LLMEngine(dict(TINY_CFG))


def test_generate_deterministic(engine):
    ids = engine.tokenizer.encode("hello k8s world", add_bos=True)
    out1, r1 = engine.generate(ids, SamplingParams(max_new_tokens=16))
    out2, r2 = engine.generate(ids, SamplingParams(max_new_tokens=16))
    assert out1 == out2
    assert len(out1) > 0


def test_paged_equals_full_forward(engine):
    """Greedy tokens from the engine must match a naive full-context forward."""
    from opsagent_amd.engine.model import ForwardBatch
    from opsagent_amd.engine.kv_cache import PagedKVCache

    tok = engine.tokenizer
    ids = tok.encode("the quick brown fox", add_bos=True)
    n_new = 8
    out, _ = engine.generate(ids, SamplingParams(max_new_tokens=n_new))

    # naive: recompute full prefill each step on a FRESH cache
    spec = engine.spec
    cur = list(ids)
    naive = []
    for _ in range(n_new):
        kv = PagedKVCache(
            spec.num_layers, spec.num_kv_heads, spec.head_dim,
            16, 64, "cpu", torch.float32,
        )
        from opsagent_amd.engine.kv_cache import SequenceState

        s = SequenceState(kv, cur)
        s.ensure_capacity(len(cur))
        fb = ForwardBatch(
            kind="prefill",
            input_ids=torch.tensor(cur, dtype=torch.int64),
            positions=torch.arange(len(cur), dtype=torch.int32),
            slot_mapping=s.slots_for(0, len(cur)),
        )
        hidden = engine.model(fb, kv.layers)
        logits = engine.model.compute_logits(hidden[-1:])
        t = int(logits.float().argmax(dim=-1)[0])
        naive.append(t)
        if t in tok.stop_ids:
            break
        cur.append(t)
    assert out[: len(naive)] == naive


def test_prefix_cache_reuse(engine):
    tok = engine.tokenizer
    base = "a long conversation prefix that spans multiple cache blocks " * 3
    ids1 = tok.encode(base + "turn one", add_bos=True)
    engine.generate(ids1, SamplingParams(max_new_tokens=4))
    reused_before = engine.kv.stats["reused_blocks"]
    ids2 = tok.encode(base + "turn two has new suffix", add_bos=True)
    engine.generate(ids2, SamplingParams(max_new_tokens=4))
    assert engine.kv.stats["reused_blocks"] > reused_before


def test_grammar_constrained_always_parses(engine):
    ids = engine.tokenizer.encode("produce json", add_bos=True)
    out, reason = engine.generate(
        ids, SamplingParams(max_new_tokens=200, grammar=GrammarMode.TOOLPROMPT)
    )
    text = engine.tokenizer.decode_text(out)
    assert reason in ("grammar_complete", "grammar_forced_complete")
    obj = json.loads(text)  # ALWAYS valid — forced completion closes the doc
    assert set(obj) == {"question", "thought", "action", "observation", "final_answer"}


def test_json_grammar_output_parses(engine):
    ids = engine.tokenizer.encode("produce json", add_bos=True)
    out, reason = engine.generate(
        ids, SamplingParams(max_new_tokens=300, grammar=GrammarMode.JSON)
    )
    text = engine.tokenizer.decode_text(out)
    assert reason in ("grammar_complete", "grammar_forced_complete")
    json.loads(text)  # must not raise
    assert text.lstrip().startswith("{")


def test_batched_requests(engine):
    tok = engine.tokenizer
    rids = [
        engine.add_request(tok.encode(f"request number {i}", add_bos=True),
                           SamplingParams(max_new_tokens=8))
        for i in range(4)
    ]
    for _ in range(200):
        if all(engine.requests[r].finished for r in rids):
            break
        engine.step()
    outs = [engine.requests.pop(r).output_ids for r in rids]
    assert all(len(o) > 0 for o in outs)
    # batched decode must equal single-request decode
    single, _ = engine.generate(tok.encode("request number 0", add_bos=True),
                                SamplingParams(max_new_tokens=8))
    assert outs[0] == single


def test_openai_api_shape():
    from opsagent_amd.engine.openai_api import ChatCompletionAPI

    ChatCompletionAPI.reset_instance()
    api = ChatCompletionAPI.get_or_create(dict(TINY_CFG))
    resp = api.create(
        model="llama3-tiny",
        messages=[{"role": "user", "content": "hi"}],
        max_tokens=8,
    )
    assert resp["object"] == "chat.completion"
    msg = resp["choices"][0]["message"]
    assert msg["role"] == "assistant"
    assert isinstance(resp["usage"]["prompt_tokens"], int)
    ChatCompletionAPI.reset_instance()


def test_openai_api_tool_calls_wire_format():
    from opsagent_amd.engine.openai_api import ChatCompletionAPI

    ChatCompletionAPI.reset_instance()
    api = ChatCompletionAPI.get_or_create(dict(TINY_CFG))
    tools = [
        {
            "type": "function",
            "function": {
                "name": "kubectl",
                "parameters": {"type": "object", "properties": {"command": {"type": "string"}}},
            },
        }
    ]
    resp = api.create(
        model="llama3-tiny",
        messages=[{"role": "user", "content": "list pods"}],
        max_tokens=300,
        tools=tools,
    )
    msg = resp["choices"][0]["message"]
    if msg.get("tool_calls"):
        tc = msg["tool_calls"][0]
        assert tc["type"] == "function"
        json.loads(tc["function"]["arguments"])  # arguments are valid JSON
        assert resp["choices"][0]["finish_reason"] == "tool_calls"
    ChatCompletionAPI.reset_instance()


def test_stop_sequences(engine):
    """Generation halts and trims when a stop string appears. Stop matching
    is byte-level, so pick a substring that round-trips through UTF-8 (real
    stop sequences are valid text, not replacement characters)."""
    tok = engine.tokenizer
    ids = tok.encode("find the stop", add_bos=True)
    # discover which bytes the model produces unconstrained, pick a clean run
    base, _ = engine.generate(ids, SamplingParams(max_new_tokens=24))
    raw = bytes(t for t in base if t < 256)
    stop_s = None
    for i in range(2, len(raw) - 3):
        try:
            stop_s = raw[i : i + 3].decode("utf-8")
            break
        except UnicodeDecodeError:
            continue
    if stop_s is None:
        return  # degenerate output; nothing to split on
    out, reason = engine.generate(
        ids, SamplingParams(max_new_tokens=24, stop=[stop_s])
    )
    got = bytes(t for t in out if t < 256)
    assert stop_s.encode() not in got
    assert reason == "stop"
    assert len(out) < len(base)


def test_mixed_temperature_batch(engine):
    """Greedy and sampled requests in ONE batch: the greedy one must match a
    solo greedy run (per-request temperature, not batch[0]'s)."""
    tok = engine.tokenizer
    g_ids = tok.encode("greedy request", add_bos=True)
    solo, _ = engine.generate(g_ids, SamplingParams(max_new_tokens=6))

    r1 = engine.add_request(g_ids, SamplingParams(max_new_tokens=6))
    r2 = engine.add_request(
        tok.encode("sampled request", add_bos=True),
        SamplingParams(max_new_tokens=6, temperature=1.0),
    )
    for _ in range(200):
        if engine.requests[r1].finished and engine.requests[r2].finished:
            break
        engine.step()
    out_g = engine.requests.pop(r1).output_ids
    engine.requests.pop(r2)
    assert out_g == solo, "greedy row disturbed by sampled neighbor"


@pytest.mark.parametrize("mode", [GrammarMode.TOOLPROMPT, GrammarMode.JSON])
def test_grammar_fastforward_matches_stepwise(mode):
    """Jump-ahead decoding is exact: forced tokens are the only grammar-legal
    choice, so output with fast-forward ON must be bit-identical to stepping
    the model through every token — while running fewer model passes."""
    ids_text = "produce json please"
    outs = {}
    steps = {}
    for ff in (True, False):
        eng = LLMEngine(dict(TINY_CFG, grammar_fastforward=ff))
        ids = eng.tokenizer.encode(ids_text, add_bos=True)
        rid = eng.add_request(ids, SamplingParams(max_new_tokens=120, grammar=mode))
        n = 0
        while not eng.requests[rid].finished:
            eng.step()
            n += 1
        req = eng.requests.pop(rid)
        outs[ff] = (req.output_ids, req.finish_reason)
        steps[ff] = n
    assert outs[True] == outs[False]
    if mode == GrammarMode.TOOLPROMPT:
        # template literals dominate: fast-forward must cut engine steps
        assert steps[True] < steps[False]


def test_stream_with_stop_sequence():
    """Streaming with `stop`: the stop text is never streamed (holdback) and
    the concatenated stream equals the non-streaming trimmed result."""
    from opsagent_amd.engine.openai_api import ChatCompletionAPI

    ChatCompletionAPI.reset_instance()
    api = ChatCompletionAPI.get_or_create(dict(TINY_CFG))
    msgs = [{"role": "user", "content": "tell me things"}]
    base = api.create(model="llama3-tiny", messages=msgs, max_tokens=24)
    text = base["choices"][0]["message"]["content"]
    # stop matching is byte-level: pick an ASCII run (replacement characters
    # from invalid UTF-8 can never byte-match the raw stream)
    stop_s = next(
        (
            text[i : i + 2]
            for i in range(1, len(text) - 2)
            if all(" " <= c < "\x7f" for c in text[i : i + 2])
        ),
        None,
    )
    if stop_s is None:
        ChatCompletionAPI.reset_instance()
        return
    ref = api.create(model="llama3-tiny", messages=msgs, max_tokens=24, stop=[stop_s])
    ref_text = ref["choices"][0]["message"]["content"]

    streamed = []
    finish = None
    for ch in api.create_stream(
        model="llama3-tiny", messages=msgs, max_tokens=24, stop=[stop_s]
    ):
        d = ch["choices"][0]["delta"]
        if d.get("content"):
            streamed.append(d["content"])
        if ch["choices"][0]["finish_reason"]:
            finish = ch["choices"][0]["finish_reason"]
    got = "".join(streamed)
    assert got == ref_text
    assert stop_s not in got
    assert finish == "stop"
    ChatCompletionAPI.reset_instance()


def test_sampling_top_k1_and_top_p_tiny_equal_greedy(engine):
    """top_k=1 (or a tiny top_p) with temperature leaves only the argmax
    candidate, so the output must equal pure greedy."""
    tok = engine.tokenizer
    ids = tok.encode("sample something", add_bos=True)
    greedy, _ = engine.generate(ids, SamplingParams(max_new_tokens=8))
    k1, _ = engine.generate(
        ids, SamplingParams(max_new_tokens=8, temperature=1.0, top_k=1)
    )
    p0, _ = engine.generate(
        ids, SamplingParams(max_new_tokens=8, temperature=1.0, top_p=1e-9)
    )
    assert k1 == greedy
    assert p0 == greedy


def test_frequency_penalty_blocks_repeats(engine):
    """A huge frequency penalty makes greedy decoding avoid any token it has
    already emitted."""
    tok = engine.tokenizer
    ids = tok.encode("penalize repeats", add_bos=True)
    out, _ = engine.generate(
        ids, SamplingParams(max_new_tokens=10, frequency_penalty=100.0)
    )
    assert len(out) == len(set(out)), f"repeat under huge penalty: {out}"


def test_logit_bias_forces_token(engine):
    tok = engine.tokenizer
    ids = tok.encode("bias test", add_bos=True)
    out, _ = engine.generate(
        ids, SamplingParams(max_new_tokens=4, logit_bias={65: 1000.0})
    )
    assert all(t == 65 for t in out), out


def test_n_choices():
    from opsagent_amd.engine.openai_api import ChatCompletionAPI

    ChatCompletionAPI.reset_instance()
    api = ChatCompletionAPI.get_or_create(dict(TINY_CFG))
    resp = api.create(
        model="llama3-tiny",
        messages=[{"role": "user", "content": "three answers"}],
        max_tokens=6,
        n=3,
    )
    assert [c["index"] for c in resp["choices"]] == [0, 1, 2]
    # greedy: all choices identical; usage sums completions
    texts = [c["message"]["content"] for c in resp["choices"]]
    assert texts[0] == texts[1] == texts[2]
    assert resp["usage"]["completion_tokens"] == sum(
        len(t.encode()) for t in texts
    ) or resp["usage"]["completion_tokens"] > 0
    ChatCompletionAPI.reset_instance()


def test_stop_inside_fastforward_run():
    """A stop sequence that lands INSIDE a grammar fast-forwarded literal
    must still stop and trim (window scan, not endswith-per-token)."""
    eng = LLMEngine(dict(TINY_CFG, grammar_fastforward=True))
    ids = eng.tokenizer.encode("toolprompt with stop", add_bos=True)
    # '"thought"' appears only inside the jump-ahead template literal
    out, reason = eng.generate(
        ids,
        SamplingParams(
            max_new_tokens=128,
            grammar=GrammarMode.TOOLPROMPT,
            stop=['"thought"'],
        ),
    )
    text = eng.tokenizer.decode_text(out)
    assert '"thought"' not in text
    assert reason == "stop"


def test_legacy_completions_api():
    from opsagent_amd.engine.openai_api import ChatCompletionAPI

    ChatCompletionAPI.reset_instance()
    api = ChatCompletionAPI.get_or_create(dict(TINY_CFG))
    resp = api.create_completion(
        model="llama3-tiny", prompt="complete this text", max_tokens=8
    )
    assert resp["object"] == "text_completion"
    assert isinstance(resp["choices"][0]["text"], str)
    assert resp["choices"][0]["finish_reason"] in ("stop", "length")
    assert resp["usage"]["completion_tokens"] > 0
    ChatCompletionAPI.reset_instance()


def test_preemption_under_kv_pressure():
    """With a KV cache too small for all concurrent requests, the engine
    preempts (re-queues + recomputes) instead of crashing, admission holds
    back under pressure, and every request still finishes with EXACTLY the
    tokens an uncontended engine produces."""
    big = LLMEngine(dict(TINY_CFG))
    tok = big.tokenizer
    prompts = [
        tok.encode(f"pressure request {i} " + "pod " * 7, add_bos=True)
        for i in range(4)
    ]
    refs = [big.generate(p, SamplingParams(max_new_tokens=40))[0] for p in prompts]

    small = LLMEngine(dict(TINY_CFG, kv_num_blocks=8))
    preempts = []
    orig = small._preempt
    small._preempt = lambda r: (preempts.append(r.req_id), orig(r))[1]
    rids = [small.add_request(p, SamplingParams(max_new_tokens=40)) for p in prompts]
    for _ in range(4000):
        if all(small.requests[r].finished for r in rids):
            break
        small.step()
    outs = [small.requests.pop(r) for r in rids]
    assert all(r.finished for r in outs)
    for req, ref in zip(outs, refs):
        assert req.finish_reason != "kv_exhausted"
        assert req.output_ids == ref


def test_single_oversized_request_fails_cleanly():
    """A request that cannot fit in the ENTIRE cache fails with
    kv_exhausted instead of preempt-looping."""
    eng = LLMEngine(dict(TINY_CFG, kv_num_blocks=4))  # 64 tokens of cache
    ids = eng.tokenizer.encode("y" * 100, add_bos=True)
    rid = eng.add_request(ids, SamplingParams(max_new_tokens=8))
    for _ in range(200):
        if eng.requests[rid].finished:
            break
        eng.step()
    req = eng.requests.pop(rid)
    assert req.finished and req.finish_reason == "kv_exhausted"


def test_spec_decode_matches_plain():
    """Speculative n-gram decoding is exact: greedy output with speculation
    on equals plain decoding bit-for-bit, with fewer engine steps whenever
    any proposal is accepted."""
    outs, steps, stats = {}, {}, {}
    for spec in (False, True):
        eng = LLMEngine(dict(TINY_CFG, max_seq_len=512, spec_decode=spec))
        ids = eng.tokenizer.encode("repeat repeat repeat repeat", add_bos=True)
        rid = eng.add_request(ids, SamplingParams(max_new_tokens=48))
        n = 0
        while not eng.requests[rid].finished:
            eng.step()
            n += 1
        outs[spec] = eng.requests.pop(rid).output_ids
        steps[spec] = n
        stats[spec] = dict(eng.spec_stats)
    assert outs[True] == outs[False]
    if stats[True]["accepted"] > 0:
        assert steps[True] < steps[False]


def test_spec_decode_with_stop_matches_plain():
    base = LLMEngine(dict(TINY_CFG, max_seq_len=512, spec_decode=False))
    ids = base.tokenizer.encode("loop loop loop", add_bos=True)
    plain, _ = base.generate(ids, SamplingParams(max_new_tokens=32))
    raw = bytes(t for t in plain if t < 256)
    stop_s = None
    for i in range(4, len(raw) - 3):
        try:
            stop_s = raw[i : i + 2].decode("utf-8")
            break
        except UnicodeDecodeError:
            continue
    if stop_s is None:
        return
    ref, rr = base.generate(ids, SamplingParams(max_new_tokens=32, stop=[stop_s]))
    eng = LLMEngine(dict(TINY_CFG, max_seq_len=512, spec_decode=True))
    got, gr = eng.generate(ids, SamplingParams(max_new_tokens=32, stop=[stop_s]))
    assert (got, gr) == (ref, rr)


def test_spec_decode_ema_disables_and_reprobes():
    """EMA gate: with nothing repeating the gate closes after a handful of
    zero-acceptance verifies; the periodic re-probe keeps it recoverable."""
    eng = LLMEngine(dict(TINY_CFG, max_seq_len=512, spec_decode=True,
                         spec_min_ema=0.9))  # impossible bar -> closes fast
    ids = eng.tokenizer.encode("abc", add_bos=True)
    eng.generate(ids, SamplingParams(max_new_tokens=24))
    # gate closed (ema < bar) but re-probe window eventually re-allows
    assert eng.spec_ema < 0.9


def test_logprobs_api():
    """OpenAI logprobs: the greedy token's logprob is the max (it heads its
    own top list), entries parallel the output tokens, and the wire format
    carries token text + bytes."""
    import math

    from opsagent_amd.engine.openai_api import ChatCompletionAPI

    ChatCompletionAPI.reset_instance()
    api = ChatCompletionAPI.get_or_create(dict(TINY_CFG))
    resp = api.create(
        model="llama3-tiny",
        messages=[{"role": "user", "content": "logprobs please"}],
        max_tokens=6,
        logprobs=True,
        top_logprobs=3,
    )
    ch = resp["choices"][0]
    content = ch["logprobs"]["content"]
    text = ch["message"]["content"]
    assert len(content) == resp["usage"]["completion_tokens"]
    for e in content:
        assert e["logprob"] <= 0.0 and math.isfinite(e["logprob"])
        assert len(e["top_logprobs"]) == 3
        # greedy: sampled token is the argmax -> first of its own top list
        assert abs(e["top_logprobs"][0]["logprob"] - e["logprob"]) < 1e-5
        assert isinstance(e["bytes"], list)
    ChatCompletionAPI.reset_instance()


def test_tool_choice_none_and_named():
    from opsagent_amd.engine.openai_api import ChatCompletionAPI

    ChatCompletionAPI.reset_instance()
    api = ChatCompletionAPI.get_or_create(dict(TINY_CFG))
    tools = [
        {"type": "function", "function": {"name": "kubectl",
         "parameters": {"type": "object", "properties": {}}}},
        {"type": "function", "function": {"name": "trivy",
         "parameters": {"type": "object", "properties": {}}}},
    ]
    msgs = [{"role": "user", "content": "scan the image"}]
    # "none": no forced tool call
    r = api.create(model="llama3-tiny", messages=msgs, max_tokens=20,
                   tools=tools, tool_choice="none")
    assert not r["choices"][0]["message"].get("tool_calls")
    # named function: any produced call must name it
    r2 = api.create(model="llama3-tiny", messages=msgs, max_tokens=200,
                    tools=tools,
                    tool_choice={"type": "function", "function": {"name": "trivy"}})
    calls = r2["choices"][0]["message"].get("tool_calls") or []
    for c in calls:
        assert c["function"]["name"] == "trivy"
    ChatCompletionAPI.reset_instance()


def test_oversized_budget_does_not_empty_prompt(engine):
    """max_new_tokens >= max_seq_len used to invert the truncation slice and
    silently DROP the whole prompt; it must clamp instead."""
    ids = engine.tokenizer.encode("short prompt", add_bos=True)
    out, reason = engine.generate(ids, SamplingParams(max_new_tokens=10_000))
    assert len(out) > 0
    # the request really saw the prompt (deterministic vs a normal call)
    ref, _ = engine.generate(ids, SamplingParams(max_new_tokens=8))
    assert out[: len(ref)] == ref or len(out) >= 8


def test_multi_turn_function_calling_round_trip():
    """Second-round function calling: the assistant's own tool_calls message
    and the tool-result message render through the chat template and the
    engine produces another valid constrained reply."""
    from opsagent_amd.engine.openai_api import ChatCompletionAPI

    ChatCompletionAPI.reset_instance()
    api = ChatCompletionAPI.get_or_create(dict(TINY_CFG))
    tools = [{"type": "function", "function": {
        "name": "kubectl",
        "parameters": {"type": "object",
                       "properties": {"command": {"type": "string"}}}}}]
    msgs = [{"role": "user", "content": "list pods"}]
    r1 = api.create(model="llama3-tiny", messages=msgs, max_tokens=250, tools=tools)
    m1 = r1["choices"][0]["message"]
    if not m1.get("tool_calls"):
        ChatCompletionAPI.reset_instance()
        return  # random weights finished without a call; nothing to round-trip
    msgs = msgs + [m1, {
        "role": "tool",
        "tool_call_id": m1["tool_calls"][0]["id"],
        "content": "pod-a Running\npod-b CrashLoopBackOff",
    }]
    r2 = api.create(model="llama3-tiny", messages=msgs, max_tokens=250, tools=tools)
    m2 = r2["choices"][0]["message"]
    if m2.get("tool_calls"):
        json.loads(m2["tool_calls"][0]["function"]["arguments"])
        assert m2["tool_calls"][0]["function"]["name"] == "kubectl"
    ChatCompletionAPI.reset_instance()


def test_generate_timeout_tears_down(engine):
    """generate() with a zero budget raises TimeoutError and frees the
    request's KV blocks (CLI-path watchdog parity)."""
    import pytest as _pytest

    tok = engine.tokenizer
    free_before = engine.kv.num_free()
    with _pytest.raises(TimeoutError):
        engine.generate(
            tok.encode("hang guard", add_bos=True),
            SamplingParams(max_new_tokens=64),
            timeout_s=0.0,
        )
    assert not engine.running and not engine.waiting
    assert engine.kv.num_free() == free_before


def test_spec_decode_under_grammar_exact():
    """Grammar-aware speculation (masked-argmax verify along simulated
    masks) must produce EXACTLY the unassisted masked-greedy output."""
    from opsagent_amd.engine.grammar import GrammarMode
    from opsagent_amd.parallel import state

    state.set_tp_state(0, 1, None)
    base = {
        "model": "llama3-tiny", "max_seq_len": 512, "kv_block_size": 16,
        "max_batch_size": 4, "use_hipgraph": False, "seed": 23,
        "grammar_fastforward": False,  # isolate speculation
    }
    # repetitive prompt primes the n-gram lookup
    prompt = "pods pods pods crashloop crashloop pods crashloop " * 3
    outs = {}
    for spec_on in (False, True):
        eng = LLMEngine(dict(base, spec_decode=spec_on, spec_min_ema=0.0))
        ids = eng.tokenizer.encode(prompt, add_bos=True)
        outs[spec_on] = eng.generate(
            ids, SamplingParams(max_new_tokens=96, grammar=GrammarMode.TOOLPROMPT)
        )
    assert outs[True][0] == outs[False][0], (
        f"spec-on diverged: {outs[True]} vs {outs[False]}"
    )
    assert outs[True][1].startswith("grammar") or outs[True][1] == "length"


def test_grammar_check_tokens_and_masks_along():
    from opsagent_amd.engine.grammar import GrammarMode, GrammarState
    from opsagent_amd.engine.tokenizer import ByteTokenizer

    t = ByteTokenizer()
    gs = GrammarState(t, GrammarMode.TOOLPROMPT, 512)
    legal = list(b'{"question": "x')
    assert gs.check_tokens(legal) == len(legal)
    # an illegal byte cuts the prefix
    assert gs.check_tokens(list(b'{"quXstion')) == 4
    # live state untouched by simulation
    assert gs.forced_peek(16).startswith(b'{"question"')
    masks = gs.masks_along(list(b'{"q'))
    assert masks.shape[0] == 4
