"""BPE tokenizer support: a trained byte-level BPE loaded from tokenizer.json
drives the same engine interface (encode/decode/token_bytes for the grammar
FSM), with the byte-level-only fast paths switching themselves off."""

import json

import pytest

from opsagent_amd.engine.tokenizer import BPETokenizer, get_tokenizer


@pytest.fixture(scope="module")
def bpe_path(tmp_path_factory):
    from tokenizers import Tokenizer, models, pre_tokenizers, decoders, trainers

    tok = Tokenizer(models.BPE(unk_token=None))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tok.decoder = decoders.ByteLevel()
    trainer = trainers.BpeTrainer(
        vocab_size=384,
        special_tokens=["<|begin_of_text|>", "<|end_of_text|>", "<|eot_id|>"],
        initial_alphabet=pre_tokenizers.ByteLevel.alphabet(),
    )
    corpus = [
        '{"question": "why is the pod failing", "thought": "check events"}',
        "kubectl get pods -n default",
        "the quick brown fox jumps over the lazy dog",
    ] * 20
    tok.train_from_iterator(corpus, trainer)
    p = tmp_path_factory.mktemp("tok") / "tokenizer.json"
    tok.save(str(p))
    return str(p)


def test_bpe_roundtrip_and_bytes(bpe_path):
    t = get_tokenizer(bpe_path)
    assert isinstance(t, BPETokenizer)
    assert t.byte_level_ids is False
    text = 'kubectl get pods {"a": 1}'
    ids = t.encode(text)
    assert t.decode_text(ids) == text
    # token_bytes must reconstruct the exact byte stream
    assert b"".join(t.token_bytes(i) for i in ids) == text.encode()
    # specials resolve
    assert t.bos_id != t.eot_id or t.vocab_size > 0
    assert t.token_bytes(t.bos_id) == b""


def test_bpe_grammar_fsm_compat(bpe_path):
    """The grammar FSM works over multi-byte BPE tokens: simulate a masked
    greedy walk picking the first allowed token each step — the result must
    be valid JSON."""
    from opsagent_amd.engine.grammar import GrammarMode, GrammarState

    t = get_tokenizer(bpe_path)
    gs = GrammarState(t, GrammarMode.JSON, t.vocab_size)
    out = []
    for _ in range(200):
        if gs.is_complete():
            break
        allowed = gs.allowed_bool()
        idx = int(allowed.float().argmax())
        assert allowed[idx], "no token allowed"
        assert gs.accept(idx)
        out.extend(t.token_bytes(idx))
    else:
        comp = gs.completion_bytes()
        assert comp is not None
        out.extend(comp)
    json.loads(bytes(out).decode())


def test_bpe_engine_generation(bpe_path):
    """End-to-end: the engine runs with a real BPE (tiny model, CPU) and
    grammar-constrained output still always parses; jump-ahead stays ON
    (token-aligned forced-byte runs)."""
    from opsagent_amd.engine.engine import LLMEngine, SamplingParams
    from opsagent_amd.engine.grammar import GrammarMode

    eng = LLMEngine(
        {"model": "llama3-tiny", "max_seq_len": 256, "kv_block_size": 16,
         "max_batch_size": 4, "use_hipgraph": False, "seed": 11,
         "tokenizer": bpe_path}
    )
    assert eng.grammar_fastforward is True
    ids = eng.tokenizer.encode("produce json", add_bos=True)
    out, reason = eng.generate(
        ids, SamplingParams(max_new_tokens=120, grammar=GrammarMode.JSON)
    )
    assert reason.startswith("grammar")
    json.loads(eng.tokenizer.decode_text(out))


def test_bpe_forced_peek_nonadvancing(bpe_path):
    """forced_peek returns the unique byte continuation without moving the
    state: peeking twice gives the same bytes, and the allowed mask is
    unchanged."""
    from opsagent_amd.engine.grammar import GrammarMode, GrammarState

    t = get_tokenizer(bpe_path)
    gs = GrammarState(t, GrammarMode.TOOLPROMPT, t.vocab_size)
    before = gs.allowed_bool().clone()
    r1 = gs.forced_peek(64)
    r2 = gs.forced_peek(64)
    assert r1 == r2
    # the ToolPrompt template opens with a forced literal run
    assert r1.startswith(b'{"question"')
    assert (gs.allowed_bool() == before).all()


def test_bpe_jump_ahead_token_aligned(bpe_path):
    """BPE jump-ahead: the engine appends whole BPE tokens
    for forced template literals without model passes; output still parses
    as the exact ToolPrompt schema and forced-run catch-ups were used."""
    from opsagent_amd.engine.engine import LLMEngine, SamplingParams
    from opsagent_amd.engine.grammar import GrammarMode
    from opsagent_amd.utils.perf import get_perf_stats

    eng = LLMEngine(
        {"model": "llama3-tiny", "max_seq_len": 512, "kv_block_size": 16,
         "max_batch_size": 4, "use_hipgraph": False, "seed": 11,
         "tokenizer": bpe_path}
    )
    get_perf_stats().reset()
    ids = eng.tokenizer.encode("analyze the pod", add_bos=True)
    out, reason = eng.generate(
        ids, SamplingParams(max_new_tokens=200, grammar=GrammarMode.TOOLPROMPT)
    )
    assert reason.startswith("grammar")
    obj = json.loads(eng.tokenizer.decode_text(out))
    assert set(obj) == {"question", "thought", "action", "observation",
                        "final_answer"}
    stats = get_perf_stats().get_stats()
    assert "engine_ff_catchup_tokens" in stats, (
        "BPE jump-ahead never fired on a fully templated document"
    )


def test_bpe_without_special_tokens(tmp_path):
    """A tokenizer.json with NO recognized specials still loads: bos/eos
    fall back deterministically and generation interfaces stay usable."""
    from tokenizers import Tokenizer, models, pre_tokenizers, decoders, trainers

    tok = Tokenizer(models.BPE(unk_token=None))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tok.decoder = decoders.ByteLevel()
    trainer = trainers.BpeTrainer(
        vocab_size=300, special_tokens=[],
        initial_alphabet=pre_tokenizers.ByteLevel.alphabet(),
    )
    tok.train_from_iterator(["plain text only"] * 10, trainer)
    p = tmp_path / "plain.json"
    tok.save(str(p))

    t = get_tokenizer(str(p))
    assert isinstance(t.bos_id, int) and isinstance(t.eot_id, int)
    ids = t.encode("plain text", add_bos=True)
    assert t.decode_text(ids[1:]) == "plain text"
    assert isinstance(t.stop_ids, set) and t.stop_ids
