from opsagent_amd.config import Config, DEFAULTS, get_global, load_config, set_global
from opsagent_amd.llm.tokens import constrict_messages, constrict_prompt, get_token_limits


def test_defaults_load(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    cfg = load_config()
    assert cfg.get("server.port") == 8080
    assert cfg.get("engine.model") == "llama3-8b"
    assert cfg.get("engine.tp") == 1


def test_yaml_override(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    (tmp_path / "config.yaml").write_text("server:\n  port: 9999\nengine:\n  tp: 4\n")
    cfg = load_config()
    assert cfg.get("server.port") == 9999
    assert cfg.get("engine.tp") == 4
    # untouched keys keep defaults
    assert cfg.get("jwt.expire") == 24


def test_env_override(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    monkeypatch.setenv("OPSAGENT_SERVER_PORT", "7070")
    monkeypatch.setenv("OPSAGENT_PERF_ENABLED", "false")
    cfg = load_config()
    assert cfg.get("server.port") == 7070
    assert cfg.get("perf.enabled") is False


def test_set_and_section():
    cfg = Config({"a": {"b": 1}})
    cfg.set("a.c", 2)
    assert cfg.get("a.c") == 2
    assert cfg.section("a") == {"b": 1, "c": 2}
    assert cfg.section("missing") == {}


def test_global_store():
    set_global("k", 42)
    assert get_global("k") == 42
    assert get_global("missing", "d") == "d"


def test_token_limits():
    assert get_token_limits("llama3-8b") == 8192
    assert get_token_limits("unknown-model") == 4096


def test_constrict_prompt_drops_leading_lines():
    prompt = "\n".join(f"line {i} with some padding text here" for i in range(600))
    out = constrict_prompt(prompt, "llama3-8b", 100)
    assert "line 599" in out
    assert "line 0 " not in out
    assert len(out) < len(prompt)


def test_constrict_prompt_single_line():
    out = constrict_prompt("x" * 100000, "llama3-8b", 10)
    assert len(out) <= 10 * 4


def test_constrict_messages_drops_oldest_nonsystem():
    msgs = [{"role": "system", "content": "sys"}] + [
        {"role": "user", "content": "filler " * 500} for _ in range(10)
    ]
    out = constrict_messages(msgs, "llama3-8b", 7000)
    assert out[0]["role"] == "system"
    assert len(out) < len(msgs)


def test_logging_file_rotation(tmp_path, monkeypatch):
    import logging

    from opsagent_amd.utils.logging import get_logger, init_logging

    monkeypatch.chdir(tmp_path)
    init_logging(level="info", fmt="json", output="both", log_dir=str(tmp_path / "logs"))
    lg = get_logger("rotation-test")
    lg.info("hello file sink")
    for h in logging.getLogger("opsagent").handlers:
        h.flush()
    import datetime
    import glob
    import json as _json

    files = glob.glob(str(tmp_path / "logs" / "opsagent-*.log"))
    assert files, "daily log file created"
    line = open(files[0]).readline()
    rec = _json.loads(line)
    assert rec["msg"] == "hello file sink"
    assert datetime.date.today().strftime("%Y%m%d") in files[0]
    init_logging()  # restore default stderr-only config


def test_remote_model_counting_uses_bundled_bpe():
    """Remote models count with the bundled 32k BPE (subword counts), not
    raw bytes — parity in spirit with the reference's per-model tiktoken
    counting (ref tokens.go:60-107)."""
    from opsagent_amd.llm.tokens import count_text_tokens

    text = ("the pod is failing because the container keeps restarting "
            "check resource limits and requests then inspect events") * 4
    local = count_text_tokens(text, "llama3-8b")
    remote = count_text_tokens(text, "gpt-4")
    # byte-level counts ~= len(text); BPE merges words -> far fewer tokens
    assert local == len(text)
    assert remote < local / 2
    # and better than the old ~4 chars/token estimate on this prose
    assert remote != max(1, len(text) // 4)


def test_remote_model_counting_in_constrict():
    from opsagent_amd.llm.tokens import constrict_prompt, count_text_tokens

    prompt = "\n".join("observation line with words here" for _ in range(200))
    out = constrict_prompt(prompt, "gpt-4", 64)
    assert count_text_tokens(out, "gpt-4") <= 64
    assert len(out) > 0


class TestKernelDispatchGates:
    """The Python mirrors of the C dispatch eligibility (measured win
    regions) — lock the policy so a refactor can't silently widen it."""

    def test_bf16_mfma_gate(self):
        from opsagent_amd.ops import _bf16_mfma_ok
        assert _bf16_mfma_ok(4, 4096)            # 8B-class win region
        assert _bf16_mfma_ok(8, 4096)
        assert not _bf16_mfma_ok(2, 4096)        # VALU keeps M<=2
        assert not _bf16_mfma_ok(16, 4096)       # engine A/B: M>8 loses
        assert not _bf16_mfma_ok(4, 8192)        # deep K: hipBLASLt wins
        assert not _bf16_mfma_ok(4, 1000)        # K % 1024
        assert _bf16_mfma_ok(4, 2048, gateup=True)

    def test_fp8_mfma_gate(self):
        from opsagent_amd.ops import _fp8_mfma_ok
        assert _fp8_mfma_ok(2, 4096)
        assert _fp8_mfma_ok(16, 4096)            # fp8 keeps the wide-M range
        assert _fp8_mfma_ok(16, 8192)            # 131072-byte image boundary
        assert not _fp8_mfma_ok(1, 4096)         # M=1: RW2 VALU stream wins
        assert not _fp8_mfma_ok(17, 4096)
        assert not _fp8_mfma_ok(4, 768)          # (K/512) % 8
        assert _fp8_mfma_ok(4, 2048, gateup=True)
