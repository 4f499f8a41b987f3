"""HTTP /v1/chat/completions over the real (tiny) in-process engine, plus
k8s helper tests with a fake kubectl."""

import json
import os
import stat

import pytest
from fastapi.testclient import TestClient

from opsagent_amd.config import Config, DEFAULTS
from opsagent_amd.engine.openai_api import ChatCompletionAPI
from opsagent_amd.server.app import create_app

TINY = {
    "model": "llama3-tiny",
    "max_seq_len": 512,
    "kv_block_size": 16,
    "max_batch_size": 4,
    "use_hipgraph": False,
    "seed": 13,
    "grammar": "auto",
    "open_api": True,  # tests drive /v1 without a token; gate tested below
}


@pytest.fixture(scope="module")
def client():
    ChatCompletionAPI.reset_instance()
    cfg = Config(json.loads(json.dumps(DEFAULTS)))
    cfg._data["engine"] = dict(TINY)
    app = create_app(cfg)
    with TestClient(app) as c:
        yield c
    ChatCompletionAPI.reset_instance()


def test_v1_chat_completions_plain(client):
    r = client.post(
        "/v1/chat/completions",
        json={"model": "llama3-tiny", "messages": [{"role": "user", "content": "hi"}],
              "max_tokens": 8},
    )
    assert r.status_code == 200
    body = r.json()
    assert body["object"] == "chat.completion"
    assert body["usage"]["completion_tokens"] > 0


def test_v1_chat_completions_json_mode(client):
    r = client.post(
        "/v1/chat/completions",
        json={
            "model": "llama3-tiny",
            "messages": [{"role": "user", "content": "produce json"}],
            "max_tokens": 64,
            "response_format": {"type": "json_object"},
        },
    )
    assert r.status_code == 200
    content = r.json()["choices"][0]["message"]["content"]
    json.loads(content)  # grammar-constrained: always parses


def test_v1_chat_completions_tools(client):
    tools = [{"type": "function", "function": {"name": "kubectl", "parameters": {
        "type": "object", "properties": {"command": {"type": "string"}}}}}]
    r = client.post(
        "/v1/chat/completions",
        json={"model": "llama3-tiny", "messages": [{"role": "user", "content": "pods"}],
              "max_tokens": 200, "tools": tools},
    )
    assert r.status_code == 200
    msg = r.json()["choices"][0]["message"]
    if msg.get("tool_calls"):
        json.loads(msg["tool_calls"][0]["function"]["arguments"])


def test_health_reports_engine_ready(client):
    client.post(
        "/v1/chat/completions",
        json={"model": "llama3-tiny", "messages": [{"role": "user", "content": "x"}],
              "max_tokens": 2},
    )
    r = client.get("/api/health")
    assert r.json()["engine"] in ("ready", "unhealthy")


def test_k8s_helpers(tmp_path, monkeypatch):
    from opsagent_amd import k8s

    script = tmp_path / "kubectl"
    script.write_text(
        "#!/bin/bash\n"
        'if [[ "$1" == "get" ]]; then echo "kind: Pod"; exit 0; fi\n'
        'if [[ "$1" == "apply" ]]; then echo "pod/x serverside-applied"; exit 0; fi\n'
        "exit 1\n"
    )
    script.chmod(script.stat().st_mode | stat.S_IEXEC)
    monkeypatch.setenv("PATH", str(tmp_path) + os.pathsep + os.environ["PATH"])
    assert "kind: Pod" in k8s.get_yaml("pod", "x", "default")
    assert "serverside-applied" in k8s.apply_yaml("kind: Pod\n")


def test_temperature_sampling_runs():
    from opsagent_amd.engine.engine import LLMEngine, SamplingParams

    eng = LLMEngine(dict(TINY))
    ids = eng.tokenizer.encode("sample with temperature", add_bos=True)
    out, _ = eng.generate(ids, SamplingParams(max_new_tokens=8, temperature=0.8))
    assert len(out) > 0


def test_streaming_chat_completions(client):
    r = client.post(
        "/v1/chat/completions",
        json={"model": "llama3-tiny", "messages": [{"role": "user", "content": "hi"}],
              "max_tokens": 12, "stream": True},
    )
    assert r.status_code == 200
    assert "text/event-stream" in r.headers["content-type"]
    chunks = [ln for ln in r.text.split("\n\n") if ln.startswith("data: ")]
    assert chunks[-1] == "data: [DONE]"
    parsed = [json.loads(c[len("data: "):]) for c in chunks[:-1]]
    assert parsed[0]["choices"][0]["delta"].get("role") == "assistant"
    assert parsed[-1]["choices"][0]["finish_reason"] in ("stop", "length")
    text = "".join(p["choices"][0]["delta"].get("content", "") for p in parsed)
    assert isinstance(text, str)


def test_engine_stats_endpoint(client):
    from opsagent_amd.server.auth import create_token

    tok = create_token("admin", "novastar-secret-key")
    r = client.get("/api/engine/stats", headers={"Authorization": f"Bearer {tok}"})
    assert r.status_code == 200
    body = r.json()
    assert body["model"] == "llama3-tiny"
    assert "kv" in body and "healthy" in body


def test_v1_completions_endpoint(client):
    r = client.post(
        "/v1/completions",
        json={"model": "llama3-tiny", "prompt": "complete me", "max_tokens": 8},
    )
    assert r.status_code == 200
    data = r.json()
    assert data["object"] == "text_completion"
    assert isinstance(data["choices"][0]["text"], str)
    assert data["usage"]["completion_tokens"] > 0


def test_v1_chat_completions_sampling_params(client):
    """top_p/top_k/penalties accepted over the wire; logit_bias steers."""
    r = client.post(
        "/v1/chat/completions",
        json={
            "model": "llama3-tiny",
            "messages": [{"role": "user", "content": "biased"}],
            "max_tokens": 4,
            "temperature": 0.9,
            "top_p": 0.9,
            "top_k": 10,
            "presence_penalty": 0.1,
            "logit_bias": {"65": 1000.0},
        },
    )
    assert r.status_code == 200
    content = r.json()["choices"][0]["message"]["content"]
    assert set(content) == {"A"}  # byte 65 forced by the bias


def test_stream_with_tools_emits_complete_call(client):
    """stream + tools: one complete chunk with a parseable tool_calls delta
    (the grammar guarantees whole-document validity — no partial deltas)."""
    tools = [{"type": "function", "function": {
        "name": "kubectl",
        "parameters": {"type": "object", "properties": {"command": {"type": "string"}}}}}]
    with client.stream(
        "POST", "/v1/chat/completions",
        json={"model": "llama3-tiny", "messages": [{"role": "user", "content": "list pods"}],
              "max_tokens": 200, "tools": tools, "stream": True},
    ) as r:
        assert r.status_code == 200
        body = "".join(r.iter_text())
    chunks = [json.loads(l[6:]) for l in body.splitlines()
              if l.startswith("data: ") and l != "data: [DONE]"]
    assert chunks, body[:200]
    delta = chunks[0]["choices"][0]["delta"]
    if delta.get("tool_calls"):
        json.loads(delta["tool_calls"][0]["function"]["arguments"])


def test_metrics_prometheus_format(client):
    """/metrics emits parseable Prometheus summary lines."""
    import re

    r = client.get("/metrics")
    assert r.status_code == 200
    lines = [l for l in r.text.splitlines() if l and not l.startswith("#")]
    pat = re.compile(r'^opsagent_operation_ms(?:_count)?\{[^}]*\} [-0-9.eE]+$')
    assert lines, "no metric samples"
    for l in lines:
        assert pat.match(l), l


def test_engine_stats_exposes_spec_decode(client):
    tok = client.post(
        "/login", json={"username": "admin", "password": "novastar"}
    ).json()["token"]
    r = client.get("/api/engine/stats", headers={"Authorization": f"Bearer {tok}"})
    assert r.status_code == 200
    st = r.json()
    assert "spec_decode" in st and "ema" in st["spec_decode"]
    assert "kv" in st and "healthy" in st
