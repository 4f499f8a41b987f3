"""End-to-end: the ReAct agent driven by the in-process engine (tiny model,
CPU). With the ToolPrompt grammar active, EVERY assistant turn must parse as
the exact ToolPrompt schema — the tool-call JSON validity metric is 100% by
construction even with random weights.
"""

import json

import pytest

from opsagent_amd.agent import prompts, react
from opsagent_amd.engine.openai_api import ChatCompletionAPI
from opsagent_amd.llm.client import LocalEngineClient
from opsagent_amd.tools import TOOLS

TINY_CFG = {
    "model": "llama3-tiny",
    "max_seq_len": 512,
    "kv_block_size": 16,
    "max_batch_size": 4,
    "use_hipgraph": False,
    "seed": 11,
    "grammar": "auto",
}


@pytest.fixture(scope="module")
def client():
    ChatCompletionAPI.reset_instance()
    c = LocalEngineClient(dict(TINY_CFG))
    yield c
    ChatCompletionAPI.reset_instance()


def test_react_loop_with_local_engine(monkeypatch, client):
    monkeypatch.setitem(TOOLS, "kubectl", lambda s: "default\nkube-system")
    messages = [
        {"role": "system", "content": prompts.execute_system_prompt(TOOLS.keys())},
        {"role": "user", "content": "count namespaces"},
    ]
    result, history = react.assistant(
        client, "llama3-tiny", messages, max_tokens=150, max_iterations=2
    )
    assert isinstance(result, str)
    # every assistant message the engine produced must parse as ToolPrompt JSON
    assistant_msgs = [m for m in history if m["role"] == "assistant"]
    assert assistant_msgs
    n_valid = 0
    for m in assistant_msgs:
        obj = json.loads(m["content"])  # must not raise — grammar-constrained
        assert set(obj) == {"question", "thought", "action", "observation", "final_answer"}
        n_valid += 1
    assert n_valid == len(assistant_msgs)  # 100% validity


def test_json_validity_metric_is_structural(client):
    """Direct chat with the ToolPrompt-demanding system prompt: reply parses."""
    reply = client.chat(
        "llama3-tiny",
        120,
        [
            {"role": "system", "content": prompts.execute_system_prompt(["kubectl"])},
            {"role": "user", "content": "anything"},
        ],
    )
    obj = json.loads(reply["content"])
    assert "final_answer" in obj
