"""Function-calling workflow tests with a scripted LLM emitting tool_calls
(the native replacement for the reference's swarm-go flows)."""

import json

import pytest

from opsagent_amd.agent.workflows import (
    TOOL_SCHEMAS,
    analysis_flow,
    assistant_flow,
    audit_flow,
    generator_flow,
    run_tool_flow,
)
from opsagent_amd.llm.client import ScriptedLLM
from opsagent_amd.tools import TOOLS


def tc(name, args, cid="call_1"):
    return {
        "id": cid,
        "type": "function",
        "function": {"name": name, "arguments": json.dumps(args)},
    }


def test_tool_flow_dispatch_and_result(monkeypatch):
    calls = []
    monkeypatch.setitem(TOOLS, "kubectl", lambda s: calls.append(s) or "3 pods running")
    llm = ScriptedLLM(
        [
            {"role": "assistant", "content": None,
             "tool_calls": [tc("kubectl", {"command": "get pods"})]},
            "All 3 pods are running fine.",
        ]
    )
    out = run_tool_flow(llm, "stub", "system", "check pods", tool_names=["kubectl"])
    assert out == "All 3 pods are running fine."
    assert calls == ["get pods"]
    # tool result was delivered back as a tool-role message
    msgs = llm.calls[-1]["messages"]
    assert any(m.get("role") == "tool" and "3 pods running" in m["content"] for m in msgs)
    # the schemas were sent on the first call
    assert llm.calls[0]["tools"] == [TOOL_SCHEMAS["kubectl"]]


def test_tool_flow_unknown_tool_and_bad_args(monkeypatch):
    llm = ScriptedLLM(
        [
            {"role": "assistant", "content": None,
             "tool_calls": [tc("nope", {}), {"id": "c2", "type": "function",
                                            "function": {"name": "kubectl",
                                                         "arguments": "not-json"}}]},
            "done",
        ]
    )
    monkeypatch.setitem(TOOLS, "kubectl", lambda s: f"ran:{s}")
    out = run_tool_flow(llm, "stub", "sys", "u", tool_names=["kubectl"])
    assert out == "done"
    msgs = llm.calls[-1]["messages"]
    tool_msgs = [m["content"] for m in msgs if m.get("role") == "tool"]
    assert any("not available" in c for c in tool_msgs)
    assert any(c.startswith("ran:") for c in tool_msgs)  # raw-arg fallback


def test_tool_flow_turn_limit():
    looping = {"role": "assistant", "content": None,
               "tool_calls": [tc("kubectl", {"command": "get ns"})]}
    llm = ScriptedLLM([looping] * 3, fallback="forced final")
    out = run_tool_flow(
        llm, "stub", "sys", "u", tool_names=["kubectl"], max_turns=3,
        tools_map={"kubectl": lambda s: "ns"},
    )
    assert out == "forced final"
    # the final-answer nudge was injected
    assert any("Turn limit" in str(m.get("content")) for m in llm.calls[-1]["messages"])


def test_named_flows_compose(monkeypatch):
    monkeypatch.setitem(TOOLS, "kubectl", lambda s: "yaml-ish")
    monkeypatch.setitem(TOOLS, "trivy", lambda s: "0 CVEs")
    llm = ScriptedLLM(["analysis result"])
    assert analysis_flow(llm, "stub", "kind: Pod") == "analysis result"
    llm2 = ScriptedLLM(["audit result"])
    assert audit_flow(llm2, "stub", "default", "pod-1") == "audit result"
    llm3 = ScriptedLLM(["```yaml\nkind: Deployment\n```"])
    assert "Deployment" in generator_flow(llm3, "stub", "make a deployment")
    llm4 = ScriptedLLM(["formatted"])
    assert assistant_flow(llm4, "stub", "raw") == "formatted"
    # the audit flow advertises kubectl AND trivy schemas
    assert {t["function"]["name"] for t in llm2.calls[0]["tools"]} == {"kubectl", "trivy"}


def test_cli_execute_with_scripted_llm(monkeypatch, tmp_path):
    """Full CLI `execute` path through typer with a scripted local 'engine'."""
    from typer.testing import CliRunner

    import opsagent_amd.cli as cli_mod

    reply = json.dumps(
        {"question": "q", "thought": "t", "action": {"name": "", "input": ""},
         "observation": "", "final_answer": "there are 4 namespaces"}
    )
    llm = ScriptedLLM([reply, "formatted: 4 namespaces"])
    monkeypatch.setattr(cli_mod, "new_client", lambda *a, **k: llm)
    monkeypatch.chdir(tmp_path)
    runner = CliRunner()
    result = runner.invoke(cli_mod.app, ["execute", "count namespaces"])
    assert result.exit_code == 0, result.output
    assert "4 namespaces" in result.output
