"""CLI command tests (typer CliRunner) with a scripted LLM and mocked k8s —
the reference leaves its CLI subcommands unregistered (SURVEY known
inconsistencies); here every command is wired and tested."""

import json

import pytest
from typer.testing import CliRunner

from opsagent_amd import cli as cli_mod
from opsagent_amd.llm.client import ScriptedLLM

runner = CliRunner()


def tp(thought="", action=None, final=""):
    return json.dumps(
        {
            "question": "q",
            "thought": thought,
            "action": action or {"name": "", "input": ""},
            "observation": "",
            "final_answer": final,
        }
    )


@pytest.fixture()
def scripted(monkeypatch):
    """Route the CLI's LLM client to a scripted one; returns a setter."""
    holder = {}

    def set_script(script, fallback="formatted result"):
        llm = ScriptedLLM(script, fallback=fallback)
        monkeypatch.setattr(
            cli_mod, "new_client", lambda **kw: llm
        )
        holder["llm"] = llm
        return llm

    return set_script


def test_version_command():
    res = runner.invoke(cli_mod.app, ["version"])
    assert res.exit_code == 0
    assert "opsagent-amd" in res.output


def test_execute_command(scripted, monkeypatch):
    from opsagent_amd.tools import TOOLS

    monkeypatch.setitem(TOOLS, "kubectl", lambda cmd: "default\nkube-system")
    scripted(
        [
            tp(thought="list", action={"name": "kubectl", "input": "get ns"}),
            tp(thought="done", final="There are 2 namespaces."),
            "**There are 2 namespaces.**",  # assistant_flow formatting turn
        ]
    )
    res = runner.invoke(cli_mod.app, ["execute", "count namespaces"])
    assert res.exit_code == 0, res.output
    assert "2 namespaces" in res.output


def test_analyze_command(scripted, monkeypatch):
    from opsagent_amd import k8s

    monkeypatch.setattr(
        k8s, "get_yaml", lambda r, n, ns: "kind: Pod\nmetadata:\n  name: web"
    )
    scripted(["analysis: the pod looks healthy"])
    res = runner.invoke(
        cli_mod.app, ["analyze", "--name", "web", "--namespace", "prod"]
    )
    assert res.exit_code == 0, res.output
    assert "healthy" in res.output


def test_diagnose_command(scripted, monkeypatch):
    from opsagent_amd.tools import TOOLS

    monkeypatch.setitem(TOOLS, "kubectl", lambda cmd: "Events: OOMKilled")
    scripted(
        [
            tp(thought="check", action={"name": "kubectl", "input": "describe pod broken"}),
            tp(thought="oom", final="The pod was OOMKilled; raise its memory limit."),
            "The pod was OOMKilled; raise its memory limit.",
        ]
    )
    res = runner.invoke(
        cli_mod.app, ["diagnose", "--name", "broken", "--namespace", "default"]
    )
    assert res.exit_code == 0, res.output
    assert "OOMKilled" in res.output


def test_generate_command_writes_without_apply(scripted, monkeypatch):
    from opsagent_amd import k8s

    applied = []
    monkeypatch.setattr(k8s, "apply_yaml", lambda y: applied.append(y) or "applied")
    scripted(
        ["```yaml\napiVersion: v1\nkind: Namespace\nmetadata:\n  name: demo\n```"]
    )
    # no --yes and non-tty stdin: must NOT apply
    res = runner.invoke(cli_mod.app, ["generate", "a namespace called demo"])
    assert res.exit_code == 0, res.output
    assert "kind: Namespace" in res.output
    assert not applied


def test_audit_command(scripted, monkeypatch):
    """audit: pod YAML → misconfig analysis → trivy image scan → summary
    (ref audit.go flow), with both tools mocked."""
    from opsagent_amd.tools import TOOLS

    calls = []
    monkeypatch.setitem(
        TOOLS, "kubectl",
        lambda cmd: calls.append(("kubectl", cmd)) or
        "kind: Pod\nspec:\n  containers:\n  - image: nginx:1.19",
    )
    monkeypatch.setitem(
        TOOLS, "trivy",
        lambda img: calls.append(("trivy", img)) or
        "nginx:1.19 - CVE-2021-23017 HIGH resolver off-by-one",
    )
    def call(name, args):
        return {
            "role": "assistant",
            "content": None,
            "tool_calls": [{
                "id": "call_1", "type": "function",
                "function": {"name": name, "arguments": json.dumps(args)},
            }],
        }

    llm = scripted(
        [
            call("kubectl", {"command": "get pod web-1 -o yaml"}),
            call("trivy", {"image": "nginx:1.19"}),
            "Security summary: nginx:1.19 carries CVE-2021-23017 (HIGH); "
            "bump the image and drop root.",
        ],
        fallback="Security summary: done",
    )
    res = runner.invoke(cli_mod.app, ["audit", "--name", "web-1"])
    assert res.exit_code == 0, res.output
    assert "CVE-2021-23017" in res.output or "Security summary" in res.output
