"""Function-calling workflows.

The reference delegates analyze/audit/generate/format to the external
swarm-go library (OpenAI-native function-calling agents,
/root/reference/pkg/workflows/*.go via SimpleFlow.Run). Here the same flows
run on a native function-calling loop against any LLMClient — including the
in-process MI355X engine, whose grammar-constrained sampler emits tool_calls
in the same OpenAI wire format (SURVEY.md §2b "hard part (e)").

Flows (ref files):
  analysis_flow   — pkg/workflows/analyze.go:47  (kubectl tool, MaxTurns 30)
  audit_flow      — pkg/workflows/audit.go:58    (kubectl + trivy)
  generator_flow  — pkg/workflows/generate.go:56 (no tools)
  assistant_flow  — pkg/workflows/assistant.go:69 (formatting; the reference
                    buggily reuses the analysis prompt there — we use a real
                    formatting prompt, per SURVEY.md "known inconsistencies")
"""

from __future__ import annotations

import json
from typing import Dict, List, Optional, Sequence

from opsagent_amd.agent import prompts
from opsagent_amd.llm.client import ChatMessage, LLMClient
from opsagent_amd.tools import TOOLS, Tool, ToolError
from opsagent_amd.utils.logging import get_logger
from opsagent_amd.utils.perf import get_perf_stats

log = get_logger("workflows")

# OpenAI tools wire-format schemas for the agent functions
# (ref pkg/workflows/swarm.go:14-77 kubectlFunc/trivyFunc/pythonFunc).
TOOL_SCHEMAS: Dict[str, dict] = {
    "kubectl": {
        "type": "function",
        "function": {
            "name": "kubectl",
            "description": "Run a kubectl command against the cluster and return its output. "
            "Input is the command without the leading 'kubectl'. Pipes are allowed.",
            "parameters": {
                "type": "object",
                "properties": {
                    "command": {"type": "string", "description": "e.g. 'get pods -n kube-system'"}
                },
                "required": ["command"],
            },
        },
    },
    "trivy": {
        "type": "function",
        "function": {
            "name": "trivy",
            "description": "Scan a container image for vulnerabilities with trivy.",
            "parameters": {
                "type": "object",
                "properties": {"image": {"type": "string", "description": "image reference"}},
                "required": ["image"],
            },
        },
    },
    "python": {
        "type": "function",
        "function": {
            "name": "python",
            "description": "Execute a python3 script and return stdout.",
            "parameters": {
                "type": "object",
                "properties": {"script": {"type": "string"}},
                "required": ["script"],
            },
        },
    },
}

_ARG_KEY = {"kubectl": "command", "trivy": "image", "python": "script"}


def _dispatch(name: str, arguments: str, tools_map: Dict[str, Tool]) -> str:
    fn = tools_map.get(name)
    if fn is None:
        return f"tool {name} is not available"
    try:
        args = json.loads(arguments) if arguments else {}
    except json.JSONDecodeError:
        args = {_ARG_KEY.get(name, "input"): arguments}
    value = args.get(_ARG_KEY.get(name, "input"), "")
    if not isinstance(value, str):
        value = json.dumps(value)
    try:
        return fn(value)
    except ToolError as e:
        return f"tool {name} failed: {e}"
    except Exception as e:  # noqa: BLE001
        return f"tool {name} crashed: {e}"


def run_tool_flow(
    client: LLMClient,
    model: str,
    system: str,
    user: str,
    tool_names: Sequence[str] = (),
    max_turns: int = 30,
    max_tokens: int = 2048,
    tools_map: Optional[Dict[str, Tool]] = None,
) -> str:
    """Native function-calling loop (replaces swarm.SimpleFlow.Run)."""
    perf = get_perf_stats()
    tools_map = tools_map or TOOLS
    schemas = [TOOL_SCHEMAS[n] for n in tool_names if n in TOOL_SCHEMAS]
    messages: List[ChatMessage] = [
        {"role": "system", "content": system},
        {"role": "user", "content": user},
    ]
    with perf.trace("workflow_run"):
        for _ in range(max_turns):
            reply = client.chat(model, max_tokens, messages, tools=schemas or None)
            tool_calls = reply.get("tool_calls") or []
            if not tool_calls:
                return reply.get("content") or ""
            messages.append(
                {
                    "role": "assistant",
                    "content": reply.get("content"),
                    "tool_calls": tool_calls,
                }
            )
            for tc in tool_calls:
                fn = tc.get("function", {})
                result = _dispatch(fn.get("name", ""), fn.get("arguments", ""), tools_map)
                messages.append(
                    {
                        "role": "tool",
                        "tool_call_id": tc.get("id", ""),
                        "content": result,
                    }
                )
        # turn limit: ask for a final answer without tools
        messages.append(
            {"role": "user", "content": "Turn limit reached — give your best final answer now."}
        )
        reply = client.chat(model, max_tokens, messages)
        return reply.get("content") or ""


def analysis_flow(client: LLMClient, model: str, manifest_yaml: str, max_tokens: int = 2048) -> str:
    return run_tool_flow(
        client,
        model,
        prompts.ANALYSIS_PROMPT,
        f"Analyze this Kubernetes manifest:\n```yaml\n{manifest_yaml}\n```",
        tool_names=["kubectl"],
        max_turns=30,
        max_tokens=max_tokens,
    )


def audit_flow(client: LLMClient, model: str, namespace: str, pod: str, max_tokens: int = 2048) -> str:
    return run_tool_flow(
        client,
        model,
        prompts.AUDIT_PROMPT,
        f"Audit pod {pod} in namespace {namespace}.",
        tool_names=["kubectl", "trivy"],
        max_turns=30,
        max_tokens=max_tokens,
    )


def generator_flow(client: LLMClient, model: str, instructions: str, max_tokens: int = 4096) -> str:
    return run_tool_flow(
        client,
        model,
        prompts.GENERATE_PROMPT,
        instructions,
        tool_names=[],
        max_turns=2,
        max_tokens=max_tokens,
    )


def assistant_flow(client: LLMClient, model: str, raw_result: str, max_tokens: int = 2048) -> str:
    """Reformat a raw agent result as clean markdown (ref assistant.go:69)."""
    return run_tool_flow(
        client,
        model,
        prompts.ASSISTANT_FORMAT_PROMPT,
        f"Reformat this result for the user:\n\n{raw_result}",
        tool_names=[],
        max_turns=2,
        max_tokens=max_tokens,
    )
