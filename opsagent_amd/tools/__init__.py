"""Tool plugin registry.

Capability parity with /root/reference/pkg/tools/tool.go:17-38: a tool is a
callable `str -> str` (raising ToolError on failure), registered by name in
TOOLS. Default registry matches the reference's CopilotTools map
(tool.go:20-26): search, python, trivy, kubectl, jq.
"""

from __future__ import annotations

from typing import Callable, Dict

Tool = Callable[[str], str]


class ToolError(Exception):
    """Raised by a tool on failure; the agent converts it to an observation."""


TOOLS: Dict[str, Tool] = {}


def register_tool(name: str, fn: Tool) -> None:
    TOOLS[name] = fn


def get_tool(name: str):
    return TOOLS.get(name)


def _register_defaults() -> None:
    from opsagent_amd.tools.kubectl import kubectl
    from opsagent_amd.tools.python_repl import python_repl
    from opsagent_amd.tools.trivy import trivy
    from opsagent_amd.tools.jq import jq
    from opsagent_amd.tools.search import google_search

    register_tool("kubectl", kubectl)
    register_tool("python", python_repl)
    register_tool("trivy", trivy)
    register_tool("jq", jq)
    register_tool("search", google_search)


_register_defaults()
