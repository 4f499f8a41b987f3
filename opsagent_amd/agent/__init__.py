from opsagent_amd.agent.react import ToolPrompt, assistant
from opsagent_amd.agent.workflows import (
    analysis_flow,
    assistant_flow,
    audit_flow,
    generator_flow,
)

__all__ = [
    "ToolPrompt",
    "assistant",
    "analysis_flow",
    "assistant_flow",
    "audit_flow",
    "generator_flow",
]
