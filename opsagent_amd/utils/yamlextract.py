"""Extract YAML manifests from LLM output (ref /root/reference/pkg/utils/yaml.go:22-36)."""

from __future__ import annotations

import re

_YAML_FENCE = re.compile(r"```yaml\s*\n(.*?)```", re.DOTALL)
_ANY_FENCE = re.compile(r"```\s*\n(.*?)```", re.DOTALL)


def extract_yaml(text: str) -> str:
    """Return YAML content from ```yaml fences, else any fence, else the text itself."""
    m = _YAML_FENCE.search(text)
    if m:
        return m.group(1).strip()
    m = _ANY_FENCE.search(text)
    if m:
        return m.group(1).strip()
    return text.strip()
