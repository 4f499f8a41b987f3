"""JSONPath-style extraction helper (ref /root/reference/pkg/tools/jsonpath.go:10-60).

The reference ships an unexported helper that pulls namespace/name/images out
of a Kubernetes List JSON; here it is a usable utility with a tiny
dotted-path evaluator (the jq tool handles full expressions)."""

from __future__ import annotations

import json
import re
from typing import Any, List


def json_path(obj: Any, path: str) -> Any:
    """Evaluate a simple path like `items[0].metadata.name` ('$.' prefix ok)."""
    path = path.lstrip("$").lstrip(".")
    node = obj
    for key, idx in re.findall(r"([A-Za-z_][A-Za-z0-9_\-]*)|\[(\d+)\]", path):
        try:
            node = node[key] if key else node[int(idx)]
        except (KeyError, IndexError, TypeError):
            return None
    return node


def extract_pod_summaries(list_json: str) -> List[dict]:
    """From a k8s List JSON, extract [{namespace, name, images}] per item
    (the reference helper's behavior, jsonpath.go:10-60)."""
    try:
        obj = json.loads(list_json)
    except json.JSONDecodeError:
        return []
    out = []
    for item in obj.get("items", []) if isinstance(obj, dict) else []:
        meta = item.get("metadata", {})
        spec = item.get("spec", {})
        images = [
            c.get("image", "")
            for c in spec.get("containers", []) + spec.get("initContainers", [])
            if c.get("image")
        ]
        out.append(
            {
                "namespace": meta.get("namespace", ""),
                "name": meta.get("name", ""),
                "images": images,
            }
        )
    return out
