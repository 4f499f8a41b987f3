"""jq tool (ref /root/reference/pkg/tools/jq.go:25-143).

Input convention matches the reference: "<JSON> | <jq-expression>" split on
the first pipe that is outside the JSON (we split on the last top-level '|').
Falls back to a pure-Python subset (.field access chains) when the jq binary
is missing, so CPU tests are hermetic.
"""

from __future__ import annotations

import json
import subprocess

from opsagent_amd.tools import ToolError
from opsagent_amd.utils.perf import get_perf_stats

DEFAULT_TIMEOUT = 30


def _split_input(inp: str):
    # The JSON document may itself contain '|' inside strings; find the first
    # '|' after the end of the balanced JSON value.
    s = inp.strip()
    try:
        obj, end = json.JSONDecoder().raw_decode(s)
        rest = s[end:].lstrip()
        if rest.startswith("|"):
            return json.dumps(obj), rest[1:].strip()
    except json.JSONDecodeError:
        pass
    # plain first-pipe split (ref jq.go:39-45)
    if "|" in s:
        doc, expr = s.split("|", 1)
        return doc.strip(), expr.strip()
    raise ToolError('jq input must be "<JSON> | <jq-expression>"')


def _python_jq_subset(doc: str, expr: str) -> str:
    """Evaluate simple `.a.b[0].c` expressions without the jq binary."""
    obj = json.loads(doc)
    expr = expr.strip()
    if expr == ".":
        return json.dumps(obj, indent=2)
    if not expr.startswith("."):
        raise ToolError(f"jq binary not available and expression unsupported by fallback: {expr}")
    node = obj
    import re as _re

    for part in _re.findall(r"\.([A-Za-z_][A-Za-z0-9_]*)|\[(\d+)\]", expr):
        key, idx = part
        try:
            if key:
                node = node[key]
            else:
                node = node[int(idx)]
        except (KeyError, IndexError, TypeError):
            return "null"
    return json.dumps(node, indent=2) if isinstance(node, (dict, list)) else json.dumps(node)


def jq(inp: str, timeout: int = DEFAULT_TIMEOUT) -> str:
    doc, expr = _split_input(inp)
    try:
        json.loads(doc)
    except json.JSONDecodeError as e:
        raise ToolError(f"invalid JSON input: {e}")
    perf = get_perf_stats()
    perf.record_metric("jq_expression_length", float(len(expr)))
    with perf.trace("jq_command"):
        try:
            proc = subprocess.run(
                ["jq", expr],
                input=doc,
                capture_output=True,
                text=True,
                timeout=timeout,
            )
        except FileNotFoundError:
            return _python_jq_subset(doc, expr)
        except subprocess.TimeoutExpired:
            raise ToolError(f"jq timed out after {timeout}s")
    if proc.returncode != 0:
        raise ToolError(proc.stderr.strip() or f"jq exited with code {proc.returncode}")
    return proc.stdout.strip()
