"""JSON repair utilities for LLM output.

Capability parity with /root/reference/pkg/utils/json.go:16-190 (CleanJSON /
ParseJSON / ExtractField): extract the outermost JSON object from surrounding
prose or code fences, escape raw newlines inside strings, drop trailing
commas, and fall back to regex field extraction.

Note: when decoding runs on the local MI355X engine with grammar-constrained
sampling (opsagent_amd.engine.grammar), the model *cannot* emit invalid JSON
and these repairs are a no-op safety net. They remain load-bearing when the
agent is pointed at a remote/unconstrained endpoint.
"""

from __future__ import annotations

import json
import re
from typing import Any, Optional


def _strip_code_fences(s: str) -> str:
    # ```json ... ``` or ``` ... ```
    m = re.search(r"```(?:json)?\s*(.*?)```", s, re.DOTALL)
    if m:
        return m.group(1)
    return s


def _extract_braced(s: str) -> str:
    """Return the first balanced {...} region, or s unchanged."""
    start = s.find("{")
    if start < 0:
        return s
    depth = 0
    in_str = False
    esc = False
    for i in range(start, len(s)):
        c = s[i]
        if in_str:
            if esc:
                esc = False
            elif c == "\\":
                esc = True
            elif c == '"':
                in_str = False
        else:
            if c == '"':
                in_str = True
            elif c == "{":
                depth += 1
            elif c == "}":
                depth -= 1
                if depth == 0:
                    return s[start : i + 1]
    # unbalanced: take from first { to last }
    end = s.rfind("}")
    if end > start:
        return s[start : end + 1]
    return s


def _escape_newlines_in_strings(s: str) -> str:
    out = []
    in_str = False
    esc = False
    for c in s:
        if in_str:
            if esc:
                out.append(c)
                esc = False
                continue
            if c == "\\":
                out.append(c)
                esc = True
                continue
            if c == '"':
                in_str = False
                out.append(c)
                continue
            if c == "\n":
                out.append("\\n")
                continue
            if c == "\t":
                out.append("\\t")
                continue
            if c == "\r":
                out.append("\\r")
                continue
            out.append(c)
        else:
            if c == '"':
                in_str = True
            out.append(c)
    return "".join(out)


_TRAILING_COMMA = re.compile(r",\s*([}\]])")


def clean_json(s: str) -> str:
    """Best-effort repair of an LLM reply into parseable JSON text."""
    s = _strip_code_fences(s)
    s = _extract_braced(s)
    s = s.strip()
    s = _escape_newlines_in_strings(s)
    s = _TRAILING_COMMA.sub(r"\1", s)
    return s


def parse_json(s: str) -> Optional[Any]:
    """Parse with repair fallback; None if unrecoverable."""
    try:
        return json.loads(s)
    except (json.JSONDecodeError, TypeError):
        pass
    try:
        return json.loads(clean_json(s))
    except (json.JSONDecodeError, TypeError):
        pass
    # quote fixing (ref json.go CleanJSON): a document written with single
    # quotes and (nearly) no double quotes — swap conservatively
    if "'" in s and s.count('"') <= 1:
        try:
            return json.loads(clean_json(s.replace("'", '"')))
        except (json.JSONDecodeError, TypeError):
            pass
    return None


def extract_field(s: str, field: str) -> str:
    """Extract a top-level string field, with regex fallback (ref json.go:155-190)."""
    obj = parse_json(s)
    if isinstance(obj, dict):
        v = obj.get(field)
        if isinstance(v, str):
            return v
        if v is not None:
            return json.dumps(v, ensure_ascii=False)
    # regex fallback: "field"\s*:\s*"..."
    m = re.search(
        r'"' + re.escape(field) + r'"\s*:\s*"((?:[^"\\]|\\.)*)"',
        s,
        re.DOTALL,
    )
    if m:
        raw = m.group(1)
        try:
            return json.loads('"' + raw + '"')
        except json.JSONDecodeError:
            return raw
    return ""
