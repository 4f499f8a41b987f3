"""Google Custom Search tool (ref /root/reference/pkg/tools/googlesearch.go:28-44).

Uses GOOGLE_API_KEY / GOOGLE_CSE_ID env vars; plain HTTPS via httpx. Raises a
clear ToolError when credentials or network are unavailable (offline clusters).
"""

from __future__ import annotations

import os

from opsagent_amd.tools import ToolError


def google_search(query: str, timeout: float = 15.0) -> str:
    query = query.strip()
    if not query:
        raise ToolError("empty search query")
    api_key = os.environ.get("GOOGLE_API_KEY", "")
    cse_id = os.environ.get("GOOGLE_CSE_ID", "")
    if not api_key or not cse_id:
        raise ToolError("google search unavailable: GOOGLE_API_KEY/GOOGLE_CSE_ID not set")
    try:
        import httpx

        resp = httpx.get(
            "https://customsearch.googleapis.com/customsearch/v1",
            params={"key": api_key, "cx": cse_id, "q": query, "num": 5},
            timeout=timeout,
        )
        resp.raise_for_status()
        data = resp.json()
    except Exception as e:  # noqa: BLE001 — any transport failure becomes an observation
        raise ToolError(f"google search failed: {e}")
    items = data.get("items", [])
    if not items:
        return "(no results)"
    out = []
    for it in items[:5]:
        out.append(f"{it.get('title', '')}\n{it.get('link', '')}\n{it.get('snippet', '')}")
    return "\n\n".join(out)
