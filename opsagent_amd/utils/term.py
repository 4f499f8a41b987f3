"""Terminal markdown rendering (ref /root/reference/pkg/utils/term.go:11-30)."""

from __future__ import annotations

import shutil
import sys


def render_markdown(text: str) -> str:
    """Render markdown to ANSI at terminal width via rich; plain text fallback."""
    if not sys.stdout.isatty():
        return text
    try:
        import io

        from rich.console import Console
        from rich.markdown import Markdown

        width = shutil.get_terminal_size((100, 24)).columns
        buf = io.StringIO()
        console = Console(file=buf, width=width, force_terminal=True)
        console.print(Markdown(text))
        return buf.getvalue()
    except Exception:
        return text
