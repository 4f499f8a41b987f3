"""Python REPL tool (ref /root/reference/pkg/tools/python.go:25-77).

The reference shells into a hardcoded venv (`~/k8s/python-cli/k8s-env`) built
in its Dockerfile; we run the current interpreter with `python3 -c` (the
kubernetes SDK, when installed, is importable the same way) and honor an
optional OPSAGENT_PYTHON_BIN override for a dedicated environment.
"""

from __future__ import annotations

import os
import subprocess
import sys

from opsagent_amd.tools import ToolError

DEFAULT_TIMEOUT = 60


def python_repl(script: str, timeout: int = DEFAULT_TIMEOUT) -> str:
    script = script.strip()
    if not script:
        raise ToolError("empty python script")
    py = os.environ.get("OPSAGENT_PYTHON_BIN", sys.executable or "python3")
    try:
        proc = subprocess.run(
            [py, "-c", script],
            capture_output=True,
            text=True,
            timeout=timeout,
        )
    except subprocess.TimeoutExpired:
        raise ToolError(f"python script timed out after {timeout}s")
    if proc.returncode != 0:
        raise ToolError(proc.stderr.strip() or f"python exited with code {proc.returncode}")
    return proc.stdout.strip() or "(no output)"
