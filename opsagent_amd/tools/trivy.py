"""Trivy image-scan tool (ref /root/reference/pkg/tools/trivy.go:23-53)."""

from __future__ import annotations

import subprocess

from opsagent_amd.tools import ToolError

DEFAULT_TIMEOUT = 300


def trivy(image: str, timeout: int = DEFAULT_TIMEOUT) -> str:
    image = image.strip()
    # strip "image " prefix (ref trivy.go:30-34)
    if image.startswith("image "):
        image = image[len("image "):].strip()
    if not image:
        raise ToolError("empty image name")
    try:
        proc = subprocess.run(
            ["trivy", "image", image, "--scanners", "vuln"],
            capture_output=True,
            text=True,
            timeout=timeout,
        )
    except subprocess.TimeoutExpired:
        raise ToolError(f"trivy scan timed out after {timeout}s")
    except FileNotFoundError:
        raise ToolError("trivy binary not found on PATH")
    if proc.returncode != 0:
        raise ToolError(proc.stderr.strip() or f"trivy exited with code {proc.returncode}")
    return proc.stdout.strip() or "(no output)"
