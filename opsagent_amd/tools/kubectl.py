"""kubectl tool.

Capability parity with /root/reference/pkg/tools/kubectl.go:21-194:
runs through `bash -c` so pipes work, auto-prefixes "kubectl " when missing,
classifies common failure modes (not found / forbidden / cluster unreachable),
records a per-verb perf metric, and filters metrics-server noise and
klog error lines from the output.
"""

from __future__ import annotations

import re
import subprocess

from opsagent_amd.tools import ToolError
from opsagent_amd.utils.perf import get_perf_stats

# klog lines like "E0307 12:34:56.789012 ..." and metrics-server noise
_KLOG_LINE = re.compile(r"^[EWIF]\d{4} \d{2}:\d{2}:\d{2}\.\d+")
_NOISE_SUBSTRINGS = (
    "metrics.k8s.io",
    "couldn't get current server API group list",
    "the server is currently unable to handle the request",
)

DEFAULT_TIMEOUT = 60


def _filter_output(text: str) -> str:
    lines = []
    for line in text.splitlines():
        if _KLOG_LINE.match(line.strip()):
            continue
        if any(s in line for s in _NOISE_SUBSTRINGS):
            continue
        lines.append(line)
    return "\n".join(lines).strip()


def classify_error(stderr: str) -> str:
    low = stderr.lower()
    if "notfound" in low or "not found" in low:
        return "resource not found — check the name and namespace"
    if "forbidden" in low:
        return "forbidden — the service account lacks RBAC permission for this verb"
    if "unable to connect" in low or "connection refused" in low or "no such host" in low:
        return "cluster unreachable — check kubeconfig and API server availability"
    return ""


def kubectl(command: str, timeout: int = DEFAULT_TIMEOUT) -> str:
    command = command.strip()
    if not command:
        raise ToolError("empty kubectl command")
    # auto-prefix (ref kubectl.go:75-77)
    if not command.startswith("kubectl"):
        command = "kubectl " + command
    verb = (command.split() + ["", ""])[1]
    perf = get_perf_stats()
    with perf.trace(f"kubectl_command_{verb}"):
        try:
            proc = subprocess.run(
                ["bash", "-c", command],
                capture_output=True,
                text=True,
                timeout=timeout,
            )
        except subprocess.TimeoutExpired:
            raise ToolError(f"kubectl command timed out after {timeout}s: {command}")
        except FileNotFoundError:
            raise ToolError("bash not available to run kubectl")
    if proc.returncode != 0:
        hint = classify_error(proc.stderr)
        msg = _filter_output(proc.stderr) or f"kubectl exited with code {proc.returncode}"
        raise ToolError(f"{msg}" + (f" ({hint})" if hint else ""))
    out = _filter_output(proc.stdout)
    return out if out else "(no output)"
