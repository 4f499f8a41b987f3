"""Kubernetes helpers.

Capability parity with /root/reference/pkg/kubernetes/ (GetKubeConfig
apply.go:24, GetYaml get.go:30, ApplyYaml apply.go:38). The reference uses
client-go; here we use the kubectl binary (already a hard dependency of the
kubectl tool), honoring in-cluster service accounts and ~/.kube/config the
same way kubectl itself does. `apply_yaml` uses server-side apply to match
the reference's dynamic Apply with a field manager (apply.go:55-99).
"""

from __future__ import annotations

import subprocess

from opsagent_amd.tools import ToolError

FIELD_MANAGER = "opsagent-amd"


def _run(args, input_text: str = "", timeout: int = 60) -> str:
    try:
        proc = subprocess.run(
            args, input=input_text or None, capture_output=True, text=True, timeout=timeout
        )
    except FileNotFoundError:
        raise ToolError("kubectl binary not found on PATH")
    except subprocess.TimeoutExpired:
        raise ToolError(f"kubectl timed out after {timeout}s")
    if proc.returncode != 0:
        raise ToolError(proc.stderr.strip() or f"kubectl exited with {proc.returncode}")
    return proc.stdout


def get_yaml(resource: str, name: str, namespace: str = "default") -> str:
    """Fetch one resource as YAML (ref get.go:30-88)."""
    args = ["kubectl", "get", resource, name, "-o", "yaml"]
    if namespace:
        args += ["-n", namespace]
    return _run(args)


def apply_yaml(manifests: str) -> str:
    """Server-side apply of one or more YAML documents (ref apply.go:38-99).

    Manifests are piped over stdin (`-f -`) so secret-bearing YAML never
    touches the filesystem.
    """
    return _run(
        [
            "kubectl",
            "apply",
            "--server-side",
            f"--field-manager={FIELD_MANAGER}",
            "-f",
            "-",
        ],
        input_text=manifests,
    )
