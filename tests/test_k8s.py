"""k8s helper tests with a fake kubectl binary on PATH (hermetic — the
reference's python tool test is NOT hermetic)."""

import os
import stat

import pytest

from opsagent_amd import k8s
from opsagent_amd.tools import ToolError


@pytest.fixture()
def fake_kubectl(tmp_path, monkeypatch):
    """Install a fake kubectl that records argv and replays canned output."""
    log = tmp_path / "calls.log"
    script = tmp_path / "kubectl"
    script.write_text(
        "#!/bin/bash\n"
        f'echo "$@" >> {log}\n'
        'if [[ "$1" == "get" ]]; then echo "kind: Pod"; exit 0; fi\n'
        'if [[ "$1" == "apply" ]]; then cat "${@: -1}" >/dev/null; '
        'echo "namespace/demo serverside-applied"; exit 0; fi\n'
        'echo "unknown" >&2; exit 1\n'
    )
    script.chmod(script.stat().st_mode | stat.S_IEXEC)
    monkeypatch.setenv("PATH", f"{tmp_path}:{os.environ['PATH']}")
    return log


def test_get_yaml(fake_kubectl):
    out = k8s.get_yaml("pod", "web", "prod")
    assert out.strip() == "kind: Pod"
    call = fake_kubectl.read_text()
    assert "get pod web -o yaml -n prod" in call


def test_apply_yaml_server_side(fake_kubectl):
    out = k8s.apply_yaml("apiVersion: v1\nkind: Namespace\nmetadata:\n  name: demo")
    assert "serverside-applied" in out
    call = fake_kubectl.read_text()
    assert "--server-side" in call and "--field-manager=opsagent-amd" in call


def test_kubectl_error_raises(tmp_path, monkeypatch):
    bad = tmp_path / "kubectl"
    bad.write_text("#!/bin/bash\necho 'forbidden: denied' >&2\nexit 1\n")
    bad.chmod(bad.stat().st_mode | stat.S_IEXEC)
    monkeypatch.setenv("PATH", f"{tmp_path}:{os.environ['PATH']}")
    with pytest.raises(ToolError, match="forbidden"):
        k8s.get_yaml("pod", "x")
