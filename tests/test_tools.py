import json
import os
import stat

import pytest

from opsagent_amd.tools import TOOLS, ToolError, register_tool
from opsagent_amd.tools.jq import jq
from opsagent_amd.tools.kubectl import classify_error, kubectl
from opsagent_amd.tools.python_repl import python_repl


def test_registry_defaults():
    for name in ("kubectl", "python", "trivy", "jq", "search"):
        assert name in TOOLS


def test_register_tool():
    register_tool("echo", lambda s: s)
    assert TOOLS["echo"]("hi") == "hi"
    del TOOLS["echo"]


class TestPythonRepl:
    def test_hello(self):
        assert python_repl("print('hello world')") == "hello world"

    def test_syntax_error(self):
        with pytest.raises(ToolError):
            python_repl("print(")

    def test_empty(self):
        with pytest.raises(ToolError):
            python_repl("   ")


class TestJq:
    def test_simple_field(self):
        out = jq('{"a": {"b": 5}} | .a.b')
        assert out.strip() == "5"

    def test_identity(self):
        out = jq('{"x": 1} | .')
        assert json.loads(out) == {"x": 1}

    def test_json_with_pipe_in_string(self):
        out = jq('{"cmd": "a | b"} | .cmd')
        assert json.loads(out) == "a | b"

    def test_invalid_json(self):
        with pytest.raises(ToolError):
            jq("not-json | .")

    def test_missing_pipe(self):
        with pytest.raises(ToolError):
            jq('{"a": 1}')


class TestKubectl:
    @pytest.fixture()
    def fake_kubectl(self, tmp_path, monkeypatch):
        """Install a fake kubectl on PATH that echoes canned output."""
        script = tmp_path / "kubectl"
        script.write_text(
            "#!/bin/bash\n"
            'if [[ "$1" == "get" && "$2" == "namespaces" ]]; then\n'
            '  printf "default\\nkube-system\\nkube-public\\n"\n'
            "  exit 0\n"
            "fi\n"
            'if [[ "$1" == "get" && "$2" == "pods" ]]; then\n'
            '  echo "E0307 10:00:00.000000 1 memcache.go] noise" >&2\n'
            '  printf "pod-a Running\\npod-b CrashLoopBackOff\\n"\n'
            "  exit 0\n"
            "fi\n"
            'echo "error: resource NotFound" >&2\n'
            "exit 1\n"
        )
        script.chmod(script.stat().st_mode | stat.S_IEXEC)
        monkeypatch.setenv("PATH", str(tmp_path) + os.pathsep + os.environ["PATH"])
        return script

    def test_auto_prefix_and_output(self, fake_kubectl):
        out = kubectl("get namespaces")
        assert "kube-system" in out

    def test_explicit_prefix(self, fake_kubectl):
        out = kubectl("kubectl get pods")
        assert "pod-a" in out

    def test_error_classified(self, fake_kubectl):
        with pytest.raises(ToolError) as ei:
            kubectl("get nonexistent thing")
        assert "not found" in str(ei.value).lower()

    def test_empty_command(self):
        with pytest.raises(ToolError):
            kubectl("")


def test_classify_error():
    assert "not found" in classify_error('pods "x" NotFound')
    assert "forbidden" in classify_error("Error: Forbidden")
    assert "unreachable" in classify_error("unable to connect to the server")
    assert classify_error("something else") == ""


def test_kubectl_filters_klog_noise(tmp_path, monkeypatch):
    script = tmp_path / "kubectl"
    script.write_text(
        "#!/bin/bash\n"
        'printf "E0307 10:00:00.000001 1 memcache.go] klog noise\\nreal output\\n"\n'
    )
    script.chmod(0o755)
    monkeypatch.setenv("PATH", str(tmp_path) + os.pathsep + os.environ["PATH"])
    out = kubectl("get x")
    assert "klog noise" not in out
    assert "real output" in out


def test_jsonpath_helpers():
    from opsagent_amd.tools.jsonpath import extract_pod_summaries, json_path

    doc = {
        "items": [
            {
                "metadata": {"namespace": "prod", "name": "web-1"},
                "spec": {"containers": [{"image": "nginx:1.25"}],
                         "initContainers": [{"image": "busybox:1"}]},
            }
        ]
    }
    import json as _json

    out = extract_pod_summaries(_json.dumps(doc))
    assert out == [{"namespace": "prod", "name": "web-1", "images": ["nginx:1.25", "busybox:1"]}]
    assert json_path(doc, "items[0].metadata.name") == "web-1"
    assert json_path(doc, "$.items[0].spec.containers[0].image") == "nginx:1.25"
    assert json_path(doc, "items[5].x") is None
    assert extract_pod_summaries("not json") == []
