"""RemoteOpenAIClient behavior against a local stub HTTP server: retry with
backoff on 429/5xx, fail-fast on 401, Azure URL mapping (hermetic — the
reference's client behavior at pkg/llms/openai.go:58-101)."""

import json
import threading
from http.server import BaseHTTPRequestHandler, HTTPServer

import pytest

from opsagent_amd.llm.client import LLMError, RemoteOpenAIClient


class _Handler(BaseHTTPRequestHandler):
    plan = []          # list of status codes to serve in order
    requests = []      # recorded (path, body)

    def do_POST(self):  # noqa: N802
        body = self.rfile.read(int(self.headers.get("Content-Length", 0)))
        _Handler.requests.append((self.path, json.loads(body or b"{}")))
        code = _Handler.plan.pop(0) if _Handler.plan else 200
        if code != 200:
            self.send_response(code)
            self.end_headers()
            self.wfile.write(b'{"error": "nope"}')
            return
        self.send_response(200)
        self.send_header("Content-Type", "application/json")
        self.end_headers()
        self.wfile.write(json.dumps({
            "choices": [{"message": {"role": "assistant", "content": "ok"},
                         "finish_reason": "stop"}],
            "usage": {"prompt_tokens": 1, "completion_tokens": 1},
        }).encode())

    def log_message(self, *a):  # quiet
        pass


@pytest.fixture()
def stub_server():
    srv = HTTPServer(("127.0.0.1", 0), _Handler)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    _Handler.plan = []
    _Handler.requests = []
    yield f"http://127.0.0.1:{srv.server_port}"
    srv.shutdown()


MSGS = [{"role": "user", "content": "hi"}]


def test_retry_on_429_then_success(stub_server):
    _Handler.plan = [429, 500]
    c = RemoteOpenAIClient("key", stub_server, retries=5, backoff_s=0.01)
    reply = c.chat("gpt-4", 64, MSGS)
    content = reply["content"] if isinstance(reply, dict) else reply
    assert "ok" in str(content)
    assert len(_Handler.requests) == 3  # two failures + the success


def test_fail_fast_on_401(stub_server):
    _Handler.plan = [401, 401, 401]
    c = RemoteOpenAIClient("key", stub_server, retries=5, backoff_s=0.01)
    with pytest.raises(LLMError):
        c.chat("gpt-4", 64, MSGS)
    assert len(_Handler.requests) == 1, "401 must not retry"


def test_retries_exhausted_raises(stub_server):
    _Handler.plan = [500, 500, 500]
    c = RemoteOpenAIClient("key", stub_server, retries=2, backoff_s=0.01)
    with pytest.raises(LLMError):
        c.chat("gpt-4", 64, MSGS)
    assert len(_Handler.requests) >= 2


def test_azure_url_mapping():
    c = RemoteOpenAIClient("key", "https://foo.azure.example.com/openai")
    assert c.is_azure
    url = c._url("gpt-4")
    assert "api-version=" in url


def test_connection_error_retries_then_raises():
    """Connection refusals retry with backoff and surface as LLMError (not a
    raw httpx exception) — ref openai.go treats transport errors as
    retryable."""
    c = RemoteOpenAIClient("key", "http://127.0.0.1:1", retries=2, backoff_s=0.01)
    with pytest.raises(LLMError):
        c.chat("gpt-4", 16, MSGS)
