import json

import pytest
from fastapi.testclient import TestClient

from opsagent_amd.config import Config, DEFAULTS
from opsagent_amd.llm.client import ScriptedLLM
from opsagent_amd.server import app as app_module
from opsagent_amd.server.auth import create_token, verify_token


def make_app(monkeypatch, script=None):
    cfg = Config(json.loads(json.dumps(DEFAULTS)))
    llm = ScriptedLLM(script or [])
    monkeypatch.setattr(app_module, "new_client", lambda *a, **k: llm)
    return app_module.create_app(cfg), llm


@pytest.fixture()
def client_and_llm(monkeypatch):
    script = [
        json.dumps(
            {
                "question": "q",
                "thought": "t",
                "action": {"name": "", "input": ""},
                "observation": "",
                "final_answer": "the final answer from the scripted model",
            }
        )
    ]
    app, llm = make_app(monkeypatch, script)
    return TestClient(app), llm


def login(client):
    r = client.post("/login", json={"username": "admin", "password": "novastar"})
    assert r.status_code == 200
    return r.json()["token"]


def test_login_ok_and_bad(client_and_llm):
    client, _ = client_and_llm
    token = login(client)
    assert verify_token(token, "novastar-secret-key")["username"] == "admin"
    r = client.post("/login", json={"username": "admin", "password": "wrong"})
    assert r.status_code == 401


def test_version_and_health(client_and_llm):
    client, _ = client_and_llm
    r = client.get("/api/version")
    assert r.status_code == 200 and r.json()["version"].startswith("v")
    r = client.get("/api/health")
    assert r.status_code == 200 and r.json()["status"] == "ok"


def test_execute_requires_auth(client_and_llm):
    client, _ = client_and_llm
    r = client.post("/api/execute", json={"instructions": "count namespaces"})
    assert r.status_code == 401


def test_execute_with_token(client_and_llm):
    client, llm = client_and_llm
    token = login(client)
    r = client.post(
        "/api/execute",
        json={"instructions": "count namespaces"},
        headers={"Authorization": f"Bearer {token}"},
    )
    assert r.status_code == 200
    body = r.json()
    assert body["status"] == "success"
    assert "final answer" in body["message"]


def test_execute_show_thought(monkeypatch):
    script = [
        json.dumps(
            {
                "question": "count",
                "thought": "I will count",
                "action": {"name": "", "input": ""},
                "observation": "",
                "final_answer": "a sufficiently long final answer",
            }
        )
    ]
    app, _llm = make_app(monkeypatch, script)
    client = TestClient(app)
    token = login(client)
    r = client.post(
        "/api/execute?show_thought=true",
        json={"instructions": "count"},
        headers={"Authorization": f"Bearer {token}"},
    )
    assert r.status_code == 200
    assert "tools_history" in r.json()


def test_perf_endpoints(client_and_llm):
    client, _ = client_and_llm
    token = login(client)
    hdr = {"Authorization": f"Bearer {token}"}
    r = client.get("/api/perf/stats", headers=hdr)
    assert r.status_code == 200 and "stats" in r.json()
    r = client.post("/api/perf/reset", headers=hdr)
    assert r.status_code == 200


def test_metrics_endpoint(client_and_llm):
    client, _ = client_and_llm
    client.get("/api/version")
    r = client.get("/metrics")
    assert r.status_code == 200
    assert "opsagent_operation_ms" in r.text


def test_expired_token_rejected(client_and_llm):
    client, _ = client_and_llm
    tok = create_token("admin", "novastar-secret-key", expire_hours=-1)
    r = client.post(
        "/api/execute",
        json={"instructions": "x"},
        headers={"Authorization": f"Bearer {tok}"},
    )
    assert r.status_code == 401


def test_jwt_tamper_rejected():
    tok = create_token("admin", "key1")
    assert verify_token(tok, "key1") is not None
    assert verify_token(tok, "key2") is None
    assert verify_token(tok + "x", "key1") is None
    assert verify_token("garbage", "key1") is None


def test_v1_models_listing(client_and_llm):
    client, _ = client_and_llm
    # /v1 is gated: no credentials -> 401
    r = client.get("/v1/models")
    assert r.status_code == 401
    token = login(client)
    r = client.get("/v1/models", headers={"Authorization": f"Bearer {token}"})
    assert r.status_code == 200
    data = r.json()
    assert data["object"] == "list"
    ids = [m["id"] for m in data["data"]]
    assert "llama3-8b" in ids and "llama3-70b" in ids


def test_v1_gate_api_key(client_and_llm):
    """X-API-Key matching engine.api_key also passes the /v1 gate."""
    import json as _json

    from opsagent_amd.config import Config, DEFAULTS
    from opsagent_amd.server.app import create_app

    cfg = Config(_json.loads(_json.dumps(DEFAULTS)))
    cfg.set("engine.api_key", "sekrit")
    from fastapi.testclient import TestClient

    with TestClient(create_app(cfg)) as c:
        assert c.get("/v1/models").status_code == 401
        assert c.get("/v1/models", headers={"X-API-Key": "wrong"}).status_code == 401
        assert c.get("/v1/models", headers={"X-API-Key": "sekrit"}).status_code == 200
        # chat+completions also gated
        r = c.post("/v1/chat/completions", json={"messages": []})
        assert r.status_code == 401
