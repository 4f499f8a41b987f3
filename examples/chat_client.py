"""Minimal OpenAI-compatible client against the local server.

Start the server first:
    python -m opsagent_amd.cli serve --port 8080
Then:
    python examples/chat_client.py
"""

import json
import urllib.request

BASE = "http://127.0.0.1:8080"


def post(path: str, body: dict) -> dict:
    req = urllib.request.Request(
        BASE + path,
        data=json.dumps(body).encode(),
        headers={"Content-Type": "application/json"},
    )
    with urllib.request.urlopen(req) as r:
        return json.loads(r.read())


resp = post(
    "/v1/chat/completions",
    {
        "model": "llama3-8b",
        "messages": [
            {"role": "user", "content": "Why might a pod be in CrashLoopBackOff?"}
        ],
        "max_tokens": 256,
    },
)
print(resp["choices"][0]["message"]["content"])

# function calling: the engine CONSTRAINS output to a valid call of a
# declared tool (name included) — json.loads below cannot fail
resp = post(
    "/v1/chat/completions",
    {
        "model": "llama3-8b",
        "messages": [{"role": "user", "content": "list the pods in prod"}],
        "max_tokens": 256,
        "tools": [
            {
                "type": "function",
                "function": {
                    "name": "kubectl",
                    "description": "run a kubectl command",
                    "parameters": {
                        "type": "object",
                        "properties": {"command": {"type": "string"}},
                    },
                },
            }
        ],
    },
)
msg = resp["choices"][0]["message"]
if msg.get("tool_calls"):
    call = msg["tool_calls"][0]["function"]
    print("tool:", call["name"], "args:", json.loads(call["arguments"]))
