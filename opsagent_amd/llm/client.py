"""LLM client layer — the boundary the MI355X engine replaces.

The reference's model layer is an HTTPS call to a remote OpenAI-compatible
endpoint with retry/backoff (/root/reference/pkg/llms/openai.go:29-104). Here
the same `chat()` interface has three implementations:

  * LocalEngineClient — the in-process MI355X inference engine
    (opsagent_amd.engine), selected with base_url == "local". No network, no
    serialization: messages go straight to the engine's chat-completions
    entry point, which runs grammar-constrained sampling on-GPU.
  * RemoteOpenAIClient — parity with the reference: POST /chat/completions
    with 5 retries, 1 s base exponential backoff on 429/5xx, fail-fast on 401
    (ref openai.go:58-101), Azure detection by "azure" in base_url.
  * ScriptedLLM — deterministic canned replies for hermetic agent tests
    (BASELINE config #1: stub echo-LLM).

All return an OpenAI-shaped assistant message dict:
  {"role": "assistant", "content": str|None, "tool_calls": [...]|None}
"""

from __future__ import annotations

import time
from typing import Any, Dict, List, Optional, Sequence

from opsagent_amd.utils.logging import get_logger

log = get_logger("llm")

ChatMessage = Dict[str, Any]


class LLMError(Exception):
    def __init__(self, message: str, status: int = 0):
        super().__init__(message)
        self.status = status


class LLMClient:
    """Interface: chat() for plain completion, with optional function-calling."""

    def chat(
        self,
        model: str,
        max_tokens: int,
        messages: Sequence[ChatMessage],
        tools: Optional[List[dict]] = None,
        temperature: float = 0.0,
        response_format: Optional[dict] = None,
    ) -> ChatMessage:
        raise NotImplementedError

    def chat_text(self, model: str, max_tokens: int, messages: Sequence[ChatMessage], **kw) -> str:
        reply = self.chat(model, max_tokens, messages, **kw)
        return reply.get("content") or ""


class RemoteOpenAIClient(LLMClient):
    """OpenAI/Azure-compatible HTTP client (ref openai.go:38-104)."""

    def __init__(self, api_key: str, base_url: str, retries: int = 5, backoff_s: float = 1.0):
        self.api_key = api_key
        self.base_url = (base_url or "https://api.openai.com/v1").rstrip("/")
        self.retries = retries
        self.backoff_s = backoff_s
        self.is_azure = "azure" in self.base_url.lower()
        self.azure_api_version = "2024-06-01"

    def _url(self, model: str) -> str:
        if self.is_azure:
            deployment = model.replace(".", "")  # ref openai.go:49-55 model-name mapper
            return (
                f"{self.base_url}/openai/deployments/{deployment}/chat/completions"
                f"?api-version={self.azure_api_version}"
            )
        return f"{self.base_url}/chat/completions"

    def chat(
        self,
        model: str,
        max_tokens: int,
        messages: Sequence[ChatMessage],
        tools: Optional[List[dict]] = None,
        temperature: float = 0.0,
        response_format: Optional[dict] = None,
    ) -> ChatMessage:
        import httpx

        payload: Dict[str, Any] = {
            "model": model,
            "max_tokens": max_tokens,
            # ref openai.go:74: Temperature = math.SmallestNonzeroFloat32 (≈ greedy)
            "temperature": temperature if temperature > 0 else 1e-8,
            "messages": list(messages),
        }
        if tools:
            payload["tools"] = tools
        if response_format:
            payload["response_format"] = response_format
        headers = {"Content-Type": "application/json"}
        if self.is_azure:
            headers["api-key"] = self.api_key
        else:
            headers["Authorization"] = f"Bearer {self.api_key}"

        backoff = self.backoff_s
        last_err: Optional[Exception] = None
        for attempt in range(self.retries):
            try:
                resp = httpx.post(self._url(model), json=payload, headers=headers, timeout=120.0)
            except Exception as e:  # transport error — retry
                last_err = e
                time.sleep(backoff)
                backoff *= 2
                continue
            if resp.status_code == 401:
                raise LLMError("unauthorized (401) — check the API key", 401)  # fail fast
            if resp.status_code in (429,) or resp.status_code >= 500:
                last_err = LLMError(f"HTTP {resp.status_code}: {resp.text[:200]}", resp.status_code)
                time.sleep(backoff)
                backoff *= 2
                continue
            if resp.status_code != 200:
                raise LLMError(f"HTTP {resp.status_code}: {resp.text[:500]}", resp.status_code)
            data = resp.json()
            choices = data.get("choices") or []
            if not choices:
                raise LLMError("empty choices in response")
            return choices[0].get("message", {"role": "assistant", "content": ""})
        raise LLMError(f"exhausted {self.retries} retries: {last_err}")


class ScriptedLLM(LLMClient):
    """Deterministic scripted LLM for hermetic tests.

    `script` is a list of assistant replies (str or message dict) returned in
    order; after exhaustion returns `fallback`. Records received prompts in
    `.calls` for assertions.
    """

    def __init__(self, script: Sequence[Any], fallback: str = ""):
        self.script = list(script)
        self.fallback = fallback
        self.calls: List[dict] = []
        self._idx = 0

    def chat(
        self,
        model: str,
        max_tokens: int,
        messages: Sequence[ChatMessage],
        tools: Optional[List[dict]] = None,
        temperature: float = 0.0,
        response_format: Optional[dict] = None,
    ) -> ChatMessage:
        self.calls.append(
            {"model": model, "max_tokens": max_tokens, "messages": list(messages), "tools": tools}
        )
        if self._idx < len(self.script):
            item = self.script[self._idx]
            self._idx += 1
        else:
            item = self.fallback
        if isinstance(item, dict):
            return item
        return {"role": "assistant", "content": str(item)}


class LocalEngineClient(LLMClient):
    """In-process MI355X engine client. Lazily builds the engine on first use.

    This is the replacement for the reference's process/network boundary at
    pkg/llms/openai.go:79 — messages go directly to
    opsagent_amd.engine.openai_api.ChatCompletionAPI with zero serialization,
    and the engine keeps per-session KV cache across ReAct iterations.
    """

    def __init__(self, engine_config: Optional[dict] = None):
        self._engine_config = engine_config or {}
        self._api = None

    def _get_api(self):
        if self._api is None:
            from opsagent_amd.engine.openai_api import ChatCompletionAPI

            self._api = ChatCompletionAPI.get_or_create(self._engine_config)
        return self._api

    def chat(
        self,
        model: str,
        max_tokens: int,
        messages: Sequence[ChatMessage],
        tools: Optional[List[dict]] = None,
        temperature: float = 0.0,
        response_format: Optional[dict] = None,
    ) -> ChatMessage:
        api = self._get_api()
        resp = api.create(
            model=model,
            messages=list(messages),
            max_tokens=max_tokens,
            tools=tools,
            temperature=temperature,
            response_format=response_format,
        )
        return resp["choices"][0]["message"]


def new_client(
    api_key: str = "",
    base_url: str = "local",
    engine_config: Optional[dict] = None,
) -> LLMClient:
    """Factory (ref NewOpenAIClient, openai.go:38): "local" → in-process engine."""
    if base_url in ("", "local", "engine", "in-process"):
        return LocalEngineClient(engine_config)
    return RemoteOpenAIClient(api_key, base_url)
