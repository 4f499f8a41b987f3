"""Token budgeting utilities.

Capability parity with /root/reference/pkg/llms/tokens.go:26-144:
per-model context-length table, token counting with per-message overhead,
`constrict_messages` (drop oldest non-system messages until the budget fits),
and `constrict_prompt` (drop the first third of lines until under the limit).

Counting is PER-MODEL (the reference counts with tiktoken per model,
tokens.go:60-107; tiktoken's assets are network-fetched, which this offline
build does not assume):
  * local engine models — the running engine's own tokenizer (exact), or the
    byte-level tokenizer before the engine is up (exact for the byte vocab);
  * remote models (gpt-*, anything unknown) — the bundled 32k byte-level BPE
    (assets/tokenizer-32k.json). It is not cl100k, but as a trained subword
    vocabulary its counts track a provider tokenizer far closer than the
    previous ~4-chars/token estimate; the estimate remains the last-resort
    fallback if the asset is missing.
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional

TOKEN_LIMITS: Dict[str, int] = {
    # local engine models
    "llama3-8b": 8192,
    "llama3-70b": 8192,
    "llama3-tiny": 8192,
    "deepseek-moe-small": 8192,
    "deepseek-moe": 8192,
    # remote-model table kept for compatibility (ref tokens.go:26-46)
    "gpt-4": 8192,
    "gpt-4-32k": 32768,
    "gpt-4o": 128000,
    "gpt-4o-mini": 128000,
    "gpt-3.5-turbo": 16385,
    "gpt-3.5-turbo-instruct": 4096,
}

DEFAULT_TOKEN_LIMIT = 4096  # ref tokens.go:49-56
_PER_MESSAGE_OVERHEAD = 4   # role + separators, ref tokens.go:60-107 idea


def get_token_limits(model: str) -> int:
    return TOKEN_LIMITS.get(model, DEFAULT_TOKEN_LIMIT)


_LOCAL_PREFIXES = ("llama3-", "deepseek-")

_tokenizer = None          # byte-level (local default)
_remote_tokenizer = None   # bundled BPE for remote-model counting


def _get_tokenizer():
    global _tokenizer
    if _tokenizer is None:
        try:
            from opsagent_amd.engine.tokenizer import ByteTokenizer

            _tokenizer = ByteTokenizer()
        except Exception:
            _tokenizer = False
    return _tokenizer or None


def _get_remote_tokenizer():
    global _remote_tokenizer
    if _remote_tokenizer is None:
        try:
            from opsagent_amd.engine.tokenizer import BPETokenizer

            path = os.path.join(
                os.path.dirname(os.path.dirname(os.path.dirname(
                    os.path.abspath(__file__)))),
                "assets", "tokenizer-32k.json",
            )
            _remote_tokenizer = (
                BPETokenizer(path) if os.path.isfile(path) else False
            )
        except Exception:
            _remote_tokenizer = False
    return _remote_tokenizer or None


def _counting_tokenizer(model: Optional[str]):
    if model and any(model.startswith(p) for p in _LOCAL_PREFIXES):
        try:
            from opsagent_amd.engine.openai_api import ChatCompletionAPI

            api = ChatCompletionAPI.instance()
            if api is not None:
                return api.engine.tokenizer  # exact: what the engine consumes
        except Exception:
            pass
        return _get_tokenizer()
    return _get_remote_tokenizer() or _get_tokenizer()


def count_text_tokens(text: str, model: Optional[str] = None) -> int:
    tok = _counting_tokenizer(model)
    if tok is not None:
        return len(tok.encode(text, add_bos=False))
    return max(1, len(text) // 4)


def count_tokens(messages: List[dict], model: Optional[str] = None) -> int:
    total = 0
    for m in messages:
        total += _PER_MESSAGE_OVERHEAD
        total += count_text_tokens(str(m.get("content", "")), model)
    return total + 2


def constrict_messages(messages: List[dict], model: str, max_tokens: int) -> List[dict]:
    """Drop oldest non-system messages until the conversation + reply budget fits
    the model context (ref tokens.go:110-125)."""
    limit = get_token_limits(model)
    budget = limit - max_tokens
    if budget <= 0:
        return messages[-1:]
    msgs = list(messages)
    while len(msgs) > 1 and count_tokens(msgs, model) > budget:
        # drop the first non-system message
        for i, m in enumerate(msgs):
            if m.get("role") != "system":
                del msgs[i]
                break
        else:
            break
    return msgs


def constrict_prompt(prompt: str, model: str, max_tokens: int) -> str:
    """Drop the first third of lines until the prompt fits (ref tokens.go:128-144)."""
    while count_text_tokens(prompt, model) > max_tokens:
        lines = prompt.splitlines()
        if len(lines) <= 1:
            # single huge line: hard truncate from the front
            approx_chars = max_tokens * 4
            return prompt[-approx_chars:]
        prompt = "\n".join(lines[len(lines) // 3:])
    return prompt
