from opsagent_amd.llm.client import (
    ChatMessage,
    LLMClient,
    LLMError,
    RemoteOpenAIClient,
    ScriptedLLM,
    new_client,
)
from opsagent_amd.llm.tokens import (
    constrict_messages,
    constrict_prompt,
    count_tokens,
    get_token_limits,
)

__all__ = [
    "ChatMessage",
    "LLMClient",
    "LLMError",
    "RemoteOpenAIClient",
    "ScriptedLLM",
    "new_client",
    "constrict_messages",
    "constrict_prompt",
    "count_tokens",
    "get_token_limits",
]
