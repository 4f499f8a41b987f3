"""Byte-level tokenizer + Llama-3-style chat template.

The reference never tokenizes for generation (its only token math is tiktoken
counting, ref pkg/llms/tokens.go); the local engine needs a real tokenizer.
With no network for tokenizer assets, the default is a deterministic,
reversible byte-level tokenizer: ids 0..255 are raw bytes, followed by the
special tokens. A trained BPE can be dropped in via `tokenizers` (installed)
when a tokenizer.json is available — the engine only uses this interface:

    encode(text, add_bos) -> List[int]
    decode(ids) -> str
    token_bytes(id) -> bytes  (for the grammar FSM)
    apply_chat_template(messages, tools) -> str
"""

from __future__ import annotations

import json
from typing import Dict, List, Optional, Sequence

BOS = "<|begin_of_text|>"
EOS = "<|end_of_text|>"
START_HEADER = "<|start_header_id|>"
END_HEADER = "<|end_header_id|>"
EOT = "<|eot_id|>"  # end of turn
PAD = "<|pad|>"

SPECIAL_TOKENS = [BOS, EOS, START_HEADER, END_HEADER, EOT, PAD]


class ByteTokenizer:
    """ids 0..255 = bytes; 256.. = special tokens.

    `template` selects the chat rendering: "llama3" (header-id turns) or
    "deepseek" (User:/Assistant: turns with <|end_of_text|> separators) —
    SURVEY.md §2b "Tokenizer + chat templating: Llama-3 / DeepSeek templates".
    """

    def __init__(self, template: str = "llama3") -> None:
        self.template = template
        self.special: Dict[str, int] = {s: 256 + i for i, s in enumerate(SPECIAL_TOKENS)}
        self.special_rev = {v: k for k, v in self.special.items()}
        self.bos_id = self.special[BOS]
        self.eos_id = self.special[EOS]
        self.eot_id = self.special[EOT]
        self.pad_id = self.special[PAD]
        # realizable vocab (max id + 1); the MODEL vocab may be larger (padded)
        self.vocab_size = 256 + len(SPECIAL_TOKENS)
        self.stop_ids = {self.eos_id, self.eot_id}

    def encode(self, text: str, add_bos: bool = False) -> List[int]:
        ids: List[int] = [self.bos_id] if add_bos else []
        i = 0
        n = len(text)
        while i < n:
            if text[i] == "<":
                matched = False
                for s, sid in self.special.items():
                    if text.startswith(s, i):
                        ids.append(sid)
                        i += len(s)
                        matched = True
                        break
                if matched:
                    continue
            ch = text[i]
            ids.extend(ch.encode("utf-8"))
            i += 1
        return ids

    def decode(self, ids: Sequence[int]) -> str:
        out: List[str] = []
        buf = bytearray()
        for t in ids:
            if t < 256:
                buf.append(t)
            else:
                if buf:
                    out.append(buf.decode("utf-8", errors="replace"))
                    buf.clear()
                name = self.special_rev.get(t)
                if name and name not in (PAD,):
                    out.append(name)
        if buf:
            out.append(buf.decode("utf-8", errors="replace"))
        return "".join(out)

    def decode_text(self, ids: Sequence[int]) -> str:
        """Decode, dropping ALL special tokens (for user-facing content)."""
        buf = bytearray(t for t in ids if t < 256)
        return buf.decode("utf-8", errors="replace")

    def token_bytes(self, token_id: int) -> bytes:
        """Raw bytes a token contributes to the text stream ('' for specials)."""
        if token_id < 256:
            return bytes([token_id])
        return b""

    # ---- chat templating (Llama-3 style) ---------------------------------
    def apply_chat_template(
        self,
        messages: Sequence[dict],
        tools: Optional[List[dict]] = None,
        add_generation_prompt: bool = True,
    ) -> str:
        """Render an OpenAI-style message list into the prompt string.

        Tool declarations are injected into the system turn; tool results
        appear as `tool` turns; an assistant message with tool_calls is
        serialized as its JSON wire form so the model sees its own calls.
        """
        msgs = list(messages)
        if self.template == "deepseek":
            return self._deepseek_template(msgs, tools, add_generation_prompt)
        parts: List[str] = [BOS]
        if tools:
            tool_decl = (
                "\n\nYou may call tools. Available tools (JSON schema):\n"
                + json.dumps(tools, ensure_ascii=False)
                + '\nTo call a tool respond with JSON: {"tool_calls": [{"name": '
                '"<tool>", "arguments": {...}}]}. Otherwise answer normally.'
            )
            if msgs and msgs[0].get("role") == "system":
                msgs[0] = dict(msgs[0])
                msgs[0]["content"] = (msgs[0].get("content") or "") + tool_decl
            else:
                msgs.insert(0, {"role": "system", "content": "You are a helpful assistant." + tool_decl})
        for m in msgs:
            role = m.get("role", "user")
            content = m.get("content")
            if role == "assistant" and m.get("tool_calls"):
                calls = [
                    {
                        "name": tc.get("function", {}).get("name", ""),
                        "arguments": tc.get("function", {}).get("arguments", ""),
                    }
                    for tc in m["tool_calls"]
                ]
                content = json.dumps({"tool_calls": calls}, ensure_ascii=False)
            elif role == "tool":
                content = json.dumps(
                    {"tool_call_id": m.get("tool_call_id", ""), "result": content},
                    ensure_ascii=False,
                )
            parts.append(f"{START_HEADER}{role}{END_HEADER}\n\n{content or ''}{EOT}")
        if add_generation_prompt:
            parts.append(f"{START_HEADER}assistant{END_HEADER}\n\n")
        return "".join(parts)


    def _deepseek_template(self, msgs, tools, add_generation_prompt: bool) -> str:
        parts: List[str] = [BOS]
        if tools:
            decl = (
                "You may call tools. Available tools (JSON schema):\n"
                + json.dumps(tools, ensure_ascii=False)
                + '\nTo call a tool respond with JSON: {"tool_calls": [{"name": '
                '"<tool>", "arguments": {...}}]}.'
            )
            msgs = list(msgs)
            if msgs and msgs[0].get("role") == "system":
                msgs[0] = dict(msgs[0])
                msgs[0]["content"] = (msgs[0].get("content") or "") + "\n\n" + decl
            else:
                msgs.insert(0, {"role": "system", "content": decl})
        role_map = {"system": "", "user": "User: ", "assistant": "Assistant: ",
                    "tool": "Observation: "}
        for m in msgs:
            role = m.get("role", "user")
            content = m.get("content")
            if role == "assistant" and m.get("tool_calls"):
                calls = [
                    {"name": tc.get("function", {}).get("name", ""),
                     "arguments": tc.get("function", {}).get("arguments", "")}
                    for tc in m["tool_calls"]
                ]
                content = json.dumps({"tool_calls": calls}, ensure_ascii=False)
            parts.append(f"{role_map.get(role, '')}{content or ''}\n\n")
        if add_generation_prompt:
            parts.append("Assistant: ")
        return "".join(parts)


def get_tokenizer(path: Optional[str] = None, template: str = "llama3") -> ByteTokenizer:
    # future: load a real BPE via `tokenizers` when a tokenizer.json is given
    return ByteTokenizer(template=template)
