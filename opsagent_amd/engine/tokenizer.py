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

    byte_level_ids = True  # ids 0..255 ARE output bytes (fast paths rely on it)
    max_token_bytes = 1    # longest byte image of any single token

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

    def bytes_to_ids(self, data: bytes) -> List[int]:
        """Token ids that produce exactly `data` (byte-level: identity)."""
        return list(data)

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


def _bytes_to_unicode() -> Dict[int, str]:
    """GPT-2/Llama-3 byte-level BPE alphabet: the reversible byte → printable
    unicode mapping used inside tokenizer.json vocabularies."""
    bs = (
        list(range(ord("!"), ord("~") + 1))
        + list(range(0xA1, 0xAD))
        + list(range(0xAE, 0x100))
    )
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return {b: chr(c) for b, c in zip(bs, cs)}


class BPETokenizer(ByteTokenizer):
    """A real trained BPE loaded from a HF `tokenizer.json` (via the
    `tokenizers` library), exposing the same interface the engine uses.
    Byte-level BPE merges mean token ids are NOT raw bytes, so the grammar
    FSM gets per-token byte expansions via token_bytes() and the engine's
    byte-level-only fast paths (jump-ahead) switch themselves off on
    `byte_level_ids = False`."""

    byte_level_ids = False

    _BOS_CANDIDATES = ["<|begin_of_text|>", "<s>", "<|endoftext|>"]
    _EOS_CANDIDATES = ["<|end_of_text|>", "</s>", "<|endoftext|>"]
    _EOT_CANDIDATES = ["<|eot_id|>", "<|im_end|>", "</s>", "<|end_of_text|>"]

    def __init__(self, path: str, template: str = "llama3") -> None:
        from tokenizers import Tokenizer  # offline wheelhouse

        super().__init__(template=template)
        self._tok = Tokenizer.from_file(path)
        self.vocab_size = self._tok.get_vocab_size()
        vocab = self._tok.get_vocab()

        def find(cands, default):
            for c in cands:
                if c in vocab:
                    return vocab[c]
            return default

        self.bos_id = find(self._BOS_CANDIDATES, 0)
        self.eos_id = find(self._EOS_CANDIDATES, 0)
        self.eot_id = find(self._EOT_CANDIDATES, self.eos_id)
        self.stop_ids = {self.eos_id, self.eot_id}
        # byte expansion per token for the grammar FSM
        byte_dec = {c: b for b, c in _bytes_to_unicode().items()}
        self._token_bytes: List[bytes] = []
        id_to_tok = {i: t for t, i in vocab.items()}
        special_ids = {
            vocab[c]
            for c in (
                self._BOS_CANDIDATES + self._EOS_CANDIDATES + self._EOT_CANDIDATES
                + SPECIAL_TOKENS
            )
            if c in vocab
        }
        for i in range(self.vocab_size):
            t = id_to_tok.get(i, "")
            if i in special_ids or (t.startswith("<|") and t.endswith("|>")):
                self._token_bytes.append(b"")
            elif t and all(ch in byte_dec for ch in t):
                self._token_bytes.append(bytes(byte_dec[ch] for ch in t))
            else:
                # sentencepiece-style piece: '▁' marks a space
                self._token_bytes.append(t.replace("▁", " ").encode("utf-8"))
        self.max_token_bytes = max(
            (len(b) for b in self._token_bytes), default=1
        ) or 1

    def encode(self, text: str, add_bos: bool = False) -> List[int]:
        ids = self._tok.encode(text, add_special_tokens=False).ids
        return ([self.bos_id] + ids) if add_bos else ids

    def bytes_to_ids(self, data: bytes) -> List[int]:
        return self._tok.encode(
            data.decode("utf-8", errors="replace"), add_special_tokens=False
        ).ids

    def decode(self, ids: Sequence[int]) -> str:
        return self._tok.decode(list(ids), skip_special_tokens=False)

    def decode_text(self, ids: Sequence[int]) -> str:
        return b"".join(self.token_bytes(t) for t in ids).decode(
            "utf-8", errors="replace"
        )

    def token_bytes(self, token_id: int) -> bytes:
        if 0 <= token_id < len(self._token_bytes):
            return self._token_bytes[token_id]
        return b""


def get_tokenizer(path: Optional[str] = None, template: str = "llama3") -> ByteTokenizer:
    """Byte-level tokenizer by default; a trained BPE when a tokenizer.json
    path is configured (`engine.tokenizer` in config.yaml)."""
    if path:
        return BPETokenizer(path, template=template)
    return ByteTokenizer(template=template)
