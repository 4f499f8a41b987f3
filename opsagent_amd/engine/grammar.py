"""Python wrapper for the C++ grammar FSM (ops/csrc/grammar_fsm.cpp).

Per-sequence constrained-decoding state: each decode step `fill_mask` writes
the allowed-token bitmask (pinned int32 tensor), which is uploaded to the GPU
and fused into the sampling kernel (ops.greedy_sample_masked). Invalid JSON
becomes unrepresentable — replacing the reference's post-hoc JSON repair
(/root/reference/pkg/utils/json.go:16-190).
"""

from __future__ import annotations

import ctypes
import os
from enum import IntEnum
from typing import Optional

import numpy as np
import torch

_GRAMMAR_LIB_PATH = os.environ.get(
    "OPSAGENT_GRAMMAR_LIB",
    os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "ops",
        "libopsagent_grammar.so",
    ),
)

_lib: Optional[ctypes.CDLL] = None


class GrammarMode(IntEnum):
    JSON = 0        # any JSON object
    TOOLPROMPT = 1  # the agent's ToolPrompt schema (ref pkg/tools/tool.go:29-38)
    TOOLCALLS = 2   # OpenAI tool_calls wire schema


def _get_lib() -> ctypes.CDLL:
    global _lib
    if _lib is None:
        if not os.path.exists(_GRAMMAR_LIB_PATH):
            from opsagent_amd.ops.build import build_grammar

            build_grammar()
        lib = ctypes.CDLL(_GRAMMAR_LIB_PATH)
        lib.oa_vocab_create.restype = ctypes.c_void_p
        lib.oa_vocab_create.argtypes = [
            ctypes.c_void_p,
            ctypes.c_void_p,
            ctypes.c_int,
            ctypes.c_int,
        ]
        lib.oa_vocab_destroy.argtypes = [ctypes.c_void_p]
        lib.oa_grammar_create.restype = ctypes.c_void_p
        lib.oa_grammar_create.argtypes = [ctypes.c_int, ctypes.c_void_p]
        lib.oa_grammar_create_names.restype = ctypes.c_void_p
        lib.oa_grammar_create_names.argtypes = [
            ctypes.c_int,
            ctypes.c_void_p,
            ctypes.c_void_p,
            ctypes.c_void_p,
            ctypes.c_int,
        ]
        lib.oa_grammar_destroy.argtypes = [ctypes.c_void_p]
        lib.oa_grammar_reset.argtypes = [ctypes.c_void_p]
        lib.oa_grammar_is_complete.argtypes = [ctypes.c_void_p]
        lib.oa_grammar_is_complete.restype = ctypes.c_int
        lib.oa_grammar_accept_token.argtypes = [ctypes.c_void_p, ctypes.c_int]
        lib.oa_grammar_accept_token.restype = ctypes.c_int
        lib.oa_grammar_fill_mask.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
        lib.oa_grammar_completion.argtypes = [
            ctypes.c_void_p,
            ctypes.c_void_p,
            ctypes.c_int,
        ]
        lib.oa_grammar_completion.restype = ctypes.c_int
        lib.oa_grammar_forced_run.argtypes = [
            ctypes.c_void_p,
            ctypes.c_void_p,
            ctypes.c_int,
            ctypes.c_int,
        ]
        lib.oa_grammar_forced_run.restype = ctypes.c_int
        lib.oa_grammar_forced_bytes.argtypes = [
            ctypes.c_void_p,
            ctypes.c_void_p,
            ctypes.c_int,
        ]
        lib.oa_grammar_forced_bytes.restype = ctypes.c_int
        lib.oa_grammar_check_tokens.argtypes = [
            ctypes.c_void_p,
            ctypes.c_void_p,
            ctypes.c_int,
        ]
        lib.oa_grammar_check_tokens.restype = ctypes.c_int
        lib.oa_grammar_masks_along.argtypes = [
            ctypes.c_void_p,
            ctypes.c_void_p,
            ctypes.c_int,
            ctypes.c_void_p,
        ]
        lib.oa_grammar_masks_along.restype = ctypes.c_int
        _lib = lib
    return _lib


def _get_vocab_handle(tokenizer, model_vocab: int) -> int:
    """Shared C++ vocab table, built once per (tokenizer INSTANCE, model
    vocab) — stored on the instance so two different tokenizer.json files
    can never collide on a cache key."""
    cache = getattr(tokenizer, "_oa_vocab_handles", None)
    if cache is None:
        cache = {}
        tokenizer._oa_vocab_handles = cache
    h = cache.get(model_vocab)
    if h is not None:
        return h
    lib = _get_lib()
    lens = np.zeros(model_vocab, dtype=np.int32)
    chunks = []
    for t in range(min(tokenizer.vocab_size, model_vocab)):
        b = tokenizer.token_bytes(t)
        lens[t] = len(b)
        if b:
            chunks.append(b)
    concat = b"".join(chunks)
    buf = (ctypes.c_uint8 * max(1, len(concat))).from_buffer_copy(concat or b"\x00")
    h = lib.oa_vocab_create(
        lens.ctypes.data_as(ctypes.c_void_p),
        ctypes.cast(buf, ctypes.c_void_p),
        model_vocab,
        tokenizer.eot_id,
    )
    if not h:
        raise RuntimeError("vocab create failed")
    cache[model_vocab] = h
    return h


class GrammarState:
    """One sequence's constrained-decoding FSM (references the shared vocab)."""

    def __init__(
        self,
        tokenizer,
        mode: GrammarMode,
        model_vocab: int,
        tool_names: Optional[list] = None,
    ):
        lib = _get_lib()
        self._lib = lib
        self.vocab = model_vocab
        self.eos_id = tokenizer.eot_id
        vh = _get_vocab_handle(tokenizer, model_vocab)
        names = [str(n).encode("utf-8") for n in (tool_names or []) if n]
        if names and len(names) <= 32:
            # the template's name field only accepts one of the DECLARED
            # tool names — tool_choice / hallucinated-tool protection is
            # structural, not post-hoc
            concat = b"".join(names)
            lens = np.array([len(n) for n in names], dtype=np.int32)
            buf = (ctypes.c_uint8 * max(1, len(concat))).from_buffer_copy(
                concat or b"\x00"
            )
            self._h = lib.oa_grammar_create_names(
                int(mode), vh, ctypes.cast(buf, ctypes.c_void_p),
                lens.ctypes.data_as(ctypes.c_void_p), len(names),
            )
        else:
            self._h = lib.oa_grammar_create(int(mode), vh)
        if not self._h:
            raise RuntimeError("grammar create failed")
        self.mask_words = (model_vocab + 31) // 32
        self._mask_np = np.zeros(self.mask_words, dtype=np.uint32)

    def __del__(self):
        h = getattr(self, "_h", None)
        if h:
            self._lib.oa_grammar_destroy(h)
            self._h = None

    def reset(self) -> None:
        self._lib.oa_grammar_reset(self._h)

    def accept(self, token_id: int) -> bool:
        return self._lib.oa_grammar_accept_token(self._h, int(token_id)) == 0

    def is_complete(self) -> bool:
        return bool(self._lib.oa_grammar_is_complete(self._h))

    def fill_mask_np(self) -> np.ndarray:
        self._lib.oa_grammar_fill_mask(
            self._h, self._mask_np.ctypes.data_as(ctypes.c_void_p)
        )
        return self._mask_np

    def fill_mask_into(self, out_row: torch.Tensor) -> None:
        """Write the mask into an int32 CPU tensor row [mask_words]."""
        m = self.fill_mask_np()
        out_row.copy_(torch.from_numpy(m.view(np.int32)))

    def completion_bytes(self, max_len: int = 4096) -> Optional[bytes]:
        """Shortest byte sequence that completes the document from the current
        state (closing strings/containers, finishing template literals)."""
        buf = (ctypes.c_uint8 * max_len)()
        n = self._lib.oa_grammar_completion(self._h, ctypes.cast(buf, ctypes.c_void_p), max_len)
        if n < 0:
            return None
        return bytes(buf[:n])

    def forced_run(self, max_tokens: int, min_tokens: int) -> bytes:
        """Jump-ahead decoding: the run of grammar-FORCED bytes from the
        current state (singleton allowed set — the masked argmax could only
        ever pick them). ADVANCES the state past the returned bytes; returns
        b"" (state untouched) when fewer than min_tokens are forced, since a
        short jump is not worth a KV catch-up pass."""
        if max_tokens <= 0:
            return b""
        buf = (ctypes.c_uint8 * max_tokens)()
        n = self._lib.oa_grammar_forced_run(
            self._h, ctypes.cast(buf, ctypes.c_void_p), max_tokens, min_tokens
        )
        return bytes(buf[:n])

    def forced_peek(self, max_bytes: int) -> bytes:
        """BPE jump-ahead: the run of grammar-forced bytes from the current
        state, WITHOUT advancing it. The engine tokenizes the run and
        advances via accept() for each whole token kept."""
        if max_bytes <= 0:
            return b""
        buf = (ctypes.c_uint8 * max_bytes)()
        n = self._lib.oa_grammar_forced_bytes(
            self._h, ctypes.cast(buf, ctypes.c_void_p), max_bytes
        )
        return bytes(buf[:n])

    def check_tokens(self, ids: list) -> int:
        """Length of the grammar-legal prefix of `ids`, simulated from the
        current state (state untouched) — pre-filters speculative proposals."""
        if not ids:
            return 0
        arr = np.asarray(ids, dtype=np.int32)
        return int(self._lib.oa_grammar_check_tokens(
            self._h, arr.ctypes.data_as(ctypes.c_void_p), len(arr)
        ))

    def masks_along(self, ids: list) -> np.ndarray:
        """Allowed-token masks along a proposal path: row i = the mask at
        the state reached after accepting ids[:i] (state untouched). Shape
        [m, mask_words] uint32 with m = legal-prefix + 1 rows valid."""
        n = len(ids)
        arr = np.asarray(ids, dtype=np.int32) if n else np.zeros(1, np.int32)
        out = np.zeros((n + 1, self.mask_words), dtype=np.uint32)
        m = self._lib.oa_grammar_masks_along(
            self._h, arr.ctypes.data_as(ctypes.c_void_p), n,
            out.ctypes.data_as(ctypes.c_void_p),
        )
        return out[:m]

    def allowed_bool(self) -> torch.Tensor:
        """Bool [vocab] tensor (CPU path / tests)."""
        m = self.fill_mask_np()
        bits = np.unpackbits(m.view(np.uint8), bitorder="little")[: self.vocab]
        return torch.from_numpy(bits.astype(bool))
