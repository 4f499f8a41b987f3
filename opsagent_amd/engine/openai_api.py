"""OpenAI chat-completions + function-calling API over the local engine.

The wire format matches what the reference's Go client and swarm-style flows
expect (SURVEY.md hard part (e)): request {model, messages, max_tokens,
tools?, temperature?, response_format?} → response {choices: [{message:
{role, content, tool_calls?}, finish_reason}], usage}.

Grammar selection (engine.grammar = auto):
  * tools present            → TOOLCALLS schema (tool_calls JSON forced)
  * response_format json     → generic JSON grammar
  * system prompt demands the ToolPrompt schema (contains '"final_answer"')
                             → TOOLPROMPT schema
  * otherwise                → unconstrained

With any grammar active the emitted text ALWAYS parses — the tool-call JSON
validity metric is structural, not statistical (vs the reference's post-hoc
repair in pkg/utils/json.go).
"""

from __future__ import annotations

import json
import threading
import time
import uuid
from typing import Any, Dict, List, Optional

from opsagent_amd.engine.engine import LLMEngine, SamplingParams
from opsagent_amd.engine.grammar import GrammarMode
from opsagent_amd.utils.logging import get_logger
from opsagent_amd.utils.perf import get_perf_stats

log = get_logger("openai_api")

_instance: Optional["ChatCompletionAPI"] = None
_instance_lock = threading.Lock()

_FINISH_MAP = {
    "stop": "stop",
    "grammar_complete": "stop",
    "grammar_forced_complete": "stop",
    "length": "length",
    "max_seq_len": "length",
    "grammar_dead_end": "stop",
}


class ChatCompletionAPI:
    def __init__(self, engine_cfg: Optional[dict] = None):
        from opsagent_amd.engine.serving import EngineLoop

        self.cfg = dict(engine_cfg or {})
        self.grammar_mode_cfg = str(self.cfg.get("grammar", "auto"))
        self.engine = LLMEngine(self.cfg)
        self.loop = EngineLoop(self.engine)

    # -- singleton ------------------------------------------------------
    @classmethod
    def get_or_create(cls, engine_cfg: Optional[dict] = None) -> "ChatCompletionAPI":
        global _instance
        if _instance is None:
            with _instance_lock:
                if _instance is None:
                    _instance = cls(engine_cfg)
        return _instance

    @classmethod
    def instance(cls) -> Optional["ChatCompletionAPI"]:
        return _instance

    @classmethod
    def reset_instance(cls) -> None:
        global _instance
        _instance = None

    # -- grammar choice --------------------------------------------------
    def _pick_grammar(
        self,
        messages: List[dict],
        tools: Optional[List[dict]],
        response_format: Optional[dict],
        tool_choice: Any = None,
    ) -> Optional[GrammarMode]:
        mode = self.grammar_mode_cfg
        if mode == "off":
            return None
        if mode == "json":
            return GrammarMode.JSON
        if mode == "toolprompt":
            return GrammarMode.TOOLPROMPT
        # auto
        if tools and tool_choice != "none":
            # "auto"/"required"/named function all constrain to the
            # tool_calls schema; "none" keeps the declarations in context
            # but generation unconstrained
            return GrammarMode.TOOLCALLS
        if response_format and response_format.get("type") in ("json_object", "json_schema"):
            return GrammarMode.JSON
        sys_text = " ".join(
            str(m.get("content", "")) for m in messages if m.get("role") == "system"
        )
        if '"final_answer"' in sys_text:
            return GrammarMode.TOOLPROMPT
        return None

    def create_stream(
        self,
        model: str,
        messages: List[dict],
        max_tokens: int = 1024,
        temperature: float = 0.0,
        response_format: Optional[dict] = None,
        stop: Optional[List[str]] = None,
        top_p: float = 1.0,
        top_k: int = 0,
        presence_penalty: float = 0.0,
        frequency_penalty: float = 0.0,
        logit_bias: Optional[dict] = None,
    ):
        """Streaming chat completion: yields OpenAI `chat.completion.chunk`
        dicts as tokens are sampled (tool-call streaming is not offered; the
        tools path buffers for a complete, parseable call).

        With `stop` sequences active, the longest stop length is HELD BACK
        from the stream so matched stop text is never emitted (OpenAI
        semantics); the holdback is reconciled against the engine's trimmed
        final output when generation ends."""
        tok = self.engine.tokenizer
        grammar = self._pick_grammar(messages, None, response_format)
        prompt = tok.apply_chat_template(messages)
        prompt_ids = tok.encode(prompt)
        stops = [stop] if isinstance(stop, str) else (stop or [])
        params = SamplingParams(
            max_new_tokens=max_tokens,
            temperature=temperature if temperature and temperature > 1e-5 else 0.0,
            top_p=float(top_p),
            top_k=int(top_k),
            presence_penalty=float(presence_penalty),
            frequency_penalty=float(frequency_penalty),
            logit_bias=logit_bias,
            grammar=grammar,
            stop=stops or None,
        )
        # Holdback covers the worst case trim: the engine trims WHOLE tokens
        # when a stop matches, so up to (max stop bytes - 1) + (max token
        # bytes - 1) already-queued bytes can disappear from the final output
        # (ADVICE r1). Byte-level vocab: max_token_bytes == 1 → plain stop len.
        holdback = max((len(s.encode("utf-8")) for s in stops), default=0)
        if holdback:
            holdback += max(0, getattr(tok, "max_token_bytes", 1) - 1)
        token_iter, fut = self.loop.submit_stream(prompt_ids, params)
        cid = f"chatcmpl-{uuid.uuid4().hex[:16]}"
        created = int(time.time())

        def chunk(delta: Dict[str, Any], finish=None):
            return {
                "id": cid,
                "object": "chat.completion.chunk",
                "created": created,
                "model": model or self.engine.spec.name,
                "choices": [{"index": 0, "delta": delta, "finish_reason": finish}],
            }

        yield chunk({"role": "assistant", "content": ""})
        # decode incrementally; buffer partial UTF-8 sequences + stop holdback
        buf = bytearray()
        emitted = 0  # bytes emitted so far
        for t in token_iter:
            b = tok.token_bytes(t)
            if not b:
                continue
            buf.extend(b)
            flush_n = len(buf) - holdback
            if flush_n <= 0:
                continue
            # flush up to flush_n bytes, backing off to a UTF-8 boundary
            k = flush_n
            text = None
            while k > max(0, flush_n - 4):
                try:
                    text = bytes(buf[:k]).decode("utf-8")
                    break
                except UnicodeDecodeError:
                    k -= 1
            if text is None or not text:
                continue
            del buf[:k]
            emitted += k
            yield chunk({"content": text})
        _ids, reason = fut.result()
        # reconcile: the engine's final output is authoritative (stop text
        # trimmed); emit whatever trails the bytes already streamed
        final_bytes = tok.decode_text(_ids).encode("utf-8")
        if len(final_bytes) > emitted:
            yield chunk(
                {"content": final_bytes[emitted:].decode("utf-8", errors="replace")}
            )
        yield chunk({}, finish=_FINISH_MAP.get(reason, "stop"))

    def stats(self) -> Dict[str, Any]:
        """Engine observability (backs GET /api/engine/stats)."""
        eng = self.engine
        return {
            "model": eng.spec.name,
            "device": eng.device,
            "tp": eng.tp,
            "dtype": str(eng.dtype),
            "max_batch_size": eng.max_batch,
            "running": len(eng.running),
            "waiting": len(eng.waiting),
            "kv": eng.cache_stats(),
            "kv_blocks_total": eng.kv.num_blocks,
            "kv_block_tokens": eng.kv.block_size,
            "graphs_captured": sorted(eng._graphs.keys()),
            "spec_decode": dict(eng.spec_stats, ema=round(eng.spec_ema, 4)),
            "healthy": self.loop.healthy,
            "last_step_ms": round(self.loop.last_step_ms, 3),
        }

    def create_completion(
        self,
        model: str,
        prompt: str,
        max_tokens: int = 256,
        temperature: float = 0.0,
        stop: Optional[List[str]] = None,
        top_p: float = 1.0,
        n: int = 1,
    ) -> Dict[str, Any]:
        """Legacy `/v1/completions` (raw text in, text out — no chat
        template), for clients that still use the completions API."""
        tok = self.engine.tokenizer
        prompt_ids = tok.encode(str(prompt), add_bos=True)
        params = SamplingParams(
            max_new_tokens=max_tokens,
            temperature=temperature if temperature and temperature > 1e-5 else 0.0,
            top_p=float(top_p),
            stop=[stop] if isinstance(stop, str) else stop,
        )
        n = max(1, int(n))
        futs = [self.loop.submit(prompt_ids, params) for _ in range(n)]
        results = [f.result() for f in futs]
        total = sum(len(ids) for ids, _ in results)
        return {
            "id": f"cmpl-{uuid.uuid4().hex[:16]}",
            "object": "text_completion",
            "created": int(time.time()),
            "model": model or self.engine.spec.name,
            "choices": [
                {
                    "index": i,
                    "text": tok.decode_text(ids),
                    "finish_reason": _FINISH_MAP.get(reason, "stop"),
                    "logprobs": None,
                }
                for i, (ids, reason) in enumerate(results)
            ],
            "usage": {
                "prompt_tokens": len(prompt_ids),
                "completion_tokens": total,
                "total_tokens": len(prompt_ids) + total,
            },
        }

    # -- main entry -------------------------------------------------------
    def create(
        self,
        model: str,
        messages: List[dict],
        max_tokens: int = 1024,
        tools: Optional[List[dict]] = None,
        temperature: float = 0.0,
        response_format: Optional[dict] = None,
        stop: Optional[List[str]] = None,
        top_p: float = 1.0,
        top_k: int = 0,
        presence_penalty: float = 0.0,
        frequency_penalty: float = 0.0,
        logit_bias: Optional[dict] = None,
        n: int = 1,
        logprobs: bool = False,
        top_logprobs: int = 0,
        tool_choice: Any = None,
    ) -> Dict[str, Any]:
        perf = get_perf_stats()
        t0 = time.perf_counter()
        tok = self.engine.tokenizer
        if isinstance(tool_choice, dict):
            # named function: narrow the declared tools to that function so
            # the constrained tool_calls JSON can only name it
            want = tool_choice.get("function", {}).get("name")
            if want and tools:
                narrowed = [
                    t for t in tools
                    if t.get("function", {}).get("name") == want
                ]
                tools = narrowed or tools
        grammar = self._pick_grammar(messages, tools, response_format, tool_choice)
        prompt = tok.apply_chat_template(messages, tools=tools)
        prompt_ids = tok.encode(prompt)
        params = SamplingParams(
            max_new_tokens=max_tokens,
            temperature=temperature if temperature and temperature > 1e-5 else 0.0,
            top_p=float(top_p),
            top_k=int(top_k),
            presence_penalty=float(presence_penalty),
            frequency_penalty=float(frequency_penalty),
            logit_bias=logit_bias,
            grammar=grammar,
            stop=[stop] if isinstance(stop, str) else stop,
            logprobs=bool(logprobs),
            top_logprobs=int(top_logprobs or 0),
            tool_names=(
                [
                    t.get("function", {}).get("name")
                    for t in (tools or [])
                    if t.get("function", {}).get("name")
                ]
                if grammar == GrammarMode.TOOLCALLS
                else None
            ),
        )
        # concurrent callers (and the n>1 fan-out) batch together in the
        # engine loop's continuous batches
        n = max(1, int(n))
        futs = [self.loop.submit(prompt_ids, params) for _ in range(n)]
        results = [f.result() for f in futs]
        lp_contents = [getattr(f, "logprob_content", None) for f in futs]
        perf.record_metric("engine_chat_ms", (time.perf_counter() - t0) * 1000.0)

        choices = []
        total_completion = 0
        for idx, (out_ids, finish_reason) in enumerate(results):
            total_completion += len(out_ids)
            perf.record_metric("engine_completion_tokens", float(len(out_ids)))
            text = tok.decode_text(out_ids)
            message: Dict[str, Any] = {"role": "assistant", "content": text}
            if grammar == GrammarMode.TOOLCALLS:
                try:
                    obj = json.loads(text)
                    calls = obj.get("tool_calls", [])
                    message = {
                        "role": "assistant",
                        "content": None,
                        "tool_calls": [
                            {
                                "id": f"call_{uuid.uuid4().hex[:12]}",
                                "type": "function",
                                "function": {
                                    "name": c.get("name", ""),
                                    "arguments": json.dumps(c.get("arguments", {}))
                                    if not isinstance(c.get("arguments"), str)
                                    else c["arguments"],
                                },
                            }
                            for c in calls
                        ],
                    }
                except json.JSONDecodeError:
                    # only reachable when generation stopped early (length)
                    message = {"role": "assistant", "content": text}
            finish = _FINISH_MAP.get(finish_reason, "stop")
            if message.get("tool_calls"):
                finish = "tool_calls"
            choice: Dict[str, Any] = {
                "index": idx, "message": message, "finish_reason": finish,
            }
            if logprobs and lp_contents[idx] is not None:
                def _wire(e: dict) -> Dict[str, Any]:
                    b = tok.token_bytes(e["token_id"])
                    return {
                        "token": b.decode("utf-8", errors="replace"),
                        "logprob": e["logprob"],
                        "bytes": list(b),
                    }

                choice["logprobs"] = {
                    "content": [
                        dict(
                            _wire(e),
                            top_logprobs=[_wire(t) for t in e.get("top", [])],
                        )
                        for e in lp_contents[idx]
                    ]
                }
            choices.append(choice)

        usage = {
            "prompt_tokens": len(prompt_ids),
            "completion_tokens": total_completion,
            "total_tokens": len(prompt_ids) + total_completion,
        }
        truncated = max(
            (getattr(f, "truncated_prompt_tokens", 0) or 0) for f in futs
        )
        if truncated:
            # non-standard field: tells the caller the engine dropped the
            # oldest `truncated` prompt tokens to fit the context window
            usage["prompt_tokens_truncated"] = truncated
        return {
            "id": f"chatcmpl-{uuid.uuid4().hex[:16]}",
            "object": "chat.completion",
            "created": int(time.time()),
            "model": model or self.engine.spec.name,
            "choices": choices,
            "usage": usage,
        }
