"""ReAct agent core.

Capability parity with /root/reference/pkg/assistants/simple.go:287-616:

  * `assistant(...)` sends the conversation, parses the reply into a
    ToolPrompt (question/thought/action/observation/final_answer), and
    iterates: dispatch tool → truncate observation → append the re-serialized
    ToolPrompt as a *user* message → chat again — up to max_iterations.
  * An unparsable first reply is treated as the final answer (simple.go:367-382).
  * A final answer is accepted only if it is not a template placeholder and an
    observation exists (simple.go:414-419); template detection mirrors
    isTemplateValue (simple.go:624-657).
  * Tool failures become observations so the model can self-correct
    ("Tool X failed ... refine the inputs", simple.go:449-455,473-481).
  * Observations are truncated to 1024 tokens (simple.go:495).
  * On a mid-loop parse failure the loop falls back to a summarize turn
    (simple.go:558-600).

MI355X note: when the client is the LocalEngineClient, each `chat` call hits
the in-process engine, which caches the KV prefix of the growing conversation
across iterations (prefix reuse), so ReAct iterations pay only for the new
tokens — the agent-turn-latency lever named in SURVEY.md §2b.
"""

from __future__ import annotations

import dataclasses
import json
import re
from typing import List, Optional, Sequence, Tuple

from opsagent_amd.llm.client import ChatMessage, LLMClient
from opsagent_amd.llm.tokens import constrict_prompt
from opsagent_amd.tools import TOOLS, ToolError
from opsagent_amd.utils.jsonrepair import clean_json, extract_field, parse_json
from opsagent_amd.utils.logging import get_logger
from opsagent_amd.utils.perf import get_perf_stats

log = get_logger("agent")

OBSERVATION_TOKEN_LIMIT = 1024  # ref simple.go:495


@dataclasses.dataclass
class ToolPrompt:
    """The agent's constrained-JSON schema (ref pkg/tools/tool.go:29-38)."""

    question: str = ""
    thought: str = ""
    action_name: str = ""
    action_input: str = ""
    observation: str = ""
    final_answer: str = ""

    @classmethod
    def from_obj(cls, obj: dict) -> "ToolPrompt":
        action = obj.get("action") or {}
        if not isinstance(action, dict):
            action = {}

        def _s(v) -> str:
            if v is None:
                return ""
            if isinstance(v, str):
                return v
            return json.dumps(v, ensure_ascii=False)

        return cls(
            question=_s(obj.get("question")),
            thought=_s(obj.get("thought")),
            action_name=_s(action.get("name")),
            action_input=_s(action.get("input")),
            observation=_s(obj.get("observation")),
            final_answer=_s(obj.get("final_answer")),
        )

    def to_json(self) -> str:
        return json.dumps(
            {
                "question": self.question,
                "thought": self.thought,
                "action": {"name": self.action_name, "input": self.action_input},
                "observation": self.observation,
                "final_answer": self.final_answer,
            },
            ensure_ascii=False,
        )


_TEMPLATE_PATTERNS = re.compile(r"<[^>]{0,80}>|\.\.\.|\bTODO\b|\byour answer\b", re.IGNORECASE)


def is_template_value(s: str) -> bool:
    """Placeholder detector (ref simple.go:624-657): short or <...>-style stubs."""
    s = s.strip()
    if not s:
        return True
    if len(s) < 10:
        return True
    return bool(_TEMPLATE_PATTERNS.fullmatch(s)) or (
        s.startswith("<") and s.endswith(">")
    )


def _parse_reply(content: str) -> Optional[ToolPrompt]:
    obj = parse_json(content)
    if isinstance(obj, dict) and (
        "final_answer" in obj or "action" in obj or "thought" in obj
    ):
        return ToolPrompt.from_obj(obj)
    return None


def assistant(
    client: LLMClient,
    model: str,
    messages: Sequence[ChatMessage],
    max_tokens: int = 2048,
    verbose: bool = False,
    max_iterations: int = 10,
) -> Tuple[str, List[ChatMessage]]:
    """Run the ReAct loop. Returns (final_answer, chat_history).

    `messages` must already contain the system prompt and the user instruction
    (the callers in cli.py / server build them — ref execute.go:225-234).
    """
    perf = get_perf_stats()
    history: List[ChatMessage] = list(messages)

    with perf.trace("assistant_total"):
        perf.start_timer("assistant_first_chat")
        reply = client.chat(model, max_tokens, history)
        perf.stop_timer("assistant_first_chat")
        content = reply.get("content") or ""
        history.append({"role": "assistant", "content": content})

        tp = _parse_reply(content)
        if tp is None:
            # unparsable first reply ⇒ the whole reply is the answer (ref :367-382)
            return content, history

        iterations = 0
        while True:
            iterations += 1
            if iterations > max_iterations:
                # give back whatever final answer we have (ref :407-412)
                return tp.final_answer or tp.thought or content, history

            if tp.final_answer and not is_template_value(tp.final_answer):
                # accept only when evidence exists or no tool was requested (ref :414-419)
                if tp.observation or not tp.action_name:
                    return tp.final_answer, history

            if tp.action_name:
                tool = TOOLS.get(tp.action_name)
                if tool is None:
                    observation = (
                        f"Tool {tp.action_name} is not available. "
                        f"Available tools: {', '.join(sorted(TOOLS))}."
                    )
                else:
                    if verbose:
                        log.info("tool %s(%r)", tp.action_name, tp.action_input[:200])
                    perf.start_timer(f"assistant_tool_{tp.action_name}")
                    try:
                        observation = tool(tp.action_input)
                    except ToolError as e:
                        observation = (
                            f"Tool {tp.action_name} failed: {e}. "
                            "Consider refining the inputs."
                        )
                    except Exception as e:  # noqa: BLE001 — defensive: any tool crash → observation
                        observation = f"Tool {tp.action_name} crashed: {e}."
                    perf.stop_timer(f"assistant_tool_{tp.action_name}")
            else:
                observation = "No action specified and no final answer given; please provide final_answer."

            tp.observation = constrict_prompt(observation, model, OBSERVATION_TOKEN_LIMIT)
            # re-serialized ToolPrompt goes back as a USER message (ref :497-501)
            history.append({"role": "user", "content": tp.to_json()})

            perf.start_timer("assistant_intermediate_chat")
            reply = client.chat(model, max_tokens, history)
            perf.stop_timer("assistant_intermediate_chat")
            content = reply.get("content") or ""
            history.append({"role": "assistant", "content": content})

            next_tp = _parse_reply(content)
            if next_tp is None:
                # summarize fallback (ref :558-600)
                history.append(
                    {
                        "role": "user",
                        "content": "Summarize all the chat history and respond to the "
                        "original question with the ToolPrompt JSON format, filling final_answer.",
                    }
                )
                perf.start_timer("assistant_summarize")
                reply = client.chat(model, max_tokens, history)
                perf.stop_timer("assistant_summarize")
                content = reply.get("content") or ""
                history.append({"role": "assistant", "content": content})
                fa = extract_field(content, "final_answer")
                if fa:
                    return fa, history
                obj = parse_json(clean_json(content))
                if isinstance(obj, dict) and obj.get("final_answer"):
                    return str(obj["final_answer"]), history
                return content, history
            tp = next_tp
