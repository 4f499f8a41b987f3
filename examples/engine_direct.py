"""Drive the engine directly (no HTTP): grammar-constrained generation,
speculative decoding stats, and the prefix cache across turns."""

import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from opsagent_amd.engine.engine import LLMEngine, SamplingParams
from opsagent_amd.engine.grammar import GrammarMode

# llama3-8b bf16 on GPU; swap to "llama3-tiny" to try this on CPU
eng = LLMEngine({"model": "llama3-tiny", "use_hipgraph": False})
tok = eng.tokenizer

convo = "SYSTEM: reply with the ToolPrompt JSON only.\nUSER: check the pods"
ids = tok.encode(convo, add_bos=True)
out, reason = eng.generate(
    ids, SamplingParams(max_new_tokens=200, grammar=GrammarMode.TOOLPROMPT)
)
doc = json.loads(tok.decode_text(out))  # ALWAYS parses
print("finish:", reason)
print("action:", doc["action"])

# second turn reuses the first turn's KV blocks (prefix cache)
ids2 = tok.encode(convo + tok.decode_text(out) + "\nUSER: and the events?",
                  add_bos=True)
eng.generate(ids2, SamplingParams(max_new_tokens=50, grammar=GrammarMode.TOOLPROMPT))
print("kv stats:", eng.cache_stats())
print("speculative decode:", eng.spec_stats)
