"""Flagship serving benchmark — BASELINE.json metric: agent-turn latency +
tool-call JSON validity %, synthetic k8s-state prompts, random-init weights.

Modes (BASELINE configs):
  turn        (default, config #2/#4): one "step" = one agent turn — prefill a
              synthetic analyze-pod prompt (shared ~250-token system prefix +
              fresh per-turn observation body) + decode a grammar-constrained
              ToolPrompt JSON reply.
  multiturn   (config #3): one conversation; each step appends a synthetic
              tool observation and decodes the next ToolPrompt — the paged-KV
              prefix cache makes each iteration prefill only the new suffix.
  concurrent  (config #5): C concurrent sessions submit turns to the engine
              loop (continuous batching, one hipGraph replay per token for
              the whole batch); value is still per-turn p50 latency and
              throughput is reported in config.

Contract (driver): `python bench.py --gpus N --steps K --warmup W`; for N>1
launched under torch.distributed.run with one rank per GPU (TP=N over RCCL).
W untimed warmup steps, then EXACTLY K timed steps bracketed by barrier +
torch.cuda.synchronize on both sides; MAX over ranks; rank 0 prints ONE JSON
line. TP splits one fixed model -> scaling is "strong".
"""

from __future__ import annotations

import argparse
import json
import os
import random
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch  # noqa: E402


def synthetic_pod_state(rng: random.Random, idx: int) -> str:
    """Fresh kubectl-flavored cluster state per call."""
    pods = []
    for i in range(14):
        name = f"app-{rng.randrange(100,999)}-{rng.choice('abcdef')}{i}"
        status = rng.choice(
            ["Running", "CrashLoopBackOff", "Pending", "ImagePullBackOff", "Running", "Running"]
        )
        restarts = rng.randrange(0, 40)
        pods.append(f"{name:<28} 1/1   {status:<18} {restarts:<4} {rng.randrange(1,200)}d")
    events = [
        f"{rng.randrange(1,59)}m  Warning  {rng.choice(['BackOff','FailedScheduling','Unhealthy','OOMKilling'])}"
        f"  pod/{pods[rng.randrange(len(pods))].split()[0]}  "
        + rng.choice(
            [
                "Back-off restarting failed container",
                "0/3 nodes are available: insufficient memory",
                "Liveness probe failed: HTTP 500",
                "Memory cgroup out of memory",
            ]
        )
        for _ in range(6)
    ]
    return (
        f"turn {idx}: diagnose this cluster state.\n\n$ kubectl get pods -n prod\n"
        + "\n".join(pods)
        + "\n\n$ kubectl get events -n prod --sort-by=lastTimestamp\n"
        + "\n".join(events)
    )


SYSTEM = (
    "You are a Kubernetes operations agent. Diagnose the cluster state "
    "using the tools kubectl, python, trivy, jq, search. Respond with a "
    'single ToolPrompt JSON object {"question", "thought", "action": '
    '{"name", "input"}, "observation", "final_answer"} and nothing else.'
)


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--mode", choices=["turn", "multiturn", "concurrent"], default="turn")
    ap.add_argument("--concurrency", type=int, default=8, help="sessions in concurrent mode")
    ap.add_argument("--moe-dtype", default=None, help="fp8 to quantize MoE experts")
    ap.add_argument("--decode-tokens", type=int, default=128)
    ap.add_argument("--prompt-tokens", type=int, default=1024)
    ap.add_argument("--breakdown", action="store_true", help="print stage perf table to stderr")
    ap.add_argument("--no-fastforward", action="store_true",
                    help="disable grammar jump-ahead decoding (A/B)")
    ap.add_argument("--quantize", default=None,
                    help="fp8: quantize DENSE weights (extra mode; the "
                         "headline bench stays bf16)")
    ap.add_argument("--tokenizer", default="auto",
                    help="'auto' = assets/tokenizer-32k.json when present "
                         "(trained BPE, scripts/train_tokenizer.py), "
                         "'byte' = byte-level, or a tokenizer.json path")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    n_gpus = max(args.gpus, world)
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")

    from opsagent_amd.engine.engine import LLMEngine, SamplingParams
    from opsagent_amd.engine.grammar import GrammarMode
    from opsagent_amd.parallel import state as pstate

    have_gpu = torch.cuda.is_available()
    # real-tokenizer default: the committed 32k byte-level
    # BPE; --tokenizer byte reverts to the raw byte vocabulary
    tok_path = None
    if args.tokenizer == "auto":
        from opsagent_amd.engine.config import get_model_spec

        cand = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                            "assets", "tokenizer-32k.json")
        # the 32k BPE needs a model vocab that covers its ids (tiny CPU
        # test models have 512-entry embeddings)
        if os.path.isfile(cand) and get_model_spec(args.model).vocab_size >= 32000:
            tok_path = cand
    elif args.tokenizer not in ("byte", "none", ""):
        tok_path = args.tokenizer
    eng_cfg = {
        "model": args.model,
        "dtype": "bf16",
        "max_seq_len": 8192,
        "kv_block_size": 32,
        "max_batch_size": max(16, args.concurrency),
        "use_hipgraph": True,
        "seed": 1234,
        "grammar_fastforward": not args.no_fastforward,
        "tokenizer": tok_path,
    }
    if args.quantize:
        eng_cfg["quantize"] = args.quantize
    if args.moe_dtype:
        eng_cfg["moe_dtype"] = args.moe_dtype
    eng = LLMEngine(eng_cfg)
    tok = eng.tokenizer
    rng = random.Random(0)

    def make_prompt(i: int) -> list:
        body = synthetic_pod_state(rng, i)
        while True:
            text = tok.apply_chat_template(
                [
                    {"role": "system", "content": SYSTEM},
                    {"role": "user", "content": body},
                ]
            )
            ids = tok.encode(text)
            if len(ids) >= args.prompt_tokens:
                break
            body += "\n" + synthetic_pod_state(rng, i * 1000 + len(ids))
        return ids[: args.prompt_tokens]

    params = SamplingParams(
        max_new_tokens=args.decode_tokens, grammar=GrammarMode.TOOLPROMPT
    )

    def barrier_sync():
        if world > 1:
            pstate.barrier()
        if have_gpu:
            torch.cuda.synchronize()

    valid = 0
    total_prefill_tokens = 0

    if world > 1 and rank != 0:
        # Followers never see the request stream: rank 0
        # broadcasts tokenized admissions inside engine.step(); marks from
        # bcast_mark() delimit the warmup and timed regions here.
        reason = eng.follower_loop()            # ... until end-of-warmup mark
        barrier_sync()
        t0 = time.perf_counter()
        if reason == "mark":
            reason = eng.follower_loop()        # ... until end-of-timed mark
        barrier_sync()
        total_ms = (time.perf_counter() - t0) * 1000.0
        t = torch.tensor([total_ms] + [0.0] * args.steps, dtype=torch.float64)
        if have_gpu:
            t = t.cuda()
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        return

    def check_valid(out) -> None:
        nonlocal valid
        try:
            json.loads(tok.decode_text(out))
            valid += 1
        except json.JSONDecodeError:
            pass

    # ---- mode drivers ----------------------------------------------------
    if args.mode == "turn":
        def warmup():
            for i in range(args.warmup):
                eng.generate(make_prompt(10_000 + i), params)

        def run_steps():
            nonlocal total_prefill_tokens
            times = []
            for i in range(args.steps):
                ts = time.perf_counter()
                out, _ = eng.generate(make_prompt(i), params)
                if have_gpu:
                    torch.cuda.synchronize()
                times.append((time.perf_counter() - ts) * 1000.0)
                check_valid(out)
                total_prefill_tokens += args.prompt_tokens
            return times

    elif args.mode == "multiturn":
        # one growing ReAct conversation: each step = append an observation,
        # decode the next ToolPrompt. Conversation resets when near max_seq.
        convo = {"ids": make_prompt(0)}

        def one_iteration(i: int):
            nonlocal total_prefill_tokens
            obs = (
                '{"observation": '
                + json.dumps(synthetic_pod_state(rng, 50_000 + i)[:600])
                + "}"
            )
            obs_ids = tok.encode(
                f"<|start_header_id|>user<|end_header_id|>\n\n{obs}<|eot_id|>"
                "<|start_header_id|>assistant<|end_header_id|>\n\n"
            )
            if len(convo["ids"]) + len(obs_ids) + args.decode_tokens + 8 >= eng.max_seq_len:
                convo["ids"] = make_prompt(60_000 + i)
            convo["ids"] = convo["ids"] + obs_ids
            total_prefill_tokens += len(obs_ids)
            out, _ = eng.generate(convo["ids"], params)
            convo["ids"] = convo["ids"] + out + tok.encode("<|eot_id|>")
            return out

        def warmup():
            for i in range(args.warmup):
                one_iteration(-1 - i)

        def run_steps():
            times = []
            for i in range(args.steps):
                ts = time.perf_counter()
                out = one_iteration(i)
                if have_gpu:
                    torch.cuda.synchronize()
                times.append((time.perf_counter() - ts) * 1000.0)
                check_valid(out)
            return times

    else:  # concurrent
        from opsagent_amd.engine.serving import EngineLoop

        loop = EngineLoop(eng)

        def submit_block(base: int, n: int):
            futs = [loop.submit(make_prompt(base + j), params) for j in range(n)]
            return [f.result(timeout=600) for f in futs]

        def warmup():
            submit_block(20_000, min(args.concurrency, max(1, args.warmup)))

        def run_steps():
            nonlocal total_prefill_tokens
            # steps = total turns, issued in waves of `concurrency`
            times = []
            done = 0
            while done < args.steps:
                n = min(args.concurrency, args.steps - done)
                ts = time.perf_counter()
                results = submit_block(done, n)
                if have_gpu:
                    torch.cuda.synchronize()
                dt = (time.perf_counter() - ts) * 1000.0
                times.extend([dt] * n)  # per-turn latency of the wave
                for out, _ in results:
                    check_valid(out)
                total_prefill_tokens += n * args.prompt_tokens
                done += n
            return times

    warmup()
    if world > 1:
        eng.bcast_mark()            # release followers to the timing barrier
    barrier_sync()
    t0 = time.perf_counter()
    step_ms = run_steps()
    if world > 1:
        eng.bcast_mark()            # close the followers' timed region
    barrier_sync()
    total_ms = (time.perf_counter() - t0) * 1000.0

    # MAX over ranks of the timed region and per-step times
    if world > 1:
        t = torch.tensor([total_ms] + step_ms, dtype=torch.float64)
        if have_gpu:
            t = t.cuda()
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        total_ms = float(t[0])
        step_ms = [float(x) for x in t[1:]]

    if rank == 0:
        ms_per_step = total_ms / args.steps
        p50 = sorted(step_ms)[len(step_ms) // 2]
        total_tokens = total_prefill_tokens + args.steps * args.decode_tokens
        result = {
            "metric": "agent_turn_p50_latency_ms",
            "value": round(p50, 2),
            "unit": "ms",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": False,
            "scaling": "strong",
            "vs_baseline": None,
            # honest dtype: bf16 headline; --quantize fp8 is an EXTRA mode
            "dtype": args.quantize or "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "mode": args.mode,
                "parallelism": f"tp{n_gpus}",
                "prompt_tokens": args.prompt_tokens,
                "decode_tokens": args.decode_tokens,
                "global_batch": args.concurrency if args.mode == "concurrent" else 1,
                "seq_len": args.prompt_tokens + args.decode_tokens,
                "grammar": "toolprompt",
                "tokenizer": "bpe-32k" if tok_path else "byte-level",
                "json_validity_pct": round(100.0 * valid / max(1, args.steps), 1),
                "tokens_per_s": round(total_tokens / (total_ms / 1000.0), 1),
                "turns_per_s": round(args.steps / (total_ms / 1000.0), 3),
                "prefix_cache_blocks": eng.cache_stats()["reused_blocks"],
                "spec": dict(eng.spec_stats, ema=round(eng.spec_ema, 3)),
            },
        }
        if args.moe_dtype:
            result["config"]["moe_dtype"] = args.moe_dtype
        print(json.dumps(result))
        if args.breakdown:
            from opsagent_amd.utils.perf import get_perf_stats

            print(get_perf_stats().format_table(), file=sys.stderr)


if __name__ == "__main__":
    main()
