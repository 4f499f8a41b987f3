"""Train the repo's offline byte-level BPE tokenizer (assets/tokenizer-32k.json).

There is no network access for real Llama-3 tokenizer assets, so the
benchmark's "real tokenizer"  is a byte-level BPE trained
here on a synthetic corpus shaped like the workload: kubectl output, k8s
events/manifests, ToolPrompt JSON documents, and English ops prose. The
tokenizer.json is Llama-3-compatible in structure (HF `tokenizers` format,
ByteLevel alphabet, <|begin_of_text|>/<|end_of_text|>/<|eot_id|> + header
specials) and loads through opsagent_amd.engine.tokenizer.BPETokenizer.

Deterministic: fixed RNG seed, fixed corpus -> identical tokenizer.json.

Usage: python scripts/train_tokenizer.py [--vocab 32000] [--out assets/tokenizer-32k.json]
"""

from __future__ import annotations

import argparse
import json
import os
import random
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def synthetic_corpus(rng: random.Random, n_docs: int = 60000):
    """Workload-shaped text: the same distribution bench.py prompts draw from."""
    statuses = ["Running", "CrashLoopBackOff", "Pending", "ImagePullBackOff",
                "Completed", "Error", "Terminating", "ContainerCreating"]
    reasons = ["BackOff", "FailedScheduling", "Unhealthy", "OOMKilling",
               "FailedMount", "NodeNotReady", "Evicted", "Killing"]
    msgs = [
        "Back-off restarting failed container",
        "0/3 nodes are available: insufficient memory",
        "Liveness probe failed: HTTP probe failed with statuscode: 500",
        "Memory cgroup out of memory: Killed process",
        "MountVolume.SetUp failed for volume \"config\": configmap not found",
        "Readiness probe failed: connection refused",
        "Failed to pull image: manifest unknown",
    ]
    verbs = ["get", "describe", "logs", "top", "rollout status", "get events"]
    kinds = ["pods", "deployments", "services", "nodes", "namespaces",
             "configmaps", "secrets", "statefulsets", "daemonsets", "pvc"]
    base_words = ("the pod is failing because its container keeps restarting after "
                  "an out of memory kill check the resource limits and requests "
                  "then inspect recent events for scheduling pressure on the node "
                  "consider increasing memory or adding a liveness probe delay "
                  "cluster capacity network policy image registry authentication "
                  "certificate rotation kubelet api server etcd scheduler controller").split()
    # synthetic lexicon: BPE needs lexical diversity to reach a 32k merge
    # table — syllable-composed words give it realistic subword structure
    onsets = ["b", "c", "d", "f", "g", "h", "j", "k", "l", "m", "n", "p", "r",
              "s", "t", "v", "w", "z", "ch", "sh", "th", "tr", "st", "pl", "br", ""]
    nuclei = ["a", "e", "i", "o", "u", "ai", "ea", "io", "ou", "ee", "ar", "er", "or"]
    codas = ["", "n", "r", "s", "t", "l", "m", "d", "k", "st", "nd", "ng", "tion", "ment"]
    lexicon = []
    for _ in range(30000):
        w = "".join(
            rng.choice(onsets) + rng.choice(nuclei) + rng.choice(codas)
            for _ in range(rng.randrange(1, 4))
        )
        lexicon.append(w)
    words = base_words * 40 + lexicon  # common ops terms stay frequent

    for i in range(n_docs):
        kind = i % 5
        if kind == 0:  # kubectl table output
            rows = []
            for j in range(rng.randrange(4, 16)):
                rows.append(
                    f"app-{rng.randrange(100, 999)}-{rng.choice('abcdef')}{j}"
                    f"{'':<6}1/1   {rng.choice(statuses):<18} "
                    f"{rng.randrange(0, 40):<4} {rng.randrange(1, 200)}d"
                )
            yield (f"$ kubectl {rng.choice(verbs)} {rng.choice(kinds)} -n "
                   f"{rng.choice(['prod', 'default', 'kube-system', 'staging'])}\n"
                   + "\n".join(rows))
        elif kind == 1:  # events
            evs = [
                f"{rng.randrange(1, 59)}m  Warning  {rng.choice(reasons)}  "
                f"pod/web-{rng.randrange(10, 99)}  {rng.choice(msgs)}"
                for _ in range(rng.randrange(3, 8))
            ]
            yield "\n".join(evs)
        elif kind == 2:  # ToolPrompt JSON
            yield json.dumps({
                "question": " ".join(rng.choices(words, k=rng.randrange(4, 10))),
                "thought": " ".join(rng.choices(words, k=rng.randrange(6, 14))),
                "action": {
                    "name": rng.choice(["kubectl", "python", "trivy", "jq", "search"]),
                    "input": f"kubectl {rng.choice(verbs)} {rng.choice(kinds)} -n prod",
                },
                "observation": " ".join(rng.choices(words, k=rng.randrange(4, 12))),
                "final_answer": " ".join(rng.choices(words, k=rng.randrange(6, 20))),
            })
        elif kind == 3:  # manifest-ish YAML
            yield (f"apiVersion: apps/v1\nkind: Deployment\nmetadata:\n"
                   f"  name: web-{rng.randrange(10, 99)}\n  namespace: prod\n"
                   f"spec:\n  replicas: {rng.randrange(1, 9)}\n  template:\n"
                   f"    spec:\n      containers:\n      - name: app\n"
                   f"        image: registry.local/app:v{rng.randrange(1, 30)}\n"
                   f"        resources:\n          limits:\n"
                   f"            memory: {rng.choice(['256Mi', '512Mi', '1Gi', '2Gi'])}\n"
                   f"            cpu: {rng.choice(['250m', '500m', '1'])}")
        else:  # prose
            yield " ".join(rng.choices(words, k=rng.randrange(20, 60))).capitalize() + "."


SPECIALS = [
    "<|begin_of_text|>", "<|end_of_text|>", "<|start_header_id|>",
    "<|end_header_id|>", "<|eot_id|>", "<|pad|>",
]


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--vocab", type=int, default=32000)
    ap.add_argument("--out", default="assets/tokenizer-32k.json")
    args = ap.parse_args()

    from tokenizers import Tokenizer, decoders, models, pre_tokenizers, trainers

    tok = Tokenizer(models.BPE(unk_token=None))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tok.decoder = decoders.ByteLevel()
    trainer = trainers.BpeTrainer(
        vocab_size=args.vocab,
        special_tokens=SPECIALS,
        initial_alphabet=pre_tokenizers.ByteLevel.alphabet(),
        show_progress=False,
    )
    rng = random.Random(20260914)
    tok.train_from_iterator(synthetic_corpus(rng), trainer)
    os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
    tok.save(args.out)
    print(f"saved {args.out}: vocab={tok.get_vocab_size()}")


if __name__ == "__main__":
    main()
