"""Driver-contract tests for bench.py: single-rank JSON output and the
multi-rank torchrun path (rank 0 drives, followers mirror via the engine's
admission broadcast) — exactly how the driver launches the scaling bench."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(cmd, timeout=240):
    return subprocess.run(
        cmd, cwd=ROOT, capture_output=True, text=True, timeout=timeout,
        env=dict(os.environ, MASTER_ADDR="127.0.0.1"),
    )


def _parse_json_line(stdout: str) -> dict:
    for line in stdout.splitlines():
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{stdout[-2000:]}")


def test_bench_single_rank_contract():
    out = _run([sys.executable, "bench.py", "--gpus", "1", "--steps", "2",
                "--warmup", "1", "--model", "llama3-tiny",
                "--prompt-tokens", "64", "--decode-tokens", "8"])
    assert out.returncode == 0, out.stderr[-2000:]
    d = _parse_json_line(out.stdout)
    assert d["metric"] == "agent_turn_p50_latency_ms"
    assert d["steps"] == 2 and d["warmup"] == 1 and d["n_gpus"] == 1
    assert d["higher_is_better"] is False
    assert d["config"]["json_validity_pct"] == 100.0


@pytest.mark.timeout(300)
def test_bench_torchrun_world2_contract():
    """The driver launches N>1 as torchrun; followers must mirror rank 0
    through the admission broadcast and the run must emit ONE JSON line."""
    out = _run([sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
                "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
                "--master-port", "29741", "bench.py", "--gpus", "2",
                "--steps", "2", "--warmup", "1", "--model", "llama3-tiny-w4",
                "--prompt-tokens", "64", "--decode-tokens", "8"])
    assert out.returncode == 0, out.stderr[-3000:]
    d = _parse_json_line(out.stdout)
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "tp2"
    assert d["config"]["json_validity_pct"] == 100.0
