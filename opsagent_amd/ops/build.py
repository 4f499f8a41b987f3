"""Build the HIP kernel library for gfx950 (MI355X) — in-tree, so the .so
travels with the repo snapshot to GPU boxes.

Usage: python -m opsagent_amd.ops.build [--check]
Called by __graft_entry__.build().
"""

from __future__ import annotations

import os
import subprocess
import sys

CSRC = os.path.join(os.path.dirname(os.path.abspath(__file__)), "csrc")
OUT = os.path.join(os.path.dirname(os.path.abspath(__file__)), "libopsagent_kernels.so")

SOURCES = [
    "rmsnorm.hip",
    "gemv.hip",
    "rope.hip",
    "rope_kv.hip",
    "elementwise.hip",
    "sampling.hip",
    "attention_decode.hip",
    "attention_prefill.hip",
    "fp8_moe.hip",
    "moe_gemv.hip",
]

HIPCC = os.environ.get("HIPCC", "hipcc")
ARCH = os.environ.get("OPSAGENT_GPU_ARCH", "gfx950")

GRAMMAR_OUT = os.path.join(
    os.path.dirname(os.path.abspath(__file__)), "libopsagent_grammar.so"
)
GRAMMAR_SRC = os.path.join(CSRC, "grammar_fsm.cpp")


def _mtime(path: str) -> float:
    try:
        return os.path.getmtime(path)
    except OSError:
        return 0.0


def needs_rebuild() -> bool:
    out_t = _mtime(OUT)
    if out_t == 0.0:
        return True
    deps = [os.path.join(CSRC, s) for s in SOURCES] + [os.path.join(CSRC, "common.h")]
    return any(_mtime(d) > out_t for d in deps)


def build(verbose: bool = True, force: bool = False) -> str:
    if not force and not needs_rebuild():
        return OUT
    srcs = [os.path.join(CSRC, s) for s in SOURCES]
    cmd = [
        HIPCC,
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-shared",
        "-fPIC",
        "-o",
        OUT,
        *srcs,
    ]
    if verbose:
        print("[opsagent build]", " ".join(cmd), file=sys.stderr)
    proc = subprocess.run(cmd, capture_output=True, text=True)
    if proc.returncode != 0:
        raise RuntimeError(
            f"hipcc failed ({proc.returncode}):\n{proc.stdout}\n{proc.stderr}"
        )
    return OUT


def build_grammar(verbose: bool = True, force: bool = False) -> str:
    """Build the CPU grammar-FSM library (plain g++; used on CPU and GPU)."""
    if not force and _mtime(GRAMMAR_OUT) > _mtime(GRAMMAR_SRC):
        return GRAMMAR_OUT
    cmd = ["g++", "-O2", "-std=c++17", "-shared", "-fPIC", "-o", GRAMMAR_OUT, GRAMMAR_SRC]
    if verbose:
        print("[opsagent build]", " ".join(cmd), file=sys.stderr)
    proc = subprocess.run(cmd, capture_output=True, text=True)
    if proc.returncode != 0:
        raise RuntimeError(f"g++ failed ({proc.returncode}):\n{proc.stderr}")
    return GRAMMAR_OUT


def build_all(force: bool = False) -> None:
    build_grammar(force=force)
    build(force=force)


if __name__ == "__main__":
    build_all(force="--force" in sys.argv)
    print(OUT)
    print(GRAMMAR_OUT)
