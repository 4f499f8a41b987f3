#!/bin/bash
# AddressSanitizer run of the C++ grammar FSM (SURVEY §5 "race detection /
# sanitizers"): builds an ASAN-instrumented libopsagent_grammar.so and runs
# the full grammar + BPE test files against it. The FSM is the host-side
# component with the most raw pointer math (token byte tables, mask buffers,
# simulated-state walks); the HIP kernels are covered by the race-repeat
# GPU tests (tests/test_kernels_gpu.py) and the zero-scratch resource gate.
#
# Usage: bash scripts/sanitize_grammar.sh   (CPU only, ~1 min)
set -euo pipefail
cd "$(dirname "$0")/.."

CLANG=/opt/rocm/lib/llvm/bin/clang++
OUT=/tmp/libopsagent_grammar_asan.so
$CLANG -O1 -g -std=c++17 -fsanitize=address -fno-omit-frame-pointer \
  -shared -fPIC -o "$OUT" opsagent_amd/ops/csrc/grammar_fsm.cpp
ASAN_RT=$($CLANG -print-file-name=libclang_rt.asan-x86_64.so)

LD_PRELOAD="$ASAN_RT" \
ASAN_OPTIONS=detect_leaks=0:abort_on_error=1 \
OPSAGENT_GRAMMAR_LIB="$OUT" \
python -m pytest tests/test_grammar.py tests/test_tokenizer_bpe.py -q -p no:cacheprovider

echo "ASAN grammar run: PASS"
