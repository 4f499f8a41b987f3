"""Kernel resource-usage regression gate: the perf-critical HIP kernels must
compile for gfx950 with ZERO scratch (register spills). A spill in the GEMV
or attention inner loops silently costs 2-10x — catch it at compile time
(hipcc cross-compiles without a GPU; mirrors .github/workflows/build.yaml)."""

import os
import shutil
import subprocess

import pytest

CSRC = os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    "opsagent_amd", "ops", "csrc",
)

FILES = ["gemv.hip", "attention_prefill.hip", "attention_decode.hip", "fp8_moe.hip"]

# Known, measured-benign exceptions: kernel name fragment -> max scratch
# bytes/lane. gemm_fp8_kernel_v2<false> (the deep-K register-staged MX GEMM)
# carries 80 B of prologue/epilogue spill from its 64-AGPR accumulators +
# 32-VGPR staging ring; it measures 735-764 TF at its dispatch shapes
# (profiles/README.md), i.e. the spill is not on the K-loop path.
ALLOWED_SCRATCH = {"gemm_fp8_kernel_v2ILb0E": 80}


@pytest.mark.skipif(shutil.which("hipcc") is None, reason="hipcc not on PATH")
@pytest.mark.parametrize("fname", FILES)
def test_no_register_spills(fname, tmp_path):
    out = subprocess.run(
        ["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17",
         "-Rpass-analysis=kernel-resource-usage",
         "-c", os.path.join(CSRC, fname), "-o", str(tmp_path / "k.o")],
        capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    spills = []
    current_fn = ""
    for ln in out.stderr.splitlines():
        if "Function Name:" in ln:
            current_fn = ln.rsplit(None, 2)[-2]
        if "ScratchSize" in ln and "ScratchSize [bytes/lane]: 0" not in ln:
            bytes_lane = int(ln.split("ScratchSize [bytes/lane]:")[1].split()[0])
            cap = next(
                (v for k, v in ALLOWED_SCRATCH.items() if k in current_fn), 0
            )
            if bytes_lane > cap:
                spills.append(f"{current_fn}: {bytes_lane} B/lane")
    assert not spills, f"register spills in {fname}:\n" + "\n".join(spills)
