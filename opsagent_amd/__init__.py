"""opsagent_amd — MI355X-native Kubernetes ops agent framework.

A from-scratch rebuild of the capabilities of myysophia/OpsAgent (a Go
LLM-driven k8s ops agent whose "model layer" is an HTTPS call to a remote
OpenAI-compatible endpoint) as an MI355X-first framework:

  * the agent shell (CLI, HTTP API, ReAct loop, tool plugins, config) lives in
    Python (reference: Go `cmd/kube-copilot`, `pkg/{assistants,tools,api,...}`),
  * the model layer is an in-process inference engine on PyTorch-ROCm with
    hand-written HIP/CDNA4 kernels (gfx950 MFMA, LDS-tiled) for the hot ops,
    paged KV cache sized for 288 GB HBM3E, hipGraph-captured decode, and
    tensor parallelism over RCCL/xGMI — replacing the reference's network
    boundary at `pkg/llms/openai.go:69`.

Layer map (mirrors SURVEY.md §1):
  cli.py / server/     — front ends              (ref: cmd/, pkg/api, pkg/handlers)
  agent/               — ReAct core + workflows  (ref: pkg/assistants, pkg/workflows)
  llm/                 — OpenAI-compatible client (ref: pkg/llms)
  tools/               — tool plugins            (ref: pkg/tools)
  k8s.py               — kubernetes helpers      (ref: pkg/kubernetes)
  utils/               — perf/log/json/yaml      (ref: pkg/utils)
  engine/              — the MI355X inference engine        (new; vacant in ref)
  ops/                 — HIP/CDNA4 kernels + C++ grammar FSM (new)
  parallel/            — TP layers + RCCL collectives        (new)
"""

from opsagent_amd.version import VERSION, __version__  # noqa: F401
