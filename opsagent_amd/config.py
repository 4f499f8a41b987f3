"""Configuration system.

Keeps the reference's `config.yaml` layout (ref /root/reference/configs/
config.yaml:1-20 and pkg/utils/config.go:10-53: jwt.key/expire, server.port/
host, log.level/format/output, perf.enabled/reset_interval) and adds the
`engine:` section for the in-process MI355X inference engine (model, dtype,
tensor-parallel degree, KV budget, grammar mode) per SURVEY.md §5.

Search order mirrors the reference: ./configs/config.yaml then ./config.yaml,
then coded defaults. Env overrides: OPSAGENT_<SECTION>_<KEY>.
"""

from __future__ import annotations

import copy
import os
import threading
from typing import Any, Dict, Optional

import yaml

DEFAULTS: Dict[str, Any] = {
    "jwt": {"key": "novastar-secret-key", "expire": 24},  # hours
    "server": {"port": 8080, "host": "0.0.0.0"},
    "log": {"level": "info", "format": "console", "output": "stderr", "dir": "logs"},
    "perf": {"enabled": True, "reset_interval": 0},
    "llm": {
        # "local" = in-process MI355X engine; any http(s) URL = remote
        # OpenAI-compatible endpoint (the reference's only mode).
        "base_url": "local",
        "api_key": "",
        "model": "llama3-8b",
        "max_tokens": 2048,
        "max_iterations": 10,
        "temperature": 1e-45,  # ref openai.go:74 uses math.SmallestNonzeroFloat32 ≈ greedy
    },
    "engine": {
        "model": "llama3-8b",          # name in opsagent_amd.engine.config.MODEL_REGISTRY
        "dtype": "bf16",
        "tp": 1,                        # tensor-parallel degree (1..8, one rank per GPU)
        "kv_cache_gb": 0,               # 0 = auto-size from free HBM (288 GB/GPU on MI355X)
        "kv_block_size": 32,            # tokens per paged-KV block
        "max_seq_len": 8192,
        "grammar": "auto",              # auto|off|json|toolprompt — constrained sampling mode
        "max_batch_size": 64,           # continuous batching limit
        "weights": "random",            # "random" or a safetensors path
        "use_hipgraph": True,
        "seed": 1234,
        # /v1 endpoint access control: require a JWT bearer or X-API-Key ==
        # api_key; open_api=true disables the gate (localhost sidecars only).
        "open_api": False,
        "api_key": "",
    },
}


def _deep_update(dst: Dict[str, Any], src: Dict[str, Any]) -> Dict[str, Any]:
    for k, v in src.items():
        if isinstance(v, dict) and isinstance(dst.get(k), dict):
            _deep_update(dst[k], v)
        else:
            dst[k] = v
    return dst


class Config:
    def __init__(self, data: Dict[str, Any]):
        self._data = data

    def get(self, dotted: str, default: Any = None) -> Any:
        node: Any = self._data
        for part in dotted.split("."):
            if not isinstance(node, dict) or part not in node:
                return default
            node = node[part]
        return node

    def section(self, name: str) -> Dict[str, Any]:
        v = self._data.get(name, {})
        return v if isinstance(v, dict) else {}

    def set(self, dotted: str, value: Any) -> None:
        parts = dotted.split(".")
        node = self._data
        for p in parts[:-1]:
            node = node.setdefault(p, {})
        node[parts[-1]] = value

    def as_dict(self) -> Dict[str, Any]:
        return copy.deepcopy(self._data)


def load_config(path: Optional[str] = None) -> Config:
    data = copy.deepcopy(DEFAULTS)
    candidates = [path] if path else ["configs/config.yaml", "config.yaml"]
    for cand in candidates:
        if cand and os.path.isfile(cand):
            with open(cand, "r") as f:
                loaded = yaml.safe_load(f) or {}
            if not isinstance(loaded, dict):
                raise ValueError(f"config file {cand} must contain a mapping")
            _deep_update(data, loaded)
            break
    # env overrides: OPSAGENT_SERVER_PORT=9090 etc.
    for key, val in os.environ.items():
        if not key.startswith("OPSAGENT_"):
            continue
        parts = key[len("OPSAGENT_"):].lower().split("_", 1)
        if len(parts) != 2:
            continue
        section, name = parts
        if section in data and isinstance(data[section], dict):
            old = data[section].get(name)
            if isinstance(old, bool):
                data[section][name] = val.lower() in ("1", "true", "yes", "on")
            elif isinstance(old, int):
                try:
                    data[section][name] = int(val)
                except ValueError:
                    data[section][name] = val
            elif isinstance(old, float):
                try:
                    data[section][name] = float(val)
                except ValueError:
                    data[section][name] = val
            else:
                data[section][name] = val
    return Config(data)


# -- global var store (ref pkg/utils/global.go:15-41) -------------------------
_globals: Dict[str, Any] = {}
_globals_lock = threading.RLock()


def set_global(key: str, value: Any) -> None:
    with _globals_lock:
        _globals[key] = value


def get_global(key: str, default: Any = None) -> Any:
    with _globals_lock:
        return _globals.get(key, default)
