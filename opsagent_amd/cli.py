"""CLI front-end.

Capability parity with /root/reference/cmd/kube-copilot/: subcommands
execute, analyze, audit, diagnose, generate, serve(r), version, with the
persistent flags --model --max-tokens --count-tokens --verbose
--max-iterations (ref main.go:28-32). Unlike the reference snapshot — which
registers only `server` on the root command (main.go:34, a known bug per
SURVEY.md §1) — every subcommand here is wired.

Entry point: `python -m opsagent_amd.cli <command>` or the `opsagent` script.
"""

from __future__ import annotations

import sys
from typing import Optional

import typer

from opsagent_amd import VERSION
from opsagent_amd.agent import react, workflows
from opsagent_amd.agent import prompts
from opsagent_amd.config import load_config, set_global
from opsagent_amd.llm.client import new_client
from opsagent_amd.tools import TOOLS
from opsagent_amd.utils.logging import init_logging
from opsagent_amd.utils.perf import get_perf_stats
from opsagent_amd.utils.term import render_markdown
from opsagent_amd.utils.yamlextract import extract_yaml

app = typer.Typer(
    name="opsagent",
    help="MI355X-native Kubernetes ops agent (OpsAgent capability parity).",
    add_completion=False,
)

_state = {"cfg": None}


def _setup(model: Optional[str], verbose: bool, config_path: Optional[str] = None):
    cfg = load_config(config_path)
    _state["cfg"] = cfg
    init_logging(
        level="debug" if verbose else cfg.get("log.level", "info"),
        fmt=cfg.get("log.format", "console"),
        output=cfg.get("log.output", "stderr"),
        log_dir=cfg.get("log.dir", "logs"),
    )
    get_perf_stats().enabled = bool(cfg.get("perf.enabled", True))
    client = new_client(
        api_key=cfg.get("llm.api_key", ""),
        base_url=cfg.get("llm.base_url", "local"),
        engine_config=cfg.section("engine"),
    )
    mdl = model or cfg.get("llm.model", "llama3-8b")
    return cfg, client, mdl


@app.command()
def execute(
    instructions: str = typer.Argument(..., help="what to do on the cluster"),
    model: Optional[str] = typer.Option(None, "--model"),
    max_tokens: int = typer.Option(2048, "--max-tokens"),
    max_iterations: int = typer.Option(10, "--max-iterations"),
    count_tokens: bool = typer.Option(False, "--count-tokens", help="print token usage"),
    verbose: bool = typer.Option(False, "--verbose"),
    config: Optional[str] = typer.Option(None, "--config"),
):
    """Execute operations based on prompt instructions (ref execute.go:189-317)."""
    cfg, client, mdl = _setup(model, verbose, config)
    perf = get_perf_stats()
    with perf.trace("execute_total_time"):
        messages = [
            {"role": "system", "content": prompts.execute_system_prompt(TOOLS.keys())},
            {"role": "user", "content": instructions},
        ]
        with perf.trace("execute_assistant"):
            result, _history = react.assistant(
                client, mdl, messages, max_tokens=max_tokens,
                verbose=verbose, max_iterations=max_iterations,
            )
        with perf.trace("execute_format_results"):
            formatted = workflows.assistant_flow(client, mdl, result)
    typer.echo(render_markdown(formatted or result))
    if count_tokens:
        from opsagent_amd.llm.tokens import count_tokens as _ct

        typer.echo(f"[tokens] conversation ≈ {_ct(messages)} prompt tokens", err=True)
    if verbose:
        typer.echo(perf.format_table(), err=True)


@app.command()
def analyze(
    resource: str = typer.Option("pod", "--resource"),
    name: str = typer.Option(..., "--name"),
    namespace: str = typer.Option("default", "--namespace"),
    model: Optional[str] = typer.Option(None, "--model"),
    verbose: bool = typer.Option(False, "--verbose"),
    config: Optional[str] = typer.Option(None, "--config"),
):
    """Analyze issues for a live resource (ref analyze.go:42-85)."""
    from opsagent_amd import k8s

    _cfg, client, mdl = _setup(model, verbose, config)
    manifest = k8s.get_yaml(resource, name, namespace)
    result = workflows.analysis_flow(client, mdl, manifest)
    typer.echo(render_markdown(result))


@app.command()
def audit(
    name: str = typer.Option(..., "--name"),
    namespace: str = typer.Option("default", "--namespace"),
    model: Optional[str] = typer.Option(None, "--model"),
    verbose: bool = typer.Option(False, "--verbose"),
    config: Optional[str] = typer.Option(None, "--config"),
):
    """Audit pod security issues (ref audit.go:37-70)."""
    _cfg, client, mdl = _setup(model, verbose, config)
    result = workflows.audit_flow(client, mdl, namespace, name)
    typer.echo(render_markdown(result))


@app.command()
def diagnose(
    name: str = typer.Option(..., "--name"),
    namespace: str = typer.Option("default", "--namespace"),
    model: Optional[str] = typer.Option(None, "--model"),
    max_tokens: int = typer.Option(2048, "--max-tokens"),
    max_iterations: int = typer.Option(10, "--max-iterations"),
    verbose: bool = typer.Option(False, "--verbose"),
    config: Optional[str] = typer.Option(None, "--config"),
):
    """Diagnose problems for a pod (ref diagnose.go:85-139)."""
    _cfg, client, mdl = _setup(model, verbose, config)
    messages = [
        {"role": "system", "content": prompts.diagnose_system_prompt(TOOLS.keys())},
        {
            "role": "user",
            "content": f"Diagnose the pod {name} in namespace {namespace}. "
            "Read-only: never delete, edit, scale or apply anything.",
        },
    ]
    result, _history = react.assistant(
        client, mdl, messages, max_tokens=max_tokens,
        verbose=verbose, max_iterations=max_iterations,
    )
    formatted = workflows.assistant_flow(client, mdl, result)
    typer.echo(render_markdown(formatted or result))


@app.command()
def generate(
    instructions: str = typer.Argument(..., help="what manifests to generate"),
    model: Optional[str] = typer.Option(None, "--model"),
    verbose: bool = typer.Option(False, "--verbose"),
    yes: bool = typer.Option(False, "--yes", help="apply without confirmation"),
    config: Optional[str] = typer.Option(None, "--config"),
):
    """Generate manifests and optionally apply them (ref generate.go:36-94)."""
    from opsagent_amd import k8s

    _cfg, client, mdl = _setup(model, verbose, config)
    result = workflows.generator_flow(client, mdl, instructions)
    manifests = extract_yaml(result)
    typer.echo(manifests)
    # interactive y/n gate (ref generate.go:77-92)
    if yes or (sys.stdin.isatty() and typer.confirm("Apply these manifests to the cluster?", default=False)):
        out = k8s.apply_yaml(manifests)
        typer.echo(out)


@app.command()
def serve(
    port: int = typer.Option(None, "--port"),
    host: str = typer.Option(None, "--host"),
    jwt_key: Optional[str] = typer.Option(None, "--jwt-key"),
    show_thought: bool = typer.Option(False, "--show-thought"),
    model: Optional[str] = typer.Option(None, "--model"),
    verbose: bool = typer.Option(False, "--verbose"),
    config: Optional[str] = typer.Option(None, "--config"),
):
    """Run the HTTP API server (ref server.go:68-113).

    TP>1 (launched under torchrun, one rank per GPU): rank 0 serves HTTP
    and drives the engine; every other rank builds its model shard and
    mirrors rank 0's steps via the engine's request broadcast — the server
    surface is unchanged at tp=2..8.
    """
    import uvicorn

    cfg, _client, _mdl = _setup(model, verbose, config)
    if jwt_key:
        cfg.set("jwt.key", jwt_key)
    set_global("show_thought", show_thought)

    import os as _os

    if int(_os.environ.get("WORLD_SIZE", "1")) > 1 and int(_os.environ.get("RANK", "0")) != 0:
        from opsagent_amd.engine.openai_api import ChatCompletionAPI

        api = ChatCompletionAPI.get_or_create(cfg.section("engine"))
        while api.engine.follower_loop() == "mark":
            pass  # marks are harness sync points; keep following
        return

    from opsagent_amd.server.app import create_app

    application = create_app(cfg)
    # build the engine (and release followers' model-build barrier) BEFORE
    # accepting traffic when a local engine is configured
    if str(cfg.get("llm.base_url", "local")) == "local":
        from opsagent_amd.engine.openai_api import ChatCompletionAPI

        ChatCompletionAPI.get_or_create(cfg.section("engine"))
    uvicorn.run(
        application,
        host=host or cfg.get("server.host", "0.0.0.0"),
        port=port or int(cfg.get("server.port", 8080)),
        log_level="info",
    )


@app.command()
def version():
    """Print version (ref version.go:29-41)."""
    typer.echo(f"opsagent-amd {VERSION}")


def main():
    app()


if __name__ == "__main__":
    main()
