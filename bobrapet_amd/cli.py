"""bobrapet command-line interface.

Role parity with the reference's process entry + kubectl surface
(reference: cmd/main.go — flags, manager wiring, health endpoints): the
CLI loads CRD-style YAML, validates it, runs Stories in-process on the
engine (native core when eligible), and serves the HTTP ingress.

  bobrapet validate -f stories.yaml
  bobrapet run -f stories.yaml --story my-story --inputs '{"x": 1}'
  bobrapet serve -f resources.yaml --port 8080
  bobrapet engrams
"""
from __future__ import annotations

import json
import sys
import typing as _t

import typer

app = typer.Typer(add_completion=False, no_args_is_help=True, help=__doc__)


def _load_files(files: _t.List[str]):
    from .specs import load_path

    objs = []
    for f in files:
        objs.extend(load_path(f))
    return objs


@app.command()
def validate(
    file: _t.List[str] = typer.Option(..., "-f", "--file", help="YAML file(s)"),
):
    """Validate spec documents (admission-webhook parity)."""
    from .specs import types as T
    from .specs import validation as V

    objs = _load_files(file)
    failed = False
    for obj in objs:
        kind = type(obj).__name__
        if isinstance(obj, T.Story):
            res = V.validate_story(obj)
        elif isinstance(obj, T.Engram):
            res = V.validate_engram(obj)
        elif isinstance(obj, T.Impulse):
            res = V.validate_impulse(obj)
        elif isinstance(obj, T.EngramTemplate):
            res = V.validate_engram_template(obj)
        elif isinstance(obj, T.ImpulseTemplate):
            res = V.validate_impulse_template(obj)
        elif isinstance(obj, T.Transport):
            res = V.validate_transport(obj)
        else:
            typer.echo(f"  {kind}/{obj.name}: ok (no validator)")
            continue
        status = "ok" if res.ok else "INVALID"
        typer.echo(f"  {kind}/{getattr(obj, 'name', '?')}: {status}")
        for e in res.errors:
            typer.echo(f"    error: {e}")
            failed = True
        for w in res.warnings:
            typer.echo(f"    warning: {w}")
    raise typer.Exit(1 if failed else 0)


@app.command()
def run(
    file: _t.List[str] = typer.Option(..., "-f", "--file", help="YAML file(s)"),
    story: _t.Optional[str] = typer.Option(None, help="story name (default: first Story)"),
    inputs: str = typer.Option("{}", help="JSON inputs"),
    timeout: float = typer.Option(300.0),
    engine_impl: str = typer.Option("auto", help="auto | python | native"),
    output_json: bool = typer.Option(False, "--json", help="print the run record as JSON"),
):
    """Run one Story end-to-end and print its output."""
    from .engine import EngineConfig, RunEngine
    from .specs import types as T

    objs = _load_files(file)
    eng = RunEngine(EngineConfig()).start()
    try:
        target = None
        for obj in objs:
            eng.apply(obj)
            if isinstance(obj, T.Story) and (story is None or obj.name == story):
                target = target or obj
        if story is not None:
            matches = [o for o in objs if isinstance(o, T.Story) and o.name == story]
            if not matches:
                typer.echo(f"story {story!r} not found", err=True)
                raise typer.Exit(2)
            target = matches[0]
        if target is None:
            typer.echo("no Story in the given files", err=True)
            raise typer.Exit(2)

        in_val = json.loads(inputs)
        use_native = False
        if engine_impl in ("auto", "native"):
            try:
                from .runtime.native import NativeRunner, story_supported

                if story_supported(target) is None:
                    use_native = True
                elif engine_impl == "native":
                    typer.echo(f"native core cannot run this story: {story_supported(target)}", err=True)
                    raise typer.Exit(2)
            except RuntimeError as exc:
                if engine_impl == "native":
                    typer.echo(str(exc), err=True)
                    raise typer.Exit(2)

        if use_native:
            from .runtime.native import NativeRunner

            nr = NativeRunner.from_run_engine(eng)
            try:
                status = nr.run_story(target, in_val, timeout=timeout)
            finally:
                nr.stop()
            record = status
            phase = status["phase"]
        else:
            r = eng.run_story(target, in_val, timeout=timeout)
            phase = str(r.phase)
            record = {
                "phase": phase,
                "output": r.output,
                "error": r.error.to_dict() if r.error else None,
                "steps": {
                    k: {"phase": str(v.phase), "output": v.output}
                    for k, v in r.step_states.items()
                },
            }
        if output_json:
            typer.echo(json.dumps(record, indent=2, default=str))
        else:
            typer.echo(f"phase: {phase}")
            typer.echo(f"output: {json.dumps(record.get('output'), default=str)}")
        raise typer.Exit(0 if phase == "Succeeded" else 1)
    finally:
        eng.stop()


@app.command()
def serve(
    file: _t.List[str] = typer.Option([], "-f", "--file", help="YAML file(s) to apply"),
    host: str = typer.Option("127.0.0.1"),
    port: int = typer.Option(8080),
    start_impulses: bool = typer.Option(True, help="start all applied Impulses"),
    checkpoint: _t.Optional[str] = typer.Option(
        None, help="state snapshot path: restored on boot, saved periodically"
    ),
    checkpoint_interval: float = typer.Option(30.0, help="snapshot period (s)"),
    grpc_port: _t.Optional[int] = typer.Option(
        None, help="also serve the gRPC ingress (triggers + streaming packets)"
    ),
    otlp: _t.Optional[str] = typer.Option(
        None, help="OTLP span export: a file path or http://host:4318/v1/traces"
    ),
):
    """Start the engine + REST control plane / impulse ingress."""
    import os

    from .engine import EngineConfig, RunEngine
    from .engine.impulses import serve_http
    from .specs import types as T

    cfg = EngineConfig()
    if checkpoint:
        cfg.checkpoint_path = checkpoint
        cfg.checkpoint_interval_seconds = checkpoint_interval
    if otlp:
        cfg.otlp_endpoint = otlp
    eng = RunEngine(cfg).start()
    objs = _load_files(file) if file else []
    for obj in objs:
        eng.apply(obj)
    if start_impulses:
        for obj in objs:
            if isinstance(obj, T.Impulse):
                eng.impulses.start(obj.key)
                typer.echo(f"impulse {obj.key} started")
    if checkpoint and os.path.exists(checkpoint):
        n = eng.load_state(checkpoint)
        typer.echo(f"restored {n} runs from {checkpoint}")
    if grpc_port is not None:
        from .engine.ingress_grpc import serve_grpc

        _server, bound = serve_grpc(eng, port=grpc_port)
        typer.echo(f"gRPC ingress on 127.0.0.1:{bound}")
    typer.echo(f"serving on http://{host}:{port}")
    serve_http(eng, host=host, port=port)


@app.command()
def engrams():
    """List the builtin engram implementations."""
    from .engrams import registry

    for name in registry.known():
        typer.echo(name)


@app.command()
def version():
    import bobrapet_amd

    from . import ops

    typer.echo(f"bobrapet_amd {bobrapet_amd.__version__}")
    typer.echo(f"  hip kernels: {'built' if ops.hip_available() else 'NOT built'}")
    try:
        from .runtime.native import core_available

        typer.echo(f"  native core: {'built' if core_available() else 'NOT built'}")
    except Exception:
        typer.echo("  native core: NOT built")


def main():
    app()


if __name__ == "__main__":
    main()
