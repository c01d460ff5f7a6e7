"""unionml-amd CLI — project scaffolding and lifecycle commands.

Command-set parity with the reference CLI (unionml/cli.py:1-331):
``init / deploy / train / predict / serve / activate-schedules /
deactivate-schedules / list-model-versions / list-prediction-ids /
list-scheduled-training-runs / list-scheduled-prediction-runs /
fetch-model / fetch-predictions``.

Reference quirks deliberately fixed here:

- ``deactivate-schedules`` actually deactivates (reference cli.py:124
  calls the activate method);
- ``list-scheduled-prediction-runs`` lists prediction runs (reference
  cli.py:230 lists training runs);
- ``fetch-predictions`` opens its output file for writing (reference
  cli.py:273 opens it read-only).

``init`` renders the in-package templates under
``unionml_amd/templates/`` with a dependency-free renderer instead of
cookiecutter (not available offline); generated projects are
git-initialised like the reference's post-gen hook
(templates/common/hooks/post_gen_project.py:1-9).
"""

import importlib
import json
import os
import shutil
import subprocess
import sys
from pathlib import Path
from typing import List, Optional

import typer

app = typer.Typer(
    name="unionml-amd",
    help="MI355X-native ML-microservice framework CLI.",
    no_args_is_help=True,
)


def _version_callback(value: bool):
    if value:
        try:
            from importlib.metadata import version as _v

            typer.echo(f"unionml-amd {_v('unionml_amd')}")
        except Exception:
            typer.echo("unionml-amd 0.1.0")
        raise typer.Exit()


@app.callback()
def _main(
    version: bool = typer.Option(
        False, "--version", callback=_version_callback, is_eager=True,
        help="print the framework version and exit",
    ),
):
    pass


@app.callback(invoke_without_command=True)
def _main_callback(
    version: bool = typer.Option(False, "--version", help="print version and exit"),
):
    if version:
        import unionml_amd

        typer.echo(f"unionml-amd {unionml_amd.__version__}")
        raise typer.Exit()

TEMPLATES_DIR = Path(__file__).parent / "templates"


def get_model(model_spec: str):
    """Import ``module:variable`` and return the Model instance
    (reference: remote.py:30-35)."""
    if ":" not in model_spec:
        raise typer.BadParameter(
            f"model spec must look like 'module:variable', got {model_spec!r}"
        )
    module_name, var = model_spec.split(":", 1)
    sys.path.insert(0, os.getcwd())
    module = importlib.import_module(module_name)
    try:
        return getattr(module, var)
    except AttributeError:
        raise typer.BadParameter(f"module {module_name!r} has no attribute {var!r}")


def _parse_inputs(inputs: Optional[str]) -> dict:
    if not inputs:
        return {}
    try:
        parsed = json.loads(inputs)
    except json.JSONDecodeError as exc:
        raise typer.BadParameter(f"--inputs is not valid JSON: {exc}")
    if not isinstance(parsed, dict):
        raise typer.BadParameter("--inputs must be a JSON object")
    return parsed


# ----------------------------------------------------------------------
# init
# ----------------------------------------------------------------------


def _available_templates() -> List[str]:
    if not TEMPLATES_DIR.is_dir():
        return []
    return sorted(p.name for p in TEMPLATES_DIR.iterdir() if p.is_dir())


def _render(text: str, context: dict) -> str:
    for key, value in context.items():
        text = text.replace("{{" + key + "}}", str(value))
    return text


@app.command()
def init(
    app_name: str = typer.Argument(..., help="name of the app directory to create"),
    template: str = typer.Option("basic", "--template", "-t", help="project template"),
    output_dir: Path = typer.Option(Path("."), "--dir", "-d", help="parent directory"),
):
    """Scaffold a new unionml_amd project from a template
    (reference: cli.py:38-56)."""
    templates = _available_templates()
    if template not in templates:
        typer.echo(f"unknown template {template!r}; available: {', '.join(templates)}")
        raise typer.Exit(code=1)
    src = TEMPLATES_DIR / template
    dest = output_dir / app_name
    if dest.exists():
        typer.echo(f"destination {dest} already exists")
        raise typer.Exit(code=1)
    context = {"app_name": app_name}
    for path in sorted(src.rglob("*")):
        rel = _render(str(path.relative_to(src)), context)
        target = dest / rel
        if path.is_dir():
            target.mkdir(parents=True, exist_ok=True)
        else:
            target.parent.mkdir(parents=True, exist_ok=True)
            if path.suffix in {".py", ".md", ".txt", ".cfg", ".toml", ".yaml", ".yml", ""}:
                target.write_text(_render(path.read_text(), context))
            else:
                shutil.copy2(path, target)
    # git-init the generated project (reference post-gen hook behavior)
    if shutil.which("git"):
        subprocess.run(
            ["git", "init", "-q"], cwd=dest, check=False, capture_output=True
        )
        subprocess.run(["git", "add", "."], cwd=dest, check=False, capture_output=True)
        subprocess.run(
            ["git", "commit", "-q", "-m", f"initialize {app_name} from template {template}"],
            cwd=dest,
            check=False,
            capture_output=True,
        )
    typer.echo(f"created project {dest} from template {template!r}")


# ----------------------------------------------------------------------
# deploy / train / predict
# ----------------------------------------------------------------------


@app.command()
def deploy(
    model_spec: str = typer.Argument(..., help="'module:model' of the app"),
    app_version: Optional[str] = typer.Option(None, "--app-version", "-v"),
    allow_uncommitted: bool = typer.Option(False, "--allow-uncommitted"),
    patch: bool = typer.Option(False, "--patch", help="redeploy code only, same image"),
):
    """Register the app's workflows + schedules with the backend
    (reference: cli.py:59-92)."""
    model = get_model(model_spec)
    version = model.remote_deploy(
        app_version=app_version, allow_uncommitted=allow_uncommitted, patch=patch
    )
    typer.echo(f"deployed {model.name} app_version={version}")


@app.command()
def train(
    model_spec: str = typer.Argument(...),
    inputs: Optional[str] = typer.Option(None, "--inputs", "-i", help="JSON inputs"),
    app_version: Optional[str] = typer.Option(None, "--app-version", "-v"),
    local: bool = typer.Option(
        False, "--local", help="train in-process instead of on the backend"
    ),
    output: Optional[Path] = typer.Option(
        None, "--output", "-o", help="save the trained artifact to this path (--local)"
    ),
):
    """Run the training workflow (reference: cli.py:127-145)."""
    model = get_model(model_spec)
    kwargs = _parse_inputs(inputs)
    if local:
        model_obj, metrics = model.train(**kwargs)
        typer.echo(f"trained {model.name}: metrics={metrics}")
        if output is not None:
            model.save(output)
            typer.echo(f"saved artifact to {output}")
        return
    execution = model.remote_train(app_version=app_version, wait=True, **kwargs)
    typer.echo(f"execution {execution.id} SUCCEEDED")
    typer.echo(f"metrics: {model.artifact.metrics}")


@app.command()
def predict(
    model_spec: str = typer.Argument(...),
    inputs: Optional[str] = typer.Option(None, "--inputs", "-i", help="JSON reader kwargs"),
    features: Optional[Path] = typer.Option(
        None, "--features", "-f", help="path to a JSON features file"
    ),
    app_version: Optional[str] = typer.Option(None, "--app-version", "-v"),
    model_version: str = typer.Option("latest", "--model-version", "-m"),
    local: bool = typer.Option(False, "--local", help="predict in-process"),
):
    """Run the prediction workflow (reference: cli.py:148-169)."""
    model = get_model(model_spec)
    feats = None
    if features is not None:
        feats = model._dataset.get_features(features)
    if local:
        if feats is not None:
            predictions = model.predict(features=feats)
        else:
            predictions = model.predict(**_parse_inputs(inputs))
    else:
        predictions = model.remote_predict(
            app_version=app_version,
            model_version=model_version,
            wait=True,
            features=feats,
            **_parse_inputs(inputs),
        )
    typer.echo(json.dumps(_jsonable(predictions)))


def _jsonable(obj):
    import numpy as np

    if hasattr(obj, "detach"):
        return obj.detach().cpu().tolist()
    if isinstance(obj, np.ndarray):
        return obj.tolist()
    if isinstance(obj, np.generic):
        return obj.item()
    if isinstance(obj, (list, tuple)):
        return [_jsonable(o) for o in obj]
    if isinstance(obj, dict):
        return {k: _jsonable(v) for k, v in obj.items()}
    return obj


# ----------------------------------------------------------------------
# schedules
# ----------------------------------------------------------------------


@app.command("activate-schedules")
def activate_schedules(
    model_spec: str = typer.Argument(...),
    names: Optional[List[str]] = typer.Argument(None, help="schedule names (all if omitted)"),
):
    """Activate schedules on the backend (reference: cli.py:95-109)."""
    model = get_model(model_spec)
    activated = model.remote_activate_schedules(list(names) if names else None)
    typer.echo(f"activated schedules: {activated}")


@app.command("deactivate-schedules")
def deactivate_schedules(
    model_spec: str = typer.Argument(...),
    names: Optional[List[str]] = typer.Argument(None),
):
    """Deactivate schedules on the backend (reference: cli.py:112-124;
    the reference's version activates by mistake — fixed here)."""
    model = get_model(model_spec)
    deactivated = model.remote_deactivate_schedules(list(names) if names else None)
    typer.echo(f"deactivated schedules: {deactivated}")


# ----------------------------------------------------------------------
# listings
# ----------------------------------------------------------------------


@app.command("list-model-versions")
def list_model_versions(
    model_spec: str = typer.Argument(...),
    app_version: Optional[str] = typer.Option(None, "--app-version", "-v"),
    limit: int = typer.Option(10, "--limit", "-n"),
):
    """List trained model versions (reference: cli.py:172-185)."""
    model = get_model(model_spec)
    for version in model.remote_list_model_versions(app_version=app_version, limit=limit):
        typer.echo(version)


@app.command("list-prediction-ids")
def list_prediction_ids(
    model_spec: str = typer.Argument(...),
    app_version: Optional[str] = typer.Option(None, "--app-version", "-v"),
    limit: int = typer.Option(10, "--limit", "-n"),
):
    """List prediction execution ids (reference: cli.py:188-201)."""
    model = get_model(model_spec)
    for pid in model.remote_list_prediction_ids(app_version=app_version, limit=limit):
        typer.echo(pid)


@app.command("list-scheduled-training-runs")
def list_scheduled_training_runs(
    model_spec: str = typer.Argument(...),
    schedule_name: str = typer.Argument(...),
    limit: int = typer.Option(10, "--limit", "-n"),
):
    """List runs of a training schedule (reference: cli.py:204-217)."""
    model = get_model(model_spec)
    for run in model.remote_list_scheduled_training_runs(schedule_name, limit=limit):
        typer.echo(run)


@app.command("list-scheduled-prediction-runs")
def list_scheduled_prediction_runs(
    model_spec: str = typer.Argument(...),
    schedule_name: str = typer.Argument(...),
    limit: int = typer.Option(10, "--limit", "-n"),
):
    """List runs of a prediction schedule (reference: cli.py:220-231;
    the reference's version lists training runs — fixed here)."""
    model = get_model(model_spec)
    for run in model.remote_list_scheduled_prediction_runs(schedule_name, limit=limit):
        typer.echo(run)


@app.command("run-scheduler")
def run_scheduler(
    model_spec: str = typer.Argument(...),
    app_version: Optional[str] = typer.Option(None, "--app-version", "-v"),
    iterations: Optional[int] = typer.Option(
        None, "--iterations", "-n", help="bound the loop (default: run forever)"
    ),
    poll_s: float = typer.Option(1.0, "--poll-s"),
):
    """Run the backend scheduler loop, firing active cron / fixed-rate
    launch plans (the reference delegates this to Flyte; here the
    framework owns it — see docs/scheduling.md)."""
    model = get_model(model_spec)
    typer.echo("scheduler running (ctrl-c to stop)")
    model._backend().run_scheduler(
        model, app_version=app_version, iterations=iterations, poll_s=poll_s
    )


# ----------------------------------------------------------------------
# fetch
# ----------------------------------------------------------------------


@app.command("fetch-model")
def fetch_model(
    model_spec: str = typer.Argument(...),
    output_file: Path = typer.Argument(..., help="where to save the artifact"),
    model_version: str = typer.Option("latest", "--model-version", "-m"),
    app_version: Optional[str] = typer.Option(None, "--app-version", "-v"),
):
    """Fetch a trained model artifact from the backend and save it
    locally (reference: cli.py:234-251)."""
    model = get_model(model_spec)
    model.artifact = model._backend().fetch_model_artifact(
        model, app_version=app_version, model_version=model_version
    )
    model.save(output_file)
    typer.echo(f"saved model artifact to {output_file}")


@app.command("fetch-predictions")
def fetch_predictions(
    model_spec: str = typer.Argument(...),
    execution_id: str = typer.Argument(...),
    output_file: Path = typer.Option(Path("predictions.json"), "--output", "-o"),
):
    """Fetch a prediction execution's outputs and write them as JSON
    (reference: cli.py:254-277; its ``open(output_file)`` misses the
    write mode — fixed here)."""
    model = get_model(model_spec)
    backend = model._backend()
    execution = backend.get_execution(execution_id)
    predictions = backend.fetch_output(execution)
    with open(output_file, "w") as fh:
        json.dump(_jsonable(predictions), fh)
    typer.echo(f"wrote predictions to {output_file}")


# ----------------------------------------------------------------------
# serve
# ----------------------------------------------------------------------


@app.command()
def serve(
    app_spec: str = typer.Argument(..., help="'module:app' FastAPI app spec"),
    model_path: Optional[Path] = typer.Option(
        None, "--model-path", help="artifact to load at startup"
    ),
    host: str = typer.Option("127.0.0.1", "--host"),
    port: int = typer.Option(8000, "--port", "-p"),
    workers: int = typer.Option(1, "--workers"),
    reload: bool = typer.Option(False, "--reload"),
    reuse_port_worker: bool = typer.Option(
        False, "--reuse-port-worker", hidden=True,
        help="internal: run as one SO_REUSEPORT worker of a multi-worker serve",
    ),
):
    """Serve a unionml_amd FastAPI app with uvicorn
    (reference: cli.py:285-320).

    ``--workers N`` uses the framework's own supervisor: N independent
    single-worker processes sharing the port via SO_REUSEPORT (kernel
    load balancing), each with TCP_NODELAY set on the listening socket.
    uvicorn's bundled multiprocess mode is NOT used — its explicit-socket
    path leaves Nagle enabled, which measures as a uniform ~40 ms
    delayed-ACK penalty per request on this stack (see
    profiles/r01_resnet_and_serving.md).
    """
    if model_path is not None:
        if os.environ.get("UNIONML_MODEL_PATH"):
            typer.echo(
                "UNIONML_MODEL_PATH is already set; unset it or drop --model-path"
            )
            raise typer.Exit(code=1)
        os.environ["UNIONML_MODEL_PATH"] = str(model_path)
    sys.path.insert(0, os.getcwd())
    import uvicorn

    if reuse_port_worker:
        import socket as socket_mod

        sock = socket_mod.socket(socket_mod.AF_INET, socket_mod.SOCK_STREAM)
        sock.setsockopt(socket_mod.SOL_SOCKET, socket_mod.SO_REUSEADDR, 1)
        sock.setsockopt(socket_mod.SOL_SOCKET, socket_mod.SO_REUSEPORT, 1)
        # accepted sockets inherit this on Linux; uvicorn's passed-socket
        # path never sets it itself
        sock.setsockopt(socket_mod.IPPROTO_TCP, socket_mod.TCP_NODELAY, 1)
        sock.bind((host, port))
        sock.listen(2048)
        config = uvicorn.Config(app_spec, log_level="warning", access_log=False)
        uvicorn.Server(config).run(sockets=[sock])
        return

    if workers > 1:
        import signal
        import subprocess

        children = []
        cmd = [
            sys.executable, "-m", "unionml_amd.cli", "serve", app_spec,
            "--host", host, "--port", str(port), "--reuse-port-worker",
        ]
        env = dict(os.environ)
        for _ in range(workers):
            children.append(subprocess.Popen(cmd, env=env))

        def shutdown(signum, frame):
            for c in children:
                c.terminate()

        signal.signal(signal.SIGTERM, shutdown)
        signal.signal(signal.SIGINT, shutdown)
        try:
            for c in children:
                c.wait()
        finally:
            for c in children:
                if c.poll() is None:
                    c.terminate()
        return

    uvicorn.run(app_spec, host=host, port=port, reload=reload)


def main():
    app()


if __name__ == "__main__":
    main()
