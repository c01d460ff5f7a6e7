"""Remote execution backend — app packaging, versioning, execution,
scheduling, and artifact registry.

The reference delegates remote execution to a Flyte cluster
(unionml/remote.py). This build carries its own backend with the same
user-visible semantics, designed for a single MI355X node (SURVEY.md
§2b: multi-node fan-out stays out of scope):

- **App registry**: ``deploy`` records, per git-sha app version, the
  resolver args of the three workflows plus the compiled launch plans.
- **Executions**: ``execute`` launches a *worker subprocess* that
  rehydrates the app via the task resolver and runs the named workflow;
  inputs/outputs pass through a blob directory (the Flyte blob-store
  analog). Tasks whose resources request GPUs get ``HIP_VISIBLE_DEVICES``
  assigned from a per-node device allocator so scheduled batch jobs fan
  out across the 8 GPUs of the node.
- **Model registry**: a training execution's output artifact is
  addressable by execution id; ``latest`` resolves to the newest
  SUCCEEDED training run (reference: remote.py:200-233).
- **Scheduler**: an in-process loop fires active launch plans by cron /
  fixed-rate (reference registers Flyte LaunchPlans instead).
- **Docker packaging** (reference: remote.py:62-122) is provided for
  completeness and gated on a reachable Docker daemon.
"""

import datetime
import json
import os
import pickle
import subprocess
import sys
import time
import uuid
from dataclasses import dataclass
from pathlib import Path
from typing import Any, Dict, List, Optional

from unionml_amd._logging import logger
from unionml_amd.artifact import ModelArtifact
from unionml_amd.exceptions import ModelArtifactNotFound, VersionFetchError

STATUS_QUEUED = "QUEUED"
STATUS_RUNNING = "RUNNING"
STATUS_SUCCEEDED = "SUCCEEDED"
STATUS_FAILED = "FAILED"


def get_app_version(allow_uncommitted: bool = False, repo_path: str = ".") -> str:
    """App version = git sha of the user repo; dirty tree raises
    (reference: remote.py:45-59)."""
    try:
        sha = (
            subprocess.run(
                ["git", "rev-parse", "HEAD"],
                cwd=repo_path,
                capture_output=True,
                text=True,
                check=True,
            ).stdout.strip()
        )
    except (subprocess.CalledProcessError, FileNotFoundError) as exc:
        raise VersionFetchError(f"cannot determine git sha in {repo_path!r}: {exc}") from exc
    dirty = subprocess.run(
        ["git", "status", "--porcelain"], cwd=repo_path, capture_output=True, text=True
    ).stdout.strip()
    if dirty and not allow_uncommitted:
        raise VersionFetchError(
            "uncommitted changes in the app repo; commit them or pass allow_uncommitted=True"
        )
    return sha[:12] if not dirty else f"{sha[:12]}-dirty"


def get_image_fqn(registry: str, image_name: str, model_name: str, version: str) -> str:
    """(reference: remote.py:62-68)"""
    return f"{registry}/{image_name}:{model_name.replace('_', '-')}-{version}"


def docker_build_push(image_fqn: str, dockerfile: str = "Dockerfile", context: str = ".") -> str:
    """Build & push the app image (reference: remote.py:104-122); needs a
    Docker daemon."""
    for cmd in (
        ["docker", "build", "-t", image_fqn, "-f", dockerfile, context],
        ["docker", "push", image_fqn],
    ):
        try:
            proc = subprocess.run(cmd, capture_output=True, text=True)
        except FileNotFoundError as exc:
            raise RuntimeError(f"docker is not available: {exc}") from exc
        if proc.returncode != 0:
            raise RuntimeError(f"{' '.join(cmd)} failed:\n{proc.stderr[-2000:]}")
    return image_fqn


def _pid_alive(pid: int) -> bool:
    try:
        os.kill(pid, 0)
    except ProcessLookupError:
        return False
    except PermissionError:
        return True
    # an unreaped child (zombie) is dead for our purposes
    try:
        with open(f"/proc/{pid}/stat") as fh:
            return fh.read().split(") ", 1)[1].split()[0] != "Z"
    except (FileNotFoundError, IndexError, OSError):
        return True


@dataclass
class Execution:
    """Handle to a backend execution."""

    id: str
    workflow: str
    path: str

    @property
    def status(self) -> str:
        status_file = Path(self.path) / "status"
        return status_file.read_text().strip() if status_file.exists() else STATUS_QUEUED


class GpuAllocator:
    """Round-robin HIP_VISIBLE_DEVICES assignment for worker processes."""

    def __init__(self, root: Path):
        self._counter_file = root / "gpu_counter"
        self.n_devices = self._detect()

    @staticmethod
    def _detect() -> int:
        if os.environ.get("HIP_VISIBLE_DEVICES"):
            return len(os.environ["HIP_VISIBLE_DEVICES"].split(","))
        try:
            import torch

            return torch.cuda.device_count()
        except Exception:
            return 0

    def assign(self, n_gpus: int) -> Optional[str]:
        if n_gpus <= 0 or self.n_devices == 0:
            return None
        # read-modify-write under an exclusive flock: concurrent execute()
        # calls (scheduled fan-out across 8 GPUs) must not hand two
        # workers the same device
        import fcntl

        self._counter_file.parent.mkdir(parents=True, exist_ok=True)
        with open(self._counter_file, "a+") as f:
            fcntl.flock(f, fcntl.LOCK_EX)
            f.seek(0)
            try:
                counter = int(f.read().strip() or 0)
            except ValueError:
                counter = 0
            devices = [
                str((counter + i) % self.n_devices) for i in range(min(n_gpus, self.n_devices))
            ]
            f.seek(0)
            f.truncate()
            f.write(str((counter + n_gpus) % self.n_devices))
            f.flush()
        return ",".join(devices)


class Backend:
    def __init__(
        self,
        project: str,
        domain: str = "development",
        backend_path: Optional[str] = None,
        registry: Optional[str] = None,
        image_name: Optional[str] = None,
        dockerfile: str = "Dockerfile",
        **_: Any,
    ):
        self.project = project
        self.domain = domain
        self.registry = registry
        self.image_name = image_name or project
        self.dockerfile = dockerfile
        self.root = Path(
            backend_path or Path.home() / ".unionml_amd" / project / domain
        )
        (self.root / "apps").mkdir(parents=True, exist_ok=True)
        (self.root / "executions").mkdir(parents=True, exist_ok=True)
        self.gpus = GpuAllocator(self.root)

    # ------------------------------------------------------------------
    # deploy
    # ------------------------------------------------------------------

    def deploy(
        self,
        model,
        app_version: Optional[str] = None,
        allow_uncommitted: bool = False,
        patch: bool = False,
    ) -> str:
        """Register the app's workflows + launch plans under a version
        (reference: model.py:983-1083). ``patch=True`` re-registers the
        manifest without a new image build."""
        from unionml_amd.task_resolver import loader_args

        if app_version is None:
            try:
                app_version = get_app_version(allow_uncommitted=allow_uncommitted)
            except VersionFetchError:
                if not allow_uncommitted:
                    raise
                app_version = uuid.uuid4().hex[:12]

        workflows = {}
        for wf_name, task in (
            ("train", model.train_task()),
            ("predict", model.predict_task()),
            ("predict_from_features", model.predict_from_features_task()),
        ):
            workflows[wf_name] = loader_args(task)

        launchplans = []
        for lp in model.launchplans():
            if lp.schedule.activate_on_deploy:
                lp.activate()
            launchplans.append(
                {
                    "name": lp.name,
                    "workflow": lp.workflow_name.split(".")[-1],
                    "active": lp.active,
                    "type": lp.schedule.type.value,
                    "expression": lp.schedule.expression,
                    "fixed_rate_s": lp.schedule.fixed_rate.total_seconds()
                    if lp.schedule.fixed_rate
                    else None,
                    "time_arg": lp.schedule.time_arg,
                    "inputs": {},  # non-picklable inputs stored in blob below
                }
            )

        app_dir = self.root / "apps" / app_version
        app_dir.mkdir(parents=True, exist_ok=True)
        manifest = {
            "model_name": model.name,
            "app_module": model.app_module,
            "object_name": model.find_lhs(),
            "module_file": model._module_file,
            "workflows": workflows,
            "launchplans": launchplans,
            "deployed_at": datetime.datetime.now().isoformat(),
            "gpu_resources": getattr(
                model._trainer_task_kwargs.get("resources"), "gpu", 0
            ),
        }
        # fast-registration code snapshot (reference remote.py:138-152
        # zips the source and uploads it so containers run the VERSIONED
        # code): workers import the app from this copy, so editing the
        # live file after deploy does not change a deployed version, and
        # patch=True re-registers code without any image build.
        code_file = self._snapshot_code(manifest["app_module"], manifest["module_file"], app_dir)
        if code_file:
            manifest["code_file"] = code_file
        (app_dir / "manifest.json").write_text(json.dumps(manifest, indent=2))

        import cloudpickle

        lp_inputs = {lp.name: lp.fixed_inputs for lp in model.launchplans()}
        (app_dir / "launchplan_inputs.pkl").write_bytes(cloudpickle.dumps(lp_inputs))

        if self.registry and not patch:
            image_fqn = get_image_fqn(self.registry, self.image_name, model.name, app_version)
            os.environ["UNIONML_INTERNAL_IMAGE"] = image_fqn
            # an explicitly configured registry means the user asked for
            # an image: a failed build/push is a deploy failure, not a
            # warning to swallow
            docker_build_push(image_fqn, self.dockerfile)

        logger.info(
            "deployed app version %s to %s%s", app_version, app_dir,
            " (patch: code-only)" if patch else "",
        )
        return app_version

    def _snapshot_code(self, app_module: str, module_file: Optional[str], app_dir: Path):
        """Copy the app's source into the version dir so workers run the
        registered code. Framework-internal apps (``unionml_amd.*``) are
        provided by the installed framework (the reference's analog: the
        container image provides the library; fast-registration ships
        only user code)."""
        import shutil

        if not module_file or app_module.split(".")[0] == "unionml_amd":
            return None
        mf = Path(module_file)
        if not mf.exists():
            return None
        code_dir = app_dir / "code"
        if code_dir.exists():
            shutil.rmtree(code_dir)
        code_dir.mkdir(parents=True)
        parts = app_module.split(".")
        if len(parts) == 1:
            shutil.copy2(mf, code_dir / mf.name)
            return str(Path("code") / mf.name)
        # module inside a package: copy the whole root package
        root_pkg = mf.parents[len(parts) - 2]
        shutil.copytree(
            root_pkg, code_dir / root_pkg.name,
            ignore=shutil.ignore_patterns("__pycache__", ".git", "*.so"),
        )
        return str(Path("code") / root_pkg.name / Path(*parts[1:]).with_suffix(".py"))

    def latest_app_version(self) -> Optional[str]:
        apps = sorted(
            (self.root / "apps").iterdir(),
            key=lambda p: (p / "manifest.json").stat().st_mtime if (p / "manifest.json").exists() else 0,
        )
        return apps[-1].name if apps else None

    def _manifest(self, app_version: Optional[str]) -> Dict:
        manifest, _ = self._manifest_with_dir(app_version)
        return manifest

    def _manifest_with_dir(self, app_version: Optional[str]):
        app_version = app_version or self.latest_app_version()
        if app_version is None:
            raise ModelArtifactNotFound(f"no app deployed in project '{self.project}'")
        app_dir = self.root / "apps" / app_version
        path = app_dir / "manifest.json"
        if not path.exists():
            raise ModelArtifactNotFound(f"app version '{app_version}' is not deployed")
        return json.loads(path.read_text()), app_dir

    # ------------------------------------------------------------------
    # execute
    # ------------------------------------------------------------------

    def execute(
        self,
        model,
        workflow: str,
        app_version: Optional[str] = None,
        inputs: Optional[Dict[str, Any]] = None,
        schedule_name: Optional[str] = None,
        wait_process: bool = False,
        n_gpus: Optional[int] = None,
    ) -> Execution:
        """Run ``{model.name}.{workflow}`` in a worker subprocess."""
        import cloudpickle

        try:
            manifest, app_dir = self._manifest_with_dir(app_version)
        except ModelArtifactNotFound:
            # auto-deploy for local-first ergonomics
            self.deploy(model, allow_uncommitted=True)
            manifest, app_dir = self._manifest_with_dir(None)

        # prefer the fast-registration code snapshot: the worker runs the
        # REGISTERED code, not whatever the live tree looks like now
        module_file = manifest.get("module_file")
        code_file = manifest.get("code_file")
        if code_file and (app_dir / code_file).exists():
            module_file = str(app_dir / code_file)

        exec_id = f"{workflow}-{datetime.datetime.now():%Y%m%d%H%M%S}-{uuid.uuid4().hex[:6]}"
        exec_dir = self.root / "executions" / exec_id
        exec_dir.mkdir(parents=True)
        (exec_dir / "status").write_text(STATUS_QUEUED)
        job = {
            "workflow": workflow,
            "app_module": manifest["app_module"],
            "object_name": manifest["object_name"],
            "module_file": module_file,
            "schedule_name": schedule_name,
            "model_name": manifest["model_name"],
        }
        (exec_dir / "job.json").write_text(json.dumps(job))
        from unionml_amd.utils.serialization import tensors_to_cpu

        (exec_dir / "inputs.pkl").write_bytes(cloudpickle.dumps(tensors_to_cpu(inputs or {})))

        env = dict(os.environ)
        if n_gpus is None:
            n_gpus = manifest.get("gpu_resources", 0)
        devices = self.gpus.assign(n_gpus)
        if devices is not None:
            env["HIP_VISIBLE_DEVICES"] = devices
        # make the app module AND this framework importable in the worker
        # (the worker may run from any cwd; pytest tmp dirs, cron, etc.)
        import unionml_amd

        pythonpath = [str(Path(unionml_amd.__file__).parent.parent)]
        if module_file:
            pythonpath.insert(0, str(Path(module_file).parent))
        if env.get("PYTHONPATH"):
            pythonpath.append(env["PYTHONPATH"])
        env["PYTHONPATH"] = os.pathsep.join(pythonpath)

        with open(exec_dir / "worker.log", "wb") as log:
            proc = subprocess.Popen(
                [sys.executable, "-m", "unionml_amd.runner", str(exec_dir)],
                stdout=log,
                stderr=subprocess.STDOUT,
                env=env,
            )
        (exec_dir / "pid").write_text(str(proc.pid))
        execution = Execution(id=exec_id, workflow=workflow, path=str(exec_dir))
        if wait_process:
            proc.wait()
        return execution

    def get_execution(self, execution_id: str) -> Execution:
        """Rehydrate an execution handle by id (reference fetches from
        the Flyte admin: remote.py:236-269)."""
        path = self.root / "executions" / execution_id
        if not path.is_dir():
            raise ModelArtifactNotFound(f"execution '{execution_id}' not found")
        return Execution(id=execution_id, workflow=execution_id.split("-")[0], path=str(path))

    def wait(self, execution: Execution, timeout: Optional[float] = None) -> Execution:
        deadline = time.monotonic() + (timeout or 3600)
        while time.monotonic() < deadline:
            status = execution.status
            if status in (STATUS_SUCCEEDED, STATUS_FAILED):
                if status == STATUS_FAILED:
                    log = Path(execution.path) / "worker.log"
                    tail = log.read_text()[-3000:] if log.exists() else ""
                    raise RuntimeError(f"execution {execution.id} FAILED:\n{tail}")
                return execution
            # fail fast if the worker died before reaching a terminal status
            # (e.g. crashed during interpreter startup)
            pid_file = Path(execution.path) / "pid"
            if pid_file.exists() and not _pid_alive(int(pid_file.read_text())):
                # re-check: the worker may have written the status just
                # before exiting
                if execution.status not in (STATUS_SUCCEEDED, STATUS_FAILED):
                    log = Path(execution.path) / "worker.log"
                    tail = log.read_text()[-3000:] if log.exists() else ""
                    (Path(execution.path) / "status").write_text(STATUS_FAILED)
                    raise RuntimeError(
                        f"execution {execution.id} worker died without a status:\n{tail}"
                    )
            time.sleep(0.05)
        raise TimeoutError(f"execution {execution.id} did not finish in time")

    def fetch_output(self, execution: Execution):
        out = Path(execution.path) / "outputs.pkl"
        if not out.exists():
            raise ModelArtifactNotFound(f"execution {execution.id} has no outputs")
        with open(out, "rb") as f:
            return pickle.load(f)

    # ------------------------------------------------------------------
    # model registry
    # ------------------------------------------------------------------

    def _executions(self, workflow_prefix: Optional[str] = None) -> List[Execution]:
        execs = []
        for d in sorted((self.root / "executions").iterdir()):
            if workflow_prefix and not d.name.startswith(workflow_prefix):
                continue
            execs.append(Execution(id=d.name, workflow=d.name.split("-")[0], path=str(d)))
        return execs

    def fetch_model_artifact(
        self, model, app_version: Optional[str] = None, model_version: str = "latest"
    ) -> ModelArtifact:
        """latest (or id-addressed) SUCCEEDED training run's outputs
        (reference: remote.py:200-233, 272-280)."""
        if model_version == "latest":
            candidates = [
                e for e in self._executions("train") if e.status == STATUS_SUCCEEDED
            ]
            if not candidates:
                raise ModelArtifactNotFound(
                    f"no successful training execution in project '{self.project}'"
                )
            execution = candidates[-1]
        else:
            execution = Execution(
                id=model_version,
                workflow="train",
                path=str(self.root / "executions" / model_version),
            )
            if execution.status != STATUS_SUCCEEDED:
                raise ModelArtifactNotFound(
                    f"execution '{model_version}' not found or not SUCCEEDED"
                )
        model_obj, hp, metrics = self.fetch_output(execution)
        return ModelArtifact(model_obj, hp, metrics)

    def list_model_versions(self, model, app_version: Optional[str] = None, limit: int = 10):
        return [
            e.id for e in self._executions("train") if e.status == STATUS_SUCCEEDED
        ][-limit:][::-1]

    def list_prediction_ids(self, model, app_version: Optional[str] = None, limit: int = 10):
        return [
            e.id
            for e in self._executions("predict")
            if e.status == STATUS_SUCCEEDED
        ][-limit:][::-1]

    def list_scheduled_runs(self, model, schedule_name: str, kind: str, limit: int = 10):
        out = []
        for e in self._executions():
            job_file = Path(e.path) / "job.json"
            if job_file.exists():
                job = json.loads(job_file.read_text())
                if job.get("schedule_name") == schedule_name:
                    out.append(e.id)
        return out[-limit:][::-1]

    # ------------------------------------------------------------------
    # schedules
    # ------------------------------------------------------------------

    def set_schedules_active(
        self, model, schedule_names: Optional[List[str]], active: bool, app_version: Optional[str] = None
    ):
        app_version = app_version or self.latest_app_version()
        manifest = self._manifest(app_version)
        changed = []
        for lp in manifest["launchplans"]:
            if schedule_names is None or lp["name"] in schedule_names:
                lp["active"] = active
                changed.append(lp["name"])
        (self.root / "apps" / (app_version or "") / "manifest.json").write_text(
            json.dumps(manifest, indent=2)
        )
        return changed

    def run_scheduler(
        self,
        model,
        app_version: Optional[str] = None,
        iterations: Optional[int] = None,
        poll_s: float = 1.0,
        now_fn=datetime.datetime.now,
    ):
        """Fire active launch plans when due. ``iterations`` bounds the
        loop for tests/CLI; None = run forever."""
        from unionml_amd.schedule import Schedule, next_fire_time

        app_version = app_version or self.latest_app_version()
        manifest = self._manifest(app_version)
        import cloudpickle

        lp_inputs = {}
        lp_inputs_file = self.root / "apps" / (app_version or "") / "launchplan_inputs.pkl"
        if lp_inputs_file.exists():
            lp_inputs = cloudpickle.loads(lp_inputs_file.read_bytes())

        next_fire: Dict[str, datetime.datetime] = {}
        it = 0
        while iterations is None or it < iterations:
            it += 1
            now = now_fn()
            for lp in manifest["launchplans"]:
                if not lp["active"]:
                    continue
                sched = Schedule(
                    type=lp["type"],
                    name=lp["name"],
                    expression=lp["expression"],
                    fixed_rate=datetime.timedelta(seconds=lp["fixed_rate_s"])
                    if lp["fixed_rate_s"]
                    else None,
                    time_arg=lp["time_arg"],
                )
                due = next_fire.get(lp["name"])
                if due is None:
                    next_fire[lp["name"]] = next_fire_time(sched, now)
                    continue
                if now >= due:
                    next_fire[lp["name"]] = next_fire_time(sched, now)
                    inputs = dict(lp_inputs.get(lp["name"], {}))
                    if lp["time_arg"]:
                        inputs[lp["time_arg"]] = now
                    wf = lp["workflow"]
                    if wf == "train":
                        # schedule inputs may carry workflow-level stage
                        # kwargs (hyperparameters, trainer_kwargs, ...);
                        # everything else is a reader kwarg (incl. the
                        # injected time_arg)
                        exec_inputs = {
                            k: inputs.pop(k, None)
                            for k in (
                                "hyperparameters",
                                "loader_kwargs",
                                "splitter_kwargs",
                                "parser_kwargs",
                                "trainer_kwargs",
                            )
                        }
                        explicit = inputs.pop("reader_kwargs", None)
                        exec_inputs["reader_kwargs"] = {**inputs, **(explicit or {})}
                    else:
                        model_obj = inputs.pop("model_object", None)
                        if model_obj is None:
                            model_obj = self.fetch_model_artifact(model).model_object
                        if "features" in inputs:
                            exec_inputs = dict(
                                model_object=model_obj, features=inputs["features"]
                            )
                            wf = "predict_from_features"
                        else:
                            exec_inputs = dict(model_object=model_obj, reader_kwargs=inputs)
                    logger.info("schedule %s firing workflow %s", lp["name"], wf)
                    self.execute(
                        model,
                        workflow=wf,
                        app_version=app_version,
                        inputs=exec_inputs,
                        schedule_name=lp["name"],
                    )
            if iterations is None or it < iterations:
                time.sleep(poll_s)
