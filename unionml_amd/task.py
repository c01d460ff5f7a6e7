"""Task & workflow compilation — the framework's own execution layer.

The reference compiles user callables into flytekit tasks and imperative
workflows executed either in-process or on a Flyte cluster
(unionml/utils.py:11-60, unionml/model.py:425-653). This build replaces
that with a self-contained layer:

- :class:`Task` — a named, cacheable, resource-annotated wrapper around
  a user callable, carrying resolver metadata so a worker process can
  rehydrate it by ``(app_module, object_name, task_builder)``.
- :class:`Workflow` — a small static DAG of tasks with named inputs and
  outputs, executed in-process by a deterministic topological executor.

The local executor is what runs for ``model.train()``/``model.predict()``;
the remote backend (unionml_amd/remote.py) ships the same Task objects to
worker processes. A task's :class:`~unionml_amd.defaults.Resources` may
request MI355X devices (``gpu=N``), which the local executor exposes via
``HIP_VISIBLE_DEVICES`` scheduling in the remote backend.
"""

import functools
import hashlib
import inspect
import pickle
import time
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Callable, Dict, List, Optional, Sequence, Tuple

from unionml_amd._logging import logger
from unionml_amd.defaults import DEFAULT_RESOURCES, Resources


class _TaskCache:
    """Tiny content-addressed on-disk memo cache for cache=True tasks.

    Mirrors the reference's Flyte-cache passthrough semantics
    (``@dataset.reader(cache=True, cache_version="1")`` —
    SURVEY.md §5 checkpoint/resume) without a cluster: keyed on
    (task name, cache_version, pickled inputs).
    """

    def __init__(self, root: Optional[Path] = None):
        self.root = root or (Path.home() / ".cache" / "unionml_amd" / "task_cache")

    def _key(self, task_name: str, cache_version: str, args: tuple, kwargs: dict) -> Optional[str]:
        try:
            blob = pickle.dumps((task_name, cache_version, args, sorted(kwargs.items())))
        except Exception:
            return None
        return hashlib.sha256(blob).hexdigest()

    def get(self, key: str):
        path = self.root / f"{key}.pkl"
        if path.exists():
            with open(path, "rb") as f:
                return True, pickle.load(f)
        return False, None

    def put(self, key: str, value: Any) -> None:
        try:
            self.root.mkdir(parents=True, exist_ok=True)
            tmp = self.root / f".{key}.tmp"
            with open(tmp, "wb") as f:
                pickle.dump(value, f)
            tmp.rename(self.root / f"{key}.pkl")
        except Exception as exc:  # cache failures must never fail the task
            logger.warning("task cache write failed: %s", exc)


_GLOBAL_TASK_CACHE = _TaskCache()


@dataclass
class ResolverArgs:
    """How a worker process finds this task again (reference:
    unionml/task_resolver.py:23-31 loader_args)."""

    app_module: str
    object_name: str
    task_builder: str

    def as_list(self) -> List[str]:
        return [
            "app-module",
            self.app_module,
            "unionml-obj-name",
            self.object_name,
            "task-name",
            self.task_builder,
        ]


class Task:
    """A named callable compiled from a user function.

    Unlike the reference's flytekit task, this object is a plain Python
    callable with metadata; serialization for remote execution goes
    through :class:`ResolverArgs`, never pickling of the function itself.
    """

    def __init__(
        self,
        fn: Callable,
        name: str,
        *,
        resources: Resources = DEFAULT_RESOURCES,
        cache: bool = False,
        cache_version: str = "0",
        retries: int = 0,
        retry_delay_s: float = 0.0,
        resolver_args: Optional[ResolverArgs] = None,
        signature: Optional[inspect.Signature] = None,
        **extra_task_kwargs,
    ):
        self.fn = fn
        self.name = name
        self.resources = resources
        self.cache = cache
        self.cache_version = cache_version
        self.retries = retries
        self.retry_delay_s = retry_delay_s
        # unknown task kwargs are kept as metadata (the reference forwards
        # arbitrary kwargs to flytekit's task(); we stay tolerant the same way)
        self.extra_task_kwargs = extra_task_kwargs
        self.resolver_args = resolver_args
        self._signature = signature or inspect.signature(fn)
        functools.update_wrapper(self, fn, updated=[])

    @property
    def signature(self) -> inspect.Signature:
        return self._signature

    def __call__(self, *args, **kwargs):
        if self.cache:
            key = _GLOBAL_TASK_CACHE._key(self.name, self.cache_version, args, kwargs)
            if key is not None:
                hit, value = _GLOBAL_TASK_CACHE.get(key)
                if hit:
                    logger.info("task %s: cache hit", self.name)
                    return value
        for attempt in range(self.retries + 1):
            try:
                out = self.fn(*args, **kwargs)
                break
            except Exception as exc:
                if attempt >= self.retries:
                    raise
                logger.warning(
                    "task %s failed (attempt %d/%d): %s — retrying",
                    self.name, attempt + 1, self.retries + 1, exc,
                )
                if self.retry_delay_s > 0:
                    time.sleep(self.retry_delay_s)
        if self.cache and key is not None:
            _GLOBAL_TASK_CACHE.put(key, out)
        return out

    def __repr__(self):
        return f"Task(name={self.name!r}, resources={self.resources})"


@dataclass
class WorkflowNode:
    """One node of a workflow DAG."""

    task: Task
    # mapping of task-parameter name -> source: either ("input", wf_input_name)
    # or ("node", node_index, output_index_or_None)
    bindings: Dict[str, Tuple] = field(default_factory=dict)
    # parameters forwarded as **kwargs from a dict-valued workflow input
    kwargs_from: Optional[str] = None


class Workflow:
    """A static DAG of tasks with named inputs/outputs, executed in-process.

    Nodes execute in insertion order (the builder adds them already
    topologically sorted); each node's inputs are bound either to
    workflow inputs or to upstream node outputs.
    """

    def __init__(self, name: str, inputs: Sequence[str], outputs: Sequence[Tuple[str, Tuple]]):
        self.name = name
        self.input_names = list(inputs)
        # outputs: list of (output_name, source) with source like bindings
        self.outputs = list(outputs)
        self.nodes: List[WorkflowNode] = []

    def add_node(self, task: Task, bindings: Dict[str, Tuple], kwargs_from: Optional[str] = None) -> int:
        self.nodes.append(WorkflowNode(task=task, bindings=bindings, kwargs_from=kwargs_from))
        return len(self.nodes) - 1

    def _resolve(self, source: Tuple, wf_inputs: Dict[str, Any], node_results: List[Any]):
        kind = source[0]
        if kind == "input":
            return wf_inputs[source[1]]
        if kind == "node":
            _, node_idx, out_idx = source
            result = node_results[node_idx]
            return result if out_idx is None else result[out_idx]
        raise ValueError(f"unknown binding source {source!r}")

    def __call__(self, **wf_inputs):
        missing = set(self.input_names) - set(wf_inputs)
        if missing:
            raise TypeError(f"workflow '{self.name}' missing inputs: {sorted(missing)}")
        node_results: List[Any] = []
        for node in self.nodes:
            kwargs = {
                pname: self._resolve(src, wf_inputs, node_results)
                for pname, src in node.bindings.items()
            }
            if node.kwargs_from is not None:
                extra = wf_inputs.get(node.kwargs_from) or {}
                kwargs.update(extra)
            t0 = time.perf_counter()
            node_results.append(node.task(**kwargs))
            logger.debug(
                "workflow %s: task %s took %.3fs", self.name, node.task.name, time.perf_counter() - t0
            )
        outs = tuple(self._resolve(src, wf_inputs, node_results) for _, src in self.outputs)
        return outs[0] if len(outs) == 1 else outs

    def __repr__(self):
        return (
            f"Workflow(name={self.name!r}, inputs={self.input_names}, "
            f"nodes={[n.task.name for n in self.nodes]}, "
            f"outputs={[o for o, _ in self.outputs]})"
        )


def inner_task(
    fn: Callable,
    *,
    owner,
    name: Optional[str] = None,
    resources: Resources = DEFAULT_RESOURCES,
    cache: bool = False,
    cache_version: str = "0",
    task_builder: Optional[str] = None,
    signature: Optional[inspect.Signature] = None,
    **task_kwargs,
) -> Task:
    """Wrap a closure into a :class:`Task` named ``{owner.name}.{fn name}``
    with resolver metadata (reference: unionml/utils.py:11-60).

    ``owner`` is the Dataset/Model the task belongs to; resolver args are
    filled lazily (the owner's lhs may not be bound yet at decoration
    time), so we store the builder name and let the remote layer call
    :meth:`Task.resolver_args` when packaging.
    """
    task_name = f"{getattr(owner, 'name', 'app')}.{name or fn.__name__}"
    resolver = None
    if task_builder is not None:
        try:
            resolver = ResolverArgs(
                app_module=owner.app_module,
                object_name=owner.find_lhs(),
                task_builder=task_builder,
            )
        except Exception:
            resolver = None  # resolved lazily at deploy time
    task = Task(
        fn,
        task_name,
        resources=resources,
        cache=cache,
        cache_version=cache_version,
        resolver_args=resolver,
        signature=signature,
        **task_kwargs,
    )
    task.__unionml_object__ = owner
    return task
