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


class WorkflowError(ValueError):
    """Raised for malformed workflow DAGs (bad bindings, cycles,
    unknown inputs) — at graph-build time, not execution time."""


class Workflow:
    """A static DAG of tasks with named (optionally typed) inputs and
    outputs.

    Every binding is validated when the node is added: "input" sources
    must name a declared workflow input, "node" sources must reference
    an already-added node (so the graph is acyclic by construction, and
    a bad output index fails before execution, not mid-run). Execution
    is dependency-driven: nodes are grouped into ready waves, and a
    wave with multiple independent nodes runs its branches concurrently
    when the workflow is built with ``parallel=True``.

    ``inputs`` may be a sequence of names, or a mapping name -> type
    for runtime input type checking.
    """

    def __init__(
        self,
        name: str,
        inputs,
        outputs: Sequence[Tuple[str, Tuple]],
        parallel: bool = False,
    ):
        self.name = name
        if isinstance(inputs, dict):
            self.input_names = list(inputs)
            self.input_types: Dict[str, Any] = dict(inputs)
        else:
            self.input_names = list(inputs)
            self.input_types = {}
        self.outputs = list(outputs)
        self.parallel = parallel
        self.nodes: List[WorkflowNode] = []

    def _check_source(self, source: Tuple, n_nodes: int, where: str) -> None:
        if not isinstance(source, tuple) or not source:
            raise WorkflowError(f"workflow '{self.name}': malformed binding {source!r} in {where}")
        kind = source[0]
        if kind == "input":
            if source[1] not in self.input_names:
                raise WorkflowError(
                    f"workflow '{self.name}': {where} references unknown input "
                    f"{source[1]!r} (declared: {self.input_names})"
                )
        elif kind == "node":
            _, node_idx, _out_idx = source
            if not (0 <= node_idx < n_nodes):
                raise WorkflowError(
                    f"workflow '{self.name}': {where} references node {node_idx}, but "
                    f"only {n_nodes} node(s) exist upstream — bindings may only point "
                    "at already-added nodes (this keeps the DAG acyclic by construction)"
                )
        else:
            raise WorkflowError(f"workflow '{self.name}': unknown binding kind {kind!r} in {where}")

    def add_node(self, task: Task, bindings: Dict[str, Tuple], kwargs_from: Optional[str] = None) -> int:
        for pname, src in bindings.items():
            self._check_source(src, len(self.nodes), f"node '{task.name}' param '{pname}'")
        if kwargs_from is not None and kwargs_from not in self.input_names:
            raise WorkflowError(
                f"workflow '{self.name}': node '{task.name}' kwargs_from references "
                f"unknown input {kwargs_from!r}"
            )
        self.nodes.append(WorkflowNode(task=task, bindings=bindings, kwargs_from=kwargs_from))
        return len(self.nodes) - 1

    def validate(self) -> None:
        """Validate the finished graph (outputs may reference any node)."""
        for out_name, src in self.outputs:
            self._check_source(src, len(self.nodes), f"output '{out_name}'")

    def _deps(self, node: WorkflowNode) -> set:
        return {src[1] for src in node.bindings.values() if src[0] == "node"}

    def _waves(self) -> List[List[int]]:
        """Group node indices into dependency waves (every node's deps
        are in strictly earlier waves)."""
        depth: Dict[int, int] = {}
        for i, node in enumerate(self.nodes):
            deps = self._deps(node)
            depth[i] = (max(depth[d] for d in deps) + 1) if deps else 0
        waves: Dict[int, List[int]] = {}
        for i, d in depth.items():
            waves.setdefault(d, []).append(i)
        return [waves[d] for d in sorted(waves)]

    def _resolve(self, source: Tuple, wf_inputs: Dict[str, Any], node_results: List[Any]):
        kind = source[0]
        if kind == "input":
            return wf_inputs[source[1]]
        if kind == "node":
            _, node_idx, out_idx = source
            result = node_results[node_idx]
            if out_idx is not None:
                try:
                    return result[out_idx]
                except (IndexError, KeyError, TypeError) as exc:
                    raise WorkflowError(
                        f"workflow '{self.name}': node {node_idx} "
                        f"({self.nodes[node_idx].task.name}) returned "
                        f"{type(result).__name__} without output index {out_idx!r}"
                    ) from exc
            return result
        raise WorkflowError(f"unknown binding source {source!r}")

    def _check_input_types(self, wf_inputs: Dict[str, Any]) -> None:
        for name, expected in self.input_types.items():
            if expected is None or expected is Any or name not in wf_inputs:
                continue
            value = wf_inputs[name]
            if value is None:
                continue
            origin = getattr(expected, "__origin__", expected)
            if isinstance(origin, type) and not isinstance(value, origin):
                raise TypeError(
                    f"workflow '{self.name}' input '{name}' expects "
                    f"{expected}, got {type(value).__name__}"
                )

    def _run_node(self, node: WorkflowNode, wf_inputs, node_results):
        kwargs = {
            pname: self._resolve(src, wf_inputs, node_results)
            for pname, src in node.bindings.items()
        }
        if node.kwargs_from is not None:
            extra = wf_inputs.get(node.kwargs_from) or {}
            kwargs.update(extra)
        t0 = time.perf_counter()
        out = node.task(**kwargs)
        logger.debug(
            "workflow %s: task %s took %.3fs", self.name, node.task.name, time.perf_counter() - t0
        )
        return out

    def __call__(self, **wf_inputs):
        missing = set(self.input_names) - set(wf_inputs)
        if missing:
            raise TypeError(f"workflow '{self.name}' missing inputs: {sorted(missing)}")
        self.validate()
        self._check_input_types(wf_inputs)
        node_results: List[Any] = [None] * len(self.nodes)
        for wave in self._waves():
            if self.parallel and len(wave) > 1:
                import concurrent.futures as cf

                with cf.ThreadPoolExecutor(max_workers=len(wave)) as pool:
                    futures = {
                        i: pool.submit(self._run_node, self.nodes[i], wf_inputs, node_results)
                        for i in wave
                    }
                    for i, fut in futures.items():
                        node_results[i] = fut.result()
            else:
                for i in wave:
                    node_results[i] = self._run_node(self.nodes[i], wf_inputs, node_results)
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
