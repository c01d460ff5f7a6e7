"""Task/Workflow executor + cache + tracker/resolver tests (net-new
coverage for this build's own L3 layer)."""

import pandas as pd
import pytest

from unionml_amd.task import Task, Workflow, _TaskCache


def test_task_call_and_metadata():
    t = Task(lambda x: x + 1, "inc")
    assert t(x=1) == 2
    assert t.name == "inc"


def test_task_cache(tmp_path, monkeypatch):
    import unionml_amd.task as task_mod

    monkeypatch.setattr(task_mod, "_GLOBAL_TASK_CACHE", _TaskCache(tmp_path))
    calls = []

    def expensive(n: int):
        calls.append(n)
        return n * 2

    t = Task(expensive, "exp", cache=True, cache_version="1")
    assert t(n=4) == 8
    assert t(n=4) == 8
    assert calls == [4]  # second call served from cache
    t2 = Task(expensive, "exp", cache=True, cache_version="2")
    assert t2(n=4) == 8
    assert calls == [4, 4]  # version bump invalidates


def test_workflow_dag_execution():
    wf = Workflow(
        name="wf",
        inputs=["a", "b"],
        outputs=[("sum", ("node", 1, None)), ("prod", ("node", 2, None))],
    )
    n0 = wf.add_node(Task(lambda a, b: (a + b, a * b), "both"), {"a": ("input", "a"), "b": ("input", "b")})
    wf.add_node(Task(lambda x: x, "pick_sum"), {"x": ("node", n0, 0)})
    wf.add_node(Task(lambda x: x, "pick_prod"), {"x": ("node", n0, 1)})
    assert wf(a=3, b=4) == (7, 12)


def test_workflow_missing_input():
    wf = Workflow(name="wf", inputs=["a"], outputs=[("out", ("node", 0, None))])
    wf.add_node(Task(lambda a: a, "id"), {"a": ("input", "a")})
    with pytest.raises(TypeError):
        wf()


def test_tracker_find_lhs(sklearn_model):
    # fixtures instantiate in model_fixtures; lhs lookup scans that module
    import model_fixtures  # noqa: F401

    name = None
    try:
        name = sklearn_model.find_lhs()
    except ValueError:
        pass  # fixture-local instances are not module-level; that's legal
    assert name is None or isinstance(name, str)


def test_resolver_roundtrip(tmp_path):
    """Write an app module, resolve its task by loader args."""
    app = tmp_path / "resolver_app.py"
    app.write_text(
        """
import pandas as pd
from unionml_amd import Dataset

ds = Dataset(name="resolver_ds", targets=["y"])

@ds.reader
def reader(n: int = 5) -> pd.DataFrame:
    return pd.DataFrame({"x": range(n), "y": [i % 2 for i in range(n)]})
"""
    )
    import sys

    sys.path.insert(0, str(tmp_path))
    try:
        from unionml_amd.task_resolver import load_task

        task = load_task(
            ["app-module", "resolver_app", "unionml-obj-name", "ds", "task-name", "dataset_task"]
        )
        out = task(n=3)
        assert len(out) == 3
        assert task.name == "resolver_ds.reader"
    finally:
        sys.path.remove(str(tmp_path))
        sys.modules.pop("resolver_app", None)


def test_loader_args_from_module_level_instance(tmp_path):
    app = tmp_path / "resolver_app2.py"
    app.write_text(
        """
import pandas as pd
from unionml_amd import Dataset

ds2 = Dataset(name="resolver_ds2", targets=["y"])

@ds2.reader
def reader(n: int = 5) -> pd.DataFrame:
    return pd.DataFrame({"x": range(n), "y": [0] * n})
"""
    )
    import sys

    sys.path.insert(0, str(tmp_path))
    try:
        import importlib

        mod = importlib.import_module("resolver_app2")
        from unionml_amd.task_resolver import loader_args

        args = loader_args(mod.ds2.dataset_task())
        assert args == [
            "app-module",
            "resolver_app2",
            "unionml-obj-name",
            "ds2",
            "task-name",
            "dataset_task",
        ]
    finally:
        sys.path.remove(str(tmp_path))
        sys.modules.pop("resolver_app2", None)


def test_task_retries():
    """retries=N re-runs a failing task body (reference forwards `retries`
    to flytekit tasks; our executor honors it directly)."""
    from unionml_amd.task import Task

    calls = []

    def flaky(x: int) -> int:
        calls.append(x)
        if len(calls) < 3:
            raise RuntimeError("transient")
        return x * 2

    t = Task(flaky, "flaky", retries=3)
    assert t(5) == 10
    assert len(calls) == 3

    calls.clear()
    t0 = Task(flaky, "flaky0", retries=0)
    import pytest as _pytest

    with _pytest.raises(RuntimeError):
        t0(5)
    assert len(calls) == 1


def test_reader_retries_kwarg_passthrough():
    """@dataset.reader(retries=2) must reach the compiled Task."""
    import pandas as pd

    from unionml_amd import Dataset

    ds = Dataset(name="r", targets=["y"])
    attempts = []

    @ds.reader(retries=2)
    def reader(n: int = 3) -> pd.DataFrame:
        attempts.append(n)
        if len(attempts) < 2:
            raise IOError("flaky source")
        return pd.DataFrame({"a": range(n), "y": [0] * n})

    task = ds.dataset_task()
    assert task.retries == 2
    out = task(n=4)
    assert len(out) == 4 and len(attempts) == 2


# ------------------- workflow DAG hardening (r02) ---------------------


def test_workflow_rejects_unknown_input_binding():
    from unionml_amd.task import Task, Workflow, WorkflowError

    wf = Workflow("w", inputs=["a"], outputs=[("out", ("node", 0, None))])
    t = Task(lambda x: x, "t")
    with pytest.raises(WorkflowError, match="unknown input"):
        wf.add_node(t, {"x": ("input", "nope")})


def test_workflow_rejects_forward_node_reference():
    from unionml_amd.task import Task, Workflow, WorkflowError

    wf = Workflow("w", inputs=["a"], outputs=[("out", ("node", 0, None))])
    t = Task(lambda x: x, "t")
    with pytest.raises(WorkflowError, match="acyclic"):
        wf.add_node(t, {"x": ("node", 3, None)})  # node 3 doesn't exist


def test_workflow_validates_outputs():
    from unionml_amd.task import Task, Workflow, WorkflowError

    wf = Workflow("w", inputs=["a"], outputs=[("out", ("node", 9, None))])
    wf.add_node(Task(lambda x: x, "t"), {"x": ("input", "a")})
    with pytest.raises(WorkflowError):
        wf(a=1)


def test_workflow_bad_output_index_is_clear_error():
    from unionml_amd.task import Task, Workflow, WorkflowError

    wf = Workflow("w", inputs=["a"], outputs=[("out", ("node", 0, 5))])
    wf.add_node(Task(lambda x: (x,), "t"), {"x": ("input", "a")})
    with pytest.raises(WorkflowError, match="output index"):
        wf(a=1)


def test_workflow_typed_inputs():
    from unionml_amd.task import Task, Workflow

    wf = Workflow("w", inputs={"n": int}, outputs=[("out", ("node", 0, None))])
    wf.add_node(Task(lambda n: n * 2, "t"), {"n": ("input", "n")})
    assert wf(n=4) == 8
    with pytest.raises(TypeError, match="expects"):
        wf(n="four")


def test_workflow_parallel_branches():
    """Independent nodes in one dependency wave run concurrently when
    parallel=True: two 0.2s sleepers finish in well under 0.4s."""
    import time as _time

    from unionml_amd.task import Task, Workflow

    def slow(tag):
        def fn(x):
            _time.sleep(0.2)
            return f"{tag}:{x}"

        return fn

    wf = Workflow(
        "w",
        inputs=["a"],
        outputs=[("out", ("node", 2, None))],
        parallel=True,
    )
    n0 = wf.add_node(Task(slow("l"), "left"), {"x": ("input", "a")})
    n1 = wf.add_node(Task(slow("r"), "right"), {"x": ("input", "a")})
    wf.add_node(
        Task(lambda l, r: (l, r), "join"),
        {"l": ("node", n0, None), "r": ("node", n1, None)},
    )
    t0 = _time.perf_counter()
    out = wf(a=1)
    wall = _time.perf_counter() - t0
    assert out == ("l:1", "r:1")
    assert wall < 0.38, f"branches did not run concurrently ({wall:.2f}s)"

    # same graph, sequential executor: takes >= 0.4s
    wf.parallel = False
    t0 = _time.perf_counter()
    wf(a=1)
    assert _time.perf_counter() - t0 >= 0.39


def test_workflow_dependency_order_not_insertion_order():
    """A node added early but depending on a later wave's result is
    impossible by construction; conversely, waves compute correct
    depths for diamond graphs."""
    from unionml_amd.task import Task, Workflow

    wf = Workflow("diamond", inputs=["a"], outputs=[("out", ("node", 3, None))])
    top = wf.add_node(Task(lambda x: x + 1, "top"), {"x": ("input", "a")})
    l = wf.add_node(Task(lambda x: x * 2, "l"), {"x": ("node", top, None)})
    r = wf.add_node(Task(lambda x: x * 3, "r"), {"x": ("node", top, None)})
    wf.add_node(Task(lambda p, q: p + q, "join"), {"p": ("node", l, None), "q": ("node", r, None)})
    assert wf(a=1) == 2 * 2 + 2 * 3
    assert [sorted(w) for w in wf._waves()] == [[0], [1, 2], [3]]


def test_workflow_parallel_branch_error_propagates():
    """A branch raising inside a parallel wave must fail the workflow
    call with the ORIGINAL exception (no hang, no swallowed error), and
    healthy sibling branches must not mask it."""
    from unionml_amd.task import Task, Workflow

    class Boom(RuntimeError):
        pass

    def bad(x):
        raise Boom("branch exploded")

    wf = Workflow(
        "w_err",
        inputs=["a"],
        outputs=[("out", ("node", 2, None))],
        parallel=True,
    )
    n0 = wf.add_node(Task(lambda x: x + 1, "ok"), {"x": ("input", "a")})
    n1 = wf.add_node(Task(bad, "bad"), {"x": ("input", "a")})
    wf.add_node(
        Task(lambda l, r: (l, r), "join"),
        {"l": ("node", n0, None), "r": ("node", n1, None)},
    )
    with pytest.raises(Boom, match="branch exploded"):
        wf(a=1)
