"""Task resolver — rehydrate a compiled task in a worker process.

Loader args name the app module, the Dataset/Model variable, and the
task-builder method; ``load_task`` imports the module and calls e.g.
``model.train_task()`` (reference: unionml/task_resolver.py:16-31).
"""

import importlib
from typing import List, Sequence

from unionml_amd.task import ResolverArgs, Task


def loader_args(task: Task) -> List[str]:
    if task.resolver_args is None:
        owner = getattr(task, "__unionml_object__", None)
        if owner is None:
            raise ValueError(f"task {task.name} has no resolver metadata")
        task.resolver_args = ResolverArgs(
            app_module=owner.app_module,
            object_name=owner.find_lhs(),
            task_builder=task.name.split(".")[-1],
        )
    return task.resolver_args.as_list()


def load_task(args: Sequence[str]) -> Task:
    """``["app-module", M, "unionml-obj-name", O, "task-name", T]`` ->
    the rehydrated Task."""
    kv = dict(zip(args[::2], args[1::2]))
    module = importlib.import_module(kv["app-module"])
    obj = getattr(module, kv["unionml-obj-name"])
    builder = getattr(obj, kv["task-name"])
    return builder()


def load_object(app_module: str, object_name: str):
    """Rehydrate a Dataset/Model instance by module + variable name."""
    module = importlib.import_module(app_module)
    return getattr(module, object_name)
