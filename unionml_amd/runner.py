"""Worker-process entrypoint for backend executions.

``python -m unionml_amd.runner <execution_dir>`` rehydrates the app via
the task resolver (the reference's container-side
``TaskResolver.load_task`` analog, unionml/task_resolver.py:16-21),
runs the named workflow on the stored inputs, and writes outputs +
status back to the execution directory.
"""

import json
import pickle
import sys
import traceback
from pathlib import Path


def main(exec_dir: str) -> int:
    exec_path = Path(exec_dir)
    (exec_path / "status").write_text("RUNNING")
    try:
        import cloudpickle

        job = json.loads((exec_path / "job.json").read_text())
        inputs = cloudpickle.loads((exec_path / "inputs.pkl").read_bytes())

        module_file = job.get("module_file")
        if module_file and Path(module_file).parent.exists():
            sys.path.insert(0, str(Path(module_file).parent))

        from unionml_amd.task_resolver import load_object

        try:
            model = load_object(job["app_module"], job["object_name"])
        except (ImportError, AttributeError):
            if not module_file:
                raise
            from unionml_amd.tracker import import_module_from_file

            module = import_module_from_file(Path(module_file).stem, module_file)
            model = getattr(module, job["object_name"])

        workflow = job["workflow"]
        if workflow == "train":
            wf = model.train_workflow()
            outputs = wf(
                hyperparameters=inputs.get("hyperparameters"),
                loader_kwargs=inputs.get("loader_kwargs"),
                splitter_kwargs=inputs.get("splitter_kwargs"),
                parser_kwargs=inputs.get("parser_kwargs"),
                trainer_kwargs=inputs.get("trainer_kwargs"),
                reader_kwargs=inputs.get("reader_kwargs") or {},
            )
        elif workflow == "predict":
            wf = model.predict_workflow()
            outputs = wf(
                model_object=inputs["model_object"],
                reader_kwargs=inputs.get("reader_kwargs") or {},
            )
        elif workflow == "predict_from_features":
            wf = model.predict_from_features_workflow()
            outputs = wf(model_object=inputs["model_object"], features=inputs["features"])
        else:
            raise ValueError(f"unknown workflow {workflow!r}")

        from unionml_amd.utils.serialization import tensors_to_cpu

        # device payloads must be CPU-portable across the process
        # boundary (plain pickle of CUDA storages does not round-trip)
        (exec_path / "outputs.pkl").write_bytes(
            cloudpickle.dumps(tensors_to_cpu(outputs, _copy_modules=False))
        )
        (exec_path / "status").write_text("SUCCEEDED")
        return 0
    except Exception:
        traceback.print_exc()
        (exec_path / "status").write_text("FAILED")
        return 1


if __name__ == "__main__":
    sys.exit(main(sys.argv[1]))
