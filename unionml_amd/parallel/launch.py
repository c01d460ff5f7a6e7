"""Process-per-GPU launcher for ``Model.train(dp=N)``.

Spawns one worker process per GPU, initializes ``torch.distributed``
(RCCL over xGMI on ROCm devices, gloo on CPU), row-shards the train
split per rank, runs the registered trainer body in every rank, and
returns rank 0's artifact. The trainer body itself is unchanged — the
"one code path" contract (SURVEY.md §7 hard-part 5): trainers opt into
gradient synchronization via :func:`unionml_amd.parallel.maybe_wrap`
(the framework's built-in trainers do).

If a process group already exists (launched under
``torch.distributed.run``), the current process acts as its own rank
and no spawning happens.
"""

import os
import pickle
import socket
import tempfile
import traceback
from pathlib import Path
from typing import Any, Dict, Optional, Tuple

import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from unionml_amd._logging import logger
from unionml_amd.parallel.ddp import shard


def _free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _comm_backend() -> str:
    return "nccl" if torch.cuda.is_available() else "gloo"


def run_sharded_train_body(
    model,
    *,
    hyperparameters=None,
    loader_kwargs=None,
    splitter_kwargs=None,
    parser_kwargs=None,
    trainer_kwargs=None,
    reader_kwargs=None,
) -> Tuple[Any, Any, Dict[str, float]]:
    """Rank-local train body: read, split, shard this rank's train rows,
    train, evaluate (on the full splits, rank 0 only)."""
    ds = model._dataset
    raw = ds.dataset_task()(**(reader_kwargs or {}))
    data = ds.get_data(
        raw,
        loader_kwargs=loader_kwargs,
        splitter_kwargs=splitter_kwargs,
        parser_kwargs=parser_kwargs,
    )
    hyperparameters = model._coerce_hyperparameters(hyperparameters)
    train_split = [shard(el) if _shardable(el) else el for el in data["train"]]
    model_obj = model._call_init(hyperparameters)
    model_obj = model._trainer(model_obj, *train_split, **(trainer_kwargs or {}))
    metrics: Dict[str, float] = {}
    rank = dist.get_rank() if dist.is_initialized() else 0
    if rank == 0 and model._evaluator is not None:
        for split_name, split in data.items():
            metrics[split_name] = model._evaluator(model_obj, *split)
    return model_obj, hyperparameters, metrics


def _shardable(el) -> bool:
    try:
        len(el)
        return not isinstance(el, (str, bytes, dict))
    except TypeError:
        return False


def _dp_worker(rank: int, world: int, port: int, model_blob: bytes, kwargs_blob: bytes, out_dir: str):
    try:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ["MASTER_PORT"] = str(port)
        if torch.cuda.is_available():
            torch.cuda.set_device(rank % torch.cuda.device_count())
        dist.init_process_group(_comm_backend(), rank=rank, world_size=world)
        import cloudpickle

        model = cloudpickle.loads(model_blob)
        kwargs = pickle.loads(kwargs_blob)
        model_obj, hp, metrics = run_sharded_train_body(model, **kwargs)
        if rank == 0:
            from unionml_amd.artifact import ModelArtifact

            model.artifact = ModelArtifact(model_obj, hp, metrics)
            model.save(str(Path(out_dir) / "artifact.bin"))
            with open(Path(out_dir) / "meta.pkl", "wb") as f:
                pickle.dump({"hyperparameters": hp, "metrics": metrics}, f)
        dist.barrier()
    except Exception:
        with open(Path(out_dir) / f"error_rank{rank}.txt", "w") as f:
            f.write(traceback.format_exc())
        raise
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def train_data_parallel(
    model,
    dp: int,
    *,
    hyperparameters=None,
    loader_kwargs=None,
    splitter_kwargs=None,
    parser_kwargs=None,
    trainer_kwargs=None,
    reader_kwargs=None,
) -> Tuple[Any, Any, Dict[str, float]]:
    """Entry point used by ``Model.train(dp=N)``."""
    kwargs = dict(
        hyperparameters=hyperparameters,
        loader_kwargs=loader_kwargs,
        splitter_kwargs=splitter_kwargs,
        parser_kwargs=parser_kwargs,
        trainer_kwargs=trainer_kwargs,
        reader_kwargs=reader_kwargs,
    )

    if dist.is_available() and dist.is_initialized():
        # already inside a torchrun-style launch: act as our own rank
        return run_sharded_train_body(model, **kwargs)

    # RCCL needs one distinct device per rank: dp > visible GPUs would
    # die deep inside communicator init with an opaque error — fail
    # here with an actionable one instead (CPU/gloo has no such limit)
    if torch.cuda.is_available() and dp > torch.cuda.device_count():
        raise ValueError(
            f"dp={dp} exceeds the {torch.cuda.device_count()} visible GPU(s): "
            "RCCL requires one device per rank. Reduce dp or widen "
            "HIP_VISIBLE_DEVICES."
        )

    import cloudpickle

    model_blob = cloudpickle.dumps(model)
    kwargs_blob = pickle.dumps(kwargs)
    port = _free_port()

    with tempfile.TemporaryDirectory(prefix="unionml_amd_dp_") as out_dir:
        try:
            mp.start_processes(
                _dp_worker,
                args=(dp, port, model_blob, kwargs_blob, out_dir),
                nprocs=dp,
                join=True,
                start_method="spawn",
            )
        except Exception as exc:
            errors = sorted(Path(out_dir).glob("error_rank*.txt"))
            detail = errors[0].read_text() if errors else str(exc)
            raise RuntimeError(f"dp worker failed:\n{detail}") from exc
        errors = sorted(Path(out_dir).glob("error_rank*.txt"))
        if errors:
            raise RuntimeError(f"dp worker failed:\n{errors[0].read_text()}")
        model_obj = model._loader(str(Path(out_dir) / "artifact.bin"))
        with open(Path(out_dir) / "meta.pkl", "rb") as f:
            meta = pickle.load(f)
    return model_obj, meta["hyperparameters"], meta["metrics"]
