"""Model API — registers model-lifecycle callables and compiles them
into train/predict pipelines, serving apps, and deployable tasks.

Capability parity with the reference Model (unionml/model.py:59-1566):
``@model.init/.trainer/.predictor/.evaluator/.saver/.loader`` decorators,
``train/predict/save/load/serve``, task & workflow builders, schedules,
artifact resolution, and remote deploy/train/predict.

MI355X-native additions:

- ``model.train(..., dp=N)`` runs the training body data-parallel across
  N GPUs of one node: one process per GPU over RCCL/xGMI, shards the
  train split per rank, and all-reduces gradients through
  :mod:`unionml_amd.parallel` (SURVEY.md §7 stage 5).
- Trainer bodies can opt into the hand-written CDNA4 tabular hot path
  via :mod:`unionml_amd.ops` without changing the decorator API.
"""

import dataclasses
import inspect
import os
from dataclasses import dataclass, field, make_dataclass
from inspect import signature
from pathlib import Path
from typing import Any, Callable, Dict, IO, List, Optional, Tuple, Type, Union

from unionml_amd import type_guards
from unionml_amd._logging import logger
from unionml_amd.artifact import ModelArtifact, default_loader, default_saver
from unionml_amd.dataset import Dataset
from unionml_amd.defaults import DEFAULT_RESOURCES, Resources
from unionml_amd.exceptions import ModelArtifactNotFound
from unionml_amd.schedule import LaunchPlan, Schedule, ScheduleType, create_scheduled_launchplan
from unionml_amd.task import Task, Workflow, inner_task
from unionml_amd.tracker import TrackedInstance


@dataclass
class BaseHyperparameters:
    """Base class users may subclass for typed hyperparameters
    (reference: model.py:35-43)."""


class Model(TrackedInstance):
    """Declarative model lifecycle for a unionml_amd app."""

    def __init__(
        self,
        name: str = "model",
        *,
        init: Optional[Union[Type, Callable]] = None,
        hyperparameter_type: Optional[Type] = None,
        hyperparameter_config: Optional[Dict[str, Type]] = None,
        dataset: Optional[Dataset] = None,
    ):
        self.name = name
        if dataset is None:
            raise ValueError("Model requires a dataset=Dataset(...) argument")
        self._dataset = dataset
        self._init_cls_or_fn: Optional[Union[Type, Callable]] = init
        self._hyperparameter_type: Optional[Type] = hyperparameter_type
        self._hyperparameter_config: Optional[Dict[str, Type]] = hyperparameter_config
        if hyperparameter_type is not None and hyperparameter_config is not None:
            raise ValueError(
                "pass either hyperparameter_type or hyperparameter_config, not both"
            )

        self._init_fn: Optional[Callable] = None
        self._trainer: Optional[Callable] = None
        self._trainer_task_kwargs: Dict[str, Any] = {}
        self._predictor: Optional[Callable] = None
        self._predictor_task_kwargs: Dict[str, Any] = {}
        self._prediction_callbacks: Tuple[Callable, ...] = ()
        self._evaluator: Optional[Callable] = None
        self._saver: Callable = self._default_saver
        self._loader: Callable = self._default_loader

        self.artifact: Optional[ModelArtifact] = None

        self._train_task: Optional[Task] = None
        self._predict_task: Optional[Task] = None
        self._predict_from_features_task: Optional[Task] = None

        self._training_schedules: List[Schedule] = []
        self._prediction_schedules: List[Schedule] = []

        # remote backend configuration (set by .remote(...))
        self._remote_config: Optional[Dict[str, Any]] = None
        self.__remote = None

    # ------------------------------------------------------------------
    # basic properties
    # ------------------------------------------------------------------

    @property
    def dataset(self) -> Dataset:
        return self._dataset

    def _declared_model_type(self) -> Optional[type]:
        """Model type as declared by init (class, or an init function's
        return annotation) — None when nothing is declared, so guards
        skip the model-type compat checks rather than comparing against
        a guessed ``object``."""
        if inspect.isclass(self._init_cls_or_fn):
            return self._init_cls_or_fn
        for fn in (self._init_fn, self._init_cls_or_fn):
            if fn is not None:
                ret = signature(fn).return_annotation
                if ret is not inspect.Signature.empty and inspect.isclass(ret):
                    return ret
        return None

    @property
    def model_type(self) -> type:
        """Infer the model object's type from init/trainer annotations
        (reference: model.py:1420-1423)."""
        if inspect.isclass(self._init_cls_or_fn):
            return self._init_cls_or_fn
        for fn in (self._init_cls_or_fn, self._init_fn):
            if fn is not None:
                ret = signature(fn).return_annotation
                if ret is not inspect.Signature.empty and inspect.isclass(ret):
                    return ret
        if self._trainer is not None:
            ret = signature(self._trainer).return_annotation
            if ret is not inspect.Signature.empty and inspect.isclass(ret):
                return ret
        return object

    @property
    def hyperparameter_type(self) -> Type:
        """Resolve the hyperparameter container type (reference 4-branch
        logic: model.py:168-204): explicit > dataclass annotation on the
        init arg > synthesized from init keyword annotations > dict."""
        if self._hyperparameter_type is not None:
            return self._hyperparameter_type
        if self._hyperparameter_config is not None:
            # dict-of-types branch (reference model.py:66,180-186): the
            # test fixtures of the reference use this path
            return make_dataclass(
                f"{type(self).__name__}Hyperparameters",
                [(name, typ) for name, typ in self._hyperparameter_config.items()],
            )

        init = self._init_fn or self._init_cls_or_fn
        if init is not None and not inspect.isclass(init):
            params = signature(init).parameters
            hp_param = params.get("hyperparameters")
            if hp_param is not None and dataclasses.is_dataclass(hp_param.annotation):
                return hp_param.annotation
            # synthesize from an init FUNCTION's keyword-only annotations
            # (reference branch 3, model.py:168-204)
            kw_fields = []
            for pname, p in params.items():
                if p.kind is not p.KEYWORD_ONLY or pname == "hyperparameters":
                    continue
                ann = p.annotation if p.annotation is not inspect.Parameter.empty else Any
                if p.default is not inspect.Parameter.empty:
                    kw_fields.append((pname, ann, field(default=p.default)))
                else:
                    kw_fields.append((pname, ann))
            if kw_fields:
                return make_dataclass(f"{type(self).__name__}Hyperparameters", kw_fields)

        if inspect.isclass(init):
            fields = []
            for pname, p in signature(init).parameters.items():
                if p.kind in (p.VAR_KEYWORD, p.VAR_POSITIONAL):
                    continue
                ann = p.annotation if p.annotation is not inspect.Parameter.empty else Any
                if p.default is not inspect.Parameter.empty:
                    fields.append((pname, ann, field(default=p.default)))
                else:
                    fields.append((pname, ann))
            if fields:
                return make_dataclass(f"{type(self).__name__}Hyperparameters", fields)

        return dict

    @property
    def trainer_params(self) -> Dict[str, inspect.Parameter]:
        """Keyword-only parameters of the trainer = per-run trainer kwargs
        (reference: model.py:416-423)."""
        if self._trainer is None:
            return {}
        return {
            name: p
            for name, p in signature(self._trainer).parameters.items()
            if p.kind == p.KEYWORD_ONLY
        }

    @property
    def training_schedules(self) -> List[Schedule]:
        return list(self._training_schedules)

    @property
    def prediction_schedules(self) -> List[Schedule]:
        return list(self._prediction_schedules)

    # ------------------------------------------------------------------
    # decorators
    # ------------------------------------------------------------------

    def init(self, fn: Callable) -> Callable:
        """Register the function creating a fresh model object from
        hyperparameters (reference: model.py:256-259)."""
        self._init_fn = fn
        return fn

    def trainer(
        self,
        fn: Optional[Callable] = None,
        *,
        cache: bool = False,
        cache_version: str = "0",
        resources: Resources = DEFAULT_RESOURCES,
        schedule: Optional[Schedule] = None,
        schedules: Optional[List[Schedule]] = None,
        **task_kwargs,
    ):
        """Register the training function (reference: model.py:261-303).

        The trainer takes ``(model, *split_data, **hyperparameter/custom
        kwargs)`` and returns the trained model. Keyword-only args become
        per-run trainer parameters.
        """

        def decorator(f: Callable) -> Callable:
            type_guards.guard_trainer(
                f,
                self._declared_model_type(),
                self._dataset.n_parser_outputs,
                self._dataset.parser_return_types,
            )
            self._trainer = f
            self._trainer_task_kwargs = dict(
                cache=cache, cache_version=cache_version, resources=resources, **task_kwargs
            )
            f.__unionml_model__ = self  # reference: model.py:292-294
            for s in [schedule] if schedule else (schedules or []):
                self.add_trainer_schedule(s)
            self._train_task = None
            return f

        return decorator(fn) if fn is not None else decorator

    def predictor(
        self,
        fn: Optional[Callable] = None,
        *,
        callbacks: Optional[List[Callable]] = None,
        cache: bool = False,
        cache_version: str = "0",
        resources: Resources = DEFAULT_RESOURCES,
        schedule: Optional[Schedule] = None,
        schedules: Optional[List[Schedule]] = None,
        **task_kwargs,
    ):
        """Register the prediction function (reference: model.py:319-367)."""

        def decorator(f: Callable) -> Callable:
            type_guards.guard_predictor(
                f, self._declared_model_type(), self._dataset.feature_type
            )
            for cb in callbacks or []:
                type_guards.guard_prediction_callback(cb)
            self._predictor = f
            self._prediction_callbacks = tuple(callbacks or ())
            self._predictor_task_kwargs = dict(
                cache=cache, cache_version=cache_version, resources=resources, **task_kwargs
            )
            f.__unionml_model__ = self
            for s in [schedule] if schedule else (schedules or []):
                self.add_predictor_schedule(s)
            self._predict_task = None
            self._predict_from_features_task = None
            return f

        return decorator(fn) if fn is not None else decorator

    def evaluator(self, fn: Callable) -> Callable:
        """Register the evaluation function (reference: model.py:387-404)."""
        type_guards.guard_evaluator(
            fn,
            self._declared_model_type(),
            self._dataset.n_parser_outputs,
            self._dataset.parser_return_types,
        )
        self._evaluator = fn
        return fn

    def saver(self, fn: Callable) -> Callable:
        type_guards.guard_saver(fn)
        self._saver = fn
        return fn

    def loader(self, fn: Callable) -> Callable:
        type_guards.guard_loader_fn(fn)
        self._loader = fn
        return fn

    # ------------------------------------------------------------------
    # hyperparameter handling
    # ------------------------------------------------------------------

    def _coerce_hyperparameters(self, hyperparameters):
        hp_type = self.hyperparameter_type
        if hyperparameters is None:
            return None
        if dataclasses.is_dataclass(hp_type) and isinstance(hyperparameters, dict):
            try:
                return hp_type(**hyperparameters)
            except TypeError:
                return hyperparameters
        return hyperparameters

    def _hyperparameters_as_kwargs(self, hyperparameters) -> Dict[str, Any]:
        if hyperparameters is None:
            return {}
        if dataclasses.is_dataclass(hyperparameters):
            return dataclasses.asdict(hyperparameters)
        if isinstance(hyperparameters, dict):
            return dict(hyperparameters)
        return {"hyperparameters": hyperparameters}

    def _call_init(self, hyperparameters) -> Any:
        """Create a fresh model object (reference: model.py:1425-1430).

        Function inits are called ``init(hyperparameters=dict)`` when
        they declare a ``hyperparameters`` parameter, else with the
        hyperparameters spread as keyword arguments (the synthesized-
        dataclass style, reference branch 3)."""
        hp_kwargs = self._hyperparameters_as_kwargs(hyperparameters)
        init = self._init_fn or self._init_cls_or_fn
        if init is None:
            raise ValueError(
                f"model '{self.name}' has no init: pass init= to Model(...) or use @model.init"
            )
        if inspect.isclass(init):
            return init(**hp_kwargs)
        if "hyperparameters" in signature(init).parameters:
            return init(hyperparameters=hp_kwargs)
        return init(**hp_kwargs)

    # ------------------------------------------------------------------
    # compiled tasks
    # ------------------------------------------------------------------

    def train_task(self) -> Task:
        """Compile the train body into a Task (reference: model.py:512-578)."""
        if self._train_task is not None:
            return self._train_task
        if self._trainer is None:
            raise ValueError(f"model '{self.name}' has no @trainer registered")

        def train_task(
            *,
            hyperparameters=None,
            loader_kwargs=None,
            splitter_kwargs=None,
            parser_kwargs=None,
            trainer_kwargs=None,
            raw_data=None,
        ):
            return self._train_body(
                raw_data,
                hyperparameters=hyperparameters,
                loader_kwargs=loader_kwargs,
                splitter_kwargs=splitter_kwargs,
                parser_kwargs=parser_kwargs,
                trainer_kwargs=trainer_kwargs,
            )

        self._train_task = inner_task(
            train_task,
            owner=self,
            name="train_task",
            task_builder="train_task",
            **self._trainer_task_kwargs,
        )
        return self._train_task

    def _train_body(
        self,
        raw_data,
        *,
        hyperparameters=None,
        loader_kwargs=None,
        splitter_kwargs=None,
        parser_kwargs=None,
        trainer_kwargs=None,
    ) -> Tuple[Any, Any, Dict[str, float]]:
        """The task body shared by local and remote execution
        (reference: model.py:560-575)."""
        trainer_kwargs = trainer_kwargs or {}
        hyperparameters = self._coerce_hyperparameters(hyperparameters)
        data = self._dataset.get_data(
            raw_data,
            loader_kwargs=loader_kwargs,
            splitter_kwargs=splitter_kwargs,
            parser_kwargs=parser_kwargs,
        )
        model_obj = self._call_init(hyperparameters)
        model_obj = self._trainer(model_obj, *data["train"], **trainer_kwargs)
        metrics: Dict[str, float] = {}
        if self._evaluator is not None:
            for split_name, split in data.items():
                metrics[split_name] = self._evaluator(model_obj, *split)
        return model_obj, hyperparameters, metrics

    def predict_task(self) -> Task:
        """Compile the raw-data predict body (reference: model.py:580-617):
        re-runs parser + feature_transformer on reader output, then the
        predictor + callbacks."""
        if self._predict_task is not None:
            return self._predict_task
        if self._predictor is None:
            raise ValueError(f"model '{self.name}' has no @predictor registered")

        def predict_task(*, model_object, raw_data, parser_kwargs=None):
            ds = self._dataset
            pk = {**ds.parser_kwargs, **(parser_kwargs or {})}
            loaded = ds._loader(raw_data)
            parsed = list(ds._parser(loaded, **pk))
            features = ds._feature_transformer(parsed[ds._parser_feature_key])
            return self._run_predictor(model_object, features)

        self._predict_task = inner_task(
            predict_task,
            owner=self,
            name="predict_task",
            task_builder="predict_task",
            **self._predictor_task_kwargs,
        )
        return self._predict_task

    def predict_from_features_task(self) -> Task:
        """Compile the model-ready-features predict body (reference:
        model.py:619-653)."""
        if self._predict_from_features_task is not None:
            return self._predict_from_features_task
        if self._predictor is None:
            raise ValueError(f"model '{self.name}' has no @predictor registered")

        def predict_from_features_task(*, model_object, features):
            return self._run_predictor(model_object, features)

        self._predict_from_features_task = inner_task(
            predict_from_features_task,
            owner=self,
            name="predict_from_features_task",
            task_builder="predict_from_features_task",
            **self._predictor_task_kwargs,
        )
        return self._predict_from_features_task

    def _run_predictor(self, model_object, features):
        predictions = self._predictor(model_object, features)
        for cb in self._prediction_callbacks:
            try:
                cb(model_object, features, predictions)
            except Exception:  # callbacks must never fail serving
                logger.exception(
                    "prediction callback %s raised; swallowing (reference: model.py:608-612)",
                    getattr(cb, "__name__", cb),
                )
        return predictions

    # ------------------------------------------------------------------
    # workflows
    # ------------------------------------------------------------------

    def train_workflow(self) -> Workflow:
        """dataset_task -> train_task (reference: model.py:425-471)."""
        wf = Workflow(
            name=f"{self.name}.train",
            inputs=[
                "hyperparameters",
                "loader_kwargs",
                "splitter_kwargs",
                "parser_kwargs",
                "trainer_kwargs",
                "reader_kwargs",
            ],
            outputs=[
                ("model_object", ("node", 1, 0)),
                ("hyperparameters", ("node", 1, 1)),
                ("metrics", ("node", 1, 2)),
            ],
        )
        n0 = wf.add_node(self._dataset.dataset_task(), bindings={}, kwargs_from="reader_kwargs")
        wf.add_node(
            self.train_task(),
            bindings={
                "raw_data": ("node", n0, None),
                "hyperparameters": ("input", "hyperparameters"),
                "loader_kwargs": ("input", "loader_kwargs"),
                "splitter_kwargs": ("input", "splitter_kwargs"),
                "parser_kwargs": ("input", "parser_kwargs"),
                "trainer_kwargs": ("input", "trainer_kwargs"),
            },
        )
        return wf

    def predict_workflow(self) -> Workflow:
        """dataset_task -> predict_task (reference: model.py:473-495)."""
        wf = Workflow(
            name=f"{self.name}.predict",
            inputs=["model_object", "reader_kwargs"],
            outputs=[("predictions", ("node", 1, None))],
        )
        n0 = wf.add_node(self._dataset.dataset_task(), bindings={}, kwargs_from="reader_kwargs")
        wf.add_node(
            self.predict_task(),
            bindings={"model_object": ("input", "model_object"), "raw_data": ("node", n0, None)},
        )
        return wf

    def predict_from_features_workflow(self) -> Workflow:
        """single predict_from_features_task node (reference: model.py:497-510)."""
        wf = Workflow(
            name=f"{self.name}.predict_from_features",
            inputs=["model_object", "features"],
            outputs=[("predictions", ("node", 0, None))],
        )
        wf.add_node(
            self.predict_from_features_task(),
            bindings={"model_object": ("input", "model_object"), "features": ("input", "features")},
        )
        return wf

    # ------------------------------------------------------------------
    # local entrypoints
    # ------------------------------------------------------------------

    def train(
        self,
        hyperparameters: Optional[Union[dict, Any]] = None,
        loader_kwargs: Optional[dict] = None,
        splitter_kwargs: Optional[dict] = None,
        parser_kwargs: Optional[dict] = None,
        trainer_kwargs: Optional[dict] = None,
        dp: int = 1,
        **reader_kwargs,
    ) -> Tuple[Any, Dict[str, float]]:
        """Train locally (reference: model.py:655-709).

        ``dp > 1`` runs the train body data-parallel across ``dp`` GPUs
        of this node (one process per GPU, RCCL gradient all-reduce over
        xGMI); the returned artifact is rank 0's.
        """
        if dp > 1:
            from unionml_amd.parallel.launch import train_data_parallel

            model_obj, hp, metrics = train_data_parallel(
                self,
                dp=dp,
                hyperparameters=hyperparameters,
                loader_kwargs=loader_kwargs,
                splitter_kwargs=splitter_kwargs,
                parser_kwargs=parser_kwargs,
                trainer_kwargs=trainer_kwargs,
                reader_kwargs=reader_kwargs,
            )
        else:
            wf = self.train_workflow()
            model_obj, hp, metrics = wf(
                hyperparameters=hyperparameters,
                loader_kwargs=loader_kwargs,
                splitter_kwargs=splitter_kwargs,
                parser_kwargs=parser_kwargs,
                trainer_kwargs=trainer_kwargs,
                reader_kwargs=reader_kwargs,
            )
        self.artifact = ModelArtifact(model_obj, hp, metrics)
        return model_obj, metrics

    def predict(self, features: Any = None, **reader_kwargs):
        """Predict locally from features or reader kwargs
        (reference: model.py:711-741)."""
        if self.artifact is None:
            raise ModelArtifactNotFound(
                f"model '{self.name}' has no artifact: train or load a model first"
            )
        if features is not None and reader_kwargs:
            raise ValueError("pass either features=... or reader kwargs, not both")
        if features is not None:
            features = self._dataset.get_features(features)
            wf = self.predict_from_features_workflow()
            return wf(model_object=self.artifact.model_object, features=features)
        wf = self.predict_workflow()
        return wf(model_object=self.artifact.model_object, reader_kwargs=reader_kwargs)

    # ------------------------------------------------------------------
    # persistence
    # ------------------------------------------------------------------

    def save(self, file: Union[str, Path, IO], *args, **kwargs):
        """Save the current artifact (reference: model.py:743-757)."""
        if self.artifact is None:
            raise ModelArtifactNotFound(f"model '{self.name}' has no artifact to save")
        return self._saver(
            self.artifact.model_object, self.artifact.hyperparameters, file, *args, **kwargs
        )

    def load(self, file: Union[str, Path, IO], *args, **kwargs):
        """Load a model artifact and set ``self.artifact``
        (reference: model.py:758-769)."""
        model_obj = self._loader(file, *args, **kwargs)
        self.artifact = ModelArtifact(model_obj)
        return model_obj

    def load_from_env(self, env_var: str = "UNIONML_MODEL_PATH", *args, **kwargs):
        path = os.environ.get(env_var)
        if not path:
            raise ModelArtifactNotFound(f"environment variable {env_var} is not set")
        return self.load(path, *args, **kwargs)

    def _default_saver(self, model_obj, hyperparameters, file, **kwargs):
        return default_saver(model_obj, hyperparameters, file, **kwargs)

    def _default_loader(self, file, **kwargs):
        mt = self.model_type
        return default_loader(
            file,
            model_type=mt if mt is not object else None,
            init=self._init_fn
            or (None if inspect.isclass(self._init_cls_or_fn) else self._init_cls_or_fn),
            hyperparameter_type=self.hyperparameter_type
            if self.hyperparameter_type is not dict
            else None,
            **kwargs,
        )

    # ------------------------------------------------------------------
    # serving
    # ------------------------------------------------------------------

    def serve(
        self,
        app,
        *,
        remote: bool = False,
        app_version: Optional[str] = None,
        model_version: str = "latest",
        batch: bool = False,
        max_batch_size: int = 64,
        max_delay_ms: float = 0.0,
    ):
        """Attach serving routes for this model to a FastAPI app
        (reference: model.py:771-784). ``batch=True`` enables the
        MI355X dynamic batcher + hipGraph-captured inference."""
        from unionml_amd.fastapi import serving_app

        serving_app(
            self,
            app,
            remote=remote,
            app_version=app_version,
            model_version=model_version,
            batch=batch,
            max_batch_size=max_batch_size,
            max_delay_ms=max_delay_ms,
        )

    # ------------------------------------------------------------------
    # schedules
    # ------------------------------------------------------------------

    def add_trainer_schedule(self, schedule: Schedule):
        if schedule.type != ScheduleType.trainer:
            raise ValueError(f"schedule '{schedule.name}' is not a trainer schedule")
        if any(s.name == schedule.name for s in self._training_schedules):
            raise ValueError(f"duplicate training schedule name '{schedule.name}'")
        self._training_schedules.append(schedule)

    def add_predictor_schedule(self, schedule: Schedule):
        if schedule.type != ScheduleType.predictor:
            raise ValueError(f"schedule '{schedule.name}' is not a predictor schedule")
        if any(s.name == schedule.name for s in self._prediction_schedules):
            raise ValueError(f"duplicate prediction schedule name '{schedule.name}'")
        self._prediction_schedules.append(schedule)

    def schedule_training(
        self,
        name: str,
        *,
        expression: Optional[str] = None,
        offset: Optional[str] = None,
        fixed_rate=None,
        time_arg: Optional[str] = None,
        inputs: Optional[dict] = None,
        reader_time_arg: Optional[str] = None,
        activate_on_deploy: bool = True,
        launchplan_kwargs: Optional[dict] = None,
    ):
        """Register a training schedule (reference: model.py:786-855)."""
        self.add_trainer_schedule(
            Schedule(
                type=ScheduleType.trainer,
                name=name,
                expression=expression,
                offset=offset,
                fixed_rate=fixed_rate,
                time_arg=time_arg,
                inputs=inputs,
                reader_time_arg=reader_time_arg,
                activate_on_deploy=activate_on_deploy,
                launchplan_kwargs=launchplan_kwargs,
            )
        )

    def schedule_prediction(
        self,
        name: str,
        *,
        expression: Optional[str] = None,
        offset: Optional[str] = None,
        fixed_rate=None,
        time_arg: Optional[str] = None,
        inputs: Optional[dict] = None,
        reader_time_arg: Optional[str] = None,
        activate_on_deploy: bool = True,
        launchplan_kwargs: Optional[dict] = None,
        model_object: Optional[Any] = None,
        model_version: Optional[str] = None,
        model_file: Optional[Union[str, Path]] = None,
    ):
        """Register a prediction schedule; the model artifact to predict
        with is resolved eagerly (reference: model.py:857-934, 915-921)."""
        inputs = dict(inputs or {})
        if any(x is not None for x in (model_object, model_version, model_file)):
            artifact = self.resolve_model_artifact(
                model_object=model_object, model_version=model_version, model_file=model_file
            )
            inputs["model_object"] = artifact.model_object
        self.add_predictor_schedule(
            Schedule(
                type=ScheduleType.predictor,
                name=name,
                expression=expression,
                offset=offset,
                fixed_rate=fixed_rate,
                time_arg=time_arg,
                inputs=inputs,
                reader_time_arg=reader_time_arg,
                activate_on_deploy=activate_on_deploy,
                launchplan_kwargs=launchplan_kwargs,
            )
        )

    def launchplans(self) -> List[LaunchPlan]:
        """Compile all schedules into launch plans."""
        plans = []
        for s in self._training_schedules:
            plans.append(create_scheduled_launchplan(f"{self.name}.train", s.name, s))
        for s in self._prediction_schedules:
            plans.append(create_scheduled_launchplan(f"{self.name}.predict", s.name, s))
        return plans

    # ------------------------------------------------------------------
    # artifact resolution
    # ------------------------------------------------------------------

    def resolve_model_artifact(
        self,
        model_object: Optional[Any] = None,
        model_version: Optional[str] = None,
        model_file: Optional[Union[str, Path]] = None,
        app_version: Optional[str] = None,
        loader_kwargs: Optional[dict] = None,
    ) -> ModelArtifact:
        """Resolve an artifact from exactly one source, falling back to
        ``self.artifact`` (reference: model.py:1521-1566)."""
        provided = [x is not None for x in (model_object, model_version, model_file)]
        if sum(provided) > 1:
            raise ValueError(
                "model_object, model_version and model_file are mutually exclusive"
            )
        if model_object is not None:
            return ModelArtifact(model_object)
        if model_file is not None:
            model_obj = self._loader(model_file, **(loader_kwargs or {}))
            return ModelArtifact(model_obj)
        if model_version is not None:
            return self._backend().fetch_model_artifact(
                self, app_version=app_version, model_version=model_version
            )
        if self.artifact is not None:
            return self.artifact
        raise ModelArtifactNotFound(
            f"model '{self.name}': no artifact resolved — train, load, or pass "
            "model_object/model_version/model_file"
        )

    # ------------------------------------------------------------------
    # remote execution (backend in unionml_amd/remote.py)
    # ------------------------------------------------------------------

    def remote(
        self,
        registry: Optional[str] = None,
        image_name: Optional[str] = None,
        dockerfile: str = "Dockerfile",
        project: Optional[str] = None,
        domain: Optional[str] = None,
        backend_path: Optional[Union[str, Path]] = None,
        **kwargs,
    ):
        """Configure the remote backend (reference: model.py:936-965).

        Without a cluster, the default backend is this build's local
        process cluster rooted at ``backend_path`` (defaults to
        ``~/.unionml_amd/<project>``)."""
        self._remote_config = dict(
            registry=registry,
            image_name=image_name,
            dockerfile=dockerfile,
            project=project or self.name,
            domain=domain or "development",
            backend_path=backend_path,
            **kwargs,
        )
        self.__remote = None

    def _backend(self):
        from unionml_amd.remote import Backend

        if self.__remote is None:
            self.__remote = Backend(**(self._remote_config or {"project": self.name}))
        return self.__remote

    def remote_deploy(
        self, app_version: Optional[str] = None, allow_uncommitted: bool = False, patch: bool = False
    ) -> str:
        """Package and register this app's workflows + launch plans with
        the backend (reference: model.py:983-1083)."""
        return self._backend().deploy(
            self, app_version=app_version, allow_uncommitted=allow_uncommitted, patch=patch
        )

    def remote_train(
        self,
        app_version: Optional[str] = None,
        wait: bool = True,
        *,
        hyperparameters: Optional[dict] = None,
        loader_kwargs: Optional[dict] = None,
        splitter_kwargs: Optional[dict] = None,
        parser_kwargs: Optional[dict] = None,
        trainer_kwargs: Optional[dict] = None,
        **reader_kwargs,
    ):
        """Execute the train workflow on the backend (reference:
        model.py:1085-1158)."""
        execution = self._backend().execute(
            self,
            workflow="train",
            app_version=app_version,
            inputs=dict(
                hyperparameters=hyperparameters,
                loader_kwargs=loader_kwargs,
                splitter_kwargs=splitter_kwargs,
                parser_kwargs=parser_kwargs,
                trainer_kwargs=trainer_kwargs,
                reader_kwargs=reader_kwargs,
            ),
        )
        if wait:
            execution = self.remote_wait(execution)
            self.remote_load(execution)
        return execution

    def remote_predict(
        self,
        app_version: Optional[str] = None,
        model_version: Optional[str] = None,
        wait: bool = True,
        *,
        features: Any = None,
        **reader_kwargs,
    ):
        """Execute the predict workflow on the backend (reference:
        model.py:1160-1226)."""
        backend = self._backend()
        artifact = self.resolve_model_artifact(
            model_version=model_version, app_version=app_version
        )
        if features is not None:
            features = self._dataset.get_features(features)
            execution = backend.execute(
                self,
                workflow="predict_from_features",
                app_version=app_version,
                inputs=dict(model_object=artifact.model_object, features=features),
            )
        else:
            execution = backend.execute(
                self,
                workflow="predict",
                app_version=app_version,
                inputs=dict(model_object=artifact.model_object, reader_kwargs=reader_kwargs),
            )
        if wait:
            execution = self.remote_wait(execution)
            return backend.fetch_output(execution)
        return execution

    def remote_wait(self, execution, timeout: Optional[float] = None):
        return self._backend().wait(execution, timeout=timeout)

    def remote_load(self, execution):
        """Load a remote training execution's outputs into
        ``self.artifact`` (reference: model.py:1247-1270)."""
        outputs = self._backend().fetch_output(execution)
        model_obj, hp, metrics = outputs
        self.artifact = ModelArtifact(model_obj, hp, metrics)
        return self.artifact

    def remote_list_model_versions(self, app_version: Optional[str] = None, limit: int = 10):
        return self._backend().list_model_versions(self, app_version=app_version, limit=limit)

    def remote_list_prediction_ids(self, app_version: Optional[str] = None, limit: int = 10):
        return self._backend().list_prediction_ids(self, app_version=app_version, limit=limit)

    def remote_list_scheduled_training_runs(self, schedule_name: str, limit: int = 10):
        return self._backend().list_scheduled_runs(
            self, schedule_name, kind="training", limit=limit
        )

    def remote_list_scheduled_prediction_runs(self, schedule_name: str, limit: int = 10):
        return self._backend().list_scheduled_runs(
            self, schedule_name, kind="prediction", limit=limit
        )

    def remote_activate_schedules(self, schedule_names: Optional[List[str]] = None):
        return self._backend().set_schedules_active(self, schedule_names, active=True)

    def remote_deactivate_schedules(self, schedule_names: Optional[List[str]] = None):
        return self._backend().set_schedules_active(self, schedule_names, active=False)

    def __repr__(self):
        return f"Model(name={self.name!r}, dataset={self._dataset.name!r})"
