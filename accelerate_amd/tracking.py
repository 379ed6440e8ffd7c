"""Experiment tracking (reference: tracking.py).

`GeneralTracker` protocol + built-ins: TensorBoard, WandB, MLflow and a
zero-dependency JSONL tracker (always available; writes one JSON object per
log call — convenient for benchmark drivers). All tracker calls are
main-process-gated.
"""

import functools
import importlib.util
import json
import os
import time
from typing import Any, Dict, List, Optional, Union

from .logging import get_logger
from .state import PartialState
from .utils.imports import is_mlflow_available, is_tensorboard_available, is_wandb_available

logger = get_logger(__name__)

_available_trackers = []


def on_main_process(function):
    """Run only on the main process (reference: tracking.py:78)."""

    @functools.wraps(function)
    def execute_on_main_process(self, *args, **kwargs):
        if getattr(self, "main_process_only", False):
            return PartialState().on_main_process(function)(self, *args, **kwargs)
        else:
            return function(self, *args, **kwargs)

    return execute_on_main_process


def get_available_trackers():
    return _available_trackers


class GeneralTracker:
    """Tracker protocol: name/requires_logging_directory/tracker properties;
    start, store_init_configuration, log, finish (reference: tracking.py:102)."""

    main_process_only = True
    name = "base"
    requires_logging_directory = False

    def __init__(self, _blank=False):
        if not _blank:
            err = ""
            if not hasattr(self, "name"):
                err += "`name`"
            if not hasattr(self, "requires_logging_directory"):
                if len(err) > 0:
                    err += ", "
                err += "`requires_logging_directory`"
            if "tracker" not in dir(self):
                if len(err) > 0:
                    err += ", "
                err += "`tracker`"
            if len(err) > 0:
                raise NotImplementedError(
                    f"The implementation for this tracker class is missing the following required attributes: {err}"
                )

    @property
    def tracker(self):
        return None

    def start(self):
        pass

    def store_init_configuration(self, values: dict):
        pass

    def log(self, values: dict, step: Optional[int] = None, **kwargs):
        pass

    def finish(self):
        pass


class JSONLTracker(GeneralTracker):
    """Dependency-free tracker writing JSON lines under the logging dir."""

    name = "jsonl"
    requires_logging_directory = True
    main_process_only = True

    @on_main_process
    def __init__(self, run_name: str, logging_dir: Union[str, os.PathLike] = ".", **kwargs):
        super().__init__()
        self.run_name = run_name
        os.makedirs(os.path.join(logging_dir, run_name), exist_ok=True)
        self.path = os.path.join(logging_dir, run_name, "metrics.jsonl")
        self._fh = open(self.path, "a")

    @property
    def tracker(self):
        return self._fh

    @on_main_process
    def store_init_configuration(self, values: dict):
        self._fh.write(json.dumps({"_config": values, "_time": time.time()}, default=str) + "\n")
        self._fh.flush()

    @on_main_process
    def log(self, values: dict, step: Optional[int] = None, **kwargs):
        payload = dict(values)
        if step is not None:
            payload["_step"] = step
        payload["_time"] = time.time()
        self._fh.write(json.dumps(payload, default=str) + "\n")
        self._fh.flush()

    @on_main_process
    def finish(self):
        self._fh.close()


class TensorBoardTracker(GeneralTracker):
    """(reference: tracking.py:179)"""

    name = "tensorboard"
    requires_logging_directory = True

    @on_main_process
    def __init__(self, run_name: str, logging_dir: Union[str, os.PathLike], **kwargs):
        super().__init__()
        try:
            from torch.utils import tensorboard
        except ImportError:
            import tensorboardX as tensorboard
        self.run_name = run_name
        self.logging_dir = os.path.join(logging_dir, run_name)
        self.writer = tensorboard.SummaryWriter(self.logging_dir, **kwargs)
        logger.debug(f"Initialized TensorBoard project {self.run_name} logging to {self.logging_dir}")

    @property
    def tracker(self):
        return self.writer

    @on_main_process
    def store_init_configuration(self, values: dict):
        self.writer.add_hparams(values, metric_dict={})
        self.writer.flush()
        import yaml

        with open(os.path.join(self.logging_dir, "hparams.yml"), "w") as outfile:
            try:
                yaml.dump(values, outfile)
            except yaml.representer.RepresenterError:
                logger.error("Serialization to store hyperparameters failed")
                raise

    @on_main_process
    def log(self, values: dict, step: Optional[int] = None, **kwargs):
        values = listify_values(values)
        for k, v in values.items():
            if isinstance(v, (int, float)):
                self.writer.add_scalar(k, v, global_step=step, **kwargs)
            elif isinstance(v, str):
                self.writer.add_text(k, v, global_step=step, **kwargs)
            elif isinstance(v, dict):
                self.writer.add_scalars(k, v, global_step=step, **kwargs)
        self.writer.flush()

    @on_main_process
    def finish(self):
        self.writer.close()


class WandBTracker(GeneralTracker):
    """(reference: tracking.py:294)"""

    name = "wandb"
    requires_logging_directory = False
    main_process_only = True

    @on_main_process
    def __init__(self, run_name: str, **kwargs):
        super().__init__()
        import wandb

        self.run_name = run_name
        self.run = wandb.init(project=self.run_name, **kwargs)

    @property
    def tracker(self):
        return self.run

    @on_main_process
    def store_init_configuration(self, values: dict):
        import wandb

        wandb.config.update(values, allow_val_change=True)

    @on_main_process
    def log(self, values: dict, step: Optional[int] = None, **kwargs):
        self.run.log(values, step=step, **kwargs)

    @on_main_process
    def finish(self):
        self.run.finish()


class MLflowTracker(GeneralTracker):
    """(reference: tracking.py:693)"""

    name = "mlflow"
    requires_logging_directory = False

    @on_main_process
    def __init__(self, experiment_name: str = None, logging_dir=None, run_id=None, tags=None, nested_run=False,
                 run_name=None, description=None):
        super().__init__()
        import mlflow

        experiment_name = os.environ.get("MLFLOW_EXPERIMENT_NAME", experiment_name)
        run_id = os.environ.get("MLFLOW_RUN_ID", run_id)
        tags = os.environ.get("MLFLOW_TAGS", tags)
        if isinstance(tags, str):
            tags = json.loads(tags)
        nested_run = os.environ.get("MLFLOW_NESTED_RUN", nested_run)
        exps = mlflow.search_experiments(filter_string=f"name = '{experiment_name}'")
        if len(exps) > 0:
            experiment_id = exps[0].experiment_id
        else:
            experiment_id = mlflow.create_experiment(name=experiment_name, artifact_location=logging_dir, tags=tags)
        self.active_run = mlflow.start_run(
            run_id=run_id, experiment_id=experiment_id, run_name=run_name, nested=nested_run,
            tags=tags, description=description,
        )

    @property
    def tracker(self):
        return self.active_run

    @on_main_process
    def store_init_configuration(self, values: dict):
        import mlflow

        for name, value in list(values.items()):
            if len(str(value)) > mlflow.utils.validation.MAX_PARAM_VAL_LENGTH:
                logger.warning_once(f'Trainer is attempting to log a value of "{value}" for key "{name}" as a parameter.')
                del values[name]
        values_list = list(values.items())
        for i in range(0, len(values_list), mlflow.utils.validation.MAX_PARAMS_TAGS_PER_BATCH):
            mlflow.log_params(dict(values_list[i : i + mlflow.utils.validation.MAX_PARAMS_TAGS_PER_BATCH]))

    @on_main_process
    def log(self, values: dict, step: Optional[int] = None, **kwargs):
        import mlflow

        metrics = {}
        for k, v in values.items():
            if isinstance(v, (int, float)):
                metrics[k] = v
        mlflow.log_metrics(metrics, step=step)

    @on_main_process
    def finish(self):
        import mlflow

        mlflow.end_run()


class CometMLTracker(GeneralTracker):
    """Comet ML (reference: tracking.py:496). Lazy import; the comet_ml
    package is not bundled — constructing without it raises ImportError."""

    name = "comet_ml"
    requires_logging_directory = False

    @on_main_process
    def __init__(self, run_name: str, **kwargs):
        super().__init__()
        from comet_ml import ExperimentConfig, start

        self.run_name = run_name
        self.writer = start(experiment_config=ExperimentConfig(name=run_name, **kwargs))

    @property
    def tracker(self):
        return self.writer

    @on_main_process
    def store_init_configuration(self, values: dict):
        self.writer.log_parameters(values)

    @on_main_process
    def log(self, values: dict, step: Optional[int] = None, **kwargs):
        if step is not None:
            self.writer.set_step(step)
        for k, v in listify_values(values).items():
            if isinstance(v, (int, float)):
                self.writer.log_metric(k, v, step=step, **kwargs)
            elif isinstance(v, str):
                self.writer.log_other(k, v, **kwargs)
            elif isinstance(v, dict):
                self.writer.log_metrics(v, step=step, **kwargs)

    @on_main_process
    def finish(self):
        self.writer.end()


class AimTracker(GeneralTracker):
    """Aim (reference: tracking.py:590)."""

    name = "aim"
    requires_logging_directory = True

    @on_main_process
    def __init__(self, run_name: str, logging_dir=".", **kwargs):
        super().__init__()
        from aim import Run

        self.writer = Run(repo=str(logging_dir), **kwargs)
        self.writer.name = run_name

    @property
    def tracker(self):
        return self.writer

    @on_main_process
    def store_init_configuration(self, values: dict):
        self.writer["hparams"] = values

    @on_main_process
    def log(self, values: dict, step: Optional[int] = None, **kwargs):
        for k, v in listify_values(values).items():
            self.writer.track(v, name=k, step=step, **kwargs)

    @on_main_process
    def finish(self):
        self.writer.close()


class ClearMLTracker(GeneralTracker):
    """ClearML (reference: tracking.py:902)."""

    name = "clearml"
    requires_logging_directory = False

    @on_main_process
    def __init__(self, run_name: str = None, **kwargs):
        super().__init__()
        from clearml import Task

        current = Task.current_task()
        self._initialized_externally = current is not None
        self.task = current or Task.init(
            project_name=os.environ.get("CLEARML_PROJECT", run_name),
            task_name=os.environ.get("CLEARML_TASK", run_name),
            **kwargs,
        )

    @property
    def tracker(self):
        return self.task

    @on_main_process
    def store_init_configuration(self, values: dict):
        self.task.connect_configuration(values)

    @on_main_process
    def log(self, values: dict, step: Optional[int] = None, **kwargs):
        for k, v in listify_values(values).items():
            if isinstance(v, (int, float)) and step is None:
                self.task.get_logger().report_single_value(name=k, value=v, **kwargs)
            elif isinstance(v, (int, float)):
                title, _, series = k.rpartition("/") if "/" in k else ("train", "", k)
                self.task.get_logger().report_scalar(
                    title=title or "train", series=series, value=v, iteration=step, **kwargs
                )

    @on_main_process
    def finish(self):
        if not self._initialized_externally:
            self.task.close()


class DVCLiveTracker(GeneralTracker):
    """DVCLive (reference: tracking.py:1060)."""

    name = "dvclive"
    requires_logging_directory = False

    @on_main_process
    def __init__(self, run_name: str = None, live=None, **kwargs):
        super().__init__()
        from dvclive import Live

        self.live = live if live is not None else Live(**kwargs)

    @property
    def tracker(self):
        return self.live

    @on_main_process
    def store_init_configuration(self, values: dict):
        self.live.log_params(values)

    @on_main_process
    def log(self, values: dict, step: Optional[int] = None, **kwargs):
        if step is not None:
            self.live.step = step
        for k, v in listify_values(values).items():
            if isinstance(v, (int, float)):
                self.live.log_metric(k, v, **kwargs)
        self.live.next_step()

    @on_main_process
    def finish(self):
        self.live.end()


class SwanLabTracker(GeneralTracker):
    """SwanLab (reference: tracking.py:1148)."""

    name = "swanlab"
    requires_logging_directory = False

    @on_main_process
    def __init__(self, run_name: str, **kwargs):
        super().__init__()
        import swanlab

        self.run = swanlab.init(project=run_name, **kwargs)

    @property
    def tracker(self):
        return self.run

    @on_main_process
    def store_init_configuration(self, values: dict):
        import swanlab

        swanlab.config.update(values, allow_val_change=True)

    @on_main_process
    def log(self, values: dict, step: Optional[int] = None, **kwargs):
        self.run.log(listify_values(values), step=step)

    @on_main_process
    def finish(self):
        self.run.finish()


class TrackioTracker(GeneralTracker):
    """Trackio (reference: tracking.py:419) — wandb-compatible local API."""

    name = "trackio"
    requires_logging_directory = False

    @on_main_process
    def __init__(self, run_name: str, **kwargs):
        super().__init__()
        import trackio

        self.run = trackio.init(project=run_name, **kwargs)

    @property
    def tracker(self):
        return self.run

    @on_main_process
    def store_init_configuration(self, values: dict):
        import trackio

        trackio.config.update(values, allow_val_change=True)

    @on_main_process
    def log(self, values: dict, step: Optional[int] = None, **kwargs):
        self.run.log(listify_values(values))

    @on_main_process
    def finish(self):
        self.run.finish()


def listify_values(values: dict) -> dict:
    import torch

    out = {}
    for k, v in values.items():
        if isinstance(v, torch.Tensor):
            v = v.item() if v.numel() == 1 else v.tolist()
        out[k] = v
    return out


LOGGER_TYPE_TO_CLASS = {
    "tensorboard": TensorBoardTracker,
    "wandb": WandBTracker,
    "mlflow": MLflowTracker,
    "jsonl": JSONLTracker,
    "comet_ml": CometMLTracker,
    "aim": AimTracker,
    "clearml": ClearMLTracker,
    "dvclive": DVCLiveTracker,
    "swanlab": SwanLabTracker,
    "trackio": TrackioTracker,
}


def filter_trackers(log_with, logging_dir=None):
    """Resolve 'all'/names/instances into usable tracker classes
    (reference: tracking.py:1311)."""
    loggers = []
    if log_with is not None:
        if not isinstance(log_with, (list, tuple)):
            log_with = [log_with]
        if "all" in [str(l) for l in log_with]:
            candidates = []
            if is_tensorboard_available():
                candidates.append("tensorboard")
            if is_wandb_available():
                candidates.append("wandb")
            for name, mod in (("comet_ml", "comet_ml"), ("aim", "aim"), ("clearml", "clearml"),
                              ("dvclive", "dvclive"), ("swanlab", "swanlab"), ("trackio", "trackio")):
                if importlib.util.find_spec(mod) is not None:
                    candidates.append(name)
            candidates.append("jsonl")
            log_with = candidates
        for log_type in log_with:
            if issubclass(type(log_type), GeneralTracker):
                loggers.append(log_type)
                continue
            log_type = str(log_type)
            if log_type not in LOGGER_TYPE_TO_CLASS:
                raise ValueError(f"Unknown tracker {log_type}; choose from {list(LOGGER_TYPE_TO_CLASS)}")
            tracker_cls = LOGGER_TYPE_TO_CLASS[log_type]
            if tracker_cls.requires_logging_directory and logging_dir is None:
                raise ValueError(f"Logging with `{log_type}` requires a `logging_dir` to be passed in.")
            loggers.append(tracker_cls)
    return loggers
