"""Tracker integration registry (reference tracking.py built-ins)."""

import pytest


def test_all_reference_tracker_integrations_registered():
    """Parity with the reference's 9 built-ins (reference tracking.py:179-1244)
    plus our JSONL tracker; missing packages raise ImportError at construction
    (lazy imports), never at module import."""
    from accelerate_amd.tracking import LOGGER_TYPE_TO_CLASS

    expected = {
        "tensorboard", "wandb", "mlflow", "jsonl",
        "comet_ml", "aim", "clearml", "dvclive", "swanlab", "trackio",
    }
    assert expected == set(LOGGER_TYPE_TO_CLASS)
    for name, cls in LOGGER_TYPE_TO_CLASS.items():
        assert cls.name == name
        assert isinstance(cls.requires_logging_directory, bool)


def test_unavailable_tracker_raises_at_construction():
    import pytest as _pytest

    from accelerate_amd.tracking import AimTracker

    with _pytest.raises(ImportError):
        AimTracker("run", logging_dir=".")


# ---------------------------------------------------------------------------
# Behavioral tests for every third-party integration via injected fake
# modules (the packages aren't installed offline): each tracker must drive
# the documented API through its full lifecycle — init, config, log, finish.
# ---------------------------------------------------------------------------

import sys
import types


class Recorder:
    def __init__(self):
        self.calls = []

    def rec(self, name):
        def f(*a, **k):
            self.calls.append((name, a, k))
            return self
        return f


def _fake_module(name, **attrs):
    mod = types.ModuleType(name)
    for k, v in attrs.items():
        setattr(mod, k, v)
    return mod


def _lifecycle(tracker):
    tracker.store_init_configuration({"lr": 0.1, "model": "bert"})
    tracker.log({"loss": 1.5, "note": "s"}, step=3)
    tracker.finish()
    assert tracker.tracker is not None


def test_wandb_tracker_lifecycle(monkeypatch):
    from accelerate_amd.tracking import WandBTracker

    r = Recorder()
    run = types.SimpleNamespace(log=r.rec("run.log"), finish=r.rec("run.finish"))
    config = types.SimpleNamespace(update=r.rec("config.update"))
    monkeypatch.setitem(sys.modules, "wandb", _fake_module("wandb", init=lambda **k: run, config=config))
    t = WandBTracker("proj")
    _lifecycle(t)
    names = [c[0] for c in r.calls]
    assert names == ["config.update", "run.log", "run.finish"]
    assert r.calls[1][2].get("step") == 3


def test_mlflow_tracker_lifecycle(monkeypatch):
    from accelerate_amd.tracking import MLflowTracker

    r = Recorder()
    validation = types.SimpleNamespace(MAX_PARAM_VAL_LENGTH=250, MAX_PARAMS_TAGS_PER_BATCH=100)
    mod = _fake_module(
        "mlflow",
        search_experiments=lambda filter_string="": [],
        create_experiment=r.rec("create_experiment"),
        start_run=r.rec("start_run"),
        log_params=r.rec("log_params"),
        log_metrics=r.rec("log_metrics"),
        end_run=r.rec("end_run"),
        utils=types.SimpleNamespace(validation=validation),
    )
    monkeypatch.setitem(sys.modules, "mlflow", mod)
    t = MLflowTracker(experiment_name="exp")
    _lifecycle(t)
    names = [c[0] for c in r.calls]
    assert names == ["create_experiment", "start_run", "log_params", "log_metrics", "end_run"]
    # non-numeric values are filtered from metrics
    assert r.calls[3][1][0] == {"loss": 1.5}


def test_comet_tracker_lifecycle(monkeypatch):
    from accelerate_amd.tracking import CometMLTracker

    r = Recorder()
    writer = types.SimpleNamespace(
        log_parameters=r.rec("log_parameters"), set_step=r.rec("set_step"),
        log_metric=r.rec("log_metric"), log_other=r.rec("log_other"),
        log_metrics=r.rec("log_metrics"), end=r.rec("end"),
    )
    mod = _fake_module("comet_ml", ExperimentConfig=lambda **k: k, start=lambda experiment_config=None: writer)
    monkeypatch.setitem(sys.modules, "comet_ml", mod)
    t = CometMLTracker("run")
    _lifecycle(t)
    names = [c[0] for c in r.calls]
    assert names[0] == "log_parameters" and names[1] == "set_step" and "log_metric" in names and names[-1] == "end"


def test_aim_tracker_lifecycle(monkeypatch, tmp_path):
    from accelerate_amd.tracking import AimTracker

    r = Recorder()

    class FakeRun:
        def __init__(self, repo=None, **k):
            self.repo = repo
            self.store = {}
        def __setitem__(self, k, v):
            self.store[k] = v
        track = property(lambda self: r.rec("track"))
        close = property(lambda self: r.rec("close"))

    monkeypatch.setitem(sys.modules, "aim", _fake_module("aim", Run=FakeRun))
    t = AimTracker("run", logging_dir=str(tmp_path))
    _lifecycle(t)
    assert t.writer.store["hparams"] == {"lr": 0.1, "model": "bert"}
    assert [c[0] for c in r.calls] == ["track", "track", "close"]


def test_clearml_tracker_lifecycle(monkeypatch):
    from accelerate_amd.tracking import ClearMLTracker

    r = Recorder()
    logger_obj = types.SimpleNamespace(
        report_single_value=r.rec("report_single_value"), report_scalar=r.rec("report_scalar")
    )
    task = types.SimpleNamespace(
        connect_configuration=r.rec("connect_configuration"),
        get_logger=lambda: logger_obj,
        close=r.rec("close"),
    )
    Task = types.SimpleNamespace(current_task=lambda: None, init=lambda **k: task)
    monkeypatch.setitem(sys.modules, "clearml", _fake_module("clearml", Task=Task))
    t = ClearMLTracker("run")
    _lifecycle(t)
    names = [c[0] for c in r.calls]
    assert names == ["connect_configuration", "report_scalar", "close"]


def test_dvclive_tracker_lifecycle(monkeypatch):
    from accelerate_amd.tracking import DVCLiveTracker

    r = Recorder()

    class FakeLive:
        def __init__(self, **k):
            self.step = 0
        log_params = property(lambda self: r.rec("log_params"))
        log_metric = property(lambda self: r.rec("log_metric"))
        next_step = property(lambda self: r.rec("next_step"))
        end = property(lambda self: r.rec("end"))

    monkeypatch.setitem(sys.modules, "dvclive", _fake_module("dvclive", Live=FakeLive))
    t = DVCLiveTracker("run")
    _lifecycle(t)
    assert t.live.step == 3  # log(step=3) sets the live step
    assert [c[0] for c in r.calls] == ["log_params", "log_metric", "next_step", "end"]


def test_swanlab_tracker_lifecycle(monkeypatch):
    from accelerate_amd.tracking import SwanLabTracker

    r = Recorder()
    run = types.SimpleNamespace(log=r.rec("run.log"), finish=r.rec("run.finish"))
    config = types.SimpleNamespace(update=r.rec("config.update"))
    monkeypatch.setitem(sys.modules, "swanlab", _fake_module("swanlab", init=lambda **k: run, config=config))
    t = SwanLabTracker("proj")
    _lifecycle(t)
    assert [c[0] for c in r.calls] == ["config.update", "run.log", "run.finish"]


def test_trackio_tracker_lifecycle(monkeypatch):
    from accelerate_amd.tracking import TrackioTracker

    r = Recorder()
    run = types.SimpleNamespace(
        log=r.rec("run.log"), finish=r.rec("run.finish"),
        config=types.SimpleNamespace(update=r.rec("config.update")),
    )
    config = types.SimpleNamespace(update=r.rec("config.update"))
    monkeypatch.setitem(sys.modules, "trackio", _fake_module("trackio", init=lambda **k: run, config=config))
    t = TrackioTracker("proj")
    _lifecycle(t)
    names = [c[0] for c in r.calls]
    assert "run.log" in names and names[-1] == "run.finish"
