"""Experiment tracker integrations + tune Callback plumbing
(reference: air/integrations/{wandb,mlflow}.py, tune/callback.py).
Tracker libraries are optional in this image; fakes cover OUR glue."""
import sys
import types

import pytest

import ray_amd as ray  # noqa: F401
from ray_amd import tune
from ray_amd.tune import TuneConfig, Tuner


def _trainable(config):
    for i in range(2):
        tune.report({"score": config["x"] * (i + 1)})


def test_callbacks_fire_through_tuner(ray_start_regular, tmp_path):
    from ray_amd.train import RunConfig
    from ray_amd.tune.impl import Callback

    events = []

    class Recorder(Callback):
        def setup(self, **info):
            events.append("setup")

        def on_trial_start(self, iteration, trials, trial, **info):
            events.append(("start", trial["name"]))

        def on_trial_result(self, iteration, trials, trial, result, **info):
            events.append(("result", trial["name"], result["score"]))

        def on_trial_complete(self, iteration, trials, trial, **info):
            events.append(("complete", trial["name"]))

        def on_experiment_end(self, trials, **info):
            events.append("end")

    Tuner(
        _trainable,
        param_space={"x": tune.grid_search([1.0, 2.0])},
        tune_config=TuneConfig(metric="score", mode="max"),
        run_config=RunConfig(name="cb", storage_path=str(tmp_path),
                             callbacks=[Recorder()]),
    ).fit()

    assert events[0] == "setup" and events[-1] == "end"
    starts = {e[1] for e in events if e[0] == "start"}
    completes = {e[1] for e in events if e[0] == "complete"}
    assert starts == completes and len(starts) == 2
    assert any(e[0] == "result" for e in events)


class _FakeWandbRun:
    def __init__(self):
        self.rows = []
        self.finished = False

    def log(self, row):
        self.rows.append(row)

    def finish(self):
        self.finished = True


def test_wandb_logger_callback(monkeypatch):
    runs = []
    fake = types.ModuleType("wandb")

    def init(**kw):
        r = _FakeWandbRun()
        r.kw = kw
        runs.append(r)
        return r

    fake.init = init
    monkeypatch.setitem(sys.modules, "wandb", fake)

    from ray_amd.air.integrations.wandb import WandbLoggerCallback

    cb = WandbLoggerCallback(project="proj", excludes=["noise"])
    cb.setup()
    trial = {"name": "trial_00000", "config": {"lr": 0.1}}
    cb.on_trial_start(0, [], trial)
    cb.on_trial_result(1, [], trial, {"score": 5, "noise": 1,
                                      "config/lr": 0.1})
    cb.on_trial_complete(2, [], trial)
    assert len(runs) == 1
    assert runs[0].kw["project"] == "proj"
    assert runs[0].kw["config"] == {"lr": 0.1}
    assert runs[0].rows == [{"score": 5}]
    assert runs[0].finished


def test_mlflow_logger_callback(monkeypatch):
    logged = {"params": [], "metrics": [], "ended": []}
    fake = types.ModuleType("mlflow")

    class _Run:
        class info:
            run_id = "r1"

    fake.set_tracking_uri = lambda uri: logged.setdefault("uri", uri)
    fake.set_experiment = lambda name: logged.setdefault("exp", name)
    fake.start_run = lambda **kw: _Run()
    fake.log_param = lambda k, v, run_id=None: logged["params"].append(
        (k, v))
    fake.log_metric = lambda k, v, step=None, run_id=None: (
        logged["metrics"].append((k, v, step)))
    fake.end_run = lambda run_id=None: logged["ended"].append(run_id)
    monkeypatch.setitem(sys.modules, "mlflow", fake)

    from ray_amd.air.integrations.mlflow import MLflowLoggerCallback

    cb = MLflowLoggerCallback(tracking_uri="file:/tmp/mlruns",
                              experiment_name="e1")
    cb.setup()
    trial = {"name": "trial_00000", "config": {"lr": 0.5}}
    cb.on_trial_start(0, [], trial)
    cb.on_trial_result(1, [], trial,
                       {"score": 2.0, "training_iteration": 3})
    cb.on_trial_complete(2, [], trial)
    assert logged["uri"] == "file:/tmp/mlruns"
    assert ("lr", 0.5) in logged["params"]
    assert ("score", 2.0, 3) in logged["metrics"]
    assert logged["ended"] == ["r1"]


def test_missing_trackers_raise_lazily():
    if "wandb" in sys.modules or "mlflow" in sys.modules:
        pytest.skip("tracker installed/faked")
    from ray_amd.air.integrations.mlflow import MLflowLoggerCallback
    from ray_amd.air.integrations.wandb import WandbLoggerCallback

    with pytest.raises(ImportError, match="wandb"):
        WandbLoggerCallback().setup()
    with pytest.raises(ImportError, match="mlflow"):
        MLflowLoggerCallback().setup()
