"""MLflow integration (reference: python/ray/air/integrations/
mlflow.py — MLflowLoggerCallback mirrors trial results into MLflow
runs; setup_mlflow configures the client inside a trainable)."""
from __future__ import annotations

from typing import Dict, Optional

from ray_amd.tune.impl import Callback


def _import_mlflow():
    try:
        import mlflow

        return mlflow
    except ImportError as e:
        raise ImportError(
            "mlflow integration requires the `mlflow` package") from e


class MLflowLoggerCallback(Callback):
    def __init__(self, tracking_uri: Optional[str] = None,
                 experiment_name: Optional[str] = None,
                 tags: Optional[dict] = None,
                 save_artifact: bool = False):
        self.tracking_uri = tracking_uri
        self.experiment_name = experiment_name
        self.tags = tags or {}
        self.save_artifact = save_artifact
        self._mlflow = None
        self._runs: Dict[str, object] = {}

    def setup(self, **info):
        self._mlflow = _import_mlflow()
        if self.tracking_uri:
            self._mlflow.set_tracking_uri(self.tracking_uri)
        if self.experiment_name:
            self._mlflow.set_experiment(self.experiment_name)

    def on_trial_start(self, iteration, trials, trial, **info):
        if trial["name"] in self._runs:
            return
        run = self._mlflow.start_run(run_name=trial["name"], nested=True,
                                     tags=self.tags)
        self._runs[trial["name"]] = run
        for k, v in trial["config"].items():
            self._mlflow.log_param(k, v, run_id=run.info.run_id)

    def on_trial_result(self, iteration, trials, trial, result, **info):
        run = self._runs.get(trial["name"])
        if run is None:
            return
        step = int(result.get("training_iteration", iteration) or 0)
        for k, v in result.items():
            if isinstance(v, (int, float)) and not k.startswith("config/"):
                self._mlflow.log_metric(k, float(v), step=step,
                                        run_id=run.info.run_id)

    def on_trial_complete(self, iteration, trials, trial, **info):
        run = self._runs.pop(trial["name"], None)
        if run is not None:
            self._mlflow.end_run(run_id=run.info.run_id)

    on_trial_error = on_trial_complete

    def on_experiment_end(self, trials, **info):
        for run in self._runs.values():
            self._mlflow.end_run(run_id=run.info.run_id)
        self._runs.clear()


def setup_mlflow(config: Optional[dict] = None, *, tracking_uri=None,
                 experiment_name=None, run_name=None, **kwargs):
    """Configure mlflow inside a trainable; rank-0 only on train
    workers (reference parity)."""
    mlflow = _import_mlflow()
    try:
        from ray_amd import train

        ctx = train.get_context()
        if ctx.get_world_rank() not in (None, 0):
            return mlflow  # non-zero ranks: no active run
        run_name = run_name or ctx.get_trial_name()
    except Exception:
        pass
    if tracking_uri:
        mlflow.set_tracking_uri(tracking_uri)
    if experiment_name:
        mlflow.set_experiment(experiment_name)
    mlflow.start_run(run_name=run_name, **kwargs)
    if config:
        for k, v in config.items():
            mlflow.log_param(k, v)
    return mlflow
