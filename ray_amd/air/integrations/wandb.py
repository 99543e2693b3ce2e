"""Weights & Biases integration (reference:
python/ray/air/integrations/wandb.py — WandbLoggerCallback logs every
trial's results to a wandb run; setup_wandb initializes a run inside a
trainable)."""
from __future__ import annotations

import os
from typing import Dict, Optional

from ray_amd.tune.impl import Callback


def _import_wandb():
    try:
        import wandb

        return wandb
    except ImportError as e:
        raise ImportError(
            "wandb integration requires the `wandb` package") from e


class WandbLoggerCallback(Callback):
    """One wandb run per trial; every reported result becomes a
    wandb.log row tagged with the trial's config."""

    def __init__(self, project: Optional[str] = None,
                 group: Optional[str] = None, api_key: Optional[str] = None,
                 excludes: Optional[list] = None, log_config: bool = True,
                 **kwargs):
        self.project = project
        self.group = group
        self.api_key = api_key
        self.excludes = set(excludes or [])
        self.log_config = log_config
        self.kwargs = kwargs
        self._runs: Dict[str, object] = {}
        self._wandb = None

    def setup(self, **info):
        self._wandb = _import_wandb()
        if self.api_key:
            os.environ.setdefault("WANDB_API_KEY", self.api_key)

    def on_trial_start(self, iteration, trials, trial, **info):
        if trial["name"] in self._runs:
            return
        self._runs[trial["name"]] = self._wandb.init(
            project=self.project, group=self.group, name=trial["name"],
            config=dict(trial["config"]) if self.log_config else None,
            reinit=True, **self.kwargs)

    def on_trial_result(self, iteration, trials, trial, result, **info):
        run = self._runs.get(trial["name"])
        if run is None:
            return
        row = {k: v for k, v in result.items()
               if k not in self.excludes and not k.startswith("config/")}
        run.log(row)

    def on_trial_complete(self, iteration, trials, trial, **info):
        run = self._runs.pop(trial["name"], None)
        if run is not None:
            run.finish()

    on_trial_error = on_trial_complete

    def on_experiment_end(self, trials, **info):
        for run in self._runs.values():
            run.finish()
        self._runs.clear()


def setup_wandb(config: Optional[dict] = None, *, project=None,
                trial_id=None, trial_name=None, **kwargs):
    """Initialize a wandb run from inside a trainable (reference:
    air/integrations/wandb.py setup_wandb). Rank-nonzero train workers
    get a disabled run."""
    wandb = _import_wandb()
    try:
        from ray_amd import train

        ctx = train.get_context()
        if ctx.get_world_rank() not in (None, 0):
            kwargs.setdefault("mode", "disabled")
        trial_id = trial_id or ctx.get_trial_id()
        trial_name = trial_name or ctx.get_trial_name()
    except Exception:
        pass
    return wandb.init(project=project, id=trial_id, name=trial_name,
                      config=config, **kwargs)
