"""Train/AIR configs (reference: python/ray/air/config.py,
ray/train/v2/api/config.py)."""
from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Any, Dict, Optional

from .checkpoint import Checkpoint


@dataclass
class ScalingConfig:
    num_workers: int = 1
    use_gpu: bool = False
    resources_per_worker: Optional[Dict[str, float]] = None
    placement_strategy: str = "PACK"
    trainer_resources: Optional[Dict[str, float]] = None
    # elastic training (reference: v2 scaling_policy/elastic.py:27):
    # on failure-retry the group restarts with as many workers as the
    # cluster can currently place, down to min_workers.
    elastic: bool = False
    min_workers: int = 1

    def worker_resources(self) -> Dict[str, float]:
        res = dict(self.resources_per_worker or {})
        res.setdefault("CPU", 1)
        if self.use_gpu:
            res.setdefault("GPU", 1)
        return res


@dataclass
class CheckpointConfig:
    num_to_keep: Optional[int] = None
    checkpoint_score_attribute: Optional[str] = None
    checkpoint_score_order: str = "max"
    checkpoint_frequency: int = 0
    checkpoint_at_end: Optional[bool] = None


@dataclass
class FailureConfig:
    max_failures: int = 0


@dataclass
class RunConfig:
    name: Optional[str] = None
    storage_path: Optional[str] = None
    failure_config: Optional[FailureConfig] = None
    checkpoint_config: Optional[CheckpointConfig] = None
    verbose: int = 1
    log_to_file: bool = False
    # experiment callbacks (reference: air.RunConfig.callbacks —
    # tune.Callback hooks; wandb/mlflow loggers plug in here)
    callbacks: Optional[list] = None

    def resolved_storage_path(self) -> str:
        base = self.storage_path or os.path.expanduser("~/ray_amd_results")
        name = self.name or "train_run"
        return os.path.join(base, name)


@dataclass
class Result:
    metrics: Optional[Dict[str, Any]] = None
    checkpoint: Optional[Checkpoint] = None
    path: Optional[str] = None
    error: Optional[BaseException] = None
    metrics_dataframe: Any = None
    best_checkpoints: list = field(default_factory=list)

    @property
    def config(self):
        return None
