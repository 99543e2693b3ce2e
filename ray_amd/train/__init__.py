"""ray_amd.train — distributed training (reference: python/ray/train/).

Public API parity: ScalingConfig/RunConfig/CheckpointConfig/FailureConfig
(air/config.py), Checkpoint (train/_checkpoint.py:56, dir-based, same
layout), session report/get_context/get_checkpoint
(train/_internal/session.py), TorchTrainer (train/torch/torch_trainer.py)
over an actor WorkerGroup with RCCL process groups
(train/torch/config.py:96 _setup_torch_process_group).
"""
from .checkpoint import Checkpoint  # noqa: F401
from .config import (  # noqa: F401
    CheckpointConfig,
    FailureConfig,
    Result,
    RunConfig,
    ScalingConfig,
)
from .session import (  # noqa: F401
    get_checkpoint,
    get_context,
    get_dataset_shard,
    report,
)
from .trainer import DataParallelTrainer  # noqa: F401


def __getattr__(name):
    if name == "torch":
        from . import torch as _t

        return _t
    raise AttributeError(name)


try:  # usage tagging (local-only; util/usage_stats.py)
    from ray_amd.util.usage_stats import record_library_usage

    record_library_usage("train")
except Exception:  # pragma: no cover
    pass
