"""ray_amd.air — shared Train/Tune plumbing (reference: python/ray/air).
The canonical configs live in ray_amd.train; re-exported here for API
parity with `from ray.air import ScalingConfig, RunConfig`. """
from ..train.checkpoint import Checkpoint  # noqa: F401
from ..train.config import (  # noqa: F401
    CheckpointConfig,
    FailureConfig,
    Result,
    RunConfig,
    ScalingConfig,
)
from ..train.session import get_checkpoint, get_context, report  # noqa: F401


def session():  # legacy alias namespace
    from ..train import session as s

    return s
