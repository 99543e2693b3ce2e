"""Lazy construction of the Lightning integration classes.

Reference surface: python/ray/train/lightning/_lightning_utils.py —
RayDDPStrategy (DDP wired to the session's process group),
RayLightningEnvironment (ClusterEnvironment answering rank/world-size
from the ray_amd train session), RayTrainReportCallback (per-epoch
metrics + checkpoint into ray_amd.train.report), prepare_trainer
(validates the Trainer was built with the Ray pieces).
"""
from __future__ import annotations

import os
import tempfile
from types import SimpleNamespace

_cache = None


def _import_pl():
    try:
        import pytorch_lightning as pl  # noqa
        return pl
    except ImportError:
        pass
    try:
        from lightning import pytorch as pl  # noqa
        return pl
    except ImportError:
        raise ImportError(
            "ray_amd.train.lightning requires `pytorch_lightning` or "
            "`lightning` to be installed"
        )


def build():
    global _cache
    if _cache is not None:
        return _cache
    pl = _import_pl()
    ClusterEnvironment = pl.plugins.environments.ClusterEnvironment

    from ray_amd import train

    class RayLightningEnvironment(ClusterEnvironment):
        """Answers Lightning's topology questions from the ray_amd
        train session (reference: _lightning_utils.py
        RayLightningEnvironment)."""

        def __init__(self):
            super().__init__()
            self._ctx = train.get_context()

        def world_size(self) -> int:
            return self._ctx.get_world_size()

        def global_rank(self) -> int:
            return self._ctx.get_world_rank()

        def local_rank(self) -> int:
            return self._ctx.get_local_rank()

        def node_rank(self) -> int:
            return self._ctx.get_node_rank()

        @property
        def creates_processes_externally(self) -> bool:
            return True  # ray_amd TorchTrainer launched the workers

        @property
        def main_address(self) -> str:
            return os.environ.get("MASTER_ADDR", "127.0.0.1")

        @property
        def main_port(self) -> int:
            return int(os.environ.get("MASTER_PORT", "0"))

        @staticmethod
        def detect() -> bool:
            from ray_amd.train.session import _get_session

            return _get_session() is not None

        def set_world_size(self, size: int) -> None:
            pass  # fixed by the ray_amd worker group

        def set_global_rank(self, rank: int) -> None:
            pass

        def teardown(self):
            pass

    DDPStrategy = pl.strategies.DDPStrategy

    class RayDDPStrategy(DDPStrategy):
        """DDP against the process group ray_amd already initialized
        (setup_dist ran before the train loop)."""

        def __init__(self, *args, **kwargs):
            kwargs.setdefault("cluster_environment",
                              RayLightningEnvironment())
            super().__init__(*args, **kwargs)

        @property
        def root_device(self):
            import torch

            if torch.cuda.is_available():
                return torch.device(
                    "cuda", train.get_context().get_local_rank())
            return torch.device("cpu")

        @property
        def distributed_sampler_kwargs(self):
            ctx = train.get_context()
            return dict(num_replicas=ctx.get_world_size(),
                        rank=ctx.get_world_rank())

    class RayTrainReportCallback(pl.Callback):
        """Reports Lightning's logged metrics (+ a checkpoint) to
        ray_amd.train at every train-epoch end (reference:
        _lightning_utils.py RayTrainReportCallback)."""

        CHECKPOINT_NAME = "checkpoint.ckpt"

        def on_train_epoch_end(self, trainer, pl_module):
            metrics = {k: (v.item() if hasattr(v, "item") else v)
                       for k, v in trainer.callback_metrics.items()}
            metrics["epoch"] = trainer.current_epoch
            metrics["step"] = trainer.global_step
            with tempfile.TemporaryDirectory() as tmp:
                ckpt_path = os.path.join(tmp, self.CHECKPOINT_NAME)
                trainer.save_checkpoint(ckpt_path, weights_only=False)
                ckpt = train.Checkpoint.from_directory(tmp)
                train.report(metrics, checkpoint=ckpt)

    def prepare_trainer(trainer):
        """Validate the Trainer uses the Ray strategy/environment
        (reference: lightning/_lightning_utils.py prepare_trainer)."""
        strategy = getattr(trainer, "strategy", None)
        env = getattr(strategy, "cluster_environment", None)
        if strategy is not None and not isinstance(
                strategy, (RayDDPStrategy,)) and env is not None and \
                not isinstance(env, RayLightningEnvironment):
            raise RuntimeError(
                "Trainer must be configured with RayDDPStrategy (or a "
                "strategy holding RayLightningEnvironment) to run under "
                "ray_amd TorchTrainer"
            )
        return trainer

    _cache = SimpleNamespace(
        RayLightningEnvironment=RayLightningEnvironment,
        RayDDPStrategy=RayDDPStrategy,
        RayTrainReportCallback=RayTrainReportCallback,
        prepare_trainer=prepare_trainer,
    )
    return _cache
