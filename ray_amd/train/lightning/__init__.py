"""PyTorch Lightning integration (reference: python/ray/train/lightning/
_lightning_utils.py — RayDDPStrategy, RayLightningEnvironment,
RayTrainReportCallback, prepare_trainer).

Lightning is an optional dependency: the classes are built lazily
against `pytorch_lightning` (or `lightning.pytorch`) on first access so
importing ray_amd.train.lightning without the library only fails when a
class is actually used.
"""

__all__ = [
    "RayDDPStrategy",
    "RayLightningEnvironment",
    "RayTrainReportCallback",
    "prepare_trainer",
]


def __getattr__(name):
    if name in __all__:
        from . import _impl

        return getattr(_impl.build(), name)
    raise AttributeError(name)
