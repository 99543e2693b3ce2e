from .transformers import (  # noqa: F401
    RayTrainReportCallback,
    prepare_trainer,
)
