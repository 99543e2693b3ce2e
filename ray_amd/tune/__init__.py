"""ray_amd.tune — hyperparameter tuning (reference: python/ray/tune/).

Tuner.fit (tuner.py:332) over actor-based trial execution
(execution/tune_controller.py), basic-variant search (grid + random),
ASHA scheduler (schedulers/async_hyperband.py), ResultGrid.
"""
from .impl import (  # noqa: F401
    ASHAScheduler,
    Callback,
    FIFOScheduler,
    HyperBandScheduler,
    MedianStoppingRule,
    PB2,
    PopulationBasedTraining,
    ResultGrid,
    TuneConfig,
    Tuner,
    choice,
    grid_search,
    loguniform,
    qrandint,
    randint,
    randn,
    report,
    run,
    sample_from,
    uniform,
    with_parameters,
    with_resources,
)
from . import search  # noqa: F401
from .search import (  # noqa: F401
    BasicVariantGenerator,
    BayesOptSearch,
    ConcurrencyLimiter,
    HyperOptSearch,
    OptunaSearch,
    Searcher,
)
from ..train.checkpoint import Checkpoint  # noqa: F401
from ..train.session import get_checkpoint, get_context  # noqa: F401


try:  # usage tagging (local-only; util/usage_stats.py)
    from ray_amd.util.usage_stats import record_library_usage

    record_library_usage("tune")
except Exception:  # pragma: no cover
    pass
