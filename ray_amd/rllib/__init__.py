"""ray_amd.rllib — reinforcement learning (reference: rllib/)."""
from .algorithm import Algorithm, AlgorithmConfig  # noqa: F401
from .env import register_env  # noqa: F401


try:  # usage tagging (local-only; util/usage_stats.py)
    from ray_amd.util.usage_stats import record_library_usage

    record_library_usage("rllib")
except Exception:  # pragma: no cover
    pass
