"""ray_amd.rllib — reinforcement learning (reference: rllib/)."""
from .algorithm import Algorithm, AlgorithmConfig  # noqa: F401
from .env import register_env  # noqa: F401
