"""ray_amd.runtime_context (reference: python/ray/runtime_context.py)."""
from .api import RuntimeContext, get_runtime_context  # noqa: F401
