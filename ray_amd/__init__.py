"""ray_amd — an MI355X-native distributed compute engine with the
capabilities of ray-project/ray: task/actor model, shared-memory object
store, collectives over RCCL/xGMI, and the AIR libraries (data, train,
tune, serve, rllib).

Public API mirrors `ray` (python/ray/__init__.py of the reference):

    import ray_amd as ray
    ray.init()

    @ray.remote
    def f(x): return x + 1

    ray.get(f.remote(1))
"""
from . import exceptions  # noqa: F401
from .api import (  # noqa: F401
    ActorClass,
    ActorHandle,
    ActorMethod,
    ObjectRef,
    RemoteFunction,
    available_resources,
    cancel,
    cluster_resources,
    drain_node,
    get,
    get_actor,
    get_gpu_ids,
    timeline,
    get_runtime_context,
    init,
    is_initialized,
    kill,
    method,
    nodes,
    put,
    remote,
    shutdown,
    wait,
)

__version__ = "0.1.0"
__commit__ = "dev"


class _ActorExitHelper:
    @staticmethod
    def exit_actor():
        raise SystemExit(0)


def exit_actor():
    """Terminate the current actor from inside one of its methods."""
    raise SystemExit(0)


# Lazy submodule access so `import ray_amd` stays light (no torch import).
_LAZY = {
    "data",
    "train",
    "tune",
    "serve",
    "rllib",
    "util",
    "dag",
    "experimental",
    "cluster_utils",
    "runtime_context",
    "air",
}


def __getattr__(name):
    if name in _LAZY:
        import importlib

        mod = importlib.import_module(f".{name}", __name__)
        globals()[name] = mod
        return mod
    raise AttributeError(f"module 'ray_amd' has no attribute {name!r}")
