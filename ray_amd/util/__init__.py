"""ray_amd.util — utility APIs (reference: python/ray/util/)."""
from .actor_pool import ActorPool  # noqa: F401
from .placement_group import (  # noqa: F401
    PlacementGroup,
    get_current_placement_group,
    placement_group,
    placement_group_table,
    remove_placement_group,
)
from .queue import Queue  # noqa: F401
from .scheduling_strategies import (  # noqa: F401
    NodeAffinitySchedulingStrategy,
    NodeLabelSchedulingStrategy,
    PlacementGroupSchedulingStrategy,
)


def __getattr__(name):
    if name in ("collective", "state", "metrics"):
        import importlib

        mod = importlib.import_module(f".{name}", __name__)
        globals()[name] = mod
        return mod
    raise AttributeError(name)
