"""Autoscaler (reference: python/ray/autoscaler/ v1+v2).

ray_amd round-1 scope: demand-driven scaling of local raylet "nodes"
(the FakeMultiNodeProvider-style harness the reference uses for
autoscaler tests, autoscaler/_private/fake_multi_node/node_provider.py)
plus the `sdk.request_resources` API. Cloud node providers are a
later-round item; the scaling loop, demand signals (queued leases
reported by raylets + explicit requests) and node lifecycle are real.
"""
from . import sdk  # noqa: F401
from .autoscaler import LocalAutoscaler  # noqa: F401
from .v2 import (  # noqa: F401
    AutoscalerV2,
    LocalNodeProvider,
    NodeProvider,
    NodeType,
)
