"""ray_amd.serve — model serving (reference: python/ray/serve/).

@serve.deployment / .bind() / serve.run / DeploymentHandle with
power-of-two-choices routing (request_router/pow_2_router.py:27),
ServeController actor reconciling replica sets
(_private/deployment_state.py), per-node HTTP proxy (uvicorn/ASGI,
_private/proxy.py:1046), serve.batch, and request-rate autoscaling.
"""
from .api import (  # noqa: F401
    get_multiplexed_model_id,
    multiplexed,
    Application,
    Deployment,
    DeploymentHandle,
    batch,
    delete,
    deployment,
    get_app_handle,
    get_deployment_handle,
    ingress,
    run,
    shutdown,
    start,
    status,
)


try:  # usage tagging (local-only; util/usage_stats.py)
    from ray_amd.util.usage_stats import record_library_usage

    record_library_usage("serve")
except Exception:  # pragma: no cover
    pass
