from .tracing_helper import (  # noqa: F401
    current_span_context,
    get_trace_events,
    tracing_enabled,
    enable_tracing,
    span,
)
