"""Experimental: GPU-native tensor transport (RDT), channels, object
locations, dynamic resources."""
from .rdt import (  # noqa: F401
    GpuObjectRef,
    GpuObjectStore,
    get_gpu_object_store,
)
from .locations import (  # noqa: F401
    get_object_locations,
    set_resource,
)
