"""Experimental: GPU-native tensor transport (RDT) and channels."""
from .rdt import (  # noqa: F401
    GpuObjectRef,
    GpuObjectStore,
    get_gpu_object_store,
)
