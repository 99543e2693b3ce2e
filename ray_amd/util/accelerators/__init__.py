"""Accelerator type constants + helpers (reference:
python/ray/util/accelerators/accelerators.py). MI355X-first: the AMD
types are the ones this framework targets; requesting one maps to the
`accelerator_type:<T>` custom resource like the reference."""

AMD_INSTINCT_MI355X = "AMD-Instinct-MI355X"
AMD_INSTINCT_MI300X = "AMD-Instinct-MI300X"
AMD_INSTINCT_MI250X = "AMD-Instinct-MI250X"
AMD_INSTINCT_MI250 = "AMD-Instinct-MI250X-MI250"
AMD_INSTINCT_MI210 = "AMD-Instinct-MI210"
AMD_INSTINCT_MI100 = "AMD-Instinct-MI100"
AMD_RADEON_R9_200_HD_7900 = "AMD-Radeon-R9-200-HD-7900"
AMD_RADEON_HD_7900 = "AMD-Radeon-HD-7900"
# non-AMD names kept for API compatibility with the reference
NVIDIA_TESLA_V100 = "V100"
NVIDIA_TESLA_A100 = "A100"
NVIDIA_H100 = "H100"


def accelerator_resource(accelerator_type: str) -> str:
    """Custom-resource key a node advertises for a given accelerator."""
    return f"accelerator_type:{accelerator_type}"
