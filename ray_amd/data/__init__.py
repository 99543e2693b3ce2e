"""ray_amd.data — distributed datasets (reference: python/ray/data/).

Blocks are pyarrow Tables stored in the shm object store; transforms
run as ray_amd tasks (task-pool) or actor pools (map_batches
concurrency), with a windowed streaming iterator for consumption
(reference: _internal/execution/streaming_executor.py:107 — a bounded
in-flight window provides the same backpressure effect for the v1
pipeline).
"""
from . import expressions  # noqa: F401
from .expressions import col, lit  # noqa: F401
from .dataset import (  # noqa: F401
    ActorPoolStrategy,
    Dataset,
    DataContext,
    from_arrow,
    from_items,
    from_numpy,
    from_pandas,
    range,  # noqa: A001  (API parity with ray.data.range)
    range_tensor,
    from_huggingface,
    read_binary_files,
    read_text,
    read_csv,
    read_json,
    read_parquet,
    read_numpy,
    read_webdataset,
    read_datasource,
    Datasource,
    Datasink,
)
from .preprocessors import (  # noqa: F401
    Concatenator,
    LabelEncoder,
    MinMaxScaler,
    StandardScaler,
    TorchVisionNormalizer,
)


try:  # usage tagging (local-only; util/usage_stats.py)
    from ray_amd.util.usage_stats import record_library_usage

    record_library_usage("data")
except Exception:  # pragma: no cover
    pass
