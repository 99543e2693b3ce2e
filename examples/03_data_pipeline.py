"""Data pipeline: transform, shuffle, groupby, torch batches."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import ray_amd as ray
import ray_amd.data as rd

ray.init()
ds = (
    rd.range(1000, parallelism=8)
    .map_batches(lambda b: {"id": b["id"], "bucket": b["id"] % 5})
    .random_shuffle(seed=0)
)
print("count:", ds.count())
print("per-bucket sums:", ds.groupby("bucket").sum("id").take_all())
for batch in ds.iter_torch_batches(batch_size=256, device="cpu"):
    print("torch batch:", batch["id"].shape, batch["id"].dtype)
    break
ray.shutdown()
