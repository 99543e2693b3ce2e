"""ray_amd.train.torch — TorchTrainer and torch utilities.

Reference: train/torch/torch_trainer.py, train/torch/train_loop_utils.py
(prepare_model :159 — DDP wrap; prepare_data_loader :203 —
DistributedSampler + device transfer; get_device).

MI355X specifics: DDP bucket_cap_mb defaults to 128 (xGMI rings are
per-link bound, ~153 GB/s/link — large buckets amortize; SURVEY.md
§5.8), gradient_as_bucket_view=True to avoid a grad copy.
"""
from __future__ import annotations

from typing import Optional

import torch

from .session import get_context
from .trainer import DataParallelTrainer

from .._config import config as _cfg

XGMI_BUCKET_CAP_MB = _cfg.ddp_bucket_cap_mb


class TorchTrainer(DataParallelTrainer):
    _default_backend = "gloo"


class TorchConfig:
    def __init__(self, backend: Optional[str] = None, timeout_s: int = 1800):
        self.backend = backend
        self.timeout_s = timeout_s


def get_device() -> torch.device:
    if torch.cuda.is_available():
        import os

        idx = int(os.environ.get("LOCAL_RANK", "0"))
        if idx >= torch.cuda.device_count():
            idx = 0
        return torch.device("cuda", idx)
    return torch.device("cpu")


def get_devices():
    return [get_device()]


def prepare_model(
    model: torch.nn.Module,
    move_to_device: bool = True,
    parallel_strategy: Optional[str] = "ddp",
    parallel_strategy_kwargs: Optional[dict] = None,
) -> torch.nn.Module:
    """Move to this worker's device and wrap in DDP over RCCL."""
    import torch.distributed as dist

    device = get_device()
    if move_to_device:
        model = model.to(device)
        # move non-persistent buffers too (rope tables etc.)
        for name, buf in model.named_buffers():
            if buf.device != device:
                parts = name.split(".")
                mod = model
                for p in parts[:-1]:
                    mod = getattr(mod, p)
                setattr(mod, parts[-1], buf.to(device))
    world = get_context().get_world_size()
    if parallel_strategy and world > 1 and dist.is_initialized():
        kwargs = dict(parallel_strategy_kwargs or {})
        if parallel_strategy == "ddp":
            from torch.nn.parallel import DistributedDataParallel as DDP

            kwargs.setdefault("bucket_cap_mb", XGMI_BUCKET_CAP_MB)
            kwargs.setdefault("gradient_as_bucket_view", True)
            if device.type == "cuda":
                kwargs.setdefault("device_ids", [device.index or 0])
            model = DDP(model, **kwargs)
        elif parallel_strategy == "fsdp":
            from torch.distributed.fsdp import FullyShardedDataParallel as FSDP

            model = FSDP(model, **kwargs)
        else:
            raise ValueError(f"unknown parallel_strategy {parallel_strategy}")
    return model


def prepare_data_loader(
    data_loader: torch.utils.data.DataLoader,
    add_dist_sampler: bool = True,
    move_to_device: bool = True,
    auto_transfer: bool = True,
) -> torch.utils.data.DataLoader:
    ctx = get_context()
    world = ctx.get_world_size()
    rank = ctx.get_world_rank()
    if world > 1 and add_dist_sampler and not isinstance(
        data_loader.sampler, torch.utils.data.DistributedSampler
    ):
        sampler = torch.utils.data.DistributedSampler(
            data_loader.dataset, num_replicas=world, rank=rank
        )
        data_loader = torch.utils.data.DataLoader(
            data_loader.dataset,
            batch_size=data_loader.batch_size,
            sampler=sampler,
            num_workers=data_loader.num_workers,
            collate_fn=data_loader.collate_fn,
            pin_memory=data_loader.pin_memory,
            drop_last=data_loader.drop_last,
        )
    if move_to_device:
        device = get_device()

        class _DeviceLoader:
            def __init__(self, dl):
                self._dl = dl

            def __iter__(self):
                for batch in self._dl:
                    yield _move(batch, device)

            def __len__(self):
                return len(self._dl)

            def __getattr__(self, item):
                return getattr(self._dl, item)

        return _DeviceLoader(data_loader)
    return data_loader


def _move(batch, device):
    if isinstance(batch, torch.Tensor):
        return batch.to(device, non_blocking=True)
    if isinstance(batch, (list, tuple)):
        return type(batch)(_move(b, device) for b in batch)
    if isinstance(batch, dict):
        return {k: _move(v, device) for k, v in batch.items()}
    return batch


def enable_reproducibility(seed: int = 0):
    import random

    import numpy as np

    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)


def backward(loss):  # AMP-era helper parity
    loss.backward()
