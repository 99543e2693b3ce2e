"""Per-worker training session (reference:
python/ray/train/_internal/session.py:698 report, :815 get_checkpoint,
:1149 get_dataset_shard; ray/train/context.py get_context)."""
from __future__ import annotations

import os
import queue
import shutil
import threading
from typing import Any, Dict, Optional

from .checkpoint import Checkpoint

_session: Optional["TrainSession"] = None


class TrainContext:
    def __init__(self, s: "TrainSession"):
        self._s = s

    def get_world_size(self) -> int:
        return self._s.world_size

    def get_world_rank(self) -> int:
        return self._s.rank

    def get_local_rank(self) -> int:
        return self._s.local_rank

    def get_local_world_size(self) -> int:
        return self._s.local_world_size

    def get_node_rank(self) -> int:
        return 0

    def get_trial_name(self) -> str:
        return self._s.run_name

    def get_experiment_name(self) -> str:
        return self._s.run_name

    def get_trial_id(self) -> str:
        return self._s.run_name

    def get_trial_resources(self):
        return None

    def get_storage(self):
        return self._s.storage_dir


class TrainSession:
    def __init__(self, rank, world_size, local_rank, local_world_size,
                 storage_dir, run_name, latest_checkpoint=None,
                 dataset_shards=None):
        self.rank = rank
        self.world_size = world_size
        self.local_rank = local_rank
        self.local_world_size = local_world_size
        self.storage_dir = storage_dir
        self.run_name = run_name
        self.results_queue: "queue.Queue" = queue.Queue()
        self.latest_checkpoint = latest_checkpoint
        self.dataset_shards = dataset_shards or {}
        self.iteration = 0
        self.stop_requested = threading.Event()

    def report(self, metrics: Dict[str, Any],
               checkpoint: Optional[Checkpoint] = None):
        self.iteration += 1
        ckpt_path = None
        if checkpoint is not None:
            # persist the checkpoint into run storage (all ranks may
            # report; rank-0 layout matches the reference:
            # <storage>/checkpoint_NNNNNN/)
            dest = os.path.join(
                self.storage_dir,
                f"checkpoint_{self.iteration:06d}"
                + ("" if self.rank == 0 else f"_rank{self.rank}"),
            )
            os.makedirs(dest, exist_ok=True)
            if os.path.abspath(checkpoint.path) != os.path.abspath(dest):
                shutil.copytree(checkpoint.path, dest, dirs_exist_ok=True)
            ckpt_path = dest
            self.latest_checkpoint = Checkpoint(dest)
        self.results_queue.put(
            {"metrics": dict(metrics), "checkpoint_path": ckpt_path,
             "iteration": self.iteration, "rank": self.rank}
        )


def _set_session(s: Optional[TrainSession]):
    global _session
    _session = s


def _get_session() -> Optional[TrainSession]:
    return _session


def report(metrics: Dict[str, Any], checkpoint: Optional[Checkpoint] = None,
           checkpoint_dir_name: Optional[str] = None):
    if _session is None:
        raise RuntimeError("ray_amd.train.report() called outside a train loop")
    _session.report(metrics, checkpoint)


def get_checkpoint() -> Optional[Checkpoint]:
    if _session is None:
        return None
    return _session.latest_checkpoint


def get_context() -> TrainContext:
    if _session is None:
        # driver-side context (world of 1)
        return TrainContext(
            TrainSession(0, 1, 0, 1, "/tmp", "driver")
        )
    return TrainContext(_session)


def get_dataset_shard(dataset_name: str = "train"):
    if _session is None:
        raise RuntimeError("get_dataset_shard() called outside a train loop")
    return _session.dataset_shards.get(dataset_name)
