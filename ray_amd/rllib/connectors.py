"""Connector pipelines (reference: rllib/connectors/ — composable
env->module and learner transform chains; GAE ships as a learner
connector, connectors/learner/general_advantage_estimation.py:21).

A connector is `__call__(batch, **kw) -> batch`; pipelines support the
reference's insert/append/prepend/remove surface. Env->module
connectors transform observation batches before inference; learner
connectors transform the train batch before the loss (GAE runs the
HIP scan kernel on GPU).
"""
from __future__ import annotations

from typing import Dict, List

import numpy as np


class Connector:
    def __call__(self, batch: Dict[str, np.ndarray], **kw):
        raise NotImplementedError

    @property
    def name(self) -> str:
        return type(self).__name__


class ConnectorPipeline(Connector):
    def __init__(self, connectors: List[Connector] = None):
        self.connectors: List[Connector] = list(connectors or [])

    def __call__(self, batch, **kw):
        for c in self.connectors:
            batch = c(batch, **kw)
        return batch

    # reference surface: insert_before/insert_after/prepend/append/remove
    def append(self, c: Connector):
        self.connectors.append(c)
        return self

    def prepend(self, c: Connector):
        self.connectors.insert(0, c)
        return self

    def _idx(self, name: str) -> int:
        for i, c in enumerate(self.connectors):
            if c.name == name or type(c).__name__ == name:
                return i
        raise ValueError(f"no connector named {name!r}")

    def insert_before(self, name: str, c: Connector):
        self.connectors.insert(self._idx(name), c)
        return self

    def insert_after(self, name: str, c: Connector):
        self.connectors.insert(self._idx(name) + 1, c)
        return self

    def remove(self, name: str):
        self.connectors.pop(self._idx(name))
        return self


# ---------------- env -> module connectors ----------------


class FlattenObservations(Connector):
    def __call__(self, batch, **kw):
        obs = batch["obs"]
        if obs.ndim > 2:
            batch = dict(batch)
            batch["obs"] = obs.reshape(obs.shape[0], -1)
        return batch


class NormalizeObservations(Connector):
    """Running mean/std normalization (Welford)."""

    def __init__(self, eps: float = 1e-8, clip: float = 10.0):
        self.eps = eps
        self.clip = clip
        self.count = 0.0
        self.mean = None
        self.m2 = None

    def __call__(self, batch, **kw):
        obs = np.asarray(batch["obs"], np.float64)
        flat = obs.reshape(-1, obs.shape[-1])
        if self.mean is None:
            self.mean = np.zeros(flat.shape[-1])
            self.m2 = np.zeros(flat.shape[-1])
        for row in flat:
            self.count += 1
            d = row - self.mean
            self.mean += d / self.count
            self.m2 += d * (row - self.mean)
        var = self.m2 / max(self.count - 1, 1)
        std = np.sqrt(var) + self.eps
        batch = dict(batch)
        batch["obs"] = np.clip(
            (obs - self.mean) / std, -self.clip, self.clip
        ).astype(np.float32)
        return batch


class FrameStacking(Connector):
    """Stack the last N observation frames along the feature axis
    (reference: connectors/learner/frame_stacking.py)."""

    def __init__(self, num_frames: int = 4):
        self.n = num_frames
        self._hist = None

    def __call__(self, batch, **kw):
        obs = np.asarray(batch["obs"])
        if self._hist is None or self._hist[0].shape != obs.shape:
            self._hist = [obs] * self.n
        else:
            self._hist.pop(0)
            self._hist.append(obs)
        batch = dict(batch)
        batch["obs"] = np.concatenate(self._hist, axis=-1)
        return batch


class ClipRewards(Connector):
    def __init__(self, limit: float = 1.0):
        self.limit = limit

    def __call__(self, batch, **kw):
        if "rewards" in batch:
            batch = dict(batch)
            batch["rewards"] = np.clip(batch["rewards"], -self.limit,
                                       self.limit)
        return batch


# ---------------- learner connectors ----------------


class GeneralAdvantageEstimation(Connector):
    """GAE as a learner connector (reference:
    general_advantage_estimation.py:21); runs the HIP warp-scan kernel
    on GPU, numpy recursion on CPU. Expects [T, B] rewards/vf (vf has
    T+1 rows: bootstrap) and writes advantages/value_targets."""

    def __init__(self, gamma: float = 0.99, lambda_: float = 0.95,
                 device="cpu"):
        self.gamma = gamma
        self.lambda_ = lambda_
        self.device = device

    def __call__(self, batch, **kw):
        import torch

        from ray_amd import ops

        rewards = torch.as_tensor(batch["rewards"], device=self.device)
        values = torch.as_tensor(batch["vf"], device=self.device)
        cont = 1.0 - torch.as_tensor(batch["dones"], device=self.device)
        adv, vtarg = ops.gae(rewards.float(), values.float(), cont.float(),
                             self.gamma, self.lambda_)
        batch = dict(batch)
        batch["advantages"] = adv
        batch["value_targets"] = vtarg
        return batch


class StandardizeAdvantages(Connector):
    def __call__(self, batch, **kw):
        adv = batch.get("advantages")
        if adv is not None:
            std, mean = adv.std(), adv.mean()
            batch = dict(batch)
            batch["advantages"] = (adv - mean) / (std + 1e-8)
        return batch
