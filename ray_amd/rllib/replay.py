"""Replay buffers (reference: rllib/utils/replay_buffers/)."""
from __future__ import annotations

from typing import Dict

import numpy as np


class ReplayBuffer:
    """Uniform FIFO transition buffer."""

    def __init__(self, capacity: int = 100_000):
        self.capacity = capacity
        self._store: Dict[str, np.ndarray] = {}
        self._size = 0
        self._idx = 0

    def add_batch(self, batch: Dict[str, np.ndarray]):
        n = len(batch["actions"])
        if not self._store:
            for k, v in batch.items():
                shape = (self.capacity,) + tuple(v.shape[1:])
                self._store[k] = np.zeros(shape, dtype=v.dtype)
        for k, v in batch.items():
            idxs = (self._idx + np.arange(n)) % self.capacity
            self._store[k][idxs] = v
        self._idx = (self._idx + n) % self.capacity
        self._size = min(self._size + n, self.capacity)

    def sample(self, batch_size: int, rng=None) -> Dict[str, np.ndarray]:
        rng = rng or np.random.default_rng()
        idxs = rng.integers(0, self._size, batch_size)
        return {k: v[idxs] for k, v in self._store.items()}

    def __len__(self):
        return self._size


class PrioritizedReplayBuffer(ReplayBuffer):
    """Proportional prioritization (reference: prioritized buffers)."""

    def __init__(self, capacity: int = 100_000, alpha: float = 0.6):
        super().__init__(capacity)
        self.alpha = alpha
        self._prio = np.zeros(capacity, dtype=np.float64)
        self._max_prio = 1.0

    def add_batch(self, batch):
        n = len(batch["actions"])
        idxs = (self._idx + np.arange(n)) % self.capacity
        super().add_batch(batch)
        self._prio[idxs] = self._max_prio

    def sample(self, batch_size: int, rng=None, beta: float = 0.4):
        rng = rng or np.random.default_rng()
        p = self._prio[: self._size] ** self.alpha
        p = p / p.sum()
        idxs = rng.choice(self._size, batch_size, p=p)
        out = {k: v[idxs] for k, v in self._store.items()}
        w = (self._size * p[idxs]) ** (-beta)
        out["weights"] = (w / w.max()).astype(np.float32)
        out["batch_indexes"] = idxs
        return out

    def update_priorities(self, idxs, td_errors):
        pr = np.abs(td_errors) + 1e-6
        self._prio[idxs] = pr
        self._max_prio = max(self._max_prio, pr.max())
