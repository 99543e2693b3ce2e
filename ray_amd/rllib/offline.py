"""Offline RL (reference: rllib/offline/ — offline data pipeline on
Ray Data; rllib/algorithms/bc behavior cloning, marwil.py).

- record_episodes: roll a policy out and materialize (obs, action,
  reward, done) rows as a ray_amd.data Dataset (the reference writes
  offline episodes through Ray Data too).
- BC: supervised policy learning from such a Dataset (cross-entropy on
  the logged actions), with optional MARWIL-style advantage weighting
  when the rows carry "advantages".
"""
from __future__ import annotations

from typing import Callable, Optional

import numpy as np
import torch

from .algorithm import Algorithm, AlgorithmConfig
from .core import TorchRLModule
from .env import VectorEnv


def record_episodes(env: str, policy_fn: Optional[Callable] = None,
                    num_steps: int = 2000, num_envs: int = 8,
                    seed: int = 0):
    """Roll `policy_fn(obs_batch) -> actions` (random if None) and
    return a Dataset of transitions."""
    from .. import data as ray_data

    vec = VectorEnv(env, num_envs, seed=seed)
    rng = np.random.default_rng(seed)
    obs = vec.reset()
    rows = []
    for _ in range(num_steps // num_envs):
        if policy_fn is None:
            acts = rng.integers(0, vec.action_space.n, num_envs)
        else:
            acts = np.asarray(policy_fn(obs))
        nobs, rew, term, trunc = vec.step(acts)
        done = np.logical_or(term, trunc)
        for i in range(num_envs):
            rows.append({
                "obs": obs[i].astype(np.float32),
                "action": int(acts[i]),
                "reward": float(rew[i]),
                "done": bool(done[i]),
            })
        obs = nobs
    return ray_data.from_items(rows)


class BCConfig(AlgorithmConfig):
    def __init__(self):
        super().__init__(algo_class=BC)
        self.lr = 1e-3
        self.train_batch_size = 512
        self.updates_per_iteration = 32
        self.input_ = None          # Dataset or parquet path
        self.beta = 0.0             # >0: MARWIL advantage weighting

    def offline_data(self, *, input_=None, **kwargs):
        if input_ is not None:
            self.input_ = input_
        return self


class BC(Algorithm):
    """Behavior cloning from a Dataset of {obs, action[, advantages]}."""

    def _setup(self, config: BCConfig):
        from .. import data as ray_data

        ds = config.input_
        if isinstance(ds, str):
            ds = ray_data.read_parquet(ds)
        if ds is None:
            raise ValueError("BCConfig.offline_data(input_=...) required")
        rows = ds.take_all()
        self._obs = np.stack([np.asarray(r["obs"], np.float32)
                              for r in rows])
        self._acts = np.asarray([r["action"] for r in rows], np.int64)
        self._adv = (
            np.asarray([r["advantages"] for r in rows], np.float32)
            if "advantages" in rows[0] else None
        )
        probe = VectorEnv(config.env, 1, seed=config.seed)
        obs_dim = int(np.prod(probe.observation_space.shape))
        self.num_actions = probe.action_space.n
        self.device = torch.device(
            "cuda:0"
            if config.num_gpus_per_learner > 0 and torch.cuda.is_available()
            else "cpu"
        )
        self.module = TorchRLModule(obs_dim, self.num_actions,
                                    hidden=config.model_hidden,
                                    device=self.device)
        self.opt = torch.optim.Adam(self.module.parameters(), lr=config.lr)
        self._rng = np.random.default_rng(config.seed)

    def training_step(self):
        cfg = self.config
        n = len(self._acts)
        stats = {}
        for _ in range(cfg.updates_per_iteration):
            idx = self._rng.integers(0, n, min(cfg.train_batch_size, n))
            obs = torch.as_tensor(self._obs[idx], device=self.device)
            act = torch.as_tensor(self._acts[idx], device=self.device)
            logits = self.module(obs)["logits"]
            logp = torch.log_softmax(logits, -1).gather(
                1, act.view(-1, 1)).squeeze(1)
            if self._adv is not None and cfg.beta > 0:
                w = torch.as_tensor(
                    np.exp(cfg.beta * self._adv[idx]), device=self.device)
                loss = -(w * logp).mean()
            else:
                loss = -logp.mean()
            self.opt.zero_grad(set_to_none=True)
            loss.backward()
            self.opt.step()
            acc = float((logits.argmax(-1) == act).float().mean())
            stats = {"bc_loss": float(loss.detach()), "action_acc": acc}
        return {"learner": stats, "num_rows": n}

    def evaluate(self, num_steps: int = 1000, num_envs: int = 8):
        """Greedy rollout of the cloned policy; returns episode stats."""
        vec = VectorEnv(self.config.env, num_envs,
                        seed=self.config.seed + 1)
        obs = vec.reset()
        for _ in range(num_steps // num_envs):
            acts = self.module.forward_inference(obs)
            obs, _, _, _ = vec.step(acts)
        rets, lens = vec.pop_episode_stats()
        return {
            "episode_reward_mean":
                float(np.mean(rets)) if len(rets) else None,
            "episodes": len(rets),
        }

    def get_weights(self):
        return self.module.get_weights()

    def set_weights(self, w):
        self.module.set_weights(w)


def record_continuous_episodes(env: str, policy_fn=None,
                               num_steps: int = 2000, num_envs: int = 8,
                               seed: int = 0):
    """Continuous-action transition recorder: rows carry next_obs too
    (what CQL/IQL consume)."""
    from .. import data as ray_data

    vec = VectorEnv(env, num_envs, seed=seed)
    rng = np.random.default_rng(seed)
    act_dim = int(np.prod(vec.action_space.shape))
    limit = float(np.max(np.abs(vec.action_space.high)))
    obs = vec.reset()
    rows = []
    for _ in range(num_steps // num_envs):
        if policy_fn is None:
            acts = rng.uniform(-limit, limit,
                               size=(num_envs, act_dim)).astype(np.float32)
        else:
            acts = np.asarray(policy_fn(obs), np.float32)
        nobs, rew, term, trunc = vec.step(acts)
        done = np.logical_or(term, trunc)
        for i in range(num_envs):
            rows.append({
                "obs": obs[i].astype(np.float32),
                "next_obs": nobs[i].astype(np.float32),
                "action": acts[i],
                "reward": float(rew[i]),
                "done": bool(done[i]),
            })
        obs = nobs
    return ray_data.from_items(rows)
