"""CQL — Conservative Q-Learning, offline continuous control
(reference: rllib/algorithms/cql/ — SAC losses + the conservative
regularizer that pushes down Q on out-of-distribution actions and up
on dataset actions; Kumar et al. 2020).

Trains from a transition Dataset ({obs, action, reward, next_obs,
done}); no environment interaction during training.
"""
from __future__ import annotations

import copy

import numpy as np
import torch

from ..algorithm import Algorithm, AlgorithmConfig
from ..env import VectorEnv
from .sac import SACModule


class CQLConfig(AlgorithmConfig):
    def __init__(self):
        super().__init__(algo_class=CQL)
        self.env = "Pendulum-v1"   # used for spaces + evaluation only
        self.lr = 3e-4
        self.gamma = 0.99
        self.tau = 0.005
        self.train_batch_size = 256
        self.updates_per_iteration = 50
        self.input_ = None
        self.cql_alpha = 1.0        # conservative penalty weight
        self.num_ood_actions = 4    # sampled actions for the logsumexp
        self.initial_alpha = 0.2    # SAC entropy temperature

    def offline_data(self, *, input_=None, **kwargs):
        if input_ is not None:
            self.input_ = input_
        return self


def _load_transitions(ds):
    rows = ds.take_all()
    return {
        "obs": np.stack([np.asarray(r["obs"], np.float32) for r in rows]),
        "next_obs": np.stack(
            [np.asarray(r["next_obs"], np.float32) for r in rows]),
        "actions": np.stack(
            [np.asarray(r["action"], np.float32).reshape(-1) for r in rows]),
        "rewards": np.asarray([r["reward"] for r in rows], np.float32),
        "dones": np.asarray([float(r["done"]) for r in rows], np.float32),
    }


class CQL(Algorithm):
    def _setup(self, config: CQLConfig):
        from ... import data as ray_data

        ds = config.input_
        if isinstance(ds, str):
            ds = ray_data.read_parquet(ds)
        if ds is None:
            raise ValueError("CQLConfig.offline_data(input_=...) required")
        self._data = _load_transitions(ds)
        probe = VectorEnv(config.env, 1, seed=config.seed)
        obs_dim = int(np.prod(probe.observation_space.shape))
        self.act_dim = int(np.prod(probe.action_space.shape))
        self.act_limit = float(np.max(np.abs(probe.action_space.high)))
        self.device = torch.device(
            "cuda:0"
            if config.num_gpus_per_learner > 0 and torch.cuda.is_available()
            else "cpu"
        )
        self.module = SACModule(obs_dim, self.act_dim, self.act_limit,
                                config.model_hidden).to(self.device)
        self.target = copy.deepcopy(self.module).to(self.device)
        for p in self.target.parameters():
            p.requires_grad_(False)
        self.pi_opt = torch.optim.Adam(self.module.actor.parameters(),
                                       lr=config.lr)
        self.q_opt = torch.optim.Adam(
            list(self.module.q1.parameters())
            + list(self.module.q2.parameters()), lr=config.lr)
        self.log_alpha = torch.tensor(
            float(np.log(config.initial_alpha)), device=self.device)
        self._rng = np.random.default_rng(config.seed)

    def _batch(self):
        n = len(self._data["rewards"])
        idx = self._rng.integers(0, n, min(self.config.train_batch_size, n))
        d = self.device
        return tuple(
            torch.as_tensor(self._data[k][idx], device=d)
            for k in ("obs", "next_obs", "actions", "rewards", "dones")
        )

    def _update_once(self):
        cfg = self.config
        obs, nobs, act, rew, done = self._batch()
        cont = 1.0 - done
        alpha = self.log_alpha.exp()

        with torch.no_grad():
            na, nlogp = self.module.pi(nobs)
            tq1, tq2 = self.target.q(nobs, na)
            target = rew + cfg.gamma * cont * (
                torch.min(tq1, tq2) - alpha * nlogp)
        q1, q2 = self.module.q(obs, act)
        bellman = ((q1 - target) ** 2).mean() + ((q2 - target) ** 2).mean()

        # conservative penalty: logsumexp over {uniform, policy} actions
        # minus Q at the dataset action (pushes OOD actions down)
        B = obs.shape[0]
        K = cfg.num_ood_actions
        rand_a = (torch.rand(K, B, self.act_dim, device=self.device) * 2
                  - 1) * self.act_limit
        with torch.no_grad():
            pol_a, _ = self.module.pi(
                obs.unsqueeze(0).expand(K, -1, -1).reshape(K * B, -1))
        cat_a = torch.cat([rand_a.reshape(K * B, -1), pol_a], 0)
        cat_o = obs.unsqueeze(0).expand(2 * K, -1, -1).reshape(2 * K * B, -1)
        cq1, cq2 = self.module.q(cat_o, cat_a)
        lse1 = torch.logsumexp(cq1.view(2 * K, B), dim=0).mean()
        lse2 = torch.logsumexp(cq2.view(2 * K, B), dim=0).mean()
        conservative = (lse1 - q1.mean()) + (lse2 - q2.mean())
        q_loss = bellman + cfg.cql_alpha * conservative
        self.q_opt.zero_grad(set_to_none=True)
        q_loss.backward()
        self.q_opt.step()

        for p in self.module.q1.parameters():
            p.requires_grad_(False)
        for p in self.module.q2.parameters():
            p.requires_grad_(False)
        a, logp = self.module.pi(obs)
        qa1, qa2 = self.module.q(obs, a)
        pi_loss = (alpha.detach() * logp - torch.min(qa1, qa2)).mean()
        self.pi_opt.zero_grad(set_to_none=True)
        pi_loss.backward()
        self.pi_opt.step()
        for p in self.module.q1.parameters():
            p.requires_grad_(True)
        for p in self.module.q2.parameters():
            p.requires_grad_(True)

        with torch.no_grad():
            for p, tp in zip(self.module.parameters(),
                             self.target.parameters()):
                tp.mul_(1 - cfg.tau).add_(cfg.tau * p)
        return {
            "q_loss": float(q_loss.detach()),
            "bellman_loss": float(bellman.detach()),
            "conservative_gap": float(conservative.detach()),
            "pi_loss": float(pi_loss.detach()),
        }

    def training_step(self):
        stats = {}
        for _ in range(self.config.updates_per_iteration):
            stats = self._update_once()
        return {"learner": stats,
                "num_rows": len(self._data["rewards"])}

    def evaluate(self, num_steps: int = 500, num_envs: int = 4):
        vec = VectorEnv(self.config.env, num_envs,
                        seed=self.config.seed + 1)
        obs = vec.reset()
        for _ in range(num_steps // num_envs):
            with torch.no_grad():
                a = self.module.pi(
                    torch.as_tensor(obs, dtype=torch.float32,
                                    device=self.device),
                    deterministic=True,
                )[0].cpu().numpy()
            obs, _, _, _ = vec.step(a)
        rets, _ = vec.pop_episode_stats()
        return {"episode_reward_mean":
                float(np.mean(rets)) if len(rets) else None}

    def get_weights(self):
        return {k: v.cpu().numpy()
                for k, v in self.module.state_dict().items()}

    def set_weights(self, w):
        self.module.load_state_dict(
            {k: torch.as_tensor(v) for k, v in w.items()})
        self.target = copy.deepcopy(self.module).to(self.device)
