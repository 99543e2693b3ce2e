"""IQL — Implicit Q-Learning, offline continuous control (reference:
rllib/algorithms/iql-class offline methods; Kostrikov et al. 2021):
expectile regression fits V to the tau-expectile of Q, Q regresses to
r + gamma V(s'), and the policy is extracted with advantage-weighted
regression — no OOD action queries at all.
"""
from __future__ import annotations

import copy

import numpy as np
import torch
import torch.nn as nn

from ..algorithm import Algorithm, AlgorithmConfig
from ..env import VectorEnv
from .cql import _load_transitions
from .sac import LOG_STD_MAX, LOG_STD_MIN, SACModule, _mlp


class IQLConfig(AlgorithmConfig):
    def __init__(self):
        super().__init__(algo_class=IQL)
        self.env = "Pendulum-v1"
        self.lr = 3e-4
        self.gamma = 0.99
        self.tau = 0.005
        self.train_batch_size = 256
        self.updates_per_iteration = 50
        self.input_ = None
        self.expectile = 0.7        # V fits this expectile of Q
        self.awr_beta = 3.0         # advantage-weighted regression temp
        self.awr_weight_clip = 100.0

    def offline_data(self, *, input_=None, **kwargs):
        if input_ is not None:
            self.input_ = input_
        return self


class IQL(Algorithm):
    def _setup(self, config: IQLConfig):
        from ... import data as ray_data

        ds = config.input_
        if isinstance(ds, str):
            ds = ray_data.read_parquet(ds)
        if ds is None:
            raise ValueError("IQLConfig.offline_data(input_=...) required")
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
        self.vnet = _mlp((obs_dim,) + tuple(config.model_hidden),
                         1).to(self.device)
        self.target = copy.deepcopy(self.module).to(self.device)
        for p in self.target.parameters():
            p.requires_grad_(False)
        self.pi_opt = torch.optim.Adam(self.module.actor.parameters(),
                                       lr=config.lr)
        self.q_opt = torch.optim.Adam(
            list(self.module.q1.parameters())
            + list(self.module.q2.parameters()), lr=config.lr)
        self.v_opt = torch.optim.Adam(self.vnet.parameters(), lr=config.lr)
        self._rng = np.random.default_rng(config.seed)

    def _batch(self):
        n = len(self._data["rewards"])
        idx = self._rng.integers(0, n, min(self.config.train_batch_size, n))
        d = self.device
        return tuple(
            torch.as_tensor(self._data[k][idx], device=d)
            for k in ("obs", "next_obs", "actions", "rewards", "dones")
        )

    def _logp(self, obs, act):
        """log pi(a|s) of the tanh-Gaussian at the DATASET action."""
        out = self.module.actor(obs)
        mean, log_std = out.chunk(2, dim=-1)
        log_std = torch.clamp(log_std, LOG_STD_MIN, LOG_STD_MAX)
        a = torch.clamp(act / self.act_limit, -0.999999, 0.999999)
        u = torch.atanh(a)
        dist = torch.distributions.Normal(mean, log_std.exp())
        logp = dist.log_prob(u).sum(-1)
        logp = logp - (
            2 * (np.log(2) - u - nn.functional.softplus(-2 * u))
        ).sum(-1)
        return logp

    def _update_once(self):
        cfg = self.config
        obs, nobs, act, rew, done = self._batch()
        cont = 1.0 - done

        # V: expectile regression against the frozen twin-Q minimum
        with torch.no_grad():
            tq1, tq2 = self.target.q(obs, act)
            q_min = torch.min(tq1, tq2)
        v = self.vnet(obs).squeeze(-1)
        diff = q_min - v
        w = torch.where(diff > 0, cfg.expectile, 1 - cfg.expectile)
        v_loss = (w * diff ** 2).mean()
        self.v_opt.zero_grad(set_to_none=True)
        v_loss.backward()
        self.v_opt.step()

        # Q: regress to r + gamma V(s') (no actor in the target)
        with torch.no_grad():
            target = rew + cfg.gamma * cont * self.vnet(nobs).squeeze(-1)
        q1, q2 = self.module.q(obs, act)
        q_loss = ((q1 - target) ** 2).mean() + ((q2 - target) ** 2).mean()
        self.q_opt.zero_grad(set_to_none=True)
        q_loss.backward()
        self.q_opt.step()

        # policy: advantage-weighted regression on dataset actions
        with torch.no_grad():
            adv = q_min - self.vnet(obs).squeeze(-1)
            wts = torch.clamp(torch.exp(cfg.awr_beta * adv),
                              max=cfg.awr_weight_clip)
        logp = self._logp(obs, act)
        pi_loss = -(wts * logp).mean()
        self.pi_opt.zero_grad(set_to_none=True)
        pi_loss.backward()
        self.pi_opt.step()

        with torch.no_grad():
            for p, tp in zip(self.module.parameters(),
                             self.target.parameters()):
                tp.mul_(1 - cfg.tau).add_(cfg.tau * p)
        return {
            "v_loss": float(v_loss.detach()),
            "q_loss": float(q_loss.detach()),
            "pi_loss": float(pi_loss.detach()),
            "adv_mean": float(adv.mean()),
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
        return {
            "module": {k: v.cpu().numpy()
                       for k, v in self.module.state_dict().items()},
            "vnet": {k: v.cpu().numpy()
                     for k, v in self.vnet.state_dict().items()},
        }

    def set_weights(self, w):
        self.module.load_state_dict(
            {k: torch.as_tensor(v) for k, v in w["module"].items()})
        self.vnet.load_state_dict(
            {k: torch.as_tensor(v) for k, v in w["vnet"].items()})
        self.target = copy.deepcopy(self.module).to(self.device)
