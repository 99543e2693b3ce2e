"""SAC — Soft Actor-Critic for continuous actions (reference:
rllib/algorithms/sac/ — twin Q critics, tanh-squashed Gaussian policy,
automatic entropy-temperature tuning).

Rollouts run on a local VectorEnv (continuous-action distributed env
runners land with the multi-agent stack); updates follow the standard
SAC losses with Polyak-averaged target critics.
"""
from __future__ import annotations

import copy

import numpy as np
import torch
import torch.nn as nn

from ..algorithm import Algorithm, AlgorithmConfig
from ..env import VectorEnv
from ..replay import ReplayBuffer

LOG_STD_MIN, LOG_STD_MAX = -20.0, 2.0


def _mlp(sizes, out_dim):
    layers = []
    last = sizes[0]
    for h in sizes[1:]:
        layers += [nn.Linear(last, h), nn.ReLU()]
        last = h
    layers.append(nn.Linear(last, out_dim))
    return nn.Sequential(*layers)


class SACModule(nn.Module):
    """Tanh-squashed Gaussian actor + twin Q critics."""

    def __init__(self, obs_dim: int, act_dim: int, act_limit: float,
                 hidden=(256, 256)):
        super().__init__()
        self.act_dim = act_dim
        self.act_limit = act_limit
        self.actor = _mlp((obs_dim,) + tuple(hidden), 2 * act_dim)
        self.q1 = _mlp((obs_dim + act_dim,) + tuple(hidden), 1)
        self.q2 = _mlp((obs_dim + act_dim,) + tuple(hidden), 1)

    def pi(self, obs, deterministic=False):
        """Returns (action, log_prob) with tanh squashing correction."""
        out = self.actor(obs)
        mean, log_std = out.chunk(2, dim=-1)
        log_std = torch.clamp(log_std, LOG_STD_MIN, LOG_STD_MAX)
        std = log_std.exp()
        dist = torch.distributions.Normal(mean, std)
        u = mean if deterministic else dist.rsample()
        logp = dist.log_prob(u).sum(-1)
        # tanh change-of-variables (numerically stable form)
        logp = logp - (2 * (np.log(2) - u - nn.functional.softplus(-2 * u))
                       ).sum(-1)
        a = torch.tanh(u) * self.act_limit
        return a, logp

    def q(self, obs, act):
        x = torch.cat([obs, act], dim=-1)
        return self.q1(x).squeeze(-1), self.q2(x).squeeze(-1)


class SACConfig(AlgorithmConfig):
    def __init__(self):
        super().__init__(algo_class=SAC)
        self.env = "Pendulum-v1"
        self.lr = 3e-4
        self.gamma = 0.99
        self.tau = 0.005                      # Polyak target rate
        self.train_batch_size = 256
        self.rollout_fragment_length = 50     # env steps per iteration/env
        self.replay_buffer_capacity = 100_000
        self.num_steps_sampled_before_learning = 1500
        self.updates_per_iteration = 50
        self.initial_alpha = 0.2
        self.target_entropy = None            # default: -act_dim


class SAC(Algorithm):
    def _setup(self, config: SACConfig):
        self.device = torch.device(
            "cuda:0"
            if config.num_gpus_per_learner > 0 and torch.cuda.is_available()
            else "cpu"
        )
        n_envs = max(1, config.num_envs_per_env_runner *
                     max(1, config.num_env_runners))
        self.vec = VectorEnv(config.env, n_envs, seed=config.seed)
        obs_dim = int(np.prod(self.vec.observation_space.shape))
        self.act_dim = int(np.prod(self.vec.action_space.shape))
        act_limit = float(np.max(np.abs(self.vec.action_space.high)))
        self.module = SACModule(obs_dim, self.act_dim, act_limit,
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
            float(np.log(config.initial_alpha)), device=self.device,
            requires_grad=True)
        self.alpha_opt = torch.optim.Adam([self.log_alpha], lr=config.lr)
        self.target_entropy = (
            config.target_entropy
            if config.target_entropy is not None else -float(self.act_dim))
        self.buffer = ReplayBuffer(config.replay_buffer_capacity)
        self._obs = self.vec.reset()
        self._env_steps = 0
        self._rng = np.random.default_rng(config.seed)

    # ---------------- rollout ----------------

    def _collect(self, n_steps: int):
        for _ in range(n_steps):
            with torch.no_grad():
                ob = torch.as_tensor(self._obs, dtype=torch.float32,
                                     device=self.device)
                if self._env_steps < self.config.num_steps_sampled_before_learning:
                    a = self._rng.uniform(
                        -self.module.act_limit, self.module.act_limit,
                        size=(self.vec.num_envs, self.act_dim),
                    ).astype(np.float32)
                else:
                    a = self.module.pi(ob)[0].cpu().numpy()
            nobs, rew, term, trunc = self.vec.step(a)
            self.buffer.add_batch({
                "obs": self._obs.astype(np.float32),
                "next_obs": nobs.astype(np.float32),
                "actions": a.reshape(self.vec.num_envs, self.act_dim),
                "rewards": rew.astype(np.float32),
                # time-limit truncation must NOT zero the bootstrap
                "dones": term.astype(np.float32),
            })
            self._obs = nobs
            self._env_steps += self.vec.num_envs

    # ---------------- update ----------------

    def _update_once(self):
        cfg = self.config
        b = self.buffer.sample(cfg.train_batch_size, self._rng)
        obs = torch.as_tensor(b["obs"], device=self.device)
        nobs = torch.as_tensor(b["next_obs"], device=self.device)
        act = torch.as_tensor(b["actions"], device=self.device)
        rew = torch.as_tensor(b["rewards"], device=self.device)
        cont = 1.0 - torch.as_tensor(b["dones"], device=self.device)
        alpha = self.log_alpha.exp().detach()

        with torch.no_grad():
            na, nlogp = self.module.pi(nobs)
            tq1, tq2 = self.target.q(nobs, na)
            target = rew + cfg.gamma * cont * (
                torch.min(tq1, tq2) - alpha * nlogp)
        q1, q2 = self.module.q(obs, act)
        q_loss = ((q1 - target) ** 2).mean() + ((q2 - target) ** 2).mean()
        self.q_opt.zero_grad(set_to_none=True)
        q_loss.backward()
        self.q_opt.step()

        for p in self.module.q1.parameters():
            p.requires_grad_(False)
        for p in self.module.q2.parameters():
            p.requires_grad_(False)
        a, logp = self.module.pi(obs)
        qa1, qa2 = self.module.q(obs, a)
        pi_loss = (alpha * logp - torch.min(qa1, qa2)).mean()
        self.pi_opt.zero_grad(set_to_none=True)
        pi_loss.backward()
        self.pi_opt.step()
        for p in self.module.q1.parameters():
            p.requires_grad_(True)
        for p in self.module.q2.parameters():
            p.requires_grad_(True)

        alpha_loss = -(
            self.log_alpha * (logp.detach() + self.target_entropy)
        ).mean()
        self.alpha_opt.zero_grad(set_to_none=True)
        alpha_loss.backward()
        self.alpha_opt.step()

        with torch.no_grad():
            for p, tp in zip(self.module.parameters(),
                             self.target.parameters()):
                tp.mul_(1 - cfg.tau).add_(cfg.tau * p)
        return {
            "q_loss": float(q_loss.detach()),
            "pi_loss": float(pi_loss.detach()),
            "alpha": float(self.log_alpha.exp()),
            "entropy": float(-logp.mean()),
        }

    def training_step(self):
        cfg = self.config
        self._collect(cfg.rollout_fragment_length)
        stats = {}
        if self._env_steps >= cfg.num_steps_sampled_before_learning:
            for _ in range(cfg.updates_per_iteration):
                stats = self._update_once()
        rets, lens = self.vec.pop_episode_stats()
        result = {
            "learner": stats,
            "num_env_steps_sampled": cfg.rollout_fragment_length
            * self.vec.num_envs,
            "num_env_steps_sampled_lifetime": self._env_steps,
            "replay_buffer_size": len(self.buffer),
            "env_runners": {
                "episode_return_mean":
                    float(np.mean(rets)) if rets else None,
                "episode_len_mean":
                    float(np.mean(lens)) if lens else None,
            },
        }
        em = result["env_runners"]["episode_return_mean"]
        if em is not None:
            result["episode_reward_mean"] = em
        return result

    # checkpointing (Algorithm.save/restore use get/set_weights)

    def get_weights(self):
        return {
            "module": {k: v.cpu().numpy()
                       for k, v in self.module.state_dict().items()},
            "log_alpha": float(self.log_alpha.detach()),
            "env_steps": self._env_steps,
        }

    def set_weights(self, state):
        self.module.load_state_dict(
            {k: torch.as_tensor(v) for k, v in state["module"].items()}
        )
        self.target = copy.deepcopy(self.module).to(self.device)
        with torch.no_grad():
            self.log_alpha.fill_(state["log_alpha"])
        self._env_steps = state["env_steps"]
