"""TQC — Truncated Quantile Critics (reference: rllib/algorithms/
tqc-class continuous control; Kuznetsov et al. 2020): SAC whose
critics are distributional — M critics each predict N return
quantiles; the TD target pools all M*N next-state quantiles, sorts
them, and DROPS the top `top_quantiles_to_drop_per_net * M` to fight
overestimation. Critic loss is the quantile Huber loss.
"""
from __future__ import annotations

import copy

import numpy as np
import torch
import torch.nn as nn

from ..algorithm import Algorithm, AlgorithmConfig
from ..env import VectorEnv
from ..replay import ReplayBuffer
from .sac import LOG_STD_MAX, LOG_STD_MIN, _mlp  # noqa: F401


class TQCModule(nn.Module):
    """Tanh-Gaussian actor + M quantile critics of N quantiles each."""

    def __init__(self, obs_dim, act_dim, act_limit, hidden=(256, 256),
                 n_critics=2, n_quantiles=25):
        super().__init__()
        self.act_dim = act_dim
        self.act_limit = act_limit
        self.n_critics = n_critics
        self.n_quantiles = n_quantiles
        self.actor = _mlp((obs_dim,) + tuple(hidden), 2 * act_dim)
        self.critics = nn.ModuleList(
            _mlp((obs_dim + act_dim,) + tuple(hidden), n_quantiles)
            for _ in range(n_critics))

    def pi(self, obs, deterministic=False):
        out = self.actor(obs)
        mean, log_std = out.chunk(2, dim=-1)
        log_std = torch.clamp(log_std, LOG_STD_MIN, LOG_STD_MAX)
        dist = torch.distributions.Normal(mean, log_std.exp())
        u = mean if deterministic else dist.rsample()
        logp = dist.log_prob(u).sum(-1)
        logp = logp - (2 * (np.log(2) - u -
                            nn.functional.softplus(-2 * u))).sum(-1)
        return torch.tanh(u) * self.act_limit, logp

    def quantiles(self, obs, act):
        """[B, M, N] return quantiles."""
        x = torch.cat([obs, act], dim=-1)
        return torch.stack([c(x) for c in self.critics], dim=1)


class TQCConfig(AlgorithmConfig):
    def __init__(self):
        super().__init__(algo_class=TQC)
        self.env = "Pendulum-v1"
        self.lr = 3e-4
        self.gamma = 0.99
        self.tau = 0.005
        self.train_batch_size = 256
        self.rollout_fragment_length = 50
        self.replay_buffer_capacity = 100_000
        self.num_steps_sampled_before_learning = 1500
        self.updates_per_iteration = 50
        self.initial_alpha = 0.2
        self.target_entropy = None
        self.n_critics = 2
        self.n_quantiles = 25
        self.top_quantiles_to_drop_per_net = 2


class TQC(Algorithm):
    def _setup(self, config: TQCConfig):
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
        self.module = TQCModule(
            obs_dim, self.act_dim, act_limit, config.model_hidden,
            config.n_critics, config.n_quantiles).to(self.device)
        self.target = copy.deepcopy(self.module).to(self.device)
        for p in self.target.parameters():
            p.requires_grad_(False)
        self.pi_opt = torch.optim.Adam(self.module.actor.parameters(),
                                       lr=config.lr)
        self.q_opt = torch.optim.Adam(self.module.critics.parameters(),
                                      lr=config.lr)
        self.log_alpha = torch.tensor(
            float(np.log(config.initial_alpha)), device=self.device,
            requires_grad=True)
        self.alpha_opt = torch.optim.Adam([self.log_alpha], lr=config.lr)
        self.target_entropy = (
            config.target_entropy
            if config.target_entropy is not None else -float(self.act_dim))
        # quantile midpoints tau_hat for the Huber loss
        n = config.n_quantiles
        self.tau_hat = ((torch.arange(n, device=self.device,
                                      dtype=torch.float32) + 0.5) / n)
        self.buffer = ReplayBuffer(config.replay_buffer_capacity)
        self._obs = self.vec.reset()
        self._env_steps = 0
        self._rng = np.random.default_rng(config.seed)

    def _collect(self, n_steps: int):
        for _ in range(n_steps):
            with torch.no_grad():
                ob = torch.as_tensor(self._obs, dtype=torch.float32,
                                     device=self.device)
                if (self._env_steps
                        < self.config.num_steps_sampled_before_learning):
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
                "dones": term.astype(np.float32),
            })
            self._obs = nobs
            self._env_steps += self.vec.num_envs

    def _update_once(self):
        cfg = self.config
        b = self.buffer.sample(cfg.train_batch_size, self._rng)
        obs = torch.as_tensor(b["obs"], device=self.device)
        nobs = torch.as_tensor(b["next_obs"], device=self.device)
        act = torch.as_tensor(b["actions"], device=self.device)
        rew = torch.as_tensor(b["rewards"], device=self.device)
        cont = 1.0 - torch.as_tensor(b["dones"], device=self.device)
        alpha = self.log_alpha.exp().detach()
        M, N = cfg.n_critics, cfg.n_quantiles

        # --- critic: truncated pooled quantile target ---
        with torch.no_grad():
            na, nlogp = self.module.pi(nobs)
            nz = self.target.quantiles(nobs, na).reshape(-1, M * N)
            nz, _ = torch.sort(nz, dim=-1)
            keep = M * N - cfg.top_quantiles_to_drop_per_net * M
            nz = nz[:, :keep]
            y = rew[:, None] + cfg.gamma * cont[:, None] * (
                nz - alpha * nlogp[:, None])         # [B, keep]
        z = self.module.quantiles(obs, act)           # [B, M, N]
        # quantile Huber loss over every (predicted, target) pair
        diff = y[:, None, None, :] - z[:, :, :, None]  # [B,M,N,keep]
        abs_diff = diff.abs()
        huber = torch.where(abs_diff <= 1.0, 0.5 * diff ** 2,
                            abs_diff - 0.5)
        q_loss = (torch.abs(self.tau_hat[None, None, :, None]
                            - (diff.detach() < 0).float())
                  * huber).mean()
        self.q_opt.zero_grad(set_to_none=True)
        q_loss.backward()
        self.q_opt.step()

        # --- actor: maximize mean of ALL quantiles - alpha*logp ---
        for p in self.module.critics.parameters():
            p.requires_grad_(False)
        a, logp = self.module.pi(obs)
        qmean = self.module.quantiles(obs, a).mean(dim=(1, 2))
        pi_loss = (alpha * logp - qmean).mean()
        self.pi_opt.zero_grad(set_to_none=True)
        pi_loss.backward()
        self.pi_opt.step()
        for p in self.module.critics.parameters():
            p.requires_grad_(True)

        # --- temperature ---
        a_loss = -(self.log_alpha
                   * (logp.detach() + self.target_entropy)).mean()
        self.alpha_opt.zero_grad(set_to_none=True)
        a_loss.backward()
        self.alpha_opt.step()

        with torch.no_grad():
            for p, tp in zip(self.module.parameters(),
                             self.target.parameters()):
                tp.mul_(1 - cfg.tau).add_(cfg.tau * p)
        return {
            "q_loss": float(q_loss.detach()),
            "pi_loss": float(pi_loss.detach()),
            "alpha": float(alpha),
        }

    def training_step(self):
        cfg = self.config
        self._collect(cfg.rollout_fragment_length)
        stats = {}
        if self._env_steps >= cfg.num_steps_sampled_before_learning:
            for _ in range(cfg.updates_per_iteration):
                stats = self._update_once()
        return {"learner": stats, "env_steps": self._env_steps}

    def evaluate(self, num_steps: int = 500, num_envs: int = 4):
        vec = VectorEnv(self.config.env, num_envs,
                        seed=self.config.seed + 1)
        obs = vec.reset()
        for _ in range(num_steps // num_envs):
            with torch.no_grad():
                a = self.module.pi(
                    torch.as_tensor(obs, dtype=torch.float32,
                                    device=self.device),
                    deterministic=True)[0].cpu().numpy()
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
