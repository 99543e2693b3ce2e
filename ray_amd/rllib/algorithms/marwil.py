"""MARWIL + BC — offline imitation (reference: rllib/algorithms/
marwil/ — Monotonic Advantage Re-Weighted Imitation Learning, Wang et
al. 2018 — and rllib/algorithms/bc/, which the reference implements as
MARWIL with beta=0: plain behavioral cloning).

Continuous control on the offline transition schema shared with
CQL/IQL (`cql._load_transitions`): maximize exp(beta * A(s,a)) *
log pi(a|s) with a learned value baseline; beta=0 drops the critic and
reduces to pure BC.
"""
from __future__ import annotations

import numpy as np
import torch
import torch.nn as nn

from ..algorithm import Algorithm, AlgorithmConfig
from ..env import VectorEnv
from .cql import _load_transitions
from .sac import LOG_STD_MAX, LOG_STD_MIN, _mlp


class MARWILConfig(AlgorithmConfig):
    def __init__(self):
        super().__init__(algo_class=MARWIL)
        self.env = "Pendulum-v1"
        self.lr = 3e-4
        self.gamma = 0.99
        self.train_batch_size = 256
        self.updates_per_iteration = 50
        self.input_ = None
        self.beta = 1.0               # 0 => behavioral cloning
        self.vf_coeff = 1.0
        self.moving_average_sqd_adv_norm_update_rate = 1e-7
        self.advantage_clip = 10.0

    def offline_data(self, *, input_=None, **kwargs):
        if input_ is not None:
            self.input_ = input_
        return self


class MARWIL(Algorithm):
    def _setup(self, config: MARWILConfig):
        from ... import data as ray_data

        ds = config.input_
        if isinstance(ds, str):
            ds = ray_data.read_parquet(ds)
        if ds is None:
            raise ValueError("MARWILConfig.offline_data(input_=...) required")
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
        hidden = tuple(config.model_hidden)
        self.actor = _mlp((obs_dim,) + hidden, 2 * self.act_dim).to(
            self.device)
        self.vnet = _mlp((obs_dim,) + hidden, 1).to(self.device)
        params = list(self.actor.parameters())
        if config.beta:
            params += list(self.vnet.parameters())
        self.opt = torch.optim.Adam(params, lr=config.lr)
        # running normalizer of squared advantages (reference:
        # marwil_torch_learner.py moving avg sqd adv norm)
        self._ma_sqd_adv = torch.tensor(1.0, device=self.device)
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
        out = self.actor(obs)
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

        if cfg.beta:
            v = self.vnet(obs).squeeze(-1)
            with torch.no_grad():
                v_next = self.vnet(nobs).squeeze(-1)
                td_target = rew + cfg.gamma * cont * v_next
                adv = td_target - v
                # normalize by the running sqd-advantage average
                self._ma_sqd_adv += (
                    cfg.moving_average_sqd_adv_norm_update_rate
                    * ((adv ** 2).mean() - self._ma_sqd_adv))
                norm_adv = adv / torch.sqrt(self._ma_sqd_adv + 1e-8)
                wts = torch.exp(torch.clamp(cfg.beta * norm_adv,
                                            max=np.log(cfg.advantage_clip)))
            v_loss = ((td_target.detach() - v) ** 2).mean()
        else:
            wts = torch.ones_like(rew)
            v_loss = torch.zeros((), device=self.device)

        logp = self._logp(obs, act)
        pi_loss = -(wts * logp).mean()
        loss = pi_loss + cfg.vf_coeff * v_loss
        self.opt.zero_grad(set_to_none=True)
        loss.backward()
        self.opt.step()
        return {
            "pi_loss": float(pi_loss.detach()),
            "v_loss": float(v_loss.detach()),
            "mean_weight": float(wts.mean()),
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
                out = self.actor(torch.as_tensor(
                    obs, dtype=torch.float32, device=self.device))
                mean, _ = out.chunk(2, dim=-1)
                a = (torch.tanh(mean) * self.act_limit).cpu().numpy()
            obs, _, _, _ = vec.step(a)
        rets, _ = vec.pop_episode_stats()
        return {"episode_reward_mean":
                float(np.mean(rets)) if len(rets) else None}

    def get_weights(self):
        return {
            "actor": {k: v.cpu().numpy()
                      for k, v in self.actor.state_dict().items()},
            "vnet": {k: v.cpu().numpy()
                     for k, v in self.vnet.state_dict().items()},
        }

    def set_weights(self, w):
        self.actor.load_state_dict(
            {k: torch.as_tensor(v) for k, v in w["actor"].items()})
        self.vnet.load_state_dict(
            {k: torch.as_tensor(v) for k, v in w["vnet"].items()})


class BCConfig(MARWILConfig):
    """Behavioral cloning = MARWIL with beta=0 (reference:
    rllib/algorithms/bc/bc.py)."""

    def __init__(self):
        super().__init__()
        self.algo_class = BC
        self.beta = 0.0


class BC(MARWIL):
    pass
