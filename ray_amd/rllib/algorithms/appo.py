"""APPO — async PPO (reference: rllib/algorithms/appo/): the IMPALA
substrate (V-trace off-policy correction over async env runners) plus
a PPO-clipped surrogate against a Polyak-averaged TARGET policy's
importance ratios and a KL penalty toward the target."""
from __future__ import annotations

import copy

import numpy as np
import torch

from ..algorithm import Algorithm, AlgorithmConfig, summarize_episodes
from ..env import VectorEnv
from ..env_runner import EnvRunnerGroup
from ..learner import ImpalaTorchLearner
from ... import ops


class APPOConfig(AlgorithmConfig):
    def __init__(self):
        super().__init__(algo_class=APPO)
        self.lr = 5e-4
        self.entropy_coeff = 0.01
        self.clip_param = 0.2
        self.kl_coeff = 0.2
        self.target_update_freq = 4   # learner updates per target sync
        self.tau = 1.0                # hard target copy (reference default)


class _AppoLearner(ImpalaTorchLearner):
    def __init__(self, *a, clip_param=0.2, kl_coeff=0.2, **kw):
        super().__init__(*a, **kw)
        self.clip = clip_param
        self.kl_coeff = kl_coeff
        self.target = copy.deepcopy(self.module)
        for p in self.target.parameters():
            p.requires_grad_(False)
        self._updates = 0

    def sync_target(self):
        self.target.load_state_dict(self.module.state_dict())

    def update(self, samples):
        batch = {
            k: np.concatenate([s[k] for s in samples], axis=1)
            for k in ("obs", "actions", "rewards", "dones", "logp", "vf")
        }
        T, B = batch["rewards"].shape
        obs = torch.as_tensor(batch["obs"], device=self.device)
        actions = torch.as_tensor(batch["actions"], device=self.device)
        behav_logp = torch.as_tensor(batch["logp"], device=self.device)
        rewards = torch.as_tensor(batch["rewards"], device=self.device)
        cont = 1.0 - torch.as_tensor(batch["dones"], device=self.device)

        out = self.module(obs.reshape(T * B, -1))
        dist = torch.distributions.Categorical(
            logits=out["logits"].reshape(T, B, -1)
        )
        tgt_logp = dist.log_prob(actions)
        values = out["vf"].reshape(T, B)
        with torch.no_grad():
            t_out = self.target(obs.reshape(T * B, -1))
            t_dist = torch.distributions.Categorical(
                logits=t_out["logits"].reshape(T, B, -1)
            )
            old_logp = t_dist.log_prob(actions)
            # V-trace targets/advantages w.r.t. the BEHAVIOUR policy
            log_rhos = (tgt_logp - behav_logp).detach()
            values_tb1 = torch.cat(
                [values.detach(),
                 torch.as_tensor(batch["vf"][-1:], device=self.device)], 0
            )
            vs, pg_adv = ops.vtrace(
                log_rhos, rewards, values_tb1, cont, self.gamma
            )
        # PPO clip against the target policy's log-probs
        ratio = (tgt_logp - old_logp).exp()
        surr = torch.min(
            ratio * pg_adv,
            ratio.clamp(1 - self.clip, 1 + self.clip) * pg_adv,
        )
        pi_loss = -surr.mean()
        kl = torch.distributions.kl_divergence(t_dist, dist).mean()
        vf_loss = (values - vs).pow(2).mean()
        entropy = dist.entropy().mean()
        loss = (pi_loss + self.vf_coeff * vf_loss
                - self.ent_coeff * entropy + self.kl_coeff * kl)
        self.opt.zero_grad()
        loss.backward()
        torch.nn.utils.clip_grad_norm_(self.module.parameters(), 40.0)
        self.opt.step()
        self._updates += 1
        return {
            "policy_loss": float(pi_loss.detach()),
            "vf_loss": float(vf_loss.detach()),
            "entropy": float(entropy.detach()),
            "kl": float(kl.detach()),
            "total_loss": float(loss.detach()),
        }


class APPO(Algorithm):
    def _setup(self, config: APPOConfig):
        probe = VectorEnv(config.env, 1, seed=config.seed)
        obs_dim = int(np.prod(probe.observation_space.shape))
        num_actions = probe.action_space.n
        self.env_runner_group = EnvRunnerGroup(
            config.env, config.num_env_runners,
            config.num_envs_per_env_runner, hidden=config.model_hidden,
        )
        self.learner = _AppoLearner(
            obs_dim, num_actions, hidden=config.model_hidden,
            lr=config.lr, gamma=config.gamma,
            ent_coeff=config.entropy_coeff,
            use_gpu=config.num_gpus_per_learner > 0,
            clip_param=config.clip_param, kl_coeff=config.kl_coeff,
        )
        self._env_steps_total = 0
        self.env_runner_group.sync_weights(self.learner.get_weights())

    def training_step(self):
        cfg = self.config
        n_runners = max(1, cfg.num_env_runners)
        steps_per_runner = max(
            1, cfg.train_batch_size
            // (n_runners * cfg.num_envs_per_env_runner),
        )
        samples = self.env_runner_group.sample(steps_per_runner)
        env_steps = sum(s["rewards"].size for s in samples)
        self._env_steps_total += env_steps
        stats = self.learner.update(samples)
        if self.learner._updates % cfg.target_update_freq == 0:
            self.learner.sync_target()
        self.env_runner_group.sync_weights(self.learner.get_weights())
        result = {
            "env_runners": summarize_episodes(samples),
            "learner": stats,
            "num_env_steps_sampled": env_steps,
            "num_env_steps_sampled_lifetime": self._env_steps_total,
        }
        em = result["env_runners"].get("episode_return_mean")
        if em is not None:
            result["episode_reward_mean"] = em
        return result

    def get_weights(self):
        return self.learner.get_weights()

    def set_weights(self, w):
        self.learner.set_weights(w)
        self.learner.sync_target()
        self.env_runner_group.sync_weights(w)

    def stop(self):
        self.env_runner_group.stop()
