"""IMPALA (reference: rllib/algorithms/impala/impala.py — async
EnvRunners feeding V-trace learners; V-trace scan = HIP kernel)."""
from __future__ import annotations

import numpy as np

from ..algorithm import Algorithm, AlgorithmConfig, summarize_episodes
from ..env import VectorEnv
from ..env_runner import EnvRunnerGroup
from ..learner import ImpalaTorchLearner


class IMPALAConfig(AlgorithmConfig):
    def __init__(self):
        super().__init__(algo_class=IMPALA)
        self.lr = 5e-4
        self.entropy_coeff = 0.01


class IMPALA(Algorithm):
    def _setup(self, config):
        probe = VectorEnv(config.env, 1, seed=config.seed)
        obs_dim = int(np.prod(probe.observation_space.shape))
        num_actions = probe.action_space.n
        self.env_runner_group = EnvRunnerGroup(
            config.env, config.num_env_runners, config.num_envs_per_env_runner,
            hidden=config.model_hidden,
        )
        self.learner = ImpalaTorchLearner(
            obs_dim, num_actions, hidden=config.model_hidden, lr=config.lr,
            gamma=config.gamma, ent_coeff=config.entropy_coeff,
            use_gpu=config.num_gpus_per_learner > 0,
        )
        self._pending = []  # async sample refs (off-policy by design)
        self._env_steps_total = 0
        self.env_runner_group.sync_weights(self.learner.get_weights())

    def training_step(self):
        import time

        cfg = self.config
        n_runners = max(1, cfg.num_env_runners)
        steps_per_runner = max(
            1,
            cfg.train_batch_size // (n_runners * cfg.num_envs_per_env_runner),
        )
        t0 = time.time()
        samples = self.env_runner_group.sample(steps_per_runner)
        env_steps = sum(s["rewards"].size for s in samples)
        self._env_steps_total += env_steps
        stats = self.learner.update(samples)
        self.env_runner_group.sync_weights(self.learner.get_weights())
        dt = time.time() - t0
        result = {
            "env_runners": summarize_episodes(samples),
            "learner": stats,
            "num_env_steps_sampled": env_steps,
            "num_env_steps_sampled_lifetime": self._env_steps_total,
            "env_steps_per_sec": env_steps / max(dt, 1e-9),
        }
        em = result["env_runners"].get("episode_return_mean")
        if em is not None:
            result["episode_reward_mean"] = em
        return result

    def get_weights(self):
        return self.learner.get_weights()

    def set_weights(self, w):
        self.learner.set_weights(w)
        self.env_runner_group.sync_weights(w)

    def stop(self):
        self.env_runner_group.stop()
