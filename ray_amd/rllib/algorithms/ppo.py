"""PPO (reference: rllib/algorithms/ppo/ppo.py:365; training_step :391:
synchronous_parallel_sample -> learner update (minibatch epochs) ->
env_runner_group.sync_weights)."""
from __future__ import annotations

import numpy as np

from ..algorithm import Algorithm, AlgorithmConfig, summarize_episodes
from ..env import VectorEnv
from ..env_runner import EnvRunnerGroup
from ..learner import PPOTorchLearner


class PPOConfig(AlgorithmConfig):
    def __init__(self):
        super().__init__(algo_class=PPO)


class PPO(Algorithm):
    def _setup(self, config: PPOConfig):
        probe = VectorEnv(config.env, 1, seed=config.seed)
        obs_dim = int(np.prod(probe.observation_space.shape))
        num_actions = probe.action_space.n
        self.env_runner_group = EnvRunnerGroup(
            config.env,
            config.num_env_runners,
            config.num_envs_per_env_runner,
            hidden=config.model_hidden,
            env_to_module_connector=getattr(
                config, "env_to_module_connector", None),
        )
        self.learner = PPOTorchLearner(
            obs_dim,
            num_actions,
            hidden=config.model_hidden,
            lr=config.lr,
            gamma=config.gamma,
            lambda_=config.lambda_,
            clip_param=config.clip_param,
            vf_clip_param=config.vf_clip_param,
            vf_loss_coeff=config.vf_loss_coeff,
            entropy_coeff=config.entropy_coeff,
            num_epochs=config.num_epochs,
            minibatch_size=config.minibatch_size,
            use_gpu=config.num_gpus_per_learner > 0,
            learner_connector=getattr(config, "learner_connector", None),
        )
        self._env_steps_total = 0
        self.env_runner_group.sync_weights(self.learner.get_weights())

    def training_step(self):
        import time

        cfg = self.config
        n_runners = max(1, cfg.num_env_runners)
        per_runner_envs = cfg.num_envs_per_env_runner
        steps_per_runner = max(
            1, cfg.train_batch_size // (n_runners * per_runner_envs)
        )
        t0 = time.time()
        samples = self.env_runner_group.sample(steps_per_runner)
        t_sample = time.time() - t0
        env_steps = sum(s["rewards"].size for s in samples)
        self._env_steps_total += env_steps
        t0 = time.time()
        stats = self.learner.update(samples)
        t_learn = time.time() - t0
        self.env_runner_group.sync_weights(self.learner.get_weights())
        result = {
            "env_runners": summarize_episodes(samples),
            "learner": stats,
            "num_env_steps_sampled": env_steps,
            "num_env_steps_sampled_lifetime": self._env_steps_total,
            "env_steps_per_sec": env_steps / max(t_sample + t_learn, 1e-9),
            "time_sample_s": t_sample,
            "time_learn_s": t_learn,
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
