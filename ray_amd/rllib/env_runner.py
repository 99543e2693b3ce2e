"""EnvRunner — sampling actor (reference:
rllib/env/single_agent_env_runner.py:68, sample :152; EnvRunnerGroup
env/env_runner_group.py:70 with sync_weights :695 and fault-tolerant
foreach :913)."""
from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np

from .core import build_module_for_env
from .env import VectorEnv


class SingleAgentEnvRunner:
    """Runs vectorized envs with an inference-only module copy."""

    def __init__(self, env_name: str, num_envs: int = 8, seed: int = 0,
                 hidden=(256, 256), env_to_module=None):
        self.vec = VectorEnv(env_name, num_envs, seed)
        self.module = build_module_for_env(self.vec, hidden=hidden)
        self.obs = self.vec.reset()
        # env->module connector pipeline (reference: rllib/connectors):
        # transforms the observation batch before every inference
        self.env_to_module = env_to_module

    def _transform_obs(self, obs):
        if self.env_to_module is None:
            return obs
        return self.env_to_module({"obs": obs})["obs"]

    def sample(self, num_steps: int) -> Dict[str, np.ndarray]:
        """Rollout num_steps per env; returns [T, B, ...] arrays."""
        T = num_steps
        B = self.vec.num_envs
        obs_buf = np.zeros((T, B) + self.vec.observation_space.shape, np.float32)
        act_buf = np.zeros((T, B), np.int64)
        rew_buf = np.zeros((T, B), np.float32)
        done_buf = np.zeros((T, B), np.float32)
        logp_buf = np.zeros((T, B), np.float32)
        vf_buf = np.zeros((T + 1, B), np.float32)
        obs = self.obs
        for t in range(T):
            a, logp, vf = self.module.forward_exploration(
                self._transform_obs(obs))
            obs_buf[t] = obs
            act_buf[t] = a
            logp_buf[t] = logp
            vf_buf[t] = vf
            obs, r, term, trunc = self.vec.step(a)
            rew_buf[t] = r
            done_buf[t] = np.logical_or(term, trunc).astype(np.float32)
        # bootstrap value
        _, _, vf = self.module.forward_exploration(
            self._transform_obs(obs))
        vf_buf[T] = vf
        self.obs = obs
        rets, lens = self.vec.pop_episode_stats()
        return {
            "obs": obs_buf,
            "actions": act_buf,
            "rewards": rew_buf,
            "dones": done_buf,
            "logp": logp_buf,
            "vf": vf_buf,
            "episode_returns": np.asarray(rets, np.float32),
            "episode_lens": np.asarray(lens, np.int64),
        }

    def set_weights(self, weights):
        self.module.set_weights(weights)

    def get_weights(self):
        return self.module.get_weights()

    def ping(self):
        return "pong"


class EnvRunnerGroup:
    """Manages N EnvRunner actors (+ a local runner when N==0)."""

    def __init__(self, env_name: str, num_runners: int, num_envs_per_runner: int,
                 hidden=(256, 256), env_to_module_connector=None):
        import ray_amd as ray

        self._ray = ray
        self.num_runners = num_runners
        e2m = (env_to_module_connector()
               if env_to_module_connector is not None else None)
        if num_runners == 0:
            self.local = SingleAgentEnvRunner(env_name, num_envs_per_runner,
                                              hidden=hidden,
                                              env_to_module=e2m)
            self.remotes: List = []
        else:
            self.local = None
            cls = ray.remote(SingleAgentEnvRunner)
            self.remotes = [
                cls.options(num_cpus=1).remote(
                    env_name, num_envs_per_runner, seed=1000 * i,
                    hidden=hidden, env_to_module=e2m,
                )
                for i in range(num_runners)
            ]

    def sample(self, num_steps_per_runner: int) -> List[Dict[str, np.ndarray]]:
        if self.local is not None:
            return [self.local.sample(num_steps_per_runner)]
        refs = [r.sample.remote(num_steps_per_runner) for r in self.remotes]
        return self._ray.get(refs, timeout=300)

    def sync_weights(self, weights: Optional[dict] = None, from_module=None):
        if weights is None and from_module is not None:
            weights = from_module.get_weights()
        if self.local is not None:
            self.local.set_weights(weights)
            return
        ref = self._ray.put(weights)
        self._ray.get([r.set_weights.remote(ref) for r in self.remotes],
                      timeout=120)

    def stop(self):
        for r in self.remotes:
            try:
                self._ray.kill(r)
            except Exception:
                pass
