"""Multi-agent RL (reference: rllib/env/multi_agent_env.py MultiAgentEnv,
rllib/env/multi_agent_env_runner.py, config.multi_agent(policies=...,
policy_mapping_fn=...)).

Scope: dict-based MultiAgentEnv API, a built-in MultiAgentCartPole,
and MultiAgentPPO — one PPO policy per policy id, agents mapped by
`policy_mapping_fn`, per-policy [T, B] batches fed to the same
PPOTorchLearner (HIP GAE) the single-agent path uses.
"""
from __future__ import annotations

from typing import Callable, Dict, List, Optional

import numpy as np

from .algorithm import Algorithm
from .algorithms.ppo import PPOConfig
from .env import CartPoleEnv, make_env
from .learner import PPOTorchLearner


class MultiAgentEnv:
    """Dict-keyed env API: reset() -> (obs_dict, infos); step(actions)
    -> (obs, rewards, terminateds, truncateds, infos) — all dicts keyed
    by agent id, with terminateds["__all__"] ending the episode."""

    agents: List[str] = []
    possible_agents: List[str] = []

    def reset(self, *, seed=None):
        raise NotImplementedError

    def step(self, action_dict: Dict):
        raise NotImplementedError


class MultiAgentCartPole(MultiAgentEnv):
    """N independent CartPoles, one per agent (reference:
    rllib/examples/envs/classes/multi_agent/__init__.py)."""

    def __init__(self, config: Optional[dict] = None):
        config = config or {}
        n = int(config.get("num_agents", 2))
        self.agents = [f"agent_{i}" for i in range(n)]
        self.possible_agents = list(self.agents)
        seed = config.get("seed")
        self._envs = {
            a: CartPoleEnv(seed=None if seed is None else seed + i)
            for i, a in enumerate(self.agents)
        }
        self._done: Dict[str, bool] = {}
        self.observation_space = self._envs[self.agents[0]].observation_space
        self.action_space = self._envs[self.agents[0]].action_space

    def reset(self, *, seed=None):
        self._done = {a: False for a in self.agents}
        obs = {a: e.reset(seed=seed)[0] for a, e in self._envs.items()}
        return obs, {a: {} for a in self.agents}

    def step(self, action_dict: Dict):
        obs, rew, term, trunc = {}, {}, {}, {}
        for a, act in action_dict.items():
            if self._done.get(a, True):
                continue
            o, r, t, tr, _ = self._envs[a].step(int(act))
            obs[a], rew[a], term[a], trunc[a] = o, r, t, tr
            if t or tr:
                self._done[a] = True
        term["__all__"] = all(self._done.values())
        trunc["__all__"] = False
        return obs, rew, term, trunc, {a: {} for a in obs}


class MultiAgentPPOConfig(PPOConfig):
    def __init__(self):
        super().__init__()
        self.algo_class = MultiAgentPPO
        self.policies: List[str] = ["default_policy"]
        self.policy_mapping_fn: Callable = (
            lambda agent_id, *a, **k: "default_policy"
        )

    def multi_agent(self, *, policies=None, policy_mapping_fn=None,
                    **kwargs):
        if policies is not None:
            self.policies = sorted(policies)
        if policy_mapping_fn is not None:
            self.policy_mapping_fn = policy_mapping_fn
        return self


class MultiAgentPPO(Algorithm):
    """PPO over a MultiAgentEnv: fixed agent→policy mapping, one
    learner per policy, per-agent auto-reset (vectorized rollout)."""

    def _setup(self, config: MultiAgentPPOConfig):
        n_envs = max(1, config.num_envs_per_env_runner)
        self._envs = [self._make(config) for _ in range(n_envs)]
        probe = self._envs[0]
        self.agents = list(probe.possible_agents)
        obs_dim = int(np.prod(probe.observation_space.shape))
        num_actions = probe.action_space.n
        self.policy_of = {
            a: config.policy_mapping_fn(a) for a in self.agents
        }
        self.learners: Dict[str, PPOTorchLearner] = {
            pid: PPOTorchLearner(
                obs_dim, num_actions, hidden=config.model_hidden,
                lr=config.lr, gamma=config.gamma, lambda_=config.lambda_,
                clip_param=config.clip_param,
                num_epochs=config.num_epochs,
                minibatch_size=config.minibatch_size,
                use_gpu=config.num_gpus_per_learner > 0,
            )
            for pid in config.policies
        }
        self._obs = [e.reset()[0] for e in self._envs]
        self._ep_ret = [dict.fromkeys(self.agents, 0.0)
                        for _ in self._envs]
        self._completed: List[float] = []
        self._env_steps = 0

    @staticmethod
    def _make(config):
        env = config.env
        if isinstance(env, str):
            return make_env(env)
        if isinstance(env, type):
            return env(getattr(config, "env_config", {}) or {})
        return env(getattr(config, "env_config", {}) or {})

    def training_step(self):
        cfg = self.config
        T = max(1, cfg.train_batch_size
                // (len(self._envs) * len(self.agents)))
        E = len(self._envs)
        A = self.agents
        obs_buf = {a: np.zeros((T + 1, E) + self._envs[0].observation_space.shape,
                               np.float32) for a in A}
        act_buf = {a: np.zeros((T, E), np.int64) for a in A}
        rew_buf = {a: np.zeros((T, E), np.float32) for a in A}
        done_buf = {a: np.zeros((T, E), np.float32) for a in A}
        logp_buf = {a: np.zeros((T, E), np.float32) for a in A}
        vf_buf = {a: np.zeros((T + 1, E), np.float32) for a in A}

        for t in range(T + 1):
            # batched forward per policy across envs
            for a in A:
                pid = self.policy_of[a]
                obs_a = np.stack([self._obs[e][a] for e in range(E)])
                act, logp, vf = self.learners[pid].raw_module.\
                    forward_exploration(obs_a)
                obs_buf[a][t] = obs_a
                vf_buf[a][t] = vf
                if t < T:
                    act_buf[a][t] = act
                    logp_buf[a][t] = logp
            if t == T:
                break
            for e in range(E):
                actions = {a: int(act_buf[a][t][e]) for a in A}
                obs, rew, term, trunc, _ = self._envs[e].step(actions)
                for a in A:
                    r = rew.get(a, 0.0)
                    self._ep_ret[e][a] += r
                    rew_buf[a][t][e] = r
                    d = term.get(a, True) or trunc.get(a, False)
                    done_buf[a][t][e] = float(d)
                    if a in obs:
                        self._obs[e][a] = obs[a]
                if term.get("__all__") or trunc.get("__all__"):
                    self._completed.extend(self._ep_ret[e].values())
                    self._ep_ret[e] = dict.fromkeys(A, 0.0)
                    self._obs[e] = self._envs[e].reset()[0]
            self._env_steps += E * len(A)

        # per-policy update over its agents' batches
        stats = {}
        for pid, learner in self.learners.items():
            samples = [
                {
                    "obs": obs_buf[a][:T], "actions": act_buf[a],
                    "rewards": rew_buf[a], "dones": done_buf[a],
                    "logp": logp_buf[a], "vf": vf_buf[a],
                }
                for a in A if self.policy_of[a] == pid
            ]
            if samples:
                stats[pid] = learner.update(samples)
        comp = self._completed[-100:]
        result = {
            "learner": stats,
            "num_env_steps_sampled_lifetime": self._env_steps,
            "episode_reward_mean":
                float(np.mean(comp)) if comp else None,
            "env_runners": {
                "agent_episode_returns_mean":
                    float(np.mean(comp)) if comp else None,
            },
        }
        return result

    def get_weights(self):
        return {pid: l.get_weights() for pid, l in self.learners.items()}

    def set_weights(self, w):
        for pid, lw in w.items():
            self.learners[pid].set_weights(lw)
