"""DQN (reference: rllib/algorithms/dqn/) — replay buffer + target
network + epsilon-greedy exploration over the EnvRunner substrate."""
from __future__ import annotations

import copy

import numpy as np
import torch

from ..algorithm import Algorithm, AlgorithmConfig, summarize_episodes
from ..core import TorchRLModule
from ..env import VectorEnv
from ..env_runner import EnvRunnerGroup
from ..replay import ReplayBuffer


class DQNConfig(AlgorithmConfig):
    def __init__(self):
        super().__init__(algo_class=DQN)
        self.lr = 5e-4
        self.train_batch_size = 32
        self.rollout_fragment_length = 4
        self.replay_buffer_capacity = 50_000
        self.num_steps_sampled_before_learning = 1000
        self.target_network_update_freq = 500
        self.epsilon_start = 1.0
        self.epsilon_end = 0.05
        self.epsilon_decay_steps = 10_000
        self.updates_per_iteration = 64


class DQN(Algorithm):
    def _setup(self, config: DQNConfig):
        probe = VectorEnv(config.env, 1, seed=config.seed)
        obs_dim = int(np.prod(probe.observation_space.shape))
        self.num_actions = probe.action_space.n
        self.device = torch.device(
            "cuda:0"
            if config.num_gpus_per_learner > 0 and torch.cuda.is_available()
            else "cpu"
        )
        self.q_net = TorchRLModule(obs_dim, self.num_actions,
                                   hidden=config.model_hidden,
                                   device=self.device)
        self.target_net = copy.deepcopy(self.q_net)
        self.opt = torch.optim.Adam(self.q_net.parameters(), lr=config.lr)
        self.env_runner_group = EnvRunnerGroup(
            config.env, config.num_env_runners,
            config.num_envs_per_env_runner, hidden=config.model_hidden,
        )
        self.buffer = ReplayBuffer(config.replay_buffer_capacity)
        self._env_steps = 0
        self._updates = 0
        self._rng = np.random.default_rng(config.seed)
        self._sync_exploration_weights()

    def _epsilon(self) -> float:
        c = self.config
        frac = min(1.0, self._env_steps / max(1, c.epsilon_decay_steps))
        return c.epsilon_start + frac * (c.epsilon_end - c.epsilon_start)

    def _sync_exploration_weights(self):
        # env runners sample with softmax over Q as a stand-in for
        # epsilon-greedy at high temperature; epsilon mixing happens
        # learner-side by injecting random actions into the buffer
        self.env_runner_group.sync_weights(self.q_net.get_weights())

    def training_step(self):
        cfg = self.config
        samples = self.env_runner_group.sample(
            max(cfg.rollout_fragment_length, 1)
        )
        eps = self._epsilon()
        for s in samples:
            T, B = s["rewards"].shape
            obs = s["obs"][: T - 1].reshape((T - 1) * B, -1)
            next_obs = s["obs"][1:].reshape((T - 1) * B, -1)
            actions = s["actions"][: T - 1].reshape(-1).copy()
            # epsilon-greedy: replace a fraction with random actions
            mask = self._rng.random(len(actions)) < eps
            actions[mask] = self._rng.integers(
                0, self.num_actions, mask.sum()
            )
            self.buffer.add_batch(
                {
                    "obs": obs.astype(np.float32),
                    "next_obs": next_obs.astype(np.float32),
                    "actions": actions,
                    "rewards": s["rewards"][: T - 1].reshape(-1),
                    "dones": s["dones"][: T - 1].reshape(-1),
                }
            )
            self._env_steps += T * B
        stats = {}
        if self._env_steps >= cfg.num_steps_sampled_before_learning:
            for _ in range(cfg.updates_per_iteration):
                stats = self._update_once()
        self._sync_exploration_weights()
        result = {
            "env_runners": summarize_episodes(samples),
            "learner": stats,
            "num_env_steps_sampled": sum(s["rewards"].size for s in samples),
            "num_env_steps_sampled_lifetime": self._env_steps,
            "epsilon": self._epsilon(),
            "replay_buffer_size": len(self.buffer),
        }
        em = result["env_runners"].get("episode_return_mean")
        if em is not None:
            result["episode_reward_mean"] = em
        return result

    def _update_once(self):
        cfg = self.config
        batch = self.buffer.sample(cfg.train_batch_size, self._rng)
        obs = torch.as_tensor(batch["obs"], device=self.device)
        next_obs = torch.as_tensor(batch["next_obs"], device=self.device)
        actions = torch.as_tensor(batch["actions"], device=self.device)
        rewards = torch.as_tensor(batch["rewards"], device=self.device)
        cont = 1.0 - torch.as_tensor(batch["dones"], device=self.device)
        q = self.q_net(obs)["logits"].gather(1, actions.view(-1, 1)).squeeze(1)
        with torch.no_grad():
            # double-DQN target
            next_a = self.q_net(next_obs)["logits"].argmax(1)
            next_q = (
                self.target_net(next_obs)["logits"]
                .gather(1, next_a.view(-1, 1))
                .squeeze(1)
            )
            target = rewards + cfg.gamma * cont * next_q
        loss = torch.nn.functional.smooth_l1_loss(q, target)
        self.opt.zero_grad()
        loss.backward()
        torch.nn.utils.clip_grad_norm_(self.q_net.parameters(), 10.0)
        self.opt.step()
        self._updates += 1
        if self._updates % cfg.target_network_update_freq == 0:
            self.target_net.load_state_dict(self.q_net.state_dict())
        return {"td_loss": float(loss.detach()),
                "mean_q": float(q.mean().detach())}

    def get_weights(self):
        return self.q_net.get_weights()

    def set_weights(self, w):
        self.q_net.set_weights(w)
        self.target_net.set_weights(w)
        self.env_runner_group.sync_weights(w)

    def stop(self):
        self.env_runner_group.stop()
