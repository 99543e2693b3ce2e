"""Environments for rllib. Ships a dependency-free CartPole-v1
(classic control physics, standard constants) since gymnasium is not in
the image; registered under the same name so configs match the
reference (BASELINE.json config #1). External gymnasium envs are used
when importable.
"""
from __future__ import annotations

from typing import Optional, Tuple

import numpy as np


class Box:
    def __init__(self, low, high, shape, dtype=np.float32):
        self.low = low
        self.high = high
        self.shape = tuple(shape)
        self.dtype = dtype


class Discrete:
    def __init__(self, n: int):
        self.n = n
        self.shape = ()
        self.dtype = np.int64


class CartPoleEnv:
    """CartPole-v1: standard classic-control dynamics."""

    max_episode_steps = 500

    def __init__(self, seed: Optional[int] = None):
        self.gravity = 9.8
        self.masscart = 1.0
        self.masspole = 0.1
        self.total_mass = self.masscart + self.masspole
        self.length = 0.5
        self.polemass_length = self.masspole * self.length
        self.force_mag = 10.0
        self.tau = 0.02
        self.x_threshold = 2.4
        self.theta_threshold = 12 * 2 * np.pi / 360
        self.rng = np.random.default_rng(seed)
        self.state = None
        self.steps = 0
        self.observation_space = Box(-np.inf, np.inf, (4,))
        self.action_space = Discrete(2)

    def reset(self, *, seed: Optional[int] = None) -> Tuple[np.ndarray, dict]:
        if seed is not None:
            self.rng = np.random.default_rng(seed)
        self.state = self.rng.uniform(-0.05, 0.05, size=4).astype(np.float32)
        self.steps = 0
        return self.state.copy(), {}

    def step(self, action: int):
        x, x_dot, theta, theta_dot = self.state
        force = self.force_mag if action == 1 else -self.force_mag
        costheta = np.cos(theta)
        sintheta = np.sin(theta)
        temp = (
            force + self.polemass_length * theta_dot**2 * sintheta
        ) / self.total_mass
        thetaacc = (self.gravity * sintheta - costheta * temp) / (
            self.length
            * (4.0 / 3.0 - self.masspole * costheta**2 / self.total_mass)
        )
        xacc = temp - self.polemass_length * thetaacc * costheta / self.total_mass
        x = x + self.tau * x_dot
        x_dot = x_dot + self.tau * xacc
        theta = theta + self.tau * theta_dot
        theta_dot = theta_dot + self.tau * thetaacc
        self.state = np.array([x, x_dot, theta, theta_dot], dtype=np.float32)
        self.steps += 1
        terminated = bool(
            x < -self.x_threshold
            or x > self.x_threshold
            or theta < -self.theta_threshold
            or theta > self.theta_threshold
        )
        truncated = self.steps >= self.max_episode_steps
        return self.state.copy(), 1.0, terminated, truncated, {}


class PendulumEnv:
    """Pendulum-v1: standard classic-control swing-up dynamics
    (continuous torque in [-2, 2]); the built-in continuous-action env
    for SAC (gymnasium is not in the image)."""

    max_episode_steps = 200

    def __init__(self, seed: Optional[int] = None):
        self.max_speed = 8.0
        self.max_torque = 2.0
        self.dt = 0.05
        self.g = 10.0
        self.m = 1.0
        self.l = 1.0
        self.rng = np.random.default_rng(seed)
        self.state = None
        self.steps = 0
        self.observation_space = Box(
            np.array([-1.0, -1.0, -self.max_speed]),
            np.array([1.0, 1.0, self.max_speed]), (3,))
        self.action_space = Box(-self.max_torque, self.max_torque, (1,))

    def _obs(self):
        th, thdot = self.state
        return np.array([np.cos(th), np.sin(th), thdot], dtype=np.float32)

    def reset(self, *, seed: Optional[int] = None):
        if seed is not None:
            self.rng = np.random.default_rng(seed)
        self.state = np.array([
            self.rng.uniform(-np.pi, np.pi),
            self.rng.uniform(-1.0, 1.0),
        ])
        self.steps = 0
        return self._obs(), {}

    def step(self, action):
        th, thdot = self.state
        u = float(np.clip(np.asarray(action).reshape(-1)[0],
                          -self.max_torque, self.max_torque))
        angle = ((th + np.pi) % (2 * np.pi)) - np.pi
        cost = angle**2 + 0.1 * thdot**2 + 0.001 * u**2
        thdot = thdot + (
            3 * self.g / (2 * self.l) * np.sin(th)
            + 3.0 / (self.m * self.l**2) * u
        ) * self.dt
        thdot = np.clip(thdot, -self.max_speed, self.max_speed)
        th = th + thdot * self.dt
        self.state = np.array([th, thdot])
        self.steps += 1
        truncated = self.steps >= self.max_episode_steps
        return self._obs(), -float(cost), False, truncated, {}


_REGISTRY = {"CartPole-v1": CartPoleEnv, "Pendulum-v1": PendulumEnv}
_USER_REGISTRY = {}


def register_env(name: str, creator):
    """reference: ray.tune.registry.register_env"""
    _USER_REGISTRY[name] = creator


def make_env(name_or_creator, seed=None):
    if callable(name_or_creator) and not isinstance(name_or_creator, str):
        return name_or_creator({})
    name = name_or_creator
    if name in _USER_REGISTRY:
        return _USER_REGISTRY[name]({})
    if name in _REGISTRY:
        return _REGISTRY[name](seed=seed)
    try:
        import gymnasium as gym

        return gym.make(name)
    except ImportError:
        raise ValueError(
            f"unknown env {name!r} (gymnasium not installed; built-ins: "
            f"{sorted(_REGISTRY)})"
        )


class VectorEnv:
    """Synchronous vectorized env (reference: gym.vector in
    single_agent_env_runner.py:152 sample loop)."""

    def __init__(self, name, num_envs: int, seed: int = 0):
        self.envs = [make_env(name, seed=seed + i) for i in range(num_envs)]
        self.num_envs = num_envs
        first = self.envs[0]
        self.observation_space = first.observation_space
        self.action_space = first.action_space
        self._episode_returns = np.zeros(num_envs)
        self._episode_lens = np.zeros(num_envs, dtype=np.int64)
        self.completed_returns = []
        self.completed_lens = []

    def reset(self):
        obs = [e.reset()[0] for e in self.envs]
        return np.stack(obs)

    def step(self, actions):
        obs, rews, terms, truncs = [], [], [], []
        discrete = isinstance(self.action_space, Discrete)
        for i, (e, a) in enumerate(zip(self.envs, actions)):
            o, r, term, trunc, _ = e.step(int(a) if discrete else a)
            self._episode_returns[i] += r
            self._episode_lens[i] += 1
            if term or trunc:
                self.completed_returns.append(self._episode_returns[i])
                self.completed_lens.append(int(self._episode_lens[i]))
                self._episode_returns[i] = 0.0
                self._episode_lens[i] = 0
                o = e.reset()[0]
            obs.append(o)
            rews.append(r)
            terms.append(term)
            truncs.append(trunc)
        return (
            np.stack(obs),
            np.asarray(rews, dtype=np.float32),
            np.asarray(terms),
            np.asarray(truncs),
        )

    def pop_episode_stats(self):
        r, l = self.completed_returns, self.completed_lens
        self.completed_returns, self.completed_lens = [], []
        return r, l
