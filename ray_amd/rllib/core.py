"""RLModule — the model abstraction (reference: rllib/core/rl_module/).

A TorchRLModule with forward_inference / forward_exploration /
forward_train, default MLP policy+value catalog for discrete-action
envs (reference: ppo_catalog.py).
"""
from __future__ import annotations

from typing import Dict

import numpy as np
import torch
import torch.nn as nn


class TorchRLModule(nn.Module):
    def __init__(self, obs_dim: int, num_actions: int,
                 hidden=(256, 256), device="cpu"):
        super().__init__()
        self.obs_dim = obs_dim
        self.num_actions = num_actions

        def mlp():
            layers = []
            d = obs_dim
            for h in hidden:
                layers += [nn.Linear(d, h), nn.Tanh()]
                d = h
            return nn.Sequential(*layers), d

        # Separate pi/vf trunks (reference PPO default:
        # vf_share_layers=False) — a shared trunk lets the large-scale
        # value loss corrupt policy features.
        self.encoder, d = mlp()
        self.vf_encoder, _ = mlp()
        self.pi_head = nn.Linear(d, num_actions)
        self.vf_head = nn.Linear(d, 1)
        nn.init.orthogonal_(self.pi_head.weight, gain=0.01)
        self.device = torch.device(device)
        self.to(self.device)

    def forward(self, obs: torch.Tensor) -> Dict[str, torch.Tensor]:
        return {
            "logits": self.pi_head(self.encoder(obs)),
            "vf": self.vf_head(self.vf_encoder(obs)).squeeze(-1),
        }

    @torch.no_grad()
    def forward_inference(self, obs: np.ndarray) -> np.ndarray:
        out = self(torch.as_tensor(obs, dtype=torch.float32, device=self.device))
        return out["logits"].argmax(-1).cpu().numpy()

    @torch.no_grad()
    def forward_exploration(self, obs: np.ndarray):
        out = self(torch.as_tensor(obs, dtype=torch.float32, device=self.device))
        logits = out["logits"]
        dist = torch.distributions.Categorical(logits=logits)
        a = dist.sample()
        return (
            a.cpu().numpy(),
            dist.log_prob(a).cpu().numpy(),
            out["vf"].cpu().numpy(),
        )

    def forward_train(self, obs: torch.Tensor) -> Dict[str, torch.Tensor]:
        return self(obs)

    def get_weights(self) -> Dict[str, np.ndarray]:
        return {k: v.detach().cpu().numpy() for k, v in self.state_dict().items()}

    def set_weights(self, weights: Dict[str, np.ndarray]):
        self.load_state_dict(
            {k: torch.as_tensor(v) for k, v in weights.items()}
        )


def build_module_for_env(env, hidden=(256, 256), device="cpu") -> TorchRLModule:
    obs_dim = int(np.prod(env.observation_space.shape))
    num_actions = env.action_space.n
    return TorchRLModule(obs_dim, num_actions, hidden, device)
