"""TorchLearner — loss/update (reference: rllib/core/learner/learner.py,
torch/torch_learner.py:67 compute_gradients :170; multi-GPU = one
Learner actor per GPU + DDP over RCCL :444).

The GAE advantage pass runs through the HIP scan kernel on GPU
(ray_amd/csrc/hip/rl_scans.hip; reference computes it as a Python loop,
value_predictions.py:7), V-trace likewise for IMPALA.
"""
from __future__ import annotations

from typing import Dict, List

import numpy as np
import torch

from ray_amd import ops

from .core import TorchRLModule


class PPOTorchLearner:
    def __init__(
        self,
        obs_dim: int,
        num_actions: int,
        hidden=(256, 256),
        lr: float = 3e-4,
        gamma: float = 0.99,
        lambda_: float = 0.95,
        clip_param: float = 0.2,
        vf_clip_param: float = 10.0,
        vf_loss_coeff: float = 0.5,
        entropy_coeff: float = 0.0,
        num_epochs: int = 8,
        minibatch_size: int = 512,
        use_gpu: bool = False,
        ddp: bool = False,
        learner_connector=None,
    ):
        self.device = torch.device(
            "cuda:0" if use_gpu and torch.cuda.is_available() else "cpu"
        )
        self.module = TorchRLModule(obs_dim, num_actions, hidden,
                                    device=self.device)
        self.raw_module = self.module
        if ddp:
            import torch.distributed as dist

            if dist.is_initialized():
                from torch.nn.parallel import DistributedDataParallel as DDP

                self.module = DDP(self.module)
        self.opt = torch.optim.Adam(self.raw_module.parameters(), lr=lr)
        self.gamma = gamma
        self.lambda_ = lambda_
        self.clip = clip_param
        self.vf_clip = vf_clip_param
        self.vf_coeff = vf_loss_coeff
        self.ent_coeff = entropy_coeff
        self.num_epochs = num_epochs
        self.minibatch_size = minibatch_size
        if learner_connector is not None:
            self.learner_connector = learner_connector()
        else:
            from .connectors import (ConnectorPipeline,
                                     GeneralAdvantageEstimation)

            self.learner_connector = ConnectorPipeline([
                GeneralAdvantageEstimation(gamma, lambda_,
                                           device=self.device),
            ])

    def _advantages(self, batch: Dict[str, np.ndarray]):
        # learner-connector pipeline (reference: GAE as a learner
        # connector, general_advantage_estimation.py:21)
        out = self.learner_connector(dict(batch))
        return out["advantages"], out["value_targets"]

    def update(self, samples: List[Dict[str, np.ndarray]]) -> Dict[str, float]:
        # concat along env axis
        batch = {
            k: np.concatenate([s[k] for s in samples], axis=1)
            for k in ("obs", "actions", "rewards", "dones", "logp", "vf")
        }
        adv, vtarg = self._advantages(batch)
        T, B = batch["rewards"].shape
        obs = torch.as_tensor(
            batch["obs"].reshape(T * B, -1), device=self.device
        )
        actions = torch.as_tensor(
            batch["actions"].reshape(-1), device=self.device
        )
        old_logp = torch.as_tensor(
            batch["logp"].reshape(-1), device=self.device
        )
        adv = adv.reshape(-1)
        vtarg = vtarg.reshape(-1)
        adv = (adv - adv.mean()) / (adv.std() + 1e-8)

        n = T * B
        idx_all = torch.arange(n, device=self.device)
        stats = {}
        for _ in range(self.num_epochs):
            perm = idx_all[torch.randperm(n, device=self.device)]
            for s in range(0, n, self.minibatch_size):
                mb = perm[s : s + self.minibatch_size]
                out = self.module(obs[mb])
                dist = torch.distributions.Categorical(logits=out["logits"])
                logp = dist.log_prob(actions[mb])
                ratio = (logp - old_logp[mb]).exp()
                surr = torch.min(
                    ratio * adv[mb],
                    ratio.clamp(1 - self.clip, 1 + self.clip) * adv[mb],
                )
                pi_loss = -surr.mean()
                vf_err = (out["vf"] - vtarg[mb]).pow(2)
                vf_loss = vf_err.clamp(max=self.vf_clip**2).mean()
                entropy = dist.entropy().mean()
                loss = pi_loss + self.vf_coeff * vf_loss - self.ent_coeff * entropy
                self.opt.zero_grad()
                loss.backward()
                torch.nn.utils.clip_grad_norm_(
                    self.raw_module.parameters(), 1.0
                )
                self.opt.step()
                stats = {
                    "policy_loss": float(pi_loss.detach()),
                    "vf_loss": float(vf_loss.detach()),
                    "entropy": float(entropy.detach()),
                    "total_loss": float(loss.detach()),
                }
        return stats

    def get_weights(self):
        return self.raw_module.get_weights()

    def set_weights(self, w):
        self.raw_module.set_weights(w)


class ImpalaTorchLearner:
    """V-trace learner (reference: impala.py + vtrace_torch_v2.py:73)."""

    def __init__(self, obs_dim, num_actions, hidden=(256, 256), lr=5e-4,
                 gamma=0.99, vf_coeff=0.5, ent_coeff=0.01, use_gpu=False):
        self.device = torch.device(
            "cuda:0" if use_gpu and torch.cuda.is_available() else "cpu"
        )
        self.module = TorchRLModule(obs_dim, num_actions, hidden,
                                    device=self.device)
        self.opt = torch.optim.Adam(self.module.parameters(), lr=lr)
        self.gamma = gamma
        self.vf_coeff = vf_coeff
        self.ent_coeff = ent_coeff

    def update(self, samples: List[Dict[str, np.ndarray]]) -> Dict[str, float]:
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
        # bootstrap with behaviour values' last row
        values_tb1 = torch.cat(
            [values, torch.as_tensor(batch["vf"][-1:], device=self.device)], 0
        )
        with torch.no_grad():
            log_rhos = (tgt_logp - behav_logp).detach()
            vs, pg_adv = ops.vtrace(
                log_rhos, rewards, values_tb1.detach(), cont, self.gamma
            )
        pi_loss = -(tgt_logp * pg_adv).mean()
        vf_loss = (values - vs).pow(2).mean()
        entropy = dist.entropy().mean()
        loss = pi_loss + self.vf_coeff * vf_loss - self.ent_coeff * entropy
        self.opt.zero_grad()
        loss.backward()
        torch.nn.utils.clip_grad_norm_(self.module.parameters(), 40.0)
        self.opt.step()
        return {
            "policy_loss": float(pi_loss.detach()),
            "vf_loss": float(vf_loss.detach()),
            "entropy": float(entropy.detach()),
            "total_loss": float(loss.detach()),
        }

    def get_weights(self):
        return self.module.get_weights()

    def set_weights(self, w):
        self.module.set_weights(w)
