"""Algorithm / AlgorithmConfig (reference: rllib/algorithms/algorithm.py:211,
algorithm_config.py builder pattern; train() -> training_step loop)."""
from __future__ import annotations

import copy
import time
from typing import Dict, Optional

import numpy as np


class AlgorithmConfig:
    def __init__(self, algo_class=None):
        self.algo_class = algo_class
        self.env: Optional[str] = None
        self.env_config: dict = {}
        self.num_env_runners = 0
        self.num_envs_per_env_runner = 8
        self.env_to_module_connector = None   # () -> ConnectorPipeline
        self.learner_connector = None         # () -> ConnectorPipeline
        self.rollout_fragment_length = 200
        self.train_batch_size = 4000
        self.minibatch_size = 512
        self.num_epochs = 8
        self.lr = 3e-4
        self.gamma = 0.99
        self.lambda_ = 0.95
        self.clip_param = 0.2
        self.vf_clip_param = 10.0
        self.vf_loss_coeff = 0.5
        self.entropy_coeff = 0.0
        self.num_learners = 0
        self.num_gpus_per_learner = 0
        self.model_hidden = (256, 256)
        self.seed = 0

    # builder methods (reference parity)
    def environment(self, env=None, *, env_config=None, **kwargs):
        if env is not None:
            self.env = env
        if env_config is not None:
            self.env_config = env_config
        return self

    def env_runners(self, *, num_env_runners=None, num_envs_per_env_runner=None,
                    rollout_fragment_length=None,
                    env_to_module_connector=None, **kwargs):
        if num_env_runners is not None:
            self.num_env_runners = num_env_runners
        if num_envs_per_env_runner is not None:
            self.num_envs_per_env_runner = num_envs_per_env_runner
        if rollout_fragment_length is not None:
            self.rollout_fragment_length = rollout_fragment_length
        if env_to_module_connector is not None:
            # factory () -> ConnectorPipeline (reference:
            # config.env_runners(env_to_module_connector=...))
            self.env_to_module_connector = env_to_module_connector
        return self

    def rollouts(self, **kwargs):  # old-stack alias
        return self.env_runners(**kwargs)

    def learner_connections(self, *, learner_connector=None, **kw):
        """reference: config.training(learner_connector=...) — factory
        () -> ConnectorPipeline run on the train batch before the
        loss."""
        if learner_connector is not None:
            self.learner_connector = learner_connector
        return self

    def training(self, *, lr=None, gamma=None, train_batch_size=None,
                 minibatch_size=None, num_epochs=None, clip_param=None,
                 vf_loss_coeff=None, entropy_coeff=None, lambda_=None,
                 model=None, **kwargs):
        if lr is not None:
            self.lr = lr
        if gamma is not None:
            self.gamma = gamma
        if train_batch_size is not None:
            self.train_batch_size = train_batch_size
        if minibatch_size is not None:
            self.minibatch_size = minibatch_size
        if num_epochs is not None:
            self.num_epochs = num_epochs
        if clip_param is not None:
            self.clip_param = clip_param
        if vf_loss_coeff is not None:
            self.vf_loss_coeff = vf_loss_coeff
        if entropy_coeff is not None:
            self.entropy_coeff = entropy_coeff
        if lambda_ is not None:
            self.lambda_ = lambda_
        if model and "fcnet_hiddens" in model:
            self.model_hidden = tuple(model["fcnet_hiddens"])
        return self

    def learners(self, *, num_learners=None, num_gpus_per_learner=None, **kw):
        if num_learners is not None:
            self.num_learners = num_learners
        if num_gpus_per_learner is not None:
            self.num_gpus_per_learner = num_gpus_per_learner
        return self

    def resources(self, **kwargs):
        return self

    def framework(self, *a, **k):
        return self

    def debugging(self, *, seed=None, **k):
        if seed is not None:
            self.seed = seed
        return self

    def api_stack(self, **k):
        return self

    def evaluation(self, **k):
        return self

    def copy(self):
        return copy.deepcopy(self)

    def build(self):
        if self.algo_class is None:
            raise ValueError("no algo_class bound to this config")
        return self.algo_class(self)

    # new-stack name
    def build_algo(self):
        return self.build()


class Algorithm:
    """Base: train() loop + checkpointing (Checkpointable parity)."""

    def __init__(self, config: AlgorithmConfig):
        self.config = config
        self.iteration = 0
        self._setup(config)

    def _setup(self, config):
        raise NotImplementedError

    def training_step(self) -> Dict:
        raise NotImplementedError

    def train(self) -> Dict:
        t0 = time.time()
        result = self.training_step()
        self.iteration += 1
        result.setdefault("training_iteration", self.iteration)
        result.setdefault("time_this_iter_s", time.time() - t0)
        return result

    def stop(self):
        pass

    # --- Checkpointable (reference: rllib/utils/checkpoints.py) ---

    def save(self, checkpoint_dir: Optional[str] = None) -> str:
        import os
        import pickle
        import tempfile

        d = checkpoint_dir or tempfile.mkdtemp(prefix="rllib_ckpt_")
        os.makedirs(d, exist_ok=True)
        with open(os.path.join(d, "algorithm_state.pkl"), "wb") as f:
            pickle.dump(
                {"weights": self.get_weights(), "iteration": self.iteration},
                f,
            )
        return d

    def restore(self, checkpoint_dir: str):
        import os
        import pickle

        with open(os.path.join(checkpoint_dir, "algorithm_state.pkl"), "rb") as f:
            state = pickle.load(f)
        self.set_weights(state["weights"])
        self.iteration = state.get("iteration", 0)

    def get_weights(self):
        raise NotImplementedError

    def set_weights(self, w):
        raise NotImplementedError


def summarize_episodes(samples) -> Dict[str, float]:
    rets = np.concatenate([s["episode_returns"] for s in samples])
    lens = np.concatenate([s["episode_lens"] for s in samples])
    out = {}
    if len(rets):
        out["episode_return_mean"] = float(np.mean(rets))
        out["episode_return_max"] = float(np.max(rets))
        out["episode_len_mean"] = float(np.mean(lens))
    return out
