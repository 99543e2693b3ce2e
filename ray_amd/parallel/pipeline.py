"""Pipeline parallelism: GPipe-style fill–drain microbatch schedule
over stage actors (reference: §2.10 — the reference delegates PP to
its libraries; here it is a first-class utility on the actor
substrate).

Each stage is an actor owning a module shard + its optimizer; the
driver chains microbatch ObjectRefs through the stages, so stage k
works on microbatch i while stage k+1 works on microbatch i-1 — the
actors' ordered execution provides the schedule with no explicit
synchronization.

    pipe = Pipeline([lambda: nn.Linear(8, 32), lambda: nn.Linear(32, 1)],
                    lr=1e-2)
    loss = pipe.step(x, y)          # one fill-drain training step
"""
from __future__ import annotations

from typing import Callable, List, Optional


def _make_stage_actor():
    import ray_amd as ray

    @ray.remote
    class PipelineStage:
        """One pipeline stage: module shard + optimizer + per-microbatch
        activation stash."""

        def __init__(self, module_factory, lr, optimizer, seed):
            import torch

            torch.manual_seed(seed)
            self.torch = torch
            self.module = module_factory()
            opt_cls = {
                "sgd": torch.optim.SGD, "adam": torch.optim.Adam,
                "adamw": torch.optim.AdamW,
            }[optimizer]
            self.opt = opt_cls(self.module.parameters(), lr=lr)
            self._acts = {}  # mb_id -> (input, output)

        def forward(self, mb_id, x):
            t = self.torch
            x = t.as_tensor(x)
            x = x.detach().clone().requires_grad_(x.dtype.is_floating_point)
            out = self.module(x)
            self._acts[mb_id] = (x, out)
            return out.detach()

        def backward(self, mb_id, grad_out):
            t = self.torch
            x, out = self._acts.pop(mb_id)
            out.backward(t.as_tensor(grad_out))
            return None if x.grad is None else x.grad.detach()

        def backward_from_loss(self, mb_id, targets, loss_scale,
                               loss_fn_name):
            """Last stage: compute the loss and start the backward."""
            t = self.torch
            x, out = self._acts.pop(mb_id)
            targets = t.as_tensor(targets)
            if loss_fn_name == "mse":
                loss = t.nn.functional.mse_loss(out, targets)
            else:
                loss = t.nn.functional.cross_entropy(out, targets)
            (loss * loss_scale).backward()
            grad_in = None if x.grad is None else x.grad.detach()
            return float(loss.detach()), grad_in

        def apply_step(self):
            self.opt.step()
            self.opt.zero_grad(set_to_none=True)
            return True

        def get_state(self):
            return {
                k: v.detach().cpu().numpy()
                for k, v in self.module.state_dict().items()
            }

    return PipelineStage


class Pipeline:
    """Driver for a GPipe fill–drain step over stage actors."""

    def __init__(self, stage_factories: List[Callable], *, lr: float = 1e-2,
                 optimizer: str = "sgd", num_microbatches: int = 4,
                 seed: int = 0, loss: str = "mse",
                 stage_options: Optional[List[dict]] = None):
        import ray_amd as ray

        self._ray = ray
        Stage = _make_stage_actor()
        opts = stage_options or [{} for _ in stage_factories]
        self.stages = [
            Stage.options(**o).remote(f, lr, optimizer, seed + i)
            for i, (f, o) in enumerate(zip(stage_factories, opts))
        ]
        self.num_microbatches = num_microbatches
        self.loss = loss

        def _grad_of(pair):
            return pair[1]

        self._grad_of = ray.remote(_grad_of)

    def step(self, x, y) -> float:
        """One synchronous training step over the full batch; returns
        the mean microbatch loss."""
        import numpy as np

        ray = self._ray
        nmb = self.num_microbatches
        xs = np.array_split(np.asarray(x), nmb)
        ys = np.array_split(np.asarray(y), nmb)
        nmb = len(xs)

        # fill: chain every microbatch through the stages; actor
        # ordered execution pipelines them
        last_outs = []
        for i, xi in enumerate(xs):
            ref = ray.put(xi)
            for st in self.stages:
                ref = st.forward.remote(i, ref)
            last_outs.append(ref)
        # drain: losses + backward chains in reverse stage order
        loss_refs = []
        for i, yi in enumerate(ys):
            lref = self.stages[-1].backward_from_loss.remote(
                i, yi, 1.0 / nmb, self.loss
            )
            loss_refs.append(lref)
        grad_chains = []
        for i, lref in enumerate(loss_refs):
            gref = self._grad_of.remote(lref)
            for st in reversed(self.stages[:-1]):
                gref = st.backward.remote(i, gref)
            grad_chains.append(gref)
        ray.get(grad_chains, timeout=600)
        ray.get([st.apply_step.remote() for st in self.stages],
                timeout=600)
        losses = [l for l, _ in ray.get(loss_refs, timeout=600)]
        return float(sum(losses) / len(losses))

    def state_dicts(self):
        return self._ray.get(
            [st.get_state.remote() for st in self.stages], timeout=600
        )

