"""TorchTrainer DDP (gloo on CPU; the same loop runs RCCL/xGMI on
MI355X with use_gpu=True)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import ray_amd as ray
from ray_amd.train import RunConfig, ScalingConfig
from ray_amd.train.torch import TorchTrainer, prepare_model


def train_loop(config):
    import ray_amd.train as train

    model = prepare_model(torch.nn.Linear(32, 4))
    opt = torch.optim.SGD(model.parameters(), lr=config["lr"])
    for epoch in range(3):
        x, y = torch.randn(64, 32), torch.randn(64, 4)
        loss = ((model(x) - y) ** 2).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
        train.report({"epoch": epoch, "loss": float(loss)})


ray.init()
result = TorchTrainer(
    train_loop,
    train_loop_config={"lr": 0.05},
    scaling_config=ScalingConfig(num_workers=2),  # use_gpu=True on MI355X
    run_config=RunConfig(name="example_ddp"),
).fit()
print("final:", result.metrics)
ray.shutdown()
