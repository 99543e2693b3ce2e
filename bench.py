"""Flagship benchmark: Llama-3-8B DDP training step (BASELINE.json
metric "samples/sec Ray Train Llama-3-8B DDP").

Synthetic data (random tokens), random-init weights, bf16 compute,
fp32 optimizer states (ray_amd.ops.FusedAdamW HIP kernel), DDP over
RCCL with xGMI-tuned buckets. Launched by the driver as
`torch.distributed.run --nproc-per-node N bench.py --gpus N ...` for
N>1; reads RANK/LOCAL_RANK/WORLD_SIZE from the env.
"""
from __future__ import annotations

import argparse
import json
import os
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--model", default="llama3-8b")
    p.add_argument("--seq-len", type=int, default=4096)
    p.add_argument("--micro-batch", type=int, default=8)
    p.add_argument("--bucket-mb", type=int, default=128,
                   help="DDP bucket size; xGMI rings favor large buckets")
    p.add_argument("--grad-checkpoint", action="store_true")
    p.add_argument("--device", default=None, help="cpu for local testing")
    p.add_argument("--profile-steps", type=int, default=0)
    return p.parse_args()


def main():
    if os.environ.get("BENCH_TRACE"):
        import faulthandler

        faulthandler.dump_traceback_later(
            int(os.environ["BENCH_TRACE"]), exit=True
        )
    args = parse_args()
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))

    use_cpu = args.device == "cpu" or not torch.cuda.is_available()
    if use_cpu:
        device = torch.device("cpu")
        dtype = torch.float32
        if args.model == "llama3-8b":
            args.model = "llama-tiny"
            args.seq_len = min(args.seq_len, 128)
    else:
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
        dtype = torch.bfloat16

    distributed = world_size > 1
    if distributed:
        import torch.distributed as dist

        dist.init_process_group(
            backend="gloo" if use_cpu else "nccl",
            rank=rank,
            world_size=world_size,
        )

    if args.model == "resnet50":
        run_resnet(args, world_size, rank, local_rank, device, use_cpu)
        return
    if args.model == "ppo":
        run_ppo(args, world_size, rank, use_cpu)
        return

    from ray_amd.models.llama import CONFIGS, LlamaModel
    from ray_amd.ops import FusedAdamW

    cfg = CONFIGS[args.model]
    seq = min(args.seq_len, cfg.max_seq_len)
    torch.manual_seed(1234 + rank)
    if use_cpu:
        model = LlamaModel(cfg, dtype=dtype,
                           gradient_checkpointing=args.grad_checkpoint)
    else:
        # build directly on the GPU: skips 16 GB of CPU init + H2D per
        # rank (8-rank scale runs would otherwise serialize minutes of
        # host-side random init)
        with torch.device(device):
            model = LlamaModel(cfg, dtype=dtype,
                               gradient_checkpointing=args.grad_checkpoint)
        model.cosT = model.cosT.to(device)
        model.sinT = model.sinT.to(device)
    n_params = model.num_params()

    if distributed:
        from torch.nn.parallel import DistributedDataParallel as DDP

        model = DDP(
            model,
            device_ids=None if use_cpu else [local_rank],
            bucket_cap_mb=args.bucket_mb,
            gradient_as_bucket_view=True,
        )

    opt = FusedAdamW(model.parameters(), lr=1e-4, weight_decay=0.1)

    B, T = args.micro_batch, seq
    vocab = cfg.vocab_size
    def make_batch():
        return (torch.randint(0, vocab, (B, T), device=device),
                torch.randint(0, vocab, (B, T), device=device))

    def step():
        tokens, targets = make_batch()
        loss = model(tokens, targets)
        loss.backward()
        opt.step()
        opt.zero_grad()
        return loss

    def sync():
        if distributed:
            import torch.distributed as dist

            dist.barrier()
        if not use_cpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        step()
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        loss = step()
    sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if distributed:
        import torch.distributed as dist

        t = torch.tensor([elapsed], dtype=torch.float64, device=device if not use_cpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    total_samples = args.steps * B * world_size
    samples_per_sec = total_samples / elapsed
    tokens_per_sec = samples_per_sec * T

    if rank == 0:
        result = {
            "metric": "train_samples_per_sec",
            "value": round(samples_per_sec, 3),
            "unit": "samples/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32" if use_cpu else "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": B * world_size,
                "micro_batch": B,
                "seq_len": T,
                "parallelism": f"dp{world_size}",
                "n_params": n_params,
                "tokens_per_sec": round(tokens_per_sec, 1),
                "final_loss": round(float(loss.detach().float().cpu()), 4),
                "bucket_mb": args.bucket_mb,
                "grad_checkpoint": bool(args.grad_checkpoint),
            },
        }
        print(json.dumps(result))

    if distributed:
        import torch.distributed as dist

        dist.destroy_process_group()


def run_ppo(args, world_size, rank, use_cpu):
    """RLlib PPO CartPole env-steps/sec (BASELINE metric 2; config #1 is
    the CPU plumbing case). Each rank runs an independent PPO learner
    (GPU if available) with its own rollout workers — weak scaling."""
    import ray_amd as ray

    ray.init(num_cpus=max(4, (os.cpu_count() or 8) // max(world_size, 1)),
             num_gpus=0 if use_cpu else 1)
    from ray_amd.rllib.algorithms.ppo import PPOConfig

    config = (
        PPOConfig()
        .environment("CartPole-v1")
        .env_runners(num_env_runners=2, num_envs_per_env_runner=8)
        .training(train_batch_size=4000, minibatch_size=512, num_epochs=8)
        .learners(num_gpus_per_learner=0 if use_cpu else 1)
    )
    algo = config.build()
    for _ in range(max(args.warmup, 1)):
        algo.train()
    if world_size > 1:
        import torch.distributed as dist

        dist.barrier()
    t0 = time.perf_counter()
    steps = 0
    for _ in range(args.steps):
        r = algo.train()
        steps += r["num_env_steps_sampled"]
    elapsed = time.perf_counter() - t0
    if world_size > 1:
        import torch.distributed as dist

        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
    total = steps * world_size
    if rank == 0:
        print(json.dumps({
            "metric": "ppo_env_steps_per_sec",
            "value": round(total / elapsed, 2),
            "unit": "env_steps/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {"model": "ppo-cartpole", "train_batch_size": 4000,
                       "env_runners_per_learner": 2,
                       "parallelism": f"dp{world_size}",
                       "reward_mean": r.get("episode_reward_mean")},
        }))
    algo.stop()
    ray.shutdown()
    if world_size > 1:
        import torch.distributed as dist

        dist.destroy_process_group()


def run_resnet(args, world_size, rank, local_rank, device, use_cpu):
    """North-star config 2: ResNet-50 DDP bf16 (autocast), synthetic
    ImageNet-shaped data."""
    import torch.nn.functional as F

    from ray_amd.models.resnet import ResNet50

    torch.manual_seed(1234 + rank)
    model = ResNet50().to(device)
    if world_size > 1:
        from torch.nn.parallel import DistributedDataParallel as DDP

        model = DDP(model, device_ids=None if use_cpu else [local_rank],
                    bucket_cap_mb=args.bucket_mb,
                    gradient_as_bucket_view=True)
    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9,
                          weight_decay=1e-4)
    B = args.micro_batch if args.micro_batch != 8 else 256
    if use_cpu:
        B = min(B, 8)
    res = 224 if not use_cpu else 64
    amp_dtype = torch.bfloat16

    def step():
        x = torch.randn(B, 3, res, res, device=device)
        y = torch.randint(0, 1000, (B,), device=device)
        with torch.autocast(device_type=device.type, dtype=amp_dtype,
                            enabled=not use_cpu):
            loss = F.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        opt.zero_grad(set_to_none=True)
        return loss

    def sync():
        if world_size > 1:
            import torch.distributed as dist

            dist.barrier()
        if not use_cpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        step()
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        loss = step()
    sync()
    elapsed = time.perf_counter() - t0
    if world_size > 1:
        import torch.distributed as dist

        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if not use_cpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
    total = args.steps * B * world_size
    if rank == 0:
        print(json.dumps({
            "metric": "train_samples_per_sec",
            "value": round(total / elapsed, 2),
            "unit": "images/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if not use_cpu else "fp32",
            "data": "synthetic",
            "config": {"model": "resnet50", "global_batch": B * world_size,
                       "resolution": res, "parallelism": f"dp{world_size}",
                       "final_loss": round(float(loss.detach().float().cpu()), 4)},
        }))
    if world_size > 1:
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    main()
