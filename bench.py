"""Flagship benchmark: Llama-3-8B DDP training step through the
framework (BASELINE.json metric "samples/sec Ray Train Llama-3-8B DDP").

The training runs through the full ray_amd stack: `ray_amd.init` ->
`TorchTrainer(ScalingConfig(num_workers=N, use_gpu=True))` -> placement
group -> one GPU actor per rank -> RCCL process group (reference path:
train/torch/torch_trainer.py over _internal/backend_executor.py:86).
Synthetic data (random tokens), random-init weights, bf16 compute,
fp32 optimizer states (ray_amd.ops.FusedAdamW HIP kernel), DDP with
xGMI-tuned buckets.

Launch contract: the driver runs `torch.distributed.run
--nproc-per-node N bench.py --gpus N ...` for N>1. Rank 0 drives the
whole framework job over all N GPUs; companion ranks only hold the
torchrun rendezvous open (the GPU work happens in the framework's own
worker actors, one per GPU, with their own RCCL process group).
`--raw` bypasses the framework (plain DDP in the torchrun ranks) to
measure framework overhead A/B.
"""
from __future__ import annotations

import argparse
import json
import os
import time
from datetime import timedelta

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
    p.add_argument("--raw", action="store_true",
                   help="bypass the framework: plain torchrun DDP "
                        "(for framework-overhead A/B only)")
    return p.parse_args()


# ---------------------------------------------------------------------------
# train_loop_per_worker bodies (run inside TorchTrainer worker actors)
# ---------------------------------------------------------------------------


def _timed_dist_loop(cfg, build, step_fn):
    """Shared harness: warmup, barrier+sync, time K steps, barrier+sync,
    MAX over ranks; reports elapsed + loss through the train session."""
    import torch.distributed as dist

    import ray_amd.train as train

    ctx = train.get_context()
    world = ctx.get_world_size()
    use_cpu = cfg["use_cpu"]
    device = torch.device("cpu") if use_cpu else train.torch.get_device()
    state = build(device)

    def sync():
        if world > 1 and dist.is_initialized():
            dist.barrier()
        if not use_cpu:
            torch.cuda.synchronize()

    for _ in range(cfg["warmup"]):
        step_fn(state)
    sync()
    t0 = time.perf_counter()
    loss = None
    for _ in range(cfg["steps"]):
        loss = step_fn(state)
    sync()
    elapsed = time.perf_counter() - t0
    if world > 1 and dist.is_initialized():
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device="cpu" if use_cpu else device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
    train.report({
        "elapsed_s": elapsed,
        "final_loss": float(loss.detach().float().cpu()),
        "n_params": state.get("n_params", 0),
        "world_size": world,
    })


def llama_train_loop(cfg):
    from ray_amd.models.llama import CONFIGS, LlamaModel
    from ray_amd.ops import FusedAdamW
    from ray_amd.train.torch import prepare_model

    import ray_amd.train as train

    ctx = train.get_context()
    mcfg = CONFIGS[cfg["model"]]
    seq = min(cfg["seq_len"], mcfg.max_seq_len)
    use_cpu = cfg["use_cpu"]
    dtype = torch.float32 if use_cpu else torch.bfloat16
    torch.manual_seed(1234 + ctx.get_world_rank())

    def build(device):
        if use_cpu:
            model = LlamaModel(mcfg, dtype=dtype,
                               gradient_checkpointing=cfg["grad_checkpoint"])
        else:
            # build directly on the GPU: skips 16 GB of CPU init + H2D
            # per rank
            with torch.device(device):
                model = LlamaModel(
                    mcfg, dtype=dtype,
                    gradient_checkpointing=cfg["grad_checkpoint"],
                )
        n_params = model.num_params()
        model = prepare_model(
            model,
            parallel_strategy_kwargs={"bucket_cap_mb": cfg["bucket_mb"]},
        )
        opt = FusedAdamW(model.parameters(), lr=1e-4, weight_decay=0.1)
        return {"model": model, "opt": opt, "device": device,
                "n_params": n_params, "vocab": mcfg.vocab_size, "seq": seq}

    B = cfg["micro_batch"]

    def step(state):
        tokens = torch.randint(0, state["vocab"], (B, state["seq"]),
                               device=state["device"])
        targets = torch.randint(0, state["vocab"], (B, state["seq"]),
                                device=state["device"])
        loss = state["model"](tokens, targets)
        loss.backward()
        state["opt"].step()
        state["opt"].zero_grad()
        return loss

    _timed_dist_loop(cfg, build, step)


def resnet_train_loop(cfg):
    import torch.nn.functional as F

    from ray_amd.models.resnet import ResNet50
    from ray_amd.train.torch import prepare_model

    import ray_amd.train as train

    ctx = train.get_context()
    use_cpu = cfg["use_cpu"]
    torch.manual_seed(1234 + ctx.get_world_rank())
    B = cfg["micro_batch"] if cfg["micro_batch"] != 8 else 256
    if use_cpu:
        B = min(B, 8)
    res = 64 if use_cpu else 224

    def build(device):
        model = ResNet50().to(device)
        n_params = sum(p.numel() for p in model.parameters())
        model = prepare_model(
            model,
            parallel_strategy_kwargs={"bucket_cap_mb": cfg["bucket_mb"]},
        )
        opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9,
                              weight_decay=1e-4)
        return {"model": model, "opt": opt, "device": device,
                "n_params": n_params}

    def step(state):
        device = state["device"]
        x = torch.randn(B, 3, res, res, device=device)
        y = torch.randint(0, 1000, (B,), device=device)
        with torch.autocast(device_type=device.type, dtype=torch.bfloat16,
                            enabled=not use_cpu):
            loss = F.cross_entropy(state["model"](x), y)
        loss.backward()
        state["opt"].step()
        state["opt"].zero_grad(set_to_none=True)
        return loss

    cfg = dict(cfg)
    cfg["_batch"] = B
    _timed_dist_loop(cfg, build, step)


# ---------------------------------------------------------------------------
# rank-0 driver: the framework job
# ---------------------------------------------------------------------------


def run_framework(args, n, use_cpu):
    import ray_amd as ray
    from ray_amd.train import RunConfig, ScalingConfig
    from ray_amd.train.torch import TorchTrainer

    ray.init(
        num_cpus=max(8, 2 * n + 4),
        num_gpus=0 if use_cpu else n,
    )
    try:
        is_resnet = args.model == "resnet50"
        loop = resnet_train_loop if is_resnet else llama_train_loop
        cfg = {
            "model": args.model,
            "seq_len": args.seq_len,
            "micro_batch": args.micro_batch,
            "bucket_mb": args.bucket_mb,
            "grad_checkpoint": args.grad_checkpoint,
            "steps": args.steps,
            "warmup": args.warmup,
            "use_cpu": use_cpu,
        }
        trainer = TorchTrainer(
            loop,
            train_loop_config=cfg,
            scaling_config=ScalingConfig(num_workers=n, use_gpu=not use_cpu),
            run_config=RunConfig(name=f"bench_{args.model}"),
        )
        result = trainer.fit()
        if result.error:
            raise result.error
        m = result.metrics
        elapsed = m["elapsed_s"]
        if is_resnet:
            B = args.micro_batch if args.micro_batch != 8 else 256
            if use_cpu:
                B = min(B, 8)
            seq = None
        else:
            B = args.micro_batch
            from ray_amd.models.llama import CONFIGS

            seq = min(args.seq_len, CONFIGS[args.model].max_seq_len)
        total_samples = args.steps * B * n
        sps = total_samples / elapsed
        out = {
            "metric": "train_samples_per_sec",
            "value": round(sps, 3),
            "unit": "images/s" if is_resnet else "samples/s",
            "n_gpus": n,
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
                "global_batch": B * n,
                "micro_batch": B,
                "seq_len": seq,
                "parallelism": f"dp{n}",
                "n_params": m.get("n_params"),
                "final_loss": round(m["final_loss"], 4),
                "bucket_mb": args.bucket_mb,
                "grad_checkpoint": bool(args.grad_checkpoint),
                "harness": "ray_amd.train.TorchTrainer",
            },
        }
        if not is_resnet and seq:
            out["config"]["tokens_per_sec"] = round(sps * seq, 1)
        print(json.dumps(out))
    finally:
        ray.shutdown()


def main():
    if os.environ.get("BENCH_TRACE"):
        import faulthandler

        faulthandler.dump_traceback_later(
            int(os.environ["BENCH_TRACE"]), exit=True
        )
    args = parse_args()
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    use_cpu = args.device == "cpu" or not torch.cuda.is_available()
    if use_cpu and args.model == "llama3-8b" and not args.raw:
        args.model = "llama-tiny"
        args.seq_len = min(args.seq_len, 128)

    if args.raw or args.model == "ppo":
        run_torchrun_paths(args, world_size, rank, use_cpu)
        return

    n = world_size if world_size > 1 else args.gpus
    if world_size > 1:
        # companion-rank protocol: the framework owns the GPUs from
        # rank 0; other torchrun ranks just hold the rendezvous open.
        import torch.distributed as dist

        dist.init_process_group(backend="gloo", rank=rank,
                                world_size=world_size,
                                timeout=timedelta(hours=2))
        if rank == 0:
            run_framework(args, n, use_cpu)
        dist.barrier()
        dist.destroy_process_group()
    else:
        run_framework(args, max(n, 1), use_cpu)


# ---------------------------------------------------------------------------
# raw torchrun paths (framework-overhead A/B + PPO)
# ---------------------------------------------------------------------------


def run_torchrun_paths(args, world_size, rank, use_cpu):
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
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

    def step():
        tokens = torch.randint(0, vocab, (B, T), device=device)
        targets = torch.randint(0, vocab, (B, T), device=device)
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

    if distributed:
        import torch.distributed as dist

        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if not use_cpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    total_samples = args.steps * B * world_size
    samples_per_sec = total_samples / elapsed

    if rank == 0:
        print(json.dumps({
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
                "tokens_per_sec": round(samples_per_sec * T, 1),
                "final_loss": round(float(loss.detach().float().cpu()), 4),
                "bucket_mb": args.bucket_mb,
                "grad_checkpoint": bool(args.grad_checkpoint),
                "harness": "raw-ddp",
            },
        }))

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


if __name__ == "__main__":
    main()
