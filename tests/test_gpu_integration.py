"""GPU integration tests (all @gpu): north-star configs exercised on
one MI355X — RLlib learners on GPU, Data GPU ingest, Serve+LLM, Train
GPU worker."""
import numpy as np
import pytest
import torch

import ray_amd as ray

GPU = pytest.mark.gpu


@GPU
def test_rllib_ppo_gpu_learner():
    ray.init(num_cpus=4, num_gpus=1, ignore_reinit_error=True)
    try:
        from ray_amd.rllib.algorithms.ppo import PPOConfig

        config = (
            PPOConfig()
            .environment("CartPole-v1")
            .env_runners(num_env_runners=2, num_envs_per_env_runner=4)
            .training(train_batch_size=1600, minibatch_size=256, num_epochs=2)
            .learners(num_learners=0, num_gpus_per_learner=1)
        )
        algo = config.build()
        assert algo.learner.device.type == "cuda"
        r = algo.train()
        assert r["num_env_steps_sampled"] == 1600
        assert np.isfinite(r["learner"]["total_loss"])
        algo.stop()
    finally:
        ray.shutdown()


@GPU
def test_rllib_impala_gpu_learner_vtrace():
    ray.init(num_cpus=4, num_gpus=1, ignore_reinit_error=True)
    try:
        from ray_amd.rllib.algorithms.impala import IMPALAConfig

        config = (
            IMPALAConfig()
            .environment("CartPole-v1")
            .env_runners(num_env_runners=2, num_envs_per_env_runner=4)
            .training(train_batch_size=800)
            .learners(num_gpus_per_learner=1)
        )
        algo = config.build()
        assert algo.learner.device.type == "cuda"
        r = algo.train()
        assert np.isfinite(r["learner"]["total_loss"])
        algo.stop()
    finally:
        ray.shutdown()


@GPU
def test_data_gpu_ingest_pipeline():
    """iter_torch_batches lands batches on the GPU; the fused HIP
    normalize kernel handles image preprocessing."""
    ray.init(num_cpus=4, num_gpus=1, ignore_reinit_error=True)
    try:
        import ray_amd.data as rd
        from ray_amd import ops

        imgs = np.random.randint(0, 256, (64, 32, 32, 3), dtype=np.uint8)
        ds = rd.from_numpy(imgs, column="image")
        n = 0
        for batch in ds.iter_torch_batches(batch_size=16, device="cuda"):
            assert batch["image"].is_cuda
            mean = torch.tensor([0.5, 0.5, 0.5], device="cuda")
            std = torch.tensor([0.25, 0.25, 0.25], device="cuda")
            out = ops.img_normalize(
                batch["image"].to(torch.uint8).contiguous(), mean, std
            )
            assert out.shape == (16, 3, 32, 32) and out.dtype == torch.bfloat16
            n += len(batch["image"])
        assert n == 64
    finally:
        ray.shutdown()


@GPU
def test_train_gpu_worker_ddp_world1():
    """TorchTrainer with a GPU worker: prepare_model moves to device;
    single worker (multi-GPU scaling is the driver's round-end job)."""
    ray.init(num_cpus=4, num_gpus=1, ignore_reinit_error=True)
    try:
        from ray_amd.train import RunConfig, ScalingConfig
        from ray_amd.train.torch import TorchTrainer

        def loop(config):
            import torch as t

            import ray_amd.train as train
            from ray_amd.train.torch import get_device, prepare_model

            dev = get_device()
            assert dev.type == "cuda"
            model = prepare_model(t.nn.Linear(16, 4))
            opt = t.optim.SGD(model.parameters(), lr=0.1)
            x = t.randn(32, 16, device=dev)
            y = t.randn(32, 4, device=dev)
            for _ in range(3):
                loss = ((model(x) - y) ** 2).mean()
                loss.backward()
                opt.step()
                opt.zero_grad()
            train.report({"loss": float(loss)})

        t = TorchTrainer(
            loop,
            scaling_config=ScalingConfig(num_workers=1, use_gpu=True),
            run_config=RunConfig(name="gpu1", storage_path="/tmp/ray_amd_gpu_train"),
        )
        res = t.fit()
        assert res.error is None
        assert np.isfinite(res.metrics["loss"])
    finally:
        ray.shutdown()


@GPU
def test_serve_llm_gpu_deployment():
    """North-star config 5 (single replica on 1 GPU): Serve deployment
    with hipGraph decode."""
    ray.init(num_cpus=4, num_gpus=1, ignore_reinit_error=True)
    try:
        from ray_amd import serve
        from ray_amd.llm import build_llm_deployment

        app = build_llm_deployment(
            {"model_id": "llama-tiny", "max_seq_len": 128,
             "use_hip_graph": True, "use_gpu": True}
        )
        h = serve.run(app, name="llm_gpu", http=False)
        out = h.generate.remote([1, 2, 3, 4], 8).result(timeout_s=300)
        assert len(out["token_ids"]) == 8
        assert out["decode_tok_s"] > 0
        serve.shutdown()
    finally:
        ray.shutdown()


@GPU
def test_collective_rccl_single_rank():
    """Native RCCL group (csrc/rccl_comm.hip) init + the full op set
    with world_size=1 (the multi-GPU path is covered by the driver's
    8-GPU scaling run). Asserts the NATIVE module is in use — a silent
    torch-PG fallback fails this test."""
    ray.init(num_cpus=4, num_gpus=1, ignore_reinit_error=True)
    try:

        @ray.remote(num_gpus=1)
        class W:
            def __init__(self):
                from ray_amd.util import collective as col

                col.init_collective_group(1, 0, backend="rccl",
                                          group_name="rccl1")
                self.col = col
                g = col.collective._groups["rccl1"]
                assert isinstance(g, col.collective.NativeRcclGroup), g

            def ops(self):
                col = self.col
                out = {}
                t = torch.ones(128, device="cuda",
                               dtype=torch.bfloat16) * 3
                col.allreduce(t, "rccl1")
                out["allreduce_bf16"] = float(t.float().sum().item())
                b = torch.full((64,), 7.0, device="cuda")
                col.broadcast(b, 0, "rccl1")
                out["broadcast"] = float(b.sum().item())
                src = torch.arange(8, device="cuda", dtype=torch.float32)
                dst = [torch.zeros(8, device="cuda")]
                col.allgather(dst, src, "rccl1")
                out["allgather"] = dst[0].tolist()
                rs_out = torch.zeros(4, device="cuda")
                col.reducescatter(rs_out,
                                  [torch.ones(4, device="cuda") * 2],
                                  "rccl1")
                out["reducescatter"] = rs_out.tolist()
                # self send/recv must be inside a group call
                g = col.collective._groups["rccl1"]
                s = torch.arange(4, device="cuda", dtype=torch.float32)
                r = torch.zeros(4, device="cuda")
                g.comm.group_start()
                g.comm.send(s, 0)
                g.comm.recv(r, 0)
                g.comm.group_end()
                g.comm.synchronize()
                out["p2p_self"] = r.tolist()
                col.barrier("rccl1")
                return out

        w = W.remote()
        out = ray.get(w.ops.remote(), timeout=180)
        assert out["allreduce_bf16"] == 3 * 128
        assert out["broadcast"] == 7 * 64
        assert out["allgather"] == list(range(8))
        assert out["reducescatter"] == [2.0] * 4
        assert out["p2p_self"] == [0.0, 1.0, 2.0, 3.0]
    finally:
        ray.shutdown()


@GPU
def test_ring_attention_gpu_flash_path():
    """Single-rank ring attention on GPU takes the fused flash kernel
    path and matches the fp32 reference."""
    torch.manual_seed(0)
    q = torch.randn(2, 8, 256, 128, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(2, 8, 256, 128, device="cuda", dtype=torch.bfloat16)
    v = torch.randn_like(k)

    class _G:
        def rank(self):
            return 0

        def size(self):
            return 1

    from ray_amd.parallel import ring_attention

    out = ring_attention(q, k, v, group=_G(), causal=True)
    from ray_amd import ops

    ref, _ = ops.flash_attention_ref(q, k, v, causal=True)
    assert torch.allclose(out.float(), ref.float(), atol=4e-2, rtol=4e-2)
