"""Jobs API, Ray-Client mode, and pipeline parallelism walkthrough.

Run: python examples/07_jobs_client_pipeline.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

import ray_amd as ray


def jobs_demo():
    from ray_amd.job_submission import JobStatus, JobSubmissionClient

    client = JobSubmissionClient()
    jid = client.submit_job(
        entrypoint=(
            f"{sys.executable} -c \"import ray_amd as ray; ray.init(); "
            "print('job sees CPUs:', ray.cluster_resources()['CPU'])\""
        ),
    )
    for chunk in client.tail_job_logs(jid, timeout_s=120):
        print(chunk, end="")
    print("job status:", client.get_job_status(jid))
    assert client.get_job_status(jid) == JobStatus.SUCCEEDED


def client_mode_demo():
    """A thin client driving the cluster over TCP (run the server part
    inside any driver; here: same process for brevity)."""
    from ray_amd.client.server import ClientServer

    port = ClientServer(port=0).start()
    print(f"client server on 127.0.0.1:{port} "
          f"(connect with ray.init('ray_amd://127.0.0.1:{port}'))")


def pipeline_demo():
    import torch

    from ray_amd.parallel.pipeline import Pipeline

    def stage0():
        torch.manual_seed(0)
        return torch.nn.Sequential(torch.nn.Linear(16, 64), torch.nn.Tanh())

    def stage1():
        torch.manual_seed(1)
        return torch.nn.Linear(64, 1)

    rng = np.random.default_rng(0)
    X = rng.normal(size=(64, 16)).astype(np.float32)
    Y = (X[:, :1] * 2 + 0.1).astype(np.float32)
    pipe = Pipeline([stage0, stage1], lr=0.05, num_microbatches=4)
    losses = [pipe.step(X, Y) for _ in range(10)]
    print(f"pipeline-parallel training: loss {losses[0]:.4f} -> "
          f"{losses[-1]:.4f}")
    assert losses[-1] < losses[0]


if __name__ == "__main__":
    ray.init(num_cpus=4)
    try:
        jobs_demo()
        client_mode_demo()
        pipeline_demo()
        print("all demos OK")
    finally:
        ray.shutdown()
