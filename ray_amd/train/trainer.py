"""DataParallelTrainer: actor WorkerGroup + backend executor.

Reference counterparts: train/data_parallel_trainer.py,
train/_internal/backend_executor.py:86 (start :159, start_training :481,
get_next_results :613), train/_internal/worker_group.py:108, and the v2
controller's failure handling (v2/.../failure_policy.py:14): on worker
failure the group restarts from the latest checkpoint up to
FailureConfig.max_failures times.
"""
from __future__ import annotations

import os
import socket
import threading
import time
import traceback
from typing import Any, Callable, Dict, Optional

import cloudpickle

from .checkpoint import Checkpoint
from .config import FailureConfig, Result, RunConfig, ScalingConfig
from .session import TrainSession, _set_session


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


class TrainWorker:
    """Actor hosting one training worker (one rank)."""

    def __init__(self, rank: int, world_size: int, use_gpu: bool):
        self.rank = rank
        self.world_size = world_size
        self.use_gpu = use_gpu
        self.session: Optional[TrainSession] = None
        self.thread: Optional[threading.Thread] = None
        self.error: Optional[str] = None
        self.done = False
        self.local_rank = 0
        self.local_world_size = 1

    def node_info(self):
        """Topology probe for rendezvous wiring: which node this worker
        landed on, which GPU device ids it was leased, and a reachable
        IP for rank-0 to host the process-group store."""
        import ray_amd as ray

        return {
            "node_id": ray.get_runtime_context().get_node_id(),
            "gpu_ids": ray.get_gpu_ids(),
            "ip": os.environ.get("RAY_AMD_NODE_IP") or "127.0.0.1",
        }

    def reserve_port(self):
        """Pick a free TCP port on THIS worker's host (rank 0 hosts the
        rendezvous store, so the port must be free here, not on the
        driver's node)."""
        s = socket.socket()
        s.bind((os.environ.get("RAY_AMD_NODE_IP") or "127.0.0.1", 0))
        p = s.getsockname()[1]
        s.close()
        return p

    def setup_dist(self, master_addr: str, master_port: int, backend: str,
                   local_rank: int = 0, local_world_size: int = 1,
                   node_visible_gpus=None):
        import torch.distributed as dist

        self.local_rank = local_rank
        self.local_world_size = local_world_size
        # Scrub inherited torch-elastic state: if the DRIVER ran under
        # torchrun, workers inherit TORCHELASTIC_USE_AGENT_STORE and
        # would try to join torchrun's agent store instead of hosting
        # their own rendezvous (init then hangs until timeout).
        for k in ("TORCHELASTIC_USE_AGENT_STORE", "TORCHELASTIC_RUN_ID",
                  "TORCHELASTIC_RESTART_COUNT", "TORCHELASTIC_MAX_RESTARTS",
                  "TORCHELASTIC_ERROR_FILE", "GROUP_RANK", "GROUP_WORLD_SIZE",
                  "ROLE_RANK", "ROLE_WORLD_SIZE", "ROLE_NAME"):
            os.environ.pop(k, None)
        os.environ["MASTER_ADDR"] = master_addr
        os.environ["MASTER_PORT"] = str(master_port)
        os.environ["RANK"] = str(self.rank)
        os.environ["WORLD_SIZE"] = str(self.world_size)
        os.environ["LOCAL_RANK"] = str(local_rank)
        os.environ["LOCAL_WORLD_SIZE"] = str(local_world_size)
        if self.use_gpu and node_visible_gpus:
            # Reference behavior (train/_internal/utils
            # share_cuda_visible_devices): colocated workers see the
            # UNION of their devices, ordered by local rank, so RCCL
            # can open direct xGMI p2p between ranks instead of
            # bouncing through host memory. Must happen before this
            # process first touches HIP.
            vis = ",".join(str(g) for g in node_visible_gpus)
            os.environ["HIP_VISIBLE_DEVICES"] = vis
            os.environ["CUDA_VISIBLE_DEVICES"] = vis
        if self.world_size > 1:
            dist.init_process_group(
                backend=backend, rank=self.rank, world_size=self.world_size,
                init_method=f"tcp://{master_addr}:{master_port}",
            )
        if self.use_gpu:
            import torch

            torch.cuda.set_device(min(local_rank,
                                      torch.cuda.device_count() - 1))
        return True

    def start_training(self, fn_bytes: bytes, config: dict, storage_dir: str,
                       run_name: str, latest_ckpt_path: Optional[str],
                       shard_bytes: Optional[bytes]):
        fn = cloudpickle.loads(fn_bytes)
        shards = cloudpickle.loads(shard_bytes) if shard_bytes else {}
        self.session = TrainSession(
            self.rank, self.world_size, self.local_rank,
            self.local_world_size, storage_dir, run_name,
            latest_checkpoint=Checkpoint(latest_ckpt_path)
            if latest_ckpt_path
            else None,
            dataset_shards=shards,
        )
        self.done = False
        self.error = None

        def run():
            _set_session(self.session)
            try:
                if fn.__code__.co_argcount >= 1:
                    fn(config)
                else:
                    fn()
            except BaseException:
                self.error = traceback.format_exc()
            finally:
                self.done = True
                _set_session(None)

        self.thread = threading.Thread(target=run, daemon=True)
        self.thread.start()
        return True

    def fetch(self):
        out = []
        if self.session is not None:
            while not self.session.results_queue.empty():
                out.append(self.session.results_queue.get_nowait())
        return {"results": out, "done": self.done, "error": self.error}

    def latest_checkpoint_path(self):
        if self.session and self.session.latest_checkpoint:
            return self.session.latest_checkpoint.path
        return None

    def shutdown_dist(self):
        try:
            import torch.distributed as dist

            if dist.is_initialized():
                dist.destroy_process_group()
        except Exception:
            pass
        return True


class DataParallelTrainer:
    _default_backend = "gloo"

    def __init__(
        self,
        train_loop_per_worker: Callable,
        *,
        train_loop_config: Optional[Dict[str, Any]] = None,
        scaling_config: Optional[ScalingConfig] = None,
        run_config: Optional[RunConfig] = None,
        datasets: Optional[Dict[str, Any]] = None,
        backend_config=None,
        metadata=None,
        resume_from_checkpoint: Optional[Checkpoint] = None,
    ):
        self._fn = train_loop_per_worker
        self._config = train_loop_config or {}
        self.scaling_config = scaling_config or ScalingConfig()
        self.run_config = run_config or RunConfig(name=f"run_{int(time.time())}")
        self.datasets = datasets or {}
        self._resume = resume_from_checkpoint

    def _backend(self) -> str:
        return "nccl" if self.scaling_config.use_gpu else self._default_backend

    def fit(self) -> Result:
        import ray_amd as ray

        failure = self.run_config.failure_config or FailureConfig()
        retries = failure.max_failures
        storage_dir = self.run_config.resolved_storage_path()
        os.makedirs(storage_dir, exist_ok=True)
        latest_ckpt = self._resume.path if self._resume else None
        last_err: Optional[str] = None
        n_workers = self.scaling_config.num_workers
        while True:
            try:
                return self._fit_once(ray, storage_dir, latest_ckpt, n_workers)
            except _WorkerGroupError as e:
                last_err = e.error
                latest_ckpt = e.latest_ckpt or latest_ckpt
                if retries == 0:
                    return Result(
                        metrics=None,
                        checkpoint=Checkpoint(latest_ckpt) if latest_ckpt else None,
                        path=storage_dir,
                        error=RuntimeError(last_err),
                    )
                if retries > 0:
                    retries -= 1
                if self.scaling_config.elastic:
                    n_workers = self._elastic_world_size(ray, n_workers)

    def _elastic_world_size(self, ray, current: int) -> int:
        """Shrink to what the cluster can place right now (elastic
        restart-from-checkpoint; reference: ElasticScalingPolicy)."""
        import time as _t

        sc = self.scaling_config
        res = sc.worker_resources()
        _t.sleep(1.0)  # let node death propagate
        avail = ray.available_resources()
        fit = current
        for k, v in res.items():
            if v > 0:
                fit = min(fit, int(avail.get(k, 0) // v))
        n = max(sc.min_workers, min(current, fit))
        return max(n, 1)

    def _fit_once(self, ray, storage_dir: str, latest_ckpt: Optional[str],
                  n_override: Optional[int] = None) -> Result:
        sc = self.scaling_config
        n = n_override or sc.num_workers
        res = sc.worker_resources()
        from ..util import PlacementGroupSchedulingStrategy, placement_group, remove_placement_group

        pg = placement_group([dict(res) for _ in range(n)],
                             strategy=sc.placement_strategy)
        if not pg.wait(120):
            raise RuntimeError("could not create placement group for training")
        WorkerCls = ray.remote(TrainWorker)
        workers = [
            WorkerCls.options(
                num_cpus=res.get("CPU", 1),
                num_gpus=res.get("GPU", 0),
                scheduling_strategy=PlacementGroupSchedulingStrategy(
                    placement_group=pg, placement_group_bundle_index=i
                ),
            ).remote(i, n, sc.use_gpu)
            for i in range(n)
        ]
        try:
            try:
                # topology: local ranks per node + shared device
                # visibility, rendezvous hosted on rank-0's node
                infos = ray.get([w.node_info.remote() for w in workers],
                                timeout=120)
                by_node: Dict[str, list] = {}
                for i, inf in enumerate(infos):
                    by_node.setdefault(inf["node_id"], []).append(i)
                local_rank = {}
                local_ws = {}
                node_vis = {}
                for idxs in by_node.values():
                    vis = [g for i in idxs for g in infos[i]["gpu_ids"]]
                    for lr, i in enumerate(idxs):
                        local_rank[i] = lr
                        local_ws[i] = len(idxs)
                        node_vis[i] = vis
                master_addr = infos[0]["ip"]
                port = ray.get(workers[0].reserve_port.remote(), timeout=60)
                ray.get([
                    w.setup_dist.remote(
                        master_addr, port, self._backend(),
                        local_rank[i], local_ws[i], node_vis[i],
                    )
                    for i, w in enumerate(workers)
                ], timeout=180)

                shard_payloads = self._make_shards(n)
                fn_bytes = cloudpickle.dumps(self._fn)
                ray.get([
                    w.start_training.remote(
                        fn_bytes, self._config, storage_dir,
                        self.run_config.name or "run", latest_ckpt,
                        shard_payloads[i],
                    )
                    for i, w in enumerate(workers)
                ], timeout=180)
            except (ray.exceptions.RayActorError, ray.exceptions.RayError) as e:
                raise _WorkerGroupError(
                    f"worker group setup failed: {e}", latest_ckpt
                )

            rows = []
            kept_checkpoints = []
            last_metrics = None
            latest_checkpoint_path = latest_ckpt
            while True:
                try:
                    states = ray.get([w.fetch.remote() for w in workers],
                                     timeout=120)
                except ray.exceptions.RayError as e:
                    raise _WorkerGroupError(
                        f"a training worker died: {e}", latest_checkpoint_path
                    )
                for st in states:
                    if st["error"]:
                        raise _WorkerGroupError(st["error"], latest_checkpoint_path)
                for r in states[0]["results"]:
                    rows.append(r["metrics"])
                    last_metrics = r["metrics"]
                    if r["checkpoint_path"]:
                        latest_checkpoint_path = r["checkpoint_path"]
                        kept_checkpoints.append(r["checkpoint_path"])
                        self._enforce_keep(kept_checkpoints)
                if all(st["done"] for st in states):
                    break
                time.sleep(0.05)
            ckpt = (
                Checkpoint(latest_checkpoint_path)
                if latest_checkpoint_path
                else None
            )
            result = Result(
                metrics=last_metrics, checkpoint=ckpt, path=storage_dir
            )
            result.metrics_dataframe = rows
            return result
        finally:
            for w in workers:
                try:
                    w.shutdown_dist.remote()
                except Exception:
                    pass
            time.sleep(0.1)
            for w in workers:
                try:
                    ray.kill(w)
                except Exception:
                    pass
            try:
                remove_placement_group(pg)
            except Exception:
                pass

    def _enforce_keep(self, kept: list):
        """CheckpointConfig.num_to_keep (reference:
        _internal/checkpoint_manager.py keep-top-k by recency here)."""
        import shutil

        cc = self.run_config.checkpoint_config
        if not cc or not cc.num_to_keep:
            return
        while len(kept) > cc.num_to_keep:
            victim = kept.pop(0)
            shutil.rmtree(victim, ignore_errors=True)

    def _make_shards(self, n: int):
        """Split datasets across workers (streaming_split equivalent)."""
        payloads = [dict() for _ in range(n)]
        for name, ds in self.datasets.items():
            if hasattr(ds, "split"):
                shards = ds.split(n)
                for i in range(n):
                    payloads[i][name] = shards[i]
            else:
                for i in range(n):
                    payloads[i][name] = ds
        return [cloudpickle.dumps(p) if p else None for p in payloads]


def _noop():
    pass


class _WorkerGroupError(Exception):
    def __init__(self, error: str, latest_ckpt: Optional[str]):
        super().__init__(error)
        self.error = error
        self.latest_ckpt = latest_ckpt
