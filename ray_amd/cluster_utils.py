"""Multi-node-on-localhost test harness.

Reference counterpart: python/ray/cluster_utils.py:141 `Cluster` —
spins up multiple raylets as separate processes on one machine
(add_node :208, remove_node :292), the reference's primary
distributed-test strategy.
"""
from __future__ import annotations

import json
import os
from typing import Dict, Optional

from ._core import node as _node


class NodeHandle:
    def __init__(self, proc, addr, node_id, resources):
        self.proc = proc
        self.addr = addr
        self.node_id = node_id
        self.resources = resources

    @property
    def node_id_hex(self):
        return self.node_id.hex()


class Cluster:
    def __init__(self, initialize_head: bool = True, head_node_args: Optional[dict] = None,
                 connect: bool = False, tcp: bool = False):
        """tcp=True runs every daemon on TCP (127.0.0.1) and gives each
        added node its OWN shm directory — the two-"machine" harness:
        object pulls must move bytes through the chunked raylet
        transfer, never a shared filesystem path."""
        self.tcp = tcp
        self.session_dir = _node.new_session_dir()
        os.environ["RAY_AMD_SHM_DIR"] = _node.session_shm_dir(self.session_dir)
        _node.export_driver_pythonpath()
        gcs_env = None
        if tcp:
            os.environ["RAY_AMD_NODE_IP"] = "127.0.0.1"
            gcs_env = dict(os.environ)
        self.gcs_proc, self.gcs_addr = _node.start_gcs(self.session_dir,
                                                       env=gcs_env)
        self.nodes = []
        self.head_node: Optional[NodeHandle] = None
        if initialize_head:
            self.add_node(**(head_node_args or {}))
        if connect:
            self.connect()

    @property
    def address(self) -> str:
        return self.session_dir

    def add_node(self, num_cpus: float = 4, num_gpus: float = 0,
                 resources: Optional[Dict[str, float]] = None,
                 object_store_memory: Optional[int] = None,
                 labels: Optional[dict] = None, **kwargs) -> NodeHandle:
        res = dict(resources or {})
        res.setdefault("CPU", num_cpus)
        res.setdefault("GPU", num_gpus)
        res.setdefault("memory", 16 * 2**30)
        name = f"node{len(self.nodes)}"
        env_extra = dict(kwargs.pop("env_extra", None) or {})
        if self.tcp:
            env_extra.setdefault("RAY_AMD_NODE_IP", "127.0.0.1")
            if self.nodes:  # non-head nodes get their own "machine" shm
                shm = os.path.join(
                    _node.session_shm_dir(self.session_dir) + f"_{name}"
                )
                os.makedirs(shm, exist_ok=True)
                env_extra.setdefault("RAY_AMD_SHM_DIR", shm)
        proc, addr, node_id = _node.start_raylet(
            self.session_dir, self.gcs_addr, res, node_name=name,
            labels=labels, object_store_memory=object_store_memory,
            env_extra=env_extra or None,
        )
        h = NodeHandle(proc, addr, node_id, res)
        self.nodes.append(h)
        if self.head_node is None:
            self.head_node = h
            with open(os.path.join(self.session_dir, "session.json"), "w") as f:
                json.dump(
                    {
                        "gcs_addr": self.gcs_addr,
                        "raylet_addr": addr,
                        "node_id": node_id.hex(),
                        "session_dir": self.session_dir,
                    },
                    f,
                )
        return h

    def remove_node(self, node: NodeHandle, allow_graceful: bool = True):
        try:
            node.proc.terminate()
            node.proc.wait(5)
        except Exception:
            try:
                node.proc.kill()
            except Exception:
                pass
        self.nodes.remove(node)

    def connect(self):
        import ray_amd as ray

        return ray.init(address=self.session_dir)

    def wait_for_nodes(self, timeout: float = 30):
        import time

        import ray_amd as ray

        deadline = time.time() + timeout
        while time.time() < deadline:
            try:
                alive = [n for n in ray.nodes() if n["Alive"]]
                if len(alive) >= len(self.nodes):
                    return
            except Exception:
                pass
            time.sleep(0.1)
        raise TimeoutError("cluster nodes did not come up")

    def shutdown(self):
        import ray_amd as ray

        try:
            if ray.is_initialized():
                ray.shutdown()
        except Exception:
            pass
        for n in list(self.nodes):
            try:
                n.proc.terminate()
            except Exception:
                pass
        for n in list(self.nodes):
            try:
                n.proc.wait(3)
            except Exception:
                try:
                    n.proc.kill()
                except Exception:
                    pass
        try:
            self.gcs_proc.terminate()
            self.gcs_proc.wait(3)
        except Exception:
            try:
                self.gcs_proc.kill()
            except Exception:
                pass
        import shutil

        try:
            import glob

            base = _node.session_shm_dir(self.session_dir)
            for d in [base] + glob.glob(base + "_*"):
                shutil.rmtree(d, ignore_errors=True)
        except Exception:
            pass
        shutil.rmtree(self.session_dir, ignore_errors=True)


class AutoscalingCluster(Cluster):
    """Placeholder parity for reference AutoscalingCluster (:26)."""
