"""Process orchestration: start/stop GCS, raylets, driver runtime.

Counterpart of the reference's python/ray/_private/node.py:68
(start_gcs_server :1358, start_raylet :1422, head vs worker startup).
"""
from __future__ import annotations

import json
import os
import shutil
import subprocess
import sys
import tempfile
import time
import uuid
from typing import Dict, Optional

RAY_AMD_TMP = os.environ.get("RAY_AMD_TMPDIR", "/tmp/ray_amd")


def _shm_base() -> str:
    if os.path.isdir("/dev/shm") and os.access("/dev/shm", os.W_OK):
        return "/dev/shm"
    return tempfile.gettempdir()


def new_session_dir() -> str:
    sid = time.strftime("%Y%m%d-%H%M%S-") + uuid.uuid4().hex[:8]
    d = os.path.join(RAY_AMD_TMP, f"session_{sid}")
    os.makedirs(os.path.join(d, "sock"), exist_ok=True)
    os.makedirs(os.path.join(d, "logs"), exist_ok=True)
    shm = os.path.join(_shm_base(), f"ray_amd_{sid}")
    os.makedirs(shm, exist_ok=True)
    with open(os.path.join(d, "shm_dir"), "w") as f:
        f.write(shm)
    return d


def session_shm_dir(session_dir: str) -> str:
    with open(os.path.join(session_dir, "shm_dir")) as f:
        return f.read().strip()


def detect_gpus() -> int:
    vis = os.environ.get("HIP_VISIBLE_DEVICES") or os.environ.get(
        "CUDA_VISIBLE_DEVICES"
    )
    if vis is not None:
        return len([x for x in vis.split(",") if x != ""])
    # ROCm: one entry per GPU under /sys/class/kfd topology with gfx target
    try:
        import glob

        n = 0
        for p in glob.glob("/sys/class/kfd/kfd/topology/nodes/*/properties"):
            with open(p) as f:
                txt = f.read()
            if "simd_count" in txt:
                for line in txt.splitlines():
                    if line.startswith("simd_count"):
                        if int(line.split()[1]) > 0:
                            n += 1
        return n
    except Exception:
        return 0


def start_gcs(session_dir: str, env=None) -> tuple:
    sock = os.path.join(session_dir, "sock", "gcs")
    logf = open(os.path.join(session_dir, "logs", "gcs.log"), "ab", buffering=0)
    persist = os.path.join(session_dir, "gcs_state.bin")
    proc = subprocess.Popen(
        [sys.executable, "-m", "ray_amd._core.gcs", sock, persist],
        stdout=logf,
        stderr=subprocess.STDOUT,
        env=env or os.environ.copy(),
        start_new_session=True,
    )
    deadline = time.time() + 20
    while not os.path.exists(sock):
        if proc.poll() is not None:
            raise RuntimeError("GCS failed to start; see logs/gcs.log")
        if time.time() > deadline:
            raise RuntimeError("GCS start timed out")
        time.sleep(0.01)
    import stat

    if stat.S_ISSOCK(os.stat(sock).st_mode):
        return proc, "unix:" + sock
    # TCP mode (RAY_AMD_NODE_IP): the file carries the address
    with open(sock) as f:
        return proc, f.read().strip()


def start_raylet(
    session_dir: str,
    gcs_addr: str,
    resources: Dict[str, float],
    node_name: str = "",
    labels: Optional[dict] = None,
    object_store_memory: Optional[int] = None,
    env_extra: Optional[dict] = None,
) -> tuple:
    ready = os.path.join(
        session_dir, "sock", f"raylet_ready_{uuid.uuid4().hex[:6]}"
    )
    env = os.environ.copy()
    env["RAY_AMD_SHM_DIR"] = session_shm_dir(session_dir)
    if env_extra:
        env.update(env_extra)
    logf = open(
        os.path.join(session_dir, "logs", f"raylet_{node_name or 'n'}.log"),
        "ab",
        buffering=0,
    )
    proc = subprocess.Popen(
        [
            sys.executable,
            "-m",
            "ray_amd._core.raylet",
            "--session-dir",
            session_dir,
            "--gcs",
            gcs_addr,
            "--resources",
            json.dumps(resources),
            "--node-name",
            node_name,
            "--labels",
            json.dumps(labels or {}),
            "--object-store-memory",
            str(object_store_memory or 0),
            "--ready-file",
            ready,
        ],
        stdout=logf,
        stderr=subprocess.STDOUT,
        env=env,
        start_new_session=True,
    )
    deadline = time.time() + 30
    while not os.path.exists(ready):
        if proc.poll() is not None:
            raise RuntimeError("raylet failed to start; see logs")
        if time.time() > deadline:
            raise RuntimeError("raylet start timed out")
        time.sleep(0.01)
    for _ in range(50):
        with open(ready) as f:
            parts = f.read().strip().split("\n")
        if len(parts) == 2 and parts[1]:
            return proc, parts[0], bytes.fromhex(parts[1])
        time.sleep(0.01)
    raise RuntimeError("raylet ready file malformed")


class LocalCluster:
    """A head 'cluster' on this machine: GCS + one raylet (+ extras via
    cluster_utils.Cluster). Written to session.json for address='auto'."""

    def __init__(self, session_dir, gcs_proc, gcs_addr, raylet_proc, raylet_addr, node_id):
        self.session_dir = session_dir
        self.gcs_proc = gcs_proc
        self.gcs_addr = gcs_addr
        self.raylet_proc = raylet_proc
        self.raylet_addr = raylet_addr
        self.node_id = node_id
        self.extra_raylets = []

    def write_session_file(self):
        with open(os.path.join(self.session_dir, "session.json"), "w") as f:
            json.dump(
                {
                    "gcs_addr": self.gcs_addr,
                    "raylet_addr": self.raylet_addr,
                    "node_id": self.node_id.hex(),
                    "session_dir": self.session_dir,
                },
                f,
            )
        latest = os.path.join(RAY_AMD_TMP, "latest_session")
        try:
            with open(latest, "w") as f:
                f.write(self.session_dir)
        except OSError:
            pass

    def shutdown(self):
        for proc in [p for p, _, _ in self.extra_raylets] + [
            self.raylet_proc,
            self.gcs_proc,
        ]:
            try:
                proc.terminate()
            except Exception:
                pass
        t0 = time.time()
        for proc in [p for p, _, _ in self.extra_raylets] + [
            self.raylet_proc,
            self.gcs_proc,
        ]:
            try:
                proc.wait(max(0.1, 3 - (time.time() - t0)))
            except Exception:
                try:
                    proc.kill()
                except Exception:
                    pass
        try:
            shm = session_shm_dir(self.session_dir)
            shutil.rmtree(shm, ignore_errors=True)
        except Exception:
            pass
        shutil.rmtree(self.session_dir, ignore_errors=True)


def export_driver_pythonpath():
    """Propagate the driver's sys.path to workers (reference parity:
    workers inherit the driver's import environment via PYTHONPATH /
    working_dir runtime env)."""
    import sys

    os.environ["RAY_AMD_PYTHONPATH"] = os.pathsep.join(
        p for p in sys.path if p
    )


def start_local_cluster(
    num_cpus: Optional[float] = None,
    num_gpus: Optional[float] = None,
    resources: Optional[Dict[str, float]] = None,
    object_store_memory: Optional[int] = None,
    labels: Optional[dict] = None,
) -> LocalCluster:
    session_dir = new_session_dir()
    os.environ["RAY_AMD_SHM_DIR"] = session_shm_dir(session_dir)
    export_driver_pythonpath()
    gcs_proc, gcs_addr = start_gcs(session_dir)
    res = dict(resources or {})
    res.setdefault("CPU", num_cpus if num_cpus is not None else os.cpu_count())
    res.setdefault("GPU", num_gpus if num_gpus is not None else detect_gpus())
    res.setdefault("memory", 64 * 2**30)
    raylet_proc, raylet_addr, node_id = start_raylet(
        session_dir,
        gcs_addr,
        res,
        node_name="head",
        labels=labels,
        object_store_memory=object_store_memory,
    )
    cluster = LocalCluster(
        session_dir, gcs_proc, gcs_addr, raylet_proc, raylet_addr, node_id
    )
    cluster.write_session_file()
    return cluster


def find_session(address: str) -> dict:
    if address in ("auto", "", None):
        latest = os.path.join(RAY_AMD_TMP, "latest_session")
        with open(latest) as f:
            session_dir = f.read().strip()
    else:
        session_dir = address
    with open(os.path.join(session_dir, "session.json")) as f:
        return json.load(f)
