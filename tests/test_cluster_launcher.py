"""`ray_amd up/down` cluster launcher, local provider (reference:
`ray up cluster.yaml` / autoscaler commands)."""
import json
import os
import subprocess
import sys
import time


def test_up_and_down_local_cluster(tmp_path):
    cfg = tmp_path / "cluster.yaml"
    cfg.write_text(
        """
cluster_name: launcher_test
provider:
  type: local
head_node:
  num_cpus: 2
worker_nodes:
  count: 2
  num_cpus: 1
  resources: {scratch: 1}
"""
    )
    out = subprocess.run(
        [sys.executable, "-m", "ray_amd", "up", str(cfg)],
        capture_output=True, text=True, timeout=120,
    )
    assert out.returncode == 0, out.stdout + out.stderr
    assert "head + 2 workers" in out.stdout
    session = out.stdout.split("session_dir=")[1].splitlines()[0].strip()
    try:
        # a fresh driver sees all three nodes + the custom resource
        script = f"""
import time
import ray_amd as ray
ray.init(address={session!r})
deadline = time.time() + 30
while time.time() < deadline:
    alive = [n for n in ray.nodes() if n["Alive"]]
    if len(alive) >= 3:
        break
    time.sleep(0.2)
print("NODES", len(alive))
print("SCRATCH", ray.cluster_resources().get("scratch"))
ray.shutdown(_exiting_interpreter=True)
"""
        r = subprocess.run([sys.executable, "-c", script],
                           capture_output=True, text=True, timeout=120)
        assert "NODES 3" in r.stdout, r.stdout + r.stderr
        assert "SCRATCH 2" in r.stdout, r.stdout
    finally:
        down = subprocess.run(
            [sys.executable, "-m", "ray_amd", "down", session],
            capture_output=True, text=True, timeout=60,
        )
        assert down.returncode == 0
        # daemons are gone
        pids = json.load(open(os.path.join(session, "head_pids")))
        deadline = time.time() + 15
        while time.time() < deadline:
            left = [p for p in pids.values()
                    if os.path.exists(f"/proc/{p}")]
            if not left:
                break
            time.sleep(0.2)
        # gcs exits after nodes die; allow the raylet teardown cascade
        assert not [p for p in (pids["gcs"], pids["raylet"])
                    if os.path.exists(f"/proc/{p}")], pids
