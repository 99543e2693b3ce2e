"""Job submission API (reference: python/ray/job_submission/ +
dashboard job manager)."""
import sys
import time

import pytest

import ray_amd as ray
from ray_amd.job_submission import JobStatus, JobSubmissionClient


def _wait_status(client, jid, want, timeout=60):
    deadline = time.time() + timeout
    while time.time() < deadline:
        st = client.get_job_status(jid)
        if st in want:
            return st
        time.sleep(0.2)
    raise AssertionError(f"job {jid} stuck in {st}, wanted {want}")


def test_job_lifecycle(ray_start_regular):
    client = JobSubmissionClient()

    # success path: the job driver joins THIS cluster via RAY_AMD_ADDRESS
    jid = client.submit_job(
        entrypoint=(
            f"{sys.executable} -c \""
            "import ray_amd as ray; ray.init();\n"
            "import os\n"
            "print('env:', os.environ['MY_FLAG'])\n"
            "@ray.remote\n"
            "def f(): return 42\n"
            "print('answer:', ray.get(f.remote()))\""
        ),
        runtime_env={"env_vars": {"MY_FLAG": "yes"}},
    )
    assert _wait_status(client, jid, {JobStatus.SUCCEEDED,
                                      JobStatus.FAILED}) == JobStatus.SUCCEEDED
    logs = client.get_job_logs(jid)
    assert "env: yes" in logs and "answer: 42" in logs

    # failure path
    bad = client.submit_job(
        entrypoint=f"{sys.executable} -c 'raise SystemExit(3)'"
    )
    assert _wait_status(client, bad, {JobStatus.FAILED}) == JobStatus.FAILED
    assert "exit code 3" in client.get_job_info(bad).message

    # stop path
    slow = client.submit_job(
        entrypoint=f"{sys.executable} -c 'import time; time.sleep(600)'"
    )
    _wait_status(client, slow, {JobStatus.RUNNING})
    assert client.stop_job(slow)
    assert _wait_status(client, slow, {JobStatus.STOPPED}) == JobStatus.STOPPED

    ids = {j.submission_id for j in client.list_jobs()}
    assert {jid, bad, slow} <= ids

    with pytest.raises(Exception):
        client.submit_job(entrypoint="true", submission_id=jid)  # dup id
    assert client.delete_job(bad)


def test_job_cli(ray_start_regular, tmp_path):
    """`ray_amd job submit/status/list` CLI against a live cluster."""
    import io
    from contextlib import redirect_stdout

    from ray_amd._core import runtime as rtmod
    from ray_amd.scripts import main as cli

    addr = rtmod.global_runtime().session_dir
    buf = io.StringIO()
    with redirect_stdout(buf):
        rc = cli(["job", "submit", "--address", addr, "--",
                  sys.executable, "-c", "print(40+2)"])
    assert rc == 0
    out = buf.getvalue()
    assert "42" in out

    buf = io.StringIO()
    with redirect_stdout(buf):
        assert cli(["job", "list", "--address", addr]) == 0
    assert "SUCCEEDED" in buf.getvalue()
