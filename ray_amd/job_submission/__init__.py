"""Job submission API (reference: python/ray/job_submission/ +
dashboard/modules/job/job_manager.py:57 — jobs run as driver
subprocesses supervised by a manager actor; clients poll status/logs).

    from ray_amd.job_submission import JobSubmissionClient, JobStatus

    client = JobSubmissionClient()          # in-cluster (or address=...)
    job_id = client.submit_job(
        entrypoint="python my_script.py",
        runtime_env={"env_vars": {"X": "1"}},
    )
    client.get_job_status(job_id)           # PENDING/RUNNING/...
    client.get_job_logs(job_id)
"""
from __future__ import annotations

import os
import subprocess
import threading
import time
import uuid
from typing import Dict, List, Optional

JOB_MANAGER_NAME = "_RAY_AMD_JOB_MANAGER"
JOB_NAMESPACE = "_jobs"


class JobStatus:
    PENDING = "PENDING"
    RUNNING = "RUNNING"
    SUCCEEDED = "SUCCEEDED"
    FAILED = "FAILED"
    STOPPED = "STOPPED"


class JobDetails:
    def __init__(self, d: dict):
        self.submission_id = d["submission_id"]
        self.entrypoint = d["entrypoint"]
        self.status = d["status"]
        self.start_time = d.get("start_time")
        self.end_time = d.get("end_time")
        self.message = d.get("message", "")

    def __repr__(self):
        return f"JobDetails({self.submission_id}, {self.status})"


class _JobManagerActor:
    """Supervisor (reference: JobManager + JobSupervisor actor): one
    subprocess per job, logs to the session log dir."""

    def __init__(self):
        self.jobs: Dict[str, dict] = {}
        self._procs: Dict[str, subprocess.Popen] = {}
        from ray_amd._core import runtime as rtmod

        rt = rtmod.global_runtime()
        self._log_dir = os.path.join(rt.session_dir, "logs", "jobs")
        os.makedirs(self._log_dir, exist_ok=True)
        self._session_dir = rt.session_dir

    def submit(self, submission_id: str, entrypoint: str,
               env_vars: Optional[dict], working_dir: Optional[str]):
        if submission_id in self.jobs:
            raise ValueError(f"job {submission_id!r} already exists")
        log_path = os.path.join(self._log_dir, f"{submission_id}.log")
        env = dict(os.environ)
        env.update({str(k): str(v) for k, v in (env_vars or {}).items()})
        # the job driver connects to THIS cluster
        env["RAY_AMD_ADDRESS"] = self._session_dir
        env["PYTHONUNBUFFERED"] = "1"
        self.jobs[submission_id] = {
            "submission_id": submission_id,
            "entrypoint": entrypoint,
            "status": JobStatus.PENDING,
            "start_time": time.time(),
            "log_path": log_path,
            "message": "",
        }
        out = open(log_path, "ab", buffering=0)
        try:
            proc = subprocess.Popen(
                entrypoint, shell=True, env=env,
                cwd=working_dir or os.getcwd(),
                stdout=out, stderr=subprocess.STDOUT,
                start_new_session=True,
            )
        except Exception as e:
            self.jobs[submission_id]["status"] = JobStatus.FAILED
            self.jobs[submission_id]["message"] = str(e)
            return submission_id
        self._procs[submission_id] = proc
        self.jobs[submission_id]["status"] = JobStatus.RUNNING
        threading.Thread(
            target=self._wait, args=(submission_id, proc), daemon=True
        ).start()
        return submission_id

    def _wait(self, submission_id: str, proc: subprocess.Popen):
        rc = proc.wait()
        j = self.jobs[submission_id]
        j["end_time"] = time.time()
        if j["status"] == JobStatus.STOPPED:
            return
        j["status"] = JobStatus.SUCCEEDED if rc == 0 else JobStatus.FAILED
        if rc != 0:
            j["message"] = f"exit code {rc}"

    def status(self, submission_id: str) -> str:
        return self.jobs[submission_id]["status"]

    def info(self, submission_id: str) -> dict:
        return dict(self.jobs[submission_id])

    def list(self) -> List[dict]:
        return [dict(j) for j in self.jobs.values()]

    def logs(self, submission_id: str) -> str:
        p = self.jobs[submission_id]["log_path"]
        try:
            with open(p, "r", errors="replace") as f:
                return f.read()
        except FileNotFoundError:
            return ""

    def stop(self, submission_id: str) -> bool:
        proc = self._procs.get(submission_id)
        j = self.jobs.get(submission_id)
        if proc is None or j is None or proc.poll() is not None:
            return False
        j["status"] = JobStatus.STOPPED
        j["message"] = "stopped by user"
        try:
            os.killpg(proc.pid, 15)  # the exact pgid we created
        except ProcessLookupError:
            pass
        return True

    def delete(self, submission_id: str) -> bool:
        if self.jobs.get(submission_id, {}).get("status") in (
            JobStatus.RUNNING, JobStatus.PENDING
        ):
            raise RuntimeError("stop the job before deleting it")
        self._procs.pop(submission_id, None)
        return self.jobs.pop(submission_id, None) is not None


class JobSubmissionClient:
    """Submit/inspect jobs (reference: job_submission/JobSubmissionClient
    — REST there; direct manager-actor calls here, same surface)."""

    def __init__(self, address: Optional[str] = None):
        import ray_amd as ray

        if not ray.is_initialized():
            ray.init(
                address=address
                or os.environ.get("RAY_AMD_ADDRESS", "auto"),
                ignore_reinit_error=True,
            )
        self._ray = ray
        self._mgr = ray.remote(_JobManagerActor).options(
            name=JOB_MANAGER_NAME, namespace=JOB_NAMESPACE,
            get_if_exists=True, lifetime="detached", max_restarts=1,
        ).remote()

    def submit_job(self, *, entrypoint: str, runtime_env: Optional[dict] =
                   None, submission_id: Optional[str] = None,
                   **kwargs) -> str:
        sid = submission_id or f"raysubmit_{uuid.uuid4().hex[:12]}"
        renv = runtime_env or {}
        return self._ray.get(
            self._mgr.submit.remote(
                sid, entrypoint, renv.get("env_vars"),
                renv.get("working_dir"),
            ),
            timeout=60,
        )

    def get_job_status(self, submission_id: str) -> str:
        return self._ray.get(
            self._mgr.status.remote(submission_id), timeout=30
        )

    def get_job_info(self, submission_id: str) -> JobDetails:
        return JobDetails(
            self._ray.get(self._mgr.info.remote(submission_id), timeout=30)
        )

    def list_jobs(self) -> List[JobDetails]:
        return [
            JobDetails(d)
            for d in self._ray.get(self._mgr.list.remote(), timeout=30)
        ]

    def get_job_logs(self, submission_id: str) -> str:
        return self._ray.get(
            self._mgr.logs.remote(submission_id), timeout=30
        )

    def stop_job(self, submission_id: str) -> bool:
        return self._ray.get(
            self._mgr.stop.remote(submission_id), timeout=30
        )

    def delete_job(self, submission_id: str) -> bool:
        return self._ray.get(
            self._mgr.delete.remote(submission_id), timeout=30
        )

    def tail_job_logs(self, submission_id: str, timeout_s: float = 600.0):
        """Generator yielding log increments until the job finishes."""
        seen = 0
        deadline = time.monotonic() + timeout_s
        while time.monotonic() < deadline:
            logs = self.get_job_logs(submission_id)
            if len(logs) > seen:
                yield logs[seen:]
                seen = len(logs)
            st = self.get_job_status(submission_id)
            if st in (JobStatus.SUCCEEDED, JobStatus.FAILED,
                      JobStatus.STOPPED):
                tail = self.get_job_logs(submission_id)
                if len(tail) > seen:
                    yield tail[seen:]
                return
            time.sleep(0.3)
