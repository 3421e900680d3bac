"""Job submission: run driver commands as supervised cluster jobs.

Role parity: reference python/ray/dashboard/modules/job/job_manager.py:538
(JobManager.submit_job → one detached JobSupervisor actor per job,
job_supervisor.py:58 — runs the entrypoint as a fate-shared subprocess,
streams logs, records status) and python/ray/job_submission (SDK). Status
and logs live in the GCS KV (namespace "job").
"""
from __future__ import annotations

import json
import os
import subprocess
import time
import uuid
from typing import Any, Dict, List, Optional


class JobSupervisor:
    """Detached actor: owns one job's driver subprocess."""

    def __init__(self, job_id: str, entrypoint: str,
                 runtime_env: Optional[dict], gcs_addr: str):
        self.job_id = job_id
        self.entrypoint = entrypoint
        self.runtime_env = runtime_env or {}
        self.gcs_addr = gcs_addr
        self.proc: Optional[subprocess.Popen] = None
        self.log_path = f"/tmp/antray/job_{job_id}.log"

    def _put_status(self, status: str, message: str = ""):
        import ant_ray_amd as ray
        from ant_ray_amd._private.worker import global_worker

        cw = global_worker.core_worker
        view = {
            "submission_id": self.job_id,
            "status": status,
            "message": message,
            "entrypoint": self.entrypoint,
            "start_time": getattr(self, "_start_ts", None),
            "end_time": time.time() if status in ("SUCCEEDED", "FAILED",
                                                  "STOPPED") else None,
        }
        cw.io.run(cw.gcs.call("kv_put", {
            "ns": "job", "key": f"status:{self.job_id}".encode(),
            "value": json.dumps(view).encode(), "overwrite": True,
        }))

    def run(self):
        """Launch the entrypoint; returns immediately (poll via status)."""
        env = dict(os.environ)
        env["RAY_ADDRESS"] = self.gcs_addr
        env["ANTRAY_ADDRESS"] = self.gcs_addr
        env["RAY_JOB_ID"] = self.job_id
        cwd = self.runtime_env.get("working_dir") or os.getcwd()
        env.update({k: str(v) for k, v in
                    (self.runtime_env.get("env_vars") or {}).items()})
        self._start_ts = time.time()
        os.makedirs("/tmp/antray", exist_ok=True)
        logf = open(self.log_path, "wb")
        self.proc = subprocess.Popen(
            self.entrypoint, shell=True, cwd=cwd, env=env,
            stdout=logf, stderr=subprocess.STDOUT, start_new_session=True,
        )
        self._put_status("RUNNING")
        import threading

        threading.Thread(target=self._wait, daemon=True).start()
        return True

    def _wait(self):
        rc = self.proc.wait()
        self._put_status("SUCCEEDED" if rc == 0 else "FAILED",
                         message=f"exit code {rc}")

    def stop(self):
        if self.proc and self.proc.poll() is None:
            import signal

            try:
                os.killpg(self.proc.pid, signal.SIGTERM)
            except Exception:
                self.proc.terminate()
            self._put_status("STOPPED")
        return True

    def logs(self) -> str:
        try:
            with open(self.log_path, "rb") as f:
                return f.read().decode(errors="replace")
        except FileNotFoundError:
            return ""


class JobSubmissionClient:
    """SDK (parity ray.job_submission.JobSubmissionClient). Talks to the
    connected cluster directly (no REST hop; the dashboard head exposes the
    same over HTTP)."""

    def __init__(self, address: Optional[str] = None):
        import ant_ray_amd as ray

        if not ray.is_initialized():
            ray.init(address=address or os.environ.get("RAY_ADDRESS", "auto"),
                     ignore_reinit_error=True)

    def _cw(self):
        from ant_ray_amd._private.worker import global_worker

        return global_worker.core_worker

    def submit_job(self, *, entrypoint: str, submission_id: Optional[str] = None,
                   runtime_env: Optional[dict] = None,
                   metadata: Optional[dict] = None, **_) -> str:
        import ant_ray_amd as ray

        job_id = submission_id or f"raysubmit_{uuid.uuid4().hex[:12]}"
        cw = self._cw()
        gcs_addr = f"{cw.gcs_addr[0]}:{cw.gcs_addr[1]}"
        Supervisor = ray.remote(JobSupervisor)
        sup = Supervisor.options(
            name=f"_job_supervisor:{job_id}", lifetime="detached",
            num_cpus=0, max_concurrency=4,
        ).remote(job_id, entrypoint, runtime_env, gcs_addr)
        ray.get(sup.run.remote(), timeout=60)
        return job_id

    def _get_status_view(self, job_id: str) -> Optional[dict]:
        cw = self._cw()
        r = cw.io.run(cw.gcs.call("kv_get", {
            "ns": "job", "key": f"status:{job_id}".encode()}))
        if not r.get("value"):
            return None
        return json.loads(r["value"].decode())

    def get_job_status(self, job_id: str) -> str:
        v = self._get_status_view(job_id)
        return v["status"] if v else "PENDING"

    def get_job_info(self, job_id: str) -> Optional[dict]:
        return self._get_status_view(job_id)

    def list_jobs(self) -> List[Dict[str, Any]]:
        cw = self._cw()
        keys = cw.io.run(cw.gcs.call("kv_keys", {"ns": "job",
                                                 "prefix": b"status:"}))["keys"]
        out = []
        for k in keys:
            r = cw.io.run(cw.gcs.call("kv_get", {"ns": "job", "key": k}))
            if r.get("value"):
                out.append(json.loads(r["value"].decode()))
        return out

    def get_job_logs(self, job_id: str) -> str:
        import ant_ray_amd as ray

        try:
            sup = ray.get_actor(f"_job_supervisor:{job_id}")
            return ray.get(sup.logs.remote(), timeout=30)
        except Exception:
            try:
                with open(f"/tmp/antray/job_{job_id}.log", "rb") as f:
                    return f.read().decode(errors="replace")
            except FileNotFoundError:
                return ""

    def stop_job(self, job_id: str) -> bool:
        import ant_ray_amd as ray

        try:
            sup = ray.get_actor(f"_job_supervisor:{job_id}")
            return ray.get(sup.stop.remote(), timeout=30)
        except Exception:
            return False


class JobStatus:
    PENDING = "PENDING"
    RUNNING = "RUNNING"
    SUCCEEDED = "SUCCEEDED"
    FAILED = "FAILED"
    STOPPED = "STOPPED"


class JobType:
    """How the job entered the cluster (parity: reference
    job_submission JobType)."""

    SUBMISSION = "SUBMISSION"  # via JobSubmissionClient / `ray job submit`
    DRIVER = "DRIVER"          # a directly-connected driver


class JobErrorType:
    """Failure classification (parity name)."""

    APPLICATION = "APPLICATION"   # user entrypoint exited non-zero
    SYSTEM = "SYSTEM"             # runtime failure (worker/node death)


from dataclasses import dataclass, field  # noqa: E402
from typing import Any, Dict, Optional  # noqa: E402


@dataclass
class DriverInfo:
    """Driver process of a job (parity: reference DriverInfo)."""

    id: str
    node_ip_address: str = "127.0.0.1"
    pid: Optional[int] = None


@dataclass
class JobInfo:
    """Submission-job metadata (parity: reference JobInfo)."""

    status: str
    entrypoint: str = ""
    message: Optional[str] = None
    error_type: Optional[str] = None
    start_time: Optional[int] = None
    end_time: Optional[int] = None
    metadata: Dict[str, str] = field(default_factory=dict)
    runtime_env: Dict[str, Any] = field(default_factory=dict)
    entrypoint_num_cpus: Optional[float] = None
    entrypoint_num_gpus: Optional[float] = None
    submission_id: Optional[str] = None


@dataclass
class JobDetails(JobInfo):
    """JobInfo + identity fields, the shape `ray job list`/the REST API
    returns (parity: reference JobDetails)."""

    job_id: Optional[str] = None
    type: str = JobType.SUBMISSION
    driver_info: Optional[DriverInfo] = None

    @classmethod
    def _from_info_dict(cls, d: Dict[str, Any]) -> "JobDetails":
        return cls(
            status=d.get("status", "PENDING"),
            entrypoint=d.get("entrypoint", ""),
            message=d.get("message"),
            start_time=d.get("start_time"),
            end_time=d.get("end_time"),
            metadata=d.get("metadata") or {},
            runtime_env=d.get("runtime_env") or {},
            submission_id=d.get("submission_id") or d.get("job_id"),
            job_id=d.get("job_id"),
        )
