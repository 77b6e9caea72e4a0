"""Load-job orchestration (cv load).

Analog of /root/reference/curvine-master/src/master/job/job_manager.rs
(:36-286) + the data-transfer planner/scheduler
(/root/reference/crates/server/curvine-data-transfer/src/transfer/):
a submitted load job lists the UFS path, plans one task per file, assigns
tasks to workers (delivered as heartbeat commands), and tracks progress
from worker task reports.
"""
from __future__ import annotations

import logging
import time

from curvine_amd import errors as err

log = logging.getLogger("curvine.jobs")


class JobStore:
    """Job persistence (transfer/store.rs analog): memory or sqlite."""

    def __init__(self, kind: str = "memory", path: str = ""):
        self.kind = kind
        self._db = None
        if kind == "sqlite":
            import sqlite3
            import os
            os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
            self._db = sqlite3.connect(path)
            self._db.execute(
                "CREATE TABLE IF NOT EXISTS jobs (job_id TEXT PRIMARY KEY,"
                " payload TEXT)")
            self._db.commit()

    def save(self, job: dict) -> None:
        if self._db is not None:
            import json
            self._db.execute(
                "INSERT OR REPLACE INTO jobs (job_id, payload) VALUES (?, ?)",
                (job["job_id"], json.dumps(job, default=str)))
            self._db.commit()

    def load_all(self) -> dict:
        if self._db is None:
            return {}
        import json
        return {jid: json.loads(p) for jid, p in
                self._db.execute("SELECT job_id, payload FROM jobs")}


class JobManager:
    """Runs embedded in the master (`master` set) OR inside the
    standalone transfer service (`master=None`; the service supplies
    mounts/workers over RPC and calls new_job/plan directly)."""

    def __init__(self, master, conf=None):
        self.master = master
        conf = conf if conf is not None else master.conf
        self.store = JobStore(conf.job.store, conf.job.store_path)
        self.jobs: dict[str, dict] = self.store.load_all()
        self.next_id = max((int(j.rsplit("-", 1)[-1]) for j in self.jobs), default=0)

    def submit(self, h: dict) -> dict:
        """Embedded path. h: {path, recursive, replicas}; the path must
        be under a mount (or an existing cache dir for re-replication
        loads)."""
        mount = self.master.mounts.lookup(h["path"])
        job = self.new_job(h, mount)
        worker_ids = [w.address.worker_id
                      for w in self.master.fs.workers.live_workers()]
        self.plan(job, mount, worker_ids)
        for task in job["tasks"].values():
            self.master.fs.workers.add_command(task["worker"], {
                "cmd": "load_task", **task})
        self.store.save(job)
        return {"job_id": job["job_id"], "state": job["state"],
                "total": job["total"]}

    def new_job(self, h: dict, mount) -> dict:
        self.next_id += 1
        job_id = f"job-{self.next_id}"
        job = {
            "job_id": job_id, "path": h["path"], "state": "planning",
            "submitted_ms": int(time.time() * 1000),
            "tasks": {}, "done": 0, "failed": 0, "total": 0,
            "mount": mount.to_dict() if mount else None,
            "replicas": h.get("replicas", 1),
            "recursive": h.get("recursive", True),
        }
        self.jobs[job_id] = job
        return job

    def plan(self, job: dict, mount, worker_ids: list[int]) -> None:
        """List the UFS and emit one load task per file round-robin over
        the workers (transfer/planner.rs + scheduler.rs analog).  The
        caller delivers each task to its worker."""
        if mount is None:
            job["state"] = "failed"
            job["error"] = f"path {job['path']} is not under a mount"
            return
        m = mount.to_dict() if hasattr(mount, "to_dict") else mount
        from curvine_amd.ufs import get_ufs
        try:
            ufs = get_ufs(m["ufs_path"], m.get("properties", {}))
            rel = job["path"][len(m["curvine_path"]):] or "/"
            files = ufs.list_files(rel, recursive=job["recursive"])
        except Exception as e:  # noqa: BLE001
            job["state"] = "failed"
            job["error"] = str(e)
            return
        if not worker_ids:
            job["state"] = "failed"
            job["error"] = "no live workers"
            return
        job["total"] = len(files)
        job["state"] = "running" if files else "completed"
        for i, f in enumerate(files):
            wid = worker_ids[i % len(worker_ids)]
            task_id = f"{job['job_id']}-t{i}"
            cv_path = m["curvine_path"].rstrip("/") + f["path"]
            task = {"task_id": task_id, "ufs_path": m["ufs_path"],
                    "ufs_rel": f["path"], "cv_path": cv_path,
                    "length": f["length"], "state": "assigned",
                    "worker": wid,
                    "properties": m.get("properties", {}),
                    "replicas": job["replicas"]}
            job["tasks"][task_id] = task

    def retry(self, job_id: str) -> dict:
        """Re-dispatch FAILED tasks of an existing job in place
        (RetryTransfer semantics — not a fresh submission)."""
        job = self.jobs.get(job_id)
        if job is None:
            raise err.JobNotFound(job_id)
        retried = 0
        for task in job["tasks"].values():
            if task["state"] == "failed":
                task["state"] = "assigned"
                job["failed"] -= 1
                if self.master is not None:
                    self.master.fs.workers.add_command(
                        task["worker"], {"cmd": "load_task", **task})
                retried += 1
        if retried and job["state"] in ("completed_with_failures", "failed"):
            job["state"] = "running"
        self.store.save(job)
        return {"job_id": job_id, "retried": retried,
                "state": job["state"]}

    def status(self, job_id: str) -> dict:
        job = self.jobs.get(job_id)
        if job is None:
            raise err.JobNotFound(job_id)
        return {k: job[k] for k in ("job_id", "state", "total", "done",
                                    "failed", "path")} | {
            "error": job.get("error", "")}

    def cancel(self, job_id: str) -> dict:
        job = self.jobs.get(job_id)
        if job is None:
            raise err.JobNotFound(job_id)
        if job["state"] in ("planning", "running"):
            job["state"] = "cancelled"
            self.store.save(job)
        return {"job_id": job_id, "state": job["state"]}

    def report_task(self, h: dict) -> None:
        job = self.jobs.get(h.get("job_id", h.get("task_id", "").rsplit("-t", 1)[0]))
        if job is None:
            return
        task = job["tasks"].get(h["task_id"])
        if task is None or task["state"] in ("done", "failed"):
            return
        if h.get("success"):
            task["state"] = "done"
            job["done"] += 1
        else:
            task["state"] = "failed"
            task["error"] = h.get("error", "")
            job["failed"] += 1
        if job["done"] + job["failed"] >= job["total"] and job["state"] == "running":
            job["state"] = "completed" if job["failed"] == 0 else "completed_with_failures"
            self.store.save(job)
