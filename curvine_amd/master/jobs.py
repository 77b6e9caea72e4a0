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
    def __init__(self, master):
        self.master = master
        self.store = JobStore(master.conf.job.store, master.conf.job.store_path)
        self.jobs: dict[str, dict] = self.store.load_all()
        self.next_id = max((int(j.rsplit("-", 1)[-1]) for j in self.jobs), default=0)

    def submit(self, h: dict) -> dict:
        """h: {path, recursive, replicas}.  The path must be under a mount
        (or an existing cache dir for re-replication loads)."""
        self.next_id += 1
        job_id = f"job-{self.next_id}"
        path = h["path"]
        mount = self.master.mounts.lookup(path)
        job = {
            "job_id": job_id, "path": path, "state": "planning",
            "submitted_ms": int(time.time() * 1000),
            "tasks": {}, "done": 0, "failed": 0, "total": 0,
            "mount": mount.to_dict() if mount else None,
            "replicas": h.get("replicas", 1),
            "recursive": h.get("recursive", True),
        }
        self.jobs[job_id] = job
        self._plan(job)
        self.store.save(job)
        return {"job_id": job_id, "state": job["state"], "total": job["total"]}

    def _plan(self, job: dict) -> None:
        """List the UFS and emit one load task per file to a worker."""
        mount = job["mount"]
        if mount is None:
            job["state"] = "failed"
            job["error"] = f"path {job['path']} is not under a mount"
            return
        from curvine_amd.ufs import get_ufs
        try:
            ufs = get_ufs(mount["ufs_path"], mount.get("properties", {}))
            rel = job["path"][len(mount["curvine_path"]):] or "/"
            files = ufs.list_files(rel, recursive=job["recursive"])
        except Exception as e:  # noqa: BLE001
            job["state"] = "failed"
            job["error"] = str(e)
            return
        workers = self.master.fs.workers.live_workers()
        if not workers:
            job["state"] = "failed"
            job["error"] = "no live workers"
            return
        job["total"] = len(files)
        job["state"] = "running" if files else "completed"
        for i, f in enumerate(files):
            w = workers[i % len(workers)]
            task_id = f"{job['job_id']}-t{i}"
            cv_path = mount["curvine_path"].rstrip("/") + f["path"]
            task = {"task_id": task_id, "ufs_path": mount["ufs_path"],
                    "ufs_rel": f["path"], "cv_path": cv_path,
                    "length": f["length"], "state": "assigned",
                    "worker": w.address.worker_id,
                    "properties": mount.get("properties", {}),
                    "replicas": job["replicas"]}
            job["tasks"][task_id] = task
            self.master.fs.workers.add_command(w.address.worker_id, {
                "cmd": "load_task", **task})

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
