"""Typed master RPC surface.

Analog of /root/reference/crates/client/curvine-client-core/src/file/
fs_client.rs: one method per master RpcCode, marshalling model types.
"""
from __future__ import annotations

import socket
from typing import Optional

from curvine_amd.conf import ClusterConf
from curvine_amd.model import FileBlocks, FileStatus, LocatedBlock, MountInfo
from curvine_amd.rpc.client import ClusterConnector
from curvine_amd.rpc.codes import RpcCode


class FsClient:
    def __init__(self, conf: ClusterConf):
        self.conf = conf
        self.connector = ClusterConnector(conf.client.master_addrs,
                                          conf.client.rpc_timeout_ms,
                                          conf.client.conn_retry)
        self.client_host = socket.gethostname()
        # in-process worker colocated with this client (embedded mode)
        self.local_worker_id = -1

    async def report_metrics(self, metrics: dict,
                             kind: str = "client") -> None:
        """Push a metrics snapshot to the master (MetricsReport code 60,
        the reference's client metrics channel)."""
        import os as _os
        await self._rpc(RpcCode.MetricsReport, {
            "client_id": f"{self.client_host}:{_os.getpid()}",
            "kind": kind, "metrics": metrics})

    async def close(self) -> None:
        await self.connector.close()

    async def _rpc(self, code: RpcCode, header: dict | None = None) -> dict:
        if not self.conf.client.audit_log:
            reply = await self.connector.rpc(code, header)
            return reply.header
        # client audit stream (unified_filesystem.rs:144-169 analog)
        import logging
        import time
        t0 = time.perf_counter()
        ok = True
        try:
            reply = await self.connector.rpc(code, header)
            return reply.header
        except Exception:
            ok = False
            raise
        finally:
            logging.getLogger("audit.client").info(
                "cmd=%s path=%s ok=%s used_us=%d", code.name,
                (header or {}).get("path", ""), ok,
                int((time.perf_counter() - t0) * 1e6))

    # ---------------- namespace ----------------
    async def mkdir(self, path: str, mode: int = 0o755,
                    create_parents: bool = True) -> FileStatus:
        h = await self._rpc(RpcCode.Mkdir, {"path": path, "mode": mode,
                                            "create_parents": create_parents})
        return FileStatus.from_dict(h["status"])

    async def create(self, path: str, overwrite: bool = False,
                     replicas: int = 0, block_size: int = 0,
                     storage_tier: str = "", mode: int = 0o644) -> FileStatus:
        h = await self._rpc(RpcCode.CreateFile, {
            "path": path, "overwrite": overwrite,
            "replicas": replicas or self.conf.client.replicas,
            "block_size": block_size or self.conf.client.block_size,
            "storage_tier": storage_tier or self.conf.client.storage_tier,
            "mode": mode})
        return FileStatus.from_dict(h["status"])

    async def append(self, path: str) -> FileBlocks:
        h = await self._rpc(RpcCode.AppendFile, {"path": path})
        return FileBlocks.from_dict(h["file_blocks"])

    async def open(self, path: str) -> FileBlocks:
        h = await self._rpc(RpcCode.OpenFile, {"path": path})
        return FileBlocks.from_dict(h["file_blocks"])

    async def add_block(self, path: str, commit_prev_len: int = -1,
                        exclude_workers: list[int] | None = None) -> LocatedBlock:
        h = await self._rpc(RpcCode.AddBlock, {
            "path": path, "commit_prev_len": commit_prev_len,
            "client_host": self.client_host,
            "client_worker_id": self.local_worker_id,
            "exclude_workers": exclude_workers or []})
        return LocatedBlock.from_dict(h["block"])

    async def complete_file(self, path: str, length: int,
                            block_lens: list[int],
                            commits: list[dict] | None = None) -> FileStatus:
        h = await self._rpc(RpcCode.CompleteFile, {
            "path": path, "length": length, "block_lens": block_lens,
            "commits": commits or []})
        return FileStatus.from_dict(h["status"])

    async def delete(self, path: str, recursive: bool = False) -> int:
        h = await self._rpc(RpcCode.Delete, {"path": path, "recursive": recursive})
        return h.get("deleted_blocks", 0)

    async def rename(self, src: str, dst: str) -> None:
        await self._rpc(RpcCode.Rename, {"src": src, "dst": dst})

    async def file_status(self, path: str) -> FileStatus:
        h = await self._rpc(RpcCode.FileStatus, {"path": path})
        return FileStatus.from_dict(h["status"])

    async def exists(self, path: str) -> bool:
        h = await self._rpc(RpcCode.Exists, {"path": path})
        return h["exists"]

    async def list_status(self, path: str) -> list[FileStatus]:
        h = await self._rpc(RpcCode.ListStatus, {"path": path})
        return [FileStatus.from_dict(s) for s in h["statuses"]]

    async def set_attr(self, path: str, **attrs) -> FileStatus:
        h = await self._rpc(RpcCode.SetAttr, {"path": path, **attrs})
        return FileStatus.from_dict(h["status"])

    async def symlink(self, path: str, target: str) -> FileStatus:
        h = await self._rpc(RpcCode.Symlink, {"path": path, "target": target})
        return FileStatus.from_dict(h["status"])

    async def link(self, src: str, dst: str) -> FileStatus:
        h = await self._rpc(RpcCode.Link, {"src": src, "dst": dst})
        return FileStatus.from_dict(h["status"])

    async def resize(self, path: str, length: int) -> FileStatus:
        h = await self._rpc(RpcCode.ResizeFile, {"path": path, "length": length})
        return FileStatus.from_dict(h["status"])

    async def get_block_locations(self, path: str) -> FileBlocks:
        h = await self._rpc(RpcCode.GetBlockLocations, {"path": path})
        return FileBlocks.from_dict(h["file_blocks"])

    async def free(self, path: str, recursive: bool = False) -> int:
        h = await self._rpc(RpcCode.Free, {"path": path, "recursive": recursive})
        return h.get("freed_blocks", 0)

    async def get_master_info(self) -> dict:
        # the client handshake: report our version (component_info=1000
        # analog), warn (deduped) about incompatible workers the master
        # advertises (worker_precheck.rs behavior — always diagnose on
        # the client side, never rejects)
        from curvine_amd.compat import (CompatibilityPolicy, PeerWarnDedup,
                                        component_info)
        info = await self._rpc(RpcCode.GetFilesystemInfo,
                               {"component_info": component_info("client")})
        if not hasattr(self, "_worker_precheck"):
            self._worker_precheck = PeerWarnDedup("worker")
            self._precheck_policy = CompatibilityPolicy()
        pol = self._precheck_policy
        for w in info.get("live_workers", []):
            ci = w.get("component_info")
            if pol.should_evaluate(ci is not None) and ci is not None:
                self._worker_precheck.warn(
                    w.get("address", {}).get("worker_id"),
                    pol.check_worker(ci))
        return info

    # ---------------- mounts / jobs ----------------
    async def mount(self, curvine_path: str, ufs_path: str,
                    properties: dict | None = None, cache_mode: str = "cache",
                    auto_cache: bool = True) -> MountInfo:
        h = await self._rpc(RpcCode.Mount, {
            "curvine_path": curvine_path, "ufs_path": ufs_path,
            "properties": properties or {}, "cache_mode": cache_mode,
            "auto_cache": auto_cache})
        return MountInfo.from_dict(h["mount"])

    async def unmount(self, curvine_path: str) -> None:
        await self._rpc(RpcCode.UnMount, {"curvine_path": curvine_path})

    async def get_mount_table(self) -> list[MountInfo]:
        h = await self._rpc(RpcCode.GetMountTable, {})
        return [MountInfo.from_dict(m) for m in h["mounts"]]

    async def get_mount_info(self, path: str) -> Optional[MountInfo]:
        h = await self._rpc(RpcCode.GetMountInfo, {"path": path})
        return MountInfo.from_dict(h["mount"]) if h.get("mount") else None

    async def submit_job(self, path: str, recursive: bool = True,
                         replicas: int = 1) -> dict:
        return await self._rpc(RpcCode.SubmitJob, {
            "path": path, "recursive": recursive, "replicas": replicas})

    async def job_status(self, job_id: str) -> dict:
        return await self._rpc(RpcCode.GetJobStatus, {"job_id": job_id})

    async def cancel_job(self, job_id: str) -> dict:
        return await self._rpc(RpcCode.CancelJob, {"job_id": job_id})

    async def report_task(self, report: dict) -> None:
        await self._rpc(RpcCode.ReportTask, report)
