"""FUSE daemon composition: asyncio client loop + channels + (optionally)
an embedded worker owning the local GPU's HBM arena.

The MI355X deployment model: one FUSE daemon per node (or per GPU),
embedding the worker so READs on HBM-cached blocks are in-process arena
reads.  `mount()` is the programmatic entry; `cv-fuse` CLI wraps it.
"""
from __future__ import annotations

import asyncio
import os
import logging
import threading
from typing import Optional

from curvine_amd.client.filesystem import CurvineFileSystem
from curvine_amd.conf import ClusterConf
from curvine_amd.fuse.ops import CurvineFuseFs
from curvine_amd.fuse.session import FuseSession

log = logging.getLogger("curvine.fuse.daemon")


class FuseDaemon:
    def __init__(self, conf: ClusterConf, mnt_path: str | None = None,
                 embed_worker: bool = False, device_id: int = -1):
        self.conf = conf
        self.mnt_path = mnt_path or conf.fuse.mnt_path
        self.embed_worker = embed_worker
        self.device_id = device_id
        self.loop = asyncio.new_event_loop()
        self._loop_thread = threading.Thread(
            target=self._run_loop, daemon=True, name="curvine-fuse-loop")
        self.worker = None
        self.fs: Optional[CurvineFileSystem] = None
        self.fuse_fs: Optional[CurvineFuseFs] = None
        self.session: Optional[FuseSession] = None

    def _run_loop(self):
        asyncio.set_event_loop(self.loop)
        self.loop.run_forever()

    def call(self, coro, timeout: float = 120.0):
        return asyncio.run_coroutine_threadsafe(coro, self.loop).result(timeout)

    def control_socket_path(self) -> str:
        import hashlib
        h = hashlib.sha1(self.mnt_path.encode()).hexdigest()[:10]
        return f"/tmp/cv-fuse-ctl-{h}.sock"

    def start(self, takeover: bool = False) -> "FuseDaemon":
        self._loop_thread.start()
        if self.embed_worker:
            from curvine_amd.worker.server import Worker

            async def mkworker():
                return await Worker(self.conf, device_id=self.device_id).start()
            self.worker = self.call(mkworker())

        async def mkfs():
            return CurvineFileSystem(self.conf)
        self.fs = self.call(mkfs())
        if self.worker is not None:
            self.fs.client.local_worker_id = self.worker.worker_id
        self.fuse_fs = CurvineFuseFs(self.fs, self.conf, self.loop)
        session_fd = None
        if takeover:
            session_fd, state = self._request_takeover()
            self.fuse_fs.restore_state(state)
        self.session = FuseSession(
            self.fuse_fs, self.mnt_path,
            channels=self.conf.fuse.mnt_number,
            max_write=self.conf.fuse.max_write,
            native_loop=self.conf.fuse.native_loop).start(
                session_fd=session_fd)
        self._start_control_server()
        if self.conf.fuse.metrics_report_s > 0:
            self._metrics_task = asyncio.run_coroutine_threadsafe(
                self._metrics_report_loop(), self.loop)
        return self

    async def _metrics_report_loop(self) -> None:
        """Push the per-op FUSE stats to the master every interval
        (MetricsReport code 60 — cluster-wide FUSE visibility)."""
        interval = self.conf.fuse.metrics_report_s
        while True:
            await asyncio.sleep(interval)
            try:
                if self.session is not None:
                    await self.fs.client.report_metrics(
                        {"mnt": self.mnt_path,
                         "fuse_op_stats": self.session.stats()},
                        kind="fuse")
            except asyncio.CancelledError:
                return
            except Exception:  # noqa: BLE001 — master briefly away
                pass

    # ---------------- hot upgrade (fd + state handover) ----------------
    def _request_takeover(self) -> tuple[int, dict]:
        """Connect to the running daemon's control socket; receive the
        /dev/fuse session fd (SCM_RIGHTS) + serialized handle state."""
        import json
        import socket as sock
        import struct
        s = sock.socket(sock.AF_UNIX, sock.SOCK_STREAM)
        s.connect(self.control_socket_path())
        s.sendall(b"TAKEOVER")
        msg, fds, _flags, _addr = sock.recv_fds(s, 8, 1)
        (length,) = struct.unpack(">I", msg[:4]) if len(msg) >= 4 else (0,)
        buf = msg[4:]
        while len(buf) < length:
            chunk = s.recv(65536)
            if not chunk:
                break
            buf += chunk
        s.close()
        if not fds:
            raise RuntimeError("takeover: no fd received")
        state = json.loads(buf.decode()) if buf else {}
        log.info("takeover: received session fd %d + %d handles",
                 fds[0], len(state.get("handles", [])))
        return fds[0], state

    def _start_control_server(self) -> None:
        import json
        import socket as sock
        import struct
        import threading
        path = self.control_socket_path()
        try:
            os.unlink(path)
        except OSError:
            pass
        srv = sock.socket(sock.AF_UNIX, sock.SOCK_STREAM)
        srv.bind(path)
        srv.listen(1)
        self._ctl_sock = srv

        def serve():
            while True:
                try:
                    conn, _ = srv.accept()
                except OSError:
                    return
                try:
                    req = conn.recv(64)
                    if req.startswith(b"TAKEOVER"):
                        state = self.fuse_fs.dump_state()
                        payload = json.dumps(state).encode()
                        sock.send_fds(conn,
                                      [struct.pack(">I", len(payload)) + payload],
                                      [self.session.session_fd])
                        log.info("handed session fd to successor; draining")
                        self.handed_over = True
                    elif req.startswith(b"STATS"):
                        conn.sendall(json.dumps(self.session.stats()).encode())
                except Exception as e:  # noqa: BLE001
                    log.warning("control socket: %s", e)
                finally:
                    conn.close()

        threading.Thread(target=serve, daemon=True,
                         name="cv-fuse-ctl").start()

    handed_over = False

    def stop(self) -> None:
        if getattr(self, "_ctl_sock", None):
            try:
                self._ctl_sock.close()
                os.unlink(self.control_socket_path())
            except OSError:
                pass
        if self.session:
            # after a handover the successor owns the mount: never umount
            self.session.stop(umount_fs=not self.handed_over)
        if self.fs:
            try:
                self.call(self.fs.close(), timeout=10)
            except Exception:  # noqa: BLE001
                pass
        if self.worker:
            try:
                self.call(self.worker.stop(), timeout=10)
            except Exception:  # noqa: BLE001
                pass
        self.loop.call_soon_threadsafe(self.loop.stop)
        self._loop_thread.join(timeout=5)
