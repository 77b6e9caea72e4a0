"""Master/worker HTTP endpoints: overview, browse, workers, mounts, jobs,
prometheus metrics.

Analog of the reference's curvine-web (axum WebServer + Vue dashboard,
/root/reference/curvine-web/src/: master overview/browse routers,
router/load_handler.rs) — a dependency-free asyncio HTTP/1.1 server with a
minimal single-page dashboard.
"""
from __future__ import annotations

import asyncio
import html
import json
import logging
import urllib.parse
from typing import Optional

log = logging.getLogger("curvine.web")

_PAGE = """<!doctype html><html><head><title>curvine-amd</title>
<style>body{font-family:monospace;margin:2em}table{border-collapse:collapse}
td,th{border:1px solid #999;padding:4px 8px;text-align:left}</style></head>
<body><h2>curvine-amd cluster</h2><div id=info></div>
<h3>workers</h3><table id=w><tr><th>id</th><th>addr</th><th>device</th>
<th>used</th><th>capacity</th><th>tiers</th></tr></table>
<script>
fetch('/api/info').then(r=>r.json()).then(d=>{
 document.getElementById('info').innerText =
  `inodes ${d.inode_num}  blocks ${d.block_num}  used ${(d.used/2**30).toFixed(2)} / ${(d.capacity/2**30).toFixed(2)} GiB`;
 const t=document.getElementById('w');
 for(const w of d.live_workers){const r=t.insertRow();
  const a=w.address;
  r.insertCell().innerText=a.worker_id;
  r.insertCell().innerText=`${a.hostname}:${a.rpc_port}`;
  r.insertCell().innerText=a.device_id;
  const used=w.storages.reduce((s,x)=>s+x.used,0);
  const cap=w.storages.reduce((s,x)=>s+x.capacity,0);
  r.insertCell().innerText=(used/2**30).toFixed(2)+' GiB';
  r.insertCell().innerText=(cap/2**30).toFixed(2)+' GiB';
  r.insertCell().innerText=w.storages.map(s=>`${s.tier}:${(s.capacity/2**30).toFixed(0)}G`).join(' ');
 }});
</script></body></html>"""


class WebServer:
    def __init__(self, conf, master=None, worker=None, fuse_session=None,
                 port: int | None = None, host: str | None = None):
        self.conf = conf
        self.master = master
        self.worker = worker
        self.fuse_session = fuse_session
        self.host = host if host is not None else conf.master.hostname
        self.port = port if port is not None else (
            conf.master.web_port if master else conf.worker.web_port)
        self._server: Optional[asyncio.AbstractServer] = None

    async def start(self) -> "WebServer":
        self._server = await asyncio.start_server(
            self._on_conn, self.host, self.port,
            reuse_address=True)
        if self.port == 0:
            self.port = self._server.sockets[0].getsockname()[1]
        log.info("web server on :%d", self.port)
        return self

    async def stop(self) -> None:
        if self._server:
            self._server.close()
            await self._server.wait_closed()

    async def _on_conn(self, reader, writer):
        try:
            line = await asyncio.wait_for(reader.readline(), 10)
            parts = line.decode("latin1").split()
            if len(parts) < 2:
                return
            path = parts[1]
            while True:   # drain headers
                h = await reader.readline()
                if h in (b"\r\n", b"\n", b""):
                    break
            status, ctype, body = self._route(path)
            writer.write(
                f"HTTP/1.1 {status}\r\nContent-Type: {ctype}\r\n"
                f"Content-Length: {len(body)}\r\nConnection: close\r\n\r\n"
                .encode() + body)
            await writer.drain()
        except (asyncio.TimeoutError, ConnectionError):
            pass
        finally:
            try:
                writer.close()
                await writer.wait_closed()
            except Exception:  # noqa: BLE001
                pass

    def _route(self, raw_path: str) -> tuple[str, str, bytes]:
        parsed = urllib.parse.urlsplit(raw_path)
        path = parsed.path
        q = dict(urllib.parse.parse_qsl(parsed.query))
        try:
            if path == "/" or path == "/index.html":
                return "200 OK", "text/html", _PAGE.encode()
            if path == "/api/info" and self.master:
                return self._json(self.master.fs.master_info())
            if path == "/api/browse" and self.master:
                p = q.get("path", "/")
                sts = self.master.fs.list_status(p)
                return self._json([s.to_dict() for s in sts])
            if path == "/api/mounts" and self.master:
                return self._json([m.to_dict()
                                   for m in self.master.mounts.table()])
            if path == "/api/jobs" and self.master:
                return self._json({jid: {k: j[k] for k in
                                         ("state", "total", "done", "failed")}
                                   for jid, j in self.master.jobs.jobs.items()})
            if path == "/api/raft" and self.master and self.master.raft:
                r = self.master.raft
                return self._json({
                    "id": r.id, "state": r.state, "term": r.term,
                    "leader": r.leader_id, "commit": r.commit_index,
                    "last_index": r.log.last_index})
            if path == "/api/storage" and self.worker:
                return self._json([s.__dict__
                                   for s in self.worker.store.storages()])
            if path == "/api/fuse" and self.fuse_session:
                return self._json(self.fuse_session.stats())
            if path == "/metrics":
                return "200 OK", "text/plain", self._prometheus()
            return "404 Not Found", "text/plain", b"not found"
        except Exception as e:  # noqa: BLE001
            return "500 Internal Server Error", "text/plain", str(e).encode()

    @staticmethod
    def _json(obj) -> tuple[str, str, bytes]:
        return ("200 OK", "application/json",
                json.dumps(obj, default=str).encode())

    def _prometheus(self) -> bytes:
        lines = []
        if self.master:
            info = self.master.fs.master_info()
            for k in ("inode_num", "block_num", "capacity", "used"):
                lines.append(f"curvine_master_{k} {info[k]}")
            lines.append(
                f"curvine_master_live_workers {len(info['live_workers'])}")
        if self.worker:
            for s in self.worker.store.storages():
                lbl = f'{{tier="{s.tier}",dir="{s.dir_id}"}}'
                lines.append(f"curvine_worker_capacity_bytes{lbl} {s.capacity}")
                lines.append(f"curvine_worker_used_bytes{lbl} {s.used}")
                lines.append(f"curvine_worker_blocks{lbl} {s.block_num}")
        if self.fuse_session:
            for op, d in self.fuse_session.stats().items():
                if not isinstance(d, dict) or "count" not in d:
                    continue  # e.g. "_native_loop" raw counters
                lines.append(f'curvine_fuse_ops_total{{op="{op}"}} {d["count"]}')
                lines.append(
                    f'curvine_fuse_op_seconds_total{{op="{op}"}} {d["time_s"]}')
        return ("\n".join(lines) + "\n").encode()
