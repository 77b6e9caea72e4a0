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
<meta charset=utf-8>
<style>
body{font-family:-apple-system,'Segoe UI',Roboto,monospace;margin:0;background:#0d1117;color:#c9d1d9}
header{background:#161b22;padding:12px 24px;display:flex;align-items:center;gap:16px;border-bottom:1px solid #30363d}
header h1{font-size:16px;margin:0;color:#e6edf3}
header .pill{background:#1f6feb22;color:#58a6ff;border:1px solid #1f6feb66;border-radius:12px;padding:2px 10px;font-size:12px}
main{padding:16px 24px;display:grid;grid-template-columns:1fr 1fr;gap:16px;max-width:1200px}
section{background:#161b22;border:1px solid #30363d;border-radius:8px;padding:12px 16px}
section h2{font-size:13px;margin:0 0 8px;color:#8b949e;text-transform:uppercase;letter-spacing:.08em}
table{border-collapse:collapse;width:100%;font-size:13px}
td,th{border-bottom:1px solid #21262d;padding:4px 8px;text-align:left}
th{color:#8b949e;font-weight:600}
.bar{background:#21262d;border-radius:4px;height:8px;overflow:hidden}
.bar i{display:block;height:100%;background:#238636}
.crumb a{color:#58a6ff;text-decoration:none;cursor:pointer}
#browse td a{color:#58a6ff;text-decoration:none;cursor:pointer}
.num{text-align:right;font-variant-numeric:tabular-nums}
.wide{grid-column:1/3}
</style></head><body>
<header><h1>curvine-amd</h1><span class=pill id=role>master</span>
<span id=summary style="font-size:13px;color:#8b949e"></span></header>
<main>
<section class=wide><h2>cluster</h2><div id=capbar class=bar><i style="width:0"></i></div>
<div id=info style="margin-top:6px;font-size:13px"></div></section>
<section class=wide><h2>workers</h2><table id=w><thead><tr><th>id</th><th>addr</th><th>gpu</th><th class=num>used</th><th class=num>capacity</th><th>tiers</th></tr></thead><tbody></tbody></table></section>
<section class=wide><h2>namespace <span class=crumb id=crumb></span></h2>
<table id=browse><thead><tr><th>name</th><th>type</th><th class=num>size</th><th>tier</th><th class=num>mtime</th></tr></thead><tbody></tbody></table></section>
<section><h2>mounts</h2><table id=mounts><tbody></tbody></table></section>
<section><h2>jobs</h2><table id=jobs><tbody></tbody></table></section>
<section><h2>raft</h2><div id=raft style="font-size:13px">standalone</div></section>
<section><h2>fuse</h2><div id=fuse style="font-size:13px">-</div></section>
</main>
<script>
const gib=b=>(b/2**30).toFixed(2)+' GiB';
const J=u=>fetch(u).then(r=>r.ok?r.json():null).catch(()=>null);
let cwd='/';
async function refresh(){
 const d=await J('/api/info');
 if(d){
  const pct=d.capacity? (100*d.used/d.capacity):0;
  document.querySelector('#capbar i').style.width=pct.toFixed(1)+'%';
  document.getElementById('info').innerText=
   `${d.inode_num} inodes, ${d.block_num} blocks, ${gib(d.used)} of ${gib(d.capacity)} used (${pct.toFixed(1)}%)`;
  document.getElementById('summary').innerText=
   `${(d.live_workers||[]).length} workers live`;
  const tb=document.querySelector('#w tbody');tb.innerHTML='';
  for(const w of d.live_workers||[]){const r=tb.insertRow();const a=w.address;
   const used=w.storages.reduce((s,x)=>s+x.used,0),cap=w.storages.reduce((s,x)=>s+x.capacity,0);
   r.innerHTML=`<td>${a.worker_id}</td><td>${a.hostname}:${a.rpc_port}</td><td>${a.device_id}</td>`+
    `<td class=num>${gib(used)}</td><td class=num>${gib(cap)}</td>`+
    `<td>${w.storages.map(s=>s.tier+':'+(s.capacity/2**30).toFixed(0)+'G').join(' ')}</td>`;}
 }
 const m=await J('/api/mounts');
 if(m){const tb=document.querySelector('#mounts tbody');tb.innerHTML='';
  for(const x of m){const r=tb.insertRow();
   r.innerHTML=`<td>${x.cv_path||x.path||''}</td><td>${x.ufs_path||''}</td><td>${x.cache_mode||''}</td>`;}
  if(!m.length)tb.innerHTML='<tr><td style=color:#8b949e>none</td></tr>';}
 const j=await J('/api/jobs');
 if(j){const tb=document.querySelector('#jobs tbody');tb.innerHTML='';
  const ids=Object.keys(j);
  for(const id of ids.slice(-10)){const x=j[id];const r=tb.insertRow();
   r.innerHTML=`<td>${id}</td><td>${x.state}</td><td class=num>${x.done}/${x.total}</td>`;}
  if(!ids.length)tb.innerHTML='<tr><td style=color:#8b949e>none</td></tr>';}
 const rf=await J('/api/raft');
 if(rf)document.getElementById('raft').innerText=
  `node ${rf.id} ${rf.state}, term ${rf.term}, leader ${rf.leader}, commit ${rf.commit}/${rf.last_index}`;
 const fu=await J('/api/fuse');
 if(fu)document.getElementById('fuse').innerText=JSON.stringify(fu).slice(0,400);
 browse(cwd);
}
async function browse(p){
 cwd=p;
 const parts=p.split('/').filter(x=>x);let acc='';
 let html='<a onclick="browse(String.fromCharCode(47))">/</a> ';
 for(const x of parts){acc+='/'+x;html+=`<a onclick="browse('${acc}')">${x}</a> / `;}
 document.getElementById('crumb').innerHTML=html;
 const sts=await J('/api/browse?path='+encodeURIComponent(p));
 const tb=document.querySelector('#browse tbody');tb.innerHTML='';
 if(!sts)return;
 for(const s of sts){const r=tb.insertRow();
  const dir=s.file_type===1;
  const name=dir?`<a onclick="browse('${s.path}')">${s.name}/</a>`:s.name;
  r.innerHTML=`<td>${name}</td><td>${dir?'dir':(s.file_type===2?'link':'file')}</td>`+
   `<td class=num>${dir?'':gib(s.length)}</td><td>${s.storage_tier||''}</td>`+
   `<td class=num>${new Date(s.mtime_ms).toISOString().slice(0,19)}</td>`;}
}
refresh();setInterval(refresh,3000);
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
            if path == "/api/client-metrics" and self.master:
                return self._json(getattr(self.master, "client_metrics", {}))
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
        if self.master and getattr(self.master, "native_meta", None):
            for k, v in self.master.native_meta.stats().items():
                if isinstance(v, (int, float)):
                    lines.append(f"curvine_meta_frontend_{k} {v}")
        if self.worker:
            for s in self.worker.store.storages():
                lbl = f'{{tier="{s.tier}",dir="{s.dir_id}"}}'
                lines.append(f"curvine_worker_capacity_bytes{lbl} {s.capacity}")
                lines.append(f"curvine_worker_used_bytes{lbl} {s.used}")
                lines.append(f"curvine_worker_blocks{lbl} {s.block_num}")
            stats_fn = getattr(self.worker.rpc, "stats", None)
            if stats_fn:
                try:
                    for k, v in stats_fn().items():
                        if isinstance(v, (int, float)):
                            lines.append(f"curvine_data_plane_{k} {v}")
                except Exception:  # noqa: BLE001
                    pass
        if self.fuse_session:
            for op, d in self.fuse_session.stats().items():
                if not isinstance(d, dict) or "count" not in d:
                    continue  # e.g. "_native_loop" raw counters
                lines.append(f'curvine_fuse_ops_total{{op="{op}"}} {d["count"]}')
                lines.append(
                    f'curvine_fuse_op_seconds_total{{op="{op}"}} {d["time_s"]}')
        return ("\n".join(lines) + "\n").encode()
