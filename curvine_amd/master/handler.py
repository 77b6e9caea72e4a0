"""Master RPC dispatch.

Analog of the reference's `MasterHandler`
(/root/reference/curvine-master/src/master/master_handler.rs:1030-1151):
per-connection dispatch over the RpcCode surface.  Header field names follow
proto/master.proto & worker.proto.
"""
from __future__ import annotations

import asyncio
import logging
import time
from typing import Optional

from curvine_amd import errors as err
from curvine_amd.master.filesystem import MasterFilesystem
from curvine_amd.model import WorkerInfo
from curvine_amd.rpc.codes import RpcCode
from curvine_amd.rpc.message import Message

log = logging.getLogger("curvine.master.handler")
audit = logging.getLogger("audit")


_MUTATIONS = {
    RpcCode.Mkdir, RpcCode.Delete, RpcCode.CreateFile, RpcCode.AppendFile,
    RpcCode.Rename, RpcCode.AddBlock, RpcCode.CompleteFile, RpcCode.SetAttr,
    RpcCode.Symlink, RpcCode.Link, RpcCode.ResizeFile, RpcCode.Free,
    RpcCode.CreateFilesBatch, RpcCode.AddBlocksBatch,
    RpcCode.CompleteFilesBatch, RpcCode.Mount, RpcCode.UnMount,
    RpcCode.UpdateMount, RpcCode.SubmitJob,
}


class MasterHandler:
    # code int -> (RpcCode, op method name); built once at class init
    _DISPATCH: dict = {}

    def __init__(self, master):
        self.master = master
        self.fs: MasterFilesystem = master.fs
        if not MasterHandler._DISPATCH:
            d = {}
            for code in RpcCode:
                name = f"op_{code.name.lower()}"
                if hasattr(MasterHandler, name):
                    d[int(code)] = (code, name)
            MasterHandler._DISPATCH = d
        # bound-method dispatch: one dict hit per op instead of a
        # getattr on the QPS hot path
        self._bound = {c: (code, getattr(self, name))
                       for c, (code, name) in MasterHandler._DISPATCH.items()}
        self._audit = master.fs.conf.master.audit_log

    async def handle(self, msg: Message, conn) -> Optional[Message]:
        ent = self._bound.get(msg.code)
        if ent is None:
            raise err.Unsupported(f"rpc code {msg.code}")
        code, fn = ent
        pbuf = conn.state.get("pbuf", False)
        if msg.raw_header:
            # candidate protobuf header (reference-client wire compat):
            # decode via the transcribed schema and pin the connection
            from curvine_amd.rpc import proto as _proto
            decoded = _proto.decode_request(msg.code, msg.raw_header)
            if decoded is None:
                raise err.InvalidArgument(
                    f"undecodable header for code {msg.code}")
            msg.header = decoded
            msg.raw_header = b""
            pbuf = conn.state["pbuf"] = True
        raft = self.master.raft
        if raft is not None and not raft.is_leader and code not in _READONLY_OK:
            raise err.NotLeader(f"leader={raft.leader_addr or ''}")
        if raft is not None and raft.is_leader and not raft.read_ready \
                and code not in _READONLY_OK:
            # ReadIndex rule: a fresh leader may lag entries committed by
            # the old leader; serve nothing until our no-op commits
            await raft.wait_commit(raft.term_start_index)
        # mutation retry cache (fs_retry_cache.rs analog): a replayed
        # request (connector retry after timeout/failover) returns the
        # original reply instead of re-executing (double add_block etc.)
        rkey = None
        if code in _MUTATIONS and msg.header.get("cid") is not None:
            rkey = (msg.header["cid"], msg.header.get("rid"))
            cached = self.master.retry_cache.get(rkey)
            if cached is not None:
                return msg.reply(cached)
        auditing = self._audit
        t0 = time.perf_counter() if auditing else 0.0
        try:
            if pbuf:
                # reference peers expect errors in the DATA section in
                # the ErrorEncoder layout, not a msgpack header
                try:
                    return await self._handle_pbuf(msg, fn, raft, rkey)
                finally:
                    if auditing:
                        audit.info("cmd=%s used_us=%d", code.name,
                                   int((time.perf_counter() - t0) * 1e6))
            op_before = self.master.journal.op_id
            reply = fn(msg.header, msg.data)
            if asyncio.iscoroutine(reply):
                reply = await reply
            if raft is not None and raft.is_leader \
                    and self.master.journal.op_id > op_before:
                # withhold the reply until the mutation's entries commit
                await raft.wait_commit(self.master.journal.op_id)
            if rkey is not None:
                self.master.retry_cache.put(rkey, reply or {})
            out = msg.reply(reply or {})
            if pbuf:
                from curvine_amd.rpc import proto as _proto
                enc = _proto.encode_response(msg.code, out.header)
                if enc is not None:
                    out.header = {}
                    out.raw_header = enc
            return out
        finally:
            if auditing:
                audit.info("cmd=%s used_us=%d", code.name,
                           int((time.perf_counter() - t0) * 1e6))

    async def _handle_pbuf(self, msg, fn, raft, rkey):
        from curvine_amd.rpc import proto as _proto
        from curvine_amd.rpc.message import Status
        try:
            op_before = self.master.journal.op_id
            reply = fn(msg.header, msg.data)
            if asyncio.iscoroutine(reply):
                reply = await reply
            if raft is not None and raft.is_leader \
                    and self.master.journal.op_id > op_before:
                await raft.wait_commit(self.master.journal.op_id)
        except Exception as e:  # noqa: BLE001 — reference error wire
            out = msg.reply(resp_status=Status.Error)
            out.data = _proto.encode_error(e)
            return out
        if rkey is not None:
            self.master.retry_cache.put(rkey, reply or {})
        out = msg.reply(reply or {})
        enc = _proto.encode_response(msg.code, out.header)
        if enc is not None:
            out.header = {}
            out.raw_header = enc
        return out

    # ---------------- filesystem ----------------
    def op_heartbeat(self, h, d):
        return {"ts": int(time.time() * 1000)}

    def op_mkdir(self, h, d):
        st = self.fs.mkdir(h["path"], h.get("mode", 0o755),
                           h.get("create_parents", True))
        return {"status": st.to_dict()}

    def op_createfile(self, h, d):
        st = self.fs.create_dict(h["path"], h.get("block_size", 0),
                                 h.get("replicas", 0),
                                 h.get("storage_tier", ""),
                                 h.get("overwrite", False),
                                 h.get("mode", 0o644))
        return {"status": st}

    def op_appendfile(self, h, d):
        return {"file_blocks": self.fs.append(h["path"]).to_dict()}

    def op_openfile(self, h, d):
        return {"file_blocks": self.fs.open(h["path"]).to_dict()}

    def op_filestatus(self, h, d):
        return {"status": self.fs.file_status(h["path"]).to_dict()}

    def op_liststatus(self, h, d):
        return {"statuses": [s.to_dict() for s in self.fs.list_status(h["path"])]}

    def op_exists(self, h, d):
        return {"exists": self.fs.exists(h["path"])}

    def op_delete(self, h, d):
        n = self.fs.delete(h["path"], h.get("recursive", False))
        return {"deleted_blocks": n}

    def op_rename(self, h, d):
        self.fs.rename(h["src"], h["dst"])
        return {}

    def op_addblock(self, h, d):
        lb = self.fs.add_block(h["path"], h.get("commit_prev_len", -1),
                               h.get("client_host", ""),
                               h.get("client_worker_id", -1),
                               h.get("exclude_workers"))
        return {"block": lb.to_dict()}

    def op_completefile(self, h, d):
        st = self.fs.complete_file_dict(h["path"], h["length"],
                                        h.get("block_lens"),
                                        h.get("commits"))
        return {"status": st}

    def op_getblocklocations(self, h, d):
        return {"file_blocks": self.fs.get_block_locations(h["path"]).to_dict()}

    def op_getfilesysteminfo(self, h, d):
        # client handshake carries its version report here (the
        # reference's GetFilesystemInfoRequest.component_info = 1000)
        ci = h.get("component_info")
        pol = self.master.compat_policy
        if pol.should_evaluate(ci is not None):
            v = pol.check_client(ci)
            self.master.compat_warn_clients.warn(
                (ci or {}).get("component", "client"), v)
            if v.rejects(pol.mode):
                raise err.IncompatibleVersion(f"client: {v.describe()}")
        out = self.fs.master_info()
        from curvine_amd.compat import component_info
        out["component_info"] = component_info("master")
        return out

    def op_setattr(self, h, d):
        st = self.fs.set_attr(h["path"], **{
            k: h.get(k) for k in ("mode", "uid", "gid", "atime_ms", "mtime_ms",
                                  "ttl_ms", "ttl_action", "replicas",
                                  "storage_tier", "xattrs")})
        return {"status": st.to_dict()}

    def op_symlink(self, h, d):
        return {"status": self.fs.symlink(h["path"], h["target"]).to_dict()}

    def op_link(self, h, d):
        return {"status": self.fs.link(h["src"], h["dst"]).to_dict()}

    def op_resizefile(self, h, d):
        return {"status": self.fs.resize(h["path"], h["length"]).to_dict()}

    def op_free(self, h, d):
        return {"freed_blocks": self.fs.free(h["path"], h.get("recursive", False))}

    def op_createfilesbatch(self, h, d):
        out = []
        for req in h["files"]:
            st = self.fs.create(req["path"], req.get("block_size", 0),
                                req.get("replicas", 0),
                                req.get("storage_tier", ""),
                                req.get("overwrite", False))
            out.append(st.to_dict())
        return {"statuses": out}

    def op_addblocksbatch(self, h, d):
        out = []
        for req in h["blocks"]:
            lb = self.fs.add_block(req["path"], req.get("commit_prev_len", -1),
                                   h.get("client_host", ""),
                                   h.get("client_worker_id", -1))
            out.append(lb.to_dict())
        return {"blocks": out}

    def op_completefilesbatch(self, h, d):
        out = []
        for req in h["files"]:
            st = self.fs.complete_file(req["path"], req["length"],
                                       req.get("block_lens"),
                                       req.get("commits"))
            out.append(st.to_dict())
        return {"statuses": out}

    # ---------------- mounts ----------------
    def op_mount(self, h, d):
        mi = self.master.mounts.mount(h["curvine_path"], h["ufs_path"],
                                      h.get("properties", {}),
                                      h.get("cache_mode", "cache"),
                                      h.get("auto_cache", True))
        return {"mount": mi.to_dict()}

    def op_unmount(self, h, d):
        self.master.mounts.unmount(h["curvine_path"])
        return {}

    def op_updatemount(self, h, d):
        mi = self.master.mounts.update(h["curvine_path"], h.get("properties", {}),
                                       h.get("cache_mode"), h.get("auto_cache"))
        return {"mount": mi.to_dict()}

    def op_getmounttable(self, h, d):
        return {"mounts": [m.to_dict() for m in self.master.mounts.table()]}

    def op_getmountinfo(self, h, d):
        mi = self.master.mounts.lookup(h["path"])
        return {"mount": mi.to_dict() if mi else None}

    # ---------------- jobs ----------------
    async def _forward_transfer(self, code, h):
        """Standalone transfer-service deployment: the master proxies the
        job surface to the external service (conf.job.service_addr)."""
        conn = self.master.transfer_conn()
        r = await conn.rpc(code, h)
        return r.header

    def op_submitjob(self, h, d):
        if self.master.conf.job.service_addr:
            return self._forward_transfer(RpcCode.SubmitJob, h)
        return self.master.jobs.submit(h)

    def op_getjobstatus(self, h, d):
        if self.master.conf.job.service_addr:
            return self._forward_transfer(RpcCode.GetJobStatus, h)
        return self.master.jobs.status(h["job_id"])

    def op_canceljob(self, h, d):
        if self.master.conf.job.service_addr:
            return self._forward_transfer(RpcCode.CancelJob, h)
        return self.master.jobs.cancel(h["job_id"])

    def op_reporttask(self, h, d):
        if self.master.conf.job.service_addr:
            return self._forward_transfer(RpcCode.ReportTask, h)
        self.master.jobs.report_task(h)
        return {}

    def op_submittask(self, h, d):
        """Transfer-service -> master: enqueue one worker command for
        delivery on that worker's next heartbeat (scheduler.rs:43
        dispatch analog)."""
        self.fs.workers.add_command(h["worker_id"], h["command"])
        return {}

    # ---------------- worker plane ----------------
    def op_workerheartbeat(self, h, d):
        info = WorkerInfo.from_dict(h["worker"])
        pol = self.master.compat_policy
        if pol.should_evaluate(info.component_info is not None):
            v = pol.check_worker(info.component_info)
            self.master.compat_warn_workers.warn(
                info.address.worker_id, v)
            if v.rejects(pol.mode):
                raise err.IncompatibleVersion(
                    f"worker {info.address.worker_id}: {v.describe()}")
        cmds = self.fs.worker_heartbeat(info, h.get("added_blocks", []),
                                        h.get("removed_blocks", []))
        return {"commands": cmds}

    def op_workerblockreport(self, h, d):
        to_delete = self.fs.block_report(h["worker_id"], h.get("blocks", []))
        return {"delete_blocks": to_delete}

    # ---------------- replication ----------------
    def op_reportblockreplicationresult(self, h, d):
        self.master.replication.report_result(h)
        return {}

    def op_requestreplacementworker(self, h, d):
        ws = self.fs.workers.choose_workers(
            1, "load_based", exclude=set(h.get("exclude", [])))
        return {"worker": ws[0].address.to_dict()}

    def op_reportunderreplicatedblocks(self, h, d):
        for bid in h.get("block_ids", []):
            self.master.replication.mark_under_replicated(bid)
        return {}

    def op_metricsreport(self, h, d):
        """Clients push their metrics snapshots (MetricsReport, code 60)
        for cluster-wide visibility; bounded per-client store exposed on
        the master's /metrics and /api/client-metrics."""
        cid = str(h.get("client_id", ""))[:128]
        if cid:
            store = self.master.client_metrics
            store[cid] = {"ts_ms": int(time.time() * 1000),
                          "kind": str(h.get("kind", "client"))[:32],
                          "metrics": h.get("metrics") or {}}
            while len(store) > 256:   # bounded: drop the oldest
                store.pop(min(store, key=lambda k: store[k]["ts_ms"]))
        return {}

    def op_getmetadatasnapshotpage(self, h, d):
        """Page through the full inode table (CV-metadata sync,
        master_filesystem.rs:836-875 analog)."""
        offset = h.get("page_token", 0)
        limit = min(h.get("limit", 1000), 10_000)
        ids = sorted(self.fs.fs_dir.inodes)
        page = [self.fs.fs_dir.inodes[i].to_state()
                for i in ids[offset:offset + limit]]
        nxt = offset + limit if offset + limit < len(ids) else None
        return {"inodes": page, "next_token": nxt,
                "op_id": self.master.journal.op_id}

    def op_getmetadatadeltapage(self, h, d):
        """Journal entries since op_id; requests older than the retained
        window must fall back to a snapshot page sweep."""
        since = h.get("since_op_id", 0)
        recent = self.master.recent_entries
        if recent and recent[0]["op_id"] > since + 1:
            return {"snapshot_required": True,
                    "op_id": self.master.journal.op_id}
        limit = min(h.get("limit", 1000), 10_000)
        out = [e for e in recent if e["op_id"] > since][:limit]
        return {"entries": out, "op_id": self.master.journal.op_id,
                "snapshot_required": False}

    def op_decommissionworker(self, h, d):
        self.fs.workers.decommission(h["worker_id"])
        # proactively queue re-replication of its blocks
        for bid, locs in self.fs.workers.block_locs.items():
            if h["worker_id"] in locs:
                self.master.replication.mark_under_replicated(bid)
        return {"state": "decommissioning"}

    # ---------------- transfer service (codes 46-54) ----------------
    # the standalone curvine-data-transfer surface; backed by the same
    # job manager (transfer/mod.rs analog with the in-memory store)
    def op_submittransfer(self, h, d):
        return self.master.jobs.submit(h)

    def op_gettransferstatus(self, h, d):
        return self.master.jobs.status(h.get("job_id") or h.get("transfer_id"))

    def op_canceltransfer(self, h, d):
        return self.master.jobs.cancel(h.get("job_id") or h.get("transfer_id"))

    def op_reporttransfertask(self, h, d):
        self.master.jobs.report_task(h)
        return {}

    def op_querytransfertask(self, h, d):
        job = self.master.jobs.jobs.get(h.get("job_id"))
        if job is None:
            raise err.JobNotFound(str(h.get("job_id")))
        return {"tasks": list(job["tasks"].values())}

    def op_listtransfers(self, h, d):
        return {"transfers": [
            {k: j[k] for k in ("job_id", "state", "total", "done", "failed",
                               "path")}
            for j in self.master.jobs.jobs.values()]}

    def op_retrytransfer(self, h, d):
        if self.master.conf.job.service_addr:
            return self._forward_transfer(RpcCode.RetryTransfer, h)
        return self.master.jobs.retry(h.get("job_id", ""))

    # ---------------- raft protocol ----------------
    def op_raftvote(self, h, d):
        return self.master.raft.on_vote(h)

    def op_raftappendentries(self, h, d):
        return self.master.raft.on_append(h)

    def op_raftinstallsnapshot(self, h, d):
        return self.master.raft.on_install_snapshot(h, d)

    def op_rafttransferleader(self, h, d):
        if h.get("leader") is not None:   # inbound TimeoutNow from leader
            return self.master.raft.on_transfer_leader(h)
        # client/CLI request: we are the leader, hand off to target
        return self.master.raft.transfer_leadership(h["target"])
