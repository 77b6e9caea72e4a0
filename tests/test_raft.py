"""Raft master HA: election, replicated namespace, leader failover,
restart catch-up, snapshot install."""
import asyncio
import copy

import pytest

from curvine_amd.conf import ClusterConf
from curvine_amd.master.server import Master
from curvine_amd.rpc.client import ClusterConnector
from curvine_amd.rpc.codes import RpcCode


@pytest.fixture
def loop():
    loop = asyncio.new_event_loop()
    asyncio.set_event_loop(loop)
    yield loop
    loop.close()


def run(loop, coro):
    return loop.run_until_complete(coro)


async def start_group(tmp_path, n=3, eto=400, hb=100):
    import socket
    ports = []
    socks = []
    for _ in range(n):
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        ports.append(s.getsockname()[1])
        socks.append(s)
    for s in socks:
        s.close()
    peers = [f"{i + 1}@127.0.0.1:{p}" for i, p in enumerate(ports)]
    masters = []
    for i, p in enumerate(ports):
        conf = ClusterConf()
        conf.master.rpc_port = p
        conf.journal.journal_dir = str(tmp_path / f"m{i + 1}")
        conf.journal.peers = list(peers)
        conf.journal.node_id = i + 1
        conf.journal.election_timeout_ms = eto
        conf.journal.heartbeat_interval_ms = hb
        conf.master.heartbeat_check_ms = 500
        masters.append(await Master(conf).start())
    return masters, ports, peers


async def wait_leader(masters, timeout=10.0):
    for _ in range(int(timeout / 0.1)):
        leaders = [m for m in masters if m.raft.is_leader]
        if len(leaders) == 1:
            return leaders[0]
        await asyncio.sleep(0.1)
    raise AssertionError("no single leader elected")


def test_election_and_replication(loop, tmp_path):
    async def main():
        masters, ports, _ = await start_group(tmp_path)
        leader = await wait_leader(masters)
        conn = ClusterConnector([f"127.0.0.1:{p}" for p in ports],
                                timeout_ms=8000)
        await conn.rpc(RpcCode.Mkdir, {"path": "/ha/dir", "create_parents": True})
        r = await conn.rpc(RpcCode.CreateFile,
                           {"path": "/ha/f", "block_size": 1 << 20,
                            "replicas": 1, "storage_tier": "MEM",
                            "overwrite": False, "mode": 0o644})
        await conn.rpc(RpcCode.CompleteFile, {"path": "/ha/f", "length": 0,
                                              "block_lens": []})
        await asyncio.sleep(0.5)   # let followers apply
        for m in masters:
            assert m.fs.fs_dir.resolve("/ha/f") is not None, \
                f"node {m.conf.journal.node_id} missing entry"
        await conn.close()
        for m in masters:
            await m.stop()
    run(loop, main())


def test_leader_failover(loop, tmp_path):
    async def main():
        masters, ports, _ = await start_group(tmp_path)
        leader = await wait_leader(masters)
        conn = ClusterConnector([f"127.0.0.1:{p}" for p in ports],
                                timeout_ms=8000, retries=6)
        await conn.rpc(RpcCode.Mkdir, {"path": "/pre", "create_parents": True})
        # kill the leader
        await leader.stop()
        rest = [m for m in masters if m is not leader]
        new_leader = await wait_leader(rest)
        assert new_leader is not leader
        # old data survived; new mutations work
        r = await conn.rpc(RpcCode.Exists, {"path": "/pre"})
        assert r.header["exists"]
        await conn.rpc(RpcCode.Mkdir, {"path": "/post", "create_parents": True})
        await asyncio.sleep(0.5)
        for m in rest:
            assert m.fs.fs_dir.resolve("/post") is not None
        await conn.close()
        for m in rest:
            await m.stop()
    run(loop, main())


def test_restart_catches_up(loop, tmp_path):
    async def main():
        masters, ports, peers = await start_group(tmp_path)
        leader = await wait_leader(masters)
        conn = ClusterConnector([f"127.0.0.1:{p}" for p in ports],
                                timeout_ms=8000, retries=6)
        # stop a follower
        follower = next(m for m in masters if not m.raft.is_leader)
        fid = follower.conf.journal.node_id
        fconf = copy.deepcopy(follower.conf)
        await follower.stop()
        live = [m for m in masters if m is not follower]
        # mutate while it is down
        for i in range(10):
            await conn.rpc(RpcCode.Mkdir, {"path": f"/while_down/{i}",
                                           "create_parents": True})
        # restart it
        fconf.master.rpc_port = ports[fid - 1]
        restarted = await Master(fconf).start()
        for _ in range(100):
            await asyncio.sleep(0.1)
            if restarted.fs.fs_dir.resolve("/while_down/9") is not None:
                break
        assert restarted.fs.fs_dir.resolve("/while_down/9") is not None
        await conn.close()
        for m in live + [restarted]:
            await m.stop()
    run(loop, main())


def test_chunked_snapshot_install(loop, tmp_path, monkeypatch):
    """A follower far behind a compacted log catches up via the CHUNKED
    snapshot stream (multi-frame; single-frame installs would cap the
    namespace at the 16 MiB frame limit)."""
    from curvine_amd.master.raft import RaftNode
    monkeypatch.setattr(RaftNode, "SNAP_CHUNK", 4096)   # force many chunks

    async def main():
        masters, ports, peers = await start_group(tmp_path)
        leader = await wait_leader(masters)
        conn = ClusterConnector([f"127.0.0.1:{p}" for p in ports],
                                timeout_ms=8000, retries=6)
        follower = next(m for m in masters if not m.raft.is_leader)
        fid = follower.conf.journal.node_id
        fconf = copy.deepcopy(follower.conf)
        await follower.stop()
        live = [m for m in masters if m is not follower]

        # grow the namespace (long names -> a multi-chunk snapshot blob),
        # then compact the leader's log so append catch-up is impossible
        for i in range(40):
            await conn.rpc(RpcCode.Mkdir, {
                "path": f"/snapdir/{'x' * 200}-{i}",
                "create_parents": True})
        leader = await wait_leader(live)
        leader.checkpoint()
        assert leader.raft.log.snapshot_index > 0

        fconf.master.rpc_port = ports[fid - 1]
        restarted = await Master(fconf).start()
        for _ in range(150):
            await asyncio.sleep(0.1)
            if restarted.fs.fs_dir.resolve(
                    f"/snapdir/{'x' * 200}-39") is not None:
                break
        assert restarted.fs.fs_dir.resolve(
            f"/snapdir/{'x' * 200}-39") is not None
        # and it keeps participating
        await conn.rpc(RpcCode.Mkdir, {"path": "/after_snap",
                                       "create_parents": True})
        for _ in range(50):
            await asyncio.sleep(0.1)
            if restarted.fs.fs_dir.resolve("/after_snap") is not None:
                break
        assert restarted.fs.fs_dir.resolve("/after_snap") is not None
        await conn.close()
        for m in live + [restarted]:
            await m.stop()
    run(loop, main())


def test_prevote_blocks_term_inflation(loop, tmp_path):
    """A node without quorum must not inflate its term while isolated
    (pre-vote): when peers return, the cluster resumes at a sane term."""
    async def main():
        masters, ports, peers = await start_group(tmp_path)
        leader = await wait_leader(masters)
        term0 = leader.raft.term
        survivor = next(m for m in masters if not m.raft.is_leader)
        others = [m for m in masters if m is not survivor]
        confs = [copy.deepcopy(m.conf) for m in others]
        for m in others:
            await m.stop()
        # several election timeouts pass with no quorum
        await asyncio.sleep(2.5)
        assert survivor.raft.term <= term0 + 1, \
            f"term inflated to {survivor.raft.term}"
        assert not survivor.raft.is_leader
        # peers return: a leader emerges without a huge term jump
        restarted = []
        for c, p in zip(confs, [m.conf.master.rpc_port for m in others]):
            restarted.append(await Master(c).start())
        leader2 = await wait_leader([survivor] + restarted, timeout=15)
        assert leader2.raft.term <= term0 + 3
        for m in [survivor] + restarted:
            await m.stop()
    run(loop, main())


def test_leadership_transfer(loop, tmp_path):
    """RaftTransferLeader: the leader hands off to a chosen follower
    (TimeoutNow analog) and steps down when its term arrives."""
    async def main():
        masters, ports, peers = await start_group(tmp_path)
        leader = await wait_leader(masters)
        conn = ClusterConnector([f"127.0.0.1:{p}" for p in ports],
                                timeout_ms=8000, retries=6)
        await conn.rpc(RpcCode.Mkdir, {"path": "/xfer", "create_parents": True})
        target = next(m for m in masters if not m.raft.is_leader)
        tid = target.conf.journal.node_id
        r = await conn.rpc(RpcCode.RaftTransferLeader, {"target": tid})
        assert r.header.get("accepted") is True
        for _ in range(100):
            await asyncio.sleep(0.1)
            if target.raft.is_leader:
                break
        assert target.raft.is_leader
        # the cluster keeps mutating under the new leader
        await conn.rpc(RpcCode.Mkdir, {"path": "/xfer/after",
                                       "create_parents": True})
        assert target.fs.fs_dir.resolve("/xfer/after") is not None
        await conn.close()
        for m in masters:
            await m.stop()
    run(loop, main())


def test_crashed_leader_uncommitted_tail_converges(loop, tmp_path):
    """a leader that applied an uncommitted tail
    optimistically, crashed, and restarts as a follower must NOT keep the
    phantom state.  Boot replays only up to the persisted commit
    watermark; the new leader's conflicting entries truncate the tail and
    the replica converges."""
    async def main():
        masters, ports, peers = await start_group(tmp_path)
        leader = await wait_leader(masters)
        confs = {m.conf.journal.node_id: copy.deepcopy(m.conf)
                 for m in masters}
        lid = leader.conf.journal.node_id
        # isolate the leader: stop both followers so nothing commits
        followers = [m for m in masters if m is not leader]
        for f in followers:
            await f.stop()
        # leader applies locally at append time; never reaches a majority
        leader.fs.fs_dir.mkdir("/phantom")
        assert leader.fs.fs_dir.resolve("/phantom") is not None
        await leader.stop()
        # restart the two followers; they elect a leader without /phantom
        live = []
        for f in followers:
            c = confs[f.conf.journal.node_id]
            c.master.rpc_port = ports[c.journal.node_id - 1]
            live.append(await Master(c).start())
        new_leader = await wait_leader(live)
        conn = ClusterConnector(
            [f"127.0.0.1:{ports[m.conf.journal.node_id - 1]}" for m in live],
            timeout_ms=8000, retries=6)
        for i in range(5):
            await conn.rpc(RpcCode.Mkdir, {"path": f"/real/{i}",
                                           "create_parents": True})
        # restart the crashed leader: boot must NOT apply the phantom tail
        lc = confs[lid]
        lc.master.rpc_port = ports[lid - 1]
        restarted = await Master(lc).start()
        assert restarted.fs.fs_dir.resolve("/phantom") is None, \
            "boot replayed an uncommitted tail"
        for _ in range(100):
            await asyncio.sleep(0.1)
            if restarted.fs.fs_dir.resolve("/real/4") is not None:
                break
        assert restarted.fs.fs_dir.resolve("/real/4") is not None
        assert restarted.fs.fs_dir.resolve("/phantom") is None, \
            "conflicting-entry truncation left phantom state applied"
        await conn.close()
        for m in live + [restarted]:
            await m.stop()
    run(loop, main())


def test_native_reads_commit_gated(loop, tmp_path):
    """the C++ meta mirror must only serve COMMITTED
    state.  With followers down, a leader-side optimistic apply must not
    be visible through the native frontend; once a majority returns and
    the entry commits, it must be."""
    async def main():
        masters, ports, peers = await start_group(tmp_path)
        leader = await wait_leader(masters)
        nm = leader.native_meta
        if nm is None:
            pytest.skip("native meta frontend unavailable")
        conn = ClusterConnector([f"127.0.0.1:{p}" for p in ports],
                                timeout_ms=8000, retries=6)
        await conn.rpc(RpcCode.Mkdir, {"path": "/gated",
                                       "create_parents": True})
        # committed state is natively visible
        r = await conn.rpc(RpcCode.FileStatus, {"path": "/gated"})
        assert r.header.get("status", r.header).get("file_type") is not None \
            or r.header  # served
        confs = {m.conf.journal.node_id: copy.deepcopy(m.conf)
                 for m in masters}
        followers = [m for m in masters if m is not leader]
        for f in followers:
            await f.stop()
        # leader applies optimistically; commit can't advance
        leader.fs.fs_dir.mkdir("/gated/uncommitted")
        assert leader.fs.fs_dir.resolve("/gated/uncommitted") is not None
        assert nm.mirror_gate is not None and nm.mirror_gate.pending, \
            "mirror op was not gated"
        # the native tree must NOT have it: ask through the wire on a
        # fresh connection straight at the leader
        from curvine_amd.rpc.client import RpcClient
        c = await RpcClient("127.0.0.1", leader.rpc.port,
                            timeout_ms=3000).connect()
        r = await c.rpc(RpcCode.Exists, {"path": "/gated/uncommitted"})
        assert r.header.get("exists") is False, \
            "native frontend served uncommitted state"
        await c.close()
        # bring a follower back: majority -> commit -> flushed to mirror
        fc = confs[followers[0].conf.journal.node_id]
        fc.master.rpc_port = ports[fc.journal.node_id - 1]
        back = await Master(fc).start()
        for _ in range(100):
            await asyncio.sleep(0.1)
            if leader.raft.commit_index >= leader.raft.log.last_index \
                    and not nm.mirror_gate.pending:
                break
        c = await RpcClient("127.0.0.1", leader.rpc.port,
                            timeout_ms=3000).connect()
        r = await c.rpc(RpcCode.Exists, {"path": "/gated/uncommitted"})
        assert r.header.get("exists") is True
        await c.close()
        await conn.close()
        for m in [leader, back, followers[1]]:
            try:
                await m.stop()
            except Exception:  # noqa: BLE001 — follower[1] already stopped
                pass
    run(loop, main())


def test_learner_replicates_without_voting(loop, tmp_path):
    """A learner (non-voting member) receives the replicated log but
    never becomes leader, never counts toward the commit quorum, and a
    2-voter+1-learner group keeps committing with the learner down."""
    async def main():
        import socket
        ports = []
        for _ in range(3):
            s = socket.socket()
            s.bind(("127.0.0.1", 0))
            ports.append(s.getsockname()[1])
            s.close()
        peers = [f"{i + 1}@127.0.0.1:{p}" for i, p in enumerate(ports)]
        masters = []
        for i, p in enumerate(ports):
            conf = ClusterConf()
            conf.master.rpc_port = p
            conf.journal.journal_dir = str(tmp_path / f"m{i + 1}")
            conf.journal.peers = list(peers)
            conf.journal.node_id = i + 1
            conf.journal.learners = [3]      # node 3 is the learner
            conf.journal.election_timeout_ms = 400
            conf.journal.heartbeat_interval_ms = 100
            masters.append(await Master(conf).start())
        try:
            for _ in range(60):
                leaders = [m for m in masters if m.raft.is_leader]
                if leaders:
                    break
                await asyncio.sleep(0.1)
            assert leaders, "no leader elected"
            leader = leaders[0]
            assert leader.raft.id != 3, "a learner must never lead"

            conn = ClusterConnector(
                [f"127.0.0.1:{p}" for p in ports], 5000)
            await conn.rpc(RpcCode.Mkdir, {"path": "/lrn"})
            await conn.rpc(RpcCode.CreateFile, {"path": "/lrn/a"})
            await conn.rpc(RpcCode.CompleteFile,
                           {"path": "/lrn/a", "length": 0})
            # the learner replicates the committed namespace
            learner = masters[2]
            for _ in range(50):
                if learner.fs.fs_dir.resolve("/lrn/a") is not None:
                    break
                await asyncio.sleep(0.1)
            assert learner.fs.fs_dir.resolve("/lrn/a") is not None

            # learner down: the 2 voters still commit (quorum excludes it)
            await learner.stop()
            await conn.rpc(RpcCode.Mkdir, {"path": "/lrn2"})
            r = await conn.rpc(RpcCode.Exists, {"path": "/lrn2"})
            assert r.header["exists"]
            await conn.close()
        finally:
            for m in masters:
                try:
                    await m.stop()
                except Exception:  # noqa: BLE001
                    pass
    run(loop, main())
