"""Metadata service: namespace ops, journal replay, snapshot, placement."""
import os

import pytest

from curvine_amd import errors as err
from curvine_amd.conf import ClusterConf
from curvine_amd.master.filesystem import MasterFilesystem
from curvine_amd.master.journal import JournalWriter
from curvine_amd.model import StorageInfo, WorkerAddress, WorkerInfo


def make_fs(tmp_path, name="j"):
    conf = ClusterConf()
    conf.journal.journal_dir = str(tmp_path / name)
    conf.master.worker_expire_ms = 100
    return MasterFilesystem(conf)


def add_worker(fs, wid=1, host="127.0.0.1", cap=1 << 30):
    info = WorkerInfo(address=WorkerAddress(worker_id=wid, hostname=host, rpc_port=9000 + wid),
                      storages=[StorageInfo(tier="MEM", capacity=cap)])
    fs.worker_heartbeat(info, [], [])
    return info


def test_mkdir_create_list(tmp_path):
    fs = make_fs(tmp_path)
    fs.mkdir("/a/b/c", create_parents=True)
    assert fs.file_status("/a/b/c").is_dir
    st = fs.create("/a/b/c/f1.txt")
    assert not st.is_dir and not st.is_complete
    fs.complete_file("/a/b/c/f1.txt", 0)
    names = [s.name for s in fs.list_status("/a/b/c")]
    assert names == ["f1.txt"]
    with pytest.raises(err.FileAlreadyExists):
        fs.create("/a/b/c/f1.txt")
    st2 = fs.create("/a/b/c/f1.txt", overwrite=True)
    assert st2.inode_id != st.inode_id


def test_write_flow_and_locations(tmp_path):
    fs = make_fs(tmp_path)
    add_worker(fs, 1)
    add_worker(fs, 2, host="10.0.0.2")
    fs.create("/f", block_size=100, replicas=1)
    lb1 = fs.add_block("/f")
    assert lb1.block.block_id > 0 and len(lb1.locations) == 1
    lb2 = fs.add_block("/f", commit_prev_len=100)
    assert lb2.offset == 100
    # worker reports the blocks
    w = fs.workers.get(lb1.locations[0].worker_id)
    fs.worker_heartbeat(fs.workers.get(1), [{"block_id": lb1.block.block_id, "tier": "MEM"},
                                            {"block_id": lb2.block.block_id, "tier": "MEM"}], [])
    st = fs.complete_file("/f", 150, [100, 50])
    assert st.length == 150 and st.is_complete
    fb = fs.open("/f")
    assert [b.block.length for b in fb.blocks] == [100, 50]
    assert fb.blocks[0].locations[0].worker_id == 1


def test_rename_delete(tmp_path):
    fs = make_fs(tmp_path)
    fs.mkdir("/src", create_parents=True)
    fs.create("/src/f")
    fs.complete_file("/src/f", 0)
    fs.mkdir("/dst")
    fs.rename("/src/f", "/dst/g")
    assert fs.exists("/dst/g") and not fs.exists("/src/f")
    with pytest.raises(err.DirNotEmpty):
        fs.delete("/dst")
    fs.delete("/dst", recursive=True)
    assert not fs.exists("/dst")


def test_symlink_link_resize(tmp_path):
    fs = make_fs(tmp_path)
    add_worker(fs, 1)
    fs.create("/data", block_size=10)
    for i in range(3):
        fs.add_block("/data", commit_prev_len=10 if i else -1)
    fs.complete_file("/data", 25)
    st = fs.symlink("/lnk", "/data")
    assert st.is_symlink and st.symlink_target == "/data"
    fs.link("/data", "/hard")
    assert fs.file_status("/hard").nlink == 2
    st = fs.resize("/data", 15)
    assert st.length == 15
    fb = fs.open("/data")
    assert [b.block.length for b in fb.blocks] == [10, 5]


def test_journal_replay(tmp_path):
    fs = make_fs(tmp_path)
    add_worker(fs, 1)
    fs.mkdir("/d", create_parents=True)
    fs.create("/d/f", block_size=64)
    fs.add_block("/d/f")
    fs.complete_file("/d/f", 42)
    fs.set_attr("/d/f", mode=0o600, ttl_ms=99)
    fs.journal.flush()

    fs2 = make_fs(tmp_path)   # same journal dir
    fs2.restore()
    st = fs2.file_status("/d/f")
    assert st.length == 42 and st.mode == 0o600 and st.ttl_ms == 99
    assert fs2.fs_dir.next_block_id == fs.fs_dir.next_block_id
    assert fs2.fs_dir.next_inode_id == fs.fs_dir.next_inode_id


def test_snapshot_and_incremental_replay(tmp_path):
    fs = make_fs(tmp_path)
    fs.mkdir("/pre", create_parents=True)
    fs.checkpoint()
    fs.create("/pre/after_snap")
    fs.complete_file("/pre/after_snap", 7)
    fs.journal.flush()

    fs2 = make_fs(tmp_path)
    fs2.restore()
    assert fs2.exists("/pre")
    assert fs2.file_status("/pre/after_snap").length == 7


def test_placement_policies(tmp_path):
    fs = make_fs(tmp_path)
    for i in range(1, 5):
        add_worker(fs, i, host=f"10.0.0.{i}", cap=(1 << 30) * i)
    ws = fs.workers.choose_workers(2, "local", client_host="10.0.0.3", client_worker_id=3)
    assert ws[0].address.worker_id == 3
    ws = fs.workers.choose_workers(4, "load_based")
    assert len(ws) == 4
    with pytest.raises(err.NoAvailableWorker):
        fs.workers.choose_workers(1, "local", exclude={1, 2, 3, 4})


def test_worker_expiry_drops_locations(tmp_path):
    import time
    fs = make_fs(tmp_path)
    add_worker(fs, 1)
    fs.create("/f", replicas=1)
    lb = fs.add_block("/f")
    fs.worker_heartbeat(fs.workers.get(1), [{"block_id": lb.block.block_id, "tier": "MEM"}], [])
    fs.complete_file("/f", 10)
    assert fs.open("/f").blocks[0].locations
    time.sleep(0.15)
    lost = fs.workers.check_expired()
    assert lost == [1]
    under = fs.handle_lost_workers(lost)
    assert lb.block.block_id in under
    assert not fs.open("/f").blocks[0].locations


def test_free_keeps_metadata(tmp_path):
    fs = make_fs(tmp_path)
    add_worker(fs, 1)
    fs.create("/f")
    lb = fs.add_block("/f")
    fs.complete_file("/f", 10)
    n = fs.free("/f")
    assert n == 1
    st = fs.file_status("/f")
    assert st.length == 10
    assert fs.open("/f").blocks == []


def test_metadata_snapshot_and_delta_pages(tmp_path):
    """CV-metadata sync paging (codes 28/29)."""
    import asyncio

    from curvine_amd.master.server import Master
    from curvine_amd.conf import ClusterConf
    from curvine_amd.rpc.client import RpcClient
    from curvine_amd.rpc.codes import RpcCode

    async def main():
        conf = ClusterConf()
        conf.master.rpc_port = 0
        conf.journal.journal_dir = str(tmp_path / "j2")
        m = await Master(conf).start()
        c = await RpcClient("127.0.0.1", m.rpc.port).connect()
        for i in range(25):
            await c.rpc(RpcCode.Mkdir, {"path": f"/pg/d{i}",
                                        "create_parents": True})
        # snapshot pages
        seen = []
        token = 0
        while token is not None:
            r = await c.rpc(RpcCode.GetMetadataSnapshotPage,
                            {"page_token": token, "limit": 10})
            seen += r.header["inodes"]
            token = r.header["next_token"]
        assert len(seen) == len(m.fs.fs_dir.inodes)
        op_now = r.header["op_id"]
        # delta since op_now: empty
        r = await c.rpc(RpcCode.GetMetadataDeltaPage, {"since_op_id": op_now})
        assert r.header["entries"] == []
        # mutate, delta catches it
        await c.rpc(RpcCode.Mkdir, {"path": "/pg/extra",
                                    "create_parents": True})
        r = await c.rpc(RpcCode.GetMetadataDeltaPage, {"since_op_id": op_now})
        assert any(e["op"] == "mkdir" and e.get("name") == "extra"
                   for e in r.header["entries"])
        await c.close()
        await m.stop()
    asyncio.new_event_loop().run_until_complete(main())


def test_replay_reproduces_timestamps(tmp_path):
    """Journal entries carry a log-time stamp: a WAL replay yields the
    SAME mtimes/create times as the live run (deterministic replay —
    raft followers, sqlite partial-flush restarts and crash recovery all
    converge bit-for-bit on time fields)."""
    import asyncio
    import copy

    from curvine_amd.master.server import Master
    from curvine_amd.testing import test_conf as _tc

    conf = _tc(str(tmp_path))
    conf.master.inode_db = False       # force pure WAL replay

    async def main():
        m = await Master(conf).start()
        m.fs.mkdir("/ts/dir", 0o755, True)
        m.fs.create("/ts/f", 0, 1, "", False)
        m.fs.complete_file("/ts/f", 7, [7])
        m.fs.rename("/ts/f", "/ts/g")
        live = {p: (m.fs.file_status(p).mtime_ms)
                for p in ("/ts", "/ts/dir", "/ts/g")}
        await m.stop()
        await asyncio.sleep(0.05)      # ensure the clock moved on
        m2 = await Master(copy.deepcopy(conf)).start()
        for p, mt in live.items():
            assert m2.fs.file_status(p).mtime_ms == mt, p
        await m2.stop()

    asyncio.new_event_loop().run_until_complete(main())
