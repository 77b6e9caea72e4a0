"""Protobuf wire compatibility (rpc/proto.py): golden bytes, codec round
trips, and a simulated reference client speaking protobuf headers to a
live master over the real socket."""
import asyncio
import struct

import pytest

from curvine_amd.rpc import proto
from curvine_amd.rpc.codes import RpcCode

_HDR = struct.Struct(">IIBBQI")


# ---------------------------------------------------------------- golden
# Hand-derived from the protobuf wire spec (field<<3|wiretype, varints,
# LEN-prefixed strings) for the reference's proto/master.proto messages —
# byte-identical to what prost serializes for the same values.

GOLDEN = [
    # ExistsRequest{path:"/a"} -> field1 LEN 2 '/a'
    (RpcCode.Exists, "req", {"path": "/a"}, "0a022f61"),
    # GetFileStatusRequest{path:"/x/y"}
    (RpcCode.FileStatus, "req", {"path": "/x/y"}, "0a042f782f79"),
    # DeleteRequest{path:"/d", recursive:true} -> 0a 02 2f64 10 01
    (RpcCode.Delete, "req", {"path": "/d", "recursive": True},
     "0a022f641001"),
    # RenameRequest{src:"/a", dst:"/b", flags:0}
    (RpcCode.Rename, "req", {"src": "/a", "dst": "/b"},
     "0a022f6112022f621800"),
    # ExistsResponse{exists:true} -> field1 varint 1
    (RpcCode.Exists, "resp", {"exists": True}, "0801"),
    # RenameResponse{result:true}
    (RpcCode.Rename, "resp", {}, "0801"),
    # BlockReadRequest{id:7, off:0, len:4096, chunk_size:1048576,
    #   short_circuit:false, enable_read_ahead:true,
    #   read_ahead_len:4194304, drop_cache_len:1048576}
    (RpcCode.ReadBlock, "req",
     {"block_id": 7, "offset": 0, "length": 4096, "chunk_size": 1 << 20},
     "080710001880202080804028004001488080800250808040"),
]


@pytest.mark.parametrize("code,direction,header,want", GOLDEN)
def test_golden_bytes(code, direction, header, want):
    if direction == "req":
        got = proto.encode_request(int(code), header)
    else:
        got = proto.encode_response(int(code), header)
    assert got == bytes.fromhex(want), got.hex()


def test_request_roundtrips():
    cases = {
        RpcCode.Mkdir: {"path": "/p", "create_parents": True, "mode": 0o755},
        RpcCode.CreateFile: {"path": "/f", "block_size": 1 << 20,
                             "replicas": 2, "storage_tier": "SSD",
                             "overwrite": True, "mode": 0o600},
        RpcCode.Exists: {"path": "/f"},
        RpcCode.FileStatus: {"path": "/f"},
        RpcCode.ListStatus: {"path": "/"},
        RpcCode.Delete: {"path": "/f", "recursive": False},
        RpcCode.Rename: {"src": "/a", "dst": "/b"},
        RpcCode.Free: {"path": "/f", "recursive": True},
        RpcCode.ReadBlock: {"block_id": 9, "offset": 5, "length": 100,
                            "chunk_size": 4096},
        RpcCode.WriteBlock: {"block_id": 9, "reserve": 1 << 20,
                             "tier": "SSD"},
    }
    for code, hdr in cases.items():
        raw = proto.encode_request(int(code), hdr)
        back = proto.decode_request(int(code), raw)
        for k, v in hdr.items():
            assert back.get(k) == v, (code, k, back)


def test_status_response_roundtrip():
    st = {"inode_id": 42, "path": "/p/f", "name": "f", "file_type": 0,
          "length": 12345, "is_complete": True, "block_size": 1 << 26,
          "replicas": 3, "storage_tier": "SSD", "mtime_ms": 111,
          "atime_ms": 222, "mode": 0o640, "uid": 7, "gid": 8,
          "ttl_ms": 9000, "ttl_action": "delete", "symlink_target": "",
          "nlink": 2, "xattrs": {"user.k": b"v"}}
    raw = proto.encode_response(int(RpcCode.FileStatus), {"status": st})
    back = proto.decode_response(int(RpcCode.FileStatus), raw)["status"]
    for k in ("inode_id", "path", "name", "file_type", "length",
              "is_complete", "block_size", "replicas", "storage_tier",
              "mtime_ms", "atime_ms", "mode", "uid", "gid", "ttl_ms",
              "ttl_action", "nlink"):
        assert back[k] == st[k], (k, back[k], st[k])
    assert back["xattrs"] == {"user.k": b"v"}


def test_error_wire_golden():
    """ErrorEncoder layout (error_encoder.rs:24-50): i32 kind BE,
    u32 len BE, msg, u32 data_len BE — FileNotFound kind is 8
    (fs_error.rs:47)."""
    from curvine_amd import errors as cverr
    raw = proto.encode_error(cverr.FileNotFound("/x"))
    assert raw == bytes.fromhex("00000008" "00000002" "2f78" "00000000"), \
        raw.hex()
    back = proto.decode_error(raw)
    assert isinstance(back, cverr.FileNotFound) and str(back) == "/x"
    raw2 = proto.encode_error(RuntimeError("boom"))
    assert raw2[:4] == (10000).to_bytes(4, "big")


def test_hbm_tier_maps_to_mem_on_wire():
    st = {"inode_id": 1, "path": "/f", "name": "f", "file_type": 0,
          "length": 0, "is_complete": True, "block_size": 1, "replicas": 1,
          "storage_tier": "HBM", "mtime_ms": 0, "atime_ms": 0, "mode": 0,
          "uid": 0, "gid": 0, "ttl_ms": 0, "ttl_action": "none",
          "symlink_target": "", "nlink": 1, "xattrs": {}}
    raw = proto.encode_response(int(RpcCode.FileStatus), {"status": st})
    back = proto.decode_response(int(RpcCode.FileStatus), raw)["status"]
    assert back["storage_tier"] == "MEM"


# ---------------------------------------------------------------- live

class _PbClient:
    """Minimal reference-client stand-in: 22-byte frames with protobuf
    headers over a plain socket."""

    def __init__(self, reader, writer):
        self.r, self.w = reader, writer
        self.req_id = 1000

    async def call(self, code: RpcCode, header: dict) -> dict:
        raw = proto.encode_request(int(code), header)
        self.req_id += 1
        frame = _HDR.pack(len(raw), 0, int(code), 0, self.req_id, 0) + raw
        self.w.write(frame)
        await self.w.drain()
        hdr = await self.r.readexactly(22)
        hlen, dlen, rcode, status, req_id, seq = _HDR.unpack(hdr)
        body = await self.r.readexactly(hlen + dlen)
        assert req_id == self.req_id
        if (status >> 4) == 5:
            # reference error wire: DATA section, ErrorEncoder layout
            assert hlen == 0, "error reply must not carry a header"
            raise proto.decode_error(body)
        out = proto.decode_response(int(code), body[:hlen])
        assert out is not None, "reply did not parse as protobuf"
        return out


def test_protobuf_client_against_live_master(tmp_path):
    """A protobuf-speaking peer drives mkdir/create/complete/stat/
    rename/exists/delete against a real master (native meta frontend
    forwards the frames; replies come back protobuf-encoded), while a
    msgpack client on a separate connection is unaffected."""
    from curvine_amd.testing import SyncMiniCluster

    smc = SyncMiniCluster(tmp_dir=str(tmp_path / "cv")).start()
    try:
        port = smc.master.rpc.port

        async def run():
            r, w = await asyncio.open_connection("127.0.0.1", port)
            c = _PbClient(r, w)
            st = (await c.call(RpcCode.Mkdir,
                               {"path": "/pb", "create_parents": True,
                                "mode": 0o755}))["status"]
            assert st["file_type"] == 1 and st["path"] == "/pb"
            st = (await c.call(RpcCode.CreateFile,
                               {"path": "/pb/f", "overwrite": True,
                                "mode": 0o644}))["status"]
            assert st["path"] == "/pb/f" and not st["is_complete"]
            await c.call(RpcCode.CompleteFile, {"path": "/pb/f",
                                                "length": 0})
            st = (await c.call(RpcCode.FileStatus,
                               {"path": "/pb/f"}))["status"]
            assert st["is_complete"] and st["length"] == 0
            assert (await c.call(RpcCode.Exists,
                                 {"path": "/pb/f"}))["exists"] is True
            await c.call(RpcCode.Rename, {"src": "/pb/f", "dst": "/pb/g"})
            assert (await c.call(RpcCode.Exists,
                                 {"path": "/pb/g"}))["exists"] is True
            ls = await c.call(RpcCode.ListStatus, {"path": "/pb"})
            assert [s["name"] for s in ls["statuses"]] == ["g"]
            await c.call(RpcCode.Delete, {"path": "/pb/g",
                                          "recursive": False})
            assert (await c.call(RpcCode.Exists,
                                 {"path": "/pb/g"}))["exists"] is False
            # typed errors cross the wire in the reference binary layout
            from curvine_amd import errors as cverr
            try:
                await c.call(RpcCode.FileStatus, {"path": "/pb/missing"})
                raise AssertionError("expected FileNotFound")
            except cverr.FileNotFound as e:
                assert "missing" in str(e)
            w.close()
            await w.wait_closed()

        smc.call(run())
        # msgpack path still healthy on the same master
        from curvine_amd.client.filesystem import SyncFs
        sf = SyncFs(smc.client_conf())
        sf.mkdir("/after-pb")
        assert sf.exists("/after-pb")
        sf.shutdown()
    finally:
        smc.stop()


def test_protobuf_block_read_stream(tmp_path):
    """A protobuf peer streams a block read from the worker: protobuf
    BlockReadRequest open, protobuf BlockReadResponse ack, raw data
    frames, Complete."""
    import asyncio
    import os

    from curvine_amd.testing import MiniCluster

    async def main():
        async with MiniCluster(tmp_dir=str(tmp_path / "cv")) as mc:
            fs = mc.fs()
            data = os.urandom(2 << 20)
            await fs.write_all("/pbw/f.bin", data)
            info = await fs.client.open("/pbw/f.bin")
            lb = info.blocks[0]
            r, w = await asyncio.open_connection(
                "127.0.0.1", lb.locations[0].rpc_port)
            raw = proto.encode_request(int(RpcCode.ReadBlock), {
                "block_id": lb.block.block_id, "offset": 0,
                "length": lb.block.length, "chunk_size": 1 << 20})
            w.write(_HDR.pack(len(raw), 0, int(RpcCode.ReadBlock),
                              1, 4242, 0) + raw)
            await w.drain()
            got = b""
            acked = False
            while True:
                hdr = await r.readexactly(22)
                hlen, dlen, code, status, req_id, seq = _HDR.unpack(hdr)
                body = await r.readexactly(hlen + dlen)
                rs = status >> 4
                assert rs != 5, body
                if hlen and not acked:
                    out = proto.decode_response(int(RpcCode.ReadBlock),
                                                body[:hlen])
                    assert out and out["length"] == lb.block.length
                    acked = True
                got += body[hlen:]
                if rs == 3:
                    break
            assert acked, "no protobuf open-ack"
            assert got == data[:lb.block.length]
            w.close()
            await w.wait_closed()
            await fs.close()

    asyncio.new_event_loop().run_until_complete(main())


def test_fsinfo_handshake_pbuf_roundtrip():
    """GetFilesystemInfo over the protobuf wire: a reference-shaped
    client request with component_info=1000 decodes, and the response
    carries workers + the master's compatibility contract."""
    from curvine_amd.compat import component_info
    from curvine_amd.rpc import proto
    from curvine_amd.rpc.codes import RpcCode

    req = proto.M["GetFilesystemInfoRequest"]()
    proto._ci_fill(req.component_info, component_info("client"))
    d = proto.decode_request(int(RpcCode.GetFilesystemInfo),
                             req.SerializeToString())
    assert d["component_info"]["component"] == "client"
    assert d["component_info"]["protocol_version"] == 1
    # legacy client (no component_info): empty dict, never None
    assert proto.decode_request(
        int(RpcCode.GetFilesystemInfo),
        proto.M["GetFilesystemInfoRequest"]().SerializeToString()) == {}

    hdr = {"cluster_id": "cv", "inode_num": 5, "block_num": 2,
           "capacity": 100, "used": 10,
           "component_info": component_info("master"),
           "live_workers": [{
               "address": {"worker_id": 7, "hostname": "h", "rpc_port": 1},
               "storages": [{"tier": "HBM", "dir_id": 0, "capacity": 50,
                             "used": 5, "block_num": 1}],
               "last_heartbeat_ms": 123,
               "component_info": component_info("worker")}]}
    enc = proto.encode_response(int(RpcCode.GetFilesystemInfo), hdr)
    m = proto.M["GetFilesystemInfoResponse"]()
    m.ParseFromString(enc)
    assert m.active_master == "cv"
    assert m.capacity == 100 and m.available == 90
    w = m.live_workers[0]
    assert w.address.worker_id == 7
    assert w.component_info.release_version
    assert w.storage_map["HBM-0"].capacity == 50
    # HBM rides the MEM storage type on the wire (documented deviation)
    assert m.compatibility.server.component == "master"
    assert m.compatibility.compatibility_mode == 1   # DIAGNOSE
