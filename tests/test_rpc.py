"""RPC framing, unary + streaming round trips, error propagation,
master-failover connector."""
import asyncio

import pytest

from curvine_amd.errors import FileNotFound, FsError, NotLeader
from curvine_amd.rpc import (ClientFactory, ClusterConnector, HandlerService,
                             Message, RpcClient, RpcServer, RpcCode, Status)


def test_frame_roundtrip():
    msg = Message.request(RpcCode.Mkdir, {"path": "/a/b", "mode": 0o755}, b"payload")
    raw = msg.encode()
    hlen, dlen, decoded = Message.decode_proto(raw[:22])
    decoded.set_header_bytes(raw[22:22 + hlen])
    decoded.data = raw[22 + hlen:22 + hlen + dlen]
    assert decoded.code == int(RpcCode.Mkdir)
    assert decoded.header == {"path": "/a/b", "mode": 0o755}
    assert decoded.data == b"payload"
    assert decoded.req_id == msg.req_id


def test_status_packing():
    msg = Message.request(RpcCode.ReadBlock, req_status=Status.Open)
    reply = msg.reply(resp_status=Status.Running)
    raw = reply.encode()
    _, _, decoded = Message.decode_proto(raw[:22])
    assert decoded.req_status == Status.Open
    assert decoded.resp_status == Status.Running


class EchoHandler:
    async def handle(self, msg, conn):
        if msg.code == int(RpcCode.Mkdir):
            return msg.reply({"echo": msg.header}, msg.data)
        if msg.code == int(RpcCode.OpenFile):
            raise FileNotFound(f"no {msg.header.get('path')}")
        if msg.code == int(RpcCode.ReadBlock):
            # stream 3 chunks then complete
            for i in range(3):
                await conn.send(msg.reply({"i": i}, b"x" * 10, Status.Running))
            return msg.reply(resp_status=Status.Complete)
        return msg.reply()


class EchoService(HandlerService):
    def get_message_handler(self):
        return EchoHandler()


@pytest.fixture
def loop():
    loop = asyncio.new_event_loop()
    yield loop
    loop.close()


def run(loop, coro):
    return loop.run_until_complete(coro)


def test_unary_rpc(loop):
    async def main():
        server = RpcServer("test", "127.0.0.1", 0, EchoService())
        await server.start()
        client = await RpcClient("127.0.0.1", server.port).connect()
        reply = await client.rpc(RpcCode.Mkdir, {"path": "/x"}, b"data!")
        assert reply.header["echo"] == {"path": "/x"}
        assert reply.data == b"data!"
        await client.close()
        await server.stop()
    run(loop, main())


def test_error_propagation(loop):
    async def main():
        server = RpcServer("test", "127.0.0.1", 0, EchoService())
        await server.start()
        client = await RpcClient("127.0.0.1", server.port).connect()
        with pytest.raises(FileNotFound) as ei:
            await client.rpc(RpcCode.OpenFile, {"path": "/gone"})
        assert "/gone" in str(ei.value)
        await client.close()
        await server.stop()
    run(loop, main())


def test_streaming(loop):
    async def main():
        server = RpcServer("test", "127.0.0.1", 0, EchoService())
        await server.start()
        client = await RpcClient("127.0.0.1", server.port).connect()
        stream = client.stream(RpcCode.ReadBlock)
        await stream.send({"off": 0}, status=Status.Open)
        chunks = []
        while True:
            m = await stream.recv()
            if m.resp_status == Status.Complete:
                break
            chunks.append(m.data)
        assert chunks == [b"x" * 10] * 3
        stream.close()
        await client.close()
        await server.stop()
    run(loop, main())


def test_concurrent_rpcs(loop):
    async def main():
        server = RpcServer("test", "127.0.0.1", 0, EchoService())
        await server.start()
        client = await RpcClient("127.0.0.1", server.port).connect()
        replies = await asyncio.gather(*[
            client.rpc(RpcCode.Mkdir, {"path": f"/{i}"}) for i in range(50)])
        for i, r in enumerate(replies):
            assert r.header["echo"]["path"] == f"/{i}"
        await client.close()
        await server.stop()
    run(loop, main())


class FailoverHandler:
    def __init__(self, is_leader, leader_addr):
        self.is_leader = is_leader
        self.leader_addr = leader_addr

    async def handle(self, msg, conn):
        if not self.is_leader():
            raise NotLeader(f"leader={self.leader_addr()}")
        return msg.reply({"ok": True})


class FailoverService(HandlerService):
    def __init__(self, is_leader, leader_addr):
        self.is_leader = is_leader
        self.leader_addr = leader_addr

    def get_message_handler(self):
        return FailoverHandler(self.is_leader, self.leader_addr)


def test_cluster_connector_failover(loop):
    async def main():
        leader_idx = [1]
        servers = []
        for i in range(2):
            s = RpcServer(f"m{i}", "127.0.0.1", 0,
                          FailoverService(lambda i=i: leader_idx[0] == i,
                                          lambda: f"127.0.0.1:{servers[leader_idx[0]].port}"))
            await s.start()
            servers.append(s)
        conn = ClusterConnector([f"127.0.0.1:{s.port}" for s in servers],
                                timeout_ms=5000)
        reply = await conn.rpc(RpcCode.FileStatus, {})
        assert reply.header["ok"]
        # fail over
        leader_idx[0] = 0
        reply = await conn.rpc(RpcCode.FileStatus, {})
        assert reply.header["ok"]
        await conn.close()
        for s in servers:
            await s.stop()
    run(loop, main())
