"""Web endpoints: overview json, browse, prometheus metrics."""
import asyncio
import json
import urllib.request

import pytest

from curvine_amd.testing import MiniCluster
from curvine_amd.web.server import WebServer


def test_web_endpoints(tmp_path):
    async def main():
        async with MiniCluster(tmp_dir=str(tmp_path)) as mc:
            fs = mc.fs()
            await fs.write_all("/w/file.bin", b"x" * 1000)
            web = await WebServer(mc.conf, master=mc.master,
                                  worker=mc.workers[0], port=0).start()
            loop = asyncio.get_event_loop()

            def get(p):
                with urllib.request.urlopen(
                        f"http://127.0.0.1:{web.port}{p}", timeout=10) as r:
                    return r.read()

            info = json.loads(await loop.run_in_executor(None, get, "/api/info"))
            assert info["inode_num"] >= 2
            listing = json.loads(await loop.run_in_executor(
                None, get, "/api/browse?path=/w"))
            assert listing[0]["name"] == "file.bin"
            metrics = (await loop.run_in_executor(None, get, "/metrics")).decode()
            assert "curvine_master_inode_num" in metrics
            assert "curvine_worker_capacity_bytes" in metrics
            page = (await loop.run_in_executor(None, get, "/")).decode()
            assert "curvine-amd" in page
            await web.stop()
            await fs.close()
    asyncio.new_event_loop().run_until_complete(main())


def test_client_metrics_push_and_api(tmp_path):
    """MetricsReport (code 60): a client pushes a metrics snapshot; the
    master stores it bounded and serves it on /api/client-metrics."""
    import asyncio
    import json
    import urllib.request

    from curvine_amd.client.fs_client import FsClient
    from curvine_amd.testing import MiniCluster, test_conf
    from curvine_amd.web.server import WebServer

    async def main():
        conf = test_conf(str(tmp_path))
        async with MiniCluster(conf=conf, tmp_dir=str(tmp_path)) as mc:
            web = await WebServer(conf, master=mc.master, port=0,
                                  host="127.0.0.1").start()
            conf.client.master_addrs = [f"127.0.0.1:{mc.master.rpc.port}"]
            cl = FsClient(conf)
            await cl.report_metrics({"reads": 7, "read_bytes": 1234},
                                    kind="fuse")
            url = f"http://127.0.0.1:{web.port}/api/client-metrics"
            loop = asyncio.get_running_loop()
            body = await loop.run_in_executor(
                None, lambda: urllib.request.urlopen(url, timeout=10).read())
            data = json.loads(body)
            assert len(data) == 1
            ent = next(iter(data.values()))
            assert ent["kind"] == "fuse"
            assert ent["metrics"]["read_bytes"] == 1234
            # bounded store: 300 distinct clients cap at 256
            from curvine_amd.rpc.codes import RpcCode
            for i in range(300):
                await cl._rpc(RpcCode.MetricsReport,
                              {"client_id": f"c{i}", "metrics": {"i": i}})
            assert len(mc.master.client_metrics) <= 256
            await cl.close()
            await web.stop()

    asyncio.new_event_loop().run_until_complete(main())
