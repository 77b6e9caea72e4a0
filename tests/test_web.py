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
