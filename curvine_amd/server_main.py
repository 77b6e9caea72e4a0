"""curvine-server: master and/or worker process launcher.

Analog of the reference's curvine-server binary
(/root/reference/curvine-server/src/bin/curvine-server.rs +
curvine-server/src/lib.rs:15-30 gluing master + worker + web).

    python -m curvine_amd.server_main --service master --conf etc/cluster.toml
    python -m curvine_amd.server_main --service worker --device 0
    python -m curvine_amd.server_main --service all     # single-node
"""
from __future__ import annotations

import argparse
import asyncio
import logging
import signal
import sys

from curvine_amd.conf import ClusterConf


async def run(args) -> None:
    conf = ClusterConf.from_file(args.conf) if args.conf else ClusterConf()
    if args.master_port:
        conf.master.rpc_port = args.master_port
    servers = []
    if args.service in ("master", "all"):
        from curvine_amd.master.server import Master
        servers.append(await Master(conf).start())
    if args.service in ("worker", "all"):
        from curvine_amd.worker.server import Worker
        if args.data_dir:
            conf.worker.data_dirs = args.data_dir
        if args.worker_port >= 0:
            conf.worker.rpc_port = args.worker_port
        if args.heartbeat_ms > 0:
            conf.worker.heartbeat_interval_ms = args.heartbeat_ms
        servers.append(await Worker(conf, device_id=args.device).start())
    if args.service == "transfer":
        from curvine_amd.transfer import TransferService
        svc = await TransferService(conf).start()
        print(f"TRANSFER_PORT={svc.port}", flush=True)
        servers.append(svc)
    if args.web:
        from curvine_amd.master.server import Master
        from curvine_amd.web.server import WebServer
        for s in list(servers):
            if isinstance(s, Master):
                servers.append(await WebServer(conf, master=s).start())
            else:   # worker dashboard + /metrics on its own port
                servers.append(await WebServer(conf, worker=s).start())

    stop = asyncio.Event()
    loop = asyncio.get_event_loop()
    for sig in (signal.SIGTERM, signal.SIGINT):
        loop.add_signal_handler(sig, stop.set)
    await stop.wait()
    for s in reversed(servers):
        await s.stop()


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="curvine-server")
    p.add_argument("--service", choices=["master", "worker", "all", "transfer"],
                   default="all")
    p.add_argument("--conf", default=None)
    p.add_argument("--device", type=int, default=-1)
    p.add_argument("--master-port", type=int, default=0)
    p.add_argument("--worker-port", type=int, default=-1,
                   help="worker RPC port (0 = ephemeral)")
    p.add_argument("--heartbeat-ms", type=int, default=0)
    p.add_argument("--data-dir", action="append", default=[])
    p.add_argument("--web", action="store_true")
    p.add_argument("--log-level", default="INFO")
    args = p.parse_args(argv)
    logging.basicConfig(
        level=getattr(logging, args.log_level.upper(), logging.INFO),
        format="%(asctime)s %(name)s %(levelname)s %(message)s")
    asyncio.run(run(args))
    return 0


if __name__ == "__main__":
    sys.exit(main())
