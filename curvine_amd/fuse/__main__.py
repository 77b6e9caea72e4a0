"""cv-fuse: standalone FUSE daemon process.

Usage:
    python -m curvine_amd.fuse --mnt /mnt/curvine --master 127.0.0.1:8995 \
        [--conf cluster.toml] [--embed-worker] [--device N] [--channels K]

Runs the FUSE daemon in its own process — the production deployment shape
(and the only deadlock-free one: an in-process daemon starves its own
mount's FLUSH requests whenever the embedding app spawns subprocesses,
because fork/vfork suspends the parent with the GIL held while the child's
pre-exec fd-closing sends FUSE requests).

Prints "READY <mnt>" on stdout once mounted; exits cleanly on SIGTERM.
"""
from __future__ import annotations

import argparse
import logging
import signal
import sys
import threading

from curvine_amd.conf import ClusterConf
from curvine_amd.fuse.daemon import FuseDaemon


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="cv-fuse")
    p.add_argument("--mnt", required=True)
    p.add_argument("--master", default=None, help="host:port")
    p.add_argument("--conf", default=None, help="cluster TOML")
    p.add_argument("--embed-worker", action="store_true",
                   help="run a worker (HBM arena owner) inside this daemon")
    p.add_argument("--device", type=int, default=-1, help="GPU ordinal")
    p.add_argument("--channels", type=int, default=0)
    p.add_argument("--data-dir", action="append", default=[],
                   help='worker data dir, e.g. "[HBM:200GB:0]gpu0"')
    p.add_argument("--web-port", type=int, default=0,
                   help="serve /metrics and /api/fuse on this port")
    p.add_argument("-o", "--opt", action="append", default=[],
                   help="mount-style options (cli/mount_args.rs analog): "
                        "-o master=host:port,channels=8,web_port=9100,"
                        "embed_worker,device=0, or dotted conf overlays "
                        "like fuse.max_write=1048576")
    p.add_argument("--takeover", action="store_true",
                   help="hot upgrade: adopt the running daemon's session "
                        "fd + open-handle state instead of mounting")
    p.add_argument("--log-level", default="INFO")
    args = p.parse_args(argv)

    logging.basicConfig(
        level=getattr(logging, args.log_level.upper(), logging.INFO),
        format="%(asctime)s %(name)s %(levelname)s %(message)s")

    # -o options fold into the flag namespace / conf overlay
    overlays = {}
    for group in args.opt:
        for kv in group.split(","):
            if not kv:
                continue
            k, _, v = kv.partition("=")
            if k == "master":
                args.master = args.master or v
            elif k == "channels":
                args.channels = args.channels or int(v)
            elif k == "web_port":
                args.web_port = args.web_port or int(v)
            elif k == "embed_worker":
                args.embed_worker = True
            elif k == "device":
                args.device = int(v)
            elif k == "data_dir":
                args.data_dir.append(v)
            elif "." in k:
                overlays[k] = v
            else:
                print(f"cv-fuse: unknown -o option {k!r}", file=sys.stderr)
                return 2

    conf = ClusterConf.from_file(args.conf) if args.conf else ClusterConf()
    if overlays:
        conf.overlay(**overlays)
    if args.master:
        conf.client.master_addrs = [args.master]
        host, _, port = args.master.rpartition(":")
        conf.master.hostname, conf.master.rpc_port = host, int(port)
    if args.channels:
        conf.fuse.mnt_number = args.channels
    if args.data_dir:
        conf.worker.data_dirs = args.data_dir
    conf.worker.rpc_port = 0
    conf.fuse.mnt_path = args.mnt
    import os as _os
    if _os.environ.get("CURVINE_FUSE_WRITEBACK"):
        conf.fuse.writeback_cache = \
            _os.environ["CURVINE_FUSE_WRITEBACK"] != "0"

    daemon = FuseDaemon(conf, args.mnt, embed_worker=args.embed_worker,
                        device_id=args.device).start(takeover=args.takeover)
    web = None
    if args.web_port:
        from curvine_amd.web.server import WebServer

        async def mkweb():
            return await WebServer(conf, worker=daemon.worker,
                                   fuse_session=daemon.session,
                                   port=args.web_port,
                                   host="127.0.0.1").start()
        web = daemon.call(mkweb())
    print(f"READY {args.mnt}", flush=True)

    def dump_stats(*_a):
        import json as _json
        import sys as _sys
        print(_json.dumps({"fuse_op_stats": daemon.session.stats()}),
              file=_sys.stderr, flush=True)

    stop = threading.Event()
    signal.signal(signal.SIGTERM, lambda *a: stop.set())
    signal.signal(signal.SIGINT, lambda *a: stop.set())
    signal.signal(signal.SIGUSR1, dump_stats)
    import faulthandler
    signal.signal(signal.SIGUSR2,
                  lambda *a: faulthandler.dump_traceback(file=sys.stderr))
    stop.wait()
    dump_stats()
    if web is not None:
        try:
            daemon.call(web.stop(), timeout=5)
        except Exception:
            pass
    daemon.stop()
    return 0


if __name__ == "__main__":
    sys.exit(main())
