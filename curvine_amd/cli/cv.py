"""cv — the curvine_amd command-line tool.

Analog of the reference's `cv` binary
(/root/reference/curvine-cli/src/: fs ops ls/cat/put/get/mkdir/rm/mv/stat/
du/df/count/touch/chmod/blocks/free, load/load-status/load-cancel,
mount/umount, node list/decommission, report commands.rs:19).

Usage: python -m curvine_amd.cli <cmd> [args]   (or the `cv` wrapper)
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

from curvine_amd.conf import ClusterConf, fmt_bytes
from curvine_amd.client.filesystem import SyncFs
from curvine_amd.unified import UnifiedFileSystem


def make_fs(args) -> SyncFs:
    conf = ClusterConf.from_file(args.conf) if args.conf else ClusterConf()
    if args.master:
        conf.client.master_addrs = [args.master]
    sf = SyncFs.__new__(SyncFs)
    import asyncio
    import threading
    sf._own_loop = True
    sf.loop = asyncio.new_event_loop()
    sf._thread = threading.Thread(target=sf.loop.run_forever, daemon=True)
    sf._thread.start()

    async def mk():
        return UnifiedFileSystem(conf)
    sf.fs = sf.call(mk())
    return sf


def fmt_status(s, human=True) -> str:
    kind = "d" if s.is_dir else ("l" if s.is_symlink else "-")
    size = fmt_bytes(s.length) if human else str(s.length)
    mtime = time.strftime("%Y-%m-%d %H:%M", time.localtime(s.mtime_ms / 1000))
    return f"{kind}{s.mode & 0o7777:04o} {s.replicas:2d} {size:>12} {mtime} {s.path}"


def cmd_ls(fs, args):
    for s in fs.list_status(args.path):
        print(fmt_status(s))


def cmd_mkdir(fs, args):
    fs.mkdir(args.path, create_parents=args.parents)


def cmd_rm(fs, args):
    n = fs.delete(args.path, recursive=args.recursive)
    print(f"deleted ({n} blocks)")


def cmd_mv(fs, args):
    fs.rename(args.src, args.dst)


def cmd_touch(fs, args):
    if not fs.exists(args.path):
        fs.call(fs.fs.client.create(args.path))
        fs.call(fs.fs.client.complete_file(args.path, 0, []))
    else:
        fs.set_attr(args.path, mtime_ms=int(time.time() * 1000))


def cmd_cat(fs, args):
    data = fs.read_file(args.path)
    sys.stdout.buffer.write(data)


def cmd_put(fs, args):
    async def run():
        w = await fs.fs.create(args.dst, overwrite=args.force)
        with open(args.src, "rb") as f:
            while True:
                chunk = f.read(4 << 20)
                if not chunk:
                    break
                await w.write(chunk)
        return await w.complete()
    st = fs.call(run())
    print(f"put {args.src} -> {args.dst} ({fmt_bytes(st.length)})")


def cmd_get(fs, args):
    async def run():
        r = await fs.fs.open(args.src)
        with open(args.dst, "wb") as f:
            pos = 0
            while pos < r.length:
                chunk = await r.pread(pos, 4 << 20)
                if not chunk:
                    break
                f.write(chunk)
                pos += len(chunk)
        r.close()
        return pos
    n = fs.call(run())
    print(f"get {args.src} -> {args.dst} ({fmt_bytes(n)})")


def cmd_stat(fs, args):
    s = fs.file_status(args.path)
    print(json.dumps(s.to_dict(), indent=1, default=str))


def cmd_blocks(fs, args):
    fb = fs.call(fs.fs.client.get_block_locations(args.path))
    for b in fb.blocks:
        locs = ", ".join(f"{a.hostname}:{a.rpc_port}(w{a.worker_id},{t})"
                         for a, t in zip(b.locations, b.tiers))
        print(f"block {b.block.block_id} off={b.offset} "
              f"len={b.block.length} [{locs}]")


def cmd_du(fs, args):
    total, files = 0, 0

    def walk(path):
        nonlocal total, files
        for s in fs.list_status(path):
            if s.is_dir:
                walk(s.path)
            else:
                total += s.length
                files += 1
    st = fs.file_status(args.path)
    if st.is_dir:
        walk(args.path)
    else:
        total, files = st.length, 1
    print(f"{fmt_bytes(total)}\t{files} files\t{args.path}")


def cmd_count(fs, args):
    dirs, files, size = 0, 0, 0

    def walk(path):
        nonlocal dirs, files, size
        for s in fs.list_status(path):
            if s.is_dir:
                dirs += 1
                walk(s.path)
            else:
                files += 1
                size += s.length
    walk(args.path)
    print(f"{dirs:>8} {files:>8} {fmt_bytes(size):>14} {args.path}")


def cmd_df(fs, args):
    info = fs.get_master_info()
    cap, used = info["capacity"], info["used"]
    print(f"capacity: {fmt_bytes(cap)}  used: {fmt_bytes(used)} "
          f"({used / max(1, cap) * 100:.1f}%)  "
          f"inodes: {info['inode_num']}  blocks: {info['block_num']}")


def cmd_chmod(fs, args):
    fs.set_attr(args.path, mode=int(args.mode, 8))


def cmd_chown(fs, args):
    uid, _, gid = args.owner.partition(":")
    kw = {}
    if uid:
        kw["uid"] = int(uid)
    if gid:
        kw["gid"] = int(gid)
    fs.set_attr(args.path, **kw)


def cmd_free(fs, args):
    n = fs.free(args.path, recursive=args.recursive)
    print(f"freed {n} cached blocks (metadata kept)")


def cmd_ttl(fs, args):
    fs.set_attr(args.path, ttl_ms=parse_duration_ms(args.ttl),
                ttl_action=args.action)


def parse_duration_ms(s: str) -> int:
    units = {"ms": 1, "s": 1000, "m": 60_000, "h": 3_600_000, "d": 86_400_000}
    for u, mult in sorted(units.items(), key=lambda x: -len(x[0])):
        if s.endswith(u):
            return int(float(s[:-len(u)]) * mult)
    return int(s)


def cmd_report(fs, args):
    info = fs.get_master_info()
    print(json.dumps(info, indent=1, default=str))


def cmd_node_list(fs, args):
    info = fs.get_master_info()
    for w in info["live_workers"]:
        a = w["address"]
        cap = sum(s["capacity"] for s in w["storages"])
        used = sum(s["used"] for s in w["storages"])
        tiers = ",".join(f"{s['tier']}:{fmt_bytes(s['capacity'])}"
                         for s in w["storages"])
        print(f"w{a['worker_id']} {a['hostname']}:{a['rpc_port']} "
              f"dev={a['device_id']} used={fmt_bytes(used)}/{fmt_bytes(cap)} "
              f"[{tiers}]")


def cmd_node_decommission(fs, args):
    from curvine_amd.rpc.codes import RpcCode

    async def run():
        r = await fs.fs.client.connector.rpc(
            RpcCode.DecommissionWorker, {"worker_id": args.worker_id})
        return r.header
    print(json.dumps(fs.call(run())))


def cmd_transfer_leader(fs, args):
    from curvine_amd.rpc.codes import RpcCode

    async def run():
        r = await fs.fs.client.connector.rpc(
            RpcCode.RaftTransferLeader, {"target": args.node_id})
        return r.header
    print(json.dumps(fs.call(run())))


def cmd_load(fs, args):
    job = fs.submit_job(args.path, recursive=True, replicas=args.replicas)
    print(json.dumps(job))
    if args.wait:
        while True:
            st = fs.job_status(job["job_id"])
            print(f"\r{st['state']}: {st['done']}/{st['total']} "
                  f"(failed {st['failed']})", end="", flush=True)
            if st["state"] not in ("planning", "running"):
                print()
                break
            time.sleep(0.5)


def cmd_load_status(fs, args):
    print(json.dumps(fs.job_status(args.job_id), indent=1))


def cmd_load_cancel(fs, args):
    print(json.dumps(fs.call(fs.fs.client.cancel_job(args.job_id))))


def cmd_load_retry(fs, args):
    """Re-dispatch failed tasks of a load/transfer job (RetryTransfer)."""
    from curvine_amd.rpc.codes import RpcCode

    async def go():
        r = await fs.fs.client.connector.rpc(RpcCode.RetryTransfer,
                                             {"job_id": args.job_id})
        return r.header
    print(json.dumps(fs.call(go())))


def cmd_transfers(fs, args):
    """List transfer/load jobs (ListTransfers)."""
    from curvine_amd.rpc.codes import RpcCode

    async def go():
        r = await fs.fs.client.connector.rpc(RpcCode.ListTransfers, {})
        return r.header
    h = fs.call(go())
    for j in h.get("transfers") or h.get("jobs") or []:
        print(json.dumps(j))


def cmd_mount(fs, args):
    props = dict(kv.split("=", 1) for kv in (args.option or []))
    mi = fs.mount(args.curvine_path, args.ufs_path, props,
                  cache_mode=args.cache_mode,
                  auto_cache=not args.no_auto_cache)
    print(json.dumps(mi.to_dict(), indent=1))


def cmd_umount(fs, args):
    fs.unmount(args.curvine_path)


def cmd_mount_table(fs, args):
    for mi in fs.get_mount_table():
        print(f"{mi.curvine_path} -> {mi.ufs_path} "
              f"(mode={mi.cache_mode}, auto_cache={mi.auto_cache})")


def cmd_bench(fs, args):
    """Quick metadata + throughput microbench (curvine-bench suites analog)."""
    base = "/cv-bench"
    n = args.num
    t0 = time.perf_counter()
    for i in range(n):
        fs.call(fs.fs.client.create(f"{base}/f{i}", overwrite=True))
        fs.call(fs.fs.client.complete_file(f"{base}/f{i}", 0, []))
    create_qps = n / (time.perf_counter() - t0)
    t0 = time.perf_counter()
    for i in range(n):
        fs.file_status(f"{base}/f{i}")
    stat_qps = n / (time.perf_counter() - t0)
    t0 = time.perf_counter()
    for i in range(n):
        fs.rename(f"{base}/f{i}", f"{base}/g{i}")
    rename_qps = n / (time.perf_counter() - t0)
    t0 = time.perf_counter()
    for i in range(n):
        fs.delete(f"{base}/g{i}")
    delete_qps = n / (time.perf_counter() - t0)
    data = os.urandom(args.size)
    t0 = time.perf_counter()
    fs.write_file(f"{base}/data.bin", data, overwrite=True)
    wbps = len(data) / (time.perf_counter() - t0)
    t0 = time.perf_counter()
    back = fs.read_file(f"{base}/data.bin")
    rbps = len(back) / (time.perf_counter() - t0)
    assert back == data
    fs.delete(base, recursive=True)
    print(json.dumps({
        "create_qps": round(create_qps, 1), "stat_qps": round(stat_qps, 1),
        "rename_qps": round(rename_qps, 1), "delete_qps": round(delete_qps, 1),
        "write_MBps": round(wbps / 1e6, 1), "read_MBps": round(rbps / 1e6, 1)}))


def cmd_validate(_fs, args):
    """Offline config validation (cli/validate_run.rs analog): parse the
    TOML, check tier specs / ports / peer addresses, print a summary."""
    from curvine_amd.conf import ClusterConf, TIERS
    path = args.conf_path or args.conf
    conf = ClusterConf.from_file(path) if path else ClusterConf()
    problems = []
    try:
        dirs = conf.worker.parsed_dirs()
    except Exception as e:  # noqa: BLE001
        problems.append(f"data_dirs: {e}")
        dirs = []
    for dd in dirs:
        if dd.tier not in TIERS:
            problems.append(f"unknown tier {dd.tier!r}")
        if dd.tier == "HBM" and dd.device_id < 0:
            problems.append(f"HBM dir {dd.path!r} needs a device ordinal")
    for port_name in ("rpc_port", "web_port"):
        v = getattr(conf.master, port_name)
        if not (0 <= v < 65536):
            problems.append(f"master.{port_name}={v} out of range")
    for spec in conf.journal.peers:
        pid, sep, addr = spec.partition("@")
        if not sep or not pid.isdigit() or ":" not in addr:
            problems.append(f"journal peer {spec!r}: want 'id@host:port'")
    if conf.master.block_size <= 0:
        problems.append("master.block_size must be positive")
    if len(conf.journal.peers) not in (0, 1) and \
            len(conf.journal.peers) % 2 == 0:
        problems.append("raft peer count should be odd")
    summary = {
        "conf": path or "(defaults)",
        "master": f"{conf.master.hostname}:{conf.master.rpc_port}",
        "raft_peers": len(conf.journal.peers),
        "tiers": [f"{d.tier}:{d.capacity >> 30}GB" for d in dirs],
        "block_size_mb": conf.master.block_size >> 20,
        "fuse_mnt": conf.fuse.mnt_path,
        "native_meta": conf.master.native_meta,
        "problems": problems,
    }
    print(json.dumps(summary, indent=2))
    if problems:
        raise SystemExit(1)


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(prog="cv", description=__doc__)
    p.add_argument("--conf", default=os.environ.get("CURVINE_CONF"))
    p.add_argument("--master", default=os.environ.get("CURVINE_MASTER"))
    sub = p.add_subparsers(dest="cmd", required=True)

    def add(name, fn, *specs, **kw):
        sp = sub.add_parser(name, **kw)
        for spec in specs:
            flags, skw = spec
            sp.add_argument(*flags, **skw)
        sp.set_defaults(fn=fn)
        return sp

    A = lambda *f, **kw: (f, kw)  # noqa: E731
    add("ls", cmd_ls, A("path"))
    add("mkdir", cmd_mkdir, A("path"), A("-p", "--parents", action="store_true"))
    add("rm", cmd_rm, A("path"), A("-r", "--recursive", action="store_true"))
    add("mv", cmd_mv, A("src"), A("dst"))
    add("touch", cmd_touch, A("path"))
    add("cat", cmd_cat, A("path"))
    add("put", cmd_put, A("src"), A("dst"), A("-f", "--force", action="store_true"))
    add("get", cmd_get, A("src"), A("dst"))
    add("stat", cmd_stat, A("path"))
    add("blocks", cmd_blocks, A("path"))
    add("du", cmd_du, A("path"))
    add("count", cmd_count, A("path"))
    add("df", cmd_df)
    add("chmod", cmd_chmod, A("mode"), A("path"))
    add("chown", cmd_chown, A("owner"), A("path"))
    add("free", cmd_free, A("path"), A("-r", "--recursive", action="store_true"))
    add("ttl", cmd_ttl, A("path"), A("ttl"), A("--action", default="delete",
                                               choices=["delete", "free"]))
    add("report", cmd_report)
    add("load", cmd_load, A("path"), A("--replicas", type=int, default=1),
        A("--wait", action="store_true"))
    add("load-status", cmd_load_status, A("job_id"))
    add("load-cancel", cmd_load_cancel, A("job_id"))
    add("load-retry", cmd_load_retry, A("job_id"))
    add("transfers", cmd_transfers)
    add("mount", cmd_mount, A("ufs_path"), A("curvine_path"),
        A("-o", "--option", action="append"),
        A("--cache-mode", default="cache", choices=["cache", "fs"]),
        A("--no-auto-cache", action="store_true"))
    add("umount", cmd_umount, A("curvine_path"))
    add("mount-table", cmd_mount_table)
    add("bench", cmd_bench, A("--num", type=int, default=200),
        A("--size", type=int, default=16 << 20))
    vp = sub.add_parser("validate", help="validate a cluster TOML offline")
    vp.add_argument("conf_path", nargs="?", default=None)
    vp.set_defaults(fn=cmd_validate, offline=True)
    node = sub.add_parser("node")
    nodesub = node.add_subparsers(dest="nodecmd", required=True)
    nl = nodesub.add_parser("list")
    nl.set_defaults(fn=cmd_node_list)
    nd = nodesub.add_parser("decommission")
    nd.add_argument("worker_id", type=int)
    nd.set_defaults(fn=cmd_node_decommission)
    nt = nodesub.add_parser("transfer-leader")
    nt.add_argument("node_id", type=int)
    nt.set_defaults(fn=cmd_transfer_leader)
    return p


def main(argv=None) -> int:
    args = build_parser().parse_args(argv)
    if getattr(args, "offline", False):
        try:
            args.fn(None, args)
            return 0
        except SystemExit as e:
            return int(e.code or 0)
        except Exception as e:  # noqa: BLE001
            print(f"cv: {e}", file=sys.stderr)
            return 1
    fs = make_fs(args)
    try:
        args.fn(fs, args)
        return 0
    except Exception as e:  # noqa: BLE001
        print(f"cv: {e}", file=sys.stderr)
        return 1
    finally:
        fs.shutdown()


if __name__ == "__main__":
    sys.exit(main())
