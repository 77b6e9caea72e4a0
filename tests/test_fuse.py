"""FUSE end-to-end: real kernel mount against a MiniCluster.

Needs /dev/fuse + CAP_SYS_ADMIN (true in this container and on GPU boxes);
skipped otherwise.
"""
import asyncio
import errno
import os
import shutil
import subprocess
import threading

import pytest

requires_fuse = pytest.mark.skipif(
    not os.path.exists("/dev/fuse") or os.geteuid() != 0,
    reason="needs /dev/fuse and root")

pytestmark = requires_fuse


import sys
import time


@pytest.fixture
def mount(tmp_path):
    """MiniCluster (this process) + FUSE daemon (separate process, as in
    production — an in-process daemon deadlocks the moment the test spawns
    subprocesses, see curvine_amd/fuse/__main__.py)."""
    from curvine_amd.testing import SyncMiniCluster

    smc = SyncMiniCluster(tmp_dir=str(tmp_path / "cv")).start()
    # mountpoint OUTSIDE pytest's tmp tree: a later run's tmp cleanup must
    # never stat a dead fuse mount (uninterruptible hang)
    mnt = f"/tmp/curvine-fuse-test-{os.getpid()}"
    master = f"127.0.0.1:{smc.master.rpc.port}"
    dbg = os.environ.get("CURVINE_FUSE_DEBUG_LOG")
    proc = subprocess.Popen(
        [sys.executable, "-m", "curvine_amd.fuse", "--mnt", mnt,
         "--master", master, "--log-level",
         "DEBUG" if dbg else "WARNING"],
        stdout=subprocess.PIPE,
        stderr=open(dbg, "a") if dbg else subprocess.DEVNULL, text=True,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    line = proc.stdout.readline()
    assert line.startswith("READY"), f"fuse daemon failed: {line!r}"
    try:
        yield mnt, smc, proc
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
        smc.stop()
        try:
            os.rmdir(mnt)
        except OSError:
            pass


def test_mkdir_ls_write_read(mount):
    mnt, mc, daemon = mount
    os.makedirs(f"{mnt}/a/b")
    data = os.urandom(3 << 20)
    with open(f"{mnt}/a/b/file.bin", "wb") as f:
        f.write(data)
    assert sorted(os.listdir(f"{mnt}/a")) == ["b"]
    assert os.listdir(f"{mnt}/a/b") == ["file.bin"]
    st = os.stat(f"{mnt}/a/b/file.bin")
    assert st.st_size == len(data)
    with open(f"{mnt}/a/b/file.bin", "rb") as f:
        assert f.read() == data


def test_seek_and_partial_reads(mount):
    mnt, *_ = mount
    data = os.urandom(10 << 20)   # crosses 4MB block boundary
    with open(f"{mnt}/seek.bin", "wb") as f:
        f.write(data)
    with open(f"{mnt}/seek.bin", "rb") as f:
        f.seek(4 * 1024 * 1024 - 7)
        assert f.read(20) == data[4 * 1024 * 1024 - 7:4 * 1024 * 1024 + 13]
        f.seek(-100, os.SEEK_END)
        assert f.read() == data[-100:]


def test_rename_unlink(mount):
    mnt, *_ = mount
    with open(f"{mnt}/x.txt", "wb") as f:
        f.write(b"hello")
    os.makedirs(f"{mnt}/d")
    os.rename(f"{mnt}/x.txt", f"{mnt}/d/y.txt")
    assert not os.path.exists(f"{mnt}/x.txt")
    assert open(f"{mnt}/d/y.txt", "rb").read() == b"hello"
    os.unlink(f"{mnt}/d/y.txt")
    assert not os.path.exists(f"{mnt}/d/y.txt")
    os.rmdir(f"{mnt}/d")
    assert not os.path.exists(f"{mnt}/d")


def test_rmdir_nonempty_fails(mount):
    mnt, *_ = mount
    os.makedirs(f"{mnt}/ne")
    open(f"{mnt}/ne/f", "wb").close()
    with pytest.raises(OSError) as ei:
        os.rmdir(f"{mnt}/ne")
    assert ei.value.errno == errno.ENOTEMPTY


def test_symlink_readlink(mount):
    mnt, *_ = mount
    with open(f"{mnt}/target.txt", "wb") as f:
        f.write(b"data")
    os.symlink("target.txt", f"{mnt}/lnk")
    assert os.readlink(f"{mnt}/lnk") == "target.txt"
    assert open(f"{mnt}/lnk", "rb").read() == b"data"


def test_truncate_and_overwrite(mount):
    mnt, *_ = mount
    with open(f"{mnt}/t.bin", "wb") as f:
        f.write(b"A" * 1000)
    os.truncate(f"{mnt}/t.bin", 100)
    assert os.stat(f"{mnt}/t.bin").st_size == 100
    assert open(f"{mnt}/t.bin", "rb").read() == b"A" * 100
    # O_TRUNC overwrite
    with open(f"{mnt}/t.bin", "wb") as f:
        f.write(b"B" * 10)
    assert open(f"{mnt}/t.bin", "rb").read() == b"B" * 10


def test_xattr(mount):
    mnt, *_ = mount
    p = f"{mnt}/xa.txt"
    open(p, "wb").close()
    os.setxattr(p, "user.k1", b"v1")
    os.setxattr(p, "user.k2", b"v2")
    assert os.getxattr(p, "user.k1") == b"v1"
    assert sorted(os.listxattr(p)) == ["user.k1", "user.k2"]
    os.removexattr(p, "user.k1")
    assert os.listxattr(p) == ["user.k2"]
    with pytest.raises(OSError):
        os.getxattr(p, "user.k1")


def test_chmod_utime(mount):
    mnt, *_ = mount
    p = f"{mnt}/perm.txt"
    open(p, "wb").close()
    os.chmod(p, 0o600)
    assert (os.stat(p).st_mode & 0o7777) == 0o600
    os.utime(p, (1000000, 2000000))
    st = os.stat(p)
    assert int(st.st_mtime) == 2000000


def test_statfs(mount):
    mnt, *_ = mount
    st = os.statvfs(mnt)
    assert st.f_bsize == 4096
    assert st.f_blocks > 0


def test_append_reopen(mount):
    mnt, *_ = mount
    with open(f"{mnt}/app.txt", "wb") as f:
        f.write(b"part1-")
    with open(f"{mnt}/app.txt", "ab") as f:
        f.write(b"part2")
    assert open(f"{mnt}/app.txt", "rb").read() == b"part1-part2"


def test_concurrent_readers(mount):
    mnt, *_ = mount
    data = os.urandom(8 << 20)
    with open(f"{mnt}/conc.bin", "wb") as f:
        f.write(data)
    errors = []

    def reader():
        try:
            with open(f"{mnt}/conc.bin", "rb") as f:
                assert f.read() == data
        except Exception as e:  # noqa: BLE001
            errors.append(e)

    threads = [threading.Thread(target=reader) for _ in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errors


def test_posix_locks(mount):
    """POSIX locks are per-process: a second process must see the conflict."""
    import fcntl
    import sys
    mnt, *_ = mount
    p = f"{mnt}/lock.txt"
    open(p, "wb").close()
    f1 = open(p, "rb+")
    fcntl.lockf(f1, fcntl.LOCK_EX)
    prog = ("import fcntl,sys\n"
            f"f=open({p!r},'rb+')\n"
            "try:\n"
            "    fcntl.lockf(f, fcntl.LOCK_EX|fcntl.LOCK_NB)\n"
            "    sys.exit(1)\n"   # unexpectedly acquired
            "except OSError:\n"
            "    sys.exit(0)\n")
    r = subprocess.run([sys.executable, "-c", prog], timeout=30)
    assert r.returncode == 0, "conflicting lock was granted to second process"
    fcntl.lockf(f1, fcntl.LOCK_UN)
    r = subprocess.run([sys.executable, "-c",
                        f"import fcntl;f=open({p!r},'rb+');"
                        "fcntl.lockf(f, fcntl.LOCK_EX|fcntl.LOCK_NB)"],
                       timeout=30)
    assert r.returncode == 0, "lock not released"
    f1.close()


def test_shell_tools(mount):
    """cp/cat/dd through the mount."""
    mnt, *_ = mount
    src = f"{mnt}/shell_src.bin"
    data = os.urandom(1 << 20)
    with open(src, "wb") as f:
        f.write(data)
    subprocess.run(["cp", src, f"{mnt}/shell_cp.bin"], check=True)
    assert open(f"{mnt}/shell_cp.bin", "rb").read() == data
    out = subprocess.run(["dd", f"if={src}", "of=/dev/null", "bs=256K"],
                         capture_output=True, check=True)
    assert b"1048576 bytes" in out.stderr


def test_sparse_forward_seek_write(mount):
    """cp --sparse / seek-past-EOF writes: the hole reads back as zeros."""
    mnt, *_ = mount
    with open(f"{mnt}/sparse.bin", "wb") as f:
        f.write(b"head")
        f.seek(3 << 20)
        f.write(b"tail")
    data = open(f"{mnt}/sparse.bin", "rb").read()
    assert len(data) == (3 << 20) + 4
    assert data[:4] == b"head"
    assert data[-4:] == b"tail"
    assert data[4:3 << 20] == b"\x00" * ((3 << 20) - 4)


def test_web_metrics_endpoint(tmp_path):
    """cv-fuse --web-port serves /api/fuse stats and prometheus /metrics."""
    import json
    import socket
    import urllib.request

    from curvine_amd.testing import SyncMiniCluster

    smc = SyncMiniCluster(tmp_dir=str(tmp_path / "cv")).start()
    mnt = f"/tmp/curvine-fuse-web-{os.getpid()}"
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    proc = subprocess.Popen(
        [sys.executable, "-m", "curvine_amd.fuse", "--mnt", mnt,
         "--master", f"127.0.0.1:{smc.master.rpc.port}",
         "--web-port", str(port), "--log-level", "WARNING"],
        stdout=subprocess.PIPE, stderr=subprocess.DEVNULL, text=True,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    try:
        line = proc.stdout.readline()
        assert line.startswith("READY"), f"daemon failed: {line!r}"
        with open(f"{mnt}/web.bin", "wb") as f:
            f.write(b"x" * 65536)
        assert open(f"{mnt}/web.bin", "rb").read(16) == b"x" * 16
        stats = json.load(urllib.request.urlopen(
            f"http://127.0.0.1:{port}/api/fuse", timeout=10))
        assert any(isinstance(d, dict) and d.get("count", 0) > 0
                   for d in stats.values()), stats
        body = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/metrics", timeout=10).read().decode()
        assert "curvine" in body
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
        smc.stop()
        try:
            os.rmdir(mnt)
        except OSError:
            pass


def test_readdirplus_ls_l(mount):
    """`ls -l` via READDIRPLUS: names + sizes in one op (daemon advertises
    FUSE_DO_READDIRPLUS); entries must carry correct attrs."""
    mnt, *_ = mount
    os.makedirs(f"{mnt}/plus/sub")
    sizes = {}
    for i in range(30):
        with open(f"{mnt}/plus/f{i:02d}", "wb") as f:
            f.write(b"a" * (100 + i))
        sizes[f"f{i:02d}"] = 100 + i
    out = subprocess.run(["ls", "-l", f"{mnt}/plus"],
                         capture_output=True, text=True, check=True).stdout
    for name, sz in sizes.items():
        assert name in out
        assert f" {sz} " in out.split(name)[0].rsplit("\n", 1)[-1] + " ", \
            f"size {sz} missing for {name}"
    # entries remain stat-able (nlookup bookkeeping sane across forgets)
    with os.scandir(f"{mnt}/plus") as it:
        got = {e.name: e.stat().st_size for e in it if e.is_file()}
    assert got == sizes


def test_random_writes_same_handle(mount):
    """Backward writes within one open handle (linker pattern): rewrite
    bytes inside already-written data, spanning a block boundary."""
    mnt, *_ = mount
    p = f"{mnt}/rw.bin"
    base = bytearray(os.urandom(10 << 20))   # 3 blocks at 4 MiB
    with open(p, "wb") as f:
        f.write(base)
        # patch inside the first (committed) block
        f.seek(100)
        f.write(b"HEADER-PATCH")
        base[100:112] = b"HEADER-PATCH"
        # patch across the block 0/1 boundary
        bnd = 4 * 1024 * 1024 - 6
        f.seek(bnd)
        f.write(b"BOUNDARY-SPAN")
        base[bnd:bnd + 13] = b"BOUNDARY-SPAN"
        # patch in the still-open tail block
        f.seek(len(base) - 50)
        f.write(b"TAIL")
        base[len(base) - 50:len(base) - 46] = b"TAIL"
    assert os.path.getsize(p) == len(base)
    with open(p, "rb") as f:
        got = f.read()
    assert got == bytes(base)


def test_rewrite_overlap_extends_eof(mount):
    """A write overlapping the end and extending past it: rewrite the
    overlap in place, append the tail."""
    mnt, *_ = mount
    p = f"{mnt}/ext.bin"
    with open(p, "wb") as f:
        f.write(b"A" * 1000)
        f.seek(990)
        f.write(b"B" * 30)        # 10 overlap + 20 append
    data = open(p, "rb").read()
    assert len(data) == 1020
    assert data[:990] == b"A" * 990 and data[990:] == b"B" * 30


def test_inplace_rewrite_existing_file(mount):
    """r+b on a closed file: patch the middle without truncation
    (rsync --inplace pattern); length and surrounding bytes intact."""
    mnt, *_ = mount
    p = f"{mnt}/inplace.bin"
    base = bytearray(os.urandom(6 << 20))
    with open(p, "wb") as f:
        f.write(base)
    with open(p, "r+b") as f:
        f.seek(2 << 20)
        f.write(b"MIDDLE-REWRITE")
        base[2 << 20:(2 << 20) + 14] = b"MIDDLE-REWRITE"
        f.seek(0)
        f.write(b"FRONT")
        base[:5] = b"FRONT"
    assert os.path.getsize(p) == len(base)
    assert open(p, "rb").read() == bytes(base)


def test_mount_o_options(tmp_path):
    """cv-fuse -o option parsing (mount_args analog): master/channels and
    dotted conf overlays with type coercion."""
    from curvine_amd.testing import SyncMiniCluster

    smc = SyncMiniCluster(tmp_dir=str(tmp_path / "cv")).start()
    mnt = f"/tmp/curvine-fuse-oopt-{os.getpid()}"
    proc = subprocess.Popen(
        [sys.executable, "-m", "curvine_amd.fuse", "--mnt", mnt,
         "-o", f"master=127.0.0.1:{smc.master.rpc.port},channels=2",
         "-o", "fuse.max_write=524288,client.enable_crc=true",
         "--log-level", "WARNING"],
        stdout=subprocess.PIPE, stderr=subprocess.DEVNULL, text=True,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    try:
        line = proc.stdout.readline()
        assert line.startswith("READY"), f"daemon failed: {line!r}"
        with open(f"{mnt}/o.bin", "wb") as f:
            f.write(b"k" * 100000)
        assert open(f"{mnt}/o.bin", "rb").read() == b"k" * 100000
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
        smc.stop()
        try:
            os.rmdir(mnt)
        except OSError:
            pass
    # unknown option rejected
    from curvine_amd.fuse.__main__ import main as fmain
    assert fmain(["--mnt", "/tmp/x", "-o", "bogus=1"]) == 2
