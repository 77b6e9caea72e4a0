"""POSIX conformance corpus over the FUSE mount — the analog of the
reference's targeted repro scripts (build/tests/scripts/*.py: mmap,
renameat2 flags, O_TRUNC shared writer, read-past-EOF, hardlink
semantics...)."""
import ctypes
import errno
import mmap
import os
import stat
import subprocess
import sys

import pytest

requires_fuse = pytest.mark.skipif(
    not os.path.exists("/dev/fuse") or os.geteuid() != 0,
    reason="needs /dev/fuse and root")

pytestmark = requires_fuse

from tests.test_fuse import mount  # noqa: E402,F401 — reuse the fixture


def test_read_past_eof_small(mount):
    """fuse_read_past_eof_test.rs analog (small file; the 10 KB variant
    below covers clamped tail reads)."""
    mnt, *_ = mount
    p = f"{mnt}/eof.bin"
    with open(p, "wb") as f:
        f.write(b"12345")
    fd = os.open(p, os.O_RDONLY)
    try:
        assert os.pread(fd, 100, 0) == b"12345"
        assert os.pread(fd, 10, 5) == b""
        assert os.pread(fd, 10, 1000) == b""
    finally:
        os.close(fd)


def test_mmap_read(mount):
    """mmap over FUSE (kernel satisfies faults via READ).  The fault must
    run in a SEPARATE process: this test process embeds the cache worker,
    and a page fault holds the GIL (it is not a syscall), so a same-process
    fault would deadlock against the worker serving it — a constraint of
    embedding workers, not of the FUSE server (real apps are separate
    processes)."""
    mnt, *_ = mount
    p = f"{mnt}/mm.bin"
    data = os.urandom(128 << 10)
    with open(p, "wb") as f:
        f.write(data)
    prog = (
        "import mmap, sys\n"
        f"f = open({p!r}, 'rb')\n"
        "m = mmap.mmap(f.fileno(), 0, prot=mmap.PROT_READ)\n"
        "raw = bytes(m)\n"
        "sys.stdout.buffer.write(raw[:100] + raw[-10:])\n")
    r = subprocess.run([sys.executable, "-c", prog], capture_output=True,
                       timeout=60)
    assert r.returncode == 0, r.stderr.decode()
    assert r.stdout == data[:100] + data[-10:]


def test_renameat2_noreplace(mount):
    mnt, *_ = mount
    a, b = f"{mnt}/ra.txt", f"{mnt}/rb.txt"
    open(a, "wb").write(b"a")
    open(b, "wb").write(b"b")
    libc = ctypes.CDLL("libc.so.6", use_errno=True)
    RENAME_NOREPLACE = 1
    ret = libc.renameat2(-100, a.encode(), -100, b.encode(), RENAME_NOREPLACE)
    assert ret != 0 and ctypes.get_errno() == errno.EEXIST
    os.unlink(b)
    ret = libc.renameat2(-100, a.encode(), -100, b.encode(), RENAME_NOREPLACE)
    assert ret == 0
    assert open(b, "rb").read() == b"a"


def test_renameat2_exchange(mount):
    mnt, *_ = mount
    a, b = f"{mnt}/xa.txt", f"{mnt}/xb.txt"
    open(a, "wb").write(b"AAA")
    open(b, "wb").write(b"BBB")
    libc = ctypes.CDLL("libc.so.6", use_errno=True)
    RENAME_EXCHANGE = 2
    ret = libc.renameat2(-100, a.encode(), -100, b.encode(), RENAME_EXCHANGE)
    assert ret == 0, os.strerror(ctypes.get_errno())
    assert open(a, "rb").read() == b"BBB"
    assert open(b, "rb").read() == b"AAA"


def test_hardlink_semantics(mount):
    """link count, data shared, unlink one name keeps the data."""
    mnt, *_ = mount
    a, b = f"{mnt}/h1.txt", f"{mnt}/h2.txt"
    with open(a, "wb") as f:
        f.write(b"linked")
    os.link(a, b)
    assert os.stat(a).st_nlink == 2
    assert os.stat(a).st_ino == os.stat(b).st_ino
    os.unlink(a)
    assert open(b, "rb").read() == b"linked"


def test_o_excl(mount):
    mnt, *_ = mount
    p = f"{mnt}/excl.txt"
    fd = os.open(p, os.O_CREAT | os.O_EXCL | os.O_WRONLY, 0o600)
    os.close(fd)
    with pytest.raises(OSError) as ei:
        os.open(p, os.O_CREAT | os.O_EXCL | os.O_WRONLY)
    assert ei.value.errno == errno.EEXIST


def test_mode_bits_on_create(mount):
    mnt, *_ = mount
    p = f"{mnt}/modes.bin"
    fd = os.open(p, os.O_CREAT | os.O_WRONLY, 0o640)
    os.close(fd)
    assert stat.S_IMODE(os.stat(p).st_mode) == 0o640


def test_directory_mtime_updates_on_child_create(mount):
    mnt, *_ = mount
    d = f"{mnt}/mtdir"
    os.mkdir(d)
    m1 = os.stat(d).st_mtime_ns
    import time
    time.sleep(1.1)   # attr TTL + ms resolution
    open(f"{d}/child", "wb").close()
    m2 = os.stat(d).st_mtime_ns
    assert m2 >= m1


def test_many_small_files_listing(mount):
    mnt, *_ = mount
    d = f"{mnt}/many"
    os.mkdir(d)
    for i in range(300):
        with open(f"{d}/f{i:04d}", "wb") as f:
            f.write(b"x")
    names = sorted(os.listdir(d))
    assert len(names) == 300
    assert names[0] == "f0000" and names[-1] == "f0299"


def test_deep_paths(mount):
    mnt, *_ = mount
    path = mnt
    for i in range(20):
        path = f"{path}/d{i}"
    os.makedirs(path)
    with open(f"{path}/leaf.txt", "wb") as f:
        f.write(b"deep")
    assert open(f"{path}/leaf.txt", "rb").read() == b"deep"
    # readdir down the chain
    p = mnt
    for i in range(20):
        assert f"d{i}" in os.listdir(p)
        p = f"{p}/d{i}"


def test_fsync_and_datasync(mount):
    mnt, *_ = mount
    with open(f"{mnt}/sync.bin", "wb") as f:
        f.write(b"data")
        f.flush()
        os.fsync(f.fileno())
        os.fdatasync(f.fileno())


def test_rename_open_file_keeps_reading(mount):
    """POSIX: an open fd survives rename of its path."""
    mnt, *_ = mount
    p = f"{mnt}/moving.bin"
    data = os.urandom(2 << 20)
    with open(p, "wb") as f:
        f.write(data)
    f = open(p, "rb", buffering=0)
    first = f.read(1 << 20)
    os.rename(p, f"{mnt}/moved.bin")
    rest = f.read()
    f.close()
    assert first + rest == data


def test_read_after_unlink(mount):
    """POSIX: an open fd keeps serving data after unlink (store reader
    refcounts defer block deletion)."""
    mnt, *_ = mount
    p = f"{mnt}/unlinked.bin"
    data = os.urandom(2 << 20)
    with open(p, "wb") as f:
        f.write(data)
    f = open(p, "rb")
    head = f.read(1024)
    os.unlink(p)
    assert not os.path.exists(p)
    f.seek(0)
    assert f.read() == data          # full read after unlink
    assert head == data[:1024]
    f.close()


def test_write_fd_survives_rename(mount):
    """An open write fd keeps appending to the file under its new name."""
    mnt, *_ = mount
    src, dst = f"{mnt}/wr-a.bin", f"{mnt}/wr-b.bin"
    f = open(src, "wb")
    f.write(b"part1-")
    f.flush()
    os.rename(src, dst)
    f.write(b"part2")
    f.close()
    assert not os.path.exists(src)
    assert open(dst, "rb").read() == b"part1-part2"


def test_read_past_eof(mount):
    """fuse_read_past_eof_test analog: preads at and beyond EOF return
    empty, short tail reads clamp."""
    mnt, *_ = mount
    p = f"{mnt}/eof.bin"
    data = os.urandom(10_000)
    with open(p, "wb") as f:
        f.write(data)
    fd = os.open(p, os.O_RDONLY)
    try:
        assert os.pread(fd, 100, 10_000) == b""          # at EOF
        assert os.pread(fd, 100, 50_000) == b""          # far past EOF
        assert os.pread(fd, 1000, 9_500) == data[9_500:] # clamped tail
    finally:
        os.close(fd)


def test_truncate_storm_under_readers(mount):
    """resize_lock_p95_test analog: concurrent readers during a truncate
    storm never deadlock or corrupt (bounded by the test timeout)."""
    import threading as _th
    mnt, *_ = mount
    p = f"{mnt}/storm.bin"
    with open(p, "wb") as f:
        f.write(b"S" * 1_000_000)
    stop = _th.Event()
    errors = []

    def reader():
        while not stop.is_set():
            try:
                with open(p, "rb") as f:
                    chunk = f.read(65536)
                    assert set(chunk) <= {ord("S")}
            except FileNotFoundError:
                pass
            except OSError:
                pass   # racing a shrink is allowed to return EIO once
            except Exception as e:  # noqa: BLE001
                errors.append(e)
                return

    threads = [_th.Thread(target=reader) for _ in range(4)]
    for t in threads:
        t.start()
    try:
        for i in range(30):
            os.truncate(p, 500_000 if i % 2 else 1_000_000)
            if i % 2:
                with open(p, "ab") as f:
                    f.write(b"S" * 500_000)
    finally:
        stop.set()
        for t in threads:
            t.join(timeout=30)
    assert not errors
    assert os.path.getsize(p) == 1_000_000


def test_concurrent_write_open_same_file(mount):
    """POSIX: several threads open the SAME file O_WRONLY concurrently
    (fio rand-write shape).  The FUSE layer shares one writer across the
    handles (backend_handle.rs shared-writer analog) instead of failing
    the append lease."""
    import threading as _th
    mnt, *_ = mount
    p = f"{mnt}/shared.bin"
    with open(p, "wb") as f:
        f.write(b"\x00" * (1 << 20))
    errs = []
    slots = {}
    lk = _th.Lock()

    def w(t):
        try:
            fd = os.open(p, os.O_WRONLY)
            try:
                for k in range(8):
                    slot = (t * 8 + k) % 16
                    data = bytes([t * 16 + k]) * (64 << 10)
                    with lk:
                        slots[slot] = data
                        os.pwrite(fd, data, slot * (64 << 10))
            finally:
                os.close(fd)
        except Exception as e:  # noqa: BLE001
            errs.append(e)

    ts = [_th.Thread(target=w, args=(t,)) for t in range(4)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    assert not errs, errs[0]
    with open(p, "rb") as f:
        content = f.read()
    assert len(content) == 1 << 20
    for slot, data in slots.items():
        assert content[slot * (64 << 10):(slot + 1) * (64 << 10)] == data, \
            f"slot {slot} mismatch"


def test_setlkw_interrupted_by_signal(mount):
    """A blocked F_SETLKW whose process is signalled must be aborted in
    the DAEMON too (kernel INTERRUPT -> op_interrupt -> waiter thread
    exits with EINTR).  Without that, a ghost waiter thread grabs the
    lock for the dead process when the holder releases, and the file
    stays locked forever."""
    import fcntl
    import time as _time
    mnt, *_ = mount
    p = f"{mnt}/lk.bin"
    with open(p, "wb") as f:
        f.write(b"x" * 100)
    hold = (
        "import fcntl, sys, time\n"
        f"f = open({p!r}, 'r+b')\n"
        "fcntl.lockf(f, fcntl.LOCK_EX)\n"
        "print('LOCKED', flush=True)\n"
        "sys.stdin.readline()\n"          # release on demand
        "fcntl.lockf(f, fcntl.LOCK_UN)\n"
        "print('RELEASED', flush=True)\n"
        "time.sleep(5)\n")
    holder = subprocess.Popen([sys.executable, "-c", hold],
                              stdin=subprocess.PIPE,
                              stdout=subprocess.PIPE, text=True)
    try:
        assert holder.stdout.readline().strip() == "LOCKED"
        # waiter blocks in F_SETLKW; SIGALRM handler raises -> process
        # exits mid-wait; the kernel INTERRUPTs the in-flight request
        wprog = (
            "import fcntl, signal, sys\n"
            "def h(*a): raise KeyboardInterrupt\n"
            "signal.signal(signal.SIGALRM, h)\n"
            f"f = open({p!r}, 'r+b')\n"
            "signal.alarm(1)\n"
            "try:\n"
            "    fcntl.lockf(f, fcntl.LOCK_EX)\n"
            "    print('ACQUIRED', flush=True)\n"
            "except KeyboardInterrupt:\n"
            "    print('INTERRUPTED', flush=True)\n")
        waiter = subprocess.run([sys.executable, "-c", wprog],
                                capture_output=True, text=True, timeout=20)
        assert waiter.stdout.strip() == "INTERRUPTED", \
            (waiter.stdout, waiter.stderr)
        _time.sleep(0.3)   # let the INTERRUPT reach the daemon
        holder.stdin.write("go\n")
        holder.stdin.flush()
        assert holder.stdout.readline().strip() == "RELEASED"
        # no ghost: a fresh exclusive lock must succeed promptly
        probe = (
            "import fcntl, time\n"
            f"f = open({p!r}, 'r+b')\n"
            "t0 = time.monotonic()\n"
            "fcntl.lockf(f, fcntl.LOCK_EX)\n"
            "print('OK', round(time.monotonic()-t0, 2), flush=True)\n")
        r = subprocess.run([sys.executable, "-c", probe],
                           capture_output=True, text=True, timeout=15)
        out = r.stdout.split()
        assert out and out[0] == "OK", (r.stdout, r.stderr)
        assert float(out[1]) < 5, f"ghost waiter held the lock: {out}"
    finally:
        holder.terminate()
        holder.wait(timeout=10)


def test_truncate_extend_closed_file(mount):
    """ftruncate/truncate growth on a closed file: size grows, the tail
    reads back as zeros (hole)."""
    mnt = mount[0]
    p = f"{mnt}/extend.bin"
    with open(p, "wb") as f:
        f.write(b"abc")
    os.truncate(p, 4096)
    assert os.stat(p).st_size == 4096
    data = open(p, "rb").read()
    assert data[:3] == b"abc" and len(data) == 4096
    assert data[3:] == b"\0" * 4093


def test_ftruncate_extend_open_writer(mount):
    """Growth via an open write handle persists across close."""
    mnt = mount[0]
    p = f"{mnt}/extend2.bin"
    with open(p, "wb") as f:
        f.write(b"xy")
        f.flush()
        os.ftruncate(f.fileno(), 1 << 20)
        assert os.fstat(f.fileno()).st_size == 1 << 20
    assert os.stat(p).st_size == 1 << 20
    data = open(p, "rb").read()
    assert data[:2] == b"xy" and data[2:] == b"\0" * ((1 << 20) - 2)


def test_posix_fallocate_extends(mount):
    """posix_fallocate allocates AND grows the size; KEEP_SIZE does not
    change the size."""
    mnt = mount[0]
    p = f"{mnt}/falloc.bin"
    with open(p, "wb") as f:
        f.write(b"data")
        f.flush()
        os.posix_fallocate(f.fileno(), 0, 8192)
        assert os.fstat(f.fileno()).st_size == 8192
    assert os.stat(p).st_size == 8192
    assert open(p, "rb").read()[:4] == b"data"

    q = f"{mnt}/falloc_keep.bin"
    libc = ctypes.CDLL(None, use_errno=True)
    with open(q, "wb") as f:
        f.write(b"1234")
        f.flush()
        FALLOC_FL_KEEP_SIZE = 0x01
        rc = libc.fallocate(f.fileno(), FALLOC_FL_KEEP_SIZE,
                            ctypes.c_long(0), ctypes.c_long(1 << 16))
        assert rc == 0, os.strerror(ctypes.get_errno())
        assert os.fstat(f.fileno()).st_size == 4


def test_o_append_positioning(mount):
    """O_APPEND writes land at EOF even after another handle extends the
    file."""
    mnt = mount[0]
    p = f"{mnt}/append.bin"
    with open(p, "wb") as f:
        f.write(b"base")
    fd = os.open(p, os.O_WRONLY | os.O_APPEND)
    try:
        os.write(fd, b"-tail")
    finally:
        os.close(fd)
    assert open(p, "rb").read() == b"base-tail"


def test_seek_hole_data(mount):
    """SEEK_HOLE/SEEK_DATA reflect the cached extent map: the tail of an
    extended file is a hole; dense files report one data segment."""
    mnt = mount[0]
    p = f"{mnt}/holes.bin"
    with open(p, "wb") as f:
        f.write(b"x" * 8192)
    os.truncate(p, 1 << 20)          # tail hole [8192, 1 MiB)

    fd = os.open(p, os.O_RDONLY)
    try:
        assert os.lseek(fd, 0, os.SEEK_DATA) == 0
        hole = os.lseek(fd, 0, os.SEEK_HOLE)
        assert hole == 8192
        # SEEK_DATA inside the trailing hole: ENXIO
        with pytest.raises(OSError) as ei:
            os.lseek(fd, 8192, os.SEEK_DATA)
        assert ei.value.errno == errno.ENXIO
        # past EOF: ENXIO
        with pytest.raises(OSError):
            os.lseek(fd, 2 << 20, os.SEEK_HOLE)
    finally:
        os.close(fd)

    q = f"{mnt}/dense.bin"
    with open(q, "wb") as f:
        f.write(b"y" * 4096)
    fd = os.open(q, os.O_RDONLY)
    try:
        assert os.lseek(fd, 0, os.SEEK_HOLE) == 4096   # hole == EOF
        assert os.lseek(fd, 100, os.SEEK_DATA) == 100
    finally:
        os.close(fd)


def test_xattr_corpus(mount):
    """set/get/list/remove xattrs, ENODATA on missing, replace
    semantics, many attributes listed back."""
    mnt = mount[0]
    p = f"{mnt}/xa.bin"
    open(p, "wb").write(b"z")
    os.setxattr(p, "user.one", b"1")
    os.setxattr(p, "user.two", b"22")
    assert os.getxattr(p, "user.one") == b"1"
    os.setxattr(p, "user.one", b"replaced")
    assert os.getxattr(p, "user.one") == b"replaced"
    names = set(os.listxattr(p))
    assert {"user.one", "user.two"} <= names
    os.removexattr(p, "user.two")
    assert "user.two" not in set(os.listxattr(p))
    with pytest.raises(OSError) as ei:
        os.getxattr(p, "user.two")
    assert ei.value.errno in (errno.ENODATA, 61)
    with pytest.raises(OSError):
        os.removexattr(p, "user.gone")
    # a pile of attributes round-trips through listxattr paging
    for i in range(40):
        os.setxattr(p, f"user.k{i:02d}", str(i).encode())
    got = set(os.listxattr(p))
    assert all(f"user.k{i:02d}" in got for i in range(40))


def test_readdir_stable_under_mutation(mount):
    """Listing a directory while entries are created/deleted never
    crashes and yields each stable entry exactly once."""
    mnt = mount[0]
    d = f"{mnt}/churn"
    os.mkdir(d)
    stable = {f"s{i}" for i in range(50)}
    for name in stable:
        open(f"{d}/{name}", "wb").close()
    import threading

    stop = threading.Event()

    def churn():
        i = 0
        while not stop.is_set():
            name = f"{d}/tmp{i % 7}"
            try:
                open(name, "wb").close()
                os.unlink(name)
            except OSError:
                pass
            i += 1

    th = threading.Thread(target=churn)
    th.start()
    try:
        for _ in range(20):
            seen = os.listdir(d)
            assert len(seen) == len(set(seen))   # no duplicates
            assert stable <= set(seen)
    finally:
        stop.set()
        th.join()
