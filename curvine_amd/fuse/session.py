"""FUSE session: pure mount (no libfuse), channels, dispatch loop.

Analog of the reference's libfuse-free mount + session
(/root/reference/curvine-fuse/src/raw/fuse_pure.rs:17-160 mount via
libc::mount with fd=..., session/fuse_session.rs:154-260 multi-channel run
loop, session/channel/fuse_receiver.rs dispatch).  Channels are cloned
with FUSE_DEV_IOC_CLONE (clone_fd analog) and each gets an OS thread; the
hot READ/WRITE path never touches the asyncio loop when the target block
lives in this process (HBM short-circuit).
"""
from __future__ import annotations

import ctypes
import errno
import fcntl
import logging
import os
import struct
import threading

from curvine_amd.fuse import abi

log = logging.getLogger("curvine.fuse")

# opt-in per-op trace for protocol debugging (CURVINE_FUSE_OP_TRACE=1)
_OP_TRACE = bool(os.environ.get("CURVINE_FUSE_OP_TRACE"))

libc = ctypes.CDLL("libc.so.6", use_errno=True)
FUSE_DEV_IOC_CLONE = 0x8004E500
MNT_DETACH = 2


def mount_fuse(mnt_path: str, allow_other: bool = True,
               default_permissions: bool = False,
               rootmode: int = 0o40755, max_read: int = 1 << 20) -> int:
    """Open /dev/fuse and mount it at mnt_path; returns the session fd."""
    os.makedirs(mnt_path, exist_ok=True)
    fd = os.open("/dev/fuse", os.O_RDWR)
    opts = (f"fd={fd},rootmode={rootmode & 0o170000:o},user_id=0,group_id=0"
            f",max_read={max_read}")
    if allow_other:
        opts += ",allow_other"
    if default_permissions:
        opts += ",default_permissions"
    ret = libc.mount(b"curvinefs", mnt_path.encode(), b"fuse.curvinefs",
                     0, opts.encode())
    if ret != 0:
        e = ctypes.get_errno()
        os.close(fd)
        raise OSError(e, f"fuse mount at {mnt_path}: {os.strerror(e)}")
    return fd


def tune_readahead(mnt_path: str, readahead_kb: int) -> None:
    """Raise the mount's BDI read_ahead_kb so the kernel pipelines larger
    readahead windows (fuse_session.rs:154-190 max_readahead_kb analog).

    The device id comes from /proc/self/mountinfo, NOT from stat(mnt): the
    daemon statting its own mount before serving requests deadlocks.
    """
    try:
        dev = None
        with open("/proc/self/mountinfo") as f:
            for line in f:
                parts = line.split()
                if len(parts) > 4 and parts[4] == mnt_path:
                    dev = parts[2]   # "major:minor"
        if dev is None:
            return
        with open(f"/sys/class/bdi/{dev}/read_ahead_kb", "w") as f:
            f.write(str(readahead_kb))
        log.info("bdi %s read_ahead_kb=%d", dev, readahead_kb)
    except OSError as e:
        log.debug("bdi readahead tuning failed: %s", e)


def clone_channel(session_fd: int) -> int:
    """FUSE_DEV_IOC_CLONE: a second queue fd on the same session."""
    fd = os.open("/dev/fuse", os.O_RDWR)
    buf = struct.pack("I", session_fd)
    fcntl.ioctl(fd, FUSE_DEV_IOC_CLONE, buf)
    return fd


def umount(mnt_path: str) -> None:
    libc.umount2(mnt_path.encode(), MNT_DETACH)


class FuseChannel(threading.Thread):
    """One /dev/fuse queue consumer."""

    def __init__(self, session: "FuseSession", fd: int, idx: int):
        super().__init__(daemon=True, name=f"fuse-ch{idx}")
        self.session = session
        self.fd = fd
        self.idx = idx
        self.bufsize = session.max_write + (64 << 10)
        # pinned request/reply buffers: WRITE payloads DMA host->HBM straight
        # from the request buffer; READ replies DMA HBM->host straight into
        # the buffer handed to writev — no staging hop either way.
        from curvine_amd.native import PinnedBuffer
        from curvine_amd.metrics import OpStats
        self.req_buf = PinnedBuffer(self.bufsize)
        self.reply_pin = PinnedBuffer(self.bufsize)
        self.stats = OpStats()

    def run(self) -> None:
        req_view = self.req_buf.view
        while not self.session.stopped:
            try:
                n = os.readv(self.fd, [req_view])
            except OSError as e:
                if e.errno == errno.EINTR:
                    continue
                if e.errno in (errno.ENODEV, errno.EBADF):
                    break   # unmounted
                if e.errno == errno.ENOENT:
                    continue  # request aborted before we read it
                log.error("fuse read ch%d: %s", self.idx, e)
                break
            if n <= 0:
                break
            try:
                self.dispatch(req_view[:n])
            except Exception as e:  # noqa: BLE001
                log.exception("fuse dispatch failed: %s", e)
        log.info("fuse channel %d exiting", self.idx)

    def dispatch(self, req) -> None:
        import time as _time
        t0 = _time.perf_counter()
        (length, opcode, unique, nodeid, uid, gid, pid, _extlen, _pad) = \
            abi.IN_HEADER.unpack_from(req, 0)
        body = memoryview(req)[abi.IN_HEADER_SIZE:length]
        if _OP_TRACE:
            with open("/tmp/fuse_op_trace.log", "a") as _tf:
                _tf.write(f"{abi.Op.NAMES.get(opcode, opcode)} "
                          f"node={nodeid} len={length}\n")
        handler = self.session.fs.HANDLERS.get(opcode)
        if handler is None:
            log.debug("fuse op %s unimplemented",
                      abi.Op.NAMES.get(opcode, opcode))
            self.reply_error(unique, errno.ENOSYS)
            return
        try:
            result = handler(self.session.fs, nodeid, body,
                             (uid, gid, pid, unique, self))
        except OSError as e:
            self.reply_error(unique, e.errno or errno.EIO)
            return
        except Exception as e:  # noqa: BLE001
            from curvine_amd.errors import to_errno
            eno = to_errno(e)
            if eno == errno.EIO:
                log.warning("fuse op %s error: %s",
                            abi.Op.NAMES.get(opcode, opcode), e)
            self.reply_error(unique, eno)
            return
        if result is None:
            return   # no reply (FORGET, INTERRUPT, or handler replied itself)
        self.reply(unique, result)
        self.stats.record(abi.Op.NAMES.get(opcode, str(opcode)),
                          _time.perf_counter() - t0,
                          len(result) if not isinstance(result, (list, tuple))
                          else sum(len(p) for p in result))

    # ---------------- replies ----------------
    def reply(self, unique: int, body) -> None:
        if isinstance(body, (list, tuple)):
            parts = body
        else:
            parts = [body]
        total = abi.OUT_HEADER_SIZE + sum(len(p) for p in parts)
        hdr = abi.OUT_HEADER.pack(total, 0, unique)
        try:
            os.writev(self.fd, [hdr, *parts])
        except OSError as e:
            if e.errno not in (errno.ENOENT, errno.ENODEV):
                log.warning("fuse reply failed: %s", e)

    def reply_error(self, unique: int, eno: int) -> None:
        hdr = abi.OUT_HEADER.pack(abi.OUT_HEADER_SIZE, -eno, unique)
        try:
            os.write(self.fd, hdr)
        except OSError:
            pass


class ForwardChannel(threading.Thread):
    """Python worker draining the native loop's forward queue (metadata
    ops + unregistered reads); replies directly on the session fd."""

    def __init__(self, session: "FuseSession", loop_id: int, idx: int):
        super().__init__(daemon=True, name=f"fuse-fwd{idx}")
        self.session = session
        self.loop_id = loop_id   # NB: Thread already owns 'native_id'
        self.fd = session.session_fd
        self.idx = idx
        from curvine_amd.metrics import OpStats
        from curvine_amd.native import PinnedBuffer, load
        self._native = load()
        self.stats = OpStats()
        self.reply_pin = PinnedBuffer(session.max_write + (64 << 10))
        self.req_buf = None   # forwarded requests are plain bytes

    def run(self) -> None:
        import os as _os
        import time as _t
        trace = _os.environ.get("CV_FUSE_TRACE") == "1"
        dispatch = FuseChannel.dispatch
        while not self.session.stopped:
            origin_fd, raw = self._native.fuse_loop_next_forward(
                self.loop_id, 0.5)
            if not raw:
                continue
            self.fd = origin_fd   # replies go to the fd that read the request
            if trace:
                (ln, opc, uniq, *_r) = abi.IN_HEADER.unpack_from(raw, 0)
                t0 = _t.perf_counter()
                log.info("fwd%d pull %s u=%d", self.idx,
                         abi.Op.NAMES.get(opc, opc), uniq)
            try:
                dispatch(self, memoryview(raw))
            except Exception as e:  # noqa: BLE001
                log.exception("fuse forward dispatch: %s", e)
            if trace:
                log.info("fwd%d done %s u=%d %.1fms", self.idx,
                         abi.Op.NAMES.get(opc, opc), uniq,
                         ( _t.perf_counter() - t0) * 1000)

    # reply plumbing shared with FuseChannel
    reply = FuseChannel.reply
    reply_error = FuseChannel.reply_error


class FuseSession:
    def __init__(self, fs, mnt_path: str, channels: int = 1,
                 max_write: int = 1 << 20, allow_other: bool = True,
                 native_loop: bool = True):
        self.fs = fs
        self.mnt_path = mnt_path
        self.n_channels = max(1, channels)
        self.max_write = max_write
        self.native_loop = native_loop
        self.native_id = None
        self.stopped = False
        self.session_fd = -1
        self.channels: list = []
        fs.session = self

    def start(self, session_fd: int | None = None) -> "FuseSession":
        """Mount fresh, or adopt an existing session fd (hot upgrade:
        the kernel mount persists; we clone our channels off the passed
        fd and serve the same connection)."""
        if session_fd is None:
            self.session_fd = mount_fuse(self.mnt_path, allow_other=True,
                                         max_read=self.max_write)
            tune_readahead(
                self.mnt_path,
                getattr(self.fs.conf.fuse, "max_readahead", 8 << 20) >> 10)
        else:
            self.session_fd = session_fd
        fds = [self.session_fd]
        for i in range(1, self.n_channels):
            fds.append(clone_channel(self.session_fd))
        if self.native_loop:
            # GIL-free C++ channel threads serve registered READs; the rest
            # is forwarded to Python ForwardChannel workers
            from curvine_amd.native import load
            native = load()
            self.native_id = native.fuse_loop_create(self.max_write)
            for fd in fds:
                native.fuse_loop_add_channel(self.native_id, fd)
            n_fwd = max(2, min(4, self.n_channels))
            for i in range(n_fwd):
                ch = ForwardChannel(self, self.native_id, i)
                self.channels.append(ch)
                ch.start()
        else:
            for i, fd in enumerate(fds):
                ch = FuseChannel(self, fd, i)
                self.channels.append(ch)
                ch.start()
        log.info("fuse %s at %s (%d channels%s)",
                 "adopted" if session_fd is not None else "mounted",
                 self.mnt_path, len(fds),
                 ", native loop" if self.native_id is not None else "")
        return self

    # ---------------- native read registration ----------------
    def try_register_read(self, fh: int, srs) -> bool:
        """Register an open read handle's arena extents with the native
        loop (all blocks must be in-process arena-resident)."""
        if self.native_id is None:
            return False
        exts = []
        for idx, lb in enumerate(srs.fb.blocks):
            r = srs._local.get(idx)
            if r is None:
                r = srs._open_local(idx, lb)
            if r is None or r.meta.get("kind") != "arena":
                return False
            exts.append((lb.offset, lb.block.length,
                         r.layout.arena.handle, r.meta["offset"]))
        from curvine_amd.native import load
        load().fuse_loop_register(self.native_id, fh, srs.length, exts)
        return True

    def unregister_read(self, fh: int) -> None:
        if self.native_id is not None:
            from curvine_amd.native import load
            load().fuse_loop_unregister(self.native_id, fh)

    NOTIFY_INVAL_INODE = 2

    def notify_inval_inode(self, nodeid: int) -> None:
        """Push a kernel page/attr-cache invalidation for one inode
        (needed when content moves beneath a path, e.g. RENAME_EXCHANGE)."""
        import struct as _s
        payload = _s.pack("<Qqq", nodeid, 0, -1)
        hdr = abi.OUT_HEADER.pack(abi.OUT_HEADER_SIZE + len(payload),
                                  self.NOTIFY_INVAL_INODE, 0)
        try:
            os.write(self.session_fd, hdr + payload)
        except OSError:
            pass   # ENOENT: kernel has nothing cached for it

    def stats(self) -> dict:
        from curvine_amd.metrics import OpStats
        out = OpStats.merge([ch.stats for ch in self.channels])
        if self.native_id is not None:
            from curvine_amd.native import load
            out["_native_loop"] = load().fuse_loop_stats(self.native_id)
        return out

    def stop(self, umount_fs: bool = True) -> None:
        """umount_fs=False: hand-over shutdown — the kernel mount stays
        alive and is served by the adopting daemon."""
        self.stopped = True
        if umount_fs:
            umount(self.mnt_path)
        if self.native_id is not None:
            from curvine_amd.native import load
            load().fuse_loop_stop(self.native_id)
        else:
            for ch in self.channels:
                try:
                    os.close(ch.fd)
                except OSError:
                    pass
        for ch in self.channels:
            ch.join(timeout=3)
        self.channels = []
