"""Raw /dev/fuse kernel ABI.

Hand-declared structs and opcodes for the FUSE character-device protocol,
the analog of the reference's raw/fuse_abi.rs (457 lines,
/root/reference/curvine-fuse/src/raw/fuse_abi.rs) and
session/fuse_op_code.rs.  Layouts follow include/uapi/linux/fuse.h
(protocol 7.3x).
"""
from __future__ import annotations

import struct

FUSE_KERNEL_VERSION = 7
FUSE_KERNEL_MINOR_VERSION = 36


class Op:
    LOOKUP = 1
    FORGET = 2
    GETATTR = 3
    SETATTR = 4
    READLINK = 5
    SYMLINK = 6
    MKNOD = 8
    MKDIR = 9
    UNLINK = 10
    RMDIR = 11
    RENAME = 12
    LINK = 13
    OPEN = 14
    READ = 15
    WRITE = 16
    STATFS = 17
    RELEASE = 18
    FSYNC = 20
    SETXATTR = 21
    GETXATTR = 22
    LISTXATTR = 23
    REMOVEXATTR = 24
    FLUSH = 25
    INIT = 26
    OPENDIR = 27
    READDIR = 28
    RELEASEDIR = 29
    FSYNCDIR = 30
    GETLK = 31
    SETLK = 32
    SETLKW = 33
    ACCESS = 34
    CREATE = 35
    INTERRUPT = 36
    BMAP = 37
    DESTROY = 38
    IOCTL = 39
    POLL = 40
    NOTIFY_REPLY = 41
    BATCH_FORGET = 42
    FALLOCATE = 43
    READDIRPLUS = 44
    RENAME2 = 45
    LSEEK = 46
    COPY_FILE_RANGE = 47
    SETUPMAPPING = 48
    REMOVEMAPPING = 49
    SYNCFS = 50
    TMPFILE = 51
    STATX = 52

    NAMES = {}


Op.NAMES = {v: k for k, v in vars(Op).items() if isinstance(v, int)}

# init flags (subset)
FUSE_ASYNC_READ = 1 << 0
FUSE_POSIX_LOCKS = 1 << 1
FUSE_ATOMIC_O_TRUNC = 1 << 3
FUSE_BIG_WRITES = 1 << 5
FUSE_SPLICE_WRITE = 1 << 7
FUSE_SPLICE_MOVE = 1 << 8
FUSE_SPLICE_READ = 1 << 9
FUSE_DO_READDIRPLUS = 1 << 13
FUSE_READDIRPLUS_AUTO = 1 << 14
FUSE_ASYNC_DIO = 1 << 15
FUSE_WRITEBACK_CACHE = 1 << 16
FUSE_PARALLEL_DIROPS = 1 << 18
FUSE_HANDLE_KILLPRIV = 1 << 19
FUSE_MAX_PAGES = 1 << 22
FUSE_CACHE_SYMLINKS = 1 << 23
FUSE_HANDLE_KILLPRIV_V2 = 1 << 28   # kernel skips security.capability xattr
                                    # lookups before every WRITE

# headers
IN_HEADER = struct.Struct("<IIQQIIIHH")      # len opcode unique nodeid uid gid pid total_extlen pad
OUT_HEADER = struct.Struct("<IiQ")           # len error unique
IN_HEADER_SIZE = IN_HEADER.size              # 40
OUT_HEADER_SIZE = OUT_HEADER.size            # 16

# bodies
INIT_IN = struct.Struct("<IIII")             # major minor max_readahead flags (+flags2+unused in 7.36)
INIT_OUT = struct.Struct("<IIIIHHIIHHII24x") # major minor max_readahead flags max_bg cong max_write time_gran max_pages map_align flags2 max_stack_depth unused[6]
ATTR = struct.Struct("<QQQQQQIIIIIIIIII")    # ino size blocks atime mtime ctime atimensec mtimensec ctimensec mode nlink uid gid rdev blksize flags
ENTRY_OUT = struct.Struct("<QQQQII")         # nodeid generation entry_valid attr_valid entry_nsec attr_nsec (+ATTR)
ATTR_OUT = struct.Struct("<QII")             # attr_valid attr_valid_nsec dummy (+ATTR)
GETATTR_IN = struct.Struct("<IIQ")           # flags dummy fh
SETATTR_IN = struct.Struct("<IIQQQQQQIIIIIIII")  # valid pad fh size lock_owner atime mtime ctime atimensec mtimensec ctimensec mode unused4 uid gid unused5
OPEN_IN = struct.Struct("<II")               # flags open_flags
OPEN_OUT = struct.Struct("<QII")             # fh open_flags backing_id
CREATE_IN = struct.Struct("<IIII")           # flags mode umask open_flags (+name)
MKDIR_IN = struct.Struct("<II")              # mode umask (+name)
MKNOD_IN = struct.Struct("<IIII")            # mode rdev umask pad (+name)
RENAME_IN = struct.Struct("<Q")              # newdir (+names)
RENAME2_IN = struct.Struct("<QII")           # newdir flags pad (+names)
LINK_IN = struct.Struct("<Q")                # oldnodeid (+name)
READ_IN = struct.Struct("<QQIIQII")          # fh offset size read_flags lock_owner flags pad
WRITE_IN = struct.Struct("<QQIIQII")         # fh offset size write_flags lock_owner flags pad
WRITE_OUT = struct.Struct("<II")             # size pad
RELEASE_IN = struct.Struct("<QIIQ")          # fh flags release_flags lock_owner
FLUSH_IN = struct.Struct("<QIIQ")            # fh unused pad lock_owner
FSYNC_IN = struct.Struct("<QII")             # fh fsync_flags pad
FORGET_IN = struct.Struct("<Q")              # nlookup
BATCH_FORGET_IN = struct.Struct("<II")       # count dummy
FORGET_ONE = struct.Struct("<QQ")            # nodeid nlookup
ACCESS_IN = struct.Struct("<II")             # mask pad
INTERRUPT_IN = struct.Struct("<Q")           # unique
FALLOCATE_IN = struct.Struct("<QQQII")       # fh offset length mode pad
LSEEK_IN = struct.Struct("<QQII")            # fh offset whence pad
LSEEK_OUT = struct.Struct("<Q")              # offset
KSTATFS = struct.Struct("<QQQQQIIII24x")     # blocks bfree bavail files ffree bsize namelen frsize pad spare[6]
GETXATTR_IN = struct.Struct("<II")           # size pad
GETXATTR_OUT = struct.Struct("<II")          # size pad
SETXATTR_IN = struct.Struct("<II")           # size flags (compat, no SETXATTR_EXT)
LK_IN = struct.Struct("<QQQQIIII")           # fh owner start end type pid lk_flags pad
LK_OUT = struct.Struct("<QQII")              # start end type pid
CFR_IN = struct.Struct("<QQQQQQQ")           # fh_in off_in nodeid_out fh_out off_out len flags

DIRENT_HDR = struct.Struct("<QQII")          # ino off namelen type

# setattr valid bits
FATTR_MODE = 1 << 0
FATTR_UID = 1 << 1
FATTR_GID = 1 << 2
FATTR_SIZE = 1 << 3
FATTR_ATIME = 1 << 4
FATTR_MTIME = 1 << 5
FATTR_FH = 1 << 6
FATTR_ATIME_NOW = 1 << 7
FATTR_MTIME_NOW = 1 << 8
FATTR_LOCKOWNER = 1 << 9
FATTR_CTIME = 1 << 10

FUSE_ROOT_ID = 1

# dirent types (matches (mode >> 12))
DT_DIR = 4
DT_REG = 8
DT_LNK = 10


def pack_dirent(ino: int, off: int, name: bytes, dtype: int) -> bytes:
    ent = DIRENT_HDR.pack(ino, off, len(name), dtype) + name
    pad = (8 - len(ent) % 8) % 8
    return ent + b"\x00" * pad


def pack_attr(st) -> bytes:
    """st: FileStatus-like -> fuse_attr bytes."""
    from curvine_amd.model import FileType
    if st.file_type == FileType.DIR:
        mode = 0o040000 | (st.mode & 0o7777)
    elif st.file_type == FileType.SYMLINK:
        mode = 0o120000 | 0o777
    else:
        mode = 0o100000 | (st.mode & 0o7777)
    blocks = (st.length + 511) // 512
    mt = st.mtime_ms
    at = st.atime_ms or mt
    return ATTR.pack(st.inode_id, st.length, blocks,
                     at // 1000, mt // 1000, mt // 1000,
                     (at % 1000) * 1_000_000, (mt % 1000) * 1_000_000,
                     (mt % 1000) * 1_000_000,
                     mode, max(1, st.nlink), st.uid, st.gid, 0, 4096, 0)
