"""Error taxonomy for curvine_amd.

Mirrors the capability of the reference's `curvine-error` crate
(/root/reference/crates/common/curvine-error, FsError taxonomy) and the
cross-process error encode/decode of `crates/core/error`: every error carries
a stable numeric code so it survives the RPC boundary and can be re-raised as
the same exception class on the client.
"""
from __future__ import annotations

import enum


class ErrorCode(enum.IntEnum):
    OK = 0
    COMMON = 1
    IO = 2
    FILE_NOT_FOUND = 3
    FILE_ALREADY_EXISTS = 4
    DIR_NOT_EMPTY = 5
    NOT_DIRECTORY = 6
    IS_DIRECTORY = 7
    INVALID_PATH = 8
    INVALID_ARGUMENT = 9
    BLOCK_NOT_FOUND = 10
    NO_AVAILABLE_WORKER = 11
    WORKER_NOT_FOUND = 12
    BLOCK_IN_WRITING = 13
    FILE_IN_WRITING = 14
    LEASE_EXPIRED = 15
    CAPACITY_EXCEEDED = 16
    CHECKSUM_MISMATCH = 17
    UNSUPPORTED = 18
    TIMEOUT = 19
    CANCELLED = 20
    NOT_LEADER = 21
    EXPIRED = 22
    ABNORMAL_DATA = 23
    MOUNT_NOT_FOUND = 24
    UFS_ERROR = 25
    QUOTA_EXCEEDED = 26
    PERMISSION_DENIED = 27
    STALE_GENERATION = 28
    OUT_OF_RANGE = 29
    JOB_NOT_FOUND = 30
    RETRY = 31
    NOT_EMPTY = 32
    CONNECT = 33
    INCOMPATIBLE_VERSION = 34


class FsError(Exception):
    """Base exception; `code` crosses the wire (see rpc.message)."""

    code: ErrorCode = ErrorCode.COMMON

    def __init__(self, msg: str = "", code: ErrorCode | None = None):
        super().__init__(msg)
        if code is not None:
            self.code = ErrorCode(code)

    @property
    def message(self) -> str:
        return str(self)

    # ---- wire encode/decode (analog of core/error error_encoder.rs) ----
    def encode(self) -> tuple[int, str]:
        return int(self.code), self.message

    @staticmethod
    def decode(code: int, msg: str) -> "FsError":
        cls = _CODE_TO_CLASS.get(ErrorCode(code), FsError)
        e = cls(msg)
        e.code = ErrorCode(code)
        return e

    @staticmethod
    def from_code(code: ErrorCode, msg: str = "") -> "FsError":
        cls = _CODE_TO_CLASS.get(code, FsError)
        e = cls(msg)
        e.code = code
        return e


def _err(name: str, code: ErrorCode) -> type:
    cls = type(name, (FsError,), {"code": code})
    return cls


FileNotFound = _err("FileNotFound", ErrorCode.FILE_NOT_FOUND)
FileAlreadyExists = _err("FileAlreadyExists", ErrorCode.FILE_ALREADY_EXISTS)
DirNotEmpty = _err("DirNotEmpty", ErrorCode.DIR_NOT_EMPTY)
NotDirectory = _err("NotDirectory", ErrorCode.NOT_DIRECTORY)
IsDirectory = _err("IsDirectory", ErrorCode.IS_DIRECTORY)
InvalidPath = _err("InvalidPath", ErrorCode.INVALID_PATH)
InvalidArgument = _err("InvalidArgument", ErrorCode.INVALID_ARGUMENT)
BlockNotFound = _err("BlockNotFound", ErrorCode.BLOCK_NOT_FOUND)
NoAvailableWorker = _err("NoAvailableWorker", ErrorCode.NO_AVAILABLE_WORKER)
WorkerNotFound = _err("WorkerNotFound", ErrorCode.WORKER_NOT_FOUND)
BlockInWriting = _err("BlockInWriting", ErrorCode.BLOCK_IN_WRITING)
FileInWriting = _err("FileInWriting", ErrorCode.FILE_IN_WRITING)
CapacityExceeded = _err("CapacityExceeded", ErrorCode.CAPACITY_EXCEEDED)
ChecksumMismatch = _err("ChecksumMismatch", ErrorCode.CHECKSUM_MISMATCH)
Unsupported = _err("Unsupported", ErrorCode.UNSUPPORTED)
RpcTimeout = _err("RpcTimeout", ErrorCode.TIMEOUT)
Cancelled = _err("Cancelled", ErrorCode.CANCELLED)
NotLeader = _err("NotLeader", ErrorCode.NOT_LEADER)
MountNotFound = _err("MountNotFound", ErrorCode.MOUNT_NOT_FOUND)
UfsError = _err("UfsError", ErrorCode.UFS_ERROR)
QuotaExceeded = _err("QuotaExceeded", ErrorCode.QUOTA_EXCEEDED)
PermissionDenied = _err("PermissionDenied", ErrorCode.PERMISSION_DENIED)
StaleGeneration = _err("StaleGeneration", ErrorCode.STALE_GENERATION)
OutOfRange = _err("OutOfRange", ErrorCode.OUT_OF_RANGE)
JobNotFound = _err("JobNotFound", ErrorCode.JOB_NOT_FOUND)
RetryError = _err("RetryError", ErrorCode.RETRY)
ConnectError = _err("ConnectError", ErrorCode.CONNECT)
IncompatibleVersion = _err("IncompatibleVersion",
                           ErrorCode.INCOMPATIBLE_VERSION)

_CODE_TO_CLASS: dict[ErrorCode, type] = {
    c.code: c  # type: ignore[attr-defined]
    for c in [
        FileNotFound, FileAlreadyExists, DirNotEmpty, NotDirectory,
        IsDirectory, InvalidPath, InvalidArgument, BlockNotFound,
        NoAvailableWorker, WorkerNotFound, BlockInWriting, FileInWriting,
        CapacityExceeded, ChecksumMismatch, Unsupported, RpcTimeout,
        Cancelled, NotLeader, MountNotFound, UfsError, QuotaExceeded,
        PermissionDenied, StaleGeneration, OutOfRange, JobNotFound,
        RetryError, ConnectError, IncompatibleVersion,
    ]
}

# errno mapping for the FUSE layer
import errno as _errno  # noqa: E402

ERRNO_MAP: dict[ErrorCode, int] = {
    ErrorCode.FILE_NOT_FOUND: _errno.ENOENT,
    ErrorCode.FILE_ALREADY_EXISTS: _errno.EEXIST,
    ErrorCode.DIR_NOT_EMPTY: _errno.ENOTEMPTY,
    ErrorCode.NOT_DIRECTORY: _errno.ENOTDIR,
    ErrorCode.IS_DIRECTORY: _errno.EISDIR,
    ErrorCode.INVALID_PATH: _errno.EINVAL,
    ErrorCode.INVALID_ARGUMENT: _errno.EINVAL,
    ErrorCode.CAPACITY_EXCEEDED: _errno.ENOSPC,
    ErrorCode.QUOTA_EXCEEDED: _errno.EDQUOT,
    ErrorCode.PERMISSION_DENIED: _errno.EACCES,
    ErrorCode.UNSUPPORTED: _errno.ENOSYS,
    ErrorCode.TIMEOUT: _errno.ETIMEDOUT,
    ErrorCode.OUT_OF_RANGE: _errno.EINVAL,
    ErrorCode.IO: _errno.EIO,
}


def to_errno(e: Exception) -> int:
    if isinstance(e, FsError):
        return ERRNO_MAP.get(e.code, _errno.EIO)
    if isinstance(e, (FileNotFoundError,)):
        return _errno.ENOENT
    return _errno.EIO
