"""C ABI for non-Python SDKs (csrc/sdk_abi.cpp): the cv_* symbols the
JNI binding will wrap, driven from ctypes against a live MiniCluster —
metadata ops, multi-block streaming write, and streamed reads, all with
zero Python in the client path."""
import ctypes
import os

import pytest


class CvStatus(ctypes.Structure):
    _fields_ = [("inode_id", ctypes.c_int64),
                ("length", ctypes.c_int64),
                ("mtime_ms", ctypes.c_int64),
                ("file_type", ctypes.c_int32),
                ("is_complete", ctypes.c_int32),
                ("mode", ctypes.c_int32),
                ("nlink", ctypes.c_int32)]


@pytest.fixture
def lib():
    from curvine_amd.native import _SO, load
    load()   # ensure built
    L = ctypes.CDLL(_SO)
    L.cv_fs_new.restype = ctypes.c_int64
    L.cv_fs_new.argtypes = [ctypes.c_char_p, ctypes.c_int]
    L.cv_open.restype = ctypes.c_int64
    L.cv_create.restype = ctypes.c_int64
    L.cv_read.restype = ctypes.c_int64
    L.cv_write.restype = ctypes.c_int64
    L.cv_reader_len.restype = ctypes.c_int64
    return L


@pytest.fixture
def cluster(tmp_path):
    from curvine_amd.testing import SyncMiniCluster

    smc = SyncMiniCluster(tmp_dir=str(tmp_path / "cv")).start()
    yield smc
    smc.stop()


def _err(lib) -> str:
    buf = ctypes.create_string_buffer(512)
    lib.cv_last_error(buf, 512)
    return buf.value.decode()


def test_c_abi_end_to_end(lib, cluster):
    port = cluster.master.rpc.port
    fs = lib.cv_fs_new(b"127.0.0.1", port)
    assert fs > 0, _err(lib)
    try:
        # metadata surface
        assert lib.cv_mkdir(fs, b"/cabi/sub") == 0, _err(lib)
        assert lib.cv_exists(fs, b"/cabi/sub") == 1
        assert lib.cv_exists(fs, b"/cabi/nope") == 0
        st = CvStatus()
        assert lib.cv_get_status(fs, b"/cabi/sub", ctypes.byref(st)) == 0
        assert st.file_type == 1   # dir

        # streaming write crossing block boundaries (the writer adopts
        # the master's block size from the create reply — 4 MiB in the
        # test conf, so 9 MiB = 3 blocks)
        data = os.urandom(9 << 20)
        w = lib.cv_create(fs, b"/cabi/file.bin", 1)
        assert w > 0, _err(lib)
        pos = 0
        while pos < len(data):
            chunk = data[pos:pos + (2 << 20)]
            n = lib.cv_write(w, chunk, len(chunk))
            assert n == len(chunk), _err(lib)
            pos += len(chunk)
        assert lib.cv_close_writer(w) == 0, _err(lib)

        st = CvStatus()
        assert lib.cv_get_status(fs, b"/cabi/file.bin",
                                 ctypes.byref(st)) == 0
        assert st.length == len(data) and st.is_complete == 1

        # streamed read via the worker data plane
        r = lib.cv_open(fs, b"/cabi/file.bin")
        assert r > 0, _err(lib)
        assert lib.cv_reader_len(r) == len(data)
        out = bytearray()
        buf = ctypes.create_string_buffer(3 << 20)
        while True:
            n = lib.cv_read(r, buf, len(buf))
            assert n >= 0, _err(lib)
            if n == 0:
                break
            out += buf.raw[:n]
        assert bytes(out) == data
        # seek + ranged re-read
        assert lib.cv_seek(r, 5 << 20) == 0
        n = lib.cv_read(r, buf, 1 << 20)
        assert n == 1 << 20
        assert buf.raw[:n] == data[5 << 20:6 << 20]
        assert lib.cv_close_reader(r) == 0

        # list / rename / delete
        names = ctypes.create_string_buffer(4096)
        n = lib.cv_list_status(fs, b"/cabi", names, 4096)
        assert n > 0
        assert sorted(names.value.decode().split("\n")) == \
            ["file.bin", "sub"]
        assert lib.cv_rename(fs, b"/cabi/file.bin", b"/cabi/f2.bin") == 0
        assert lib.cv_exists(fs, b"/cabi/f2.bin") == 1
        assert lib.cv_delete(fs, b"/cabi/f2.bin", 0) == 0
        assert lib.cv_exists(fs, b"/cabi/f2.bin") == 0

        # typed errors surface through cv_last_error
        st = CvStatus()
        rc = lib.cv_get_status(fs, b"/cabi/gone", ctypes.byref(st))
        assert rc < 0
        assert "gone" in _err(lib)
    finally:
        assert lib.cv_fs_close(fs) == 0
