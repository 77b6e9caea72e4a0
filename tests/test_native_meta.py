"""Native metadata frontend (csrc/meta_server.cpp): consistency of the C++
mirror against the Python inode tree, served-vs-forwarded accounting, and
the asyncio fallback path."""
import asyncio
import random
import tempfile

import pytest


@pytest.fixture
def cluster(tmp_path):
    from curvine_amd.testing import SyncMiniCluster

    smc = SyncMiniCluster(tmp_dir=str(tmp_path / "cv")).start()
    yield smc
    smc.stop()


def test_mirror_consistency_fuzz(cluster):
    """Random mutation storm, then every path's native-served status must
    equal the Python tree's status (field-for-field, via the real RPC)."""
    from curvine_amd.client.filesystem import SyncFs

    sf = SyncFs(cluster.client_conf())
    master = cluster.master
    rng = random.Random(7)

    dirs = ["/"]
    files = []
    for i in range(300):
        op = rng.random()
        try:
            if op < 0.25:
                parent = rng.choice(dirs)
                p = f"{parent.rstrip('/')}/d{i}"
                sf.mkdir(p, create_parents=True)
                dirs.append(p)
            elif op < 0.55:
                parent = rng.choice(dirs)
                p = f"{parent.rstrip('/')}/f{i}.bin"
                sf.write_file(p, bytes(rng.randrange(256)
                                       for _ in range(rng.randrange(1, 64))))
                files.append(p)
            elif op < 0.65 and files:
                p = rng.choice(files)
                sf.set_attr(p, mode=0o640, ttl_ms=60_000)
            elif op < 0.75 and files:
                src = rng.choice(files)
                dst = src + ".mv"
                sf.rename(src, dst)
                files.remove(src)
                files.append(dst)
            elif op < 0.82 and files:
                src = rng.choice(files)
                dst = src + ".ln"
                sf.link(src, dst)
                files.append(dst)
            elif op < 0.88 and files:
                p = rng.choice(files)
                sf.delete(p)
                files.remove(p)
            elif op < 0.94:
                parent = rng.choice(dirs)
                sf.symlink(f"{parent.rstrip('/')}/s{i}", "/target")
            elif len(dirs) > 1:
                p = rng.choice(dirs[1:])
                try:
                    sf.delete(p, recursive=True)
                    pref = p.rstrip("/") + "/"
                    dirs[:] = [d for d in dirs if d != p
                               and not d.startswith(pref)]
                    files[:] = [f for f in files if not f.startswith(pref)]
                except Exception:
                    pass
        except Exception:
            pass   # racing names etc. — irrelevant to the mirror contract

    fs_dir = master.fs.fs_dir
    stats0 = master.native_meta.stats()
    assert stats0["nodes"] == len(fs_dir.inodes)

    # walk every LIVE dentry edge from the root (hardlinks give one inode
    # several paths; path_of only knows the primary one)
    live_paths = []
    dir_paths = ["/"]
    stack = [(fs_dir.inodes[1], "")]
    while stack:
        node, base = stack.pop()
        for name, cid in node.children.items():
            child = fs_dir.inodes[cid]
            p = f"{base}/{name}"
            live_paths.append(p)
            if child.is_dir:
                dir_paths.append(p)
                stack.append((child, p))

    # every live path: the RPC-served status == Python-computed status
    checked = 0
    for path in live_paths:
        expect = master.fs.file_status(path).to_dict()
        got = sf.file_status(path).to_dict()
        assert got == expect, f"mismatch at {path}"
        checked += 1
    assert checked > 50

    # list_status on every dir matches
    for path in dir_paths:
        expect = [s.to_dict() for s in master.fs.list_status(path)]
        got = [s.to_dict() for s in sf.list_status(path)]
        assert got == expect, f"ls mismatch at {path}"

    stats = master.native_meta.stats()
    assert stats["served_status"] >= checked   # reads were served natively
    assert stats["forwarded"] > 0              # mutations were forwarded
    sf.shutdown()


def test_exists_and_errors_native(cluster):
    from curvine_amd import errors as err
    from curvine_amd.client.filesystem import SyncFs

    sf = SyncFs(cluster.client_conf())
    sf.mkdir("/ex/d", create_parents=True)
    sf.write_file("/ex/f", b"abc")
    assert sf.exists("/ex/f") and sf.exists("/ex/d") and sf.exists("/")
    assert not sf.exists("/ex/missing")
    assert not sf.exists("/ex/f/below")       # walk through a file
    with pytest.raises(err.FileNotFound):
        sf.file_status("/ex/missing")
    with pytest.raises(err.FileNotFound):
        sf.list_status("/ex/missing")
    # single-file list
    ls = sf.list_status("/ex/f")
    assert len(ls) == 1 and ls[0].path == "/ex/f" and ls[0].length == 3
    st = cluster.master.native_meta.stats()
    assert st["served_exists"] >= 5 and st["served_notfound"] >= 2
    sf.shutdown()


def test_asyncio_fallback_path(tmp_path):
    """native_meta=False: the asyncio RpcServer still serves everything."""
    from curvine_amd.client.filesystem import SyncFs
    from curvine_amd.testing import SyncMiniCluster, test_conf

    conf = test_conf(str(tmp_path / "cv"))
    conf.master.native_meta = False
    smc = SyncMiniCluster(conf=conf, tmp_dir=str(tmp_path / "cv")).start()
    try:
        assert smc.master.native_meta is None
        sf = SyncFs(smc.client_conf())
        sf.mkdir("/py/d", create_parents=True)
        sf.write_file("/py/f", b"hello")
        assert sf.file_status("/py/f").length == 5
        assert [s.name for s in sf.list_status("/py")] == ["d", "f"]
        assert sf.exists("/py") and not sf.exists("/py/x")
        sf.shutdown()
    finally:
        smc.stop()


def test_xattr_and_rename_visibility(cluster):
    """Mutations that touch xattrs/rename must be visible in the very next
    native read (synchronous mirror contract)."""
    from curvine_amd.client.filesystem import SyncFs

    sf = SyncFs(cluster.client_conf())
    sf.write_file("/xv/f", b"z" * 10)
    sf.set_attr("/xv/f", xattrs={"user.k": b"v1"})
    st = sf.file_status("/xv/f")
    assert st.xattrs.get("user.k") == b"v1"
    sf.rename("/xv/f", "/xv/g")
    assert sf.file_status("/xv/g").name == "g"
    assert not sf.exists("/xv/f")
    names = [s.name for s in sf.list_status("/xv")]
    assert names == ["g"]
    sf.shutdown()


def test_encoding_boundaries(cluster):
    """Long (str8/str16) and unicode names, deep paths, >16-entry dirs —
    the C++ msgpack emitters must agree with Python's decoder."""
    from curvine_amd.client.filesystem import SyncFs

    sf = SyncFs(cluster.client_conf())
    long_name = "n" * 300                      # str16 path/name on the wire
    uni = "δοκιμή-试验-🚀"
    deep = "/b/" + "/".join(f"lvl{i}" for i in range(20))
    sf.mkdir(deep, create_parents=True)
    sf.write_file(f"/b/{long_name}", b"L")
    sf.write_file(f"/b/{uni}", b"U" * 100)
    for i in range(25):                        # array16 list in ListStatus
        sf.write_file(f"/b/many/f{i:02d}", b"x")
    st = sf.file_status(f"/b/{long_name}")
    assert st.name == long_name and st.length == 1
    st = sf.file_status(f"/b/{uni}")
    assert st.name == uni and st.length == 100
    assert sf.file_status(deep).is_dir
    names = [s.name for s in sf.list_status("/b/many")]
    assert names == sorted(f"f{i:02d}" for i in range(25))
    # open with blocks through the native path
    sf.write_file("/b/blocky", b"z" * (3 << 20))
    assert sf.read_file("/b/blocky") == b"z" * (3 << 20)
    stats = cluster.master.native_meta.stats()
    assert stats["served_status"] >= 3
    sf.shutdown()


def test_garbage_frames_do_not_wedge(cluster):
    """Malformed input on the native meta port: oversized lengths close
    the connection; valid clients keep working throughout."""
    import socket
    import struct

    from curvine_amd.client.filesystem import SyncFs

    port = cluster.master.rpc.port
    sf = SyncFs(cluster.client_conf())
    sf.mkdir("/gz", create_parents=True)

    # oversized header length -> server closes the conn
    s1 = socket.create_connection(("127.0.0.1", port), timeout=5)
    s1.sendall(struct.pack(">IIBBQI", 1 << 30, 0, 7, 0, 1, 0))
    s1.settimeout(5)
    assert s1.recv(1) == b""          # closed on us
    s1.close()

    # truncated frame then disconnect mid-header
    s2 = socket.create_connection(("127.0.0.1", port), timeout=5)
    s2.sendall(b"\x00\x00\x00\x10")
    s2.close()

    # undecodable msgpack header on a hot code -> forwarded to Python,
    # which answers with an error reply rather than hanging
    s3 = socket.create_connection(("127.0.0.1", port), timeout=5)
    junk = b"\xc1\xff\xfe"            # 0xc1 is an invalid msgpack byte
    s3.sendall(struct.pack(">IIBBQI", len(junk), 0, 7, 0, 42, 0) + junk)
    s3.settimeout(10)
    hdr = s3.recv(22)
    assert len(hdr) == 22
    _hl, _dl, code, status, req_id, _seq = struct.unpack(">IIBBQI", hdr)
    assert req_id == 42 and (status >> 4) == 5   # error reply
    s3.close()

    # the real client is unaffected
    assert sf.exists("/gz")
    assert sf.file_status("/gz").is_dir
    sf.shutdown()
