"""FUSE daemon hot upgrade: a successor daemon adopts the live mount
(session fd + open-handle state); open file descriptors in applications
keep working across the swap."""
import os
import subprocess
import sys
import time

import pytest

requires_fuse = pytest.mark.skipif(
    not os.path.exists("/dev/fuse") or os.geteuid() != 0,
    reason="needs /dev/fuse and root")

pytestmark = requires_fuse


def spawn_daemon(mnt, master, takeover=False):
    cmd = [sys.executable, "-m", "curvine_amd.fuse", "--mnt", mnt,
           "--master", master, "--log-level", "WARNING"]
    if takeover:
        cmd.append("--takeover")
    proc = subprocess.Popen(cmd, stdout=subprocess.PIPE,
                            stderr=None, text=True,
                            cwd=os.path.dirname(os.path.dirname(
                                os.path.abspath(__file__))))
    line = proc.stdout.readline()
    assert line.startswith("READY"), f"daemon failed: {line!r}"
    return proc


def test_hot_upgrade_preserves_open_fds(tmp_path):
    from curvine_amd.testing import SyncMiniCluster

    smc = SyncMiniCluster(tmp_dir=str(tmp_path / "cv")).start()
    master = f"127.0.0.1:{smc.master.rpc.port}"
    mnt = f"/tmp/cv-upgrade-test-{os.getpid()}"
    a = spawn_daemon(mnt, master)
    try:
        data = os.urandom(4 << 20)
        with open(f"{mnt}/keep.bin", "wb") as f:
            f.write(data)
        # application opens the file and reads half
        f = open(f"{mnt}/keep.bin", "rb", buffering=0)
        first = f.read(1 << 20)
        assert first == data[:1 << 20]

        # hot upgrade: daemon B adopts the session, daemon A exits
        b = spawn_daemon(mnt, master, takeover=True)
        a.terminate()
        a.wait(timeout=10)

        # the SAME open fd keeps reading (served by B now)
        rest = f.read()
        assert first + rest == data
        f.close()

        # new operations work through B
        with open(f"{mnt}/after.bin", "wb") as f2:
            f2.write(b"post-upgrade")
        assert open(f"{mnt}/after.bin", "rb").read() == b"post-upgrade"
        assert sorted(os.listdir(mnt)) == ["after.bin", "keep.bin"]

        b.terminate()
        b.wait(timeout=10)
    finally:
        for p in (a, b if "b" in dir() else None):
            try:
                if p:
                    p.kill()
            except Exception:  # noqa: BLE001
                pass
        subprocess.run(["umount", "-l", mnt], capture_output=True)
        smc.stop()
        try:
            os.rmdir(mnt)
        except OSError:
            pass
