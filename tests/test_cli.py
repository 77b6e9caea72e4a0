"""cv CLI against a MiniCluster."""
import io
import json
import os
import sys

import pytest

from curvine_amd.cli.cv import main as cv_main
from curvine_amd.testing import SyncMiniCluster


@pytest.fixture
def cluster(tmp_path):
    smc = SyncMiniCluster(tmp_dir=str(tmp_path)).start()
    yield smc
    smc.stop()


def cv(cluster, *args, capsys=None):
    master = f"127.0.0.1:{cluster.master.rpc.port}"
    rc = cv_main(["--master", master, *args])
    return rc


def test_cli_fs_ops(cluster, tmp_path, capsys):
    assert cv(cluster, "mkdir", "-p", "/a/b") == 0
    src = tmp_path / "local.bin"
    payload = os.urandom(300_000)
    src.write_bytes(payload)
    assert cv(cluster, "put", str(src), "/a/b/x.bin") == 0
    assert cv(cluster, "ls", "/a/b") == 0
    out = capsys.readouterr().out
    assert "x.bin" in out
    assert cv(cluster, "stat", "/a/b/x.bin") == 0
    st = json.loads(capsys.readouterr().out)
    assert st["length"] == len(payload)
    dst = tmp_path / "back.bin"
    assert cv(cluster, "get", "/a/b/x.bin", str(dst)) == 0
    assert dst.read_bytes() == payload
    assert cv(cluster, "blocks", "/a/b/x.bin") == 0
    assert "block" in capsys.readouterr().out
    assert cv(cluster, "mv", "/a/b/x.bin", "/a/y.bin") == 0
    assert cv(cluster, "du", "/a") == 0
    assert cv(cluster, "df") == 0
    assert cv(cluster, "chmod", "600", "/a/y.bin") == 0
    assert cv(cluster, "free", "/a/y.bin") == 0
    assert cv(cluster, "rm", "-r", "/a") == 0
    assert cv(cluster, "node", "list") == 0
    assert "w1" in capsys.readouterr().out


def test_cli_mount_and_load(cluster, tmp_path, capsys):
    ufs = tmp_path / "u"
    (ufs / "dir").mkdir(parents=True)
    (ufs / "dir" / "a.txt").write_bytes(b"hello ufs")
    assert cv(cluster, "mount", f"file://{ufs}", "/mnt") == 0
    capsys.readouterr()
    assert cv(cluster, "mount-table") == 0
    assert "/mnt" in capsys.readouterr().out
    assert cv(cluster, "load", "/mnt/dir", "--wait") == 0
    assert cv(cluster, "cat", "/mnt/dir/a.txt") == 0
    assert "hello ufs" in capsys.readouterr().out
    assert cv(cluster, "umount", "/mnt") == 0


def test_cli_bench(cluster, capsys):
    assert cv(cluster, "bench", "--num", "20", "--size", str(1 << 20)) == 0
    out = json.loads(capsys.readouterr().out)
    assert out["create_qps"] > 0 and out["read_MBps"] > 0


def test_validate_subcommand(tmp_path):
    from curvine_amd.cli.cv import main
    # valid config
    assert main(["validate", "etc/curvine-cluster.toml"]) == 0
    # broken config: bad tier + even raft peer count
    bad = tmp_path / "bad.toml"
    bad.write_text('[worker]\ndata_dirs = ["[HBM]gpu0"]\n'
                   '[journal]\npeers = ["1@h:1", "2@h:2"]\n')
    assert main(["validate", str(bad)]) == 1


def test_cv_transfers_and_retry(tmp_path, capsys):
    """cv transfers / load-retry surface (ListTransfers/RetryTransfer)."""
    import json as _json
    import os
    import sys
    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

    from curvine_amd.cli.cv import main as cv_main
    from curvine_amd.testing import SyncMiniCluster

    smc = SyncMiniCluster(tmp_dir=str(tmp_path / "cv")).start()
    try:
        master = f"127.0.0.1:{smc.master.rpc.port}"
        # mount a local UFS + load it so a job exists
        src = tmp_path / "src"
        src.mkdir()
        (src / "a.bin").write_bytes(b"x" * 1000)
        assert cv_main(["--master", master, "mount", f"file://{src}",
                        "/warm"]) == 0
        assert cv_main(["--master", master, "load", "/warm"]) == 0
        out = capsys.readouterr().out
        job_id = _json.loads(out.strip().split("\n")[-1])["job_id"]
        assert cv_main(["--master", master, "transfers"]) == 0
        out = capsys.readouterr().out
        assert any(_json.loads(l)["job_id"] == job_id
                   for l in out.strip().split("\n") if l.startswith("{"))
        assert cv_main(["--master", master, "load-retry", job_id]) == 0
        out = capsys.readouterr().out
        assert _json.loads(out.strip().split("\n")[-1])["job_id"] == job_id
    finally:
        smc.stop()
