"""Peer version-compatibility policy (curvine_amd/compat.py): verdicts,
diagnose/enforce semantics, and the wired-in master-side checks.

Models /root/reference/crates/common/curvine-model/src/compatibility.rs
behavior: lenient by default, blocked versions always reject, enforce
rejects legacy/unknown/old peers."""
import asyncio

import pytest

from curvine_amd.compat import (MODE_DIAGNOSE, MODE_ENFORCE,
                                CompatibilityPolicy, PeerWarnDedup,
                                component_info, parse_mode, parse_version)


def _run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


def test_parse_lenient():
    assert parse_version("1.2.3") == (1, 2, 3)
    assert parse_version("v2.0.0-rc1") == (2, 0, 0)
    assert parse_version("1.2.3+build7") == (1, 2, 3)
    assert parse_version("garbage") is None
    assert parse_version("") is None
    # unknown modes fall back to the lenient default, never fail closed
    assert parse_mode("enforce") == MODE_ENFORCE
    assert parse_mode("ENFORCE") == MODE_ENFORCE
    assert parse_mode("strict??") == MODE_DIAGNOSE
    assert parse_mode("") == MODE_DIAGNOSE


def test_verdicts_and_modes():
    pol = CompatibilityPolicy(min_client_version="0.1.0",
                              blocked_versions=["0.0.9"])
    ok = pol.check_client(component_info("client"))
    assert ok.compatible and not ok.rejects(MODE_ENFORCE)

    # legacy peer (no info): allowed in diagnose, rejected in enforce
    legacy = pol.check_client(None)
    assert legacy.kind == "missing_info"
    assert not legacy.rejects(MODE_DIAGNOSE) and legacy.rejects(MODE_ENFORCE)

    # blocked version: rejected regardless of mode (operator backstop)
    blocked = pol.check_client({"release_version": "0.0.9",
                                "protocol_version": 1})
    assert blocked.kind == "blocked"
    assert blocked.rejects(MODE_DIAGNOSE) and blocked.rejects(MODE_ENFORCE)

    # protocol outside the supported range
    proto = pol.check_client({"release_version": "0.1.0",
                              "protocol_version": 99})
    assert proto.kind == "protocol_mismatch"

    # too old vs the configured minimum
    old = pol.check_client({"release_version": "0.0.1",
                            "protocol_version": 1})
    assert old.kind == "version_too_old"
    assert not old.rejects(MODE_DIAGNOSE) and old.rejects(MODE_ENFORCE)

    # unparseable version cannot bypass a configured minimum
    unk = pol.check_client({"release_version": "not-a-version",
                            "protocol_version": 1})
    assert unk.kind == "version_unknown" and unk.rejects(MODE_ENFORCE)


def test_should_evaluate_skips_hot_path():
    # diagnose + nothing configured + legacy peer: evaluation could only
    # warn about something diagnose always allows — skip it
    lenient = CompatibilityPolicy()
    assert not lenient.should_evaluate(False)
    assert lenient.should_evaluate(True)
    assert CompatibilityPolicy(mode=MODE_ENFORCE).should_evaluate(False)
    assert CompatibilityPolicy(
        min_worker_version="1.0.0").should_evaluate(False)


def test_warn_dedup(caplog):
    import logging
    dedup = PeerWarnDedup("worker")
    pol = CompatibilityPolicy(min_worker_version="9.0.0")
    bad = pol.check_worker({"release_version": "0.1.0",
                            "protocol_version": 1})
    with caplog.at_level(logging.WARNING, logger="curvine.compat"):
        dedup.warn(7, bad)
        dedup.warn(7, bad)          # same verdict: no second warning
    assert sum("incompatible worker 7" in r.getMessage()
               for r in caplog.records) == 1


def test_enforce_rejects_cluster_peers(tmp_path):
    """A master in enforce mode with min bounds above our version rejects
    both the worker heartbeat and the client handshake with
    IncompatibleVersion; diagnose (default) lets the same peers in."""
    from curvine_amd import errors as err
    from curvine_amd.testing import MiniCluster, test_conf

    async def enforce():
        conf = test_conf(str(tmp_path / "a"))
        conf.compatibility.mode = "enforce"
        conf.compatibility.min_client_version = "99.0.0"
        conf.compatibility.min_worker_version = "99.0.0"
        mc = MiniCluster(conf=conf, tmp_dir=str(tmp_path / "a"))
        try:
            with pytest.raises(err.IncompatibleVersion):
                await mc.start()        # worker heartbeat is rejected
        finally:
            await mc.stop()
        # a direct client handshake is rejected too
        from curvine_amd.master.server import Master
        m = await Master(conf).start()
        try:
            from curvine_amd.client.fs_client import FsClient
            conf.client.master_addrs = [
                f"127.0.0.1:{m.rpc.port}"]
            cl = FsClient(conf)
            with pytest.raises(err.IncompatibleVersion):
                await cl.get_master_info()
            await cl.close()
        finally:
            await m.stop()

    _run(enforce())

    async def diagnose():
        conf = test_conf(str(tmp_path / "b"))
        conf.compatibility.min_client_version = "99.0.0"   # mode=diagnose
        async with MiniCluster(conf=conf,
                               tmp_dir=str(tmp_path / "b")) as mc:
            from curvine_amd.client.fs_client import FsClient
            conf.client.master_addrs = [
                f"127.0.0.1:{mc.master.rpc.port}"]
            cl = FsClient(conf)
            info = await cl.get_master_info()
            # master advertises its own version report back
            assert info["component_info"]["component"] == "master"
            assert len(info["live_workers"]) >= 1
            assert info["live_workers"][0]["component_info"][
                "component"] == "worker"
            await cl.close()

    _run(diagnose())
