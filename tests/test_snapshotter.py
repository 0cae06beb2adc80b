"""SnapShotter unit tier (ref lib/snapShotter.js): rotation keeps at
most snapshotNumber AUTO snapshots and never touches operator
snapshots (:206-272); unhealthy sitter skips the snapshot but still
cleans up (:125-145); stuck deletions escalate loudly (:274-405)."""

import asyncio

import pytest

from manatee_amd.common.httpd import HttpServer
from manatee_amd.common.logging import null_logger
from manatee_amd.snapshotter import SnapShotter
from manatee_amd.storage.dirstore import DirStore
from manatee_amd.storage.provider import is_auto_snapshot


def run(coro, timeout=60):
    return asyncio.run(asyncio.wait_for(coro, timeout))


async def _store(tmp_path):
    st = DirStore(str(tmp_path / "store"), log=null_logger())
    await st.ensure()
    with open(st.mountpoint() + "/data.bin", "w") as f:
        f.write("payload")
    return st


def test_auto_snapshot_name_discipline():
    assert is_auto_snapshot("1426541061000")
    assert not is_auto_snapshot("142654106100")      # 12 digits
    assert not is_auto_snapshot("14265410610001")    # 14 digits
    assert not is_auto_snapshot("operator-backup")
    assert not is_auto_snapshot("1426541061000x")


def test_rotation_keeps_limit_and_operator_snapshots(tmp_path):
    async def go():
        st = await _store(tmp_path)
        snap = SnapShotter(st, snapshot_number=3, log=null_logger())
        # operator snapshot (non-13-digit) must survive rotation forever
        await st.snapshot("operator-keepme")
        for i in range(6):
            await st.snapshot("%013d" % (1000000000000 + i))
        await snap.run_once()    # takes one more + cleans up
        snaps = await st.list_snapshots()
        autos = [s for s in snaps if is_auto_snapshot(s)]
        assert len(autos) == 3, snaps
        assert "operator-keepme" in snaps
        # the SURVIVORS are the newest ones
        assert autos == sorted(autos)
        assert autos[-1] > "%013d" % (1000000000000 + 5)
        assert snap.stats["deleted"] >= 4
    run(go())


def test_unhealthy_sitter_skips_snapshot_but_cleans(tmp_path):
    async def go():
        st = await _store(tmp_path)
        # a /ping endpoint that reports 503 (db unhealthy)
        httpd = HttpServer("127.0.0.1", 0, log=null_logger())

        async def ping(*a, **k):
            return 503, {"healthy": False}
        httpd.route("GET", "ping", ping)
        await httpd.start()
        try:
            url = "http://127.0.0.1:%d/ping" % httpd.port
            for i in range(5):
                await st.snapshot("%013d" % (1000000000000 + i))
            snap = SnapShotter(st, snapshot_number=2, health_url=url,
                               log=null_logger())
            name = await snap.run_once()
            assert name is None
            assert snap.stats["skipped_unhealthy"] == 1
            autos = [s for s in await st.list_snapshots()
                     if is_auto_snapshot(s)]
            assert len(autos) == 2     # cleanup still ran (ref :125-145)
        finally:
            await httpd.stop()
    run(go())


def test_healthy_sitter_takes_snapshot(tmp_path):
    async def go():
        st = await _store(tmp_path)
        httpd = HttpServer("127.0.0.1", 0, log=null_logger())

        async def ping(*a, **k):
            return 200, {"healthy": True}
        httpd.route("GET", "ping", ping)
        await httpd.start()
        try:
            url = "http://127.0.0.1:%d/ping" % httpd.port
            snap = SnapShotter(st, snapshot_number=5, health_url=url,
                               log=null_logger())
            name = await snap.run_once()
            assert name is not None and is_auto_snapshot(name)
            assert snap.stats["snapshots"] == 1
        finally:
            await httpd.stop()
    run(go())


def test_unreachable_health_url_counts_as_unhealthy(tmp_path):
    async def go():
        st = await _store(tmp_path)
        snap = SnapShotter(st, snapshot_number=5,
                           health_url="http://127.0.0.1:1/ping",
                           log=null_logger())
        assert await snap.run_once() is None
        assert snap.stats["skipped_unhealthy"] == 1
    run(go())


def test_stuck_deletion_escalates(tmp_path):
    """Failed cleanup must count consecutive failures and escalate to
    fatal after 5 (ref stuck-deletion alarm :274-405) — and recover the
    counter once a pass succeeds."""
    async def go():
        st = await _store(tmp_path)
        for i in range(4):
            await st.snapshot("%013d" % (1000000000000 + i))
        snap = SnapShotter(st, snapshot_number=1, log=null_logger())

        fails = {"n": 0}
        orig = st.destroy_snapshot

        async def broken(name):
            fails["n"] += 1
            raise RuntimeError("EBUSY: dataset is busy")

        st.destroy_snapshot = broken
        levels = []
        snap.log.fatal = lambda *a, **k: levels.append("fatal")
        snap.log.error = lambda *a, **k: levels.append("error")
        for _ in range(7):
            await snap._cleanup()
        assert fails["n"] == 7
        assert "fatal" in levels            # escalated past 5 failures
        assert levels[:5] == ["error"] * 5  # ... but not before

        st.destroy_snapshot = orig
        await snap._cleanup()
        assert snap._cleanup_failures == 0  # recovery resets the counter
        autos = [s for s in await st.list_snapshots()
                 if is_auto_snapshot(s)]
        assert len(autos) == 1
    run(go())
