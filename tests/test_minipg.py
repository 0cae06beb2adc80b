"""engine=postgres integration tier: the PostgreSQL management path
(db/postgres.py + db/manager.py) driving real separate ``initdb`` /
``postgres`` binaries (minipg) with PostgreSQL's process conventions
and wire protocol — conf regeneration, standby.signal, promote trigger
files, SIGHUP reloads, libpq probes, pg_stat_replication write gating
(ref lib/postgresMgr.js; scenarios mirror test/integ.test.js).

No real PostgreSQL distribution exists in this environment (no network,
no packages), so minipg is the engine those code paths are proven
against; see manatee_amd/db/minipg/__init__.py for the fidelity
contract.
"""

import asyncio
import os
import time

import pytest

from manatee_amd.db.pgwire import PgClient, PgError
from manatee_amd.tools.devcluster import DevCluster


def run(coro, timeout=240):
    return asyncio.run(asyncio.wait_for(coro, timeout))


async def _formed(tmp_path, shard, **kw):
    c = DevCluster(str(tmp_path / "c"), n_peers=3, shard_name=shard,
                   engine="postgres", run_snapshotter=False, **kw)
    await c.start()
    s = await c.wait_cluster(
        lambda s: s.get("sync") and len(s.get("async", [])) == 1,
        timeout_s=120, what="3-peer formation (engine=postgres)")
    await c.wait_writable(timeout_s=120)
    return c, s


def test_formation_writes_replication_chain(tmp_path):
    async def go():
        c, s = await _formed(tmp_path, "1.pgform")
        try:
            prim = c.peer_by_id(s["primary"]["id"])
            cli = prim.db_client()
            for i in range(25):
                await cli.put("row%d" % i, {"i": i})
            # batch (multi-statement simple query)
            await cli.put_many([("b%d" % i, i) for i in range(20)])
            assert await cli.count(prefix="row") == 25
            assert await cli.count(prefix="b") == 20
            # pg_stat_replication shows the sync, caught up
            st = await cli.status()
            row = st["replication"][0]
            assert row["sync_state"] == "sync"
            assert row["application_name"] == s["sync"]["id"]
            assert row["sent_lsn"] == row["flush_lsn"]
            await cli.close()
            # replicated through the chain to the async (hot standby read)
            asy = c.peer_by_id(s["async"][0]["id"])
            acli = asy.db_client()
            deadline = time.monotonic() + 30
            while True:
                try:
                    if await acli.get("row24") == {"i": 24}:
                        break
                except Exception:
                    pass
                assert time.monotonic() < deadline, "async never caught up"
                await asyncio.sleep(0.1)
            # standby rejects writes with the hot-standby error
            with pytest.raises(Exception) as ei:
                await acli.put("nope", 1)
            assert "read-only" in str(ei.value)
            await acli.close()
        finally:
            c.stop()
    run(go())


def test_primary_death_zero_loss_and_rebuild(tmp_path):
    async def go():
        c, s = await _formed(tmp_path, "1.pgfail")
        try:
            prim = c.peer_by_id(s["primary"]["id"])
            cli = prim.db_client()
            for i in range(40):
                await cli.put("w%d" % i, i)
            await cli.close()
            prim.kill9()
            s2 = await c.wait_cluster(
                lambda st: st["generation"] > s["generation"] and
                st["primary"]["id"] == s["sync"]["id"],
                timeout_s=60, what="sync takeover")
            newp = await c.wait_writable(timeout_s=60)
            cli = newp.db_client()
            assert await cli.count(prefix="w") == 40
            for i in range(40):
                assert await cli.get("w%d" % i) == i
            # new writes flow on the new timeline
            await cli.put("after-failover", True)
            await cli.close()
            assert any(d["id"] == prim.id for d in s2.get("deposed", []))
            await c.rebuild_peer(prim)
            await c.wait_writable(timeout_s=60)
        finally:
            c.stop()
    run(go())


def test_db_child_kill_is_restarted(tmp_path):
    """SIGKILL only the postgres child: the sitter must notice the
    unexpected exit and restart it; writes resume on the same peer
    (ref unexpected-exit fatal error, lib/postgresMgr.js:1711-1753)."""
    async def go():
        c, s = await _formed(tmp_path, "1.pgdbkill")
        try:
            prim = c.peer_by_id(s["primary"]["id"])
            cli = prim.db_client()
            await cli.put("before", 1)
            await cli.close()
            prim.kill_db_only()
            newp = await c.wait_writable(timeout_s=60)
            assert newp.id == prim.id, "peer should keep its role"
            cli = newp.db_client()
            assert await cli.get("before") == 1
            await cli.put("after", 2)
            await cli.close()
        finally:
            c.stop()
    run(go())


def test_libpq_protocol_surface(tmp_path):
    """Direct libpq conformance against a live minipg primary: the
    introspection queries the manager/adm run, error fields, and
    multi-statement abort semantics."""
    async def go():
        c, s = await _formed(tmp_path, "1.pgwire")
        try:
            prim = c.peer_by_id(s["primary"]["id"])
            cli = PgClient(prim.ip, prim.pg_port, "postgres")
            await cli.connect()
            assert cli.parameters.get("server_version", "").startswith("12")

            r = await cli.query("SELECT pg_is_in_recovery() as r;")
            assert r.rows[0][0] == "f"
            r = await cli.query("SELECT pg_current_wal_lsn() as loc;")
            assert "/" in r.rows[0][0]
            r = await cli.query("SELECT current_time;")
            assert r.rows and r.columns == ["current_time"]

            # error surface: code + message fields, connection survives
            with pytest.raises(PgError) as ei:
                await cli.query("SELECT frobnicate();")
            assert ei.value.code == "42601"
            r = await cli.query("SELECT pg_is_in_recovery() as r;")
            assert r.rows[0][0] == "f"

            # multi-statement: later statements abort after an error
            with pytest.raises(PgError):
                await cli.query(
                    "INSERT INTO kv (k, v) VALUES ('m1', '1');"
                    "SELECT broken();"
                    "INSERT INTO kv (k, v) VALUES ('m2', '2')")
            r = await cli.query("SELECT v FROM kv WHERE k = 'm1'")
            assert r.rows
            r = await cli.query("SELECT v FROM kv WHERE k = 'm2'")
            assert not r.rows

            # standby answers the recovery probes
            sync = c.peer_by_id(s["sync"]["id"])
            scli = PgClient(sync.ip, sync.pg_port, "postgres")
            await scli.connect()
            r = await scli.query("SELECT pg_is_in_recovery() as r;")
            assert r.rows[0][0] == "t"
            r = await scli.query("SELECT pg_last_wal_replay_lsn() as loc;")
            assert "/" in r.rows[0][0]
            with pytest.raises(PgError) as ei:
                await scli.query("SELECT pg_current_wal_lsn() as loc;")
            assert ei.value.code == "55000"
            await scli.close()
            await cli.close()
        finally:
            c.stop()
    run(go())


def test_adm_against_live_pg_cluster(tmp_path):
    """manatee-adm's live path (pg probes over libpq) against the
    engine=postgres shard: pg-status rows and verify-clean."""
    async def go():
        c, s = await _formed(tmp_path, "1.pgadm")
        try:
            import manatee_amd.adm.details as det
            from manatee_amd.adm import core as adm
            # adm's db probes speak the waldb JSON status protocol, which
            # minipg multiplexes on the postgres port — same surface
            zk = await adm.create_zk_client(c.zk_conn_str)
            try:
                deadline = time.monotonic() + 30
                while True:
                    cd = await det.load_cluster_details(
                        zk, c.shard_path, zk_conn=c.zk_conn_str)
                    if not cd.errors and not cd.warnings:
                        break
                    assert time.monotonic() < deadline, \
                        (cd.errors, cd.warnings)
                    await asyncio.sleep(0.5)
                rows = cd.table_rows(det.STATUS_COLUMNS)
                assert len(rows) == 3
                assert rows[0][2] == "ok"      # pg-online on the primary
            finally:
                await zk.close()
        finally:
            c.stop()
    run(go())


def test_pg96_recovery_conf_mode(tmp_path):
    """The pre-12 path: recovery.conf with standby_mode=on +
    trigger_file, xlog/location query spellings, *_location columns in
    pg_stat_replication (ref resolveWalTranslations :649-677,
    _updateUpstreamConf :2188-2274)."""
    async def go():
        c = DevCluster(str(tmp_path / "c"), n_peers=3,
                       shard_name="1.pg96", engine="postgres",
                       pg_version="9.6", run_snapshotter=False)
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=120, what="formation on 9.6")
            prim = await c.wait_writable(timeout_s=120)

            # the sync runs with a recovery.conf, not standby.signal
            syncp = c.peer_by_id(s["sync"]["id"])
            data = os.path.join(syncp.store_dir, "live", "data")
            assert os.path.exists(os.path.join(data, "recovery.conf"))
            assert not os.path.exists(os.path.join(data, "standby.signal"))
            with open(os.path.join(data, "PG_VERSION")) as f:
                assert f.read().strip() == "9.6"

            # 9.x query spellings over libpq
            cli = PgClient(prim.ip, prim.pg_port, "postgres")
            await cli.connect()
            r = await cli.query(
                "SELECT pg_current_xlog_location() as loc;")
            assert "/" in r.rows[0][0]
            r = await cli.query("SELECT * FROM pg_stat_replication;")
            assert "sent_location" in r.columns
            assert "sent_lsn" not in r.columns
            await cli.close()

            # failover works through the recovery.conf trigger_file path
            kcli = prim.db_client()
            for i in range(20):
                await kcli.put("v%d" % i, i)
            await kcli.close()
            prim.kill9()
            await c.wait_cluster(
                lambda st: st["generation"] > s["generation"],
                timeout_s=60, what="9.6 takeover")
            newp = await c.wait_writable(timeout_s=60)
            ncli = newp.db_client()
            assert await ncli.count(prefix="v") == 20
            await ncli.close()
        finally:
            c.stop()
    run(go())


def test_diverged_standby_reports_via_wal_receiver(tmp_path):
    """A standby whose history does not match its upstream (separate
    initdb → different system identity) must report
    pg_stat_wal_receiver.status = 'diverged' — the signal the manager's
    restore-on-divergence path consumes (ref standby failure ⇒ full
    restore, lib/postgresMgr.js:1339-1373)."""
    import subprocess
    import sys as _sys
    from manatee_amd.common import confparser

    async def go():
        base = tmp_path
        bindir = None
        c = DevCluster(str(base / "scratch"), n_peers=0,
                       engine="postgres", run_snapshotter=False)
        bindir = os.path.join(c.pg_base_dir, "12.0", "bin")

        def initdb(d):
            r = subprocess.run([os.path.join(bindir, "initdb"), "-D", d],
                               capture_output=True, text=True)
            assert r.returncode == 0, r.stderr

        prim_dir = str(base / "prim")
        stby_dir = str(base / "stby")
        initdb(prim_dir)
        initdb(stby_dir)       # separate identity ⇒ diverged from prim
        pport, sport = c.peers or None, None  # unused; pick free ports
        import socket

        def free():
            s = socket.socket()
            s.bind(("127.0.0.1", 0))
            p = s.getsockname()[1]
            s.close()
            return p
        pport, sport = free(), free()
        confparser.write(os.path.join(prim_dir, "postgresql.conf"),
                         {"listen_addresses": "'127.0.0.1'",
                          "port": str(pport)})
        confparser.write(os.path.join(stby_dir, "postgresql.conf"),
                         {"listen_addresses": "'127.0.0.1'",
                          "port": str(sport),
                          "primary_conninfo":
                          "'host=127.0.0.1 port=%d user=postgres "
                          "application_name=stby'" % pport})
        open(os.path.join(stby_dir, "standby.signal"), "w").close()
        procs = []
        try:
            for d in (prim_dir, stby_dir):
                procs.append(subprocess.Popen(
                    [os.path.join(bindir, "postgres"), "-D", d],
                    stdout=subprocess.DEVNULL,
                    stderr=subprocess.DEVNULL))
            scli = PgClient("127.0.0.1", sport, "postgres")
            deadline = time.monotonic() + 20
            status = None
            while time.monotonic() < deadline:
                try:
                    if not scli.connected:
                        await scli.connect()
                    r = await scli.query(
                        "SELECT status FROM pg_stat_wal_receiver;")
                    if r.rows and r.rows[0][0] == "diverged":
                        status = "diverged"
                        break
                except Exception:
                    await scli.close()
                await asyncio.sleep(0.2)
            assert status == "diverged", "receiver never reported diverged"
            await scli.close()
        finally:
            for p in procs:
                p.kill()
    run(go())
