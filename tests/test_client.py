"""ManateeClient (node-manatee equivalent) tests: topology events from
the shared ZK state node, dedup, late shard setup, ZK restart survival,
and a live failover observed end-to-end."""

import asyncio
import json

import pytest

from manatee_amd.client import ManateeClient, topology_from_state
from manatee_amd.coord.zkclient import ZkClient
from manatee_amd.coord.zkserver import ZkServer


def run(coro, timeout=120):
    return asyncio.run(asyncio.wait_for(coro, timeout))


def ident(n):
    return {"id": "10.0.0.%d:5432:5434" % n, "ip": "10.0.0.%d" % n,
            "zoneId": "z%d" % n,
            "pgUrl": "waldb://10.0.0.%d:5432" % n,
            "backupUrl": "http://10.0.0.%d:5434" % n}


def state(gen, p, s, asyncs=()):
    return {"generation": gen, "primary": ident(p),
            "sync": ident(s) if s else None,
            "async": [ident(a) for a in asyncs], "deposed": [],
            "initWal": "0/00000000"}


def test_topology_from_state_ordering():
    t = topology_from_state(state(3, 1, 2, [3, 4]))
    assert t["urls"] == ["waldb://10.0.0.1:5432", "waldb://10.0.0.2:5432",
                         "waldb://10.0.0.3:5432", "waldb://10.0.0.4:5432"]
    assert t["primary"] == "waldb://10.0.0.1:5432"
    assert t["generation"] == 3
    # singleton: no sync
    t = topology_from_state({"generation": 1, "primary": ident(1),
                             "sync": None, "async": [],
                             "oneNodeWriteMode": True})
    assert t["urls"] == ["waldb://10.0.0.1:5432"]
    assert t["oneNodeWriteMode"]


def test_client_topology_events_and_dedup(tmp_path):
    async def go():
        srv = ZkServer(journal_path=str(tmp_path / "zk.jsonl"))
        await srv.start()
        cli = ZkClient(srv.conn_str)
        await cli.connect()
        events = []
        mc = ManateeClient(srv.conn_str, "1.cli", session_timeout_ms=4000)
        mc.on("topology", events.append)
        try:
            # client starts BEFORE the shard exists: start() must block
            start_task = asyncio.get_running_loop().create_task(
                mc.start(timeout_s=30))
            await asyncio.sleep(0.2)
            assert not start_task.done()

            await cli.mkdirp("/manatee/1.cli")
            s1 = state(1, 1, 2, [3])
            await cli.create("/manatee/1.cli/state",
                             json.dumps(s1).encode())
            await start_task
            assert len(events) == 1
            assert events[0]["generation"] == 1

            # identical rewrite must be deduped
            await cli.set_data("/manatee/1.cli/state",
                               json.dumps(s1).encode())
            await asyncio.sleep(0.3)
            assert len(events) == 1

            # real change fires
            s2 = state(2, 2, 3, [])
            await cli.set_data("/manatee/1.cli/state",
                               json.dumps(s2).encode())
            deadline = asyncio.get_running_loop().time() + 10
            while len(events) < 2:
                assert asyncio.get_running_loop().time() < deadline
                await asyncio.sleep(0.02)
            assert events[1]["primary"] == "waldb://10.0.0.2:5432"

            # ZK restart (journaled state survives): client reconnects and
            # sees the next change
            await srv.stop()
            await asyncio.sleep(0.3)
            srv2 = ZkServer(host=srv.host, port=srv.port,
                            journal_path=str(tmp_path / "zk.jsonl"))
            await srv2.start()
            cli2 = ZkClient(srv2.conn_str)
            await cli2.connect()
            s3 = state(3, 2, 1, [3])
            await cli2.set_data("/manatee/1.cli/state",
                                json.dumps(s3).encode())
            deadline = asyncio.get_running_loop().time() + 30
            while len(events) < 3:
                assert asyncio.get_running_loop().time() < deadline, \
                    "no topology after ZK restart"
                await asyncio.sleep(0.05)
            assert events[2]["generation"] == 3
            await cli2.close()
            await srv2.stop()
        finally:
            await mc.close()
            await cli.close()
    run(go())


def test_client_sees_live_failover(tmp_path):
    """End-to-end: a client connected to a live shard learns the new
    primary after kill -9, exactly like a node-manatee application."""
    from manatee_amd.tools.devcluster import DevCluster

    async def go():
        c = DevCluster(str(tmp_path / "cluster"), n_peers=3,
                       shard_name="1.clilive")
        events = []
        mc = None
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            prim = await c.wait_writable(timeout_s=60)
            mc = ManateeClient(c.zk_conn_str, c.shard_path)
            mc.on("topology", events.append)
            await mc.start(timeout_s=30)
            assert ":%d/" % prim.pg_port in mc.topology["primary"]

            prim.kill9()
            deadline = asyncio.get_running_loop().time() + 60
            new_sync_url = None
            while asyncio.get_running_loop().time() < deadline:
                if events and events[-1]["generation"] > s["generation"]:
                    new_sync_url = events[-1]["primary"]
                    break
                await asyncio.sleep(0.05)
            assert new_sync_url is not None, "client never saw takeover"
            assert new_sync_url == s["sync"]["pgUrl"], \
                "new primary should be the old sync"
        finally:
            if mc is not None:
                await mc.close()
            c.stop()
    run(go())


def test_watch_poke_during_read_is_not_swallowed(tmp_path):
    """Regression (advisor finding): a watch notification dispatched
    BETWEEN re-arming the watch and waiting on the poke event must not
    be swallowed.  The client must clear the poke BEFORE the
    read+rewatch; a clear placed after the read would eat an event set
    mid-read, block forever on a consumed watch, and serve stale
    topology until session expiry."""
    async def go():
        srv = ZkServer()
        await srv.start()
        zk = ZkClient(srv.conn_str, session_timeout_ms=30000)
        await zk.connect()
        await zk.mkdirp("/manatee/1.poke")
        await zk.create("/manatee/1.poke/state",
                        json.dumps(state(1, 1, 2)).encode())

        cli = ManateeClient(srv.conn_str, "1.poke")
        reads = {"n": 0}
        orig = cli._read_and_watch

        async def racing_read():
            await orig()
            # simulate the io-loop dispatching a buffered notification
            # right after the watch was re-armed, before wait() runs
            if reads["n"] == 0:
                cli._poke.set()
            reads["n"] += 1

        cli._read_and_watch = racing_read
        await cli.start()
        try:
            # the mid-read poke must cause ANOTHER read cycle (the old
            # ordering swallowed it and blocked in wait() forever)
            deadline = asyncio.get_running_loop().time() + 5
            while reads["n"] < 2:
                assert asyncio.get_running_loop().time() < deadline, \
                    "poke set during read was swallowed; client blind"
                await asyncio.sleep(0.02)
            # and a real state change still comes through afterwards
            await zk.set_data("/manatee/1.poke/state",
                              json.dumps(state(2, 2, 1)).encode())
            deadline = asyncio.get_running_loop().time() + 5
            while (cli.topology or {}).get("generation") != 2:
                assert asyncio.get_running_loop().time() < deadline
                await asyncio.sleep(0.02)
        finally:
            await cli.close()
            await zk.close()
            await srv.stop()
    run(go())
