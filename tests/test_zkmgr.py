"""ZkMgr component tests, mirroring test/zookeeperMgr.test.js scenarios:
setup, create/update CAS state + history writes, membership add/remove
events, debounce, stale-session dedup, session-expiry reset.
"""

import asyncio
import json

import pytest

from manatee_amd.coord import jute
from manatee_amd.coord.zkmgr import (ZkMgr, id_lists_equal,
                                     parse_and_unique_actives)
from manatee_amd.coord.zkserver import ZkServer

SHARD = "/manatee/1.moray.test"


def peer_data(ip, pg=5432, backup=12345):
    return {
        "zoneId": "zone-" + ip,
        "ip": ip,
        "pgUrl": "tcp://postgres@%s:%d/postgres" % (ip, pg),
        "backupUrl": "http://%s:%d" % (ip, backup),
    }


def mk_mgr(srv, ip, timeout=4000):
    pid = "%s:5432:12345" % ip
    return ZkMgr(id=pid, data=peer_data(ip), path=SHARD,
                 conn_str=srv.conn_str, session_timeout_ms=timeout)


def run(coro):
    return asyncio.run(coro)


async def wait_for(pred, timeout=5.0, what="condition"):
    deadline = asyncio.get_running_loop().time() + timeout
    while not pred():
        if asyncio.get_running_loop().time() > deadline:
            raise AssertionError("timeout waiting for " + what)
        await asyncio.sleep(0.02)


# ------------------------------------------------------------------ helpers
def test_parse_and_unique_actives():
    peers = parse_and_unique_actives([
        "10.0.0.1:5432:12345-0000000003",
        "10.0.0.2:5432:12345-0000000001",
        "10.0.0.1:5432:12345-0000000005",  # stale dup: keep lowest seq
        "garbage",
    ])
    assert [(p.id, p.seq) for p in peers] == [
        ("10.0.0.2:5432:12345", 1), ("10.0.0.1:5432:12345", 3)]
    assert id_lists_equal(peers, peers)
    assert not id_lists_equal(peers, peers[:1])


# -------------------------------------------------------------------- setup
def test_init_and_state_write():
    async def go():
        srv = ZkServer()
        await srv.start()
        m1 = mk_mgr(srv, "10.0.0.1")
        inits = []
        m1.on("init", lambda ev: inits.append(ev))
        await m1.init()
        try:
            assert len(inits) == 1
            assert inits[0]["clusterState"] is None
            assert [a["id"] for a in inits[0]["active"]] == \
                ["10.0.0.1:5432:12345"]
            assert inits[0]["active"][0]["pgUrl"].startswith("tcp://")

            state = {"generation": 1,
                     "primary": dict(peer_data("10.0.0.1"),
                                     id="10.0.0.1:5432:12345"),
                     "sync": None, "async": [], "deposed": [],
                     "initWal": "0/00000000"}
            await m1.put_cluster_state(state)
            # history node written atomically with state
            kids, _ = await m1._zk.get_children(SHARD + "/history")
            assert len(kids) == 1 and kids[0].startswith("1-")
            data, _ = await m1._zk.get_data(SHARD + "/state")
            assert json.loads(data)["generation"] == 1

            # CAS update
            state2 = dict(state, generation=2)
            await m1.put_cluster_state(state2)
            kids, _ = await m1._zk.get_children(SHARD + "/history")
            assert len(kids) == 2
        finally:
            await m1.close()
            await srv.stop()
    run(go())


def test_concurrent_state_write_cas_conflict():
    """Two managers racing to write state: exactly one wins; the loser gets
    BAD_VERSION (ref testFailWriteClusterState :691)."""
    async def go():
        srv = ZkServer()
        await srv.start()
        m1 = mk_mgr(srv, "10.0.0.1")
        m2 = mk_mgr(srv, "10.0.0.2")
        await m1.init()
        await m2.init()
        try:
            base = {"generation": 1, "primary": None, "sync": None,
                    "async": [], "deposed": [], "initWal": "0/00000000"}
            await m1.put_cluster_state(base)
            # m2 learns the state via its watch
            await wait_for(lambda: m2.cluster_state is not None,
                           what="m2 state watch")
            # both try to write gen 2; m1 writes first, m2's cached version
            # is now stale
            await m1.put_cluster_state(dict(base, generation=2))
            with pytest.raises(jute.ZkError) as ei:
                await m2.put_cluster_state(dict(base, generation=2))
            assert ei.value.code == jute.ZBADVERSION
        finally:
            await m1.close()
            await m2.close()
            await srv.stop()
    run(go())


def test_membership_events_and_debounce():
    async def go():
        srv = ZkServer(tick_ms=50, min_session_timeout_ms=300)
        await srv.start()
        m1 = mk_mgr(srv, "10.0.0.1", timeout=2000)
        changes = []
        m1.on("activeChange", lambda a: changes.append([p["id"] for p in a]))
        await m1.init()
        m2 = mk_mgr(srv, "10.0.0.2", timeout=400)
        await m2.init()
        try:
            await wait_for(lambda: len(changes) == 1, what="join event")
            assert sorted(changes[0]) == ["10.0.0.1:5432:12345",
                                          "10.0.0.2:5432:12345"]
            # kill m2's session without clean close → ephemeral expires
            m2._zk._closing = True
            m2._zk._writer.close()
            await wait_for(lambda: len(changes) >= 2, timeout=8,
                           what="leave event")
            assert changes[-1] == ["10.0.0.1:5432:12345"]
        finally:
            await m1.close()
            await m2.close()
            await srv.stop()
    run(go())


def test_cluster_state_change_event():
    async def go():
        srv = ZkServer()
        await srv.start()
        m1 = mk_mgr(srv, "10.0.0.1")
        m2 = mk_mgr(srv, "10.0.0.2")
        await m1.init()
        state_events = []
        m2.on("clusterStateChange", lambda s: state_events.append(s))
        await m2.init()
        try:
            st = {"generation": 5, "primary": None, "sync": None,
                  "async": [], "deposed": [], "initWal": "0/00000000"}
            await m1.put_cluster_state(st)
            await wait_for(lambda: len(state_events) >= 1,
                           what="state change event")
            assert state_events[0]["generation"] == 5
            # a second write also fires (watch re-registered)
            await m1.put_cluster_state(dict(st, generation=6))
            await wait_for(lambda: len(state_events) >= 2,
                           what="second state change event")
            assert state_events[1]["generation"] == 6
        finally:
            await m1.close()
            await m2.close()
            await srv.stop()
    run(go())


def test_session_expiry_rebuild():
    """On session expiry the manager must build a new session, rejoin the
    election, and emit a fresh init (ref zookeeperMgr.js:500-586)."""
    async def go():
        srv = ZkServer(tick_ms=50, min_session_timeout_ms=300)
        await srv.start()
        m1 = mk_mgr(srv, "10.0.0.1", timeout=400)
        inits = []
        m1.on("init", lambda ev: inits.append(ev))
        await m1.init()
        try:
            sid1 = m1._zk.session_id
            # simulate a long partition: server expires the session while
            # the client cannot reconnect
            srv._expire_session(srv.sessions[sid1])
            await wait_for(lambda: len(inits) >= 2, timeout=8,
                           what="re-init after expiry")
            assert m1._zk.session_id != sid1
            # rejoined the election with a fresh ephemeral
            assert [a["id"] for a in m1.active] == ["10.0.0.1:5432:12345"]
        finally:
            await m1.close()
            await srv.stop()
    run(go())


# ------------------------------------------------- watch-loss regressions
def test_transient_relist_failure_does_not_lose_the_watch():
    """A one-shot watch fires, the re-list RPC fails transiently
    (connection blip): the handler must retry until the watch is
    re-armed — returning without one leaves the peer permanently blind
    to membership changes (the takeover-never-happens failure shape)."""
    async def go():
        srv = ZkServer()
        await srv.start()
        m1 = mk_mgr(srv, "10.0.0.1", timeout=2000)
        changes = []
        m1.on("activeChange", lambda a: changes.append([p["id"]
                                                       for p in a]))
        await m1.init()
        m2 = mk_mgr(srv, "10.0.0.2", timeout=2000)
        await m2.init()
        try:
            await wait_for(lambda: any("10.0.0.2:5432:12345" in c
                                       for c in changes),
                           what="m2 join observed")
            # make the NEXT re-list fail once, then work again
            orig = m1._zk.get_children
            state = {"fails": 1}

            async def flaky(path, watch=None):
                if state["fails"] > 0:
                    state["fails"] -= 1
                    raise jute.ZkError(jute.ZCONNECTIONLOSS, path)
                return await orig(path, watch=watch)

            m1._zk.get_children = flaky
            before = len(changes)
            await m2.close()     # fires m1's watch; re-list fails once
            await wait_for(lambda: len(changes) > before, timeout=10,
                           what="departure observed despite the blip")
            assert "10.0.0.2:5432:12345" not in changes[-1]
            assert state["fails"] == 0
        finally:
            await m1.close()
            await srv.stop()
    run(go())


def test_resync_heals_a_fully_lost_watch():
    """Belt and braces: even if a watch notification is lost outright
    (simulated by discarding it), the low-frequency resync loop must
    surface the membership change within ~a session timeout."""
    async def go():
        srv = ZkServer()
        await srv.start()
        m1 = mk_mgr(srv, "10.0.0.1", timeout=2000)
        changes = []
        m1.on("activeChange", lambda a: changes.append([p["id"]
                                                       for p in a]))
        await m1.init()
        m2 = mk_mgr(srv, "10.0.0.2", timeout=2000)
        await m2.init()
        try:
            await wait_for(lambda: any("10.0.0.2:5432:12345" in c
                                       for c in changes),
                           what="m2 join observed")
            # lose the watch outright: drop the client-side registration
            # so the server's notification finds nothing to dispatch and
            # nothing re-arms
            m1._zk._child_watches.clear()
            before = len(changes)
            await m2.close()
            # the resync loop (period = session timeout) must notice
            await wait_for(lambda: len(changes) > before, timeout=10,
                           what="departure observed via resync")
            assert "10.0.0.2:5432:12345" not in changes[-1]
        finally:
            await m1.close()
            await srv.stop()
    run(go())
