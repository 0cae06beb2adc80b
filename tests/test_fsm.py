"""FSM scenario tests — the hermetic analogue of the reference's
integration failover matrix (test/integ.test.js: primaryDeath, syncDeath,
asyncDeath, add4thManatee, …) plus the safety rules from SURVEY.md §2.2.
"""

import asyncio

import pytest

from manatee_amd.fsm import state as st
from tests.harness import Shard, pid


def run(coro):
    return asyncio.run(asyncio.wait_for(coro, 60))


# ----------------------------------------------------------- transition rules
def test_check_transition_rules():
    p1 = st.make_ident(pid("10.0.0.1"))
    p2 = st.make_ident(pid("10.0.0.2"))
    p3 = st.make_ident(pid("10.0.0.3"))
    s1 = {"generation": 1, "primary": p1, "sync": p2, "async": [p3],
          "deposed": [], "initWal": "0/00000000"}
    # legal: sync takes over with gen bump
    s2 = {"generation": 2, "primary": p2, "sync": p3, "async": [],
          "deposed": [p1], "initWal": "0/00000100"}
    st.check_transition(s1, s2)
    # illegal: generation decreases
    with pytest.raises(st.TransitionError):
        st.check_transition(s2, dict(s1))
    # illegal: primary change without gen bump
    with pytest.raises(st.TransitionError):
        st.check_transition(s1, dict(s1, primary=p2, sync=p1))
    # illegal: sync change without gen bump
    with pytest.raises(st.TransitionError):
        st.check_transition(s1, dict(s1, sync=p3, **{"async": [p2]}))
    # illegal: new primary was neither old primary nor old sync
    with pytest.raises(st.TransitionError):
        st.check_transition(s1, {"generation": 2, "primary": p3, "sync": p2,
                                 "async": [], "deposed": [p1],
                                 "initWal": "0/00000000"})
    # illegal: primary in deposed list
    with pytest.raises(st.TransitionError):
        st.check_transition(None, {"generation": 1, "primary": p1,
                                   "sync": None, "async": [],
                                   "deposed": [p1],
                                   "initWal": "0/00000000"})


def test_role_of():
    p1 = st.make_ident(pid("10.0.0.1"))
    p2 = st.make_ident(pid("10.0.0.2"))
    p3 = st.make_ident(pid("10.0.0.3"))
    p4 = st.make_ident(pid("10.0.0.4"))
    s = {"generation": 1, "primary": p1, "sync": p2, "async": [p3],
         "deposed": [p4], "initWal": "0/00000000"}
    assert st.role_of(s, p1["id"]) == "primary"
    assert st.role_of(s, p2["id"]) == "sync"
    assert st.role_of(s, p3["id"]) == "async"
    assert st.role_of(s, p4["id"]) == "deposed"
    assert st.role_of(s, "x:1:2") == "unassigned"
    assert st.role_of(None, p1["id"]) == "unassigned"


# -------------------------------------------------------------- cluster setup
def test_cluster_formation_3_peers():
    async def go():
        shard = await Shard().start(3)
        try:
            s = await shard.wait_state(
                lambda s: s["generation"] == 1 and len(s["async"]) == 1,
                what="gen-1 formation")
            assert s["primary"]["id"] == pid("10.0.0.1")
            assert s["sync"]["id"] == pid("10.0.0.2")
            assert s["async"][0]["id"] == pid("10.0.0.3")
            assert s["initWal"] == "0/00000000"
            assert s["primary"]["zoneId"] == "zone-10.0.0.1"
            # db configs applied per role
            await asyncio.sleep(0.3)
            assert shard.peer(0).db.role == "primary"
            assert shard.peer(0).db.current["downstream"]["pgUrl"] == \
                shard.peer(1).ident["pgUrl"]
            assert shard.peer(1).db.role == "sync"
            assert shard.peer(1).db.current["upstream"]["pgUrl"] == \
                shard.peer(0).ident["pgUrl"]
            # async chains off the sync, restores off the primary
            assert shard.peer(2).db.role == "async"
            assert shard.peer(2).db.current["upstream"]["pgUrl"] == \
                shard.peer(1).ident["pgUrl"]
            assert shard.peer(2).db.current["restorePeer"]["backupUrl"] == \
                shard.peer(0).ident["backupUrl"]
        finally:
            await shard.stop()
    run(go())


def test_lone_peer_waits():
    async def go():
        shard = await Shard().start(1)
        try:
            await asyncio.sleep(1.0)
            assert await shard.state() is None
            assert "cluster setup" in shard.peer(0).fsm.debug_state()["peerState"]
        finally:
            await shard.stop()
    run(go())


def test_add_4th_peer_becomes_async():
    async def go():
        shard = await Shard().start(3)
        try:
            await shard.wait_state(lambda s: len(s["async"]) == 1)
            await shard.add_peer("10.0.0.4")
            s = await shard.wait_state(lambda s: len(s["async"]) == 2,
                                       what="4th peer added as async")
            assert s["async"][1]["id"] == pid("10.0.0.4")
            assert s["generation"] == 1  # no gen bump for async add
        finally:
            await shard.stop()
    run(go())


# ------------------------------------------------------------------ failovers
def test_primary_death_failover():
    async def go():
        shard = await Shard().start(3)
        try:
            await shard.wait_state(lambda s: len(s["async"]) == 1)
            await shard.peer(0).kill()
            s = await shard.wait_state(
                lambda s: s["generation"] == 2, timeout=15,
                what="takeover after primary death")
            assert s["primary"]["id"] == pid("10.0.0.2")
            assert s["sync"]["id"] == pid("10.0.0.3")
            assert s["async"] == []
            assert [d["id"] for d in s["deposed"]] == [pid("10.0.0.1")]
            await asyncio.sleep(0.3)
            assert shard.peer(1).db.role == "primary"
            assert shard.peer(2).db.role == "sync"
        finally:
            await shard.stop()
    run(go())


def test_sync_death_async_promoted():
    async def go():
        shard = await Shard().start(3)
        try:
            await shard.wait_state(lambda s: len(s["async"]) == 1)
            await shard.peer(1).kill()
            s = await shard.wait_state(
                lambda s: s["generation"] == 2, timeout=15,
                what="sync replacement")
            assert s["primary"]["id"] == pid("10.0.0.1")
            assert s["sync"]["id"] == pid("10.0.0.3")
            assert s["async"] == []
            assert s["deposed"] == []  # dead sync is NOT deposed
        finally:
            await shard.stop()
    run(go())


def test_async_death_removed_without_gen_bump():
    async def go():
        shard = await Shard().start(3)
        try:
            await shard.wait_state(lambda s: len(s["async"]) == 1)
            await shard.peer(2).kill()
            s = await shard.wait_state(
                lambda s: s["async"] == [], timeout=15,
                what="async removed")
            assert s["generation"] == 1
            assert s["sync"]["id"] == pid("10.0.0.2")
        finally:
            await shard.stop()
    run(go())


def test_primary_death_two_peer_cluster_no_takeover():
    """With no async to become the new sync, the sync must NOT take over
    (no replacement ⇒ durability would be compromised)."""
    async def go():
        shard = await Shard().start(2)
        try:
            await shard.wait_state(lambda s: s["generation"] == 1)
            await shard.peer(0).kill()
            await asyncio.sleep(3.0)
            s = await shard.state()
            assert s["generation"] == 1
            assert s["primary"]["id"] == pid("10.0.0.1")
        finally:
            await shard.stop()
    run(go())


def test_frozen_cluster_no_takeover():
    async def go():
        shard = await Shard().start(3)
        try:
            s = await shard.wait_state(lambda s: len(s["async"]) == 1)
            # freeze via direct state write (what manatee-adm freeze does)
            frozen = dict(s)
            frozen["freeze"] = {"date": st.iso8601(),
                                "reason": "by test for CM-129"}
            await shard.peer(1).zk.put_cluster_state(frozen)
            await shard.peer(0).kill()
            await asyncio.sleep(3.0)
            s2 = await shard.state()
            assert s2["generation"] == 1, "frozen cluster must not fail over"
            assert s2["primary"]["id"] == pid("10.0.0.1")
            # unfreeze → takeover proceeds
            thawed = {k: v for k, v in s2.items() if k != "freeze"}
            await shard.peer(1).zk.put_cluster_state(thawed)
            s3 = await shard.wait_state(lambda s: s["generation"] == 2,
                                        timeout=15, what="post-thaw takeover")
            assert s3["primary"]["id"] == pid("10.0.0.2")
        finally:
            await shard.stop()
    run(go())


def test_initwal_fence_blocks_stale_sync():
    """A sync that has not caught up to initWal must refuse takeover."""
    async def go():
        shard = Shard()
        shard.srv = None
        from manatee_amd.coord.zkserver import ZkServer
        shard.srv = ZkServer(tick_ms=50, min_session_timeout_ms=300)
        await shard.srv.start()
        p1 = await shard.add_peer("10.0.0.1", xlog="0/00001000")
        await asyncio.sleep(0.05)
        p2 = await shard.add_peer("10.0.0.2", xlog="0/00000010")  # behind
        await asyncio.sleep(0.05)
        p3 = await shard.add_peer("10.0.0.3", xlog="0/00000010")
        try:
            s = await shard.wait_state(lambda s: len(s["async"]) == 1)
            # move the fence forward: primary re-declares with its own xlog
            bumped = dict(s, generation=2, initWal="0/00001000")
            # (simulate a generation that began at 0/00001000)
            await p1.zk.put_cluster_state(bumped)
            await asyncio.sleep(0.3)
            await p1.kill()
            await asyncio.sleep(3.0)
            s2 = await shard.state()
            assert s2["generation"] == 2, \
                "stale sync must not take over past the initWal fence"
            # now let the sync catch up: fence opens, takeover proceeds
            p2.db.xlog = "0/00002000"
            s3 = await shard.wait_state(lambda s: s["generation"] == 3,
                                        timeout=15, what="post-catchup takeover")
            assert s3["primary"]["id"] == pid("10.0.0.2")
            assert s3["initWal"] == "0/00002000"
        finally:
            await shard.stop()
    run(go())


# ------------------------------------------------------------------- promote
def test_promote_async0_to_sync():
    async def go():
        shard = await Shard().start(3)
        try:
            s = await shard.wait_state(lambda s: len(s["async"]) == 1)
            req = dict(s)
            req["promote"] = {"id": pid("10.0.0.3"), "role": "async",
                              "asyncIndex": 0, "generation": s["generation"],
                              "expireTime": st.iso8601(
                                  __import__("time").time() + 30)}
            await shard.peer(2).zk.put_cluster_state(req)
            s2 = await shard.wait_state(
                lambda s: s["generation"] == 2 and "promote" not in s,
                timeout=15, what="async promote consumed")
            assert s2["sync"]["id"] == pid("10.0.0.3")
            assert s2["async"][0]["id"] == pid("10.0.0.2")  # old sync demoted
            assert s2["primary"]["id"] == pid("10.0.0.1")
        finally:
            await shard.stop()
    run(go())


def test_promote_sync_deposes_primary():
    async def go():
        shard = await Shard().start(3)
        try:
            s = await shard.wait_state(lambda s: len(s["async"]) == 1)
            req = dict(s)
            req["promote"] = {"id": pid("10.0.0.2"), "role": "sync",
                              "generation": s["generation"],
                              "expireTime": st.iso8601(
                                  __import__("time").time() + 30)}
            await shard.peer(1).zk.put_cluster_state(req)
            s2 = await shard.wait_state(
                lambda s: s["generation"] == 2 and "promote" not in s,
                timeout=15, what="sync promote consumed")
            assert s2["primary"]["id"] == pid("10.0.0.2")
            assert s2["sync"]["id"] == pid("10.0.0.3")
            assert [d["id"] for d in s2["deposed"]] == [pid("10.0.0.1")]
        finally:
            await shard.stop()
    run(go())


def test_expired_promote_cleared():
    async def go():
        shard = await Shard().start(3)
        try:
            s = await shard.wait_state(lambda s: len(s["async"]) == 1)
            req = dict(s)
            req["promote"] = {"id": pid("10.0.0.3"), "role": "async",
                              "asyncIndex": 0, "generation": s["generation"],
                              "expireTime": st.iso8601(
                                  __import__("time").time() - 5)}
            await shard.peer(2).zk.put_cluster_state(req)
            s2 = await shard.wait_state(
                lambda s: "promote" not in s, timeout=15,
                what="expired promote cleared")
            assert s2["generation"] == 1  # nothing happened
            assert s2["sync"]["id"] == pid("10.0.0.2")
        finally:
            await shard.stop()
    run(go())


# ------------------------------------------------------------- deposed rejoin
def test_deposed_peer_rejoins_stays_deposed():
    async def go():
        shard = await Shard().start(3)
        try:
            await shard.wait_state(lambda s: len(s["async"]) == 1)
            dead = shard.peer(0)
            await dead.kill()
            await shard.wait_state(lambda s: s["generation"] == 2,
                                   timeout=15)
            # the deposed peer restarts (new sitter process)
            del shard.peers[dead.id]
            revived = await shard.add_peer("10.0.0.1")
            await asyncio.sleep(1.5)
            s = await shard.state()
            assert [d["id"] for d in s["deposed"]] == [pid("10.0.0.1")]
            assert all(a["id"] != pid("10.0.0.1") for a in s["async"]), \
                "deposed peer must not be re-added as async"
            assert revived.fsm.debug_state()["role"] == "deposed"
            assert revived.db.role == "none"
        finally:
            await shard.stop()
    run(go())


# ---------------------------------------------------------------------- ONWM
def test_singleton_onwm_formation():
    async def go():
        shard = await Shard().start(1, singleton=True)
        try:
            s = await shard.wait_state(lambda s: s.get("oneNodeWriteMode"),
                                       what="ONWM formation")
            assert s["generation"] == 1
            assert s["primary"]["id"] == pid("10.0.0.1")
            assert s["sync"] is None
            await asyncio.sleep(0.3)
            assert shard.peer(0).db.role == "primary"
            assert shard.peer(0).db.current["downstream"] is None
        finally:
            await shard.stop()
    run(go())


def test_foreign_peer_shuts_down_in_onwm():
    async def go():
        shard = await Shard().start(1, singleton=True)
        try:
            await shard.wait_state(lambda s: s.get("oneNodeWriteMode"))
            joiner = await shard.add_peer("10.0.0.9")
            deadline = asyncio.get_running_loop().time() + 10
            while not joiner.fsm.debug_state()["shutdown"]:
                assert asyncio.get_running_loop().time() < deadline
                await asyncio.sleep(0.05)
            assert joiner.db.role == "none"
        finally:
            await shard.stop()
    run(go())


# ----------------------------------------------------- races / double-failure
def test_sync_and_async_simultaneous_death():
    async def go():
        shard = await Shard().start(4)
        try:
            await shard.wait_state(lambda s: len(s["async"]) == 2)
            await shard.peer(1).kill()
            await shard.peer(2).kill()
            s = await shard.wait_state(
                lambda s: s["generation"] == 2 and s["async"] == [],
                timeout=15, what="recovery from double death")
            assert s["primary"]["id"] == pid("10.0.0.1")
            assert s["sync"]["id"] == pid("10.0.0.4")
        finally:
            await shard.stop()
    run(go())


def test_primary_and_sync_simultaneous_death_no_unsafe_takeover():
    """If both primary and sync die, the asyncs must NOT self-elect —
    only the sync may become primary (data-safety rule)."""
    async def go():
        shard = await Shard().start(3)
        try:
            await shard.wait_state(lambda s: len(s["async"]) == 1)
            await shard.peer(0).kill()
            await shard.peer(1).kill()
            await asyncio.sleep(3.0)
            s = await shard.state()
            assert s["generation"] == 1
            assert s["primary"]["id"] == pid("10.0.0.1")
        finally:
            await shard.stop()
    run(go())


def test_everyone_dies_cluster_resumes_on_return():
    """everyoneDies analogue: all peers killed; when peers return with the
    same identities, the persisted state still names the old topology and
    peers resume their roles."""
    async def go():
        shard = await Shard().start(3)
        try:
            await shard.wait_state(lambda s: len(s["async"]) == 1)
            for p in list(shard.peers.values()):
                await p.kill()
            shard.peers.clear()
            await asyncio.sleep(1.5)
            for ip in ("10.0.0.1", "10.0.0.2", "10.0.0.3"):
                await shard.add_peer(ip)
                await asyncio.sleep(0.05)
            await asyncio.sleep(1.5)
            s = await shard.state()
            assert s["generation"] == 1
            assert s["primary"]["id"] == pid("10.0.0.1")
            assert shard.peer(0).db.role == "primary"
            assert shard.peer(1).db.role == "sync"
            assert shard.peer(2).db.role == "async"
        finally:
            await shard.stop()
    run(go())


def test_initialized_db_refuses_autoformation():
    """SAFETY: peers whose databases already hold data must never
    auto-declare generation 1 when cluster state is missing — arbitrary
    election order could elect a stale peer as primary and acknowledged
    writes would be destroyed when the others re-slave to it.  This is
    the operator state-backfill situation."""
    from tests.harness import Shard

    async def go():
        shard = Shard()
        shard.srv = __import__("manatee_amd.coord.zkserver",
                               fromlist=["ZkServer"]).ZkServer(
            tick_ms=50, min_session_timeout_ms=300)
        await shard.srv.start()
        try:
            for i in range(3):
                p = await shard.add_peer("10.0.0.%d" % (i + 1))
                # override the fresh-db default: data already exists
                p.db.fire_init(setup=True)
            await asyncio.sleep(2.0)
            assert await shard.state() is None, \
                "peers auto-formed despite initialized databases"
            # the operator backfills; the cluster then proceeds
            from manatee_amd.adm import core as adm
            zk = await adm.create_zk_client(shard.srv.conn_str)
            try:
                state = await adm.state_backfill(
                    zk, "/manatee/simshard")
                assert state["generation"] == 0
                assert state["freeze"], "backfill must auto-freeze"
            finally:
                await zk.close()
            s = await shard.wait_state(
                lambda s: s.get("primary") is not None,
                what="state after backfill")
            assert s["generation"] == 0
        finally:
            await shard.stop()
    run(go())
