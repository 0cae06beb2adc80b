"""Tier-4 integration tests: full shard of real subprocesses (embedded ZK
server + sitter + backupserver + waldb per peer) on loopback.

Mirrors the reference's test/integ.test.js scenario matrix: formation,
primaryDeath, syncDeath, bootstrap of an extra peer, db-crash restart —
each verifying topology AND data (zero acknowledged-write loss).
"""

import asyncio
import time

import pytest

from manatee_amd.tools.devcluster import DevCluster


def run(coro, timeout=180):
    return asyncio.run(asyncio.wait_for(coro, timeout))


@pytest.fixture
def cluster_dir(tmp_path):
    return str(tmp_path / "cluster")


def test_onwm_single_peer(cluster_dir):
    async def go():
        c = DevCluster(cluster_dir, n_peers=1, singleton=True,
                       shard_name="1.onwm")
        try:
            await c.start()
            s = await c.wait_cluster(lambda s: s.get("oneNodeWriteMode"),
                                     what="ONWM formation")
            assert s["primary"]["id"] == c.peers[0].id
            prim = await c.wait_writable()
            cli = prim.db_client()
            await cli.put("k", "v")
            assert await cli.get("k") == "v"
            await cli.close()
            # status endpoints (/ping flips to 200 once the manager's own
            # health probe confirms the db — may lag the first write)
            deadline = time.monotonic() + 10
            while True:
                status, body = await prim.http_status("/ping")
                if status == 200:
                    break
                assert time.monotonic() < deadline, "/ping never turned 200"
                await asyncio.sleep(0.1)
            assert body["healthy"]
            status, body = await prim.http_status("/state")
            assert body["role"] == "primary"
            status, body = await prim.http_status("/restore")
            assert status == 200
        finally:
            c.stop()
    run(go())


def test_3peer_formation_writes_and_replication(cluster_dir):
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.form")
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="3-peer formation")
            prim = await c.wait_writable(timeout_s=60)
            assert prim.id == s["primary"]["id"]
            cli = prim.db_client()
            for i in range(20):
                await cli.put("row%d" % i, {"i": i})
            await cli.close()
            # replicated through the chain to the async
            asy = c.peer_by_id(s["async"][0]["id"])
            acli = asy.db_client()
            deadline = time.monotonic() + 30
            while True:
                try:
                    if await acli.get("row19") == {"i": 19}:
                        break
                except Exception:
                    pass
                assert time.monotonic() < deadline, \
                    "write did not cascade to async"
                await asyncio.sleep(0.2)
            st = await acli.status()
            assert st["upstream_status"] == "streaming"
            await acli.close()
        finally:
            c.stop()
    run(go())


def test_primary_kill9_failover_zero_write_loss(cluster_dir):
    """The north-star scenario (BASELINE.json): kill -9 the primary under
    write load; a writable primary must return and every acknowledged
    write must survive."""
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.fail")
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            prim = await c.wait_writable(timeout_s=60)
            acked = {}
            cli = prim.db_client()
            for i in range(100):
                await cli.put("pre%d" % i, i)
                acked["pre%d" % i] = i
            await cli.close()

            old_primary_id = prim.id
            prim.kill9()
            t_kill = time.monotonic()

            s2 = await c.wait_cluster(
                lambda s: s["generation"] == 2 and
                s["primary"]["id"] != old_primary_id,
                timeout_s=60, what="takeover")
            new_prim = await c.wait_writable(timeout_s=60)
            t_writable = time.monotonic()
            assert new_prim.id == s2["primary"]["id"]
            assert [d["id"] for d in s2["deposed"]] == [old_primary_id]

            # zero acknowledged-write loss
            ncli = new_prim.db_client()
            for k, v in acked.items():
                assert await ncli.get(k) == v, \
                    "acknowledged write %s lost in failover" % k
            # and the new primary truly accepts new writes
            await ncli.put("post-failover", 1)
            await ncli.close()
            failover_s = t_writable - t_kill
            print("failover-to-writable: %.2fs" % failover_s)
            # reference integ bound is 30s on one loaded host
            assert failover_s < 30
        finally:
            c.stop()
    run(go())


def test_sync_kill9_async_promoted_writes_resume(cluster_dir):
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.syncdeath")
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            prim = await c.wait_writable(timeout_s=60)
            sync_peer = c.peer_by_id(s["sync"]["id"])
            sync_peer.kill9()
            s2 = await c.wait_cluster(
                lambda s: s["generation"] == 2, timeout_s=60,
                what="sync replacement")
            assert s2["sync"]["id"] == s["async"][0]["id"]
            assert s2["deposed"] == []
            # writes work again once the new sync caught up
            new_prim = await c.wait_writable(timeout_s=60)
            assert new_prim.id == prim.id
        finally:
            c.stop()
    run(go())


def test_4th_peer_bootstraps_via_backup_restore(cluster_dir):
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.boot")
        try:
            await c.start()
            await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            prim = await c.wait_writable(timeout_s=60)
            cli = prim.db_client()
            for i in range(50):
                await cli.put("seed%d" % i, i)

            # add a 4th peer with an empty store: it must restore from the
            # primary's backup server and join as async[1]
            p4 = c.add_peer_config()
            p4.start()
            s = await c.wait_cluster(
                lambda s: len(s.get("async", [])) == 2, timeout_s=90,
                what="4th peer joined")
            assert s["async"][1]["id"] == p4.id

            # data made it over (restore + streaming)
            p4cli = p4.db_client()
            deadline = time.monotonic() + 60
            while True:
                try:
                    if await p4cli.get("seed49") == 49:
                        break
                except Exception:
                    pass
                assert time.monotonic() < deadline, \
                    "bootstrap data did not arrive on 4th peer"
                await asyncio.sleep(0.3)
            st = await p4cli.status()
            assert st["upstream_status"] == "streaming"
            await p4cli.close()
            await cli.close()
        finally:
            c.stop()
    run(go())


def test_db_crash_is_restarted_by_sitter(cluster_dir):
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.crash")
        try:
            await c.start()
            await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            prim = await c.wait_writable(timeout_s=60)
            cli = prim.db_client()
            await cli.put("before-crash", 1)
            await cli.close()
            prim.kill_db_only()
            # the sitter must notice and restart its database; no failover
            # (our ZK session is still alive)
            await asyncio.sleep(1.0)
            new_prim = await c.wait_writable(timeout_s=60)
            assert new_prim.id == prim.id, "unexpected failover"
            cli = prim.db_client()
            assert await cli.get("before-crash") == 1
            await cli.put("after-crash", 2)
            await cli.close()
            s = await c.cluster_state()
            assert s["generation"] == 1
        finally:
            c.stop()
    run(go())
