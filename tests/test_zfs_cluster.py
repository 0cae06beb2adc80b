"""Full-cluster tier on the ZFS provider (fakezfs pools, one per peer):
the reference's actual storage layout — datasets mounted at the data
path, 13-digit snapshots, ``zfs send | recv`` bootstrap over the backup
server, isolation on rebuild — driven end-to-end by real daemon
subprocesses (ref lib/zfsClient.js + test/integ.test.js)."""

import asyncio
import time

import pytest

from manatee_amd.tools.devcluster import DevCluster


def run(coro, timeout=240):
    return asyncio.run(asyncio.wait_for(coro, timeout))


def test_zfs_provider_formation_failover_rebuild(tmp_path):
    async def go():
        c = DevCluster(str(tmp_path / "c"), n_peers=3,
                       shard_name="1.zfs", storage_provider="zfs",
                       run_snapshotter=False)
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=120, what="formation on zfs provider")
            prim = await c.wait_writable(timeout_s=120)
            cli = prim.db_client()
            for i in range(30):
                await cli.put("z%d" % i, i)
            await cli.close()

            prim.kill9()
            s2 = await c.wait_cluster(
                lambda st: st["generation"] > s["generation"],
                timeout_s=60, what="takeover")
            newp = await c.wait_writable(timeout_s=60)
            cli = newp.db_client()
            assert await cli.count(prefix="z") == 30
            for i in range(30):
                assert await cli.get("z%d" % i) == i
            await cli.close()
            assert any(d["id"] == prim.id for d in s2.get("deposed", []))

            # rebuild destroys the deposed pool dataset and restores
            # from the new primary's zfs stream
            await c.rebuild_peer(prim)
            await c.wait_writable(timeout_s=60)
            rcli = prim.db_client()
            deadline = time.monotonic() + 30
            while True:
                try:
                    if await rcli.get("z29") == 29:
                        break
                except Exception:
                    pass
                assert time.monotonic() < deadline
                await asyncio.sleep(0.2)
            await rcli.close()
        finally:
            c.stop()
    run(go())


def test_zfs_provider_bootstrap_under_write_load(tmp_path):
    """4th peer joins mid-load: its zfsClient-style restore (listen
    socket + POST /backup + stream into recv) runs while the primary
    keeps acking writes (ref lib/zfsClient.js:765-886)."""
    async def go():
        c = DevCluster(str(tmp_path / "c"), n_peers=3,
                       shard_name="1.zfsboot", storage_provider="zfs",
                       run_snapshotter=False)
        try:
            await c.start()
            await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=120, what="formation")
            prim = await c.wait_writable(timeout_s=120)
            cli = prim.db_client()
            for i in range(60):
                await cli.put("seed%d" % i, i)

            p4 = c.add_peer_config()
            p4.start()
            # keep writing while the bootstrap streams
            stop = False

            async def pump():
                j = 0
                while not stop:
                    try:
                        await cli.put("live%d" % j, j, timeout_s=2.0)
                        j += 1
                    except Exception:
                        await asyncio.sleep(0.05)
                return j
            pump_task = asyncio.get_running_loop().create_task(pump())
            try:
                s = await c.wait_cluster(
                    lambda s: len(s.get("async", [])) == 2, timeout_s=120,
                    what="4th peer joined via zfs bootstrap")
                assert s["async"][1]["id"] == p4.id
            finally:
                stop = True
                wrote = await pump_task
            assert wrote > 0, "writes stalled during the bootstrap"
            await cli.close()

            p4cli = p4.db_client()
            deadline = time.monotonic() + 60
            while True:
                try:
                    if await p4cli.get("seed59") == 59 and \
                            await p4cli.get("live0") == 0:
                        break
                except Exception:
                    pass
                assert time.monotonic() < deadline, \
                    "bootstrap data did not arrive on the 4th peer"
                await asyncio.sleep(0.3)
            st = await p4cli.status()
            assert st["upstream_status"] == "streaming"
            await p4cli.close()
        finally:
            c.stop()
    run(go())


def test_zfs_provider_with_postgres_engine(tmp_path):
    """The reference's actual deployment shape: the postgres engine on
    ZFS datasets (conf per major, zfs snapshots, stream bootstrap) —
    formation + failover + zero loss with both substrates in play."""
    async def go():
        c = DevCluster(str(tmp_path / "c"), n_peers=3,
                       shard_name="1.zfspg", storage_provider="zfs",
                       engine="postgres", run_snapshotter=False)
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=120, what="formation (zfs + postgres)")
            prim = await c.wait_writable(timeout_s=120)
            cli = prim.db_client()
            for i in range(25):
                await cli.put("zp%d" % i, i)
            await cli.close()
            prim.kill9()
            await c.wait_cluster(
                lambda st: st["generation"] > s["generation"],
                timeout_s=60, what="takeover (zfs + postgres)")
            newp = await c.wait_writable(timeout_s=60)
            cli = newp.db_client()
            assert await cli.count(prefix="zp") == 25
            await cli.close()
        finally:
            c.stop()
    run(go())
