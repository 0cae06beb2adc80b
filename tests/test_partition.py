"""Network-partition failover tests (the ipdadm tier of the reference's
chaos plan, ref docs/test-plan.md:24-113).

No netns/iptables exists in this environment, so partitions are induced
in userspace: every outbound connection a peer makes is routed through a
per-directed-link proxy (tools/netproxy.LinkProxy via the
MANATEE_DIAL_MAP rewrite layer, common/dial.py), and the proxy drops
bytes per direction.  Unlike SIGSTOP, this produces the classic
split-brain shapes: a peer that is alive and reachable by CLIENTS but
cut off from ZooKeeper, and asymmetric links where A hears B but B
cannot hear A.

Safety property under test, in all scenarios: once a write is
acknowledged, it survives; and after a takeover the deposed primary can
never acknowledge another write (the synchronous-replication gate).
"""

import asyncio
import os
import sys
import time

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import bench  # noqa: E402  (Writer + exact loss verification)
from manatee_amd.db.waldb.client import WaldbClient, WaldbError  # noqa: E402
from manatee_amd.tools.devcluster import DevCluster  # noqa: E402


def run(coro, timeout=240):
    """asyncio.run with BOUNDED teardown.  asyncio.run's own
    _cancel_all_tasks sends ONE cancel and gathers without a timeout; a
    straggler task that blocks in a finally-await after that cancel
    hangs the whole pytest process (observed: this file's pg-engine
    test wedging at the 600 s pytest timeout AFTER the test body
    finished).  Cancel leftovers here with a grace period, dump any
    task that survives, and cancel it again so the runner's own pass
    finds nothing left to wait on."""
    import traceback

    async def _main():
        try:
            return await asyncio.wait_for(coro, timeout)
        finally:
            cur = asyncio.current_task()
            stragglers = [t for t in asyncio.all_tasks()
                          if t is not cur and not t.done()]
            for t in stragglers:
                t.cancel()
            if stragglers:
                _done, pending = await asyncio.wait(stragglers, timeout=15)
                for t in pending:
                    print("WEDGED TASK (survived cancel):", t,
                          file=sys.stderr)
                    for f in t.get_stack():
                        traceback.print_stack(f, file=sys.stderr)
                    t.cancel()
                if pending:
                    await asyncio.wait(pending, timeout=10)

    return asyncio.run(_main())


async def _formed(cluster_dir, shard):
    c = DevCluster(cluster_dir, n_peers=3, shard_name=shard,
                   session_timeout_ms=2000, proxied=True,
                   run_snapshotter=False)
    await c.start()
    await c.wait_cluster(
        lambda s: s.get("sync") and len(s.get("async", [])) == 1,
        timeout_s=120, what="3-peer formation (proxied)")
    await c.wait_writable(timeout_s=120)
    w = bench.Writer(c)
    w.start()
    while w.seq < 30:
        await asyncio.sleep(0.05)
    return c, w


async def _put_must_fail(peer, key, timeout_s=2.5):
    """A write against `peer` must NOT be acknowledged."""
    cli = peer.db_client()
    try:
        await cli.put(key, "x", timeout_s=timeout_s)
    except (WaldbError, OSError, asyncio.TimeoutError):
        return
    finally:
        await cli.close()
    raise AssertionError("write to %s was acknowledged but must not be"
                         % peer.id)


async def _deposed_acks_must_drain(old_prim, new_prim, grace_s=4.0):
    """The system's guarantee is NO LOSS, not zero concurrent acks: at
    the promote instant one in-flight sync ack can still let the old
    primary acknowledge a write — but that write is, by the ack's very
    existence, already ON the promoted sync.  So: any straggler ack must
    be for data present on the NEW primary, and acks must cease entirely
    within a short grace window (the severed replication link)."""
    deadline = time.monotonic() + grace_s
    i = 0
    while True:
        key = "split-brain-probe-%d" % i
        i += 1
        cli = old_prim.db_client()
        acked = False
        try:
            await cli.put(key, "x", timeout_s=1.5)
            acked = True
        except Exception:
            pass
        finally:
            await cli.close()
        if not acked:
            return          # the deposed primary's gate is closed
        ncli = new_prim.db_client()
        try:
            got = await ncli.get(key)
        finally:
            await ncli.close()
        async def _st(peer):
            c2 = peer.db_client()
            out = {}
            try:
                if hasattr(c2, "query"):        # waldb json protocol
                    return await c2.query({"q": "status"}, timeout_s=3.0)
                for label, sql in (
                        ("stat_replication",
                         "SELECT * FROM pg_stat_replication;"),
                        ("wal_receiver",
                         "SELECT * FROM pg_stat_wal_receiver;"),
                        ("in_recovery", "SELECT pg_is_in_recovery();"),
                        ("wal_lsn", "SELECT pg_current_wal_lsn();")):
                    try:
                        r = await c2._query(sql, timeout_s=3.0)
                        out[label] = getattr(r, "rows", r)
                    except Exception as exc:
                        out[label] = repr(exc)
                return out
            except Exception as exc:
                return {"error": repr(exc)}
            finally:
                await c2.close()
        if got != "x":
            raise AssertionError(
                "old primary acknowledged a write ABSENT from the new "
                "primary: acked-write loss / split brain\n"
                "old_prim status: %r\nnew_prim status: %r"
                % (await _st(old_prim), await _st(new_prim)))
        if time.monotonic() >= deadline:
            raise AssertionError(
                "old primary still acknowledging writes after the grace "
                "window\nold_prim status: %r\nnew_prim status: %r"
                % (await _st(old_prim), await _st(new_prim)))


def test_primary_cut_from_zk_but_not_from_clients(tmp_path):
    """The classic split-brain shape: the primary loses ZooKeeper while
    still running, still replicating, and still reachable by clients.
    The cluster must promote the sync (generation+1), after which the
    old primary must never acknowledge another write, and no
    acknowledged write may be lost."""
    async def go():
        c, w = await _formed(str(tmp_path / "c"), "1.partzk")
        try:
            s0 = await c.cluster_state()
            prim = c.peer_by_id(s0["primary"]["id"])
            c.partition_zk(prim)

            # session expiry (2 s) → sync takeover with a generation bump
            s1 = await c.wait_cluster(
                lambda s: s["generation"] > s0["generation"] and
                s["primary"]["id"] == s0["sync"]["id"],
                timeout_s=30, what="sync takeover after zk partition")
            newp = await c.wait_writable(timeout_s=30)
            assert newp.id == s0["sync"]["id"]

            # the deposed primary is still alive and reachable by THIS
            # client — any straggler ack must be for data already on the
            # new primary, and acks must stop within the grace window
            await _deposed_acks_must_drain(prim, newp)

            # every previously acknowledged write is intact
            v = await bench.verify_no_loss(c, s1, w)
            assert v["lost"] == 0, v

            # heal: the old primary reconnects, finds itself deposed
            c.heal_zk(prim)
            s2 = await c.wait_cluster(
                lambda s: any(d["id"] == prim.id
                              for d in s.get("deposed", [])),
                timeout_s=30, what="old primary listed as deposed")
            assert s2["primary"]["id"] == newp.id

            # rebuild it; cluster returns to full strength, still no loss
            await c.rebuild_peer(prim)
            await c.wait_writable(timeout_s=60)
            s3 = await c.cluster_state()
            v = await bench.verify_no_loss(c, s3, w)
            assert v["lost"] == 0, v
            await w.stop()
        finally:
            c.stop()
    run(go())


def test_sync_partitioned_from_primary_only(tmp_path):
    """Replication link down, ZooKeeper intact on both sides: no
    topology change may happen (liveness is ZK-based, as in the
    reference), writes stall at the sync-commit gate rather than being
    acknowledged unsafely, and after healing everything acked is
    present."""
    async def go():
        c, w = await _formed(str(tmp_path / "c"), "1.partsync")
        try:
            s0 = await c.cluster_state()
            prim = c.peer_by_id(s0["primary"]["id"])
            sync = c.peer_by_id(s0["sync"]["id"])
            acked_before = w.acked_count
            c.partition(prim, sync)

            # writes must STALL (no unsafe acks without the sync)
            await asyncio.sleep(1.0)   # let in-flight acks drain
            stall_mark = w.acked_count
            await _put_must_fail(prim, "stall-probe")
            await asyncio.sleep(2.0)
            assert w.acked_count <= stall_mark + 1, \
                "writes were acknowledged during the replication partition"

            # ... and the topology must NOT change (both sessions live)
            s1 = await c.cluster_state()
            assert s1["generation"] == s0["generation"]
            assert s1["primary"]["id"] == prim.id
            assert acked_before <= w.acked_count

            c.heal_link(prim, sync)
            await c.wait_writable(timeout_s=60)
            deadline = time.monotonic() + 30
            while w.acked_count < stall_mark + 20:
                assert time.monotonic() < deadline, \
                    "writes did not resume after healing"
                await asyncio.sleep(0.1)
            s2 = await c.cluster_state()
            v = await bench.verify_no_loss(c, s2, w)
            assert v["lost"] == 0, v
            await w.stop()
        finally:
            c.stop()
    run(go())


def test_asymmetric_partition_acks_lost_one_way(tmp_path):
    """The shape SIGSTOP cannot produce: the sync still HEARS the
    primary (WAL keeps arriving and applying) but the primary never
    hears the sync's acknowledgements.  Writes must stall unacked —
    and the probe write, though unacknowledged, is visibly present on
    the sync, proving the asymmetry is real."""
    async def go():
        c, w = await _formed(str(tmp_path / "c"), "1.partasym")
        try:
            s0 = await c.cluster_state()
            prim = c.peer_by_id(s0["primary"]["id"])
            sync = c.peer_by_id(s0["sync"]["id"])

            # drop only sync→primary bytes (acks); primary→sync flows
            c.set_link(sync, prim, drop_a2b=True)
            await asyncio.sleep(0.5)

            # an un-acknowledged write...
            await _put_must_fail(prim, "asym-probe", timeout_s=2.0)

            # ...which the sync nonetheless RECEIVED (one-way link up)
            scli = sync.db_client()
            try:
                got = None
                deadline = time.monotonic() + 10
                while got is None and time.monotonic() < deadline:
                    got = await scli.get("asym-probe")
                    if got is None:
                        await asyncio.sleep(0.1)
                assert got == "x", \
                    "sync never received the WAL for the unacked write"
            finally:
                await scli.close()

            # no topology change: both ZK sessions are healthy
            s1 = await c.cluster_state()
            assert s1["generation"] == s0["generation"]

            c.heal_link(sync, prim)
            await c.wait_writable(timeout_s=60)
            s2 = await c.cluster_state()
            v = await bench.verify_no_loss(c, s2, w)
            assert v["lost"] == 0, v
            await w.stop()
        finally:
            c.stop()
    run(go())


def test_primary_cut_from_zk_postgres_engine(tmp_path):
    """The same classic split-brain shape through the engine=postgres
    path: minipg's replication and the libpq probes all dial through
    the per-link proxies; the deposed primary must refuse acks and no
    acked write may be lost."""
    async def go():
        c = DevCluster(str(tmp_path / "c"), n_peers=3,
                       shard_name="1.pgpartzk", session_timeout_ms=2000,
                       proxied=True, engine="postgres",
                       run_snapshotter=False)
        try:
            await c.start()
            await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=120, what="formation (postgres, proxied)")
            await c.wait_writable(timeout_s=120)
            w = bench.Writer(c)
            w.start()
            while w.seq < 25:
                await asyncio.sleep(0.05)

            s0 = await c.cluster_state()
            prim = c.peer_by_id(s0["primary"]["id"])
            c.partition_zk(prim)
            s1 = await c.wait_cluster(
                lambda s: s["generation"] > s0["generation"],
                timeout_s=30, what="takeover under zk partition (pg)")
            newp = await c.wait_writable(timeout_s=30)
            await _deposed_acks_must_drain(prim, newp)
            v = await bench.verify_no_loss(c, s1, w)
            assert v["lost"] == 0, v
            c.heal_zk(prim)
            await c.wait_cluster(
                lambda s: any(d["id"] == prim.id
                              for d in s.get("deposed", [])),
                timeout_s=30, what="old primary deposed (pg)")
            await w.stop()
        finally:
            c.stop()
    run(go())
