"""Regression test for the bench's zero-acknowledged-write-loss check.

The check must be EXACT: post-failover acks must not be able to pad the
server-side count over a lost write (the writer is frozen during
verification), and an injected loss of a single acknowledged key must be
detected.  Guards the verification logic in bench.py::verify_no_loss.
"""

import asyncio
import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import bench  # noqa: E402
from manatee_amd.tools.devcluster import DevCluster  # noqa: E402


def run(coro, timeout=180):
    return asyncio.run(asyncio.wait_for(coro, timeout))


async def _formed_cluster_with_writer(cluster_dir):
    c = DevCluster(cluster_dir, n_peers=3, shard_name="1.benchloss",
                   session_timeout_ms=bench.SESSION_TIMEOUT_MS)
    await c.start()
    await c.wait_cluster(
        lambda s: s.get("sync") and len(s.get("async", [])) == 1,
        timeout_s=120, what="formation")
    await c.wait_writable(timeout_s=120)
    w = bench.Writer(c)
    w.start()
    while w.seq < 50:
        await asyncio.sleep(0.05)
    return c, w


def test_injected_loss_is_detected(tmp_path):
    """Delete one acknowledged key on the new primary after failover but
    before verification — the loss check must report it, even though the
    writer keeps producing fresh acks on the new primary up to the moment
    verification freezes it."""
    async def go():
        c, w = await _formed_cluster_with_writer(str(tmp_path / "c1"))
        try:
            async def corrupt(cluster, state):
                # an early acked key: a masked-by-count bug would hide this
                newp = cluster.peer_by_id(state["primary"]["id"])
                cli = newp.db_client()
                try:
                    await cli.delete("bench-5")
                finally:
                    await cli.close()

            bench._pre_verify_hook = corrupt
            r = await bench.one_failover(c, w)
            assert r["lost_acked_writes"] >= 1, \
                "injected loss of an acked write was NOT detected"
            await w.stop()
        finally:
            bench._pre_verify_hook = None
            c.stop()
    run(go())


def test_writer_pause_freezes_ack_count(tmp_path):
    """While paused, no new acks arrive and acked_count == seq (the exact
    set bench-0..seq-1) — the invariant the count check relies on."""
    async def go():
        c, w = await _formed_cluster_with_writer(str(tmp_path / "c2"))
        try:
            await w.pause()
            n1 = w.acked_count
            assert n1 == w.seq
            await asyncio.sleep(0.5)
            assert w.acked_count == n1, "acks arrived while paused"
            w.resume()
            deadline = asyncio.get_running_loop().time() + 10
            while w.acked_count == n1:
                assert asyncio.get_running_loop().time() < deadline
                await asyncio.sleep(0.02)
            await w.stop()
        finally:
            c.stop()
    run(go())
