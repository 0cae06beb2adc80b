"""Dedicated-box (gpu-marked) tiers: the long-running chaos and
benchmark-contract checks that need a quiet machine to themselves.

This build is the re-tiered dist_sys framework directed by BASELINE.json
(the reference manages PostgreSQL failover; there are no GPU kernels),
so these tests exercise the full-cluster control/data plane on the
dedicated box rather than device code.
"""

import asyncio
import json
import os
import random
import subprocess
import sys
import time

import pytest

from manatee_amd.tools.devcluster import DevCluster

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

pytestmark = pytest.mark.gpu


def run(coro, timeout=600):
    return asyncio.run(asyncio.wait_for(coro, timeout))


@pytest.mark.timeout(1000)
def test_chaos_random_kills_zero_acked_loss(tmp_path):
    """Randomized SIGKILL chaos under continuous synchronously-replicated
    write load (the docs/test-plan.md tier): every cycle kills a random
    peer (primary, sync, or async), waits for the shard to converge and
    become writable, verifies every previously-acknowledged write, then
    heals back to primary/sync/async."""
    async def go():
        rng = random.Random(1234)
        c = DevCluster(str(tmp_path / "chaos"), n_peers=3,
                       shard_name="1.chaos")
        acked = {}
        seq = 0
        try:
            await c.start()
            await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=120, what="formation")
            await c.wait_writable(timeout_s=120)

            async def pump(n):
                nonlocal seq
                s = await c.cluster_state()
                prim = c.peer_by_id(s["primary"]["id"])
                cli = prim.db_client()
                try:
                    for _ in range(n):
                        await cli.put("c%d" % seq, seq, timeout_s=5.0)
                        acked["c%d" % seq] = seq
                        seq += 1
                finally:
                    await cli.close()

            await pump(100)
            for cycle in range(6):
                s = await c.cluster_state()
                victims = [s["primary"]["id"], s["sync"]["id"]] + \
                    [a["id"] for a in s["async"]]
                victim_id = rng.choice(victims)
                role = ("primary" if victim_id == s["primary"]["id"] else
                        "sync" if victim_id == s["sync"]["id"] else
                        "async")
                victim = c.peer_by_id(victim_id)
                victim.kill9()
                # converge + writable
                prim = await c.wait_writable(timeout_s=120)
                # verify all acknowledged writes
                cli = prim.db_client()
                lost = 0
                for k, v in acked.items():
                    if await cli.get(k) != v:
                        lost += 1
                await cli.close()
                assert lost == 0, ("cycle %d (%s killed): lost %d/%d "
                                   "acked writes"
                                   % (cycle, role, lost, len(acked)))
                # heal: restart/rebuild the victim, wait for full shape
                s2 = await c.cluster_state()
                if any(d["id"] == victim_id
                       for d in s2.get("deposed", [])):
                    await c.rebuild_peer(victim)
                else:
                    if not victim.alive():
                        victim.start()
                await c.wait_cluster(
                    lambda s: s.get("sync")
                    and len(s.get("async", [])) >= 1
                    and not s.get("deposed"),
                    timeout_s=120, what="heal after cycle %d" % cycle)
                await pump(50)
        finally:
            c.stop()
    run(go(), timeout=900)


@pytest.mark.timeout(1800)
def test_soak_50_cycles_zero_acked_loss():
    """Driver-attested chaos depth: a ≥50-cycle randomized soak
    (SIGKILLs of every role, db-child kill, SIGSTOP, full-ZK outage,
    and real network partitions through the per-link proxy layer —
    including the asymmetric shapes) under pipelined synchronous write
    load, zero acknowledged-write loss, exact verification after every
    cycle (tools/soak.py)."""
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run(
        [sys.executable, "-m", "manatee_amd.tools.soak",
         "--cycles", "50", "--minutes", "20", "--seed", "42"],
        capture_output=True, text=True, timeout=1500, env=env, cwd=REPO)
    tail = "\n".join(r.stderr.splitlines()[-12:])
    assert r.returncode == 0, tail + "\n" + r.stdout
    stats = json.loads(r.stdout.splitlines()[-1])
    assert stats["ok"], stats
    assert stats["cycles"] >= 50, stats
    assert stats["lost"] == 0, stats
    assert not stats["failures"], stats
    # every fault family must actually have fired across 50 cycles
    assert len(stats["kills"]) >= 7, stats["kills"]


@pytest.mark.timeout(700)
def test_bench_contract_runs_and_reports():
    """bench.py must emit one valid JSON line with the BASELINE.json
    metric and zero acknowledged-write loss."""
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    t0 = time.monotonic()
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--steps", "3", "--warmup", "1"],
        capture_output=True, text=True, timeout=600, env=env, cwd=REPO)
    elapsed = time.monotonic() - t0
    assert r.returncode == 0, r.stdout + r.stderr
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["metric"].startswith("failover-to-writable")
    assert out["higher_is_better"] is False
    assert out["value"] > 0
    assert out["config"]["acked_writes_lost"] == 0
    assert "INVALID" not in out["config"]
    # sanity: the run really did the steps inside the wall clock
    assert out["ms_per_step"] * out["steps"] / 1000.0 <= elapsed + 1


@pytest.mark.timeout(700)
def test_bench_postgres_engine_failover():
    """The engine=postgres management path (minipg binaries, libpq
    writes, pg_stat_replication gating) through the same failover
    benchmark: zero acknowledged-write loss, well inside the reference's
    30 s convergence bound."""
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--engine", "postgres", "--steps", "3", "--warmup", "1"],
        capture_output=True, text=True, timeout=600, env=env, cwd=REPO)
    assert r.returncode == 0, r.stdout + r.stderr
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["config"]["engine"] == "postgres"
    assert out["config"]["acked_writes_lost"] == 0
    assert out["value"] < 30.0, "failover slower than the reference bound"


def test_native_codec_is_loaded_and_used():
    """On the dedicated box the in-tree C++ codec must be present (the
    snapshot carries the built .so) and wal.py must be using it — the
    replication path may not silently run the pure-Python fallback."""
    from manatee_amd.db.waldb import wal as walmod
    from manatee_amd.native import codec
    assert codec is not None, "native codec extension missing on the box"
    assert walmod._codec() is codec
    payload = b"x" * 1000
    frame = codec.encode_frame(payload)
    assert list(walmod.parse_frames(frame)) == [(len(frame), payload)]
    # same for the native jute wire codec: the ZK client/server must be
    # running over the C++ primitives, not the pure-Python fallback
    from manatee_amd.coord import jute
    from manatee_amd.native import jutec
    assert jutec is not None, "native _jutec extension missing on the box"
    assert jute.CODEC == "native"
    assert jute.Writer is jutec.Writer
