"""Time/cycle-bounded chaos soak: a live shard under continuous
synchronous write load with randomized faults, verifying zero
acknowledged-write loss after every cycle.

    python -m manatee_amd.tools.soak --minutes 10 [--cycles 50]
        [--seed 7] [-d DIR]

Faults drawn each cycle: SIGKILL primary / sync / async, SIGKILL just
the database child (sitter must restart it), SIGSTOP+SIGCONT the
primary, full-ZK outage — plus real NETWORK partitions through the
per-link proxy layer (primary cut from ZK only, replication link cut,
asymmetric ack loss; tools/netproxy).  After every fault the shard must
converge back to writable with every previously-acknowledged write
present (writer frozen, exact server-side count + recent-window
readback), then the shard is healed to full primary/sync/async shape.
One JSON summary line on stdout at the end; exit 1 on any lost write
or convergence failure.
"""

from __future__ import annotations

import argparse
import asyncio
import json
import random
import shutil
import sys
import tempfile
import time
from typing import Optional

from .devcluster import DevCluster

WINDOW = 4000
BURST = 25


class SoakWriter:
    def __init__(self, cluster: DevCluster):
        self.cluster = cluster
        self.window = {}
        self.acked_count = 0
        self.seq = 0
        self.stop_flag = False
        self.pause_flag = False
        self._paused = asyncio.Event()
        self.task = None

    async def _run(self):
        cli = None
        while not self.stop_flag:
            if self.pause_flag:
                self._paused.set()
                await asyncio.sleep(0.01)
                continue
            try:
                if cli is None:
                    s = await self.cluster.cluster_state()
                    if s is None:
                        await asyncio.sleep(0.05)
                        continue
                    peer = self.cluster.peer_by_id(s["primary"]["id"])
                    cli = peer.db_client()
                # pipelined burst: high sustained write pressure
                base = self.seq
                items = [("soak-%d" % (base + j), base + j)
                         for j in range(BURST)]
                await cli.put_many(items, timeout_s=2.0)
                for k, v in items:
                    self.window[k] = v
                self.acked_count += len(items)
                while len(self.window) > WINDOW:
                    self.window.pop(next(iter(self.window)))
                self.seq += len(items)
            except Exception:
                if cli is not None:
                    await cli.close()
                cli = None
                await asyncio.sleep(0.02)
        if cli is not None:
            await cli.close()

    def start(self):
        self.task = asyncio.get_running_loop().create_task(self._run())

    async def pause(self):
        """Freeze at a burst boundary so the acked set is exactly
        soak-0..seq-1 and no new acks can pad the verification count."""
        self._paused.clear()
        self.pause_flag = True
        await self._paused.wait()

    def resume(self):
        self.pause_flag = False

    async def stop(self):
        self.stop_flag = True
        self.pause_flag = False
        if self.task is not None:
            await self.task


async def verify(cluster: DevCluster, writer: SoakWriter) -> int:
    """Exact loss check: freeze the writer, count the prefix, subtract
    any committed-but-unacked keys from the single abandoned in-flight
    burst (they are exactly soak-seq..soak-seq+BURST-1), and read back
    the recent window."""
    await writer.pause()
    try:
        last_exc = None
        for _attempt in range(3):
            s = await cluster.cluster_state()
            cli = cluster.peer_by_id(s["primary"]["id"]).db_client()
            lost = 0
            try:
                present = await cli.count(prefix="soak-", timeout_s=30.0)
                for j in range(BURST):
                    if await cli.get("soak-%d" % (writer.seq + j)) \
                            is not None:
                        present -= 1
                if present < writer.acked_count:
                    lost += writer.acked_count - present
                # spot-check a sample of the recent window (the exact
                # count above already catches any missing key; this
                # guards values)
                items = list(writer.window.items())
                sample = items[-200:] + items[:50]
                for key, val in sample:
                    if await cli.get(key) != val:
                        lost += 1
                return lost
            except Exception as exc:   # mid-verify failover: retry fresh
                last_exc = exc
                await asyncio.sleep(1.0)
            finally:
                await cli.close()
        raise RuntimeError("verification failed repeatedly: %r" % last_exc)
    finally:
        writer.resume()


async def dump_stall(cluster: DevCluster, action: str) -> None:
    """Diagnostics for a cycle that failed to become writable inside
    the fault window (before the operator runbook runs): cluster state
    plus every reachable db's own view.  The rare >60 s outliers in
    long soaks are only debuggable from this moment's state."""
    try:
        s = await cluster.cluster_state()
        print("# STALL[%s] cluster_state: %s"
              % (action, json.dumps(s)[:700] if s else None),
              file=sys.stderr)
        for p in cluster.peers:
            cli = p.db_client()
            try:
                st = await asyncio.wait_for(cli.status(), 2.0)
                print("# STALL[%s] db %s: role=%s ro=%s sync=%s up=%s "
                      "cur=%s repl=%s"
                      % (action, p.id, st.get("role"),
                         st.get("read_only"), st.get("sync_standby"),
                         st.get("upstream_status"),
                         st.get("current_lsn"),
                         json.dumps(st.get("replication"))[:300]),
                      file=sys.stderr)
            except Exception as exc:
                print("# STALL[%s] db %s: unreachable (%r)"
                      % (action, p.id, exc), file=sys.stderr)
            finally:
                try:
                    await cli.close()
                except Exception:
                    pass
            try:
                _code, body = await asyncio.wait_for(
                    p.http_status("/state"), 2.0)
                st = body if isinstance(body, dict) else json.loads(body)
                print("# STALL[%s] sitter %s: role=%s state=%s "
                      "dbOnline=%s gen=%s"
                      % (action, p.id, st.get("role"),
                         st.get("peerState"), st.get("dbOnline"),
                         (st.get("clusterState") or {}).get("generation")),
                      file=sys.stderr)
            except Exception as exc:
                print("# STALL[%s] sitter %s: unreachable (%r)"
                      % (action, p.id, exc), file=sys.stderr)
    except Exception as exc:
        print("# STALL[%s] dump failed: %r" % (action, exc),
              file=sys.stderr)


ACTIONS = ["kill_primary", "kill_sync", "kill_async",
           "kill_db_only", "pause_primary", "zk_outage",
           "partition_zk_primary", "partition_repl", "partition_asym"]


async def soak(minutes: float, seed: int, workdir: str,
               cycles: Optional[int] = None,
               engine: str = "waldb") -> dict:
    rng = random.Random(seed)
    c = DevCluster(workdir, n_peers=3, shard_name="1.soak", proxied=True,
                   engine=engine)
    stats = {"cycles": 0, "kills": {}, "lost": 0, "acked": 0,
             "max_failover_s": 0.0, "failures": []}
    writer = SoakWriter(c)
    # SIGTERM (timeout-bounded runs) must unwind through the finally so
    # the cluster is torn down, not orphaned
    import signal as _signal
    loop = asyncio.get_running_loop()
    me = asyncio.current_task()
    for _sig in (_signal.SIGTERM, _signal.SIGINT):
        try:
            loop.add_signal_handler(_sig, me.cancel)
        except (NotImplementedError, RuntimeError):
            pass
    try:
        await c.start()
        await c.wait_cluster(
            lambda s: s.get("sync") and len(s.get("async", [])) == 1,
            timeout_s=120, what="formation")
        await c.wait_writable(timeout_s=120)
        writer.start()
        while writer.seq < 100:
            await asyncio.sleep(0.05)

        deadline = time.monotonic() + minutes * 60.0
        while time.monotonic() < deadline and \
                (cycles is None or stats["cycles"] < cycles):
            s = await c.cluster_state()
            action = rng.choice(ACTIONS)
            stats["kills"][action] = stats["kills"].get(action, 0) + 1
            prim = c.peer_by_id(s["primary"]["id"])
            sync = c.peer_by_id(s["sync"]["id"])
            victim = None
            t0 = time.monotonic()
            if action == "kill_primary":
                victim = prim
                victim.kill9()
            elif action == "kill_sync":
                victim = sync
                victim.kill9()
            elif action == "kill_async":
                victim = c.peer_by_id(s["async"][0]["id"])
                victim.kill9()
            elif action == "kill_db_only":
                prim.kill_db_only()
            elif action == "pause_primary":
                prim.pause()
            elif action == "zk_outage":
                c.kill_zk()
                await asyncio.sleep(rng.uniform(1.0, 4.0))
                c.start_zk()
                await c.wait_zk()
            elif action == "partition_zk_primary":
                # primary cut from ZK only: session expiry → sync
                # takeover; the old primary is deposed on heal.  A
                # takeover slower than the window is NOT a safety
                # violation (writes keep flowing or stall, nothing is
                # lost) — heal, record it, and let the generic
                # writable-wait + runbook converge the shard.
                c.partition_zk(prim)
                try:
                    await c.wait_cluster(
                        lambda st: st["generation"] > s["generation"],
                        timeout_s=45, what="takeover under zk partition")
                except AssertionError as exc:
                    stats.setdefault("slow_zk_takeovers", 0)
                    stats["slow_zk_takeovers"] += 1
                    print("# WARN: %s" % exc, file=sys.stderr)
                    await dump_stall(c, action)
                finally:
                    c.heal_zk(prim)
            elif action == "partition_repl":
                # replication link down, ZK intact: writes stall but
                # topology must hold; then heal
                c.partition(prim, sync)
                await asyncio.sleep(rng.uniform(2.0, 4.0))
                c.heal_link(prim, sync)
            elif action == "partition_asym":
                # one-way: the primary stops hearing the sync's acks
                c.set_link(sync, prim, drop_a2b=True)
                await asyncio.sleep(rng.uniform(2.0, 4.0))
                c.heal_link(sync, prim)

            try:
                await c.wait_writable(timeout_s=60)
            except AssertionError:
                await dump_stall(c, action)
                # legitimately unavailable states exist (e.g. primary
                # dead while the only other healthy peer is the sync and
                # the third is deposed) — run the operator runbook:
                # restart dead peers, rebuild deposed ones, then the
                # shard must become writable
                try:
                    if action == "pause_primary":
                        prim.resume()
                    s_stuck = await c.cluster_state()
                    deposed = {d["id"]
                               for d in (s_stuck or {}).get("deposed", [])}
                    for p in c.peers:
                        if not p.alive():
                            p.start()
                    await asyncio.sleep(2.0)
                    for p in c.peers:
                        if p.id in deposed:
                            await c.rebuild_peer(p, timeout_s=120)
                    await c.wait_writable(timeout_s=120)
                except Exception as exc:
                    stats["failures"].append("%s: %s" % (action, exc))
                    break
            if action == "pause_primary":
                prim.resume()
            failover_s = time.monotonic() - t0
            stats["max_failover_s"] = max(stats["max_failover_s"],
                                          failover_s)

            lost = await verify(c, writer)
            stats["lost"] += lost
            stats["cycles"] += 1
            print("# cycle %d: %s, %.2fs to writable, lost=%d "
                  "(acked=%d)" % (stats["cycles"], action, failover_s,
                                  lost, writer.acked_count),
                  file=sys.stderr)
            if lost:
                break

            # heal back to full shape
            s2 = await c.cluster_state()
            deposed = {d["id"] for d in s2.get("deposed", [])}
            for p in c.peers:
                if p.id in deposed:
                    await c.rebuild_peer(p, timeout_s=120)
                elif not p.alive():
                    p.start()
            await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) >= 1
                and not s.get("deposed"),
                timeout_s=120, what="heal")
            await c.wait_writable(timeout_s=120)
        await writer.stop()
        stats["acked"] = writer.acked_count
    finally:
        c.stop()
    return stats


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(prog="manatee-soak")
    ap.add_argument("--minutes", type=float, default=5.0,
                    help="hard wall-clock budget")
    ap.add_argument("--cycles", type=int, default=None,
                    help="stop after N fault cycles (within the budget)")
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--engine", choices=("waldb", "postgres"),
                    default="waldb")
    ap.add_argument("-d", "--dir", default=None)
    ns = ap.parse_args(argv)
    workdir = ns.dir or tempfile.mkdtemp(prefix="manatee-soak-")
    try:
        stats = asyncio.run(soak(ns.minutes, ns.seed, workdir,
                                 cycles=ns.cycles, engine=ns.engine))
    finally:
        if ns.dir is None:
            shutil.rmtree(workdir, ignore_errors=True)
    ok = not stats["lost"] and not stats["failures"]
    stats["ok"] = ok
    stats["engine"] = ns.engine
    print(json.dumps(stats))
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
