"""Process-level failover matrix — the scenarios of the reference's
integration suite not already covered by test_integ.py
(ref /root/reference/test/integ.test.js exports: asyncDeath,
everyoneDies, pairwise instantaneous deaths, freeze, promote, plus the
live manatee-adm paths)."""

import asyncio
import json
import os
import subprocess
import sys
import time

import pytest

from manatee_amd.adm import core as adm
from manatee_amd.tools.devcluster import DevCluster

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
ADM_BIN = os.path.join(REPO, "bin", "manatee-adm")


def run(coro, timeout=240):
    return asyncio.run(asyncio.wait_for(coro, timeout))


@pytest.fixture
def cluster_dir(tmp_path):
    return str(tmp_path / "cluster")


async def _zk(c: DevCluster):
    return await adm.create_zk_client(c.zk_conn_str)


def test_async_kill9_removed_without_gen_bump(cluster_dir):
    """ref integ.test.js asyncDeath: losing an async never bumps the
    generation — it is just dropped from the async list."""
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.adeath")
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            await c.wait_writable(timeout_s=60)
            gen = s["generation"]
            apeer = c.peer_by_id(s["async"][0]["id"])
            apeer.kill9()
            s2 = await c.wait_cluster(
                lambda s: len(s.get("async", [])) == 0, timeout_s=60,
                what="async removal")
            assert s2["generation"] == gen
            assert s2["primary"]["id"] == s["primary"]["id"]
            assert s2["sync"]["id"] == s["sync"]["id"]
            # writes still work (sync unaffected)
            await c.wait_writable(timeout_s=30)
            # the async comes back and rejoins at the same generation
            apeer.start()
            s3 = await c.wait_cluster(
                lambda s: len(s.get("async", [])) == 1, timeout_s=60,
                what="async rejoin")
            assert s3["generation"] == gen
        finally:
            c.stop()
    run(go())


def test_everyone_dies_cluster_reforms_with_data(cluster_dir):
    """ref integ.test.js everyoneDies: SIGKILL all peers at once; restart
    them; the shard must reform (same or higher generation) with every
    acknowledged write still present."""
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.alldie")
        try:
            await c.start()
            await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            prim = await c.wait_writable(timeout_s=60)
            cli = prim.db_client()
            for i in range(30):
                await cli.put("all%d" % i, i)
            await cli.close()
            s = await c.cluster_state()
            gen = s["generation"]

            for p in c.peers:
                p.kill9()
            await asyncio.sleep(1.0)
            for p in c.peers:
                p.start()

            s2 = await c.wait_cluster(
                lambda s: s.get("sync") is not None
                and s["generation"] >= gen,
                timeout_s=120, what="reformation")
            assert s2["generation"] >= gen
            newp = await c.wait_writable(timeout_s=120)
            cli = newp.db_client()
            for i in range(30):
                assert await cli.get("all%d" % i) == i
            await cli.close()
        finally:
            c.stop()
    run(go())


def test_primary_and_sync_die_no_unsafe_takeover(cluster_dir):
    """Pairwise death of primary+sync: the async must NOT take over (it
    could be missing acknowledged writes).  When the old sync returns it
    takes over as the new primary with everything intact (ref FSM safety:
    new primary must be the previous sync)."""
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.pairdie")
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            prim = await c.wait_writable(timeout_s=60)
            cli = prim.db_client()
            for i in range(20):
                await cli.put("pd%d" % i, i)
            await cli.close()
            gen = s["generation"]
            sync_peer = c.peer_by_id(s["sync"]["id"])
            async_id = s["async"][0]["id"]

            prim.kill9()
            sync_peer.kill9()

            # the async alone must never declare a new generation
            await asyncio.sleep(6.0)
            s2 = await c.cluster_state()
            assert s2["generation"] == gen
            assert s2["primary"]["id"] != async_id

            # old sync returns -> takes over as primary
            sync_peer.start()
            s3 = await c.wait_cluster(
                lambda s: s["generation"] > gen
                and s["primary"]["id"] == sync_peer.id,
                timeout_s=90, what="sync takeover on return")
            assert any(d["id"] == prim.id for d in s3["deposed"])
            newp = await c.wait_writable(timeout_s=90)
            cli = newp.db_client()
            for i in range(20):
                assert await cli.get("pd%d" % i) == i
            await cli.close()
        finally:
            c.stop()
    run(go())


def test_freeze_blocks_takeover_until_unfreeze(cluster_dir):
    """ref docs/user-guide.md:311-334 — a frozen cluster performs no
    transitions; takeover resumes after unfreeze."""
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.frz")
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            prim = await c.wait_writable(timeout_s=60)
            gen = s["generation"]
            zk = await _zk(c)
            try:
                await adm.freeze(zk, c.shard_path, "matrix test")
                prim.kill9()
                await asyncio.sleep(6.0)
                s2 = await c.cluster_state()
                assert s2["generation"] == gen, \
                    "takeover happened despite freeze"
                assert s2.get("freeze")
                await adm.unfreeze(zk, c.shard_path)
            finally:
                await zk.close()
            s3 = await c.wait_cluster(
                lambda s: s["generation"] > gen, timeout_s=90,
                what="takeover after unfreeze")
            assert s3["primary"]["id"] == s["sync"]["id"]
            await c.wait_writable(timeout_s=90)
        finally:
            c.stop()
    run(go())


def test_promote_async_to_sync_live(cluster_dir):
    """Operator promote of the async into the sync slot (gen bump; old
    sync drops to async) driven through the adm promote op, consumed by
    the live FSM (ref promote lib/adm.js:1693-2014)."""
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.promo")
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            await c.wait_writable(timeout_s=60)
            gen = s["generation"]
            old_sync = s["sync"]["id"]
            target = s["async"][0]["id"]
            zk = await _zk(c)
            try:
                await adm.request_promote(zk, c.shard_path, role="async",
                                          peer_id=target)
            finally:
                await zk.close()
            s2 = await c.wait_cluster(
                lambda s: s["generation"] > gen
                and (s.get("sync") or {}).get("id") == target,
                timeout_s=90, what="promote consumption")
            assert "promote" not in s2
            assert any(a["id"] == old_sync for a in s2["async"])
            await c.wait_writable(timeout_s=90)
        finally:
            c.stop()
    run(go())


def test_adm_cli_live_against_cluster(cluster_dir):
    """Run the real bin/manatee-adm against a live shard: pg-status,
    verify, zk-state, zk-active, history, status, freeze/unfreeze."""
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.cli")
        try:
            await c.start()
            await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            await c.wait_writable(timeout_s=60)

            env = dict(os.environ)
            env["ZK_IPS"] = c.zk_conn_str
            env["SHARD"] = c.shard_path
            env.pop("MANATEE_ADM_TEST_STATE", None)

            def cli(*args):
                return subprocess.run(
                    [sys.executable, ADM_BIN] + list(args), env=env,
                    capture_output=True, text=True, timeout=60)

            r = cli("pg-status")
            assert r.returncode == 0, r.stderr
            assert "primary" in r.stdout and "sync" in r.stdout
            assert "ok" in r.stdout and "fail" not in r.stdout

            r = cli("verify", "-v")
            assert r.returncode == 0, r.stdout + r.stderr
            assert "all checks passed" in r.stdout

            r = cli("zk-state")
            assert r.returncode == 0
            state = json.loads(r.stdout)
            assert state["generation"] == 1

            r = cli("zk-active")
            assert r.returncode == 0
            assert len(json.loads(r.stdout)) == 3

            r = cli("history")
            assert r.returncode == 0
            # reference-shaped table (TIME/G#/MODE/... with zoneId abbrs)
            assert r.stdout.splitlines()[0].startswith(
                "TIME                     G# MODE  FRZ PRIMARY")
            assert " multi " in r.stdout
            r = cli("history", "-v")
            assert r.returncode == 0
            assert "cluster setup for normal" in r.stdout   # SUMMARY col
            r = cli("history", "-j")
            assert r.returncode == 0
            ev = json.loads(r.stdout.splitlines()[0])
            assert set(ev) == {"zkSeq", "time", "state"}
            assert ev["state"]["generation"] == 1

            r = cli("status")
            assert r.returncode == 0
            js = json.loads(r.stdout)
            assert c.shard_path in js or \
                c.shard_path.rsplit("/", 1)[1] in js

            r = cli("freeze", "-r", "cli test")
            assert r.returncode == 0
            r = cli("verify")
            # frozen cluster is not an error by itself; verify just checks
            # the replication topology
            r = cli("zk-state")
            assert json.loads(r.stdout).get("freeze")
            r = cli("unfreeze")
            assert r.returncode == 0
            r = cli("zk-state")
            assert not json.loads(r.stdout).get("freeze")
        finally:
            c.stop()
    run(go())


def test_adm_cli_rebuild_live(cluster_dir):
    """Full manatee-adm rebuild of a deposed ex-primary: kill -9 the
    primary, let the sync take over, then rebuild the dead peer through
    the CLI (stop already done; --start-cmd restarts the sitter) and
    watch it rejoin as an async (ref rebuild lib/adm.js:1319-1684)."""
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.rbld")
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            prim = await c.wait_writable(timeout_s=60)
            cli0 = prim.db_client()
            for i in range(10):
                await cli0.put("rb%d" % i, i)
            await cli0.close()
            gen = s["generation"]
            prim.kill9()
            await c.wait_cluster(
                lambda s: s["generation"] > gen
                and any(d["id"] == prim.id for d in s["deposed"]),
                timeout_s=90, what="takeover deposing old primary")
            await c.wait_writable(timeout_s=90)

            env = dict(os.environ)
            env["PYTHONPATH"] = REPO + os.pathsep + \
                env.get("PYTHONPATH", "")
            env.pop("MANATEE_ADM_TEST_STATE", None)
            cfg = os.path.join(prim.dir, "sitter.json")
            pid_file = os.path.join(prim.dir, "sitter-rebuilt.pid")
            start_cmd = (
                "setsid %s -m manatee_amd.daemons.sitter -f %s "
                "--log-file %s >/dev/null 2>&1 & echo $! > %s"
                % (sys.executable, cfg,
                   os.path.join(prim.dir, "sitter-rebuilt.log.json"),
                   pid_file))
            proc = await asyncio.create_subprocess_exec(
                sys.executable, ADM_BIN, "rebuild", "-c", cfg, "-y",
                "--start-cmd", start_cmd, "--timeout", "120",
                env=env, stdout=asyncio.subprocess.PIPE,
                stderr=asyncio.subprocess.STDOUT)
            out, _ = await asyncio.wait_for(proc.communicate(), 180)
            text = out.decode()
            assert proc.returncode == 0, text
            assert "rejoined the cluster" in text

            s2 = await c.cluster_state()
            assert s2["deposed"] == []
            assert any(a["id"] == prim.id for a in s2["async"])
            # data is intact on the new primary
            newp = await c.wait_writable(timeout_s=30)
            dcli = newp.db_client()
            for i in range(10):
                assert await dcli.get("rb%d" % i) == i
            await dcli.close()
            # kill the CLI-spawned sitter (devcluster does not track it)
            import signal
            try:
                with open(pid_file) as f:
                    pid = int(f.read().strip())
                os.killpg(pid, signal.SIGKILL)
            except (OSError, ValueError):
                pass
            prim.kill9()   # kills the restored db child via its pid file
        finally:
            c.stop()
    run(go(), timeout=300)


def test_takeover_promotes_db_online_without_restart(cluster_dir):
    """The sync's database must be promoted IN PLACE on takeover (the
    pg_ctl-promote discipline): same db process before and after the
    failover, timeline bumped."""
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.online")
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            prim = await c.wait_writable(timeout_s=60)
            sync_peer = c.peer_by_id(s["sync"]["id"])
            pid_before = sync_peer.db_pid()
            assert pid_before is not None
            scli = sync_peer.db_client()
            st = await scli.status()
            tl_before = st["timeline"]
            await scli.close()

            prim.kill9()
            await c.wait_cluster(
                lambda s2: s2["generation"] > s["generation"]
                and s2["primary"]["id"] == sync_peer.id,
                timeout_s=60, what="sync takeover")
            await c.wait_writable(timeout_s=60)

            assert sync_peer.db_pid() == pid_before, \
                "takeover restarted the database process"
            scli = sync_peer.db_client()
            st = await scli.status()
            assert st["role"] == "primary"
            assert st["timeline"] == tl_before + 1
            await scli.close()
        finally:
            c.stop()
    run(go())


def test_partitioned_primary_no_split_brain(cluster_dir):
    """Partition analogue (docs/test-plan.md network-partition tier):
    SIGSTOP the whole primary peer.  Its ZK session expires and the
    sync takes over.  The frozen old primary must NOT be able to
    acknowledge writes when it wakes (its sync standby is gone — the
    remote_write gate blocks), and once its sitter resumes it must
    observe that it is deposed and stop serving as primary."""
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.part")
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            prim = await c.wait_writable(timeout_s=60)
            gen = s["generation"]

            prim.pause()
            s2 = await c.wait_cluster(
                lambda s2: s2["generation"] > gen
                and s2["primary"]["id"] == s["sync"]["id"],
                timeout_s=60, what="takeover around partitioned primary")
            assert any(d["id"] == prim.id for d in s2["deposed"])
            new_prim = await c.wait_writable(timeout_s=60)

            # wake the old primary: a write against it must NOT be
            # acknowledged (sync gate) and the peer must demote itself
            prim.resume()
            cli = prim.db_client()
            acked = False
            try:
                await cli.put("split-brain", 1, timeout_s=2.0)
                acked = True
            except Exception:
                pass
            await cli.close()
            assert not acked, \
                "deposed ex-primary acknowledged a write (split brain!)"

            # the resumed sitter sees it is deposed and stops its db
            deadline = time.monotonic() + 60
            demoted = False
            while time.monotonic() < deadline:
                cli = prim.db_client()
                try:
                    st = await asyncio.wait_for(cli.status(), 1.0)
                    if st.get("role") != "primary":
                        demoted = True
                except Exception:
                    demoted = True   # db stopped entirely — also fine
                finally:
                    await cli.close()
                if demoted:
                    break
                await asyncio.sleep(0.5)
            assert demoted, "resumed deposed primary still acts as primary"

            # the new primary still works and never lost the write path
            cli = new_prim.db_client()
            await cli.put("after-partition", 2)
            assert await cli.get("after-partition") == 2
            await cli.close()
        finally:
            c.stop()
    run(go())


def test_5peer_cascading_chain_and_middle_death(cluster_dir):
    """Five peers form primary -> sync -> async0 -> async1 -> async2
    with each standby following its chain predecessor (cascading
    replication, ref docs/user-guide.md:69-90,258-266); data reaches the
    chain tail; killing a middle async re-slaves its successor."""
    async def go():
        c = DevCluster(cluster_dir, n_peers=5, shard_name="1.chain")
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 3,
                timeout_s=90, what="5-peer formation")
            prim = await c.wait_writable(timeout_s=90)

            # verify the replication edges against the CURRENT state
            # (async ordering can be revised while peers settle): each
            # peer must stream to its chain successor
            deadline = time.monotonic() + 60
            missing = None
            while time.monotonic() < deadline:
                s = await c.cluster_state()
                chain = [s["primary"], s["sync"]] + s["async"]
                missing = None
                for up, down in zip(chain[:-1], chain[1:]):
                    cli = c.peer_by_id(up["id"]).db_client()
                    try:
                        st = await cli.status()
                    except Exception:
                        st = {}
                    finally:
                        await cli.close()
                    if not any(r["application_name"] == down["id"]
                               and r["state"] == "streaming"
                               for r in st.get("replication", [])):
                        missing = "%s -> %s" % (up["id"], down["id"])
                        break
                if missing is None:
                    break
                await asyncio.sleep(0.3)
            assert missing is None, "no stream " + missing

            # a write reaches the chain tail
            cli = prim.db_client()
            await cli.put("chain", "deep")
            await cli.close()
            tail = c.peer_by_id(s["async"][-1]["id"])
            tcli = tail.db_client()
            deadline = time.monotonic() + 30
            while time.monotonic() < deadline:
                if await tcli.get("chain") == "deep":
                    break
                await asyncio.sleep(0.1)
            assert await tcli.get("chain") == "deep"
            await tcli.close()

            # kill the middle async: successor re-slaves, no gen bump
            gen = s["generation"]
            mid = c.peer_by_id(s["async"][0]["id"])
            succ_id = s["async"][1]["id"]
            mid.kill9()
            s2 = await c.wait_cluster(
                lambda s2: len(s2["async"]) == 2
                and s2["async"][0]["id"] == succ_id,
                timeout_s=60, what="middle async removal")
            assert s2["generation"] == gen
            # the promoted-up async now streams from the sync
            sync_peer = c.peer_by_id(s["sync"]["id"])
            scli = sync_peer.db_client()
            deadline = time.monotonic() + 30
            ok = False
            while time.monotonic() < deadline and not ok:
                st = await scli.status()
                ok = any(r["application_name"] == succ_id
                         for r in st.get("replication", []))
                if not ok:
                    await asyncio.sleep(0.2)
            await scli.close()
            assert ok, "successor did not re-slave to the sync"
            await c.wait_writable(timeout_s=30)
        finally:
            c.stop()
    run(go())


def test_metrics_endpoint(cluster_dir):
    """GET /metrics serves Prometheus text with role/generation/health/
    replication series (observability beyond the reference, which has
    no metrics endpoint — SURVEY.md §5.5)."""
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.metrics")
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            prim = await c.wait_writable(timeout_s=60)
            code, text = await prim.http_status("/metrics")
            assert code == 200
            assert isinstance(text, str)
            assert "manatee_role 0" in text
            assert "manatee_db_writable 1" in text
            assert "manatee_db_healthy 1" in text
            assert "manatee_generation 1" in text
            assert "manatee_cluster_frozen 0" in text
            assert "manatee_wal_lsn_bytes" in text
            assert 'manatee_replication_unflushed_bytes{downstream="%s"' \
                % s["sync"]["id"] in text

            apeer = c.peer_by_id(s["async"][0]["id"])
            code, text = await apeer.http_status("/metrics")
            assert code == 200
            assert "manatee_role 2" in text
            assert "manatee_db_writable 0" in text
        finally:
            c.stop()
    run(go())


def test_kill_sync_before_repl_established(cluster_dir):
    """MANATEE-212 regression (ref integ.test.js
    MANATEE_212_killSyncBeforeRepl*): the sync is SIGKILLed while the
    primary is still read-only waiting for it to catch up.  The primary
    must replace it with the async and open writes — never deadlock on
    the dead sync."""
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.m212")
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            await c.wait_writable(timeout_s=60)
            gen = s["generation"]
            # force a sync swap: kill the current sync; as soon as the
            # NEXT state names the async as sync (primary read-only,
            # waiting for catch-up), kill THAT sync too before it can
            # finish catching up
            first_sync = c.peer_by_id(s["sync"]["id"])
            first_sync.kill9()
            s2 = await c.wait_cluster(
                lambda s2: s2["generation"] > gen
                and s2["sync"]["id"] == s["async"][0]["id"],
                timeout_s=60, what="first sync replacement")
            second_sync = c.peer_by_id(s2["sync"]["id"])
            second_sync.kill9()
            # no spare async: the shard must hold read-only (no unsafe
            # takeover) until a peer returns…
            await asyncio.sleep(3.0)
            s3 = await c.cluster_state()
            assert s3["primary"]["id"] == s["primary"]["id"]
            # …then the first sync comes back and the shard heals
            first_sync.start()
            await c.wait_cluster(
                lambda s4: s4.get("sync") is not None
                and s4["sync"]["id"] == first_sync.id,
                timeout_s=90, what="sync re-established")
            await c.wait_writable(timeout_s=90)
        finally:
            c.stop()
    run(go())


def test_rapid_sequenced_kills_converge(cluster_dir):
    """MANATEE-207-style no-wait kills: SIGKILL the primary, then
    SIGKILL its successor the moment it takes over, without letting the
    shard settle.  Once peers restart, the shard must converge with all
    acknowledged writes intact."""
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.m207")
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            prim = await c.wait_writable(timeout_s=60)
            cli = prim.db_client()
            for i in range(25):
                await cli.put("m207-%d" % i, i)
            await cli.close()
            gen = s["generation"]
            sync_peer = c.peer_by_id(s["sync"]["id"])

            prim.kill9()
            # the moment the sync declares itself primary, kill it too
            await c.wait_cluster(
                lambda s2: s2["generation"] > gen
                and s2["primary"]["id"] == sync_peer.id,
                timeout_s=60, what="first takeover")
            sync_peer.kill9()

            # restart both dead peers; the shard must converge writable
            await asyncio.sleep(1.0)
            prim.start()
            sync_peer.start()
            newp = await c.wait_writable(timeout_s=120)
            cli = newp.db_client()
            for i in range(25):
                assert await cli.get("m207-%d" % i) == i
            await cli.close()
        finally:
            c.stop()
    run(go(), timeout=300)


def test_full_zk_outage_shard_survives_and_recovers(cluster_dir):
    """Full-ZK-outage tier (ref docs/test-plan.md): SIGKILL the
    coordination server under a live shard.  The databases keep serving
    (reads AND acked writes — the replication chain is already
    configured), no peer declares anything during the outage, and when
    ZK returns (journal-recovered state) the peers rebuild their
    sessions and the topology converges without a generation conflict
    or data loss."""
    async def go():
        c = DevCluster(cluster_dir, n_peers=3, shard_name="1.zkout")
        try:
            await c.start()
            s = await c.wait_cluster(
                lambda s: s.get("sync") and len(s.get("async", [])) == 1,
                timeout_s=60, what="formation")
            prim = await c.wait_writable(timeout_s=60)
            cli = prim.db_client()
            for i in range(20):
                await cli.put("zo%d" % i, i)

            c.kill_zk()
            await asyncio.sleep(3.0)
            # the data plane is unaffected by the coordination outage
            for i in range(20, 40):
                await cli.put("zo%d" % i, i)
            assert await cli.get("zo5") == 5

            c.start_zk()
            await c.wait_zk()
            # state survived via the journal; peers re-form around it
            s2 = await c.wait_cluster(
                lambda s2: s2.get("sync") is not None, timeout_s=90,
                what="re-formation after ZK outage")
            assert s2["generation"] >= s["generation"]
            assert s2["primary"]["id"] == prim.id, \
                "primary must not change across a pure ZK outage"
            newp = await c.wait_writable(timeout_s=90)
            ncli = newp.db_client()
            for i in range(40):
                assert await ncli.get("zo%d" % i) == i
            await ncli.close()
            await cli.close()
        finally:
            c.stop()
    run(go())
