"""Deterministic FSM simulator + randomized invariant fuzzer.

The reference's cluster state machine ships with a discrete-event
simulator and is tested against it rather than against live clusters
(referenced at lib/adm.js:592-594).  This module is
that tool for this build: it runs real ``ManateePeer`` FSMs and the
real coordination layer (embedded ZK server) against scriptable mock
databases, drives randomized fault schedules from a seed, and checks
the safety invariants after every event:

- every recorded state transition is LEGAL (checked with the same
  annotate_history rules the CLI uses: generation never decreases, a
  new primary was the previous sync, sync changes bump the
  generation);
- at most one live peer acts as primary, and only the one the cluster
  state names (no split brain at the FSM level);
- peers listed as deposed never run a database;
- role assignments are disjoint (a peer appears at most once across
  primary/sync/async/deposed).

CLI:  python -m manatee_amd.fsm.sim --seed 7 --steps 40 --peers 3 [-v]
"""

from __future__ import annotations

import asyncio
import json
import random
from typing import Dict, List, Optional

from ..common import lsn as lsnmod
from ..coord.zkmgr import ZkMgr
from ..coord.zkserver import ZkServer
from . import state as st
from .peer import ManateePeer

SHARD_PATH = "/manatee/simshard"


class MockDb:
    """Scriptable database manager implementing the FSM-facing contract
    (ref lib/postgresMgr.js events :401-421, reconfigure :758-867)."""

    def __init__(self, xlog: str = lsnmod.ZERO):
        self._listeners: Dict[str, list] = {}
        self.configs: List[dict] = []
        self.current: Optional[dict] = None
        self.xlog = xlog
        self.fail_reconfigure = False

    def on(self, event, cb):
        self._listeners.setdefault(event, []).append(cb)

    def emit(self, event, *args):
        for cb in self._listeners.get(event, []):
            cb(*args)

    def fire_init(self, setup=False, online=False):
        # setup=False: a fresh database (the normal formation case);
        # peers with already-initialized databases refuse to auto-form
        self.emit("init", {"setup": setup, "online": online})

    async def reconfigure(self, cfg):
        if self.fail_reconfigure:
            raise RuntimeError("injected reconfigure failure")
        self.configs.append(cfg)
        self.current = cfg

    async def get_xlog_location(self):
        return self.xlog

    async def stop(self):
        pass

    @property
    def role(self):
        return self.current["role"] if self.current else None


class SimPeer:
    def __init__(self, ip: str, srv: ZkServer, singleton=False,
                 session_timeout_ms=1000, xlog=lsnmod.ZERO,
                 shard_path: str = SHARD_PATH):
        self.ip = ip
        self.id = "%s:5432:12345" % ip
        self.ident = st.make_ident(self.id, zone_id="zone-" + ip)
        self.zk = ZkMgr(id=self.id,
                        data={k: v for k, v in self.ident.items()
                              if k != "id"},
                        path=shard_path, conn_str=srv.conn_str,
                        session_timeout_ms=session_timeout_ms)
        self.db = MockDb(xlog=xlog)
        self.fsm = ManateePeer(zk=self.zk, db=self.db,
                               self_ident=self.ident,
                               singleton=singleton, tick_interval_s=0.1)

    async def start(self):
        self.fsm.start()
        await self.zk.init()
        self.db.fire_init()

    async def kill(self):
        """SIGKILL analogue: sever the ZK session without clean close."""
        await self.fsm.close()
        self.zk._closed = True
        cli = self.zk._zk
        if cli is not None:
            cli._closing = True
            if cli._writer is not None:
                cli._writer.close()
            for t in (cli._mgr_task, cli._io_task, cli._ping_task):
                if t is not None:
                    t.cancel()

    async def close(self):
        await self.fsm.close()
        await self.zk.close()


class SimShard:
    def __init__(self, session_timeout_ms=1000,
                 shard_path: str = SHARD_PATH):
        self.srv: Optional[ZkServer] = None
        self.peers: Dict[str, SimPeer] = {}
        self.session_timeout_ms = session_timeout_ms
        self.shard_path = shard_path

    async def start(self, n_peers=3, singleton=False):
        self.srv = ZkServer(tick_ms=50, min_session_timeout_ms=300)
        await self.srv.start()
        for i in range(n_peers):
            await self.add_peer("10.0.0.%d" % (i + 1), singleton=singleton)
            await asyncio.sleep(0.05)
        return self

    async def add_peer(self, ip, singleton=False, xlog=lsnmod.ZERO):
        p = SimPeer(ip, self.srv, singleton=singleton,
                    session_timeout_ms=self.session_timeout_ms, xlog=xlog,
                    shard_path=self.shard_path)
        self.peers[p.id] = p
        await p.start()
        return p

    def peer(self, i) -> SimPeer:
        return list(self.peers.values())[i]

    async def state(self) -> Optional[dict]:
        node = self.srv.nodes.get(self.shard_path + "/state")
        return json.loads(node.data) if node else None

    def history(self) -> List[dict]:
        out = []
        prefix = self.shard_path + "/history/"
        for path, node in self.srv.nodes.items():
            if path.startswith(prefix):
                seq = int(path.rsplit("-", 1)[1])
                out.append({"zkSeq": seq, "time": node.ctime,
                            "state": json.loads(node.data)})
        return sorted(out, key=lambda e: e["zkSeq"])

    async def wait_state(self, pred, timeout=10.0, what="cluster state"):
        deadline = asyncio.get_running_loop().time() + timeout
        while True:
            s = await self.state()
            if s is not None and pred(s):
                return s
            if asyncio.get_running_loop().time() > deadline:
                raise AssertionError("timeout waiting for %s; last=%r"
                                     % (what, s))
            await asyncio.sleep(0.05)

    async def stop(self):
        for p in self.peers.values():
            try:
                await p.close()
            except Exception:
                pass
        if self.srv:
            await self.srv.stop()


class InvariantViolation(AssertionError):
    pass


class Simulator:
    """Seeded random fault schedule over a SimShard with invariant
    checking after each event."""

    def __init__(self, seed: int = 0, n_peers: int = 3,
                 max_peers: int = 5, verbose: bool = False):
        self.rng = random.Random(seed)
        self.seed = seed
        self.n_peers = n_peers
        self.max_peers = max_peers
        self.verbose = verbose
        self.shard = SimShard(session_timeout_ms=1000)
        self.dead: List[str] = []        # ips of killed peers
        self.next_ip = n_peers + 1
        self.events: List[str] = []
        self.frozen = False

    def _log(self, msg: str) -> None:
        self.events.append(msg)
        if self.verbose:
            print("  [sim] " + msg)

    # ------------------------------------------------------------- events
    async def _op_kill(self) -> None:
        live = [p for p in self.shard.peers.values()
                if p.ip not in self.dead]
        if len(live) <= 1:
            return
        victim = self.rng.choice(live)
        self._log("kill %s" % victim.ip)
        await victim.kill()
        self.dead.append(victim.ip)
        del self.shard.peers[victim.id]

    async def _op_restart(self) -> None:
        if not self.dead:
            return
        ip = self.rng.choice(self.dead)
        self.dead.remove(ip)
        self._log("restart %s" % ip)
        await self.shard.add_peer(ip)

    async def _op_add(self) -> None:
        if len(self.shard.peers) + len(self.dead) >= self.max_peers:
            return
        ip = "10.0.0.%d" % self.next_ip
        self.next_ip += 1
        self._log("add %s" % ip)
        await self.shard.add_peer(ip)

    async def _op_expire_session(self) -> None:
        """Server-side session expiry of a live peer that keeps running —
        the ZK-blip case: the peer must rebuild its session, rejoin the
        election, and re-evaluate (it may discover it was deposed)."""
        live = [p for p in self.shard.peers.values()
                if p.ip not in self.dead]
        if not live:
            return
        victim = self.rng.choice(live)
        cli = victim.zk._zk
        if cli is None or not cli.session_id:
            return
        sess = self.shard.srv.sessions.get(cli.session_id)
        if sess is None:
            return
        self._log("expire-session %s" % victim.ip)
        self.shard.srv._expire_session(sess)

    async def _op_toggle_freeze(self) -> None:
        from ..adm import core as adm
        zk = await adm.create_zk_client(self.shard.srv.conn_str)
        try:
            if self.frozen:
                self._log("unfreeze")
                await adm.unfreeze(zk, self.shard.shard_path)
            else:
                self._log("freeze")
                await adm.freeze(zk, self.shard.shard_path, "sim")
            self.frozen = not self.frozen
        except adm.AdmError:
            pass
        finally:
            await zk.close()

    async def _quiesce(self, settle_s: float = 1.2,
                       timeout_s: float = 15.0) -> None:
        """Wait until the cluster state stops changing for ``settle_s``."""
        loop = asyncio.get_running_loop()
        deadline = loop.time() + timeout_s
        last = json.dumps(await self.shard.state(), sort_keys=True)
        last_change = loop.time()
        while loop.time() < deadline:
            await asyncio.sleep(0.1)
            cur = json.dumps(await self.shard.state(), sort_keys=True)
            if cur != last:
                last = cur
                last_change = loop.time()
            elif loop.time() - last_change >= settle_s:
                return

    # ---------------------------------------------------------- invariants
    def check_invariants(self) -> None:
        from ..adm.core import annotate_history

        hist = self.shard.history()
        for entry in annotate_history(hist):
            if entry["violations"]:
                raise InvariantViolation(
                    "illegal transition at history seq %s: %s\nevents: %s"
                    % (entry["zkSeq"], entry["violations"],
                       "; ".join(self.events)))

        node = self.shard.srv.nodes.get(self.shard.shard_path + "/state")
        if node is None:
            return
        s = json.loads(node.data)
        # disjoint role assignment
        ids = [s["primary"]["id"]]
        if s.get("sync"):
            ids.append(s["sync"]["id"])
        ids += [a["id"] for a in s.get("async") or []]
        ids += [d["id"] for d in s.get("deposed") or []]
        if len(ids) != len(set(ids)):
            raise InvariantViolation(
                "peer appears in multiple roles: %r\nevents: %s"
                % (s, "; ".join(self.events)))
        # no split brain among live peers; deposed peers run nothing
        deposed_ids = {d["id"] for d in s.get("deposed") or []}
        primaries = []
        for p in self.shard.peers.values():
            if p.db.role == "primary":
                primaries.append(p.id)
            if p.id in deposed_ids and p.db.role not in (None, "none"):
                raise InvariantViolation(
                    "deposed peer %s runs role %r\nevents: %s"
                    % (p.id, p.db.role, "; ".join(self.events)))
        if len(primaries) > 1:
            raise InvariantViolation(
                "split brain: %r both primary\nevents: %s"
                % (primaries, "; ".join(self.events)))
        if primaries and primaries[0] != s["primary"]["id"]:
            raise InvariantViolation(
                "live primary %s is not the declared primary %s\n"
                "events: %s" % (primaries[0], s["primary"]["id"],
                                "; ".join(self.events)))

    # --------------------------------------------------------------- runs
    async def run(self, steps: int = 30) -> dict:
        ops = [(self._op_kill, 4), (self._op_restart, 4),
               (self._op_add, 1), (self._op_toggle_freeze, 1),
               (self._op_expire_session, 2)]
        weighted = [op for op, w in ops for _ in range(w)]
        await self.shard.start(n_peers=self.n_peers)
        try:
            await self.shard.wait_state(lambda s: s.get("primary"),
                                        timeout=15, what="formation")
            await self._quiesce()
            self.check_invariants()
            for step in range(steps):
                op = self.rng.choice(weighted)
                await op()
                await self._quiesce()
                self.check_invariants()
            # final: thaw and revive everyone; shard must converge back
            if self.frozen:
                await self._op_toggle_freeze()
            while self.dead:
                await self._op_restart()
            await self._quiesce(settle_s=2.0, timeout_s=30.0)
            self.check_invariants()
            s = await self.shard.state()
            hist = self.shard.history()
            return {"seed": self.seed, "steps": steps,
                    "events": len(self.events),
                    "generations": s["generation"] if s else None,
                    "history_entries": len(hist)}
        finally:
            await self.shard.stop()


def main(argv=None) -> int:
    import argparse
    ap = argparse.ArgumentParser(
        prog="manatee-fsm-sim",
        description="randomized FSM fault simulator with invariant "
                    "checking")
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--seeds", type=int, default=1,
                    help="run seeds seed..seed+N-1")
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--peers", type=int, default=3)
    ap.add_argument("-v", "--verbose", action="store_true")
    ns = ap.parse_args(argv)

    async def go():
        for seed in range(ns.seed, ns.seed + ns.seeds):
            sim = Simulator(seed=seed, n_peers=ns.peers,
                            verbose=ns.verbose)
            try:
                res = await sim.run(steps=ns.steps)
            except InvariantViolation as exc:
                print("SEED %d: INVARIANT VIOLATION\n%s" % (seed, exc))
                return 1
            print("seed %d ok: %d events, %s generations, %d history "
                  "entries" % (seed, res["events"],
                               res["generations"],
                               res["history_entries"]))
        return 0
    return asyncio.run(go())


if __name__ == "__main__":
    import sys
    sys.exit(main())
