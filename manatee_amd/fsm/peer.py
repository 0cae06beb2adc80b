"""ManateePeer — the per-peer cluster state machine.

Clean-room re-implementation of the external ``manatee-state-machine``
dependency (wired at ref lib/shard.js:59-71; observable contract reconstructed
in SURVEY.md §2.2).  One instance runs inside each sitter and decides every
topology change for the shard:

- inputs: coordination events (``init/activeChange/clusterStateChange`` from
  ZkMgr) and database events (``init/healthy/unhealthy/error`` from the db
  manager);
- outputs: ``db.reconfigure({role, upstream, downstream, restorePeer})`` and
  ``zk.put_cluster_state(state)``.

Behavioral rules enforced (diagnosed in ref lib/adm.js
annotateHistoryNode:2296-2416 and docs/user-guide.md):

- generation never decreases; a new primary must be the previous sync; a
  sync change requires a generation bump;
- the *primary* manages membership: it appoints new asyncs, removes dead
  ones, and replaces a dead sync by promoting the first live async
  (docs/user-guide.md:69-76);
- the *sync* acts only when the primary's ZK session is gone: it takes over
  with a generation bump, deposing the old primary — but only if it has
  caught up to the generation's initWal fence and a live async exists to
  become the new sync;
- frozen clusters perform no transitions (takeover backs off with
  ClusterFrozenError, ref docs/migrate-1-to-2.md:381-401);
- ONWM/singleton: the primary runs alone; any other peer that sees an ONWM
  cluster state shuts down (docs/user-guide.md:367-387);
- promote requests ({id, role, asyncIndex?, generation, expireTime}, ref
  lib/adm.js:1915-1928) move a peer up one position: async[i]→async[i-1]
  (primary acts, no gen bump unless the sync changes), async[0]→sync
  (primary acts, gen bump), sync→primary (sync acts, deposes the primary);
  stale or expired requests are cleared by the primary;
- all state writes go through versioned CAS; a conflict simply triggers
  re-evaluation (someone else saw the world first).

Everything is processed strictly serially on one asyncio task — the same
design-level race defense as the reference's single-threaded event loop
(SURVEY.md §5.2).
"""

from __future__ import annotations

import asyncio
import time
from typing import Any, Callable, Dict, List, Optional

from ..common import lsn as lsnmod
from ..common.logging import Logger, null_logger
from . import state as st


class ClusterFrozenError(Exception):
    pass


class ManateePeer:
    def __init__(self, *, zk, db, self_ident: dict, singleton: bool = False,
                 log: Optional[Logger] = None, tick_interval_s: float = 1.0,
                 settle_s: float = 3.0):
        """zk: ZkMgr-like (on/put_cluster_state/active/cluster_state).
        db: db-manager-like (on/reconfigure/stop/get_xlog_location/status).
        self_ident: {id, zoneId, ip, pgUrl, backupUrl}."""
        self.log = (log or null_logger()).child(component="cluster",
                                                peer=self_ident["id"])
        self._zk = zk
        self._db = db
        self._ident = self_ident
        self._id = self_ident["id"]
        self._singleton = singleton
        self._tick_interval_s = tick_interval_s

        self._settle_s = settle_s
        self._absence_trusted_after = 0.0
        self._zk_inited = False
        self._db_inited = False
        self._db_setup = False
        self._db_online = False
        self._cluster_state: Optional[dict] = None
        self._actives: List[dict] = []
        self._role = st.ROLE_UNASSIGNED
        self._peer_state = "uninit"       # for debugState / logs
        self._applied_db_config: Optional[dict] = None
        self._shutdown = False
        self._closing = False
        self._events: "asyncio.Queue" = asyncio.Queue()
        self._runner: Optional[asyncio.Task] = None
        self._ticker: Optional[asyncio.Task] = None
        self._listeners: Dict[str, List[Callable]] = {}
        self._last_warn: Dict[str, float] = {}

        zk.on("init", lambda ev: self._push("zk-init", ev))
        zk.on("activeChange", lambda a: self._push("active", a))
        zk.on("clusterStateChange", lambda s: self._push("state", s))
        db.on("init", lambda ev: self._push("db-init", ev))
        db.on("healthy", lambda *a: self._push("db-healthy", None))
        db.on("unhealthy", lambda *a: self._push("db-unhealthy", None))
        db.on("error", lambda err=None: self._push("db-error", err))

    # ---------------------------------------------------------------- wiring
    def on(self, event: str, cb: Callable) -> None:
        self._listeners.setdefault(event, []).append(cb)

    def _emit(self, event: str, *args: Any) -> None:
        for cb in self._listeners.get(event, []):
            try:
                cb(*args)
            except Exception as exc:
                self.log.error("listener error", event=event, err=exc)

    def _push(self, kind: str, payload: Any) -> None:
        if not self._closing:
            self._events.put_nowait((kind, payload))

    def start(self) -> None:
        loop = asyncio.get_running_loop()
        self._runner = loop.create_task(self._run())
        self._ticker = loop.create_task(self._tick())

    async def close(self) -> None:
        self._closing = True
        for t in (self._runner, self._ticker):
            if t is not None:
                t.cancel()
                try:
                    await t
                except (asyncio.CancelledError, Exception):
                    pass

    # ------------------------------------------------------------- main loop
    async def _tick(self) -> None:
        while not self._closing:
            await asyncio.sleep(self._tick_interval_s)
            if self._events.empty():
                self._push("tick", None)

    async def _run(self) -> None:
        while not self._closing:
            kind, payload = await self._events.get()
            try:
                self._ingest(kind, payload)
                # drain everything already queued BEFORE evaluating: a
                # single evaluation can block for a long time (e.g. a
                # snapshot restore), and acting per-event would replay
                # stale decisions against a world that has since moved on
                # (fresher activeChange/clusterState events sitting in
                # the queue).  Coalescing makes every evaluation use the
                # newest known state.
                while not self._events.empty():
                    k2, p2 = self._events.get_nowait()
                    self._ingest(k2, p2)
                if self._shutdown:
                    continue
                await self._eval_cluster_state()
            except Exception as exc:
                self.log.error("evalClusterState failed", event=kind, err=exc)

    def _ingest(self, kind: str, payload: Any) -> None:
        if kind == "zk-init":
            if self._zk_inited:
                # a RE-init means our session expired and was rebuilt —
                # typically because the coordination service itself
                # bounced, expiring EVERYONE at once.  Peers re-register
                # over the next moments; acting on their transient
                # absence right now would depose live peers.  Hold a
                # settle window before trusting absence.
                self._absence_trusted_after = \
                    time.monotonic() + self._settle_s
            self._zk_inited = True
            self._actives = payload.get("active") or []
            self._cluster_state = payload.get("clusterState")
            # a fresh zk init after session expiry invalidates our belief
            # about what db config matches the world
            self.log.info("zk init", nactive=len(self._actives),
                          generation=(self._cluster_state or {}).get("generation"))
        elif kind == "active":
            self._actives = payload or []
        elif kind == "state":
            self._cluster_state = payload
        elif kind == "db-init":
            self._db_inited = True
            self._db_setup = bool(payload.get("setup"))
            self._db_online = bool(payload.get("online"))
            self.log.info("db init", setup=self._db_setup,
                          online=self._db_online)
        elif kind == "db-healthy":
            self._db_online = True
        elif kind == "db-unhealthy":
            self._db_online = False
        elif kind == "db-error":
            # fatal db error (unexpected exit): the db manager gave up; we
            # keep our zk session so the cluster does NOT fail over for a
            # restartable crash — re-applying config below restarts it
            # (ref postgresMgr emits fatal 'error' :1736-1753)
            self._db_online = False
            self._applied_db_config = None

    # --------------------------------------------------------------- helpers
    def _warn_throttled(self, key: str, msg: str, **kw) -> None:
        now = time.monotonic()
        if now - self._last_warn.get(key, 0) > 10.0:
            self._last_warn[key] = now
            self.log.warn(msg, **kw)

    def _active_ids(self) -> List[str]:
        return [a["id"] for a in self._actives]

    def _absence_settled(self) -> bool:
        """May we act on a peer's ABSENCE from the active list?  False
        during the settle window after our own session rebuild (peers
        are still re-registering after a coordination bounce)."""
        if time.monotonic() >= self._absence_trusted_after:
            return True
        self._warn_throttled(
            "settle", "own session was just rebuilt; holding off on "
            "absence-driven transitions while peers re-register")
        return False

    def _ident_for(self, peer_id: str) -> dict:
        for a in self._actives:
            if a["id"] == peer_id:
                return st.ident_from_active(a)
        s = self._cluster_state
        if s:
            for entry in ([s.get("primary"), s.get("sync")]
                          + list(s.get("async") or [])
                          + list(s.get("deposed") or [])):
                if entry and entry.get("id") == peer_id:
                    return dict(entry)
        return st.make_ident(peer_id)

    async def _write_state(self, new_state: dict, why: str) -> bool:
        """CAS-write; False on conflict (re-eval happens via watch)."""
        try:
            st.check_transition(self._cluster_state, new_state)
        except st.TransitionError as exc:
            self.log.error("BUG: illegal transition blocked", why=why,
                           err=exc)
            return False
        try:
            await self._zk.put_cluster_state(new_state)
        except Exception as exc:
            self.log.warn("cluster state write failed", why=why, err=exc)
            return False
        self._cluster_state = new_state
        self.log.info("declared new state", why=why,
                      generation=new_state["generation"])
        self._emit("stateWritten", new_state)
        return True

    async def _apply_db_config(self, cfg: dict, why: str = "") -> None:
        if cfg == self._applied_db_config:
            return
        self.log.info("reconfiguring database", role=cfg["role"], why=why,
                      upstream=(cfg.get("upstream") or {}).get("pgUrl"),
                      downstream=(cfg.get("downstream") or {}).get("pgUrl"))
        await self._db.reconfigure(cfg)
        self._applied_db_config = cfg

    def _db_cfg_primary(self, sync_ident: Optional[dict]) -> dict:
        return {"role": "primary", "upstream": None,
                "downstream": ({"pgUrl": sync_ident["pgUrl"],
                                "backupUrl": sync_ident["backupUrl"]}
                               if sync_ident else None)}

    def _db_cfg_standby(self, role: str, upstream: dict,
                        restore_peer: dict) -> dict:
        # every standby rebuilds off the PRIMARY (back-pressure on the
        # primary so the sync cannot fall hopelessly behind,
        # ref lib/postgresMgr.js:1019-1029)
        return {"role": role,
                "upstream": {"pgUrl": upstream["pgUrl"],
                             "backupUrl": upstream["backupUrl"]},
                "downstream": None,
                "restorePeer": {"backupUrl": restore_peer["backupUrl"]}}

    # ---------------------------------------------------------- evaluation
    async def _eval_cluster_state(self) -> None:
        if not (self._zk_inited and self._db_inited):
            self._peer_state = "waiting (inputs)"
            return
        s = self._cluster_state

        if s is None:
            await self._cluster_setup()
            return

        # ONWM safety: a non-singleton peer that finds an ONWM cluster with
        # someone else as primary must shut down (docs/user-guide.md:377-380)
        if s.get("oneNodeWriteMode") and s["primary"]["id"] != self._id \
                and not self._singleton:
            self.log.error("cluster is in one-node-write mode with another "
                           "primary; shutting down")
            self._peer_state = "shutdown (onwm)"
            self._shutdown = True
            await self._apply_db_config({"role": "none", "upstream": None,
                                         "downstream": None}, "onwm shutdown")
            self._emit("shutdown")
            return

        role = st.role_of(s, self._id)
        self._role = role
        if role == st.ROLE_DEPOSED:
            await self._assume_deposed()
        elif role == st.ROLE_PRIMARY:
            await self._assume_primary()
        elif role == st.ROLE_SYNC:
            await self._assume_sync()
        elif role == st.ROLE_ASYNC:
            await self._assume_async()
        else:
            await self._assume_unassigned()

    # ------------------------------------------------------- cluster setup
    async def _cluster_setup(self) -> None:
        """No cluster state exists yet (ref peer-state 'assumeUnassigned' →
        'declared new generation' on formation)."""
        if self._singleton:
            self._peer_state = "declaring generation (onwm setup)"
            new_state = {
                "generation": 1,
                "primary": dict(self._ident),
                "sync": None,
                "async": [],
                "deposed": [],
                "initWal": lsnmod.ZERO,
                "oneNodeWriteMode": True,
            }
            if await self._write_state(new_state, "onwm cluster setup"):
                await self._eval_cluster_state()
            return

        # SAFETY: a peer whose database already holds data must never
        # auto-declare a new generation — if the coordination state is
        # gone but the data is not, re-forming around arbitrary election
        # order could elect a STALE peer as primary and destroy
        # acknowledged writes when the up-to-date peers re-slave to it.
        # This is the operator's `manatee-adm state-backfill` situation
        # (ref stateBackfill lib/adm.js:1231-1312, which auto-freezes for
        # the same reason).
        if self._db_setup:
            self._peer_state = "waiting (no cluster state, but database " \
                "is initialized)"
            self._warn_throttled(
                "setup-initialized",
                "database is initialized but no cluster state exists; "
                "refusing to auto-form — run `manatee-adm state-backfill` "
                "(or clear the data dir) to proceed")
            return

        actives = self._actives
        if len(actives) < 2:
            self._peer_state = "waiting (cluster setup: need 2 peers)"
            self._warn_throttled("setup", "waiting for a second peer to "
                                 "form the cluster", nactive=len(actives))
            return
        # the longest-lived member (lowest election seq = first in the
        # active list) declares generation 1
        if actives[0]["id"] != self._id:
            self._peer_state = "waiting (cluster setup: not the founder)"
            return
        self._peer_state = "declaring generation (cluster setup)"
        new_state = {
            "generation": 1,
            "primary": st.ident_from_active(actives[0]),
            "sync": st.ident_from_active(actives[1]),
            "async": [st.ident_from_active(a) for a in actives[2:]],
            "deposed": [],
            "initWal": lsnmod.ZERO,
        }
        if await self._write_state(new_state, "cluster setup"):
            self.log.info("cluster formed",
                          sync=new_state["sync"]["id"],
                          nasync=len(new_state["async"]))
            await self._eval_cluster_state()

    # ------------------------------------------------------------- primary
    async def _assume_primary(self) -> None:
        s = self._cluster_state
        self._peer_state = "primary"
        await self._apply_db_config(self._db_cfg_primary(s.get("sync")),
                                    "assume primary")

        if st.is_frozen(s):
            # frozen: no topology management at all
            self._warn_throttled("frozen", "cluster is frozen; skipping "
                                 "topology management",
                                 reason=(s.get("freeze") or {}).get("reason"))
            return

        if await self._primary_handle_promote():
            return
        if await self._primary_replace_dead_sync():
            return
        await self._primary_manage_asyncs()

    async def _primary_handle_promote(self) -> bool:
        s = self._cluster_state
        promote = s.get("promote")
        if not promote:
            return False
        base = {k: v for k, v in s.items() if k != "promote"}

        if promote.get("generation") != s.get("generation") or \
                st.promote_expired(promote):
            self.log.info("clearing stale/expired promote request",
                          promote=promote)
            return await self._write_state(base, "clear stale promote")

        if promote.get("role") == "async":
            asyncs = list(s.get("async") or [])
            idx = promote.get("asyncIndex")
            if idx is None and len(asyncs) == 1:
                idx = 0
            if idx is None or not (0 <= idx < len(asyncs)) or \
                    asyncs[idx]["id"] != promote["id"]:
                self.log.warn("ignoring promote: async position mismatch",
                              promote=promote)
                return await self._write_state(base, "clear bad promote")
            if idx == 0:
                # async[0] → sync: sync change ⇒ generation bump
                if asyncs[0]["id"] not in self._active_ids():
                    self.log.warn("ignoring promote: target async not active")
                    return await self._write_state(base,
                                                   "clear dead-target promote")
                old_sync = s.get("sync")
                new_asyncs = list(asyncs)
                promoted = new_asyncs.pop(0)
                if old_sync:
                    new_asyncs.insert(0, old_sync)
                new_state = dict(base)
                new_state["generation"] = s["generation"] + 1
                new_state["sync"] = promoted
                new_state["async"] = new_asyncs
                new_state["initWal"] = await self._own_xlog()
                return await self._write_state(new_state,
                                               "promote async[0] to sync")
            # async[i] ↔ async[i-1]: no role-boundary change, no gen bump
            new_asyncs = list(asyncs)
            new_asyncs[idx - 1], new_asyncs[idx] = \
                new_asyncs[idx], new_asyncs[idx - 1]
            new_state = dict(base)
            new_state["async"] = new_asyncs
            return await self._write_state(new_state,
                                           "promote async[%d]" % idx)

        # promote.role == 'sync' is acted on by the sync peer, not us;
        # leave the request in place
        return False

    async def _primary_replace_dead_sync(self) -> bool:
        """Sync's ZK session gone → promote the first live async
        (docs/user-guide.md:69-76: the primary adjusts the topology)."""
        s = self._cluster_state
        sync = s.get("sync")
        active = self._active_ids()
        if s.get("oneNodeWriteMode"):
            return False
        if sync and sync["id"] in active:
            return False
        if not self._absence_settled():
            return False
        candidates = [a for a in (s.get("async") or [])
                      if a["id"] in active]
        if not candidates:
            if sync:
                self._warn_throttled(
                    "nosync", "sync is gone and no live async can replace "
                    "it; shard degraded (writes blocked)", sync=sync["id"])
            return False
        new_sync = candidates[0]
        new_asyncs = [a for a in (s.get("async") or [])
                      if a["id"] != new_sync["id"]]
        new_state = {k: v for k, v in s.items() if k != "promote"}
        new_state["generation"] = s["generation"] + 1
        new_state["sync"] = new_sync
        new_state["async"] = new_asyncs
        new_state["initWal"] = await self._own_xlog()
        self.log.warn("sync peer is gone; declaring new generation",
                      old_sync=sync["id"] if sync else None,
                      new_sync=new_sync["id"])
        return await self._write_state(new_state, "replace dead sync")

    async def _primary_manage_asyncs(self) -> bool:
        """Add newly-active unassigned peers as asyncs; drop dead asyncs.
        No generation bump (sync unchanged)."""
        s = self._cluster_state
        active = self._active_ids()
        known = set(st.all_peer_ids(s))
        additions = [st.ident_from_active(a) for a in self._actives
                     if a["id"] not in known]
        kept = [a for a in (s.get("async") or []) if a["id"] in active]
        removed = [a for a in (s.get("async") or []) if a["id"] not in active]
        if removed and not self._absence_settled():
            kept = list(s.get("async") or [])
            removed = []
        if not additions and not removed:
            return False
        new_state = dict(s)
        new_state["async"] = kept + additions
        why = []
        if additions:
            why.append("add asyncs %s" % [a["id"] for a in additions])
        if removed:
            why.append("remove dead asyncs %s" % [a["id"] for a in removed])
        return await self._write_state(new_state, "; ".join(why))

    async def _own_xlog(self) -> str:
        try:
            loc = await self._db.get_xlog_location()
            if loc and lsnmod.is_lsn(loc):
                return loc
        except Exception as exc:
            self.log.warn("could not read own xlog location", err=exc)
        return (self._cluster_state or {}).get("initWal") or lsnmod.ZERO

    # ---------------------------------------------------------------- sync
    async def _assume_sync(self) -> None:
        s = self._cluster_state
        self._peer_state = "sync"
        await self._apply_db_config(
            self._db_cfg_standby("sync", s["primary"], s["primary"]),
            "assume sync")

        promote = s.get("promote")
        primary_alive = s["primary"]["id"] in self._active_ids()
        want_promote = bool(
            promote and promote.get("role") == "sync"
            and promote.get("id") == self._id
            and promote.get("generation") == s.get("generation")
            and not st.promote_expired(promote))

        if primary_alive and not want_promote:
            return
        if not primary_alive and not want_promote \
                and not self._absence_settled():
            return
        try:
            await self._start_takeover(
                "operator promote" if want_promote else "primary gone")
        except ClusterFrozenError:
            self._warn_throttled("frozentakeover",
                                 "backing off: ClusterFrozenError: cluster "
                                 "is frozen")

    async def _start_takeover(self, why: str) -> None:
        """The sync takes over as primary with a generation bump
        (ref peer-state 'startTakeover', docs/migrate-1-to-2.md:93-101)."""
        s = self._cluster_state
        self.log.info("preparing for new generation (%s)" % why)
        if st.is_frozen(s):
            raise ClusterFrozenError()

        # initWal fence: we must have replayed everything up to the point
        # where this generation began, else an acknowledged write could be
        # lost by our promotion (SURVEY.md §7 'generation fencing')
        own = await self._own_xlog()
        init_wal = s.get("initWal") or lsnmod.ZERO
        if lsnmod.compare(own, init_wal) < 0:
            self._warn_throttled(
                "fence", "cannot take over: not caught up to initWal",
                own=own, initWal=init_wal)
            return

        active = self._active_ids()
        candidates = [a for a in (s.get("async") or []) if a["id"] in active]
        if not candidates:
            self._warn_throttled(
                "noasync", "cannot take over: no live async available to "
                "become the new sync")
            return
        new_sync = candidates[0]
        new_asyncs = [a for a in (s.get("async") or [])
                      if a["id"] != new_sync["id"]]
        new_state = {k: v for k, v in s.items() if k != "promote"}
        new_state["generation"] = s["generation"] + 1
        new_state["primary"] = self._ident_for(self._id)
        new_state["sync"] = new_sync
        new_state["async"] = new_asyncs
        new_state["deposed"] = list(s.get("deposed") or []) + [s["primary"]]
        new_state["initWal"] = own
        if await self._write_state(new_state, "takeover: " + why):
            self.log.warn("declared new generation; old primary deposed",
                          generation=new_state["generation"],
                          deposed=s["primary"]["id"])
            await self._eval_cluster_state()

    # --------------------------------------------------------------- async
    async def _assume_async(self) -> None:
        s = self._cluster_state
        idx = st.async_index(s, self._id)
        self._peer_state = "async.%d" % idx
        chain_prev = s["sync"] if idx == 0 else s["async"][idx - 1]
        if chain_prev is None:
            # degenerate: no sync; replicate straight from the primary
            chain_prev = s["primary"]
        await self._apply_db_config(
            self._db_cfg_standby("async", chain_prev, s["primary"]),
            "assume async[%d]" % idx)

    # ---------------------------------------------------- deposed/unassigned
    async def _assume_deposed(self) -> None:
        self._peer_state = "deposed"
        # deposed databases stay stopped until an operator rebuild —
        # their xlog may have diverged (docs/user-guide.md:336-365)
        await self._apply_db_config({"role": "none", "upstream": None,
                                     "downstream": None}, "deposed")

    async def _assume_unassigned(self) -> None:
        self._peer_state = "unassigned"
        await self._apply_db_config({"role": "none", "upstream": None,
                                     "downstream": None}, "unassigned")

    # ---------------------------------------------------------------- debug
    def debug_state(self) -> dict:
        """Exposed at GET /state (ref lib/shard.js:74-76,
        lib/statusServer.js:106-109)."""
        return {
            "id": self._id,
            "role": self._role,
            "peerState": self._peer_state,
            "singleton": self._singleton,
            "zkInited": self._zk_inited,
            "dbInited": self._db_inited,
            "dbOnline": self._db_online,
            "shutdown": self._shutdown,
            "active": list(self._actives),
            "clusterState": self._cluster_state,
            "appliedDbConfig": self._applied_db_config,
        }
