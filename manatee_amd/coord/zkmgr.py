"""Shard coordination manager — the ZookeeperMgr equivalent.

Clean-room re-implementation of ``lib/zookeeperMgr.js``: owns every ZK
interaction for a peer.  Namespace under ``shardPath`` (ref :78-86):

- ``election/``  — EPHEMERAL_SEQUENTIAL members named ``<id>-<SEQ>`` whose
  data is the peer's identity JSON (ref :452-456);
- ``state``      — persistent cluster-state JSON, updated with versioned CAS;
- ``history/``   — PERSISTENT_SEQUENTIAL ``<generation>-<SEQ>`` copies of
  every state written (ref :621-623).

Behaviors preserved:

- one-shot watches re-registered after every fire (ref watch() :204-264);
- active list parsed and de-duplicated by peer id keeping the *lowest*
  sequence number (stale sessions from a bounced peer linger until expiry;
  ref parseAndUniqueActives :184-200);
- debounce: activeChange only fires when the id list actually changed
  (ref idListsEqual :277-300 used at :375-379);
- ``putClusterState`` is an atomic transaction: create history node +
  versioned setData (or create) of ``state`` (ref :605-630);
- on session expiry the whole client is rebuilt and a fresh ``init`` event
  is emitted so the state machine re-evaluates the world (ref :500-586).

Events (consumed by the FSM, contract at lib/shard.js:59-71):
``init {active, clusterState}``, ``activeChange(active)``,
``clusterStateChange(state)``.
"""

from __future__ import annotations

import asyncio
import json
from typing import Any, Callable, Dict, List, Optional

from ..common.logging import Logger, null_logger
from . import jute
from .zkclient import ZkClient


class ActivePeer:
    """One live member of the election directory: id + its identity data
    (flattened exactly like the reference's ``active`` getter, :96-110)."""

    __slots__ = ("id", "seq", "data")

    def __init__(self, id: str, seq: int, data: dict):
        self.id = id
        self.seq = seq
        self.data = data

    def as_dict(self) -> dict:
        out = {"id": self.id}
        out.update(self.data)
        # older peers may lack pgUrl/backupUrl: derive from the id
        # "ip:pgPort:backupPort" like the reference's compat shim (:110-130)
        parts = self.id.split(":")
        if len(parts) == 3:
            ip, pg_port, backup_port = parts
            out.setdefault("ip", ip)
            out.setdefault("pgUrl", "tcp://postgres@%s:%s/postgres"
                           % (ip, pg_port))
            out.setdefault("backupUrl", "http://%s:%s" % (ip, backup_port))
        return out

    def __repr__(self):
        return "ActivePeer(%s seq=%d)" % (self.id, self.seq)


def parse_and_unique_actives(children: List[str]) -> List[ActivePeer]:
    """Election child names are ``<id>-<SEQ>``; de-duplicate by id keeping
    the lowest sequence, return sorted by sequence (ref :184-200)."""
    peers: Dict[str, ActivePeer] = {}
    for name in children:
        idx = name.rfind("-")
        if idx <= 0:
            continue
        pid, seq_text = name[:idx], name[idx + 1:]
        try:
            seq = int(seq_text)
        except ValueError:
            continue
        cur = peers.get(pid)
        if cur is None or seq < cur.seq:
            peers[pid] = ActivePeer(pid, seq, {})
    return sorted(peers.values(), key=lambda p: p.seq)


def id_lists_equal(a: List[ActivePeer], b: List[ActivePeer]) -> bool:
    if len(a) != len(b):
        return False
    return all(x.id == y.id and x.seq == y.seq for x, y in zip(a, b))


class ZkMgr:
    def __init__(self, *, id: str, data: dict, path: str, conn_str: str,
                 session_timeout_ms: int = 60000,
                 log: Optional[Logger] = None):
        self.log = (log or null_logger()).child(component="ZkMgr")
        self._id = id
        self._data = data
        base = path if path.endswith("/") else path + "/"
        self._path = path.rstrip("/")
        self._election_path = base + "election"
        self._ephemeral_prefix = self._election_path + "/" + id + "-"
        self._history_path = base + "history"
        self._state_path = base + "state"
        self._conn_str = conn_str
        self._session_timeout_ms = session_timeout_ms

        self._zk: Optional[ZkClient] = None
        self._active: List[ActivePeer] = []
        self._cluster_state: Optional[dict] = None
        self._state_version: Optional[int] = None
        self._inited = False
        self._closed = False
        self._listeners: Dict[str, List[Callable]] = {}
        self._tasks: "asyncio.Queue[Optional[Callable]]" = asyncio.Queue()
        self._worker: Optional[asyncio.Task] = None
        self._resync_task: Optional[asyncio.Task] = None
        self._my_election_node: Optional[str] = None

    # --------------------------------------------------------------- events
    def on(self, event: str, cb: Callable) -> None:
        self._listeners.setdefault(event, []).append(cb)

    def _emit(self, event: str, *args: Any) -> None:
        for cb in self._listeners.get(event, []):
            try:
                cb(*args)
            except Exception as exc:
                self.log.error("listener error", event=event, err=exc)

    # ------------------------------------------------------------ lifecycle
    async def init(self) -> None:
        """Connect, set up the namespace, join the election, arm watches,
        then emit ``init`` (ref init/setupData :412-586)."""
        loop = asyncio.get_running_loop()
        self._worker = loop.create_task(self._work_loop())
        self._resync_task = loop.create_task(self._resync_loop())
        await self._setup_client()

    async def _resync_loop(self) -> None:
        """Low-frequency watch-loss safety net: re-list the election and
        re-read the state every few seconds.  Both handlers debounce
        (no event is emitted unless something actually changed) and
        re-arm their one-shot watches, so ANY lost watch — a server
        bug, a dropped notification, a race this code has not imagined
        — heals within one period instead of blinding the peer until
        session expiry.  Liveness here is the product; purity of the
        watch discipline is not worth a stuck shard."""
        period = max(2.0, self._session_timeout_ms / 1000.0)
        while not self._closed:
            await asyncio.sleep(period)
            if self._zk is not None and not self._closed:
                self._enqueue(self._handle_active)
                self._enqueue(self._handle_cluster_state)

    async def _setup_client(self) -> None:
        self._zk = ZkClient(self._conn_str,
                            session_timeout_ms=self._session_timeout_ms,
                            log=self.log, on_session=self._on_session_event)
        await self._zk.connect(timeout_s=max(
            self._session_timeout_ms / 1000.0, 10.0))
        zk = self._zk
        await zk.mkdirp(self._election_path)
        await zk.mkdirp(self._history_path)
        # initial state read + watch
        state = await self._read_state_and_watch()
        # join the election
        self._my_election_node = await zk.create(
            self._ephemeral_prefix,
            json.dumps(self._data, separators=(",", ":")).encode(),
            mode=jute.EPHEMERAL_SEQUENTIAL)
        self.log.debug("joined election", node=self._my_election_node)
        # read + watch the active list
        children, _ = await zk.get_children(self._election_path,
                                            watch=self._on_children_event)
        self._active = await self._load_actives(children)
        self._cluster_state = state
        if not self._inited:
            self._inited = True
            self._emit("init", {"active": [p.as_dict() for p in self._active],
                                "clusterState": self._cluster_state})
        else:
            # session-expiry rebuild: world may have changed entirely
            self._emit("init", {"active": [p.as_dict() for p in self._active],
                                "clusterState": self._cluster_state})

    async def close(self) -> None:
        self._closed = True
        if getattr(self, "_resync_task", None) is not None:
            self._resync_task.cancel()
            self._resync_task = None
        if self._worker is not None:
            await self._tasks.put(None)
            try:
                await asyncio.wait_for(self._worker, 5)
            except asyncio.TimeoutError:
                self._worker.cancel()
        if self._zk is not None:
            await self._zk.close()

    # ------------------------------------------------- serialized work loop
    async def _work_loop(self) -> None:
        """Watch events are handled strictly serially, like the reference's
        single-threaded event loop."""
        while True:
            fn = await self._tasks.get()
            if fn is None:
                return
            try:
                await fn()
            except Exception as exc:
                self.log.error("work item failed", err=exc)

    def _enqueue(self, coro_fn: Callable) -> None:
        if not self._closed:
            self._tasks.put_nowait(coro_fn)

    # ------------------------------------------------------------- watchers
    def _on_session_event(self, event: str) -> None:
        if event == "expired" and not self._closed:
            self.log.warn("zk session expired; rebuilding client")
            self._enqueue(self._rebuild_client)
        elif event == "disconnected":
            self.log.warn("zk connection lost; client reconnecting")

    async def _rebuild_client(self) -> None:
        # ref :500-586 — full reset: new session, rejoin election, rewatch,
        # re-emit init.
        old = self._zk
        self._zk = None
        if old is not None:
            await old.close()
        while not self._closed:
            try:
                await self._setup_client()
                return
            except Exception as exc:
                self.log.error("client rebuild failed; retrying", err=exc)
                await asyncio.sleep(1.0)

    def _on_children_event(self, etype: int, path: str) -> None:
        self._enqueue(self._handle_active)

    def _on_state_event(self, etype: int, path: str) -> None:
        self._enqueue(self._handle_cluster_state)

    # -------------------------------------------------------------- actives
    async def _load_actives(self, children: List[str]) -> List[ActivePeer]:
        peers = parse_and_unique_actives(children)
        zk = self._zk
        for peer in peers:
            node = "%s/%s-%010d" % (self._election_path, peer.id, peer.seq)
            try:
                data, _ = await zk.get_data(node)
                peer.data = json.loads(data.decode("utf-8")) if data else {}
            except (jute.ZkError, ValueError):
                peer.data = {}
        return peers

    def _retry_handler(self, coro_fn: Callable, delay_s: float = 0.25
                       ) -> None:
        """A one-shot watch was just CONSUMED and re-arming it failed —
        without a retry this peer would be permanently blind to that
        node (no watch armed, nothing else re-arms it until session
        expiry, which may never come on a healthy connection)."""
        if self._closed:
            return
        asyncio.get_running_loop().call_later(
            delay_s, self._enqueue, coro_fn)

    async def _handle_active(self) -> None:
        """Re-list + re-watch the election dir; emit activeChange only when
        the membership actually changed (ref handleActive :307-386)."""
        if self._zk is None or self._closed:
            return
        try:
            children, _ = await self._zk.get_children(
                self._election_path, watch=self._on_children_event)
        except jute.ZkError as exc:
            self.log.warn("election re-list failed; retrying until the "
                          "watch is re-armed", err=exc)
            self._retry_handler(self._handle_active)
            return
        peers = parse_and_unique_actives(children)
        if id_lists_equal(peers, self._active):
            self.log.debug("active list debounced (unchanged)")
            return
        self._active = await self._load_actives(children)
        self.log.info("active peers changed",
                      active=[p.id for p in self._active])
        self._emit("activeChange", [p.as_dict() for p in self._active])

    # ---------------------------------------------------------------- state
    async def _read_state_and_watch(self) -> Optional[dict]:
        """getData + watch on ``state``; if it does not exist yet, arm an
        exists-watch so creation wakes us (ref NO_NODE handling :226-256)."""
        zk = self._zk
        try:
            data, stat = await zk.get_data(self._state_path,
                                           watch=self._on_state_event)
        except jute.ZkError as exc:
            if exc.code != jute.ZNONODE:
                raise
            await zk.exists(self._state_path, watch=self._on_state_event)
            self._state_version = None
            return None
        self._state_version = stat.version
        try:
            return json.loads(data.decode("utf-8"))
        except ValueError:
            self.log.error("unparseable cluster state",
                           data=data[:256].decode("utf-8", "replace"))
            return None

    async def _handle_cluster_state(self) -> None:
        if self._zk is None or self._closed:
            return
        try:
            state = await self._read_state_and_watch()
        except jute.ZkError as exc:
            self.log.warn("state re-read failed; retrying until the "
                          "watch is re-armed", err=exc)
            self._retry_handler(self._handle_cluster_state)
            return
        if state is None:
            return
        self._cluster_state = state
        self.log.debug("cluster state changed",
                       generation=state.get("generation"))
        self._emit("clusterStateChange", state)

    async def put_cluster_state(self, state: dict) -> None:
        """Atomic history-create + versioned state write (ref :605-630).
        Raises ZkError(BAD_VERSION) if someone else wrote state first —
        the caller (FSM) must re-evaluate."""
        if self._zk is None:
            raise jute.ZkError(jute.ZCONNECTIONLOSS)
        data = json.dumps(state, separators=(",", ":")).encode()
        hp = "%s/%s-" % (self._history_path, state["generation"])
        ops = [jute.MultiOp.create(hp, data, jute.PERSISTENT_SEQUENTIAL)]
        if self._cluster_state is not None and self._state_version is not None:
            ops.append(jute.MultiOp.set_data(self._state_path, data,
                                             version=self._state_version))
        else:
            ops.append(jute.MultiOp.create(self._state_path, data,
                                           jute.PERSISTENT))
        results = await self._zk.multi(ops)
        self._cluster_state = state
        if results[1][0] == "setData":
            self._state_version = results[1][1].version
        else:
            self._state_version = 0
        self.log.info("cluster state written",
                      generation=state.get("generation"),
                      version=self._state_version)

    # --------------------------------------------------------------- status
    @property
    def active(self) -> List[dict]:
        return [p.as_dict() for p in self._active]

    @property
    def cluster_state(self) -> Optional[dict]:
        return self._cluster_state

    def status(self) -> dict:
        """Debug/status dump (ref status :592-599)."""
        return {
            "path": self._path,
            "id": self._id,
            "inited": self._inited,
            "clusterState": self._cluster_state,
            "clusterStateVersion": self._state_version,
            "active": self.active,
            "zkState": self._zk.state if self._zk else "closed",
            "sessionId": ("0x%x" % self._zk.session_id) if self._zk else None,
        }
