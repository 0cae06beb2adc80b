"""Application client — the node-manatee equivalent.

The reference keeps its client in a separate repo (node-manatee,
ref README.md:61-89): applications watch the same ZooKeeper ``state``
node the sitters maintain and receive ``topology`` events carrying the
shard's database URLs in replication order, so they always know where
the writable primary is without polling any database.

This module provides that client natively:

    client = ManateeClient(zk_conn_str, "1.moray")
    client.on("topology", lambda t: connect_to(t["primary"]))
    await client.start()

Events:

- ``ready``    — first topology resolved (fired once, after ``topology``)
- ``topology`` — {generation, primary, sync, async: [...], urls: [...],
  freeze, oneNodeWriteMode, state} whenever the cluster state CHANGES
  (deduped; ref node-manatee emits only on actual change)
- ``error``    — unrecoverable client error (session expiry is handled
  internally by rebuilding the session and re-reading state)
"""

from __future__ import annotations

import asyncio
import json
from typing import Callable, Dict, List, Optional

from .common.logging import Logger, null_logger
from .coord import jute
from .coord.zkclient import ZkClient


def topology_from_state(state: dict) -> dict:
    """Flatten a cluster-state document into the client-facing topology
    (ordered like node-manatee's: primary, sync, asyncs)."""
    def url(ident: Optional[dict]) -> Optional[str]:
        return (ident or {}).get("pgUrl") or None

    urls: List[str] = []
    for ident in [state.get("primary"), state.get("sync")] + \
            list(state.get("async") or []):
        u = url(ident)
        if u:
            urls.append(u)
    return {
        "generation": state.get("generation"),
        "primary": url(state.get("primary")),
        "sync": url(state.get("sync")),
        "async": [url(a) for a in state.get("async") or []],
        "urls": urls,
        "freeze": state.get("freeze"),
        "oneNodeWriteMode": bool(state.get("oneNodeWriteMode")),
        "state": state,
    }


class ManateeClient:
    def __init__(self, zk_conn_str: str, shard: str,
                 session_timeout_ms: int = 30000,
                 log: Optional[Logger] = None,
                 base: str = "/manatee"):
        self.zk_conn_str = zk_conn_str
        self.shard = shard
        self.session_timeout_ms = session_timeout_ms
        self.log = (log or null_logger()).child(component="ManateeClient",
                                                shard=shard)
        path = shard if shard.startswith("/") else "%s/%s" % (base, shard)
        self._state_path = path + "/state"
        self._handlers: Dict[str, List[Callable]] = {}
        self._zk: Optional[ZkClient] = None
        self._task: Optional[asyncio.Task] = None
        self._poke: Optional[asyncio.Event] = None
        self._closing = False
        self.topology: Optional[dict] = None
        self._ready = False

    # --------------------------------------------------------------- events
    def on(self, event: str, cb: Callable) -> "ManateeClient":
        self._handlers.setdefault(event, []).append(cb)
        return self

    def _emit(self, event: str, *args) -> None:
        for cb in self._handlers.get(event, []):
            try:
                cb(*args)
            except Exception as exc:
                self.log.error("handler error", event=event, err=exc)

    # ------------------------------------------------------------ lifecycle
    async def start(self, timeout_s: float = 30.0) -> None:
        self._closing = False
        self._poke = asyncio.Event()
        self._task = asyncio.get_running_loop().create_task(self._run())
        deadline = asyncio.get_running_loop().time() + timeout_s
        while not self._ready:
            if self._task.done():
                exc = self._task.exception()
                raise exc or RuntimeError("client stopped before ready")
            if asyncio.get_running_loop().time() > deadline:
                raise asyncio.TimeoutError(
                    "no topology within %ss (shard not set up?)" % timeout_s)
            await asyncio.sleep(0.02)

    async def close(self) -> None:
        self._closing = True
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except (asyncio.CancelledError, Exception):
                pass
            self._task = None
        if self._zk is not None:
            await self._zk.close()
            self._zk = None

    # ------------------------------------------------------------ main loop
    async def _run(self) -> None:
        """Session loop: (re)build the ZK session, watch the state node,
        emit deduped topology events.  Watch fires and session events both
        poke the same event; each wakeup re-reads + re-watches (the
        one-shot-watch discipline, ref lib/zookeeperMgr.js:204-264)."""
        backoff = 0.1
        while not self._closing:
            try:
                self._zk = ZkClient(
                    self.zk_conn_str,
                    session_timeout_ms=self.session_timeout_ms,
                    log=self.log, on_session=self._on_session)
                await self._zk.connect(timeout_s=10.0)
                backoff = 0.1
                while not self._closing:
                    # Clear BEFORE re-arming the watch: a notification
                    # dispatched between watch registration and wait() must
                    # survive until wait(), or the consumed one-shot watch is
                    # never re-registered and we'd serve stale topology.
                    self._poke.clear()
                    await self._read_and_watch()
                    await self._poke.wait()
            except asyncio.CancelledError:
                return
            except jute.ZkError as exc:
                if exc.code != jute.ZSESSIONEXPIRED:
                    self.log.warn("zk error; rebuilding session", err=exc)
            except Exception as exc:
                self.log.warn("client error; rebuilding session", err=exc)
            if self._zk is not None:
                try:
                    await self._zk.close()
                except Exception:
                    pass
                self._zk = None
            await asyncio.sleep(backoff)
            backoff = min(backoff * 2, 5.0)

    def _on_session(self, event: str) -> None:
        if event == "expired" and self._poke is not None:
            self._poke.set()

    def _on_watch(self, etype: int, path: str) -> None:
        if self._poke is not None:
            self._poke.set()

    async def _read_and_watch(self) -> None:
        try:
            data, _ = await self._zk.get_data(self._state_path,
                                              watch=self._on_watch)
        except jute.ZkError as exc:
            if exc.code != jute.ZNONODE:
                raise
            # shard not set up yet: wake on creation
            await self._zk.exists(self._state_path, watch=self._on_watch)
            return
        try:
            state = json.loads(data.decode("utf-8"))
        except ValueError as exc:
            self.log.error("unparseable cluster state", err=exc)
            return
        topo = topology_from_state(state)
        if self.topology is not None and \
                topo["state"] == self.topology["state"]:
            return
        self.topology = topo
        self.log.info("topology changed", generation=topo["generation"],
                      primary=topo["primary"])
        self._emit("topology", topo)
        if not self._ready:
            self._ready = True
            self._emit("ready", topo)
