"""Asyncio ZooKeeper client (standard 3.4.x wire protocol).

Clean-room equivalent of the ``joyent-zookeeper-client`` npm dependency the
reference uses for all coordination (package.json:33, lib/zookeeperMgr.js).
Speaks the protocol in ``jute.py`` against a real ZooKeeper or our embedded
``zkserver``.  Features used by the manager layer and therefore implemented:

- session establishment/maintenance (ping at timeout/3), auto-reconnect on
  connection loss across the ensemble list, session-expiry detection
  (ref zookeeperMgr.js:500-586 resets everything on expiry);
- one-shot watches on data/children with callback delivery and re-arming via
  SetWatches on reconnect (ref watch() re-registration zookeeperMgr.js:204-264);
- create (all 4 modes), delete, exists, getData, setData (versioned CAS),
  getChildren2, and atomic multi (ref putClusterState zookeeperMgr.js:605-630).
"""

from __future__ import annotations

import asyncio
import random
import struct
import time
from typing import Callable, Dict, List, Optional, Tuple

from ..common.logging import Logger, null_logger
from ..common import dial
from . import jute
from .jute import MultiOp, Reader, Stat, Writer, ZkError, ZOK, ZCONNECTIONLOSS

WatchCallback = Callable[[int, str], None]  # (event_type, path)
SessionCallback = Callable[[str], None]     # 'connected'|'disconnected'|'expired'


class ZkClient:
    def __init__(self, conn_str: str, session_timeout_ms: int = 30000,
                 log: Optional[Logger] = None,
                 on_session: Optional[SessionCallback] = None):
        self.servers: List[Tuple[str, int]] = []
        for part in conn_str.split(","):
            part = part.strip()
            if not part:
                continue
            host, _, port = part.partition(":")
            self.servers.append((host, int(port or 2181)))
        if not self.servers:
            raise ValueError("empty connection string")
        self.session_timeout_ms = session_timeout_ms
        self.log = (log or null_logger()).child(component="ZkClient")
        self.on_session = on_session

        self.session_id = 0
        self.session_passwd = b"\x00" * 16
        self.negotiated_timeout_ms = session_timeout_ms
        self.last_zxid = 0

        self.state = "closed"   # closed|connecting|connected|expired
        self._reader: Optional[asyncio.StreamReader] = None
        self._writer: Optional[asyncio.StreamWriter] = None
        self._xid = 0
        self._pending: Dict[int, asyncio.Future] = {}
        self._data_watches: Dict[str, List[WatchCallback]] = {}
        self._exists_watches: Dict[str, List[WatchCallback]] = {}
        self._child_watches: Dict[str, List[WatchCallback]] = {}
        self._io_task: Optional[asyncio.Task] = None
        self._ping_task: Optional[asyncio.Task] = None
        self._mgr_task: Optional[asyncio.Task] = None
        self._send_lock = asyncio.Lock()
        self._connected_evt = asyncio.Event()
        self._closing = False

    # ------------------------------------------------------------ lifecycle
    async def connect(self, timeout_s: float = 10.0) -> None:
        """Establish the session; raises on timeout."""
        if self.state not in ("closed",):
            raise RuntimeError("client already started")
        self._closing = False
        self.state = "connecting"
        self._mgr_task = asyncio.get_running_loop().create_task(
            self._connection_manager())
        await asyncio.wait_for(self._connected_evt.wait(), timeout_s)

    async def close(self) -> None:
        self._closing = True
        if self.state == "connected" and self._writer is not None:
            async def _send_close():
                async with self._send_lock:
                    w = jute.encode_request_header(self._next_xid(),
                                                   jute.OP_CLOSE_SESSION)
                    self._writer.write(w.framed())
                    await self._writer.drain()
            try:
                # bounded: a wedged server/socket must not hang close()
                # (the session expires server-side anyway)
                await asyncio.wait_for(_send_close(), 5.0)
            except Exception:
                pass
        for task in (self._mgr_task, self._io_task, self._ping_task):
            if task is not None:
                task.cancel()
                try:
                    # bounded: close() sits on the failover-critical
                    # session-rebuild path; a task whose cancel was
                    # lost (bpo-42130-style) must not wedge it
                    _done, pending = await asyncio.wait({task},
                                                        timeout=5.0)
                    for t in pending:
                        t.cancel()
                    if pending:
                        await asyncio.wait(pending, timeout=2.0)
                except (asyncio.CancelledError, Exception):
                    pass
        self._teardown_conn()
        self.state = "closed"

    # ------------------------------------------------------ connection mgmt
    def _teardown_conn(self) -> None:
        if self._writer is not None:
            try:
                self._writer.close()
            except Exception:
                pass
        self._reader = None
        self._writer = None
        for fut in self._pending.values():
            if not fut.done():
                fut.set_exception(ZkError(ZCONNECTIONLOSS))
        self._pending.clear()

    async def _connection_manager(self) -> None:
        """Maintains the session across TCP connections until close/expiry."""
        backoff = 0.05
        order = list(self.servers)
        while not self._closing:
            random.shuffle(order)
            connected = False
            for host, port in order:
                try:
                    await self._connect_once(host, port)
                    connected = True
                    break
                except _SessionExpired:
                    self.state = "expired"
                    self.log.warn("session expired",
                                  sid="0x%x" % self.session_id)
                    self._notify("expired")
                    return
                except (ConnectionError, OSError, asyncio.TimeoutError) as exc:
                    self.log.debug("connect failed", server="%s:%d" % (host, port),
                                   err=exc)
            if not connected:
                await asyncio.sleep(backoff)
                backoff = min(backoff * 2, 1.0)
                continue
            backoff = 0.05
            # block until the io task dies (connection lost)
            try:
                await asyncio.shield(self._io_task)
            except asyncio.CancelledError:
                # either THIS task was cancelled (teardown that never
                # called close(), e.g. an event loop shutting down) or
                # the io task was (close() is the only caller): stop
                # managing.  Swallowing this and looping on made the
                # manager task survive cancellation and reconnect
                # forever — asyncio.run's final task-gather then hung a
                # whole pytest process for its remaining per-test budget
                self._teardown_conn()
                raise
            except Exception:
                pass
            if self._closing:
                return
            was_connected = self.state == "connected"
            self.state = "connecting"
            self._teardown_conn()
            if self._ping_task is not None:
                self._ping_task.cancel()
            if was_connected:
                self._connected_evt.clear()
                self._notify("disconnected")

    async def _connect_once(self, host: str, port: int) -> None:
        reader, writer = await asyncio.wait_for(
            dial.open_connection(host, port), 5.0)
        try:
            writer.write(jute.encode_connect_request(
                self.last_zxid, self.session_timeout_ms,
                self.session_id, self.session_passwd
                if self.session_id else b"\x00" * 16))
            await writer.drain()
            body = await asyncio.wait_for(self._read_frame_from(reader), 5.0)
            if body is None:
                raise ConnectionError("handshake EOF")
            timeout_ms, sid, passwd = jute.decode_connect_response(body)
            if sid == 0:
                raise _SessionExpired()
            fresh = self.session_id == 0
            self.session_id = sid
            self.session_passwd = passwd
            self.negotiated_timeout_ms = timeout_ms
        except BaseException:
            writer.close()
            raise
        self._reader = reader
        self._writer = writer
        self.state = "connected"
        loop = asyncio.get_running_loop()
        self._io_task = loop.create_task(self._io_loop())
        self._ping_task = loop.create_task(self._ping_loop())
        self.log.debug("session " + ("established" if fresh else "reattached"),
                       sid="0x%x" % sid, timeout_ms=timeout_ms,
                       server="%s:%d" % (host, port))
        if not fresh:
            await self._rearm_watches()
        self._connected_evt.set()
        self._notify("connected")

    def _notify(self, event: str) -> None:
        if self.on_session is not None:
            try:
                self.on_session(event)
            except Exception as exc:
                self.log.error("session callback error", err=exc)

    async def _rearm_watches(self) -> None:
        data = list(self._data_watches)
        exists = list(self._exists_watches)
        children = list(self._child_watches)
        if not (data or exists or children):
            return
        w = jute.encode_request_header(jute.XID_SET_WATCHES,
                                       jute.OP_SETWATCHES)
        w.int64(self.last_zxid)
        for paths in (data, exists, children):
            w.int32(len(paths))
            for p in paths:
                w.ustring(p)
        async with self._send_lock:
            self._writer.write(w.framed())
            await self._writer.drain()

    # ------------------------------------------------------------- io loops
    async def _read_frame_from(self, reader) -> Optional[bytes]:
        try:
            hdr = await reader.readexactly(4)
        except (asyncio.IncompleteReadError, ConnectionError):
            return None
        (n,) = struct.unpack(">i", hdr)
        if n < 0 or n > 64 * 1024 * 1024:
            return None
        try:
            return await reader.readexactly(n)
        except (asyncio.IncompleteReadError, ConnectionError):
            return None

    async def _io_loop(self) -> None:
        reader = self._reader
        while True:
            body = await self._read_frame_from(reader)
            if body is None:
                return
            r = Reader(body)
            xid, zxid, err = jute.decode_reply_header(r)
            if zxid > 0:
                self.last_zxid = zxid
            if xid == jute.XID_NOTIFICATION:
                etype, _state, path = jute.decode_watcher_event(r)
                self._dispatch_watch(etype, path)
                continue
            if xid in (jute.XID_PING, jute.XID_AUTH, jute.XID_SET_WATCHES):
                continue
            fut = self._pending.pop(xid, None)
            if fut is None or fut.done():
                continue
            if err != ZOK:
                fut.set_exception(ZkError(err))
            else:
                fut.set_result(r)

    async def _ping_loop(self) -> None:
        interval = max(self.negotiated_timeout_ms / 3000.0, 0.1)
        while True:
            await asyncio.sleep(interval)
            if self._writer is None:
                return
            try:
                async with self._send_lock:
                    w = jute.encode_request_header(jute.XID_PING, jute.OP_PING)
                    self._writer.write(w.framed())
                    await self._writer.drain()
            except (ConnectionError, OSError):
                return

    def _dispatch_watch(self, etype: int, path: str) -> None:
        cbs: List[WatchCallback] = []
        if etype in (jute.EVENT_NODE_CREATED, jute.EVENT_NODE_DELETED,
                     jute.EVENT_NODE_DATA_CHANGED):
            cbs.extend(self._data_watches.pop(path, []))
            cbs.extend(self._exists_watches.pop(path, []))
        if etype in (jute.EVENT_NODE_CHILDREN_CHANGED,
                     jute.EVENT_NODE_DELETED):
            cbs.extend(self._child_watches.pop(path, []))
        if not cbs:
            self.log.warn("watch event with no registered callback",
                          etype=etype, path=path)
        for cb in cbs:
            try:
                cb(etype, path)
            except Exception as exc:
                self.log.error("watch callback error", path=path, err=exc)

    # ------------------------------------------------------------- requests
    def _next_xid(self) -> int:
        self._xid += 1
        if self._xid > 0x7FFFFFFF:
            self._xid = 1
        return self._xid

    async def _call(self, opcode: int, build: Callable[[Writer], None],
                    timeout_s: float = 10.0) -> Reader:
        if self.state == "expired":
            raise ZkError(jute.ZSESSIONEXPIRED)
        if self.state != "connected" or self._writer is None:
            raise ZkError(ZCONNECTIONLOSS)
        xid = self._next_xid()
        w = jute.encode_request_header(xid, opcode)
        build(w)
        fut: asyncio.Future = asyncio.get_running_loop().create_future()
        self._pending[xid] = fut
        try:
            async with self._send_lock:
                self._writer.write(w.framed())
                await self._writer.drain()
        except (ConnectionError, OSError):
            self._pending.pop(xid, None)
            raise ZkError(ZCONNECTIONLOSS)
        try:
            return await asyncio.wait_for(fut, timeout_s)
        except asyncio.TimeoutError:
            self._pending.pop(xid, None)
            raise ZkError(jute.ZOPERATIONTIMEOUT, "op %d" % opcode)

    # ---------------------------------------------------------- public API
    async def create(self, path: str, data: bytes = b"",
                     mode: int = jute.PERSISTENT) -> str:
        def build(w: Writer):
            w.ustring(path).buffer(data)
            jute.write_acls(w)
            w.int32(mode)
        r = await self._call(jute.OP_CREATE, build)
        return r.ustring() or ""

    async def delete(self, path: str, version: int = -1) -> None:
        def build(w: Writer):
            w.ustring(path).int32(version)
        await self._call(jute.OP_DELETE, build)

    async def exists(self, path: str,
                     watch: Optional[WatchCallback] = None) -> Optional[Stat]:
        def build(w: Writer):
            w.ustring(path).boolean(watch is not None)
        if watch is not None:
            self._exists_watches.setdefault(path, []).append(watch)
        try:
            r = await self._call(jute.OP_EXISTS, build)
        except ZkError as exc:
            if exc.code == jute.ZNONODE:
                return None
            if watch is not None:
                lst = self._exists_watches.get(path, [])
                if watch in lst:
                    lst.remove(watch)
            raise
        return Stat.read(r)

    async def get_data(self, path: str,
                       watch: Optional[WatchCallback] = None
                       ) -> Tuple[bytes, Stat]:
        def build(w: Writer):
            w.ustring(path).boolean(watch is not None)
        # register BEFORE sending: the read loop runs as its own task, so a
        # notification frame arriving right behind the response could be
        # dispatched before this coroutine resumes (watch would be lost)
        if watch is not None:
            self._data_watches.setdefault(path, []).append(watch)
        try:
            r = await self._call(jute.OP_GETDATA, build)
        except ZkError:
            if watch is not None:
                lst = self._data_watches.get(path, [])
                if watch in lst:
                    lst.remove(watch)
            raise
        data = r.buffer() or b""
        return data, Stat.read(r)

    async def set_data(self, path: str, data: bytes,
                       version: int = -1) -> Stat:
        def build(w: Writer):
            w.ustring(path).buffer(data).int32(version)
        r = await self._call(jute.OP_SETDATA, build)
        return Stat.read(r)

    async def get_children(self, path: str,
                           watch: Optional[WatchCallback] = None
                           ) -> Tuple[List[str], Stat]:
        def build(w: Writer):
            w.ustring(path).boolean(watch is not None)
        # register BEFORE sending (see get_data): avoids losing an event
        # delivered between the response and this coroutine resuming
        if watch is not None:
            self._child_watches.setdefault(path, []).append(watch)
        try:
            r = await self._call(jute.OP_GETCHILDREN2, build)
        except ZkError:
            if watch is not None:
                lst = self._child_watches.get(path, [])
                if watch in lst:
                    lst.remove(watch)
            raise
        n = r.int32()
        children = [r.ustring() or "" for _ in range(max(n, 0))]
        return children, Stat.read(r)

    async def multi(self, ops: List[MultiOp]) -> List[tuple]:
        def build(w: Writer):
            jute.write_multi_request(w, ops)
        r = await self._call(jute.OP_MULTI, build)
        results = jute.read_multi_response(r)
        for res in results:
            if res[0] == "error" and res[1] != ZOK:
                raise ZkError(res[1])
        return results

    async def mkdirp(self, path: str) -> None:
        """Create path and parents (persistent), ignoring NODE_EXISTS —
        the manager uses this for shardPath/election/history setup
        (ref zookeeperMgr.js:412-440)."""
        parts = [p for p in path.split("/") if p]
        cur = ""
        for part in parts:
            cur += "/" + part
            try:
                await self.create(cur, b"")
            except ZkError as exc:
                if exc.code != jute.ZNODEEXISTS:
                    raise


class _SessionExpired(Exception):
    pass
