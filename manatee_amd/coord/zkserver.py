"""Embedded ZooKeeper-protocol coordination server.

The reference assumes an external ZooKeeper 3.4.x ensemble (docs/user-guide.md
"Zookeeper", lib/zookeeperMgr.js).  This image (and many deploy targets) has
no Java, so the framework ships its own coordination server speaking the same
client wire protocol (``jute.py``): sessions with timeouts, ephemeral and
sequential znodes, one-shot data/child watches, versioned setData (CAS) and
atomic multi-op transactions — the exact feature set ``zookeeperMgr.js``
consumes (ephemeral-sequential election at :452-456, one-shot watch
re-registration at :204-264, transaction with versioned setData at :605-630).

A real ZooKeeper ensemble can be substituted 1:1 — the client side
(``zkclient.py``) speaks the standard protocol and never imports this module.

The store is in-memory with an optional append-only journal for restart
durability (``journal_path``).  Coordination data is tiny (one state JSON +
election nodes + history); the journal replays in milliseconds.
"""

from __future__ import annotations

import asyncio
import json
import os
import random
import re
import struct
import time
from typing import Dict, List, Optional, Set, Tuple

from ..common.logging import Logger, null_logger
from . import jute
from .jute import (MultiOp, Reader, Stat, Writer, ZkError, ZOK, ZNONODE,
                   ZNODEEXISTS, ZBADVERSION, ZNOTEMPTY, ZSESSIONEXPIRED,
                   ZNOCHILDRENFOREPHEMERALS, ZUNIMPLEMENTED, ZMARSHALLINGERROR)


def _now_ms() -> int:
    return int(time.time() * 1000)


class _Node:
    __slots__ = ("data", "children", "stat_czxid", "stat_mzxid", "ctime",
                 "mtime", "version", "cversion", "ephemeral_owner", "next_seq",
                 "pzxid")

    def __init__(self, data: bytes, czxid: int, ephemeral_owner: int = 0):
        self.data = data
        self.children: Set[str] = set()
        self.stat_czxid = czxid
        self.stat_mzxid = czxid
        self.ctime = _now_ms()
        self.mtime = self.ctime
        self.version = 0
        self.cversion = 0
        self.ephemeral_owner = ephemeral_owner
        self.next_seq = 0
        self.pzxid = czxid

    def stat(self) -> Stat:
        return Stat(czxid=self.stat_czxid, mzxid=self.stat_mzxid,
                    ctime=self.ctime, mtime=self.mtime, version=self.version,
                    cversion=self.cversion, aversion=0,
                    ephemeralOwner=self.ephemeral_owner,
                    dataLength=len(self.data), numChildren=len(self.children),
                    pzxid=self.pzxid)


class _Session:
    __slots__ = ("sid", "passwd", "timeout_ms", "last_seen", "ephemerals",
                 "conn", "closed")

    def __init__(self, sid: int, passwd: bytes, timeout_ms: int):
        self.sid = sid
        self.passwd = passwd
        self.timeout_ms = timeout_ms
        self.last_seen = time.monotonic()
        self.ephemerals: Set[str] = set()
        self.conn: Optional[asyncio.StreamWriter] = None
        self.closed = False


_TRAILING_SEQ = re.compile(r"(\d{10})$")


def _parent(path: str) -> str:
    i = path.rfind("/")
    return path[:i] if i > 0 else "/"


def _validate_path(path: str) -> str:
    if not path.startswith("/") or (len(path) > 1 and path.endswith("/")):
        raise ZkError(jute.ZAPIERROR, path)
    return path


class ZkServer:
    """One standalone coordination server (the test/dev substitute for a
    3-node ZK ensemble)."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0,
                 log: Optional[Logger] = None,
                 tick_ms: int = 200,
                 min_session_timeout_ms: int = 1000,
                 max_session_timeout_ms: int = 120000,
                 journal_path: Optional[str] = None):
        self.host = host
        self.port = port
        self.log = (log or null_logger()).child(component="ZkServer")
        self.tick_ms = tick_ms
        self.min_to = min_session_timeout_ms
        self.max_to = max_session_timeout_ms
        self.journal_path = journal_path
        self._journal = None
        self._journal_entries = 0
        self.journal_compact_entries = 5000

        self.nodes: Dict[str, _Node] = {"/": _Node(b"", 0)}
        self.sessions: Dict[int, _Session] = {}
        self.zxid = 0
        self._next_sid = (random.getrandbits(24) << 24) | 1
        self.data_watches: Dict[str, Set[int]] = {}
        self.child_watches: Dict[str, Set[int]] = {}
        self._server: Optional[asyncio.AbstractServer] = None
        self._sweeper: Optional[asyncio.Task] = None
        self.stats = {"requests": 0, "watch_events": 0, "expired_sessions": 0}

    # ------------------------------------------------------------ lifecycle
    async def start(self) -> None:
        if self.journal_path and os.path.exists(self.journal_path):
            self._replay_journal()
        if self.journal_path:
            self._journal = open(self.journal_path, "a", buffering=1)
        self._server = await asyncio.start_server(
            self._handle_conn, self.host, self.port)
        self.port = self._server.sockets[0].getsockname()[1]
        self._sweeper = asyncio.get_running_loop().create_task(
            self._sweep_sessions())
        self.log.info("zk server listening", host=self.host, port=self.port)

    async def stop(self) -> None:
        if self._sweeper:
            self._sweeper.cancel()
            try:
                await self._sweeper
            except asyncio.CancelledError:
                pass
        if self._server:
            self._server.close()
            await self._server.wait_closed()
        for sess in list(self.sessions.values()):
            if sess.conn is not None:
                try:
                    sess.conn.close()
                except Exception:
                    pass
        if self._journal:
            self._journal.close()
            self._journal = None

    @property
    def conn_str(self) -> str:
        return "%s:%d" % (self.host, self.port)

    # ------------------------------------------------------------- journal
    def _journal_write(self, op: dict) -> None:
        if self._journal is not None:
            self._journal.write(json.dumps(op, separators=(",", ":")) + "\n")
            self._journal_entries += 1
            if self._journal_entries >= self.journal_compact_entries and \
                    self._journal_entries >= 2 * max(1, len(self.nodes)):
                try:
                    self._compact_journal()
                except OSError as exc:
                    self.log.error("journal compaction failed", err=exc)

    def _replay_journal(self) -> None:
        n = 0
        with open(self.journal_path, "r") as f:
            for line in f:
                line = line.strip()
                if not line:
                    continue
                try:
                    op = json.loads(line)
                except ValueError:
                    # torn tail from a dirty kill mid-append; nothing
                    # after it can be valid either
                    self.log.warn("journal line unparseable; stopping "
                                  "replay", entries=n)
                    break
                try:
                    if op["op"] == "create":
                        # journaled paths are FINAL (already sequenced):
                        # always replay as plain persistent creates, then
                        # restore the parent's sequence counter so new
                        # sequential nodes continue after the replayed ones
                        self._do_create(op["path"],
                                        bytes.fromhex(op["data"]),
                                        jute.PERSISTENT, 0,
                                        journal=False)
                        m = _TRAILING_SEQ.search(op["path"])
                        if m:
                            parent = self.nodes.get(_parent(op["path"]))
                            if parent is not None:
                                parent.next_seq = max(
                                    parent.next_seq, int(m.group(1)) + 1)
                    elif op["op"] == "setData":
                        self._do_set_data(op["path"],
                                          bytes.fromhex(op["data"]), -1,
                                          journal=False)
                    elif op["op"] == "delete":
                        self._do_delete(op["path"], -1, journal=False)
                except ZkError:
                    pass
                n += 1
        self._journal_entries = n
        self.log.info("journal replayed", entries=n)

    def _compact_journal(self) -> None:
        """Rewrite the journal as a snapshot of the current persistent
        tree (one create per node) — bounds the append-only journal,
        which otherwise grows with every cluster-state write forever."""
        tmp = self.journal_path + ".tmp"
        count = 0
        with open(tmp, "w") as f:
            for path in sorted(self.nodes):
                node = self.nodes[path]
                if path == "/" or node.ephemeral_owner:
                    continue
                f.write(json.dumps(
                    {"op": "create", "path": path,
                     "data": node.data.hex(), "flags": jute.PERSISTENT},
                    separators=(",", ":")) + "\n")
                count += 1
            f.flush()
            os.fsync(f.fileno())
        if self._journal is not None:
            self._journal.close()
        os.replace(tmp, self.journal_path)
        self._journal = open(self.journal_path, "a", buffering=1)
        self._journal_entries = count
        self.log.info("journal compacted", nodes=count)

    # ------------------------------------------------------- session sweeper
    async def _sweep_sessions(self) -> None:
        # sweep at tick/4 (min 20 ms): expiry granularity is pure added
        # failover-detection latency and the scan is O(sessions), tiny
        while True:
            await asyncio.sleep(max(0.02, self.tick_ms / 4000.0))
            now = time.monotonic()
            for sess in list(self.sessions.values()):
                if sess.closed:
                    continue
                if (now - sess.last_seen) * 1000.0 > sess.timeout_ms:
                    self.log.info("session expired", sid="0x%x" % sess.sid,
                                  timeout_ms=sess.timeout_ms)
                    self.stats["expired_sessions"] += 1
                    self._expire_session(sess)

    def _expire_session(self, sess: _Session) -> None:
        sess.closed = True
        self.sessions.pop(sess.sid, None)
        for path in sorted(sess.ephemerals):
            try:
                self._do_delete(path, -1, force=True)
            except ZkError:
                pass
        for watchset in list(self.data_watches.values()):
            watchset.discard(sess.sid)
        for watchset in list(self.child_watches.values()):
            watchset.discard(sess.sid)
        if sess.conn is not None:
            try:
                sess.conn.close()
            except Exception:
                pass
            sess.conn = None

    # --------------------------------------------------------- watch firing
    def _fire_data_watch(self, path: str, etype: int) -> None:
        sids = self.data_watches.pop(path, None)
        if sids:
            self._deliver(sids, etype, path)

    def _fire_child_watch(self, path: str, etype: int = jute.EVENT_NODE_CHILDREN_CHANGED) -> None:
        sids = self.child_watches.pop(path, None)
        if sids:
            self._deliver(sids, etype, path)

    def _deliver(self, sids: Set[int], etype: int, path: str) -> None:
        frame = jute.encode_watcher_event(etype, jute.STATE_SYNC_CONNECTED,
                                          path)
        for sid in sids:
            sess = self.sessions.get(sid)
            if sess is not None and sess.conn is not None:
                try:
                    sess.conn.write(frame)
                    self.stats["watch_events"] += 1
                except Exception:
                    pass

    # ------------------------------------------------------------- core ops
    def _do_create(self, path: str, data: bytes, flags: int, owner_sid: int,
                   journal: bool = True) -> str:
        _validate_path(path)
        parent_path = _parent(path)
        parent = self.nodes.get(parent_path)
        if parent is None:
            raise ZkError(ZNONODE, parent_path)
        if parent.ephemeral_owner:
            raise ZkError(ZNOCHILDRENFOREPHEMERALS, parent_path)
        if flags in (jute.PERSISTENT_SEQUENTIAL, jute.EPHEMERAL_SEQUENTIAL):
            path = "%s%010d" % (path, parent.next_seq)
            parent.next_seq += 1
        if path in self.nodes:
            raise ZkError(ZNODEEXISTS, path)
        self.zxid += 1
        ephemeral = flags in (jute.EPHEMERAL, jute.EPHEMERAL_SEQUENTIAL)
        node = _Node(data, self.zxid,
                     ephemeral_owner=owner_sid if ephemeral else 0)
        self.nodes[path] = node
        parent.children.add(path[len(parent_path):].lstrip("/"))
        parent.cversion += 1
        parent.pzxid = self.zxid
        if ephemeral and owner_sid in self.sessions:
            self.sessions[owner_sid].ephemerals.add(path)
        if journal and not ephemeral:
            self._journal_write({"op": "create", "path": path,
                                 "data": data.hex(), "flags": flags})
        self._fire_data_watch(path, jute.EVENT_NODE_CREATED)
        self._fire_child_watch(parent_path)
        return path

    def _do_delete(self, path: str, version: int, force: bool = False,
                   journal: bool = True) -> None:
        node = self.nodes.get(path)
        if node is None:
            raise ZkError(ZNONODE, path)
        if node.children:
            raise ZkError(ZNOTEMPTY, path)
        if version != -1 and version != node.version and not force:
            raise ZkError(ZBADVERSION, path)
        self.zxid += 1
        del self.nodes[path]
        parent_path = _parent(path)
        parent = self.nodes.get(parent_path)
        if parent is not None:
            parent.children.discard(path[len(parent_path):].lstrip("/"))
            parent.cversion += 1
            parent.pzxid = self.zxid
        if node.ephemeral_owner and node.ephemeral_owner in self.sessions:
            self.sessions[node.ephemeral_owner].ephemerals.discard(path)
        if journal and not node.ephemeral_owner:
            self._journal_write({"op": "delete", "path": path})
        self._fire_data_watch(path, jute.EVENT_NODE_DELETED)
        self._fire_child_watch(path, jute.EVENT_NODE_DELETED)
        self._fire_child_watch(parent_path)

    def _do_set_data(self, path: str, data: bytes, version: int,
                     journal: bool = True) -> Stat:
        node = self.nodes.get(path)
        if node is None:
            raise ZkError(ZNONODE, path)
        if version != -1 and version != node.version:
            raise ZkError(ZBADVERSION, path)
        self.zxid += 1
        node.data = data
        node.version += 1
        node.mtime = _now_ms()
        node.stat_mzxid = self.zxid
        if journal and not node.ephemeral_owner:
            self._journal_write({"op": "setData", "path": path,
                                 "data": data.hex()})
        self._fire_data_watch(path, jute.EVENT_NODE_DATA_CHANGED)
        return node.stat()

    def _do_multi(self, ops: List[MultiOp], owner_sid: int) -> List[tuple]:
        # validate-then-apply so the transaction is atomic
        # (ref zookeeperMgr.js:605-630 relies on this for putClusterState CAS)
        for op in ops:
            if op.kind == "create":
                pp = _parent(op.path)
                parent = self.nodes.get(pp)
                if parent is None:
                    raise ZkError(ZNONODE, pp)
                if op.flags == jute.PERSISTENT and op.path in self.nodes:
                    raise ZkError(ZNODEEXISTS, op.path)
            elif op.kind in ("setData", "check", "delete"):
                node = self.nodes.get(op.path)
                if node is None:
                    raise ZkError(ZNONODE, op.path)
                if op.version != -1 and node.version != op.version:
                    raise ZkError(ZBADVERSION, op.path)
                if op.kind == "delete" and node.children:
                    raise ZkError(ZNOTEMPTY, op.path)
        results: List[tuple] = []
        for op in ops:
            if op.kind == "create":
                results.append(("create",
                                self._do_create(op.path, op.data or b"",
                                                op.flags, owner_sid)))
            elif op.kind == "setData":
                results.append(("setData",
                                self._do_set_data(op.path, op.data or b"",
                                                  op.version)))
            elif op.kind == "delete":
                self._do_delete(op.path, op.version)
                results.append(("delete",))
            else:
                results.append(("check",))
        return results

    # -------------------------------------------------------- conn handling
    async def _read_frame(self, reader: asyncio.StreamReader) -> Optional[bytes]:
        try:
            hdr = await reader.readexactly(4)
        except (asyncio.IncompleteReadError, ConnectionError):
            return None
        (n,) = struct.unpack(">i", hdr)
        if n < 0 or n > 8 * 1024 * 1024:
            return None
        try:
            return await reader.readexactly(n)
        except (asyncio.IncompleteReadError, ConnectionError):
            return None

    async def _handle_conn(self, reader: asyncio.StreamReader,
                           writer: asyncio.StreamWriter) -> None:
        sess: Optional[_Session] = None
        try:
            body = await self._read_frame(reader)
            if body is None:
                return
            _zxid, timeout_ms, sid, passwd = jute.decode_connect_request(body)
            if sid != 0:
                old = self.sessions.get(sid)
                if old is not None and old.passwd == passwd and not old.closed:
                    sess = old
                    if sess.conn is not None and sess.conn is not writer:
                        try:
                            sess.conn.close()
                        except Exception:
                            pass
                    sess.conn = writer
                    sess.last_seen = time.monotonic()
                    writer.write(jute.encode_connect_response(
                        sess.timeout_ms, sess.sid, sess.passwd))
                else:
                    # expired / unknown session: sessionId 0 tells the client
                    writer.write(jute.encode_connect_response(
                        timeout_ms, 0, b"\x00" * 16))
                    await writer.drain()
                    return
            else:
                timeout_ms = max(self.min_to, min(self.max_to, timeout_ms))
                sid = self._next_sid
                self._next_sid += 1
                sess = _Session(sid, os.urandom(16), timeout_ms)
                sess.conn = writer
                self.sessions[sid] = sess
                self.log.debug("session created", sid="0x%x" % sid,
                               timeout_ms=timeout_ms)
                writer.write(jute.encode_connect_response(
                    timeout_ms, sid, sess.passwd))
            await writer.drain()

            while True:
                body = await self._read_frame(reader)
                if body is None:
                    break
                if sess.closed:
                    break
                sess.last_seen = time.monotonic()
                r = Reader(body)
                xid, opcode = jute.decode_request_header(r)
                self.stats["requests"] += 1
                if opcode == jute.OP_CLOSE_SESSION:
                    w = jute.encode_reply_header(xid, self.zxid, ZOK)
                    writer.write(w.framed())
                    await writer.drain()
                    sess.closed = True
                    self.sessions.pop(sess.sid, None)
                    for path in sorted(sess.ephemerals):
                        try:
                            self._do_delete(path, -1, force=True)
                        except ZkError:
                            pass
                    for ws in list(self.data_watches.values()):
                        ws.discard(sess.sid)
                    for ws in list(self.child_watches.values()):
                        ws.discard(sess.sid)
                    break
                frame = self._dispatch(sess, xid, opcode, r)
                writer.write(frame)
                await writer.drain()
        except (ConnectionError, asyncio.CancelledError):
            pass
        except Exception as exc:  # never let one conn kill the server
            self.log.error("connection handler error", err=exc)
        finally:
            if sess is not None and sess.conn is writer:
                sess.conn = None  # session stays until timeout (disconnect != expiry)
            try:
                writer.close()
            except Exception:
                pass

    def _dispatch(self, sess: _Session, xid: int, opcode: int,
                  r: Reader) -> bytes:
        try:
            if opcode == jute.OP_PING:
                return jute.encode_reply_header(jute.XID_PING, self.zxid,
                                                ZOK).framed()
            if opcode == jute.OP_CREATE:
                path = r.ustring() or ""
                data = r.buffer() or b""
                jute.read_acls(r)
                flags = r.int32()
                created = self._do_create(path, data, flags, sess.sid)
                w = jute.encode_reply_header(xid, self.zxid, ZOK)
                w.ustring(created)
                return w.framed()
            if opcode == jute.OP_DELETE:
                path = r.ustring() or ""
                version = r.int32()
                self._do_delete(path, version)
                return jute.encode_reply_header(xid, self.zxid, ZOK).framed()
            if opcode == jute.OP_EXISTS:
                path = r.ustring() or ""
                watch = r.boolean()
                node = self.nodes.get(path)
                if watch:
                    # exists watches register even on missing nodes
                    self.data_watches.setdefault(path, set()).add(sess.sid)
                if node is None:
                    raise ZkError(ZNONODE, path)
                w = jute.encode_reply_header(xid, self.zxid, ZOK)
                node.stat().write(w)
                return w.framed()
            if opcode == jute.OP_GETDATA:
                path = r.ustring() or ""
                watch = r.boolean()
                node = self.nodes.get(path)
                if node is None:
                    # a getData watch on a missing node is NOT registered
                    raise ZkError(ZNONODE, path)
                if watch:
                    self.data_watches.setdefault(path, set()).add(sess.sid)
                w = jute.encode_reply_header(xid, self.zxid, ZOK)
                w.buffer(node.data)
                node.stat().write(w)
                return w.framed()
            if opcode == jute.OP_SETDATA:
                path = r.ustring() or ""
                data = r.buffer() or b""
                version = r.int32()
                stat = self._do_set_data(path, data, version)
                w = jute.encode_reply_header(xid, self.zxid, ZOK)
                stat.write(w)
                return w.framed()
            if opcode in (jute.OP_GETCHILDREN, jute.OP_GETCHILDREN2):
                path = r.ustring() or ""
                watch = r.boolean()
                node = self.nodes.get(path)
                if node is None:
                    raise ZkError(ZNONODE, path)
                if watch:
                    self.child_watches.setdefault(path, set()).add(sess.sid)
                w = jute.encode_reply_header(xid, self.zxid, ZOK)
                children = sorted(node.children)
                w.int32(len(children))
                for c in children:
                    w.ustring(c)
                if opcode == jute.OP_GETCHILDREN2:
                    node.stat().write(w)
                return w.framed()
            if opcode == jute.OP_MULTI:
                ops = jute.read_multi_request(r)
                try:
                    results = self._do_multi(ops, sess.sid)
                except ZkError as exc:
                    w = jute.encode_reply_header(xid, self.zxid, ZOK)
                    jute.write_multi_response(w, [("error", exc.code)])
                    return w.framed()
                w = jute.encode_reply_header(xid, self.zxid, ZOK)
                jute.write_multi_response(w, results)
                return w.framed()
            if opcode == jute.OP_SYNC:
                path = r.ustring() or ""
                w = jute.encode_reply_header(xid, self.zxid, ZOK)
                w.ustring(path)
                return w.framed()
            if opcode == jute.OP_AUTH:
                return jute.encode_reply_header(jute.XID_AUTH, self.zxid,
                                                ZOK).framed()
            if opcode == jute.OP_SETWATCHES:
                # relativeZxid, then 3 vectors of paths
                r.int64()
                for watchmap in (self.data_watches, self.data_watches,
                                 self.child_watches):
                    n = r.int32()
                    for _ in range(max(n, 0)):
                        p = r.ustring() or ""
                        watchmap.setdefault(p, set()).add(sess.sid)
                return jute.encode_reply_header(jute.XID_SET_WATCHES,
                                                self.zxid, ZOK).framed()
            raise ZkError(ZUNIMPLEMENTED)
        except ZkError as exc:
            return jute.encode_reply_header(xid, self.zxid, exc.code).framed()
        except (ValueError, struct.error):
            return jute.encode_reply_header(xid, self.zxid,
                                            ZMARSHALLINGERROR).framed()

    # ------------------------------------------------------------ test aid
    def dump_tree(self) -> Dict[str, dict]:
        out = {}
        for path, node in sorted(self.nodes.items()):
            out[path] = {"data": node.data.decode("utf-8", "replace"),
                         "version": node.version,
                         "ephemeralOwner": node.ephemeral_owner,
                         "children": sorted(node.children)}
        return out


async def run_standalone(host: str, port: int, log: Logger,
                         journal_path: Optional[str] = None) -> None:
    srv = ZkServer(host=host, port=port, log=log, journal_path=journal_path)
    await srv.start()
    try:
        while True:
            await asyncio.sleep(3600)
    finally:
        await srv.stop()


def main(argv=None) -> int:
    import argparse

    from ..common.logging import Logger as _Logger, level_from_verbosity
    ap = argparse.ArgumentParser(prog="manatee-zk")
    ap.add_argument("-H", "--host", default="127.0.0.1")
    ap.add_argument("-p", "--port", type=int, default=2181)
    ap.add_argument("-j", "--journal", default=None)
    ap.add_argument("-v", "--verbose", action="count", default=0)
    ns = ap.parse_args(argv)
    log = _Logger("manatee-zk", level=level_from_verbosity(ns.verbose))
    try:
        asyncio.run(run_standalone(ns.host, ns.port, log,
                                   journal_path=ns.journal))
    except KeyboardInterrupt:
        pass
    return 0


if __name__ == "__main__":
    import sys
    sys.exit(main())
