"""ZooKeeper "jute" wire-protocol records.

The reference talks to ZooKeeper through the pure-JS ``joyent-zookeeper-client``
(package.json:33) and ships a prebuilt C client (``deps/zookeeper/
libzookeeper_mt.a``).  This module is the from-scratch
equivalent of that native layer: encoders/decoders for the ZooKeeper 3.4.x
client protocol, shared by both our client (``zkclient.py``) and our embedded
server (``zkserver.py``), so the client remains usable against a real
ZooKeeper ensemble and the on-ZK namespace/state stays protocol-compatible.

All integers are big-endian.  Every frame on the wire is length-prefixed with
an int32.  The primitive Writer/Reader come from the C++ extension
``manatee_amd.native._jutec`` when it is built (``__graft_entry__.build()``
compiles it in-tree; ``tests/test_native_jute.py`` fuzzes byte parity);
``MANATEE_PURE_PY=1`` forces the pure-Python codec defined below, which is
also the fallback when the extension is absent.  ``CODEC`` says which one
is active.
"""

from __future__ import annotations

import os
import struct
from typing import List, Optional, Tuple

# ---------------------------------------------------------------- op codes
OP_NOTIFICATION = 0
OP_CREATE = 1
OP_DELETE = 2
OP_EXISTS = 3
OP_GETDATA = 4
OP_SETDATA = 5
OP_GETACL = 6
OP_SETACL = 7
OP_GETCHILDREN = 8
OP_SYNC = 9
OP_PING = 11
OP_GETCHILDREN2 = 12
OP_CHECK = 13
OP_MULTI = 14
OP_AUTH = 100
OP_SETWATCHES = 101
OP_CLOSE_SESSION = -11
OP_ERROR = -1

# ---------------------------------------------------------------- xids
XID_NOTIFICATION = -1
XID_PING = -2
XID_AUTH = -4
XID_SET_WATCHES = -8

# ---------------------------------------------------------------- errors
ZOK = 0
ZSYSTEMERROR = -1
ZRUNTIMEINCONSISTENCY = -2
ZCONNECTIONLOSS = -4
ZMARSHALLINGERROR = -5
ZUNIMPLEMENTED = -6
ZOPERATIONTIMEOUT = -7
ZAPIERROR = -100
ZNONODE = -101
ZNOAUTH = -102
ZBADVERSION = -103
ZNOCHILDRENFOREPHEMERALS = -108
ZNODEEXISTS = -110
ZNOTEMPTY = -111
ZSESSIONEXPIRED = -112
ZINVALIDCALLBACK = -113
ZINVALIDACL = -114
ZAUTHFAILED = -115
ZSESSIONMOVED = -118

ERROR_NAMES = {
    ZOK: "OK", ZCONNECTIONLOSS: "CONNECTION_LOSS", ZNONODE: "NO_NODE",
    ZNOAUTH: "NO_AUTH", ZBADVERSION: "BAD_VERSION",
    ZNODEEXISTS: "NODE_EXISTS", ZNOTEMPTY: "NOT_EMPTY",
    ZSESSIONEXPIRED: "SESSION_EXPIRED", ZAUTHFAILED: "AUTH_FAILED",
    ZNOCHILDRENFOREPHEMERALS: "NO_CHILDREN_FOR_EPHEMERALS",
    ZRUNTIMEINCONSISTENCY: "RUNTIME_INCONSISTENCY",
    ZMARSHALLINGERROR: "MARSHALLING_ERROR", ZUNIMPLEMENTED: "UNIMPLEMENTED",
    ZOPERATIONTIMEOUT: "OPERATION_TIMEOUT", ZAPIERROR: "API_ERROR",
    ZSYSTEMERROR: "SYSTEM_ERROR",
}

# ---------------------------------------------------------------- create modes
PERSISTENT = 0
EPHEMERAL = 1
PERSISTENT_SEQUENTIAL = 2
EPHEMERAL_SEQUENTIAL = 3

# ---------------------------------------------------------------- watch events
EVENT_NODE_CREATED = 1
EVENT_NODE_DELETED = 2
EVENT_NODE_DATA_CHANGED = 3
EVENT_NODE_CHILDREN_CHANGED = 4

STATE_DISCONNECTED = 0
STATE_SYNC_CONNECTED = 3
STATE_AUTH_FAILED = 4
STATE_EXPIRED = -112

OPEN_ACL_UNSAFE = [(0x1F, "world", "anyone")]  # perms=ALL


class ZkError(Exception):
    def __init__(self, code: int, path: str = ""):
        self.code = code
        self.path = path
        super().__init__("%s (%d) %s"
                         % (ERROR_NAMES.get(code, "ZK_ERROR"), code, path))


# ======================================================================
# primitive codec (pure Python; C++ _jute can replace Writer/Reader)
# ======================================================================

class Writer:
    __slots__ = ("_parts",)

    def __init__(self):
        self._parts: List[bytes] = []

    def int32(self, v: int) -> "Writer":
        self._parts.append(struct.pack(">i", v))
        return self

    def int64(self, v: int) -> "Writer":
        self._parts.append(struct.pack(">q", v))
        return self

    def boolean(self, v: bool) -> "Writer":
        self._parts.append(b"\x01" if v else b"\x00")
        return self

    def buffer(self, v: Optional[bytes]) -> "Writer":
        if v is None:
            self._parts.append(struct.pack(">i", -1))
        else:
            self._parts.append(struct.pack(">i", len(v)))
            self._parts.append(v)
        return self

    def ustring(self, v: Optional[str]) -> "Writer":
        return self.buffer(None if v is None else v.encode("utf-8"))

    def raw(self, v: bytes) -> "Writer":
        self._parts.append(v)
        return self

    def tobytes(self) -> bytes:
        return b"".join(self._parts)

    def framed(self) -> bytes:
        body = self.tobytes()
        return struct.pack(">i", len(body)) + body


class Reader:
    __slots__ = ("_buf", "_pos")

    def __init__(self, buf: bytes):
        self._buf = buf
        self._pos = 0

    def int32(self) -> int:
        v = struct.unpack_from(">i", self._buf, self._pos)[0]
        self._pos += 4
        return v

    def int64(self) -> int:
        v = struct.unpack_from(">q", self._buf, self._pos)[0]
        self._pos += 8
        return v

    def boolean(self) -> bool:
        v = self._buf[self._pos] != 0
        self._pos += 1
        return v

    def buffer(self) -> Optional[bytes]:
        n = self.int32()
        if n < 0:
            return None
        v = self._buf[self._pos:self._pos + n]
        if len(v) != n:
            raise ValueError("short buffer")
        self._pos += n
        return v

    def ustring(self) -> Optional[str]:
        b = self.buffer()
        return None if b is None else b.decode("utf-8")

    def remaining(self) -> int:
        return len(self._buf) - self._pos


# keep the pure-Python implementations importable under stable names
# (the parity fuzz test compares them against the native ones)
PyWriter = Writer
PyReader = Reader

CODEC = "python"
if os.environ.get("MANATEE_PURE_PY") != "1":
    try:
        from ..native import jutec as _jutec
    except ImportError:     # pragma: no cover - package layout issue only
        _jutec = None
    if _jutec is not None:
        Writer = _jutec.Writer      # type: ignore[misc]
        Reader = _jutec.Reader      # type: ignore[misc]
        CODEC = "native"


# ======================================================================
# records
# ======================================================================

class Stat:
    __slots__ = ("czxid", "mzxid", "ctime", "mtime", "version", "cversion",
                 "aversion", "ephemeralOwner", "dataLength", "numChildren",
                 "pzxid")

    def __init__(self, czxid=0, mzxid=0, ctime=0, mtime=0, version=0,
                 cversion=0, aversion=0, ephemeralOwner=0, dataLength=0,
                 numChildren=0, pzxid=0):
        self.czxid = czxid
        self.mzxid = mzxid
        self.ctime = ctime
        self.mtime = mtime
        self.version = version
        self.cversion = cversion
        self.aversion = aversion
        self.ephemeralOwner = ephemeralOwner
        self.dataLength = dataLength
        self.numChildren = numChildren
        self.pzxid = pzxid

    def write(self, w: Writer) -> None:
        (w.int64(self.czxid).int64(self.mzxid).int64(self.ctime)
         .int64(self.mtime).int32(self.version).int32(self.cversion)
         .int32(self.aversion).int64(self.ephemeralOwner)
         .int32(self.dataLength).int32(self.numChildren).int64(self.pzxid))

    @classmethod
    def read(cls, r: Reader) -> "Stat":
        s = cls()
        s.czxid = r.int64()
        s.mzxid = r.int64()
        s.ctime = r.int64()
        s.mtime = r.int64()
        s.version = r.int32()
        s.cversion = r.int32()
        s.aversion = r.int32()
        s.ephemeralOwner = r.int64()
        s.dataLength = r.int32()
        s.numChildren = r.int32()
        s.pzxid = r.int64()
        return s

    def as_dict(self) -> dict:
        return {k: getattr(self, k) for k in self.__slots__}

    def __repr__(self):
        return "Stat(%s)" % ", ".join("%s=%r" % (k, getattr(self, k))
                                      for k in self.__slots__)


def write_acls(w: Writer, acls=OPEN_ACL_UNSAFE) -> None:
    w.int32(len(acls))
    for perms, scheme, ident in acls:
        w.int32(perms).ustring(scheme).ustring(ident)


def read_acls(r: Reader):
    n = r.int32()
    out = []
    for _ in range(max(n, 0)):
        out.append((r.int32(), r.ustring(), r.ustring()))
    return out


# ---- connect handshake -------------------------------------------------

def encode_connect_request(last_zxid: int, timeout_ms: int, session_id: int,
                           passwd: bytes) -> bytes:
    w = Writer()
    w.int32(0).int64(last_zxid).int32(timeout_ms).int64(session_id)
    w.buffer(passwd)
    return w.framed()


def decode_connect_request(body: bytes) -> Tuple[int, int, int, bytes]:
    r = Reader(body)
    r.int32()  # protocolVersion
    last_zxid = r.int64()
    timeout_ms = r.int32()
    session_id = r.int64()
    passwd = r.buffer() or b""
    return last_zxid, timeout_ms, session_id, passwd


def encode_connect_response(timeout_ms: int, session_id: int,
                            passwd: bytes) -> bytes:
    w = Writer()
    w.int32(0).int32(timeout_ms).int64(session_id).buffer(passwd)
    return w.framed()


def decode_connect_response(body: bytes) -> Tuple[int, int, bytes]:
    r = Reader(body)
    r.int32()
    timeout_ms = r.int32()
    session_id = r.int64()
    passwd = r.buffer() or b""
    return timeout_ms, session_id, passwd


# ---- request/reply headers --------------------------------------------

def encode_request_header(xid: int, opcode: int) -> Writer:
    w = Writer()
    w.int32(xid).int32(opcode)
    return w


def decode_request_header(r: Reader) -> Tuple[int, int]:
    return r.int32(), r.int32()


def encode_reply_header(xid: int, zxid: int, err: int) -> Writer:
    w = Writer()
    w.int32(xid).int64(zxid).int32(err)
    return w


def decode_reply_header(r: Reader) -> Tuple[int, int, int]:
    return r.int32(), r.int64(), r.int32()


# ---- watcher event -----------------------------------------------------

def encode_watcher_event(etype: int, state: int, path: str) -> bytes:
    w = encode_reply_header(XID_NOTIFICATION, 0, ZOK)
    w.int32(etype).int32(state).ustring(path)
    return w.framed()


def decode_watcher_event(r: Reader) -> Tuple[int, int, str]:
    return r.int32(), r.int32(), r.ustring() or ""


# ---- multi-op framing --------------------------------------------------

class MultiOp:
    """One operation inside a multi (transaction).  kind in
    {'create','delete','setData','check'}."""

    __slots__ = ("kind", "path", "data", "flags", "version")

    def __init__(self, kind: str, path: str, data: Optional[bytes] = None,
                 flags: int = PERSISTENT, version: int = -1):
        self.kind = kind
        self.path = path
        self.data = data
        self.flags = flags
        self.version = version

    @classmethod
    def create(cls, path: str, data: bytes, flags: int = PERSISTENT):
        return cls("create", path, data=data, flags=flags)

    @classmethod
    def set_data(cls, path: str, data: bytes, version: int = -1):
        return cls("setData", path, data=data, version=version)

    @classmethod
    def delete(cls, path: str, version: int = -1):
        return cls("delete", path, version=version)

    @classmethod
    def check(cls, path: str, version: int = -1):
        return cls("check", path, version=version)


_MULTI_KIND_TO_OP = {"create": OP_CREATE, "delete": OP_DELETE,
                     "setData": OP_SETDATA, "check": OP_CHECK}
_MULTI_OP_TO_KIND = {v: k for k, v in _MULTI_KIND_TO_OP.items()}


def write_multi_request(w: Writer, ops: List[MultiOp]) -> None:
    for op in ops:
        w.int32(_MULTI_KIND_TO_OP[op.kind]).boolean(False).int32(-1)
        if op.kind == "create":
            w.ustring(op.path).buffer(op.data)
            write_acls(w)
            w.int32(op.flags)
        elif op.kind == "setData":
            w.ustring(op.path).buffer(op.data).int32(op.version)
        else:  # delete / check
            w.ustring(op.path).int32(op.version)
    w.int32(-1).boolean(True).int32(-1)


def read_multi_request(r: Reader) -> List[MultiOp]:
    ops: List[MultiOp] = []
    while True:
        optype = r.int32()
        done = r.boolean()
        r.int32()  # err, unused on request
        if done:
            break
        kind = _MULTI_OP_TO_KIND.get(optype)
        if kind is None:
            raise ZkError(ZMARSHALLINGERROR)
        if kind == "create":
            path = r.ustring() or ""
            data = r.buffer()
            read_acls(r)
            flags = r.int32()
            ops.append(MultiOp.create(path, data or b"", flags))
        elif kind == "setData":
            path = r.ustring() or ""
            data = r.buffer()
            version = r.int32()
            ops.append(MultiOp.set_data(path, data or b"", version))
        else:
            path = r.ustring() or ""
            version = r.int32()
            ops.append(MultiOp(kind, path, version=version))
    return ops


def write_multi_response(w: Writer, results: List[tuple]) -> None:
    """results: list of ('create', path) | ('setData', Stat) |
    ('delete',) | ('check',) | ('error', errcode)."""
    for res in results:
        kind = res[0]
        if kind == "error":
            w.int32(OP_ERROR).boolean(False).int32(res[1])
            w.int32(res[1])
        else:
            w.int32(_MULTI_KIND_TO_OP[kind]).boolean(False).int32(ZOK)
            if kind == "create":
                w.ustring(res[1])
            elif kind == "setData":
                res[1].write(w)
    w.int32(-1).boolean(True).int32(-1)


def read_multi_response(r: Reader) -> List[tuple]:
    out: List[tuple] = []
    while True:
        optype = r.int32()
        done = r.boolean()
        r.int32()
        if done:
            break
        if optype == OP_ERROR:
            out.append(("error", r.int32()))
        elif optype == OP_CREATE:
            out.append(("create", r.ustring() or ""))
        elif optype == OP_SETDATA:
            out.append(("setData", Stat.read(r)))
        elif optype == OP_DELETE:
            out.append(("delete",))
        elif optype == OP_CHECK:
            out.append(("check",))
        else:
            raise ZkError(ZMARSHALLINGERROR)
    return out
