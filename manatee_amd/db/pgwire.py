"""Minimal PostgreSQL v3 wire-protocol client (asyncio).

The admin/health paths need exactly what the reference uses the ``pg``
module for (ref lib/postgresMgr.js:1990-2172, lib/adm.js:348-427): a
single connection running simple queries (``SELECT current_time``,
``pg_stat_replication``, LSN functions) serially.  This speaks the v3
protocol directly — StartupMessage, trust/ident auth (no password
flows), simple Query, row decoding as text — with a serialized query
queue like the reference's (one query in flight per connection,
working around node-postgres #718).
"""

from __future__ import annotations

import asyncio
import struct
from typing import Dict, List, Optional, Tuple

from ..common import dial

_I32 = struct.Struct(">i")
_HDR = struct.Struct(">cI")     # type byte + length (len includes itself)

PROTOCOL_VERSION = 196608       # 3.0


class PgError(RuntimeError):
    def __init__(self, fields: Dict[str, str]):
        self.fields = fields
        super().__init__(fields.get("M", "postgres error"))

    @property
    def code(self) -> str:
        return self.fields.get("C", "")


class PgResult:
    def __init__(self, columns: List[str], rows: List[Tuple],
                 command: str = ""):
        self.columns = columns
        self.rows = rows
        self.command = command

    def dicts(self) -> List[dict]:
        return [dict(zip(self.columns, r)) for r in self.rows]


def _cstr(b: bytes, pos: int) -> Tuple[str, int]:
    end = b.index(b"\x00", pos)
    return b[pos:end].decode("utf-8"), end + 1


class PgClient:
    """One serialized connection (queries run one at a time)."""

    def __init__(self, host: str, port: int, user: str,
                 database: str = "postgres",
                 connect_timeout_s: float = 10.0):
        self.host = host
        self.port = port
        self.user = user
        self.database = database
        self.connect_timeout_s = connect_timeout_s
        self._reader: Optional[asyncio.StreamReader] = None
        self._writer: Optional[asyncio.StreamWriter] = None
        self._lock = asyncio.Lock()
        self.parameters: Dict[str, str] = {}

    # ------------------------------------------------------------ lifecycle
    async def connect(self) -> None:
        """TCP connect + startup/auth handshake, ALL bounded by
        connect_timeout_s: a backend that accepts the connection but
        never answers the startup packet (dying process, half-open
        socket after a partition) must not hang the caller forever."""
        try:
            await asyncio.wait_for(self._connect(),
                                   self.connect_timeout_s)
        except BaseException:
            writer = self._writer
            self._writer = None
            self._reader = None
            if writer is not None:
                try:
                    writer.close()
                except Exception:
                    pass
            raise

    async def _connect(self) -> None:
        self._reader, self._writer = await dial.open_connection(
            self.host, self.port)
        params = ("user\x00%s\x00database\x00%s\x00\x00"
                  % (self.user, self.database)).encode("utf-8")
        body = _I32.pack(PROTOCOL_VERSION) + params
        self._writer.write(_I32.pack(len(body) + 4) + body)
        await self._writer.drain()
        # authentication + parameter flow until ReadyForQuery
        while True:
            t, payload = await self._recv()
            if t == b"R":
                (auth,) = _I32.unpack_from(payload)
                if auth != 0:
                    raise PgError({"M": "unsupported auth method %d (only "
                                        "trust/ident)" % auth})
            elif t == b"S":
                key, pos = _cstr(payload, 0)
                val, _ = _cstr(payload, pos)
                self.parameters[key] = val
            elif t == b"K":      # BackendKeyData
                pass
            elif t == b"Z":      # ReadyForQuery
                return
            elif t == b"E":
                raise PgError(self._err_fields(payload))

    async def close(self) -> None:
        if self._writer is not None:
            try:
                self._writer.write(b"X" + _I32.pack(4))  # Terminate
                await self._writer.drain()
            except (ConnectionError, OSError):
                pass
            self._writer.close()
            self._writer = None
            self._reader = None

    @property
    def connected(self) -> bool:
        return self._writer is not None

    # --------------------------------------------------------------- query
    async def query(self, sql: str,
                    timeout_s: float = 30.0) -> PgResult:
        """Simple-protocol query; returns the LAST result set.  Any
        transport-level failure (EOF from a dead backend, timeout,
        reset) tears the connection down so the next use reconnects —
        a stale cached connection must never masquerade as a live one."""
        async with self._lock:   # serialized, ref :1990-2172
            try:
                return await asyncio.wait_for(self._query(sql), timeout_s)
            except (asyncio.IncompleteReadError, EOFError,
                    ConnectionError, OSError, asyncio.TimeoutError):
                writer = self._writer
                self._writer = None
                self._reader = None
                if writer is not None:
                    try:
                        writer.close()
                    except Exception:
                        pass
                raise

    async def _query(self, sql: str) -> PgResult:
        if self._writer is None:
            raise PgError({"M": "not connected"})
        body = sql.encode("utf-8") + b"\x00"
        self._writer.write(b"Q" + _I32.pack(len(body) + 4) + body)
        await self._writer.drain()
        columns: List[str] = []
        rows: List[Tuple] = []
        command = ""
        error: Optional[PgError] = None
        while True:
            t, payload = await self._recv()
            if t == b"T":        # RowDescription
                (n,) = struct.unpack_from(">h", payload)
                pos = 2
                columns = []
                for _ in range(n):
                    name, pos = _cstr(payload, pos)
                    pos += 18    # table oid, attnum, type oid, len, mod, fmt
                    columns.append(name)
                rows = []
            elif t == b"D":      # DataRow
                (n,) = struct.unpack_from(">h", payload)
                pos = 2
                row = []
                for _ in range(n):
                    (ln,) = struct.unpack_from(">i", payload, pos)
                    pos += 4
                    if ln == -1:
                        row.append(None)
                    else:
                        row.append(payload[pos:pos + ln].decode("utf-8"))
                        pos += ln
                rows.append(tuple(row))
            elif t == b"C":      # CommandComplete
                command, _ = _cstr(payload, 0)
            elif t == b"E":
                error = PgError(self._err_fields(payload))
            elif t == b"Z":      # ReadyForQuery — end of cycle
                if error is not None:
                    raise error
                return PgResult(columns, rows, command)
            # 'N' (notice), 'S' (parameter), 'I' (empty) are ignored

    # ------------------------------------------------------------- framing
    async def _recv(self) -> Tuple[bytes, bytes]:
        hdr = await self._reader.readexactly(5)
        t, ln = _HDR.unpack(hdr)
        payload = await self._reader.readexactly(ln - 4)
        return t, payload

    @staticmethod
    def _err_fields(payload: bytes) -> Dict[str, str]:
        fields = {}
        pos = 0
        while pos < len(payload) and payload[pos:pos + 1] != b"\x00":
            code = payload[pos:pos + 1].decode()
            val, pos = _cstr(payload, pos + 1)
            fields[code] = val
        return fields
