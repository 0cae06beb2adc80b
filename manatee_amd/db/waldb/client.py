"""waldb client — serialized single-connection query client.

The reference works around node-postgres connection-state bugs with a
serialized single-connection query queue (ref lib/postgresMgr.js:1990-2172);
we keep the same discipline: one TCP connection, one outstanding query.
"""

from __future__ import annotations

import asyncio
import json
from typing import Optional

from ...common import dial


class WaldbError(RuntimeError):
    pass


class WaldbClient:
    def __init__(self, host: str, port: int, connect_timeout_s: float = 5.0,
                 query_timeout_s: float = 30.0):
        self.host = host
        self.port = port
        self.connect_timeout_s = connect_timeout_s
        self.query_timeout_s = query_timeout_s
        self._reader: Optional[asyncio.StreamReader] = None
        self._writer: Optional[asyncio.StreamWriter] = None
        self._rbuf = b""
        self._lock = asyncio.Lock()

    @classmethod
    def from_url(cls, url: str, **kw) -> "WaldbClient":
        """Accepts 'tcp://user@host:port/db' (pgUrl shape) or 'host:port'."""
        text = url
        if "://" in text:
            text = text.split("://", 1)[1]
        if "@" in text:
            text = text.split("@", 1)[1]
        text = text.split("/", 1)[0]
        host, _, port = text.partition(":")
        return cls(host, int(port), **kw)

    async def _ensure(self) -> None:
        if self._writer is None or self._writer.is_closing():
            self._reader, self._writer = await asyncio.wait_for(
                dial.open_connection(self.host, self.port),
                self.connect_timeout_s)
            self._rbuf = b""

    async def _readline(self, timeout_s: float) -> bytes:
        """Chunk-buffered line read: one read() syscall can serve a
        whole pipelined response batch.  Returns b"" on EOF."""
        buf = self._rbuf
        nl = buf.find(b"\n")
        while nl < 0:
            chunk = await asyncio.wait_for(self._reader.read(65536),
                                           timeout_s)
            if not chunk:
                self._rbuf = b""
                return b""
            scan_from = len(buf)        # no newline before the chunk
            buf += chunk
            nl = buf.find(b"\n", scan_from)
        line, self._rbuf = buf[:nl + 1], buf[nl + 1:]
        return line

    async def query(self, req: dict, timeout_s: Optional[float] = None
                    ) -> dict:
        async with self._lock:
            await self._ensure()
            try:
                self._writer.write((json.dumps(req) + "\n").encode())
                await self._writer.drain()
                line = await self._readline(
                    timeout_s if timeout_s is not None
                    else self.query_timeout_s)
            except (ConnectionError, OSError, asyncio.TimeoutError) as exc:
                await self._teardown()
                raise WaldbError("query failed: %r" % (exc,)) from exc
            if not line:
                await self._teardown()
                raise WaldbError("connection closed by server")
            try:
                return json.loads(line)
            except ValueError as exc:
                await self._teardown()
                raise WaldbError("bad response") from exc

    async def _teardown(self) -> None:
        if self._writer is not None:
            try:
                self._writer.close()
            except Exception:
                pass
        self._writer = None
        self._reader = None
        self._rbuf = b""

    async def close(self) -> None:
        await self._teardown()

    # ------------------------------------------------------------- helpers
    async def ping(self, timeout_s: float = 5.0) -> bool:
        resp = await self.query({"q": "ping"}, timeout_s=timeout_s)
        return bool(resp.get("ok"))

    async def put(self, key: str, value, timeout_s: Optional[float] = None
                  ) -> str:
        resp = await self.query({"q": "put", "k": key, "v": value},
                                timeout_s=timeout_s)
        if not resp.get("ok"):
            raise WaldbError(resp.get("error", "put failed"))
        return resp["lsn"]

    async def get(self, key: str):
        resp = await self.query({"q": "get", "k": key})
        if not resp.get("ok"):
            raise WaldbError(resp.get("error", "get failed"))
        return resp.get("v") if resp.get("found") else None

    async def pipeline(self, reqs, timeout_s: Optional[float] = None
                       ) -> list:
        """Send many requests back-to-back and read the responses in
        order — removes the per-request round trip (the wire protocol
        answers strictly in order)."""
        reqs = list(reqs)
        async with self._lock:
            await self._ensure()
            try:
                payload = b"".join(
                    json.dumps(r).encode() + b"\n" for r in reqs)
                self._writer.write(payload)
                await self._writer.drain()
                t = (timeout_s if timeout_s is not None
                     else self.query_timeout_s)
                out = []
                for _ in reqs:
                    line = await self._readline(t)
                    if not line:
                        raise WaldbError("connection closed by server")
                    out.append(json.loads(line.decode("utf-8")))
                return out
            except (ConnectionError, OSError, asyncio.TimeoutError,
                    ValueError) as exc:
                await self._teardown()
                raise WaldbError("pipeline failed: %r" % (exc,)) from exc

    async def put_many(self, items, timeout_s: Optional[float] = None
                       ) -> int:
        """Pipelined puts; items = iterable of (key, value).  Returns the
        number acknowledged; raises on the first failed put."""
        resps = await self.pipeline(
            ({"q": "put", "k": k, "v": v} for k, v in items),
            timeout_s=timeout_s)
        for r in resps:
            if not r.get("ok"):
                raise WaldbError(r.get("error", "put failed"))
        return len(resps)

    async def delete(self, key: str, timeout_s: Optional[float] = None
                     ) -> str:
        resp = await self.query({"q": "del", "k": key},
                                timeout_s=timeout_s)
        if not resp.get("ok"):
            raise WaldbError(resp.get("error", "del failed"))
        return resp["lsn"]

    async def batch(self, ops, timeout_s: Optional[float] = None) -> str:
        """Atomic multi-op commit: ops = [{"op": "put"|"del", "k": ...,
        "v": ...}, ...] — one WAL record, all-or-nothing."""
        resp = await self.query({"q": "batch", "ops": list(ops)},
                                timeout_s=timeout_s)
        if not resp.get("ok"):
            raise WaldbError(resp.get("error", "batch failed"))
        return resp["lsn"]

    async def scan(self, prefix: str = "", limit: int = 1000,
                   after: Optional[str] = None) -> list:
        resp = await self.query({"q": "scan", "prefix": prefix,
                                 "limit": limit, "after": after})
        if not resp.get("ok"):
            raise WaldbError(resp.get("error", "scan failed"))
        return resp["items"]

    async def count(self, prefix: Optional[str] = None,
                    timeout_s: Optional[float] = None) -> int:
        resp = await self.query({"q": "count", "prefix": prefix},
                                timeout_s=timeout_s)
        if not resp.get("ok"):
            raise WaldbError(resp.get("error", "count failed"))
        return int(resp["count"])

    async def status(self) -> dict:
        resp = await self.query({"q": "status"})
        if not resp.get("ok"):
            raise WaldbError(resp.get("error", "status failed"))
        return resp

    async def xlog(self) -> str:
        resp = await self.query({"q": "xlog"})
        if not resp.get("ok"):
            raise WaldbError(resp.get("error", "xlog failed"))
        return resp["lsn"]
