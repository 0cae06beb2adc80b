"""Minimal asyncio HTTP/1.1 server + client.

The reference uses restify servers for the backup and status services
(lib/backupServer.js:100-101, lib/statusServer.js:67-70) and restify JSON
clients for polling them (lib/zfsClient.js:638-754).  This image has no need
for a framework: the API surface is four small JSON routes, implemented here
directly over asyncio streams.
"""

from __future__ import annotations

import asyncio
import json
from typing import Awaitable, Callable, Dict, Optional, Tuple

from . import dial

from .logging import Logger, null_logger

Handler = Callable[..., Awaitable[Tuple[int, object]]]

_REASONS = {200: "OK", 201: "Created", 204: "No Content",
            400: "Bad Request", 404: "Not Found", 409: "Conflict",
            500: "Internal Server Error", 503: "Service Unavailable"}


class HttpServer:
    """Route table keyed by (METHOD, first path segment); handlers get
    (path_parts, body_json) and return (status, json_body|str)."""

    def __init__(self, host: str, port: int, log: Optional[Logger] = None):
        self.host = host
        self.port = port
        self.log = (log or null_logger()).child(component="http")
        self._routes: Dict[Tuple[str, str], Handler] = {}
        self._server: Optional[asyncio.AbstractServer] = None

    def route(self, method: str, segment: str, handler: Handler) -> None:
        self._routes[(method.upper(), segment)] = handler

    async def start(self) -> None:
        self._server = await asyncio.start_server(self._handle,
                                                  self.host, self.port)
        self.port = self._server.sockets[0].getsockname()[1]
        self.log.info("http server listening", host=self.host,
                      port=self.port)

    async def stop(self) -> None:
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()

    def routes(self):
        return sorted("%s /%s" % (m, s) for (m, s) in self._routes)

    async def _handle(self, reader: asyncio.StreamReader,
                      writer: asyncio.StreamWriter) -> None:
        try:
            while True:
                line = await asyncio.wait_for(reader.readline(), 120)
                if not line:
                    return
                try:
                    method, target, _version = \
                        line.decode("latin-1").strip().split(" ", 2)
                except ValueError:
                    return
                headers = {}
                while True:
                    hline = await reader.readline()
                    if hline in (b"\r\n", b"\n", b""):
                        break
                    name, _, value = hline.decode("latin-1").partition(":")
                    headers[name.strip().lower()] = value.strip()
                body = b""
                clen = int(headers.get("content-length", "0") or "0")
                if clen:
                    body = await reader.readexactly(clen)
                status, payload = await self._dispatch(method, target, body)
                if isinstance(payload, (dict, list)):
                    data = json.dumps(payload).encode()
                    ctype = "application/json"
                else:
                    data = str(payload).encode()
                    ctype = "text/plain"
                writer.write(
                    ("HTTP/1.1 %d %s\r\ncontent-type: %s\r\n"
                     "content-length: %d\r\nconnection: keep-alive\r\n\r\n"
                     % (status, _REASONS.get(status, "X"), ctype, len(data))
                     ).encode() + data)
                await writer.drain()
        except (ConnectionError, asyncio.IncompleteReadError,
                asyncio.TimeoutError, asyncio.CancelledError):
            pass
        except Exception as exc:
            self.log.error("http handler error", err=exc)
        finally:
            try:
                writer.close()
            except Exception:
                pass

    async def _dispatch(self, method: str, target: str,
                        body: bytes) -> Tuple[int, object]:
        path = target.split("?", 1)[0]
        parts = [p for p in path.split("/") if p]
        segment = parts[0] if parts else ""
        handler = self._routes.get((method.upper(), segment))
        if handler is None:
            return 404, {"error": "no such route", "routes": self.routes()}
        body_json = None
        if body:
            try:
                body_json = json.loads(body)
            except ValueError:
                return 400, {"error": "invalid JSON body"}
        try:
            return await handler(parts, body_json)
        except Exception as exc:
            self.log.error("route handler failed", route=path, err=exc)
            return 500, {"error": repr(exc)}


async def http_request(url: str, method: str = "GET", body: object = None,
                       timeout_s: float = 30.0) -> Tuple[int, object]:
    """Tiny JSON-over-HTTP client for http://host:port/path URLs."""
    assert url.startswith("http://"), url
    rest = url[len("http://"):]
    hostport, _, path = rest.partition("/")
    host, _, port = hostport.partition(":")
    path = "/" + path
    data = b""
    if body is not None:
        data = json.dumps(body).encode()
    req = ("%s %s HTTP/1.1\r\nhost: %s\r\ncontent-type: application/json\r\n"
           "content-length: %d\r\nconnection: close\r\n\r\n"
           % (method, path, hostport, len(data))).encode() + data
    reader, writer = await asyncio.wait_for(
        dial.open_connection(host, int(port or 80)), timeout_s)
    try:
        writer.write(req)
        await writer.drain()
        status_line = await asyncio.wait_for(reader.readline(), timeout_s)
        parts = status_line.decode("latin-1").split(" ", 2)
        status = int(parts[1])
        headers = {}
        while True:
            hline = await asyncio.wait_for(reader.readline(), timeout_s)
            if hline in (b"\r\n", b"\n", b""):
                break
            name, _, value = hline.decode("latin-1").partition(":")
            headers[name.strip().lower()] = value.strip()
        clen = headers.get("content-length")
        if clen is not None:
            payload = await asyncio.wait_for(
                reader.readexactly(int(clen)), timeout_s)
        else:
            payload = await asyncio.wait_for(reader.read(), timeout_s)
        try:
            return status, json.loads(payload) if payload else None
        except ValueError:
            return status, payload.decode("utf-8", "replace")
    finally:
        writer.close()
