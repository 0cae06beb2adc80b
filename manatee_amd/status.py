"""StatusServer — per-sitter HTTP observability.

Ref lib/statusServer.js: listens on ``postgresPort + 1`` (sitter.js:122-126)
and serves:

- ``GET /ping``     → 200/503 from the database manager's health (:90-103)
- ``GET /state``    → the FSM's debugState() dump (:106-109)
- ``GET /restore``  → restore progress object (:112-121)
- ``GET /``         → route list (:77-87)
"""

from __future__ import annotations

from typing import Optional

from .common.httpd import HttpServer
from .common.logging import Logger, null_logger


class StatusServer:
    def __init__(self, host: str, port: int, shard,
                 log: Optional[Logger] = None):
        """shard: object exposing .db_manager (health/status, restore
        object) and .peer (debug_state())."""
        self.shard = shard
        self.log = (log or null_logger()).child(component="StatusServer")
        self.http = HttpServer(host, port, log=self.log)
        self.http.route("GET", "ping", self._ping)
        self.http.route("GET", "state", self._state)
        self.http.route("GET", "restore", self._restore)
        self.http.route("GET", "", self._index)

    @property
    def port(self) -> int:
        return self.http.port

    async def start(self) -> None:
        await self.http.start()

    async def stop(self) -> None:
        await self.http.stop()

    async def _ping(self, parts, body):
        db = self.shard.db_manager
        healthy = bool(db is not None and db.healthy)
        status = db.status() if db is not None else {}
        return (200 if healthy else 503), {"healthy": healthy,
                                           "status": status}

    async def _state(self, parts, body):
        peer = self.shard.peer
        if peer is None:
            return 503, {"error": "state machine not running"}
        return 200, peer.debug_state()

    async def _restore(self, parts, body):
        db = self.shard.db_manager
        if db is None or db.restore_client is None:
            return 200, {"active": False, "done": False}
        return 200, db.restore_client.restore_object.as_dict()

    async def _index(self, parts, body):
        return 200, {"routes": ["GET /ping", "GET /state", "GET /restore",
                                "GET /"]}
