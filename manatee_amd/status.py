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
        self.http.route("GET", "metrics", self._metrics)
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

    async def _metrics(self, parts, body):
        """Prometheus text exposition — observability the reference
        lacks (SURVEY.md §5.5 'no Prometheus-style metrics endpoint')."""
        import time as _time
        from .common import lsn as lsnmod
        lines = []

        def gauge(name, value, help_text="", labels=""):
            if help_text:
                lines.append("# HELP %s %s" % (name, help_text))
                lines.append("# TYPE %s gauge" % name)
            lines.append("%s%s %s" % (name, labels, value))

        db = self.shard.db_manager
        peer = self.shard.peer
        gauge("manatee_db_healthy",
              int(bool(db is not None and db.healthy)),
              "database answered its last health check")
        gauge("manatee_db_online",
              int(bool(db is not None and db.online)),
              "database child process is running")
        gauge("manatee_db_writable",
              int(bool(db is not None and db.writable)),
              "peer accepts writes (primary with caught-up sync)")
        if peer is not None:
            ds = peer.debug_state()
            role_map = {"primary": 0, "sync": 1, "async": 2,
                        "deposed": 3, "none": 4}
            gauge("manatee_role",
                  role_map.get(ds.get("role") or "none", 4),
                  "0=primary 1=sync 2=async 3=deposed 4=none")
            state = ds.get("clusterState") or {}
            if state.get("generation") is not None:
                gauge("manatee_generation", state["generation"],
                      "cluster state generation")
            gauge("manatee_cluster_frozen",
                  int(bool(state.get("freeze"))),
                  "cluster transitions frozen")
        if db is not None and db.online:
            try:
                st = await db.engine.status()
                cur = st.get("current_lsn")
                if cur:
                    gauge("manatee_wal_lsn_bytes", lsnmod.parse(cur),
                          "current WAL position in bytes")
                lrt = st.get("last_replay_time")
                if lrt and st.get("role") == "standby":
                    gauge("manatee_replay_lag_seconds",
                          round(max(0.0, _time.time() - float(lrt)), 3),
                          "seconds since last applied record")
                for row in st.get("replication") or []:
                    labels = '{downstream="%s",sync_state="%s"}' % (
                        row.get("application_name"),
                        row.get("sync_state"))
                    sent = row.get("sent_lsn")
                    flush = row.get("flush_lsn")
                    if sent and flush:
                        gauge("manatee_replication_unflushed_bytes",
                              lsnmod.parse(sent) - lsnmod.parse(flush),
                              "bytes sent but not yet flushed downstream",
                              labels)
            except Exception:
                pass
        if db is not None and db.restore_client is not None:
            ro = db.restore_client.restore_object
            gauge("manatee_restore_active", int(ro.active),
                  "a snapshot restore is in progress")
            gauge("manatee_restore_completed_bytes", ro.completed)
        return 200, "\n".join(lines) + "\n"

    async def _index(self, parts, body):
        return 200, {"routes": ["GET /ping", "GET /state", "GET /restore",
                                "GET /metrics", "GET /"]}
