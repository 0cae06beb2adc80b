"""PgKvClient — the WaldbClient API over the libpq wire protocol.

Write-load drivers (bench.py, soak, the integ tests) speak to whichever
engine a peer runs through one key/value client interface.  For
``engine=postgres`` (minipg) that means real SQL over libpq: INSERTs
acknowledged under synchronous_commit=remote_write, reads on standbys,
server-side counts — the pgbench-writes analogue of BASELINE.json's
headline config.

Values are JSON-encoded into the ``v`` column so callers get their
types back, exactly as with WaldbClient.
"""

from __future__ import annotations

import json
from typing import Optional

from .pgwire import PgClient, PgError


class PgKvError(RuntimeError):
    pass


def _q(text: str) -> str:
    """SQL single-quote escape."""
    return text.replace("'", "''")


class PgKvClient:
    def __init__(self, host: str, port: int, user: str = "postgres",
                 connect_timeout_s: float = 5.0,
                 query_timeout_s: float = 30.0):
        self.host = host
        self.port = port
        self.query_timeout_s = query_timeout_s
        self._cli = PgClient(host, port, user,
                             connect_timeout_s=connect_timeout_s)

    async def _query(self, sql: str, timeout_s: Optional[float] = None):
        try:
            if not self._cli.connected:
                await self._cli.connect()
            return await self._cli.query(
                sql, timeout_s=timeout_s if timeout_s is not None
                else self.query_timeout_s)
        except PgError as exc:
            # connection state is fine after a server-reported error;
            # anything transport-level invalidates the connection
            raise PgKvError(str(exc)) from exc
        except Exception:
            await self.close()
            raise

    # ------------------------------------------------------------- the API
    async def ping(self, timeout_s: float = 5.0) -> bool:
        try:
            await self._query("SELECT current_time;", timeout_s=timeout_s)
            return True
        except Exception:
            await self.close()
            return False

    async def put(self, key: str, value, timeout_s: Optional[float] = None
                  ) -> str:
        res = await self._query(
            "INSERT INTO kv (k, v) VALUES ('%s', '%s')"
            % (_q(key), _q(json.dumps(value))), timeout_s=timeout_s)
        if not res.command.startswith("INSERT"):
            raise PgKvError("unexpected result %r" % res.command)
        return res.command

    async def put_many(self, items, timeout_s: Optional[float] = None
                       ) -> int:
        """Multi-statement simple query: one round trip, N INSERTs (the
        libpq batch form); the server aborts the rest on any error."""
        items = list(items)
        sql = ";".join(
            "INSERT INTO kv (k, v) VALUES ('%s', '%s')"
            % (_q(k), _q(json.dumps(v))) for k, v in items)
        res = await self._query(sql, timeout_s=timeout_s)
        if not res.command.startswith("INSERT"):
            raise PgKvError("batch aborted at %r" % res.command)
        return len(items)

    async def get(self, key: str, timeout_s: Optional[float] = None):
        res = await self._query(
            "SELECT v FROM kv WHERE k = '%s'" % _q(key),
            timeout_s=timeout_s)
        if not res.rows:
            return None
        return json.loads(res.rows[0][0])

    async def delete(self, key: str,
                     timeout_s: Optional[float] = None) -> str:
        res = await self._query("DELETE FROM kv WHERE k = '%s'" % _q(key),
                                timeout_s=timeout_s)
        return res.command

    async def count(self, prefix: Optional[str] = None,
                    timeout_s: Optional[float] = None) -> int:
        if prefix:
            sql = "SELECT count(*) AS n FROM kv WHERE k LIKE '%s%%'" \
                % _q(prefix)
        else:
            sql = "SELECT count(*) AS n FROM kv"
        res = await self._query(sql, timeout_s=timeout_s)
        return int(res.rows[0][0])

    async def status(self) -> dict:
        in_rec = (await self._query(
            "SELECT pg_is_in_recovery() as r;")).rows[0][0] in ("t", "true")
        repl = []
        res = await self._query("SELECT * FROM pg_stat_replication;")
        for row in res.dicts():
            repl.append({
                "application_name": row.get("application_name"),
                "state": row.get("state"),
                "sync_state": row.get("sync_state"),
                "sent_lsn": row.get("sent_lsn") or row.get("sent_location"),
                "write_lsn": (row.get("write_lsn")
                              or row.get("write_location")),
                "flush_lsn": (row.get("flush_lsn")
                              or row.get("flush_location")),
                "replay_lsn": (row.get("replay_lsn")
                               or row.get("replay_location")),
            })
        return {"ok": True,
                "role": "standby" if in_rec else "primary",
                "replication": repl}

    async def xlog(self) -> str:
        in_rec = (await self._query(
            "SELECT pg_is_in_recovery() as r;")).rows[0][0] in ("t", "true")
        sql = ("SELECT pg_last_wal_replay_lsn() as loc;" if in_rec
               else "SELECT pg_current_wal_lsn() as loc;")
        return (await self._query(sql)).rows[0][0]

    async def close(self) -> None:
        try:
            await self._cli.close()
        except Exception:
            pass
