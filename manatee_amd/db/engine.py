"""Database engine interface + the built-in waldb engine.

The manager layer (``manager.py``, ref lib/postgresMgr.js) drives a database
through this interface: conf generation, process argv, readiness, health,
WAL positions and replication status.  Two engines implement it:

- ``WaldbEngine``    — the built-in replicated KV engine (hermetic);
- ``PostgresEngine`` — real PostgreSQL (``postgres.py``), version-aware
  (9.x recovery.conf vs ≥12 standby.signal) like the reference.
"""

from __future__ import annotations

import os
import sys
from typing import List, Optional

from ..common import confparser
from ..common.logging import Logger, null_logger
from .waldb import server as waldb_server
from .waldb.client import WaldbClient, WaldbError


def url_to_hostport(url: str) -> tuple:
    """'tcp://postgres@10.0.0.1:5432/postgres' → ('10.0.0.1', 5432)."""
    text = url
    if "://" in text:
        text = text.split("://", 1)[1]
    if "@" in text:
        text = text.split("@", 1)[1]
    text = text.split("/", 1)[0]
    host, _, port = text.partition(":")
    return host, int(port)


def peer_id_from_urls(pg_url: str, backup_url: str) -> str:
    """Reconstruct the peer id 'ip:pgPort:backupPort' from its URLs
    (identity format, ref lib/shard.js:39-41)."""
    host, pg_port = url_to_hostport(pg_url)
    b = backup_url
    if "://" in b:
        b = b.split("://", 1)[1]
    b = b.split("/", 1)[0]
    _, _, backup_port = b.partition(":")
    return "%s:%d:%s" % (host, pg_port, backup_port or "0")


class Engine:
    name = "abstract"

    def initialized(self) -> bool:
        raise NotImplementedError

    async def init_datadir(self) -> None:
        raise NotImplementedError

    def current_conf_role(self) -> Optional[str]:
        raise NotImplementedError

    def write_conf(self, role: str, upstream_url: Optional[str] = None,
                   sync_name: Optional[str] = None,
                   read_only: bool = False) -> None:
        raise NotImplementedError

    def write_promote_trigger(self) -> None:
        raise NotImplementedError

    def post_restore_fixup(self) -> None:
        """Clean engine-managed runtime files after a dataset restore."""
        raise NotImplementedError

    def spawn_argv(self) -> List[str]:
        raise NotImplementedError

    async def ping(self, timeout_s: float = 5.0) -> bool:
        raise NotImplementedError

    async def xlog(self) -> str:
        raise NotImplementedError

    async def status(self) -> dict:
        raise NotImplementedError

    async def check_repl(self, standby_name: str) -> dict:
        """One repl-status probe for the named downstream
        (ref _checkReplStatus lib/postgresMgr.js:2478-2556).
        Returns {connected, sync_state, sent_lsn, write_lsn, flush_lsn,
        replay_lsn, caught_up}."""
        raise NotImplementedError

    async def close(self) -> None:
        pass


class WaldbEngine(Engine):
    name = "waldb"

    def __init__(self, data_dir: str, ip: str, port: int, peer_name: str,
                 log: Optional[Logger] = None):
        self.data_dir = data_dir
        self.ip = ip
        self.port = port
        self.peer_name = peer_name
        self.log = (log or null_logger()).child(component="WaldbEngine")
        self._client: Optional[WaldbClient] = None

    # --------------------------------------------------------------- setup
    def initialized(self) -> bool:
        return os.path.exists(os.path.join(self.data_dir,
                                           waldb_server.IDENT_NAME))

    async def init_datadir(self) -> None:
        waldb_server.init_data_dir(self.data_dir)
        self.log.info("data directory initialized", dataDir=self.data_dir)

    def _conf_path(self) -> str:
        return os.path.join(self.data_dir, waldb_server.CONF_NAME)

    def current_conf_role(self) -> Optional[str]:
        try:
            conf = confparser.read(self._conf_path())
        except FileNotFoundError:
            return None
        return (conf.get("role") or "").strip("'\"") or None

    def write_conf(self, role: str, upstream_url: Optional[str] = None,
                   sync_name: Optional[str] = None,
                   read_only: bool = False) -> None:
        # always regenerated from scratch, like the reference's conf
        # handling (ref lib/postgresMgr.js:2282-2336)
        conf = {
            "role": "primary" if role == "primary" else "standby",
            "listen_ip": self.ip,
            "port": str(self.port),
            "name": self.peer_name,
            "default_transaction_read_only": "on" if read_only else "off",
        }
        if upstream_url:
            host, port = url_to_hostport(upstream_url)
            conf["primary_conninfo"] = "'%s:%d'" % (host, port)
        if sync_name:
            conf["synchronous_standby_names"] = "'%s'" % sync_name
        confparser.write(self._conf_path(), conf)

    def write_promote_trigger(self) -> None:
        open(os.path.join(self.data_dir,
                          waldb_server.PROMOTE_TRIGGER), "w").close()

    def post_restore_fixup(self) -> None:
        # purge PEER-LOCAL runtime files that rode in with the snapshot:
        # the SOURCE peer's conf (its port/role!), pid files and log.
        # The conf is regenerated for THIS peer right after; a foreign
        # conf left in place could make a db adopt another peer's
        # identity (role=primary on the wrong port)
        for name in ("waldb.pid", "db_child.pid", waldb_server.CONF_NAME,
                     waldb_server.PROMOTE_TRIGGER):
            try:
                os.unlink(os.path.join(self.data_dir, name))
            except FileNotFoundError:
                pass
        try:
            os.unlink(os.path.join(os.path.dirname(self.data_dir),
                                   "waldb.log"))
        except FileNotFoundError:
            pass

    def spawn_argv(self) -> List[str]:
        return [sys.executable, "-m", "manatee_amd.db.waldb.server",
                "-D", self.data_dir, "-v", "--log-file",
                os.path.join(os.path.dirname(self.data_dir), "waldb.log")]

    # -------------------------------------------------------------- client
    def _cli(self) -> WaldbClient:
        if self._client is None:
            self._client = WaldbClient(self.ip, self.port)
        return self._client

    async def ping(self, timeout_s: float = 5.0) -> bool:
        try:
            return await self._cli().ping(timeout_s=timeout_s)
        except (WaldbError, OSError):
            return False

    async def xlog(self) -> str:
        return await self._cli().xlog()

    async def status(self) -> dict:
        return await self._cli().status()

    async def check_repl(self, standby_name: str) -> dict:
        st = await self.status()
        row = next((r for r in st.get("replication", [])
                    if r["application_name"] == standby_name), None)
        if row is None:
            return {"connected": False, "caught_up": False}
        caught_up = (row["sent_lsn"] == row["write_lsn"]
                     and st["current_lsn"] == row["sent_lsn"])
        return {"connected": True, "sync_state": row["sync_state"],
                "sent_lsn": row["sent_lsn"], "write_lsn": row["write_lsn"],
                "flush_lsn": row["flush_lsn"],
                "replay_lsn": row["replay_lsn"], "caught_up": caught_up}

    async def close(self) -> None:
        if self._client is not None:
            await self._client.close()
            self._client = None
