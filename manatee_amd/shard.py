"""Shard — wires coordination, database manager and the FSM together.

Equivalent of ``lib/shard.js``: builds the peer identity
``ip:postgresPort:backupPort`` (ref :39-41), constructs the ZkMgr and the
DbManager, and hands both to the cluster state machine (ref :59-71).

Config keeps the reference's sitter.json field names (etc/sitter.json,
schema ref lib/postgresMgr.js:60-116): ``ip``, ``postgresPort``,
``backupPort``, ``shardPath``, ``zoneId``, ``zkCfg{connStr, opts
{sessionTimeout}}``, ``postgresMgrCfg{...}``.
"""

from __future__ import annotations

import os
from typing import Optional

from .common.logging import Logger, null_logger
from .coord.zkmgr import ZkMgr
from .db.engine import WaldbEngine
from .db.manager import DbManager
from .fsm.peer import ManateePeer
from .storage import open_store

SITTER_CONFIG_SCHEMA = {
    "type": "object",
    "properties": {
        "ip": {"type": "string", "required": True},
        "postgresPort": {"type": "integer", "required": True},
        "backupPort": {"type": "integer", "required": True},
        "shardPath": {"type": "string", "required": True},
        "zoneId": {"type": "string"},
        "zkCfg": {
            "type": "object", "required": True,
            "properties": {
                "connStr": {"type": "string", "required": True},
                "opts": {"type": "object"},
            },
        },
        "postgresMgrCfg": {"type": "object", "required": True},
    },
}


def peer_identity(cfg: dict) -> dict:
    ip = cfg["ip"]
    pg_port = cfg["postgresPort"]
    backup_port = cfg["backupPort"]
    return {
        "id": "%s:%d:%d" % (ip, pg_port, backup_port),
        "zoneId": cfg.get("zoneId", "%s:%d" % (ip, pg_port)),
        "ip": ip,
        "pgUrl": "tcp://postgres@%s:%d/postgres" % (ip, pg_port),
        "backupUrl": "http://%s:%d" % (ip, backup_port),
    }


def build_engine(mgr_cfg: dict, ip: str, pg_port: int, peer_name: str,
                 data_dir: str, log: Logger, store=None):
    kind = mgr_cfg.get("engine", "waldb")
    if kind == "waldb":
        return WaldbEngine(data_dir, ip, pg_port, peer_name, log=log)
    if kind == "postgres":
        from .db.postgres import PostgresEngine
        from .storage.zfsstore import ZfsStore
        cfg = dict(mgr_cfg)
        # full_page_writes may only be relaxed on a copy-on-write store
        cfg.setdefault("storeIsCow", isinstance(store, ZfsStore))
        return PostgresEngine(data_dir, ip, pg_port, peer_name,
                              cfg=cfg, log=log)
    raise ValueError("unknown engine %r" % kind)


class Shard:
    def __init__(self, cfg: dict, log: Optional[Logger] = None):
        self.cfg = cfg
        self.log = log or null_logger()
        self.ident = peer_identity(cfg)

        mgr_cfg = cfg["postgresMgrCfg"]
        self.store = open_store(mgr_cfg["storageCfg"], log=self.log)
        data_dir = mgr_cfg.get("dataDir") or \
            os.path.join(self.store.mountpoint(), "data")
        self.engine = build_engine(mgr_cfg, cfg["ip"], cfg["postgresPort"],
                                   self.ident["id"], data_dir, self.log,
                                   store=self.store)
        self.db_manager = DbManager(
            engine=self.engine, store=self.store, ip=cfg["ip"],
            health_interval_s=mgr_cfg.get("healthChkInterval", 1000) / 1000.0,
            health_timeout_s=mgr_cfg.get("healthChkTimeout", 5000) / 1000.0,
            ops_timeout_s=mgr_cfg.get("opsTimeout", 60000) / 1000.0,
            replication_timeout_s=mgr_cfg.get("replicationTimeout",
                                              60000) / 1000.0,
            one_node_write_mode=bool(mgr_cfg.get("oneNodeWriteMode")),
            log=self.log)

        zk_cfg = cfg["zkCfg"]
        self.zk = ZkMgr(
            id=self.ident["id"],
            data={k: v for k, v in self.ident.items() if k != "id"},
            path=cfg["shardPath"],
            conn_str=zk_cfg["connStr"],
            session_timeout_ms=(zk_cfg.get("opts") or {})
            .get("sessionTimeout", 60000),
            log=self.log)

        self.peer = ManateePeer(
            zk=self.zk, db=self.db_manager, self_ident=self.ident,
            singleton=bool(mgr_cfg.get("oneNodeWriteMode")),
            log=self.log,
            tick_interval_s=mgr_cfg.get("tickInterval", 1000) / 1000.0)

    async def start(self) -> None:
        self.peer.start()
        await self.db_manager.start()
        await self.zk.init()
        self.log.info("shard started", id=self.ident["id"])

    def debug_state(self) -> dict:
        return self.peer.debug_state()

    async def shutdown(self) -> None:
        """Close coordination and kill the database DIRTY — never a clean
        shutdown, to avoid xlog divergence (ref lib/shard.js:78-93,
        MANATEE-188)."""
        await self.peer.close()
        await self.zk.close()
        await self.db_manager.close()
