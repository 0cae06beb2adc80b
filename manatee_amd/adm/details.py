"""Cluster status + diagnosis (ref lib/adm.js loadClusterDetails :652-715,
ManateeClusterDetails :760-824, loadErrors :875-928, loadReplErrors
:930-985).

Builds one in-memory picture of a shard — the on-ZK cluster state, the
live election members, and every peer's database status (the
``pg_stat_replication`` analogue served by waldb/``status``) — and runs
the reference's diagnosis rules over it to produce operator-facing
errors and warnings.

Test seam: ``MANATEE_ADM_TEST_STATE=<file>`` short-circuits all network
access and loads a JSON fixture instead (ref lib/adm.js:662-665,
721-745) — this is what the golden-output CLI tests use.
"""

from __future__ import annotations

import asyncio
import json
import os
import time
from typing import Dict, List, Optional

from ..db.waldb.client import WaldbClient

TEST_STATE_ENV = "MANATEE_ADM_TEST_STATE"

# column widths of the reference tables (ref allColumns
# bin/manatee-adm:1151-1222 — byte-compatible, incl. trailing padding)
COLUMNS = {
    "peername": {"label": "PEERNAME", "width": 36},
    "peerabbr": {"label": "PEER", "width": 8},
    "role": {"label": "ROLE", "width": 8},
    "ip": {"label": "IP", "width": 16},
    "pg-online": {"label": "PG", "width": 4},
    "pg-repl": {"label": "REPL", "width": 5},
    "pg-sent": {"label": "SENT", "width": 13},
    "pg-write": {"label": "WRITE", "width": 13},
    "pg-flush": {"label": "FLUSH", "width": 13},
    "pg-replay": {"label": "REPLAY", "width": 13},
    "pg-lag": {"label": "LAG", "width": 6},
}
# ref columnAliases bin/manatee-adm:1224-1227
COLUMN_ALIASES = {"zonename": "peername", "zoneabbr": "peerabbr"}
PG_ONLY_COLUMNS = {"pg-online", "pg-repl", "pg-sent", "pg-write",
                   "pg-flush", "pg-replay", "pg-lag"}
PEERS_COLUMNS = ["role", "peername", "ip"]
STATUS_COLUMNS = ["role", "peerabbr", "pg-online", "pg-repl", "pg-sent",
                  "pg-flush", "pg-replay", "pg-lag"]
STATUS_COLUMNS_WIDE = ["role", "peername", "pg-online", "pg-repl", "pg-sent",
                       "pg-flush", "pg-replay", "pg-lag"]
ROLES = ("primary", "sync", "async", "deposed")

_LAG_UNITS = {"days": 86400, "hours": 3600, "minutes": 60, "seconds": 1}


def lag_in_seconds(lag):
    """ref lagInSeconds lib/adm.js:2504-2541 — a postgres interval object
    ({days, hours, minutes, seconds}) to seconds; None for no lag;
    the string "?" for units we cannot interpret."""
    if not lag:
        return None
    if isinstance(lag, (int, float)):
        return lag
    seconds = 0
    for unit, val in lag.items():
        if unit not in _LAG_UNITS:
            return "?"
        seconds += val * _LAG_UNITS[unit]
    return seconds


def duration(seconds) -> str:
    """ref pgDuration bin/manatee-adm:1420-1437 — ``%dm%02ds``."""
    if seconds is None:
        return "-"
    try:
        seconds = int(seconds)
    except (TypeError, ValueError):
        return "?"
    if seconds < 0:
        return "?"
    return "%dm%02ds" % (seconds // 60, seconds % 60)


class PeerDetails:
    """One peer's row: identity + role + db status (ref pgp_* fields)."""

    def __init__(self, ident: dict, role: str):
        self.ident = dict(ident)
        self.id = ident.get("id", "?")
        self.role = role
        self.db_error: Optional[str] = None    # pgp_pgerr
        self.status: Optional[dict] = None     # raw waldb status
        self.repl_rows: List[dict] = []        # pg_stat_replication rows
        self.lag_s = None                      # upstream replay lag (or "?")

    @property
    def zone_id(self) -> str:
        """The peer's name for display: zoneId when present, id otherwise
        (ref PEERNAME = pgp_ident.zoneId, bin/manatee-adm:1379)."""
        return self.ident.get("zoneId") or self.id

    @property
    def label(self) -> str:
        """ref pgp_label = zoneId.substr(0, 8) (lib/adm.js:841)."""
        return self.zone_id[:8]

    @property
    def online(self) -> bool:
        return self.db_error is None and self.status is not None

    def repl_to(self, downstream_id: str) -> Optional[dict]:
        """The replication row for the named downstream, if connected."""
        for row in self.repl_rows:
            if row.get("application_name") == downstream_id:
                return row
        return None

    def first_repl(self) -> Optional[dict]:
        return self.repl_rows[0] if self.repl_rows else None

    def cell(self, col: str) -> str:
        """ref rowForPeer bin/manatee-adm:1376-1414 (incl. the
        ``*_lsn || *_location`` fallback for pre-PG-10 field names)."""
        if col == "peername":
            return self.zone_id
        if col == "peerabbr":
            return self.label
        if col == "role":
            return self.role
        if col == "ip":
            return self.ident.get("ip", "-") or "-"
        if col == "pg-online":
            return "ok" if self.online else "fail"
        if not self.online:
            return "-"   # ref: every pg column renders "-" on pgp_pgerr
        if col == "pg-lag":
            return duration(self.lag_s)
        row = self.first_repl()

        def lsn(prefix: str) -> str:
            r = row or {}
            return r.get(prefix + "_lsn") or r.get(prefix + "_location") \
                or "-"
        if col == "pg-repl":
            return (row or {}).get("sync_state") or "-"
        if col == "pg-sent":
            return lsn("sent")
        if col == "pg-write":
            return lsn("write")
        if col == "pg-flush":
            return lsn("flush")
        if col == "pg-replay":
            return lsn("replay")
        raise KeyError("unknown column %r" % col)


class ClusterDetails:
    """The full picture of one shard (ref ManateeClusterDetails)."""

    def __init__(self, shard: str, state: dict,
                 zk_conn: Optional[str] = None):
        self.shard = shard
        self.zk_conn = zk_conn
        self.state = state
        self.generation = state.get("generation")
        self.init_wal = state.get("initWal")
        self.freeze = state.get("freeze")
        self.singleton = bool(state.get("oneNodeWriteMode"))
        self.promote = state.get("promote")
        self.peers: Dict[str, PeerDetails] = {}
        self.order: List[str] = []             # table ordering
        self.primary_id: Optional[str] = None
        self.sync_id: Optional[str] = None
        self.async_ids: List[str] = []
        self.deposed_ids: List[str] = []
        self.errors: List[str] = []            # pgs_errors
        self.warnings: List[str] = []          # pgs_warnings

        def add(ident: Optional[dict], role: str) -> Optional[str]:
            if not ident:
                return None
            pd = PeerDetails(ident, role)
            self.peers[pd.id] = pd
            self.order.append(pd.id)
            return pd.id

        self.primary_id = add(state.get("primary"), "primary")
        if not self.singleton:
            self.sync_id = add(state.get("sync"), "sync")
        for a in state.get("async") or []:
            aid = add(a, "async")
            if aid:
                self.async_ids.append(aid)
        for d in state.get("deposed") or []:
            did = add(d, "deposed")
            if did:
                self.deposed_ids.append(did)

    # ------------------------------------------------------------ statuses
    async def add_db_status(self, timeout_s: float = 5.0,
                            now: Optional[float] = None) -> None:
        """Query every peer's database in parallel (ref _addPostgresStatus
        :348-427 — one connection per peer, vasync barrier)."""
        async def one(pd: PeerDetails) -> None:
            url = pd.ident.get("pgUrl")
            if not url:
                pd.db_error = "peer has no database URL"
                return
            cli = None
            try:
                cli = WaldbClient.from_url(url, connect_timeout_s=timeout_s)
                pd.status = await asyncio.wait_for(cli.status(), timeout_s)
            except Exception as exc:
                pd.db_error = str(exc) or exc.__class__.__name__
            finally:
                if cli is not None:
                    await cli.close()
        await asyncio.gather(*(one(p) for p in self.peers.values()))
        self._ingest_statuses(now)

    def _ingest_statuses(self, now: Optional[float] = None) -> None:
        now = time.time() if now is None else now
        for pd in self.peers.values():
            if pd.status is None:
                continue
            pd.repl_rows = list(pd.status.get("replication") or [])
            # lag only meaningful for peers replaying from an upstream
            # (ref PG_REPL_LAG lib/adm.js:69, lagInSeconds :2504-2541)
            if pd.role != "primary":
                lrt = pd.status.get("last_replay_time")
                if lrt:
                    pd.lag_s = max(0.0, now - float(lrt))

    # ------------------------------------------------------------ diagnosis
    def load_errors(self) -> None:
        """ref loadErrors lib/adm.js:875-928."""
        p = self.peers.get(self.primary_id) if self.primary_id else None
        if p is not None and not p.online:
            self.errors.append("cannot query postgres on primary")

        if self.singleton:
            extra = [pid for pid in self.peers if pid != self.primary_id]
            if extra:
                self.warnings.append("found %d peers in singleton mode"
                                     % len(self.peers))
            return

        s = self.peers.get(self.sync_id) if self.sync_id else None
        if s is not None and not s.online:
            self.errors.append("cannot query postgres on sync")

        if self.deposed_ids:
            self.warnings.append("cluster has a deposed peer")
        if not self.async_ids:
            self.warnings.append("cluster has no async peers")

        # if the sync is down, that's all we can really check for now
        if s is None or not s.online:
            return
        # the primary's downstream is checked even when the primary itself
        # is unreachable (ref loadErrors — it reports BOTH "cannot query"
        # and "downstream not connected" for a down primary)
        if p is not None:
            self.load_repl_errors(p, self.sync_id, "sync", self.errors)
        self.load_repl_errors(
            s, self.async_ids[0] if self.async_ids else None,
            "async", self.warnings)
        for i, aid in enumerate(self.async_ids):
            nxt = (self.async_ids[i + 1]
                   if i < len(self.async_ids) - 1 else None)
            apd = self.peers[aid]
            if apd.online:
                self.load_repl_errors(apd, nxt, "async", self.warnings)

    def load_repl_errors(self, peer: PeerDetails,
                         ds_id: Optional[str], kind: str,
                         out: List[str]) -> None:
        """ref loadReplErrors lib/adm.js:930-985 — check the peer's
        downstream connection: present, to the right peer, streaming, and
        with the expected sync_state."""
        if ds_id is None:
            return
        before = len(out)
        row = peer.first_repl()
        if row is None:
            out.append('peer "%s": downstream replication peer not '
                       "connected" % peer.label)
            return
        ds = self.peers.get(ds_id)
        if "client_addr" in row:
            # PostgreSQL-style row: match the downstream by IP
            # (ref loadReplErrors lib/adm.js:958-963)
            if row.get("client_addr") != (ds.ident.get("ip") if ds
                                          else None):
                out.append('peer "%s": expected downstream peer to be '
                           '"%s", but found "%s"'
                           % (peer.label,
                              (ds.ident.get("ip") if ds else ds_id),
                              row.get("client_addr")))
        elif row.get("application_name") != ds_id:
            out.append('peer "%s": expected downstream peer to be "%s", '
                       'but found "%s"'
                       % (peer.label, (ds.label if ds else ds_id),
                          str(row.get("application_name"))[:8]))
        if row.get("state") != "streaming":
            out.append('peer "%s": downstream replication not yet '
                       'established (expected state "streaming", found '
                       '"%s")' % (peer.label, row.get("state")))
        if len(out) > before:
            return
        if row.get("sync_state") != kind:
            out.append('peer "%s": expected downstream replication to be '
                       '"%s", but found "%s"'
                       % (peer.label, kind, row.get("sync_state")))

    # ------------------------------------------------------------ rendering
    def table_rows(self, columns: List[str],
                   role: Optional[str] = None) -> List[List[str]]:
        rows = []
        for pid in self.order:
            pd = self.peers[pid]
            if role and pd.role != role:
                continue
            rows.append([pd.cell(c) for c in columns])
        return rows


def render_table(columns: List[str], rows: List[List[str]],
                 header: bool = True) -> str:
    """Fixed-width columns, single-space separated, trailing pad kept —
    the reference's tab-module layout (bin/manatee-adm:1330-1374)."""
    out = []
    if header:
        out.append(" ".join(
            COLUMNS[c]["label"].ljust(COLUMNS[c]["width"])
            for c in columns))
    for row in rows:
        out.append(" ".join(
            cell.ljust(COLUMNS[c]["width"])
            for c, cell in zip(columns, row)))
    return "\n".join(out) + ("\n" if out else "")


# --------------------------------------------------------------- loading

async def load_cluster_details(zk, shard: str, *, zk_conn: str = "",
                               timeout_s: float = 5.0) -> ClusterDetails:
    """ref loadClusterDetails lib/adm.js:652-715 (live path)."""
    from . import core as adm
    state, _ = await adm.get_state(zk, shard)
    if state is None:
        raise adm.AdmError(
            "cluster state not found for shard %r (not yet set up?)" % shard)
    cd = ClusterDetails(shard, state, zk_conn=zk_conn)
    await cd.add_db_status(timeout_s=timeout_s)
    cd.load_errors()
    return cd


def load_fixture(path: str, shard: Optional[str] = None,
                 zk_conn: Optional[str] = None) -> ClusterDetails:
    """Fixture seam (ref MANATEE_ADM_TEST_STATE lib/adm.js:721-745).

    Accepts TWO formats:

    - this repo's: ``{clusterState, db: {peerId: status|null}, now?}`` —
      db statuses replace the live queries;
    - the REFERENCE's internal representation (``pgs_*``/``pgp_*`` fields,
      exactly what the reference's own golden tests feed its CLI,
      ref test/tst.manateeAdm.js MockState) — this is what makes the
      byte-compat test able to render reference-shaped fixtures.
    """
    with open(path) as f:
        fx = json.load(f)
    if "pgs_peers" in fx:
        return _load_reference_fixture(fx, shard=shard, zk_conn=zk_conn)
    cd = ClusterDetails(shard or fx.get("shard", "1.fixture"),
                        fx["clusterState"],
                        zk_conn=zk_conn or "UNUSED")
    dbmap = fx.get("db") or {}
    for pid, pd in cd.peers.items():
        st = dbmap.get(pid)
        if st is None:
            pd.db_error = "connection refused"
        else:
            pd.status = st
    cd._ingest_statuses(now=fx.get("now"))
    cd.load_errors()
    return cd


def _load_reference_fixture(fx: dict, shard: Optional[str] = None,
                            zk_conn: Optional[str] = None
                            ) -> ClusterDetails:
    """Build ClusterDetails from the reference's internal representation
    (ref ManateeClusterDetails lib/adm.js:760-875)."""
    peers = fx.get("pgs_peers") or {}

    def ident_of(pid):
        return (peers.get(pid) or {}).get("pgp_ident") or {"id": pid}

    state = {
        "generation": fx.get("pgs_generation"),
        "initWal": fx.get("pgs_initwal"),
        "oneNodeWriteMode": bool(fx.get("pgs_singleton")),
        "primary": ident_of(fx["pgs_primary"]) if fx.get("pgs_primary")
        else None,
        "sync": ident_of(fx["pgs_sync"]) if fx.get("pgs_sync") else None,
        "async": [ident_of(a) for a in fx.get("pgs_asyncs") or []],
        "deposed": [ident_of(d) for d in fx.get("pgs_deposed") or []],
    }
    if fx.get("pgs_frozen"):
        state["freeze"] = {"date": fx.get("pgs_freeze_time"),
                           "reason": fx.get("pgs_freeze_reason")}
    cd = ClusterDetails(shard or "1.fixture", state,
                        zk_conn=zk_conn or "UNUSED")
    for pid, pd in cd.peers.items():
        raw = peers.get(pid) or {}
        if raw.get("pgp_pgerr"):
            pd.db_error = "connection refused"
        else:
            pd.status = {}
        if raw.get("pgp_repl"):
            pd.repl_rows = [raw["pgp_repl"]]
        pd.lag_s = lag_in_seconds(raw.get("pgp_lag"))
    cd.errors = list(fx.get("pgs_errors") or [])
    cd.warnings = list(fx.get("pgs_warnings") or [])
    cd.load_errors()
    return cd


def fixture_path() -> Optional[str]:
    return os.environ.get(TEST_STATE_ENV) or None
