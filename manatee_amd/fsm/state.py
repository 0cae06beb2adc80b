"""Cluster-state schema helpers and transition-legality rules.

The on-ZK cluster state object (ref docs/migrate-1-to-2.md:350-379, consumed
at lib/adm.js:788-819):

    {
      "generation": <int>,
      "primary": {id, zoneId, ip, pgUrl, backupUrl},
      "sync": <peer|null>,
      "async": [<peer>...],
      "deposed": [<peer>...],
      "initWal": "H/LLLLLLLL",
      "freeze": {"date": iso8601, "reason": str}?      (optional)
      "oneNodeWriteMode": true?                        (optional)
      "promote": {id, role, asyncIndex?, generation, expireTime}?  (optional)
    }

The transition-legality rules here are the executable versions of the
invariants ``manatee-adm history -v`` diagnoses (lib/adm.js:2319-2376):
generation never decreases; a new primary must be the previous sync; a sync
change requires a generation bump.
"""

from __future__ import annotations

import time
from typing import List, Optional

from ..common import lsn as lsnmod

ROLE_PRIMARY = "primary"
ROLE_SYNC = "sync"
ROLE_ASYNC = "async"
ROLE_DEPOSED = "deposed"
ROLE_NONE = "none"
ROLE_UNASSIGNED = "unassigned"


def make_ident(id: str, zone_id: str = "", ip: str = "",
               pg_url: str = "", backup_url: str = "") -> dict:
    parts = id.split(":")
    if len(parts) == 3 and not ip:
        ip = parts[0]
    if len(parts) == 3 and not pg_url:
        pg_url = "tcp://postgres@%s:%s/postgres" % (parts[0], parts[1])
    if len(parts) == 3 and not backup_url:
        backup_url = "http://%s:%s" % (parts[0], parts[2])
    return {"id": id, "zoneId": zone_id or id, "ip": ip,
            "pgUrl": pg_url, "backupUrl": backup_url}


def ident_from_active(active_entry: dict) -> dict:
    return make_ident(active_entry["id"],
                      zone_id=active_entry.get("zoneId", ""),
                      ip=active_entry.get("ip", ""),
                      pg_url=active_entry.get("pgUrl", ""),
                      backup_url=active_entry.get("backupUrl", ""))


def role_of(state: Optional[dict], peer_id: str) -> str:
    """Which role does peer_id hold in the cluster state?"""
    if state is None:
        return ROLE_UNASSIGNED
    for dep in state.get("deposed") or []:
        if dep.get("id") == peer_id:
            return ROLE_DEPOSED
    if state.get("primary") and state["primary"].get("id") == peer_id:
        return ROLE_PRIMARY
    if state.get("sync") and state["sync"].get("id") == peer_id:
        return ROLE_SYNC
    for a in state.get("async") or []:
        if a.get("id") == peer_id:
            return ROLE_ASYNC
    return ROLE_UNASSIGNED


def async_index(state: dict, peer_id: str) -> int:
    for i, a in enumerate(state.get("async") or []):
        if a.get("id") == peer_id:
            return i
    return -1


def all_peer_ids(state: Optional[dict]) -> List[str]:
    if state is None:
        return []
    out = []
    if state.get("primary"):
        out.append(state["primary"]["id"])
    if state.get("sync"):
        out.append(state["sync"]["id"])
    out.extend(a["id"] for a in state.get("async") or [])
    out.extend(d["id"] for d in state.get("deposed") or [])
    return out


def is_frozen(state: Optional[dict]) -> bool:
    return bool(state and state.get("freeze"))


def promote_expired(promote: dict, now: Optional[float] = None) -> bool:
    """promote.expireTime is ISO8601 with ms (ref lib/adm.js:1925-1926)."""
    exp = promote.get("expireTime")
    if not exp:
        return True
    try:
        import calendar
        t = calendar.timegm(time.strptime(exp[:19], "%Y-%m-%dT%H:%M:%S"))
        frac = 0.0
        if len(exp) > 19 and exp[19] == ".":
            frac = float("0" + exp[19:].rstrip("Z"))
        return (now if now is not None else time.time()) > t + frac
    except ValueError:
        return True


def iso8601(t: Optional[float] = None) -> str:
    t = time.time() if t is None else t
    ms = int(round((t % 1) * 1000))
    if ms >= 1000:
        ms = 999
    return time.strftime("%Y-%m-%dT%H:%M:%S", time.gmtime(t)) + \
        ".%03dZ" % ms


class TransitionError(Exception):
    pass


def check_transition(old: Optional[dict], new: dict) -> None:
    """Raise TransitionError if old→new violates the FSM safety rules
    (the rules ``annotateHistoryNode`` diagnoses, lib/adm.js:2296-2416)."""
    gen_new = new.get("generation")
    if not isinstance(gen_new, int) or gen_new < 0:
        raise TransitionError("bad generation %r" % (gen_new,))
    if not new.get("primary"):
        raise TransitionError("state must name a primary")
    if new.get("initWal") is None or not lsnmod.is_lsn(new["initWal"]):
        raise TransitionError("bad initWal %r" % (new.get("initWal"),))
    # primary must not also be sync/async/deposed
    pid = new["primary"]["id"]
    if new.get("sync") and new["sync"]["id"] == pid:
        raise TransitionError("primary cannot be its own sync")
    for a in new.get("async") or []:
        if a["id"] == pid:
            raise TransitionError("primary cannot be in the async chain")
    for d in new.get("deposed") or []:
        if d["id"] == pid:
            raise TransitionError("primary cannot be deposed")
    if old is None:
        return
    gen_old = old.get("generation", 0)
    if gen_new < gen_old:
        raise TransitionError("generation moved backwards (%d -> %d)"
                              % (gen_old, gen_new))
    if gen_new == gen_old:
        # same generation: primary and sync must be unchanged
        if old.get("primary") and old["primary"]["id"] != pid:
            raise TransitionError(
                "primary changed without a generation bump")
        old_sync = (old.get("sync") or {}).get("id")
        new_sync = (new.get("sync") or {}).get("id")
        if old_sync != new_sync:
            raise TransitionError("sync changed without a generation bump")
    else:
        # new primary must be the old sync or the old primary
        old_pid = (old.get("primary") or {}).get("id")
        old_sync = (old.get("sync") or {}).get("id")
        if pid not in (old_pid, old_sync):
            raise TransitionError(
                "new primary %s was neither the old primary nor the old sync"
                % pid)
