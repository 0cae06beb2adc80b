"""Admin operations library (ref lib/adm.js).

State-access pipeline helpers mirroring the reference's:
``create_zk_client`` (:81), ``get_shards`` (:107), ``get_state`` (:132),
``put_state`` — versioned CAS write + history node (:152),
``get_cluster_states`` (:191).  Operator ops: freeze/unfreeze
(:1048/:1079), reap (:1108), promote/clear-promote (:1693/:2019),
rebuild (:1319), history (:2088), plus the status/diagnosis layer used
by pg-status/show/verify (loadClusterDetails :652, loadErrors :875).
"""

from __future__ import annotations

import asyncio
import json
import time
from typing import List, Optional, Tuple

from ..common import lsn as lsnmod
from ..common.logging import Logger, null_logger
from ..coord import jute
from ..coord.zkclient import ZkClient
from ..fsm import state as st


class AdmError(RuntimeError):
    pass


async def create_zk_client(conn_str: str, timeout_s: float = 10.0,
                           log: Optional[Logger] = None) -> ZkClient:
    cli = ZkClient(conn_str, session_timeout_ms=30000, log=log)
    await cli.connect(timeout_s=timeout_s)
    return cli


async def get_shards(zk: ZkClient, base: str = "/manatee") -> List[str]:
    """List shard names under the manatee base path (ref _getShards :107)."""
    try:
        children, _ = await zk.get_children(base)
    except jute.ZkError as exc:
        if exc.code == jute.ZNONODE:
            return []
        raise
    return sorted(children)


def shard_path(shard: str, base: str = "/manatee") -> str:
    return shard if shard.startswith("/") else "%s/%s" % (base, shard)


async def get_state(zk: ZkClient, shard: str
                    ) -> Tuple[Optional[dict], Optional[int]]:
    """Read cluster state + version for CAS (ref _getState :132)."""
    try:
        data, stat = await zk.get_data(shard_path(shard) + "/state")
    except jute.ZkError as exc:
        if exc.code == jute.ZNONODE:
            return None, None
        raise
    return json.loads(data.decode("utf-8")), stat.version


async def put_state(zk: ZkClient, shard: str, state: dict,
                    version: int) -> None:
    """Versioned CAS write + history node, atomically (ref _putState :152
    — same transaction shape as the sitter's putClusterState)."""
    data = json.dumps(state, separators=(",", ":")).encode()
    hp = "%s/history/%s-" % (shard_path(shard), state["generation"])
    await zk.multi([
        jute.MultiOp.create(hp, data, jute.PERSISTENT_SEQUENTIAL),
        jute.MultiOp.set_data(shard_path(shard) + "/state", data,
                              version=version),
    ])


async def update_state(zk: ZkClient, shard: str, mutate,
                       retries: int = 5) -> dict:
    """Read-modify-CAS-write with retry; ``mutate(state) -> new state`` or
    raises AdmError.  Returns the written state."""
    for _ in range(retries):
        state, version = await get_state(zk, shard)
        if state is None:
            raise AdmError("cluster state does not exist yet")
        new_state = mutate(dict(state))
        try:
            await put_state(zk, shard, new_state, version)
            return new_state
        except jute.ZkError as exc:
            if exc.code != jute.ZBADVERSION:
                raise
            await asyncio.sleep(0.1)
    raise AdmError("state write kept conflicting; giving up")


async def get_active(zk: ZkClient, shard: str) -> List[dict]:
    """Live election members with their identity data
    (ref _active :512-553)."""
    from ..coord.zkmgr import parse_and_unique_actives
    path = shard_path(shard) + "/election"
    try:
        children, _ = await zk.get_children(path)
    except jute.ZkError as exc:
        if exc.code == jute.ZNONODE:
            return []
        raise
    peers = parse_and_unique_actives(children)
    out = []
    for p in peers:
        node = "%s/%s-%010d" % (path, p.id, p.seq)
        try:
            data, _ = await zk.get_data(node)
            p.data = json.loads(data.decode("utf-8")) if data else {}
        except (jute.ZkError, ValueError):
            p.data = {}
        out.append(p.as_dict())
    return out


async def get_history(zk: ZkClient, shard: str) -> List[dict]:
    """All history nodes, sorted by ZK sequence (ref history :2088-2162).
    Each entry: {zkSeq, time, state}."""
    path = shard_path(shard) + "/history"
    try:
        children, _ = await zk.get_children(path)
    except jute.ZkError as exc:
        if exc.code == jute.ZNONODE:
            return []
        raise

    def seq_of(name: str) -> int:
        return int(name.rsplit("-", 1)[1])

    out = []
    for name in sorted(children, key=seq_of):
        try:
            data, stat = await zk.get_data("%s/%s" % (path, name))
            out.append({"zkSeq": seq_of(name),
                        "time": stat.ctime,
                        "state": json.loads(data.decode("utf-8"))})
        except (jute.ZkError, ValueError):
            continue
    return out


def annotate_history(entries: List[dict]) -> List[dict]:
    """Attach human annotations + legal-transition checks to each history
    entry (ref annotateHistoryNode lib/adm.js:2296-2416 — that function
    encodes the FSM's safety rules: the generation never goes backwards,
    a new primary must be the previous sync, and a sync change requires a
    generation bump).  Each returned entry gains ``notes`` and
    ``violations`` lists."""
    def pid(ident) -> Optional[str]:
        return ident.get("id") if ident else None

    def abbr(ident) -> str:
        return (pid(ident) or "-")[:8]

    out = []
    last: Optional[dict] = None
    for e in entries:
        notes: List[str] = []
        violations: List[str] = []
        nst = e["state"]
        if last is None:
            notes.append("cluster setup for %s mode"
                         % ("singleton (one-node-write)"
                            if nst.get("oneNodeWriteMode")
                            else "normal (multi-peer)"))
            out.append({**e, "notes": notes, "violations": violations})
            last = nst
            continue
        lst = last
        ngen, lgen = nst.get("generation"), lst.get("generation")
        if ngen < lgen:
            violations.append("gen number went backwards")
        elif not lst.get("oneNodeWriteMode") and nst.get("oneNodeWriteMode"):
            violations.append("unsupported transition from multi-peer mode "
                              "to singleton (one-node-write) mode")
        elif lst.get("oneNodeWriteMode") and not nst.get("oneNodeWriteMode"):
            notes.append("cluster transitioned from singleton "
                         "(one-node-write) mode to multi-peer mode")
        elif pid(nst.get("primary")) != pid(lst.get("primary")):
            if ngen == lgen:
                violations.append("new primary, but same gen number")
            elif (lst.get("sync") is None
                  or pid(nst["primary"]) != pid(lst["sync"])):
                violations.append("new primary was not previous sync")
            else:
                notes.append("sync (%s) took over as primary (from %s)"
                             % (abbr(nst["primary"]), abbr(lst["primary"])))
        elif ngen > lgen:
            if lst.get("sync") is None and not lst.get("oneNodeWriteMode"):
                notes.append('sync "%s" added' % abbr(nst.get("sync")))
            elif pid(nst.get("sync")) == pid(lst.get("sync")):
                violations.append("gen number changed, but primary and "
                                  "sync did not")
            else:
                notes.append("primary (%s) selected new sync (was %s, "
                             "now %s)" % (abbr(nst["primary"]),
                                          abbr(lst.get("sync")),
                                          abbr(nst.get("sync"))))
        elif pid(nst.get("sync")) != pid(lst.get("sync")):
            violations.append("sync changed, but gen number did not")
        else:
            if nst.get("freeze") and not lst.get("freeze"):
                notes.append("cluster frozen: %s"
                             % nst["freeze"].get("reason"))
            elif lst.get("freeze") and not nst.get("freeze"):
                notes.append("cluster unfrozen")
            news = {pid(a) for a in nst.get("async") or []}
            olds = {pid(a) for a in lst.get("async") or []}
            for a in sorted(news - olds):
                notes.append('async "%s" added' % a[:8])
            for a in sorted(olds - news):
                notes.append('async "%s" removed' % a[:8])
            newd = {pid(d) for d in nst.get("deposed") or []}
            oldd = {pid(d) for d in lst.get("deposed") or []}
            for d in sorted(newd - oldd):
                notes.append('"%s" deposed' % d[:8])
            for d in sorted(oldd - newd):
                notes.append('"%s" no longer deposed' % d[:8])
            if nst.get("promote") and not lst.get("promote"):
                notes.append('promote requested for "%s"'
                             % (nst["promote"].get("id") or "-")[:8])
            elif lst.get("promote") and not nst.get("promote"):
                notes.append("promote request cleared")
        out.append({**e, "notes": notes, "violations": violations})
        last = nst
    return out


# ----------------------------------------------------------------- operator ops

async def freeze(zk: ZkClient, shard: str, reason: str) -> dict:
    """ref freeze :1048-1077."""
    if not reason:
        raise AdmError("a reason is required (-r)")

    def mutate(s: dict) -> dict:
        s["freeze"] = {"date": st.iso8601(), "reason": reason}
        return s
    return await update_state(zk, shard, mutate)


async def unfreeze(zk: ZkClient, shard: str) -> dict:
    """ref unfreeze :1079-1106."""
    def mutate(s: dict) -> dict:
        if "freeze" not in s:
            raise AdmError("cluster is not frozen")
        s.pop("freeze", None)
        return s
    return await update_state(zk, shard, mutate)


async def reap(zk: ZkClient, shard: str, peer_id: Optional[str] = None,
               zonename: Optional[str] = None) -> dict:
    """Remove a peer from the deposed list (ref reap :1108-1146)."""
    def mutate(s: dict) -> dict:
        deposed = s.get("deposed") or []
        keep = [d for d in deposed
                if not (d.get("id") == peer_id
                        or (zonename and d.get("zoneId") == zonename))]
        if len(keep) == len(deposed):
            raise AdmError("peer not found in deposed list")
        s["deposed"] = keep
        return s
    return await update_state(zk, shard, mutate)


async def set_onwm(zk: ZkClient, shard: str, mode: str) -> dict:
    """Toggle one-node-write-mode (ref setOnwm :1148-1209)."""
    if mode not in ("on", "off"):
        raise AdmError("mode must be 'on' or 'off'")

    def mutate(s: dict) -> dict:
        if mode == "on":
            if s.get("sync") or s.get("async"):
                raise AdmError("cannot enable ONWM with standbys present")
            s["oneNodeWriteMode"] = True
        else:
            s.pop("oneNodeWriteMode", None)
        return s
    return await update_state(zk, shard, mutate)


async def request_promote(zk: ZkClient, shard: str, *, role: str,
                          peer_id: Optional[str] = None,
                          zonename: Optional[str] = None,
                          async_index: Optional[int] = None,
                          expire_s: float = 30.0) -> dict:
    """Write a promote request object (ref promote :1693-2014; object shape
    :1915-1928).  Validation of role/position mirrors the reference."""
    if role not in ("sync", "async"):
        raise AdmError("role must be 'sync' or 'async'")

    def mutate(s: dict) -> dict:
        if s.get("freeze"):
            raise AdmError("cluster is frozen")
        if s.get("oneNodeWriteMode"):
            raise AdmError("cannot promote in one-node-write mode")
        if role == "sync":
            target = s.get("sync")
            if not target:
                raise AdmError("cluster has no sync")
        else:
            asyncs = s.get("async") or []
            idx = async_index
            if idx is None:
                if len(asyncs) != 1:
                    raise AdmError("--asyncIndex required with multiple "
                                   "asyncs")
                idx = 0
            if not (0 <= idx < len(asyncs)):
                raise AdmError("asyncIndex out of range")
            target = asyncs[idx]
        if peer_id and target.get("id") != peer_id:
            raise AdmError("peer %s is not the %s" % (peer_id, role))
        if zonename and target.get("zoneId") != zonename:
            raise AdmError("zone %s is not the %s" % (zonename, role))
        promote = {"id": target["id"], "role": role,
                   "generation": s["generation"],
                   "expireTime": st.iso8601(time.time() + expire_s)}
        if role == "async":
            promote["asyncIndex"] = idx if async_index is not None else 0
        s["promote"] = promote
        return s
    return await update_state(zk, shard, mutate)


async def clear_promote(zk: ZkClient, shard: str) -> dict:
    """ref clearPromote :2019-2040."""
    def mutate(s: dict) -> dict:
        if "promote" not in s:
            raise AdmError("no promote request present")
        s.pop("promote", None)
        return s
    return await update_state(zk, shard, mutate)


async def state_backfill(zk: ZkClient, shard: str,
                         actives: Optional[List[dict]] = None) -> dict:
    """v1→v2 migration: synthesize a v2 state from the election order
    (ref stateBackfill :1231-1312): first=primary, second=sync, rest
    async, generation 0, initWal 0/0000..., auto-freeze."""
    state, version = await get_state(zk, shard)
    if state is not None:
        raise AdmError("cluster state already exists")
    if actives is None:
        actives = await get_active(zk, shard)
    if len(actives) < 2:
        raise AdmError("need at least two active peers to backfill")
    new_state = {
        "generation": 0,
        "primary": st.ident_from_active(actives[0]),
        "sync": st.ident_from_active(actives[1]),
        "async": [st.ident_from_active(a) for a in actives[2:]],
        "deposed": [],
        "initWal": lsnmod.ZERO,
        "freeze": {"date": st.iso8601(),
                   "reason": "manatee-adm state-backfill"},
    }
    data = json.dumps(new_state, separators=(",", ":")).encode()
    sp = shard_path(shard)
    await zk.multi([
        jute.MultiOp.create("%s/history/%s-" % (sp, 0), data,
                            jute.PERSISTENT_SEQUENTIAL),
        jute.MultiOp.create(sp + "/state", data, jute.PERSISTENT),
    ])
    return new_state
