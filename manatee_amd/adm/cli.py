"""manatee-adm — operator CLI (ref bin/manatee-adm).

Subcommand groups mirror the reference's (bin/manatee-adm:94-122):

  status commands:   status, show, peers, pg-status, verify, history
  state commands:    zk-state (state), zk-active (active), freeze,
                     unfreeze, set-onwm, reap, state-backfill, check-lock
  peer commands:     rebuild, promote, clear-promote
  other:             version, help

Env fallbacks: ``ZK_IPS`` for -z/--zk, ``SHARD`` for -s/--shard,
``MANATEE_SITTER_CONFIG`` for -c/--config (ref bin/manatee-adm:31-88,
docs/man/manatee-adm.md:502-514).  ``MANATEE_ADM_TEST_STATE`` loads a
fixture instead of touching ZK/db — the golden-test seam.
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import sys
import time
from typing import List, Optional

from .. import __version__
from ..common.logging import null_logger
from ..db.waldb.client import WaldbClient
from ..fsm import state as st
from . import core as adm
from . import details as det

RESTORE_ATTEMPTS = 5          # ref lib/adm.js:71


class UsageError(Exception):
    """CLI usage error: printed as ``manatee-adm: <msg>`` followed by the
    subcommand's help text, exit 2 (the reference's cmdln usage()
    behavior, which its golden tests capture byte-for-byte)."""

    def __init__(self, msg: str, help_text: Optional[str] = None):
        super().__init__(msg)
        self.help_text = help_text


# Byte-compatible help texts for the committed subcommands (rendered
# exactly as the reference's cmdln/dashdash does — captured in
# /root/reference/test/tst.manateeAdm.js.out and tst.manateeAdmUsage.js).
_HELP_OPTS = {
    "help": "    -h, --help                          Show this help.",
    "columns": ("    -o COLNAME[,...], --columns=COLNAME[,...]\n"
                "                                        Columns to print."),
    "omitHeader":
        "    -H, --omitHeader                    Omit header row from output.",
    "role":
        "    -r ROLE, --role=ROLE                Only show peers with role "
        "ROLE.",
    "shard":
        "    -s SHARD, --shard=SHARD             Name of the Manatee shard "
        "(cluster).",
    "verbose": "    -v, --verbose                       Enable verbose "
               "output.",
    "wide": "    -w, --wide, --cinematic             Show full peernames.",
    "zk": ("    -z ZK_IPS, --zk=ZK_IPS              The zookeeper connection "
           "string. (e.g.,\n"
           "                                        127.0.0.1:2181)."),
}


def _mk_help(summary: str, usage: str, opts: List[str]) -> str:
    return ("%s\n\nUsage:\n    %s\n\nOptions:\n%s\n"
            % (summary, usage, "\n".join(_HELP_OPTS[o] for o in opts)))


COMMAND_HELP = {
    "peers": _mk_help(
        "Show known peers in this cluster. ",
        "manatee-adm peers [OPTIONS]",
        ["help", "columns", "omitHeader", "role", "shard", "zk"]),
    "pg-status": _mk_help(
        "Show the postgres status of this cluster. ",
        "manatee-adm pg-status [OPTIONS] [PERIOD [COUNT]]",
        ["help", "columns", "omitHeader", "role", "shard", "wide", "zk"]),
    "show": _mk_help(
        "Show cluster summary information.",
        "manatee-adm show [OPTIONS]",
        ["help", "shard", "verbose", "zk"]),
    "verify": _mk_help(
        "Verify the health of the cluster.",
        "manatee-adm verify [OPTIONS]",
        ["help", "verbose", "shard", "zk"]),
}


def _fail(msg: str) -> int:
    print("manatee-adm: %s" % msg, file=sys.stderr)
    return 1


def _need(ns, attr: str, env: str, what: str) -> str:
    val = getattr(ns, attr, None) or os.environ.get(env)
    if not val:
        raise UsageError("%s required (or set %s)" % (what, env))
    return val


async def _with_zk(ns, fn):
    zk = await adm.create_zk_client(_need(ns, "zk", "ZK_IPS", "-z/--zk"),
                                    log=null_logger())
    try:
        return await fn(zk)
    finally:
        await zk.close()


async def _details(ns) -> det.ClusterDetails:
    fx = det.fixture_path()
    if fx:
        # shard/zk still come from flags/env for display (the reference
        # prints opts.shard / opts.zk in "show" even with a fixture)
        return det.load_fixture(
            fx,
            shard=getattr(ns, "shard", None) or os.environ.get("SHARD"),
            zk_conn=getattr(ns, "zk", None) or os.environ.get("ZK_IPS"))
    shard = _need(ns, "shard", "SHARD", "-s/--shard")

    async def go(zk):
        return await det.load_cluster_details(
            zk, shard, zk_conn=_need(ns, "zk", "ZK_IPS", "-z/--zk"))
    return await _with_zk(ns, go)


def _print_issues(cd: det.ClusterDetails, stream, leading_nl: bool) -> None:
    """ref printClusterIssues bin/manatee-adm:1487-1500."""
    if leading_nl and (cd.errors or cd.warnings):
        print("", file=stream)
    for e in cd.errors:
        print("error: %s" % e.split("\n")[0], file=stream)
    for w in cd.warnings:
        print("warning: %s" % w.split("\n")[0], file=stream)


def _columns(ns, default: List[str], no_pg: bool = False) -> List[str]:
    """ref extractColumns bin/manatee-adm:1253-1310: expand comma lists,
    resolve aliases, reject unknown columns — and, for "peers", reject
    pg-only columns (byte-compatible error messages)."""
    help_text = COMMAND_HELP.get(getattr(ns, "cmd_name", ""))
    if not getattr(ns, "columns", None):
        cols = list(default)
    else:
        cols = []
        for chunk in ns.columns:
            for c in chunk.split(","):
                if not c:
                    continue
                c = det.COLUMN_ALIASES.get(c, c)
                if c not in det.COLUMNS:
                    raise UsageError('unsupported column: "%s"' % c,
                                     help_text)
                cols.append(c)
        if not cols:
            raise UsageError("no columns selected", help_text)
    if no_pg:
        for c in cols:
            if c in det.PG_ONLY_COLUMNS:
                raise UsageError('column not available with this '
                                 'subcommand: "%s"'
                                 % det.COLUMNS[c]["label"], help_text)
    return cols


def _check_role(ns) -> Optional[str]:
    role = getattr(ns, "role", None)
    if role and role not in det.ROLES:
        raise UsageError('unsupported value for --role: "%s"' % role,
                         COMMAND_HELP.get(getattr(ns, "cmd_name", "")))
    return role


# ------------------------------------------------------------ subcommands

def cmd_version(ns) -> int:
    print(__version__)
    return 0


async def cmd_peers(ns) -> int:
    cols = _columns(ns, det.PEERS_COLUMNS, no_pg=True)
    role = _check_role(ns)
    cd = await _details(ns)
    sys.stdout.write(det.render_table(
        cols, cd.table_rows(cols, role=role),
        header=not ns.omit_header))
    return 0


async def cmd_pg_status(ns) -> int:
    cols = _columns(ns, det.STATUS_COLUMNS_WIDE if ns.wide
                    else det.STATUS_COLUMNS)
    role = _check_role(ns)
    help_text = COMMAND_HELP.get("pg-status")
    period = count = None
    if ns.period is not None:
        try:
            period = int(ns.period)
        except ValueError:
            period = -1
        if period < 1:
            raise UsageError('invalid period: "%s"' % ns.period, help_text)
        if ns.count is not None:
            try:
                count = int(ns.count)
            except ValueError:
                count = -1
            if count < 1:
                raise UsageError('invalid count: "%s"' % ns.count,
                                 help_text)
    if count is None and period is None:
        count = 1
    shown = 0
    while True:
        cd = await _details(ns)
        sys.stdout.write(det.render_table(
            cols, cd.table_rows(cols, role=role),
            header=not ns.omit_header))
        _print_issues(cd, sys.stderr, leading_nl=True)
        shown += 1
        if count is not None and shown >= count:
            break
        await asyncio.sleep(period)
    return 0


def _show_header(cd: det.ClusterDetails) -> None:
    """ref do_show: zookeeper/cluster/generation/mode/freeze lines."""
    print("zookeeper:   %s" % (cd.zk_conn or "-"))
    print("cluster:     %s" % cd.shard)
    print("generation:  %s (%s)" % (cd.generation, cd.init_wal))
    print("mode:        %s" % ("singleton (one-node-write)"
                               if cd.singleton else "normal"))
    if cd.freeze:
        print("freeze:      frozen since %s" % cd.freeze.get("date"))
        print("freeze info: %s" % cd.freeze.get("reason"))
    else:
        print("freeze:      not frozen")
    if cd.promote:
        print("promote:     pending for %s (expires %s)"
              % (cd.promote.get("id"), cd.promote.get("expireTime")))


async def cmd_show(ns) -> int:
    cd = await _details(ns)
    _show_header(cd)
    print()
    if ns.verbose:
        cols = det.PEERS_COLUMNS
        sys.stdout.write(det.render_table(cols, cd.table_rows(cols)))
        print()
    cols = det.STATUS_COLUMNS
    sys.stdout.write(det.render_table(cols, cd.table_rows(cols)))
    _print_issues(cd, sys.stdout, leading_nl=True)
    return 0


async def cmd_verify(ns) -> int:
    try:
        cd = await _details(ns)
    except Exception:
        print("error: failed to fetch cluster state")
        return 1
    _print_issues(cd, sys.stdout, leading_nl=False)
    if cd.errors or cd.warnings:
        return 1
    if ns.verbose:
        print("all checks passed")
    return 0


async def cmd_status(ns) -> int:
    """JSON summary (ref status lib/adm.js:997-1027): every shard when -s
    is not given, keyed by shard name."""
    fx = det.fixture_path()
    if fx:
        cd = det.load_fixture(fx)
        print(json.dumps({cd.shard: _status_obj(cd)}, indent=2,
                         sort_keys=True))
        return 0

    async def go(zk):
        shard = getattr(ns, "shard", None) or os.environ.get("SHARD")
        shards = [shard] if shard else await adm.get_shards(zk)
        out = {}
        for sh in shards:
            try:
                cd = await det.load_cluster_details(
                    zk, sh, zk_conn=ns.zk or os.environ.get("ZK_IPS", ""))
                out[sh] = _status_obj(cd)
            except adm.AdmError as exc:
                out[sh] = {"error": str(exc)}
        print(json.dumps(out, indent=2, sort_keys=True))
        return 0
    return await _with_zk(ns, go)


def _status_obj(cd: det.ClusterDetails) -> dict:
    """The reference's `status` shape (ref _formatState lib/adm.js:475-502
    + _addPostgresStatus :348-427): per-shard keys __FROZEN__ ("date:
    reason"), primary/sync, asyncN and deposedN (suffix empty for the
    first), each peer = its cluster-state identity + online/repl/lag/
    error from the live probes (deposed peers are not probed)."""
    def peer_obj(pid: str, probed: bool = True) -> dict:
        pd = cd.peers[pid]
        o = dict(pd.ident)
        if not probed:
            return o
        o["online"] = pd.online
        if pd.db_error:
            o["error"] = pd.db_error
        if pd.online:
            o["repl"] = pd.first_repl() or {}
            if pd.role != "primary":
                lag_s = pd.lag_s if isinstance(pd.lag_s, (int, float)) \
                    else None
                o["lag"] = {"time_lag": {
                    "minutes": int(lag_s // 60),
                    "seconds": int(lag_s % 60),
                }} if lag_s is not None else {}
        return o
    out = {}
    if cd.freeze:
        out["__FROZEN__"] = "%s: %s" % (cd.freeze.get("date"),
                                        cd.freeze.get("reason"))
    if cd.primary_id:
        out["primary"] = peer_obj(cd.primary_id)
    if cd.sync_id:
        out["sync"] = peer_obj(cd.sync_id)
    for i, aid in enumerate(cd.async_ids):
        out["async" + ("" if i == 0 else str(i))] = peer_obj(aid)
    for i, did in enumerate(cd.deposed_ids):
        out["deposed" + ("" if i == 0 else str(i))] = \
            peer_obj(did, probed=False)
    return out


async def cmd_zk_state(ns) -> int:
    fx = det.fixture_path()
    if fx:
        cd = det.load_fixture(fx)
        print(json.dumps(cd.state, separators=(",", ":")))
        return 0
    shard = _need(ns, "shard", "SHARD", "-s/--shard")

    async def go(zk):
        state, version = await adm.get_state(zk, shard)
        if state is None:
            return _fail("No state exists for shard " + shard)
        # the reference prints JSON.stringify(state): compact, key order
        # preserved, no extra fields (ref do_zk_state / zkState)
        print(json.dumps(state, separators=(",", ":")))
        return 0
    return await _with_zk(ns, go)


async def cmd_zk_active(ns) -> int:
    shard = _need(ns, "shard", "SHARD", "-s/--shard")

    async def go(zk):
        actives = await adm.get_active(zk, shard)
        print(json.dumps(actives, indent=2, sort_keys=True))
        return 0
    return await _with_zk(ns, go)


async def cmd_history(ns) -> int:
    """ref do_history bin/manatee-adm: table of TIME/G#/MODE/FRZ/
    PRIMARY/SYNC/ASYNC/DEPOSED (zoneId[:8]), -v adds SUMMARY (the
    legal-transition annotation), -j emits {zkSeq, time, state} JSON
    lines, -s sorts by zkSeq (default) or time."""
    if getattr(ns, "sort", "zkSeq") not in ("zkSeq", "time"):
        raise UsageError('-s / --sort must be one of "zkSeq" or "time"')
    shard = _need(ns, "shard", "SHARD", "-s/--shard")

    def abbr(ident) -> str:
        if not ident:
            return "-"
        return (ident.get("zoneId") or ident.get("id") or "-")[:8]

    async def go(zk):
        entries = await adm.get_history(zk, shard)
        annotated = adm.annotate_history(entries)
        if ns.sort == "time":
            annotated.sort(key=lambda e: e.get("time") or 0)
        if ns.json:
            for e in annotated:
                print(json.dumps(
                    {"zkSeq": e["zkSeq"],
                     "time": (st.iso8601(e["time"] / 1000.0)
                              if e.get("time") else None),
                     "state": e["state"]}))
            return 0
        cols = [("TIME", 24, "l"), ("G#", 2, "r"), ("MODE", 5, "l"),
                ("FRZ", 3, "l"), ("PRIMARY", 8, "l"), ("SYNC", 8, "l"),
                ("ASYNC", 8, "l"), ("DEPOSED", 8, "l")]
        if ns.verbose:
            cols.append(("SUMMARY", 0, "l"))

        def emit(row):
            cells = []
            for (label, width, align), val in zip(cols, row):
                text = str(val)
                if width:
                    text = (text.rjust(width) if align == "r"
                            else text.ljust(width))
                cells.append(text)
            print(" ".join(cells))

        emit([c[0] for c in cols])
        for e in annotated:
            s = e["state"]
            when = st.iso8601(e["time"] / 1000.0) if e.get("time") else "-"
            asyncs = ",".join(abbr(a) for a in s.get("async") or []) or "-"
            deposed = ",".join(abbr(d)
                               for d in s.get("deposed") or []) or "-"
            row = [when, s.get("generation", "-"),
                   "singl" if s.get("oneNodeWriteMode") else "multi",
                   "frz" if s.get("freeze") else "-",
                   abbr(s.get("primary")), abbr(s.get("sync")),
                   asyncs, deposed]
            if ns.verbose:
                summary = "; ".join(
                    list(e.get("notes") or []) +
                    ["VIOLATION: " + v for v in e.get("violations") or []])
                row.append(summary)
            emit(row)
        return 0
    return await _with_zk(ns, go)


async def cmd_freeze(ns) -> int:
    if not ns.reason:
        raise UsageError("freeze requires a reason (-r)")
    shard = _need(ns, "shard", "SHARD", "-s/--shard")

    async def go(zk):
        await adm.freeze(zk, shard, ns.reason)
        print("Frozen.")
        return 0
    return await _with_zk(ns, go)


async def cmd_unfreeze(ns) -> int:
    shard = _need(ns, "shard", "SHARD", "-s/--shard")

    async def go(zk):
        try:
            await adm.unfreeze(zk, shard)
        except adm.AdmError as exc:
            return _fail(str(exc))
        print("Unfrozen.")
        return 0
    return await _with_zk(ns, go)


async def cmd_reap(ns) -> int:
    shard = _need(ns, "shard", "SHARD", "-s/--shard")
    if not ns.id and not ns.zonename:
        raise UsageError("reap requires -i/--id or -n/--zonename")

    async def go(zk):
        try:
            await adm.reap(zk, shard, peer_id=ns.id, zonename=ns.zonename)
        except adm.AdmError as exc:
            return _fail(str(exc))
        print("Reaped.")
        return 0
    return await _with_zk(ns, go)


async def cmd_set_onwm(ns) -> int:
    shard = _need(ns, "shard", "SHARD", "-s/--shard")
    if ns.mode not in ("on", "off"):
        raise UsageError("set-onwm requires -m on|off")
    if not ns.yes:
        raise UsageError("set-onwm is dangerous; confirm with -y")

    async def go(zk):
        try:
            await adm.set_onwm(zk, shard, ns.mode)
        except adm.AdmError as exc:
            return _fail(str(exc))
        print("one-node-write mode: %s" % ns.mode)
        return 0
    return await _with_zk(ns, go)


async def cmd_state_backfill(ns) -> int:
    shard = _need(ns, "shard", "SHARD", "-s/--shard")
    if not ns.yes:
        raise UsageError("state-backfill rewrites cluster state; confirm "
                         "with -y")

    async def go(zk):
        try:
            state = await adm.state_backfill(zk, shard)
        except adm.AdmError as exc:
            return _fail(str(exc))
        print(json.dumps(state, indent=2, sort_keys=True))
        return 0
    return await _with_zk(ns, go)


async def cmd_check_lock(ns) -> int:
    """Exit 1 (with the lock's contents) if the named ZK path exists
    (ref checkLock lib/adm.js:2049-2061)."""
    if not ns.path:
        raise UsageError("check-lock requires -p/--path")

    async def go(zk):
        from ..coord import jute
        try:
            data, _ = await zk.get_data(ns.path)
        except jute.ZkError as exc:
            if exc.code == jute.ZNONODE:
                return 0
            raise
        print(data.decode("utf-8", "replace"))
        return 1
    return await _with_zk(ns, go)


async def cmd_promote(ns) -> int:
    role = ns.role or "sync"

    # pre-flight: cluster health + replication lag (ref promote
    # :1726-1850 — refuse when the cluster has issues or any peer lags
    # more than lagToIgnore, unless forced with -y)
    try:
        cd = await _details(ns)
    except Exception as exc:
        return _fail("cannot load cluster state: %s" % exc)
    warnings = ["cluster error: %s" % e for e in cd.errors]
    warnings += ["cluster warning: %s" % w for w in cd.warnings]
    for pd in cd.peers.values():
        if pd.lag_s is not None and pd.lag_s > ns.lag_to_ignore:
            warnings.append('"%s" has %ds of lag behind its upstream '
                            "peer" % (pd.label, int(pd.lag_s)))
    if warnings and not ns.yes:
        for w in warnings:
            print("warning: %s" % w, file=sys.stderr)
        return _fail("refusing to promote with outstanding warnings "
                     "(use -y to override, or -l to raise the lag "
                     "threshold)")
    shard = _need(ns, "shard", "SHARD", "-s/--shard")

    async def go(zk):
        try:
            state = await adm.request_promote(
                zk, shard, role=role, peer_id=ns.id, zonename=ns.zonename,
                async_index=ns.async_index)
        except adm.AdmError as exc:
            return _fail(str(exc))
        promote = state["promote"]
        print("promote requested: %s (%s), expires %s"
              % (promote["id"], promote["role"], promote["expireTime"]))
        # watch until consumed or expired (ref promote :1693-2014)
        deadline = time.monotonic() + 60
        while time.monotonic() < deadline:
            cur, _ = await adm.get_state(zk, shard)
            if cur is None:
                break
            if "promote" not in cur:
                if cur["generation"] != state["generation"]:
                    print("promotion complete (generation %s)"
                          % cur["generation"])
                else:
                    print("promote request consumed")
                return 0
            exp = cur["promote"].get("expireTime", "")
            if exp and exp < st.iso8601():
                print("promote request expired without being applied",
                      file=sys.stderr)
                return 1
            await asyncio.sleep(0.5)
        print("gave up waiting for the promotion", file=sys.stderr)
        return 1
    return await _with_zk(ns, go)


async def cmd_clear_promote(ns) -> int:
    shard = _need(ns, "shard", "SHARD", "-s/--shard")

    async def go(zk):
        try:
            await adm.clear_promote(zk, shard)
        except adm.AdmError as exc:
            return _fail(str(exc))
        print("Cleared.")
        return 0
    return await _with_zk(ns, go)


async def cmd_rebuild(ns) -> int:
    """Operator rebuild of THIS peer (ref rebuild lib/adm.js:1319-1684):
    refuse on the primary; require the peer's sitter to be stopped (its
    election ephemeral gone); isolate (or destroy, if deposed) the data
    store; remove the peer from the deposed list; restart the sitter
    (via --start-cmd) and watch the restore until the peer rejoins."""
    cfg_path = ns.config or os.environ.get("MANATEE_SITTER_CONFIG")
    if not cfg_path:
        raise UsageError("rebuild requires -c/--config (or "
                         "MANATEE_SITTER_CONFIG)")
    with open(cfg_path) as f:
        cfg = json.load(f)
    peer_id = "%s:%s:%s" % (cfg["ip"], cfg["postgresPort"],
                            cfg["backupPort"])
    shard = cfg["shardPath"]
    ns.zk = ns.zk or cfg["zkCfg"]["connStr"]
    session_ms = cfg["zkCfg"].get("opts", {}).get("sessionTimeout", 60000)
    store_cfg = cfg["postgresMgrCfg"]["storageCfg"]

    async def go(zk):
        state, _ = await adm.get_state(zk, shard)
        if state is None:
            return _fail("no cluster state for shard %r" % shard)
        if state.get("primary", {}).get("id") == peer_id:
            return _fail("refusing to rebuild the PRIMARY peer; promote "
                         "another peer first")
        deposed = any(d.get("id") == peer_id
                      for d in state.get("deposed") or [])
        if not ns.yes:
            return _fail("rebuild destroys this peer's local data%s; "
                         "confirm with -y"
                         % (" (peer is DEPOSED)" if deposed else ""))
        if ns.stop_cmd:
            proc = await asyncio.create_subprocess_shell(ns.stop_cmd)
            await proc.wait()
        # wait for the peer's ephemeral election node to expire
        # (ref :1433-1478 — up to 1.5x the session timeout)
        deadline = time.monotonic() + 1.5 * session_ms / 1000.0
        while True:
            actives = await adm.get_active(zk, shard)
            if not any(a["id"] == peer_id for a in actives):
                break
            if time.monotonic() > deadline:
                return _fail("peer %s still has a live ZK session; stop "
                             "its sitter first (or pass --stop-cmd)"
                             % peer_id)
            await asyncio.sleep(0.5)
        # isolate or destroy the local store
        from ..storage import open_store
        store = open_store(store_cfg, log=null_logger())
        if await store.exists():
            if deposed:
                await store.destroy()
                print("destroyed local store (peer was deposed)")
            else:
                moved = await store.isolate("autorebuild")
                print("isolated local store%s"
                      % (" to %s" % moved if moved else ""))
        if deposed:
            try:
                await adm.reap(zk, shard, peer_id=peer_id)
                print("removed %s from the deposed list" % peer_id)
            except adm.AdmError:
                pass
        if not ns.start_cmd:
            print("local state cleared; restart the sitter to restore "
                  "from the primary")
            return 0
        proc = await asyncio.create_subprocess_shell(ns.start_cmd)
        await proc.wait()
        # watch the peer's /restore progress + the cluster state until it
        # is back as sync/async (ref :1550-1678; ≤RESTORE_ATTEMPTS
        # restore failures before giving up)
        from ..common.httpd import http_request
        status_url = "http://%s:%d" % (cfg["ip"], cfg["postgresPort"] + 1)
        print("waiting for %s to restore and rejoin (watching %s/restore)"
              % (peer_id, status_url))
        deadline = time.monotonic() + ns.timeout
        last_pct = None
        failures = 0
        restore_active = False
        while time.monotonic() < deadline:
            cur, _ = await adm.get_state(zk, shard)
            if cur is not None:
                ids = [a.get("id") for a in cur.get("async") or []]
                if cur.get("sync"):
                    ids.append(cur["sync"].get("id"))
                if peer_id in ids:
                    print("peer %s rejoined the cluster" % peer_id)
                    return 0
            try:
                code, body = await http_request(status_url + "/restore",
                                                timeout_s=2)
                if code == 200 and isinstance(body, dict):
                    if body.get("active"):
                        restore_active = True
                        size = body.get("size") or 0
                        if size:
                            pct = int(100 * (body.get("completed") or 0)
                                      / size)
                            if pct != last_pct:
                                print("restore: %3d%% (%d/%d bytes)"
                                      % (pct, body.get("completed") or 0,
                                         size))
                                last_pct = pct
                    elif restore_active and body.get("failed"):
                        restore_active = False
                        failures += 1
                        print("restore attempt failed (%d/%d): %s"
                              % (failures, RESTORE_ATTEMPTS,
                                 body.get("error")), file=sys.stderr)
                        if failures >= RESTORE_ATTEMPTS:
                            return _fail("giving up after %d failed "
                                         "restore attempts" % failures)
            except Exception:
                pass   # status server not up yet
            await asyncio.sleep(1.0)
        return _fail("peer did not rejoin within %ds" % ns.timeout)
    return await _with_zk(ns, go)


# --------------------------------------------------------------- plumbing

def _mk_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(
        prog="manatee-adm",
        description="Administer a manatee shard (clean-room rebuild; "
                    "ref bin/manatee-adm)")
    sub = p.add_subparsers(dest="cmd")

    def add(name: str, fn, aliases=(), **kw):
        if name in COMMAND_HELP:
            # these subcommands print the reference's exact help text
            kw["add_help"] = False
        sp = sub.add_parser(name, aliases=list(aliases), **kw)
        sp.set_defaults(fn=fn, cmd_name=name)
        if name in COMMAND_HELP:
            sp.add_argument("-h", "--help", dest="want_help",
                            action="store_true")
        sp.add_argument("-z", "--zk", help="ZooKeeper connection string "
                        "(env ZK_IPS)")
        sp.add_argument("-s", "--shard", help="shard name (env SHARD)")
        return sp

    add("version", cmd_version, help="print the version")
    add("status", cmd_status, help="JSON summary of every shard")

    sp = add("peers", cmd_peers, help="table of peers")
    sp.add_argument("-H", "--omitHeader", dest="omit_header",
                    action="store_true")
    sp.add_argument("-o", "--columns", action="append")
    sp.add_argument("-r", "--role")

    sp = add("pg-status", cmd_pg_status, aliases=["db-status"],
             help="table of database status per peer")
    sp.add_argument("-H", "--omitHeader", dest="omit_header",
                    action="store_true")
    sp.add_argument("-o", "--columns", action="append")
    sp.add_argument("-r", "--role")
    sp.add_argument("-w", "--wide", "--cinematic", action="store_true")
    sp.add_argument("period", nargs="?", default=None)
    sp.add_argument("count", nargs="?", default=None)

    sp = add("show", cmd_show, help="summary + status table")
    sp.add_argument("-v", "--verbose", action="store_true")

    sp = add("verify", cmd_verify, help="health checks; exit 1 on issues")
    sp.add_argument("-v", "--verbose", action="store_true")

    add("zk-state", cmd_zk_state, aliases=["state"],
        help="raw cluster state JSON")
    add("zk-active", cmd_zk_active, aliases=["active"],
        help="live election members")

    sp = add("history", cmd_history,
             help="cluster state history with transition checks")
    sp.add_argument("-j", "--json", action="store_true")
    sp.add_argument("-v", "--verbose", action="store_true",
                    help="add the SUMMARY column (transition annotation)")
    sp.add_argument("-S", "--sort", default="zkSeq",
                    help='"zkSeq" (default) or "time"')

    sp = add("freeze", cmd_freeze, help="freeze cluster transitions")
    sp.add_argument("-r", "--reason")
    add("unfreeze", cmd_unfreeze, help="resume cluster transitions")

    sp = add("reap", cmd_reap, help="drop a peer from the deposed list")
    sp.add_argument("-i", "--id")
    sp.add_argument("-n", "--zonename")

    sp = add("set-onwm", cmd_set_onwm, help="toggle one-node-write mode")
    sp.add_argument("-m", "--mode", choices=("on", "off"))
    sp.add_argument("-y", "--yes", action="store_true")

    sp = add("state-backfill", cmd_state_backfill,
             help="synthesize v2 state from the election order")
    sp.add_argument("-y", "--yes", action="store_true")

    sp = add("check-lock", cmd_check_lock,
             help="exit 1 if the named ZK lock path exists")
    sp.add_argument("-p", "--path")

    sp = add("promote", cmd_promote, help="request a peer promotion")
    sp.add_argument("-i", "--id")
    sp.add_argument("-n", "--zonename")
    sp.add_argument("--role", choices=("sync", "async"))
    sp.add_argument("--asyncIndex", dest="async_index", type=int)
    sp.add_argument("-l", "--lagToIgnore", dest="lag_to_ignore",
                    type=int, default=60,
                    help="max acceptable replay lag in seconds")
    sp.add_argument("-y", "--yes", action="store_true",
                    help="promote despite warnings")

    add("clear-promote", cmd_clear_promote,
        help="remove a pending promote request")

    sp = add("rebuild", cmd_rebuild, help="rebuild THIS peer from the "
             "primary's backup")
    sp.add_argument("-c", "--config",
                    help="sitter config (env MANATEE_SITTER_CONFIG)")
    sp.add_argument("-y", "--yes", action="store_true")
    sp.add_argument("--stop-cmd", help="command to stop the local sitter")
    sp.add_argument("--start-cmd",
                    help="command to start the local sitter")
    sp.add_argument("--timeout", type=int, default=300)
    return p


def main(argv=None) -> int:
    parser = _mk_parser()
    ns = parser.parse_args(argv)
    if not getattr(ns, "fn", None):
        parser.print_help()
        return 2
    if getattr(ns, "want_help", False):
        sys.stdout.write(COMMAND_HELP[ns.cmd_name])
        return 0
    try:
        res = ns.fn(ns)
        if asyncio.iscoroutine(res):
            res = asyncio.run(res)
        return int(res or 0)
    except UsageError as exc:
        print("manatee-adm: %s" % exc, file=sys.stderr)
        if exc.help_text:
            sys.stderr.write(exc.help_text)
        return 2
    except KeyboardInterrupt:
        return 130
    except BrokenPipeError:
        return 0


if __name__ == "__main__":
    sys.exit(main())
