"""DbManager — the database lifecycle manager (PostgresMgr equivalent).

Clean-room re-implementation of ``lib/postgresMgr.js``: owns the database
child process and performs the role transitions the FSM requests.

Preserved behaviors (reference citations inline):

- FSM-facing contract: events ``init {setup, online}``, ``healthy``,
  ``unhealthy``, fatal ``error``; methods ``reconfigure(cfg)``, ``stop()``,
  ``get_xlog_location()`` (ref :401-421, :686-928);
- transitions are serialized by a guard — one at a time (ref
  ``_transitioning`` :690-692, 850-852), and the background
  wait-for-standby task is cancellable (ref ``_transitionFunc``
  :379-385, 1123-1131);
- ``_primary``: prepare data dir (initdb analogue) → write conf
  **read-only** → restart → snapshot → background wait until the named
  sync standby is caught up, then set ``synchronous_standby_names`` +
  reload and open writes (ref :1115-1184, 1037-1105, 2390-2556);
- ``_update_standby``: a primary whose downstream changed only rewrites
  the sync name + SIGHUP — commits then block until the new sync catches
  up (ref :1195-1260);
- ``_standby``: stop → write upstream conf → restart; on failure or on WAL
  divergence, full restore from the restorePeer (always the primary — the
  back-pressure rule, ref :1019-1029, 1282-1460);
- stop is an escalating dirty kill (SIGINT → SIGQUIT → SIGKILL, one
  ops-timeout per step) and NEVER a clean shutdown — avoiding xlog
  divergence (ref :1484-1541, MANATEE-188);
- health: periodic ping with interval/timeout semantics
  (ref :1550-1626, 2373-2383); an unexpected child exit emits a fatal
  ``error`` event (ref :1711-1753).
"""

from __future__ import annotations

import asyncio
import os
import signal
import time
from typing import Any, Callable, Dict, List, Optional

from ..backup.restore import RestoreClient
from ..common import lsn as lsnmod
from ..common import procutil
from ..common.logging import Logger, null_logger
from ..storage.provider import SnapshotStore
from .engine import Engine, peer_id_from_urls


def db_child_preexec() -> None:
    """preexec_fn for the database child.

    SIGHUP is ignored until the db installs its own reload handler (the
    default disposition would kill the interpreter during boot).

    SIGINT/SIGQUIT/SIGTERM are reset to SIG_DFL: the spawning chain may
    have them at SIG_IGN (POSIX sets exactly that for commands
    backgrounded without job control), and ignored dispositions survive
    exec — an inheriting db child would be unkillable-except-SIGKILL
    until its event loop installs handlers, turning a dirty stop that
    lands in the boot/recovery window into a 2 x ops_timeout (= 60 s)
    stall that blocks the serialized FSM mid-failover (found by long
    chaos soaks as rare ~60 s failover outliers)."""
    signal.signal(signal.SIGHUP, signal.SIG_IGN)
    for _s in (signal.SIGINT, signal.SIGQUIT, signal.SIGTERM):
        signal.signal(_s, signal.SIG_DFL)


class DbManager:
    def __init__(self, *, engine: Engine, store: SnapshotStore,
                 ip: str,
                 health_interval_s: float = 1.0,
                 health_timeout_s: float = 5.0,
                 ops_timeout_s: float = 60.0,
                 replication_timeout_s: float = 60.0,
                 repl_poll_s: float = 0.2,
                 one_node_write_mode: bool = False,
                 log: Optional[Logger] = None):
        self.engine = engine
        self.store = store
        self.ip = ip
        self.health_interval_s = health_interval_s
        self.health_timeout_s = health_timeout_s
        self.ops_timeout_s = ops_timeout_s
        self.replication_timeout_s = replication_timeout_s
        self.repl_poll_s = repl_poll_s
        self.one_node_write_mode = one_node_write_mode
        self.log = (log or null_logger()).child(component="DbManager")

        self.healthy = False
        self.online = False
        self.writable = False
        self._proc: Optional[asyncio.subprocess.Process] = None
        self._proc_monitor: Optional[asyncio.Task] = None
        self._expect_exit = False
        self._lock = asyncio.Lock()          # _transitioning guard
        self._transition_task: Optional[asyncio.Task] = None
        self._health_task: Optional[asyncio.Task] = None
        self._listeners: Dict[str, List[Callable]] = {}
        self._applied: Optional[dict] = None
        self.restore_client: Optional[RestoreClient] = None
        self._closing = False

    # --------------------------------------------------------------- events
    def on(self, event: str, cb: Callable) -> None:
        self._listeners.setdefault(event, []).append(cb)

    def _emit(self, event: str, *args: Any) -> None:
        for cb in self._listeners.get(event, []):
            try:
                cb(*args)
            except Exception as exc:
                self.log.error("listener error", event=event, err=exc)

    # ------------------------------------------------------------ lifecycle
    async def start(self) -> None:
        """Probe setup/online and emit ``init`` (ref :401-421)."""
        self._health_task = asyncio.get_running_loop().create_task(
            self._health_loop())
        setup = self.engine.initialized()
        online = await self.engine.ping(timeout_s=1.0) if setup else False
        self.online = online
        self.healthy = online
        self._emit("init", {"setup": setup, "online": online})

    async def close(self) -> None:
        self._closing = True
        self._cancel_transition()
        if self._health_task is not None:
            self._health_task.cancel()
        await self._stop_db()
        await self.engine.close()

    # --------------------------------------------------------------- health
    async def _health_loop(self) -> None:
        self._last_health_ok = time.monotonic()
        while not self._closing:
            await asyncio.sleep(self.health_interval_s)
            if self._proc is None:
                self._last_health_ok = time.monotonic()
                continue
            ok = await self.engine.ping(timeout_s=self.health_timeout_s)
            now = time.monotonic()
            if ok:
                self._last_health_ok = now
                if not self.healthy:
                    self.healthy = True
                    self.online = True
                    self.log.info("database is healthy again")
                    self._emit("healthy")
            elif self.healthy and \
                    (now - self._last_health_ok) > self.health_timeout_s:
                self.healthy = False
                self.log.warn("database unhealthy",
                              since_s=now - self._last_health_ok)
                self._emit("unhealthy")

    # ------------------------------------------------------- process control
    def _pid_paths(self) -> List[str]:
        paths = []
        try:
            paths.append(os.path.join(self.engine.data_dir,
                                      "db_child.pid"))
        except AttributeError:
            pass
        try:
            paths.append(os.path.join(
                os.path.dirname(self.store.mountpoint()), "db_child.pid"))
        except Exception:
            pass
        return paths

    async def _start_db(self) -> None:
        if self._proc is not None:
            return
        self._expect_exit = False
        argv = self.engine.spawn_argv()
        self.log.info("starting database", argv=argv)
        self._proc = await asyncio.create_subprocess_exec(
            *argv, stdout=asyncio.subprocess.DEVNULL,
            stderr=asyncio.subprocess.DEVNULL,
            start_new_session=True, preexec_fn=db_child_preexec)
        # record the child pid IMMEDIATELY: the db writes its own pid file
        # only once it is up, and anything that needs to SIGKILL the whole
        # peer (tests, operators) must not race that window.  Written both
        # inside the data dir AND outside the dataset (the data dir is
        # REPLACED by restores, which would orphan the record and let an
        # old child survive a kill, squatting on the port)
        for pid_path in self._pid_paths():
            try:
                with open(pid_path, "w") as f:
                    f.write(str(self._proc.pid))
            except OSError:
                pass
        self._proc_monitor = asyncio.get_running_loop().create_task(
            self._monitor_proc(self._proc))
        # poll until the db answers (ref _start 1 Hz poll :1760-1794;
        # we poll faster to shrink failover time)
        deadline = time.monotonic() + self.ops_timeout_s
        while True:
            if self._proc is None or self._proc.returncode is not None:
                raise RuntimeError("database exited during startup")
            if await self.engine.ping(timeout_s=1.0):
                # verify the responder IS our child: an older incarnation
                # that escaped a kill can still hold the port, answer
                # pings, and leave our fresh child dead on a bind failure
                try:
                    status = await self.engine.status()
                except Exception:
                    await asyncio.sleep(0.05)
                    continue
                serving = status.get("pid")
                if serving in (None, self._proc.pid):
                    break
                self.log.error("a previous database incarnation still "
                               "holds the port; killing it",
                               old_pid=serving, new_pid=self._proc.pid)
                try:
                    os.kill(serving, signal.SIGKILL)
                except (ProcessLookupError, PermissionError):
                    pass
                # our own child likely died on the bind conflict; respawn
                if self._proc.returncode is not None:
                    self._proc = None
                    self._expect_exit = True
                    return await self._start_db()
            if time.monotonic() > deadline:
                raise RuntimeError("database did not become ready in %ss"
                                   % self.ops_timeout_s)
            await asyncio.sleep(0.05)
        self.online = True
        self.healthy = True
        self._last_health_ok = time.monotonic()

    async def _monitor_proc(self, proc: asyncio.subprocess.Process) -> None:
        rc = await proc.wait()
        if self._proc is proc:
            self._proc = None
        if not self._expect_exit and not self._closing:
            # unexpected death → fatal error event (ref :1711-1753)
            self.log.error("database exited unexpectedly", rc=rc)
            self.online = False
            self.healthy = False
            self.writable = False
            self._applied = None
            self._emit("error", RuntimeError("db exited rc=%s" % rc))

    async def _stop_db(self) -> None:
        proc = self._proc
        if proc is None:
            return
        self._expect_exit = True
        self._proc = None
        self.online = False
        self.healthy = False
        self.writable = False
        # drop any cached engine connection: it points at the process
        # being killed, and a stale-but-"connected" client would feed
        # EOFs to the first probes against the next incarnation
        try:
            await self.engine.close()
        except Exception:
            pass
        t0 = time.monotonic()
        sig_used = 0
        if proc.returncode is None:
            try:
                sig_used = await procutil.kill_escalate(
                    proc.pid, self.ops_timeout_s, pgid=True)
            except (procutil.ExecError, ProcessLookupError):
                pass
        t_kill = time.monotonic() - t0
        try:
            await asyncio.wait_for(proc.wait(), self.ops_timeout_s)
        except asyncio.TimeoutError:
            pass
        if t_kill > 2.0:
            # a dirty stop should land on the FIRST signal in
            # milliseconds — anything slower stalls the serialized FSM
            # and deserves a trace (the ~60 s failover-outlier class)
            self.log.warn("slow database stop", seconds=round(t_kill, 2),
                          signal=sig_used, pid=proc.pid)
        self.log.info("database stopped")

    async def _restart_db(self) -> None:
        await self._stop_db()
        await self._start_db()

    # ----------------------------------------------------------- transitions
    def _cancel_transition(self) -> None:
        if self._transition_task is not None:
            self._transition_task.cancel()
            self._transition_task = None

    async def reconfigure(self, cfg: dict) -> None:
        """Apply a role configuration from the FSM (contract ref :758-867)."""
        async with self._lock:
            self._cancel_transition()
            role = cfg.get("role")
            if role == "none":
                await self._stop_db()
                self._applied = cfg
                return
            if role == "primary":
                prev = self._applied
                if (prev and prev.get("role") == "primary"
                        and self._proc is not None):
                    await self._update_standby(cfg)
                else:
                    await self._primary(cfg)
            else:
                await self._standby(cfg)
            self._applied = cfg

    # ---------------------------------------------------------- primary path
    async def _primary(self, cfg: dict) -> None:
        """Full primary transition (ref _primary :1115-1184)."""
        downstream = cfg.get("downstream")
        self.log.info("transitioning to primary",
                      downstream=(downstream or {}).get("pgUrl"))
        was_standby = self.engine.current_conf_role() == "standby"
        onwm = self.one_node_write_mode or downstream is None
        if was_standby and self.online \
                and self._proc is not None \
                and self._proc.returncode is None:
            # ONLINE promote (the pg_ctl-promote discipline): the running
            # standby switches role in place on SIGHUP — no process
            # restart on the failover critical path
            self.engine.write_conf("primary", read_only=not onwm)
            self.engine.write_promote_trigger()
            self._reload_db()
            if await self._await_promoted(timeout_s=15.0):
                self.writable = bool(onwm)
                asyncio.get_running_loop().create_task(
                    self._background_snapshot())
                if not onwm:
                    self._transition_task = \
                        asyncio.get_running_loop().create_task(
                            self._wait_for_standby(downstream))
                return
            self.log.warn("online promote did not take; falling back to "
                          "a restart")
        await self._stop_db()
        if not self.engine.initialized():
            await self.store.ensure()
            await self.engine.init_datadir()
        was_standby = self.engine.current_conf_role() == "standby"
        # start read-only unless ONWM (ref :1145-1154); sync names are set
        # only after the standby has caught up (ref :1077-1085)
        self.engine.write_conf("primary", read_only=not onwm)
        if was_standby:
            self.engine.write_promote_trigger()
        await self._restart_db()
        self.writable = bool(onwm)
        # snapshot for future bootstraps (ref :1158) — in the background:
        # the reference's zfs snapshot is O(1), ours copies data, and the
        # failover-to-writable path must not wait on it
        asyncio.get_running_loop().create_task(self._background_snapshot())
        if not onwm:
            self._transition_task = asyncio.get_running_loop().create_task(
                self._wait_for_standby(downstream))

    async def _await_promoted(self, timeout_s: float = 5.0) -> bool:
        """Poll until the engine reports role=primary after an online
        promote reload."""
        deadline = time.monotonic() + timeout_s
        while time.monotonic() < deadline:
            try:
                status = await self.engine.status()
                if status.get("role") == "primary":
                    return True
            except Exception:
                pass
            await asyncio.sleep(0.02)
        return False

    async def _background_snapshot(self) -> None:
        try:
            await self.store.snapshot()
        except Exception as exc:
            self.log.warn("post-transition snapshot failed", err=exc)

    async def _update_standby(self, cfg: dict) -> None:
        """Downstream swap on a running primary: conf + SIGHUP only
        (ref _updateStandby :1195-1260).  Commits block until the new sync
        catches up; we additionally gate writes read-only until then so
        the writable flag is accurate."""
        downstream = cfg.get("downstream")
        onwm = self.one_node_write_mode or downstream is None
        self.log.info("updating standby config on running primary",
                      downstream=(downstream or {}).get("pgUrl"))
        if onwm:
            self.engine.write_conf("primary", read_only=False)
            self._reload_db()
            self.writable = True
            return
        self.engine.write_conf("primary", read_only=True)
        self._reload_db()
        self.writable = False
        self._transition_task = asyncio.get_running_loop().create_task(
            self._wait_for_standby(downstream))

    def _reload_db(self) -> None:
        if self._proc is not None and self._proc.returncode is None:
            try:
                os.kill(self._proc.pid, signal.SIGHUP)
            except ProcessLookupError:
                pass

    async def _wait_for_standby(self, downstream: dict) -> None:
        """Poll replication status until the sync standby has caught up,
        then set synchronous_standby_names + open writes
        (ref _waitForStandby :1037-1105 + _checkRepl :2390-2475)."""
        standby_name = peer_id_from_urls(downstream["pgUrl"],
                                         downstream["backupUrl"])
        my_proc = self._proc      # the db incarnation this gate belongs to
        deadline = time.monotonic() + self.replication_timeout_s
        last_progress: Optional[str] = None

        def stale() -> bool:
            # a cancelled-but-racing gate from a previous transition must
            # never reload/flip a NEWER db incarnation (a SIGHUP during
            # interpreter startup would kill it outright)
            return (self._transition_task is not asyncio.current_task()
                    or self._proc is not my_proc)
        try:
            while True:
                try:
                    repl = await self.engine.check_repl(standby_name)
                except Exception:
                    repl = {"connected": False, "caught_up": False}
                if stale():
                    return
                if repl.get("caught_up"):
                    break
                # forward progress resets the timeout (ref :2452-2460)
                flush = repl.get("write_lsn")
                if flush is not None and flush != last_progress:
                    last_progress = flush
                    deadline = time.monotonic() + self.replication_timeout_s
                if time.monotonic() > deadline:
                    # ref :2452-2460: the shard stays read-only on a
                    # catch-up timeout — but the gate must stay ARMED so
                    # writes open the moment the standby finally catches
                    # up (e.g. after it finishes a full restore), not
                    # only on the next topology change
                    self.log.error(
                        "standby did not catch up within the replication "
                        "timeout; shard stays read-only (still watching)",
                        standby=standby_name)
                    deadline = time.monotonic() + self.replication_timeout_s
                await asyncio.sleep(self.repl_poll_s)
            if stale():
                return
            self.engine.write_conf("primary", sync_name=standby_name,
                                   read_only=False)
            self._reload_db()
            self.writable = True
            self.log.info("sync standby caught up; writes enabled",
                          standby=standby_name)
            self._emit("writable")
        except asyncio.CancelledError:
            raise

    # ---------------------------------------------------------- standby path
    async def _standby(self, cfg: dict) -> None:
        """Sync/async transition (ref _standby :1282-1460)."""
        upstream = cfg["upstream"]
        restore_peer = cfg.get("restorePeer") or upstream
        role = cfg.get("role", "sync")
        self.log.info("transitioning to standby", role=role,
                      upstream=upstream.get("pgUrl"))
        await self._stop_db()
        self.writable = False
        need_restore = not self.engine.initialized()
        if not need_restore:
            self.engine.write_conf("standby",
                                   upstream_url=upstream["pgUrl"])
            try:
                await self._start_db()
            except Exception as exc:
                self.log.warn("standby failed to start; falling back to "
                              "restore", err=exc)
                need_restore = True
        if not need_restore:
            # wait briefly for streaming; divergence ⇒ restore.  The wait is
            # short — a long wait here would stall the FSM's event loop
            # during failover while the new primary restarts.
            verdict = await self._await_streaming(timeout_s=3.0)
            if verdict == "diverged":
                self.log.warn("WAL diverged from upstream; restoring")
                need_restore = True
            elif verdict == "disconnected":
                self.log.warn("upstream not reachable yet; watching "
                              "replication in the background")
                self._transition_task = \
                    asyncio.get_running_loop().create_task(
                        self._standby_watch())
        if need_restore:
            await self._restore_and_start(upstream, restore_peer)

    async def _standby_watch(self) -> None:
        """Background: if a standby that could not reach its upstream turns
        out to be DIVERGED once the upstream is back, force the FSM to
        re-issue the transition (which will restore)."""
        deadline = time.monotonic() + self.replication_timeout_s
        try:
            while time.monotonic() < deadline:
                await asyncio.sleep(0.5)
                try:
                    status = await self.engine.status()
                except Exception:
                    continue
                ustat = status.get("upstream_status")
                if ustat == "streaming":
                    return
                if ustat == "diverged":
                    # the FSM clears its applied config on db errors and
                    # re-issues the transition, which will restore
                    self.log.warn("divergence detected by watchdog; "
                                  "forcing re-transition")
                    self._emit("error",
                               RuntimeError("standby diverged from upstream"))
                    return
        except asyncio.CancelledError:
            raise

    async def _await_streaming(self, timeout_s: float) -> str:
        deadline = time.monotonic() + timeout_s
        while time.monotonic() < deadline:
            try:
                status = await self.engine.status()
            except Exception:
                await asyncio.sleep(0.2)
                continue
            ustat = status.get("upstream_status")
            if ustat == "streaming":
                return "streaming"
            if ustat == "diverged":
                return "diverged"
            await asyncio.sleep(0.2)
        return "disconnected"

    async def _restore_and_start(self, upstream: dict,
                                 restore_peer: dict) -> None:
        """Full restore from the restore peer's backup server, then start
        as a standby (ref zfsClient.restore → _standby retry :1375-1448)."""
        await self._stop_db()
        self.restore_client = RestoreClient(self.store, self.ip,
                                            log=self.log)
        # ONE attempt per reconfigure, like the reference's _standby
        # (ref :1375-1413): a failure propagates and the FSM re-issues the
        # transition on its next evaluation with a FRESH restorePeer —
        # critical when the restore peer itself just died in a failover.
        # Only data worth keeping is isolated; an empty/uninitialized
        # dataset is simply replaced.
        isolate = self.engine.initialized()
        await self.restore_client.restore(restore_peer["backupUrl"],
                                          isolate=isolate)
        self.engine.post_restore_fixup()
        self.engine.write_conf("standby", upstream_url=upstream["pgUrl"])
        await self._start_db()
        # short wait only: blocking here stalls the FSM's whole event
        # loop, and the upstream may itself be mid-failover — the
        # background watchdog re-triggers the transition if the standby
        # turns out diverged or the upstream stays unreachable
        verdict = await self._await_streaming(timeout_s=3.0)
        if verdict != "streaming":
            self.log.warn("standby not yet streaming after restore; "
                          "watching in the background", verdict=verdict)
            self._transition_task = \
                asyncio.get_running_loop().create_task(
                    self._standby_watch())

    # --------------------------------------------------------------- queries
    async def get_xlog_location(self) -> str:
        """Current WAL position (role-aware, ref getXLogLocation :878-899)."""
        return await self.engine.xlog()

    async def stop(self) -> None:
        async with self._lock:
            self._cancel_transition()
            await self._stop_db()

    def status(self) -> dict:
        return {
            "engine": self.engine.name,
            "online": self.online,
            "healthy": self.healthy,
            "writable": self.writable,
            "role": (self._applied or {}).get("role"),
            "appliedConfig": self._applied,
            "restore": (self.restore_client.restore_object.as_dict()
                        if self.restore_client else None),
        }
