"""Fork-exec helpers and process-kill escalation.

Equivalent of the reference's ``lib/common.js`` fork-exec wrappers
(``zfsExecCommon`` env-scrubbing + 2 MB output buffer, ``lib/common.js:148-172``;
``replacefile/chown/chmod`` 22-138) and PostgresMgr's dirty-stop escalation
(SIGINT → SIGQUIT → SIGKILL, each step bounded by opsTimeout,
``lib/postgresMgr.js:1484-1541`` — deliberately *never* a clean checkpointed
shutdown, to avoid xlog divergence, MANATEE-188 / docs/xlog-diverge.md).
"""

from __future__ import annotations

import asyncio
import os
import shutil
import signal
import subprocess
import time
from typing import Dict, List, Optional, Sequence

# Reference scrubs the environment down to a minimal PATH before exec'ing
# /sbin/zfs (lib/common.js:153-158); we do the same for storage/db binaries.
SCRUBBED_ENV = {"PATH": "/usr/sbin:/usr/bin:/sbin:/bin:/usr/local/bin"}

MAX_OUTPUT = 2 * 1024 * 1024  # 2 MB, matching the reference's maxBuffer


class ExecError(RuntimeError):
    def __init__(self, argv: Sequence[str], returncode: Optional[int],
                 stdout: str, stderr: str, message: str = ""):
        self.argv = list(argv)
        self.returncode = returncode
        self.stdout = stdout
        self.stderr = stderr
        super().__init__(message or
                         "command %r failed (rc=%s): %s"
                         % (" ".join(self.argv), returncode, stderr.strip()[:512]))


class ExecResult:
    __slots__ = ("argv", "returncode", "stdout", "stderr", "duration_s")

    def __init__(self, argv, returncode, stdout, stderr, duration_s):
        self.argv = argv
        self.returncode = returncode
        self.stdout = stdout
        self.stderr = stderr
        self.duration_s = duration_s


def run(argv: Sequence[str], env: Optional[Dict[str, str]] = None,
        timeout: Optional[float] = None, check: bool = True,
        stdin_data: Optional[bytes] = None, cwd: Optional[str] = None) -> ExecResult:
    """Synchronous fork-exec with scrubbed env and bounded output."""
    t0 = time.monotonic()
    try:
        proc = subprocess.run(
            list(argv), env=env if env is not None else dict(os.environ),
            input=stdin_data, capture_output=True, timeout=timeout, cwd=cwd)
    except subprocess.TimeoutExpired as exc:
        raise ExecError(argv, None,
                        (exc.stdout or b"").decode("utf-8", "replace"),
                        (exc.stderr or b"").decode("utf-8", "replace"),
                        "command %r timed out after %ss" % (" ".join(argv), timeout))
    out = proc.stdout[:MAX_OUTPUT].decode("utf-8", "replace")
    err = proc.stderr[:MAX_OUTPUT].decode("utf-8", "replace")
    res = ExecResult(list(argv), proc.returncode, out, err, time.monotonic() - t0)
    if check and proc.returncode != 0:
        raise ExecError(argv, proc.returncode, out, err)
    return res


async def run_async(argv: Sequence[str], env: Optional[Dict[str, str]] = None,
                    timeout: Optional[float] = None, check: bool = True,
                    stdin_data: Optional[bytes] = None,
                    cwd: Optional[str] = None) -> ExecResult:
    """Async fork-exec; same contract as run()."""
    t0 = time.monotonic()
    proc = await asyncio.create_subprocess_exec(
        *argv, env=env if env is not None else dict(os.environ),
        stdin=asyncio.subprocess.PIPE if stdin_data is not None else asyncio.subprocess.DEVNULL,
        stdout=asyncio.subprocess.PIPE, stderr=asyncio.subprocess.PIPE, cwd=cwd)
    try:
        out_b, err_b = await asyncio.wait_for(proc.communicate(stdin_data), timeout)
    except asyncio.TimeoutError:
        try:
            proc.kill()
        except ProcessLookupError:
            pass
        await proc.wait()
        raise ExecError(argv, None, "", "",
                        "command %r timed out after %ss" % (" ".join(argv), timeout))
    out = out_b[:MAX_OUTPUT].decode("utf-8", "replace")
    err = err_b[:MAX_OUTPUT].decode("utf-8", "replace")
    res = ExecResult(list(argv), proc.returncode, out, err, time.monotonic() - t0)
    if check and proc.returncode != 0:
        raise ExecError(argv, proc.returncode, out, err)
    return res


def replace_file(path: str, data: str, mode: Optional[int] = None) -> None:
    """Atomic file replacement (ref lib/common.js:22-86 replacefile)."""
    tmp = "%s.tmp.%d" % (path, os.getpid())
    with open(tmp, "w") as f:
        f.write(data)
        f.flush()
        os.fsync(f.fileno())
    if mode is not None:
        os.chmod(tmp, mode)
    os.replace(tmp, path)


def chown_r(path: str, uid: int, gid: int) -> None:
    """Recursive chown (ref lib/common.js:91-115 fork-execs /usr/bin/chown -R)."""
    os.chown(path, uid, gid)
    for root, dirs, files in os.walk(path):
        for name in dirs + files:
            try:
                os.lchown(os.path.join(root, name), uid, gid)
            except FileNotFoundError:
                pass


async def kill_escalate(pid: int, ops_timeout_s: float,
                        signals: Sequence[int] = (signal.SIGINT, signal.SIGQUIT,
                                                  signal.SIGKILL),
                        poll_s: float = 0.1,
                        pgid: bool = False) -> int:
    """Dirty-stop a process: send each signal in turn, waiting up to
    ``ops_timeout_s`` for exit after each, escalating to SIGKILL
    (ref lib/postgresMgr.js:1484-1541).  Returns the signal that worked.
    The process must be a child we can probe with kill(pid, 0) — the caller
    is responsible for reaping (waitpid) its own children.
    """
    target = -pid if pgid else pid

    def alive() -> bool:
        try:
            os.kill(target, 0)
        except ProcessLookupError:
            return False
        except PermissionError:
            pass
        # a zombie (exited but unreaped by its parent) is dead for our purposes
        try:
            with open("/proc/%d/stat" % pid, "r") as f:
                stat = f.read()
            return stat.rpartition(")")[2].split()[0] != "Z"
        except (OSError, IndexError):
            return False

    def stopped() -> bool:
        try:
            with open("/proc/%d/stat" % pid, "r") as f:
                stat = f.read()
            return stat.rpartition(")")[2].split()[0] == "T"
        except (OSError, IndexError):
            return False

    for sig in signals:
        if not alive():
            return 0
        try:
            os.kill(target, sig)
            if sig != signal.SIGKILL and stopped():
                # a SIGSTOPped child QUEUES termination signals but
                # cannot run them — without a CONT the escalation
                # would burn its full per-step timeout before SIGKILL
                # (observed as rare ~60 s failovers when chaos froze a
                # db out from under a live sitter); wake it to die
                os.kill(target, signal.SIGCONT)
        except ProcessLookupError:
            return 0
        deadline = time.monotonic() + ops_timeout_s
        while time.monotonic() < deadline:
            if not alive():
                return sig
            if sig != signal.SIGKILL and stopped():
                # the child may stop AFTER the signal was sent (or the
                # stop may only become visible now); same rationale as
                # above — wake it so the queued signal can run
                try:
                    os.kill(target, signal.SIGCONT)
                except ProcessLookupError:
                    return sig
            await asyncio.sleep(poll_s)
    if alive():
        raise ExecError(["kill", str(pid)], None, "", "",
                        "process %d survived SIGKILL escalation" % pid)
    return signal.SIGKILL


def which(name: str) -> Optional[str]:
    return shutil.which(name, path=SCRUBBED_ENV["PATH"] + os.pathsep + os.environ.get("PATH", ""))
