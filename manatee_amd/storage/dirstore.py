"""DirStore — plain-directory dataset with tar-stream snapshots.

Provider for hosts without ZFS (including this CI image).  Layout:

    <base>/live/                  the dataset contents (db dataDir lives here)
    <base>/snapshots/<name>.tar   snapshots (13-digit ms-epoch names)
    <base>/isolated/<reason>-<date>/   datasets set aside by isolate()

Semantics mirror lib/zfsClient.js: isolation renames rather than deletes;
snapshots are point-in-time full copies; send/recv streams the snapshot
tarball.  tar runs via fork-exec with a scrubbed env, like every zfs
invocation in the reference (lib/common.js:148-172).
"""

from __future__ import annotations

import asyncio
import os
import shutil
import time
from typing import AsyncIterator, List, Optional

from ..common import procutil
from ..common.logging import Logger, null_logger
from .provider import SnapshotStore, is_auto_snapshot, snapshot_name_now


class DirStore(SnapshotStore):
    def __init__(self, base: str, log: Optional[Logger] = None):
        self.base = os.path.abspath(base)
        self.live = os.path.join(self.base, "live")
        self.snapdir = os.path.join(self.base, "snapshots")
        self.isodir = os.path.join(self.base, "isolated")
        self.log = (log or null_logger()).child(component="DirStore",
                                                base=self.base)

    def mountpoint(self) -> str:
        return self.live

    async def exists(self) -> bool:
        return os.path.isdir(self.live)

    async def ensure(self) -> None:
        os.makedirs(self.live, exist_ok=True)
        os.makedirs(self.snapdir, exist_ok=True)

    # ------------------------------------------------------------ snapshots
    def _snap_path(self, name: str) -> str:
        if "/" in name or name.startswith("."):
            raise ValueError("bad snapshot name %r" % name)
        return os.path.join(self.snapdir, name + ".tar")

    async def snapshot(self, name: Optional[str] = None) -> str:
        name = name or snapshot_name_now()
        await self.ensure()
        tmp = self._snap_path(name) + ".partial"
        # tar the live tree; -C so paths inside are relative.  rc 1 is
        # tar's "file changed as we read it" warning — expected while the
        # database appends to its WAL, and harmless: any WAL prefix is a
        # valid restore point (torn tails are truncated on replay).
        res = await procutil.run_async(
            ["tar", "-cf", tmp, "-C", self.live, "."],
            env=procutil.SCRUBBED_ENV, timeout=600, check=False)
        if res.returncode not in (0, 1):
            raise procutil.ExecError(res.argv, res.returncode, res.stdout,
                                     res.stderr)
        os.replace(tmp, self._snap_path(name))
        self.log.debug("snapshot created", snapshot=name)
        return name

    async def list_snapshots(self) -> List[str]:
        if not os.path.isdir(self.snapdir):
            return []
        out = [f[:-4] for f in os.listdir(self.snapdir)
               if f.endswith(".tar")]
        return sorted(out)

    async def destroy_snapshot(self, name: str) -> None:
        try:
            os.unlink(self._snap_path(name))
        except FileNotFoundError:
            pass

    async def send_size(self, name: str) -> int:
        return os.stat(self._snap_path(name)).st_size

    async def send(self, name: str) -> AsyncIterator[bytes]:
        path = self._snap_path(name)

        async def gen():
            loop = asyncio.get_running_loop()
            with open(path, "rb") as f:
                while True:
                    chunk = await loop.run_in_executor(None, f.read,
                                                       1 << 20)
                    if not chunk:
                        return
                    yield chunk
        return gen()

    async def recv(self, chunks: AsyncIterator[bytes]) -> None:
        """Receive into a STAGING directory and swap it into place only
        once the stream completed — a partial or late-cancelled stream
        must never become visible in the live dataset."""
        os.makedirs(self.base, exist_ok=True)
        staging = os.path.join(self.base,
                               ".recv-%d-%d" % (os.getpid(), time.time_ns()))
        os.makedirs(staging)
        proc = await asyncio.create_subprocess_exec(
            "tar", "-xf", "-", "-C", staging,
            stdin=asyncio.subprocess.PIPE,
            stdout=asyncio.subprocess.DEVNULL,
            stderr=asyncio.subprocess.PIPE,
            env=procutil.SCRUBBED_ENV)
        try:
            async for chunk in chunks:
                proc.stdin.write(chunk)
                await proc.stdin.drain()
            proc.stdin.close()
            rc = await proc.wait()
            if rc != 0:
                err = (await proc.stderr.read()).decode("utf-8", "replace")
                raise procutil.ExecError(["tar", "-xf"], rc, "", err)
            # atomic-ish swap: retire any current live dir, then rename
            if os.path.isdir(self.live):
                old = self.live + ".replaced-%d" % time.time_ns()
                os.rename(self.live, old)
                shutil.rmtree(old, ignore_errors=True)
            os.rename(staging, self.live)
        except BaseException:
            if proc.returncode is None:
                proc.kill()
                await proc.wait()
            shutil.rmtree(staging, ignore_errors=True)
            raise

    # ------------------------------------------------------------ isolation
    async def isolate(self, reason: str = "autorebuild") -> Optional[str]:
        if not os.path.isdir(self.live):
            return None
        os.makedirs(self.isodir, exist_ok=True)
        stamp = time.strftime("%Y-%m-%dT%H-%M-%SZ", time.gmtime())
        name = "%s-%s" % (reason, stamp)
        dest = os.path.join(self.isodir, name)
        i = 0
        while os.path.exists(dest):
            i += 1
            dest = os.path.join(self.isodir, "%s.%d" % (name, i))
        os.rename(self.live, dest)
        self.log.info("dataset isolated", to=dest)
        return dest

    async def destroy(self) -> None:
        if os.path.isdir(self.live):
            shutil.rmtree(self.live)
            self.log.info("dataset destroyed")
