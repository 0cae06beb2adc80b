"""ZfsStore — real ZFS datasets via fork-exec.

Mirrors the reference's zfs usage: env-scrubbed ``/sbin/zfs`` invocations
(lib/common.js:148-172), snapshot/rename/destroy/mount helpers
(lib/common.js:177-451), ``zfs send -v -P`` / ``zfs recv -u`` streaming
(lib/backupSender.js:172-227, lib/zfsClient.js:765-886), and isolation via
``zfs rename -p`` into ``<parent>/isolated/<reason>-<date>``
(lib/zfsClient.js:514-624).

Untestable in images without ZFS; tests gate on ``shutil.which('zfs')``.
"""

from __future__ import annotations

import asyncio
import time
from typing import AsyncIterator, List, Optional

from ..common import procutil
from ..common.logging import Logger, null_logger
from .provider import SnapshotStore, snapshot_name_now


class ZfsStore(SnapshotStore):
    def __init__(self, dataset: str, mountpoint: str,
                 zfs_path: str = "/sbin/zfs",
                 log: Optional[Logger] = None):
        self.dataset = dataset
        self._mountpoint = mountpoint
        self.zfs = zfs_path
        self.log = (log or null_logger()).child(component="ZfsStore",
                                                dataset=dataset)

    def mountpoint(self) -> str:
        return self._mountpoint

    async def _zfs(self, *args: str, check: bool = True,
                   timeout: float = 600.0) -> procutil.ExecResult:
        return await procutil.run_async([self.zfs] + list(args),
                                        env=procutil.SCRUBBED_ENV,
                                        timeout=timeout, check=check)

    async def exists(self) -> bool:
        res = await self._zfs("list", self.dataset, check=False)
        return res.returncode == 0

    async def ensure(self) -> None:
        if not await self.exists():
            await self._zfs("create", "-o",
                            "mountpoint=%s" % self._mountpoint, self.dataset)
        else:
            # verify mounted; mount if not (ref mountDataset :251-437)
            res = await self._zfs("get", "-H", "-o", "value", "mounted",
                                  self.dataset)
            if res.stdout.strip() != "yes":
                await self._zfs("set", "canmount=on", self.dataset)
                await self._zfs("set", "mountpoint=%s" % self._mountpoint,
                                self.dataset)
                await self._zfs("mount", self.dataset, check=False)

    async def snapshot(self, name: Optional[str] = None) -> str:
        name = name or snapshot_name_now()
        await self._zfs("snapshot", "%s@%s" % (self.dataset, name))
        return name

    async def list_snapshots(self) -> List[str]:
        res = await self._zfs("list", "-t", "snapshot", "-H", "-o", "name",
                              "-r", self.dataset, check=False)
        if res.returncode != 0:
            return []
        out = []
        for line in res.stdout.splitlines():
            line = line.strip()
            if "@" in line and line.startswith(self.dataset + "@"):
                out.append(line.split("@", 1)[1])
        return sorted(out)

    async def destroy_snapshot(self, name: str) -> None:
        await self._zfs("destroy", "%s@%s" % (self.dataset, name),
                        check=False)

    async def send_size(self, name: str) -> int:
        res = await self._zfs("send", "-nvP",
                              "%s@%s" % (self.dataset, name), check=False)
        for line in (res.stdout + res.stderr).splitlines():
            parts = line.split()
            if parts and parts[0] == "size":
                return int(parts[1])
        return 0

    async def send(self, name: str) -> AsyncIterator[bytes]:
        proc = await asyncio.create_subprocess_exec(
            self.zfs, "send", "%s@%s" % (self.dataset, name),
            stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.PIPE, env=procutil.SCRUBBED_ENV)

        async def gen():
            try:
                while True:
                    chunk = await proc.stdout.read(1 << 20)
                    if not chunk:
                        break
                    yield chunk
                rc = await proc.wait()
                if rc != 0:
                    err = (await proc.stderr.read()).decode("utf-8",
                                                            "replace")
                    raise procutil.ExecError([self.zfs, "send"], rc, "", err)
            finally:
                if proc.returncode is None:
                    proc.kill()
                    await proc.wait()
        return gen()

    async def recv(self, chunks: AsyncIterator[bytes]) -> None:
        # -u: don't mount on receive (ref zfs recv -v -u, zfsClient.js:787)
        proc = await asyncio.create_subprocess_exec(
            self.zfs, "recv", "-u", "-F", self.dataset,
            stdin=asyncio.subprocess.PIPE,
            stdout=asyncio.subprocess.DEVNULL,
            stderr=asyncio.subprocess.PIPE, env=procutil.SCRUBBED_ENV)
        try:
            async for chunk in chunks:
                proc.stdin.write(chunk)
                await proc.stdin.drain()
            proc.stdin.close()
            rc = await proc.wait()
            if rc != 0:
                err = (await proc.stderr.read()).decode("utf-8", "replace")
                raise procutil.ExecError([self.zfs, "recv"], rc, "", err)
            # post-receive: canmount=noauto + mountpoint + mount
            # (ref lib/zfsClient.js:152-183)
            await self._zfs("set", "canmount=noauto", self.dataset)
            await self._zfs("set", "mountpoint=%s" % self._mountpoint,
                            self.dataset)
            await self._zfs("inherit", "snapdir", self.dataset)
            await self._zfs("mount", self.dataset, check=False)
        except BaseException:
            if proc.returncode is None:
                proc.kill()
                await proc.wait()
            raise

    async def isolate(self, reason: str = "autorebuild") -> Optional[str]:
        if not await self.exists():
            return None
        parent = self.dataset.rsplit("/", 1)[0]
        stamp = time.strftime("%Y-%m-%dT%H-%M-%SZ", time.gmtime())
        target = "%s/isolated/%s-%s" % (parent, reason, stamp)
        await self._zfs("set", "canmount=off", self.dataset)
        await self._zfs("inherit", "mountpoint", self.dataset)
        # -p creates intermediate datasets (ref zfs rename -p :596-614)
        await self._zfs("rename", "-p", self.dataset, target)
        self.log.info("dataset isolated", to=target)
        return target

    async def destroy(self) -> None:
        if await self.exists():
            await self._zfs("destroy", "-r", self.dataset)
            self.log.info("dataset destroyed")
