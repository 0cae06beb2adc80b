"""Backup service: job queue, REST server and snapshot sender.

Protocol kept compatible with the reference:

- ``POST /backup`` with ``{host, port, dataset?}`` enqueues a job and
  returns ``{jobid, jobPath}`` (ref lib/backupServer.js:100-155);
- ``GET /backup/<uuid>`` returns the job object with ``done``/``size``/
  ``completed`` progress fields (polled by the restoring peer,
  ref lib/zfsClient.js:685-754);
- the sender *connects back* to the requester's host:port and streams the
  latest auto-snapshot (13-digit ms-epoch name) over that raw TCP socket
  (ref lib/backupSender.js:154-242 picks the latest snapshot and pipes
  ``zfs send`` into the socket).
"""

from __future__ import annotations

import asyncio
import uuid as uuidmod
from typing import Dict, List, Optional

from ..common import dial

from ..common.httpd import HttpServer
from ..common.logging import Logger, null_logger
from ..storage.provider import SnapshotStore, is_auto_snapshot


class BackupJob:
    def __init__(self, host: str, port: int, dataset: str = ""):
        self.uuid = str(uuidmod.uuid4())
        self.host = host
        self.port = port
        self.dataset = dataset
        self.done = False
        self.failed = False
        self.error: Optional[str] = None
        self.size = 0
        self.completed = 0
        self.snapshot: Optional[str] = None

    def as_dict(self) -> dict:
        return {"uuid": self.uuid, "host": self.host, "port": self.port,
                "dataset": self.dataset, "done": self.done,
                "failed": self.failed, "error": self.error,
                "size": self.size, "completed": self.completed,
                "snapshot": self.snapshot}


class BackupQueue:
    """FIFO of backup jobs (ref lib/backupQueue.js)."""

    def __init__(self):
        self._jobs: List[BackupJob] = []
        self._by_uuid: Dict[str, BackupJob] = {}
        self._waiters: "asyncio.Queue[BackupJob]" = asyncio.Queue()

    def push(self, job: BackupJob) -> None:
        self._jobs.append(job)
        self._by_uuid[job.uuid] = job
        self._waiters.put_nowait(job)

    def get(self, uuid: str) -> Optional[BackupJob]:
        return self._by_uuid.get(uuid)

    async def next_job(self) -> BackupJob:
        return await self._waiters.get()


class BackupSender:
    """Streams the latest auto snapshot to the requester
    (ref lib/backupSender.js:154-242)."""

    def __init__(self, store: SnapshotStore, queue: BackupQueue,
                 log: Optional[Logger] = None,
                 stall_timeout_s: float = 120.0):
        self.store = store
        self.queue = queue
        self.log = (log or null_logger()).child(component="BackupSender")
        # a receiver that stops reading (partitioned mid-restore, dead
        # process with the TCP half open) must fail the job, not wedge
        # the SERIAL sender queue forever
        self.stall_timeout_s = stall_timeout_s
        self._task: Optional[asyncio.Task] = None

    def start(self) -> None:
        self._task = asyncio.get_running_loop().create_task(self._run())

    async def stop(self) -> None:
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except (asyncio.CancelledError, Exception):
                pass

    async def _run(self) -> None:
        while True:
            job = await self.queue.next_job()
            try:
                await self._send(job)
            except asyncio.CancelledError:
                raise
            except Exception as exc:
                job.failed = True
                job.done = True
                job.error = repr(exc)
                self.log.error("backup job failed", jobid=job.uuid, err=exc)

    async def _send(self, job: BackupJob) -> None:
        snaps = [s for s in await self.store.list_snapshots()
                 if is_auto_snapshot(s)]
        if not snaps:
            # no snapshot yet: take one now so bootstrap always works
            job.snapshot = await self.store.snapshot()
        else:
            job.snapshot = snaps[-1]
        job.size = await self.store.send_size(job.snapshot)
        self.log.info("sending snapshot", jobid=job.uuid,
                      snapshot=job.snapshot, to="%s:%d" % (job.host,
                                                           job.port),
                      size=job.size)
        reader, writer = await asyncio.wait_for(
            dial.open_connection(job.host, job.port), 30)
        try:
            stream = await self.store.send(job.snapshot)
            async for chunk in stream:
                writer.write(chunk)
                try:
                    await asyncio.wait_for(writer.drain(),
                                           self.stall_timeout_s)
                except asyncio.TimeoutError:
                    raise RuntimeError(
                        "receiver stalled: no progress in %.0fs"
                        % self.stall_timeout_s)
                job.completed += len(chunk)
            writer.write_eof()
            await asyncio.wait_for(writer.drain(), self.stall_timeout_s)
            job.done = True
            self.log.info("backup job complete", jobid=job.uuid,
                          bytes=job.completed)
        finally:
            writer.close()


class BackupServer:
    """REST server (ref lib/backupServer.js)."""

    def __init__(self, host: str, port: int, queue: BackupQueue,
                 log: Optional[Logger] = None):
        self.queue = queue
        self.log = (log or null_logger()).child(component="BackupServer")
        self.http = HttpServer(host, port, log=self.log)
        self.http.route("POST", "backup", self._post_backup)
        self.http.route("GET", "backup", self._get_backup)

    @property
    def port(self) -> int:
        return self.http.port

    async def start(self) -> None:
        await self.http.start()

    async def stop(self) -> None:
        await self.http.stop()

    async def _post_backup(self, parts, body):
        if not body or "host" not in body or "port" not in body:
            return 400, {"error": "host and port are required"}
        job = BackupJob(str(body["host"]), int(body["port"]),
                        str(body.get("dataset", "")))
        self.queue.push(job)
        self.log.info("backup requested", jobid=job.uuid,
                      host=job.host, port=job.port)
        return 200, {"jobid": job.uuid, "jobPath": "/backup/" + job.uuid}

    async def _get_backup(self, parts, body):
        if len(parts) < 2:
            return 400, {"error": "job uuid required"}
        job = self.queue.get(parts[1])
        if job is None:
            return 404, {"error": "no such job"}
        return 200, job.as_dict()
