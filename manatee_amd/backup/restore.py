"""Restore client — bootstrap this peer's dataset from a remote peer.

The reference's zfsClient restore dance (lib/zfsClient.js:115-207, 765-886):

1. isolate the existing dataset under ``isolated/autorebuild-<date>``;
2. open a TCP *listen* socket locally;
3. ``POST /backup {host, port}`` to the peer's backup server — the peer
   connects back and streams the snapshot into our socket;
4. pipe the inbound stream into the store's receive path;
5. poll ``GET <jobPath>`` until ``done`` (or failed);
6. take an initial snapshot of the restored dataset.

Progress is tracked in a restore object that the status server exposes at
``GET /restore`` (ref lib/statusServer.js:112-121).
"""

from __future__ import annotations

import asyncio
import time
from typing import List, Optional

from ..common.httpd import http_request
from ..common.logging import Logger, null_logger
from ..storage.provider import SnapshotStore


class RestoreError(RuntimeError):
    pass


class RestoreObject:
    def __init__(self):
        self.active = False
        self.done = False
        self.failed = False
        self.error: Optional[str] = None
        self.size = 0
        self.completed = 0
        self.started_at: Optional[float] = None
        self.job_path: Optional[str] = None

    def as_dict(self) -> dict:
        return {"active": self.active, "done": self.done,
                "failed": self.failed, "error": self.error,
                "size": self.size, "completed": self.completed,
                "startedAt": self.started_at, "jobPath": self.job_path}


class RestoreClient:
    def __init__(self, store: SnapshotStore, listen_ip: str,
                 listen_port: int = 0, poll_interval_s: float = 1.0,
                 log: Optional[Logger] = None):
        self.store = store
        self.listen_ip = listen_ip
        self.listen_port = listen_port
        self.poll_interval_s = poll_interval_s
        self.log = (log or null_logger()).child(component="RestoreClient")
        self.restore_object = RestoreObject()

    async def restore(self, backup_url: str, isolate_reason: str = "autorebuild",
                      timeout_s: float = 3600.0, isolate: bool = True) -> None:
        """Full restore from the peer at backup_url (http://ip:port)."""
        ro = self.restore_object = RestoreObject()
        ro.active = True
        ro.started_at = time.time()
        self.log.info("starting restore", from_url=backup_url)
        try:
            if isolate:
                isolated = await self.store.isolate(isolate_reason)
                if isolated:
                    self.log.info("existing dataset isolated", to=isolated)
            await self.store.ensure()

            recv_done: asyncio.Future = asyncio.get_running_loop() \
                .create_future()
            conn_seen = asyncio.Event()
            conn_task: List[Optional[asyncio.Task]] = [None]
            abandoned = [False]

            async def on_conn(reader: asyncio.StreamReader,
                              writer: asyncio.StreamWriter):
                if conn_seen.is_set() or abandoned[0]:
                    # a late sender callback for a restore that has moved
                    # on must NEVER be allowed to touch the dataset
                    writer.close()
                    return
                conn_seen.set()
                conn_task[0] = asyncio.current_task()

                async def chunks():
                    while True:
                        chunk = await reader.read(1 << 20)
                        if not chunk:
                            return
                        ro.completed += len(chunk)
                        yield chunk
                try:
                    await self.store.recv(chunks())
                    if not recv_done.done():
                        recv_done.set_result(None)
                except Exception as exc:
                    if not recv_done.done():
                        recv_done.set_exception(exc)
                finally:
                    writer.close()

            server = await asyncio.start_server(on_conn, self.listen_ip,
                                                self.listen_port)
            port = server.sockets[0].getsockname()[1]
            try:
                # short timeout: an unresponsive restore peer (paused,
                # partitioned, dying) must fail fast so the FSM can
                # re-evaluate with a fresh restorePeer instead of
                # stalling a whole evaluation cycle
                status, resp = await http_request(
                    backup_url.rstrip("/") + "/backup", "POST",
                    {"host": self.listen_ip, "port": port}, timeout_s=10.0)
                if status != 200 or not isinstance(resp, dict):
                    raise RestoreError("backup request refused: %s %r"
                                       % (status, resp))
                ro.job_path = resp["jobPath"]
                job_url = backup_url.rstrip("/") + ro.job_path
                deadline = time.monotonic() + timeout_s
                # poll job status while the stream flows
                while True:
                    if recv_done.done():
                        recv_done.result()  # raises on stream failure
                    jstatus, job = await http_request(job_url,
                                                      timeout_s=10.0)
                    if jstatus == 200 and isinstance(job, dict):
                        ro.size = job.get("size", 0)
                        if job.get("failed"):
                            raise RestoreError("sender reported failure: %s"
                                               % job.get("error"))
                        if job.get("done"):
                            break
                    if time.monotonic() > deadline:
                        raise RestoreError("restore timed out")
                    await asyncio.sleep(self.poll_interval_s)
                # sender is done; wait for our receive side to finish
                await asyncio.wait_for(recv_done, 60)
            finally:
                # no receive may outlive this restore: a late-arriving or
                # still-running stream task writing into the dataset
                # AFTER we return would clobber whatever replaced it
                # (a newer restore, a regenerated conf, a running db)
                abandoned[0] = True
                server.close()
                task = conn_task[0]
                if task is not None and not task.done():
                    task.cancel()
                    try:
                        await task
                    except (asyncio.CancelledError, Exception):
                        pass
                await server.wait_closed()

            # initial snapshot of the restored dataset
            # (ref lib/zfsClient.js:177-183)
            await self.store.snapshot()
            ro.done = True
            self.log.info("restore complete", bytes=ro.completed)
        except BaseException as exc:
            ro.failed = True
            ro.error = repr(exc)
            self.log.error("restore failed", err=exc)
            raise
        finally:
            ro.active = False
