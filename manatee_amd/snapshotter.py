"""SnapShotter — periodic snapshot + rotation daemon logic.

Ref lib/snapShotter.js: every ``pollInterval`` check the sitter's
``GET /ping`` (skip the snapshot while the database is unhealthy, :125-145),
then snapshot the dataset with a 13-digit ms-epoch name (:146-151, 445-473).
An independent cleanup pass keeps at most ``snapshotNumber`` auto snapshots,
never touching operator snapshots (non-13-digit names, :206-272), and
escalates loudly if deletions get stuck (:274-405).
"""

from __future__ import annotations

import asyncio
from typing import Optional

from .common.httpd import http_request
from .common.logging import Logger, null_logger
from .storage.provider import SnapshotStore, is_auto_snapshot


class SnapShotter:
    def __init__(self, store: SnapshotStore, *,
                 poll_interval_s: float = 3600.0,
                 snapshot_number: int = 50,
                 health_url: Optional[str] = None,
                 log: Optional[Logger] = None):
        self.store = store
        self.poll_interval_s = poll_interval_s
        self.snapshot_number = snapshot_number
        self.health_url = health_url  # e.g. http://ip:statusPort/ping
        self.log = (log or null_logger()).child(component="SnapShotter")
        self._task: Optional[asyncio.Task] = None
        self._cleanup_failures = 0
        self.stats = {"snapshots": 0, "skipped_unhealthy": 0, "deleted": 0}

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
            try:
                await self.run_once()
            except asyncio.CancelledError:
                raise
            except Exception as exc:
                self.log.error("snapshot pass failed", err=exc)
            await asyncio.sleep(self.poll_interval_s)

    async def run_once(self) -> Optional[str]:
        """One snapshot+cleanup pass; returns the snapshot name or None if
        skipped."""
        if self.health_url is not None:
            healthy = False
            try:
                status, _ = await http_request(self.health_url, timeout_s=10)
                healthy = status == 200
            except Exception:
                healthy = False
            if not healthy:
                self.stats["skipped_unhealthy"] += 1
                self.log.warn("database unhealthy; skipping snapshot")
                await self._cleanup()
                return None
        name = await self.store.snapshot()
        self.stats["snapshots"] += 1
        self.log.info("snapshot taken", snapshot=name)
        await self._cleanup()
        return name

    async def _cleanup(self) -> None:
        try:
            snaps = [s for s in await self.store.list_snapshots()
                     if is_auto_snapshot(s)]
            excess = len(snaps) - self.snapshot_number
            for name in snaps[:max(excess, 0)]:
                await self.store.destroy_snapshot(name)
                self.stats["deleted"] += 1
                self.log.debug("snapshot rotated out", snapshot=name)
            self._cleanup_failures = 0
        except Exception as exc:
            self._cleanup_failures += 1
            level = self.log.fatal if self._cleanup_failures > 5 \
                else self.log.error
            level("snapshot cleanup failing", failures=self._cleanup_failures,
                  err=exc)
