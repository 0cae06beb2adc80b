"""SnapshotStore — the storage abstraction behind bootstrap/rebuild.

The reference is hard-wired to ZFS: datasets with snapshots named by
13-digit ms-epoch (lib/zfsClient.js:214-221, lib/snapShotter.js:146-151),
``zfs send | zfs recv`` streams for peer bootstrap (lib/backupSender.js:154-242
→ lib/zfsClient.js:765-886), and dataset isolation/rename instead of deletion
on rebuild (lib/zfsClient.js:514-624).

This build keeps those semantics behind an interface with two providers:

- ``ZfsStore``   — real ZFS via fork-exec, for hosts that have it;
- ``DirStore``   — plain-directory datasets with tar-stream snapshots, for
                   hosts (and CI) without ZFS.

Snapshot names remain 13-digit ms-epoch strings; rotation logic and the
backup protocol are provider-independent.
"""

from __future__ import annotations

import re
import time
from typing import AsyncIterator, List, Optional

AUTO_SNAPSHOT_RE = re.compile(r"^\d{13}$")


def snapshot_name_now() -> str:
    """13-digit ms epoch, e.g. '1426541061000' (ref snapshot naming,
    lib/zfsClient.js:218)."""
    return "%013d" % int(time.time() * 1000)


def is_auto_snapshot(name: str) -> bool:
    """Only 13-digit names are manatee-managed; operator snapshots are
    never touched by rotation (ref lib/snapShotter.js:206-272)."""
    return bool(AUTO_SNAPSHOT_RE.match(name))


class SnapshotStore:
    """Interface.  All methods are async; mountpoint() is where the live
    dataset contents (the database dataDir parent) are visible."""

    def mountpoint(self) -> str:
        raise NotImplementedError

    async def exists(self) -> bool:
        raise NotImplementedError

    async def ensure(self) -> None:
        """Create-if-missing and mount (ref mountDataset,
        lib/zfsClient.js:251-437)."""
        raise NotImplementedError

    async def snapshot(self, name: Optional[str] = None) -> str:
        raise NotImplementedError

    async def list_snapshots(self) -> List[str]:
        """Sorted ascending by name (oldest first)."""
        raise NotImplementedError

    async def destroy_snapshot(self, name: str) -> None:
        raise NotImplementedError

    async def send_size(self, name: str) -> int:
        """Approximate byte size of the stream for progress reporting
        (ref `zfs send -v -P` parsing, lib/backupSender.js:125-136)."""
        raise NotImplementedError

    async def send(self, name: str) -> AsyncIterator[bytes]:
        """Yield the serialized snapshot stream."""
        raise NotImplementedError

    async def recv(self, chunks: AsyncIterator[bytes]) -> None:
        """Consume a stream produced by send() into this (empty) dataset."""
        raise NotImplementedError

    async def isolate(self, reason: str = "autorebuild") -> Optional[str]:
        """Move the live dataset aside (never delete data on rebuild:
        ref isolateDataset lib/zfsClient.js:514-624).  Returns the
        isolated name or None if nothing existed."""
        raise NotImplementedError

    async def destroy(self) -> None:
        """Destroy the live dataset (only used for deposed peers during
        `manatee-adm rebuild`, ref lib/adm.js:1479-1532)."""
        raise NotImplementedError
