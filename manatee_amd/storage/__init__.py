"""Snapshot/restore storage providers (ref: lib/zfsClient.js + the zfs
helpers in lib/common.js:148-451)."""

from .provider import SnapshotStore, snapshot_name_now, is_auto_snapshot
from .dirstore import DirStore

__all__ = ["SnapshotStore", "DirStore", "snapshot_name_now",
           "is_auto_snapshot", "open_store"]


def open_store(cfg: dict, log=None) -> SnapshotStore:
    """Construct a store from config: {provider: 'dir'|'zfs', ...}."""
    provider = cfg.get("provider", "dir")
    if provider == "dir":
        return DirStore(cfg["mountpoint"], log=log)
    if provider == "zfs":
        from .zfsstore import ZfsStore
        return ZfsStore(cfg["dataset"], cfg["mountpoint"],
                        zfs_path=cfg.get("zfsPath", "/sbin/zfs"), log=log)
    raise ValueError("unknown storage provider %r" % provider)
