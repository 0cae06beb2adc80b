"""manatee-snapshotter daemon (ref snapshotter.js).

``python -m manatee_amd.daemons.snapshotter -f snapshotter.json [-v]``

Config (reference field names, etc/snapshotter.json): ``{storageCfg,
pollInterval (ms, default 3600000), snapshotNumber (default 50),
statusUrl?}``.
"""

from __future__ import annotations

import asyncio
import signal
import sys

from ..common.config import parse_daemon_args
from ..snapshotter import SnapShotter
from ..storage import open_store


async def run(cfg, log) -> int:
    store = open_store(cfg["storageCfg"], log=log)
    snap = SnapShotter(
        store,
        poll_interval_s=cfg.get("pollInterval", 3600000) / 1000.0,
        snapshot_number=cfg.get("snapshotNumber", 50),
        health_url=cfg.get("statusUrl"),
        log=log)
    snap.start()
    stop = asyncio.Event()
    loop = asyncio.get_running_loop()
    for sig in (signal.SIGINT, signal.SIGTERM):
        loop.add_signal_handler(sig, stop.set)
    await stop.wait()
    await snap.stop()
    return 0


def main(argv=None) -> int:
    cfg, log, _ns = parse_daemon_args(argv if argv is not None
                                      else sys.argv[1:],
                                      "manatee-snapshotter")
    return asyncio.run(run(cfg, log))


if __name__ == "__main__":
    sys.exit(main())
