"""manatee-backupserver daemon (ref backupserver.js).

``python -m manatee_amd.daemons.backupserver -f backupserver.json [-v]``

Config (reference field names, etc/backupserver.json):
``{backupServerCfg: {port}, backupSenderCfg: {storageCfg | dataset/...}}``.
Wires one shared queue between the REST server and the sender
(ref backupserver.js:117-123).
"""

from __future__ import annotations

import asyncio
import signal
import sys

from ..backup.service import BackupQueue, BackupSender, BackupServer
from ..common.config import parse_daemon_args
from ..storage import open_store


async def run(cfg, log) -> int:
    store = open_store(cfg["backupSenderCfg"]["storageCfg"], log=log)
    queue = BackupQueue()
    server = BackupServer(cfg.get("ip", "0.0.0.0"),
                          cfg["backupServerCfg"]["port"], queue, log=log)
    sender = BackupSender(store, queue, log=log)
    await server.start()
    sender.start()
    stop = asyncio.Event()
    loop = asyncio.get_running_loop()
    for sig in (signal.SIGINT, signal.SIGTERM):
        loop.add_signal_handler(sig, stop.set)
    await stop.wait()
    await sender.stop()
    await server.stop()
    return 0


def main(argv=None) -> int:
    cfg, log, _ns = parse_daemon_args(argv if argv is not None
                                      else sys.argv[1:],
                                      "manatee-backupserver")
    return asyncio.run(run(cfg, log))


if __name__ == "__main__":
    sys.exit(main())
