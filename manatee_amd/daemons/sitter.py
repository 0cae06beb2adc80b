"""manatee-sitter daemon (ref sitter.js).

``python -m manatee_amd.daemons.sitter -f sitter.json [-v ...]``

Starts the Shard (coordination + db manager + FSM) and the StatusServer on
``postgresPort + 1`` (ref sitter.js:122-126).  On SIGINT/SIGTERM it exits
WITHOUT cleanly shutting down the database (ref sitter.js:130-155 —
deliberate, to avoid xlog divergence): the db child is killed dirty.
"""

from __future__ import annotations

import asyncio
import signal
import sys

from ..common.config import parse_daemon_args
from ..common.schema import validate
from ..shard import Shard, SITTER_CONFIG_SCHEMA
from ..status import StatusServer


async def run(cfg, log) -> int:
    validate(cfg, SITTER_CONFIG_SCHEMA)
    shard = Shard(cfg, log=log)
    status = StatusServer(cfg["ip"], cfg["postgresPort"] + 1, shard,
                          log=log)
    await status.start()
    await shard.start()

    stop = asyncio.Event()
    loop = asyncio.get_running_loop()
    for sig in (signal.SIGINT, signal.SIGTERM):
        loop.add_signal_handler(sig, stop.set)
    await stop.wait()
    log.info("sitter exiting (database killed dirty by design)")
    await status.stop()
    await shard.shutdown()
    return 0


def main(argv=None) -> int:
    cfg, log, _ns = parse_daemon_args(argv if argv is not None
                                      else sys.argv[1:], "manatee-sitter")
    return asyncio.run(run(cfg, log))


if __name__ == "__main__":
    sys.exit(main())
