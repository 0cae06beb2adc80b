"""manatee_amd — a clean-room, Python/C++ re-implementation of the
TritonDataCenter/manatee automated-failover system for replicated databases.

Capability map (reference file:line cited per module):

- ``common``   — logging / config / conf-file / fork-exec foundations
                 (ref: lib/common.js, lib/confParser.js, sitter.js)
- ``coord``    — ZooKeeper wire protocol client + embedded server, and the
                 coordination manager (ref: lib/zookeeperMgr.js)
- ``fsm``      — the cluster state machine, re-implemented natively
                 (ref: external dep ``manatee-state-machine``; contract
                 reconstructed at SURVEY.md §2.2)
- ``db``       — database-manager layer driving PostgreSQL or the built-in
                 ``waldb`` replicated engine (ref: lib/postgresMgr.js)
- ``storage``  — snapshot/restore providers: ZFS and plain-directory
                 (ref: lib/zfsClient.js, lib/common.js zfs helpers)
- ``backup``   — backup REST server / queue / sender (ref: lib/backupServer.js,
                 lib/backupQueue.js, lib/backupSender.js)
- ``adm``      — admin library + ``manatee-adm`` CLI (ref: lib/adm.js,
                 bin/manatee-adm)
- ``daemons``  — sitter / backupserver / snapshotter entrypoints
                 (ref: sitter.js, backupserver.js, snapshotter.js)

The on-ZooKeeper state format (``/<shardPath>/state``, ``election/``,
``history/``) and the ``manatee-adm`` output contract are kept compatible with
the reference (BASELINE.json).
"""

__version__ = "2.1.1-amd0"
