"""minipg — a PostgreSQL-shaped database engine process.

There is no PostgreSQL distribution (and no network to fetch one) in
this environment, so the ``engine=postgres`` management path
(db/postgres.py — conf regeneration, recovery.conf/standby.signal,
promote trigger files, initdb/postgres binaries, libpq probes,
pg_stat_replication gating; ref lib/postgresMgr.js) is proven against
THIS engine instead: a separate binary with PostgreSQL's process
conventions and wire protocol, backed by the waldb replication core.

What is postgres-faithful:
- ``initdb -D dir`` / ``postgres -D dir`` binaries (installed under
  ``<pgBaseDir>/<version>/bin`` by tools/devcluster);
- ``postgresql.conf`` / ``recovery.conf`` (pre-12) / ``standby.signal``
  (12+), ``primary_conninfo``, ``synchronous_standby_names``,
  ``default_transaction_read_only``, promote trigger files, SIGHUP
  reload, ``PG_VERSION``, ``postmaster.pid``, dirty-kill semantics;
- the libpq v3 wire protocol (startup/auth/simple query) serving the
  introspection surface the manager and adm use —
  ``pg_is_in_recovery()``, LSN functions, ``pg_stat_replication``,
  ``pg_last_xact_replay_timestamp()`` — plus a small SQL table surface
  (INSERT/SELECT/DELETE on ``kv``) for write-load benchmarks;
- synchronous_commit=remote_write semantics: an INSERT is acknowledged
  only after the named sync standby reports it written.

What is not: it is a KV store, not SQL PostgreSQL, and peer-to-peer
WAL shipping uses waldb's stream (multiplexed on the same port, as
PostgreSQL multiplexes replication connections on its port).
"""

from .server import MinipgServer, init_data_dir, main  # noqa: F401
