"""waldb — a small write-ahead-logged, streaming-replicated KV database.

The framework's built-in database engine: it exposes exactly the surface the
manager layer (ref lib/postgresMgr.js) needs from PostgreSQL — WAL positions
(LSNs), synchronous/cascading streaming replication, a pg_stat_replication
analogue, read-only gating, promote-with-timeline-bump, and dirty-kill crash
recovery — so the whole failover system can be exercised and benchmarked on
hosts with no PostgreSQL installed.  Real PostgreSQL remains a first-class
engine (``manatee_amd.db.postgres``).
"""
