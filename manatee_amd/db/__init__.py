"""Database layer: the manager driving role transitions (ref
lib/postgresMgr.js) over pluggable engines — real PostgreSQL or the
built-in ``waldb`` replicated engine."""
