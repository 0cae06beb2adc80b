"""PostgreSQL ``key = value`` conf-file reader/writer.

Equivalent of the reference's ``lib/confParser.js`` (read via iniparser, set,
write; ``lib/confParser.js:31-57``).  Semantics preserved:

- files are flat ``key = value`` lines (no sections), ``#`` comments;
- writes regenerate the file from the in-memory dict — custom keys edited by
  hand into a live conf are lost on the next transition, exactly as the
  reference warns at ``lib/postgresMgr.js:2277-2281``;
- values are stored verbatim (including any quoting), so a caller that sets
  ``synchronous_standby_names`` to ``'"peername"'`` round-trips byte-for-byte.
"""

from __future__ import annotations

import os
import tempfile
from typing import Dict, Optional


def parse_string(text: str) -> Dict[str, str]:
    conf: Dict[str, str] = {}
    for raw in text.splitlines():
        line = raw.strip()
        if not line or line.startswith("#") or line.startswith(";"):
            continue
        if "=" in line:
            key, _, val = line.partition("=")
        else:
            # postgresql.conf allows "key value" with no '='
            parts = line.split(None, 1)
            if len(parts) != 2:
                continue
            key, val = parts
        key = key.strip()
        val = val.strip()
        # strip trailing same-line comment outside of quotes
        if val and val[0] not in "'\"":
            hash_at = val.find("#")
            if hash_at >= 0:
                val = val[:hash_at].rstrip()
        if key:
            conf[key] = val
    return conf


def read(path: str) -> Dict[str, str]:
    with open(path, "r") as f:
        return parse_string(f.read())


def get(conf: Dict[str, str], key: str) -> Optional[str]:
    return conf.get(key)


def set_value(conf: Dict[str, str], key: str, value: str) -> None:
    conf[key] = str(value)


def delete(conf: Dict[str, str], key: str) -> None:
    conf.pop(key, None)


def dump_string(conf: Dict[str, str]) -> str:
    return "".join("%s = %s\n" % (k, v) for k, v in conf.items())


def write(path: str, conf: Dict[str, str]) -> None:
    """Atomic write (tmp + rename) so a crash mid-write never leaves a torn
    conf — the reference relies on regenerating confs on every transition."""
    d = os.path.dirname(os.path.abspath(path))
    fd, tmp = tempfile.mkstemp(prefix=".conf.", dir=d)
    try:
        with os.fdopen(fd, "w") as f:
            f.write(dump_string(conf))
            f.flush()
            os.fsync(f.fileno())
        os.replace(tmp, path)
    except BaseException:
        try:
            os.unlink(tmp)
        except OSError:
            pass
        raise
