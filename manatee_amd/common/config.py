"""Daemon configuration loading.

Each daemon takes ``-f config.json`` plus ``-v`` verbosity flags
(ref sitter.js:50-94 parseOptions/readConfig).  Config is plain JSON; the
sitter propagates its logger and identity into sub-configs
(ref sitter.js:104-120).  Validation is JSON-schema based like the
reference's CONFIG_SCHEMA (lib/postgresMgr.js:60-116).
"""

from __future__ import annotations

import argparse
import json
from typing import Any, Optional, Tuple

from . import logging as mlog
from .schema import validate


def read_config(path: str, schema: Optional[dict] = None) -> dict:
    with open(path, "r") as f:
        cfg = json.load(f)
    if schema is not None:
        validate(cfg, schema)
    return cfg


def parse_daemon_args(argv, prog: str, description: str = "") -> Tuple[dict, "mlog.Logger", argparse.Namespace]:
    """Standard daemon CLI: ``prog -f config.json [-v ...]``.

    Returns (config, logger, namespace).  Verbosity stacks to TRACE like the
    reference (sitter.js:61-66).
    """
    ap = argparse.ArgumentParser(prog=prog, description=description)
    ap.add_argument("-f", "--file", required=True, help="config file (JSON)")
    ap.add_argument("-v", "--verbose", action="count", default=0,
                    help="verbose output; stack for more (-vv = trace)")
    ap.add_argument("--log-file", default=None, help="log to file instead of stderr")
    ns = ap.parse_args(argv)
    cfg = read_config(ns.file)
    level = mlog.level_from_verbosity(ns.verbose)
    log = mlog.Logger(prog, level=level, path=ns.log_file)
    return cfg, log, ns


def deep_get(cfg: dict, dotted: str, default: Any = None) -> Any:
    cur: Any = cfg
    for part in dotted.split("."):
        if not isinstance(cur, dict) or part not in cur:
            return default
        cur = cur[part]
    return cur
