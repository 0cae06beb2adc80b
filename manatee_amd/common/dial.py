"""Outbound-connection layer with an address-rewrite map for network
fault injection.

The reference's chaos plan partitions the network between hosts with
``ipdadm`` (ref docs/test-plan.md:24-113).  This build runs every peer
on one host, where no netns/iptables is available, so partitions are
induced in USERSPACE instead: when ``MANATEE_DIAL_MAP`` names a JSON
file of ``{"host:port": "host:port"}`` rewrites, every outbound
connection this process makes to a mapped address is routed through the
mapped one — the test harness puts a ``tools.netproxy.LinkProxy`` there
and can then drop bytes per DIRECTED link (A→B independent of B→A),
which produces the asymmetric-partition shapes SIGSTOP cannot.

With the variable unset (production) this is exactly
``asyncio.open_connection``.
"""

from __future__ import annotations

import asyncio
import json
import os
from typing import Dict, Optional

_dial_map: Optional[Dict[str, str]] = None


def dial_map() -> Dict[str, str]:
    global _dial_map
    if _dial_map is None:
        path = os.environ.get("MANATEE_DIAL_MAP")
        if not path:
            _dial_map = {}
        else:
            try:
                with open(path) as f:
                    _dial_map = {str(k): str(v)
                                 for k, v in json.load(f).items()}
            except (OSError, ValueError):
                _dial_map = {}
    return _dial_map


def resolve(host: str, port: int) -> tuple:
    tgt = dial_map().get("%s:%d" % (host, int(port)))
    if tgt:
        h, _, p = tgt.rpartition(":")
        return h, int(p)
    return host, int(port)


async def open_connection(host: str, port: int, **kw):
    h, p = resolve(host, port)
    return await asyncio.open_connection(h, p, **kw)
