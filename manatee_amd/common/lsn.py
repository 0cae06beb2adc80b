"""WAL/xlog location (LSN) arithmetic and PG version helpers.

Equivalents of the reference's uses of the ``pg-lsn`` npm module
(``lib/postgresMgr.js`` compares sent/flush locations at 2390-2475) and of
``pgStripMinor`` (``lib/common.js:463-484``, tested table-driven in
``test/tst.common.js:15-76``).

An LSN is rendered ``HI/LO`` where HI and LO are hex words of a 64-bit byte
position in the WAL stream (e.g. ``0/174A4D0``, ``16/B374D848``).
"""

from __future__ import annotations

import re
from typing import Optional

_LSN_RE = re.compile(r"^([0-9A-Fa-f]{1,8})/([0-9A-Fa-f]{1,8})$")

ZERO = "0/00000000"


def parse(text: str) -> int:
    m = _LSN_RE.match(text.strip())
    if not m:
        raise ValueError("invalid LSN %r" % (text,))
    return (int(m.group(1), 16) << 32) | int(m.group(2), 16)


def format_lsn(value: int) -> str:
    if value < 0:
        raise ValueError("negative LSN")
    return "%X/%08X" % (value >> 32, value & 0xFFFFFFFF)


def is_lsn(text: str) -> bool:
    return bool(_LSN_RE.match(text.strip()))


def compare(a: str, b: str) -> int:
    """-1/0/1 as a </==/> b."""
    av, bv = parse(a), parse(b)
    return (av > bv) - (av < bv)


def diff_bytes(a: str, b: str) -> int:
    """a - b in WAL bytes."""
    return parse(a) - parse(b)


def max_lsn(a: str, b: str) -> str:
    return a if parse(a) >= parse(b) else b


def pg_strip_minor(version: str) -> Optional[str]:
    """Major version of a PG version string: '9.6.3' → '9.6', '12.0' → '12'.

    Matches the reference contract (lib/common.js:463-484): versions >= 10
    have single-part majors; pre-10 majors are two-part.  Returns None for
    garbage input rather than raising, like the reference's defensive use.
    """
    if not isinstance(version, str):
        return None
    parts = version.strip().split(".")
    if not parts or not parts[0].isdigit():
        return None
    major = int(parts[0])
    if major >= 10:
        return str(major)
    if len(parts) < 2 or not parts[1].isdigit():
        return None
    return "%d.%s" % (major, parts[1])
