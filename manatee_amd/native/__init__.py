"""Native (C++) extension cores.

``from manatee_amd.native import codec`` — WAL frame codec (crc32).
``from manatee_amd.native import jutec`` — jute wire-protocol codec.

Each is the compiled extension or None if it has not been built (callers
fall back to the pure-Python paths; __graft_entry__.build() compiles
both in-tree)."""

try:
    from . import _codec as codec  # type: ignore[attr-defined]
except ImportError:    # not built yet — pure-Python fallbacks take over
    codec = None

try:
    from . import _jutec as jutec  # type: ignore[attr-defined]
except ImportError:
    jutec = None

__all__ = ["codec", "jutec"]
