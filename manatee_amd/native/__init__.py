"""Native (C++) codec core.

``from manatee_amd.native import codec`` yields the compiled extension
or None if it has not been built (callers fall back to the pure-Python
paths; __graft_entry__.build() compiles it in-tree)."""

try:
    from . import _codec as codec  # type: ignore[attr-defined]
except ImportError:    # not built yet — pure-Python fallbacks take over
    codec = None

__all__ = ["codec"]
