"""Daemon entrypoints (ref: sitter.js, backupserver.js, snapshotter.js)."""
