"""waldb write-ahead log: append-only record file with crash-safe replay.

Record framing: ``u32 len | u32 crc32(payload) | payload`` (big-endian).
The LSN of a record is the byte offset of the END of its frame — so LSNs
are monotonically increasing byte positions in the WAL stream, rendered in
PostgreSQL's ``H/LLLLLLLL`` form by ``common.lsn``.

Replay truncates a torn tail (partial frame or bad crc) — the expected
state after a ``kill -9`` (the only way this system ever stops a database;
ref lib/postgresMgr.js:1484-1541 and MANATEE-188).
"""

from __future__ import annotations

import json
import os
import struct
import zlib
from typing import Callable, Iterator, Optional, Tuple

_HDR = struct.Struct(">II")


class Wal:
    def __init__(self, path: str):
        self.path = path
        self._fd: Optional[int] = None
        self.end = 0  # next append offset == current LSN

    # -------------------------------------------------------------- open
    def open(self, replay: Optional[Callable[[int, bytes], None]] = None
             ) -> int:
        """Open + replay; returns the recovered end LSN."""
        flags = os.O_RDWR | os.O_CREAT
        self._fd = os.open(self.path, flags, 0o600)
        size = os.fstat(self._fd).st_size
        pos = 0
        while pos + _HDR.size <= size:
            hdr = os.pread(self._fd, _HDR.size, pos)
            if len(hdr) < _HDR.size:
                break
            length, crc = _HDR.unpack(hdr)
            if length > 64 * 1024 * 1024 or pos + _HDR.size + length > size:
                break
            payload = os.pread(self._fd, length, pos + _HDR.size)
            if len(payload) != length or zlib.crc32(payload) != crc:
                break
            pos += _HDR.size + length
            if replay is not None:
                replay(pos, payload)
        if pos != size:
            os.ftruncate(self._fd, pos)  # torn tail from a dirty kill
        self.end = pos
        return pos

    def close(self) -> None:
        if self._fd is not None:
            os.close(self._fd)
            self._fd = None

    # ------------------------------------------------------------ append
    def append(self, payload: bytes) -> int:
        """Append one record; returns its commit LSN."""
        frame = _HDR.pack(len(payload), zlib.crc32(payload)) + payload
        os.pwrite(self._fd, frame, self.end)
        self.end += len(frame)
        return self.end

    def append_raw(self, data: bytes, at: int) -> int:
        """Standby path: append raw replicated WAL bytes at offset ``at``
        (must equal current end)."""
        if at != self.end:
            raise ValueError("non-contiguous WAL append (%d != %d)"
                             % (at, self.end))
        os.pwrite(self._fd, data, self.end)
        self.end += len(data)
        return self.end

    def fsync(self) -> None:
        os.fsync(self._fd)

    def truncate_to(self, lsn: int) -> None:
        os.ftruncate(self._fd, lsn)
        self.end = lsn

    # -------------------------------------------------------------- read
    def read(self, start: int, max_bytes: int = 1 << 20) -> bytes:
        """Raw WAL bytes [start, min(end, start+max_bytes)) for streaming."""
        n = min(self.end - start, max_bytes)
        if n <= 0:
            return b""
        return os.pread(self._fd, n, start)

    def iter_records(self, start: int = 0
                     ) -> Iterator[Tuple[int, bytes]]:
        """Yield (commit_lsn, payload) from offset ``start`` (must be a
        record boundary)."""
        pos = start
        while pos + _HDR.size <= self.end:
            hdr = os.pread(self._fd, _HDR.size, pos)
            length, crc = _HDR.unpack(hdr)
            payload = os.pread(self._fd, length, pos + _HDR.size)
            pos += _HDR.size + length
            yield pos, payload


def parse_frames(data: bytes) -> Iterator[Tuple[int, bytes]]:
    """Split a raw replicated chunk into (frame_len, payload) records.
    The chunk always contains whole frames (senders send record-aligned)."""
    pos = 0
    while pos + _HDR.size <= len(data):
        length, crc = _HDR.unpack_from(data, pos)
        payload = data[pos + _HDR.size:pos + _HDR.size + length]
        if len(payload) != length or zlib.crc32(payload) != crc:
            raise ValueError("corrupt replicated WAL frame")
        yield _HDR.size + length, payload
        pos += _HDR.size + length
    if pos != len(data):
        raise ValueError("non-record-aligned replicated chunk")


def encode_op(op: dict) -> bytes:
    return json.dumps(op, separators=(",", ":")).encode("utf-8")


def decode_op(payload: bytes) -> dict:
    return json.loads(payload.decode("utf-8"))
