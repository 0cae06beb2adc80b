"""waldb write-ahead log: segmented append-only record log with
crash-safe replay and checkpoint-driven truncation.

Record framing: ``u32 len | u32 crc32(payload) | payload`` (big-endian).
The LSN of a record is the byte offset of the END of its frame in the
*logical WAL stream* — monotonically increasing positions rendered in
PostgreSQL's ``H/LLLLLLLL`` form by ``common.lsn``.

The stream is stored as fixed-size-ish segment files
``wal-<%016x start>.seg`` so that old WAL can be dropped (whole
segments at a time) once a checkpoint covers it — the PostgreSQL
checkpoint + WAL-recycling discipline.  Only the LAST segment can have
a torn tail (partial frame / bad crc), which is truncated on open —
the expected state after ``kill -9`` (the only way this system ever
stops a database; ref lib/postgresMgr.js:1484-1541 and MANATEE-188).

``start`` is the lowest LSN still on disk; a replica asking for older
WAL must fall back to a full restore (``wal-gone``), exactly like a
PostgreSQL standby that outlived the primary's WAL retention.
"""

from __future__ import annotations

import json
import os
import re
import struct
import zlib
from typing import Callable, Iterator, List, Optional, Tuple

from ... import native as _native_pkg


def _codec():
    """The C++ codec module, or None.

    Resolved through the package attribute at call time (not bound at
    import) so a build that happens after this module was first imported
    — e.g. the test fixture compiling the extension in a fresh checkout,
    then ``importlib.reload(manatee_amd.native)`` — is picked up."""
    return _native_pkg.codec

_HDR = struct.Struct(">II")
_SEG_RE = re.compile(r"^wal-([0-9a-f]{16})\.seg$")

DEFAULT_SEGMENT_BYTES = 16 * 1024 * 1024
LEGACY_NAME = "wal.log"          # pre-segmentation single-file layout


def _seg_name(start: int) -> str:
    return "wal-%016x.seg" % start


class Wal:
    """Segmented WAL living in a directory.

    ``path`` may be either the directory itself or the legacy
    ``<dir>/wal.log`` path (the file is adopted as the first segment)."""

    def __init__(self, path: str, segment_bytes: int = DEFAULT_SEGMENT_BYTES):
        if path.endswith(LEGACY_NAME):
            path = os.path.dirname(path)
        self.dir = path
        self.segment_bytes = segment_bytes
        self._fd: Optional[int] = None       # active (last) segment fd
        self._segs: List[int] = []           # sorted segment start LSNs
        self._active_start = 0
        self.start = 0   # lowest LSN still on disk
        self.end = 0     # next append offset == current LSN
        self._buf = bytearray()              # append_buffered frames
        self._buf_at = 0                     # file LSN of _buf[0]

    # -------------------------------------------------------------- open
    def _seg_path(self, start: int) -> str:
        return os.path.join(self.dir, _seg_name(start))

    def _list_segments(self) -> List[int]:
        out = []
        for name in os.listdir(self.dir):
            m = _SEG_RE.match(name)
            if m:
                out.append(int(m.group(1), 16))
        return sorted(out)

    def open(self, replay: Optional[Callable[[int, bytes], None]] = None,
             replay_from: int = 0) -> int:
        """Open + replay records with commit LSN > ``replay_from``;
        returns the recovered end LSN.  Segments entirely below
        ``replay_from`` are trusted (they were fsynced before the
        checkpoint that covers them); only the tail is crc-validated."""
        os.makedirs(self.dir, exist_ok=True)
        legacy = os.path.join(self.dir, LEGACY_NAME)
        if os.path.exists(legacy):
            os.rename(legacy, self._seg_path(0))
        self._segs = self._list_segments()
        if not self._segs:
            self._segs = [replay_from]
            with open(self._seg_path(replay_from), "wb"):
                pass
        self.start = self._segs[0]
        # validate + replay from the segment containing replay_from
        pos = self.start
        end = self.start
        for i, seg_start in enumerate(self._segs):
            is_last = i == len(self._segs) - 1
            path = self._seg_path(seg_start)
            size = os.path.getsize(path)
            if not is_last and seg_start + size <= replay_from:
                end = seg_start + size
                continue
            fd = os.open(path, os.O_RDWR)
            try:
                buf = os.pread(fd, size, 0)
                valid, records = _scan(buf)
                pos = seg_start + valid
                if replay is not None:
                    for off, length in records:
                        commit = seg_start + off + length
                        if commit > replay_from:
                            replay(commit, buf[off:off + length])
                if pos != seg_start + size:
                    if is_last:
                        os.ftruncate(fd, pos - seg_start)  # torn tail
                        end = pos
                        break
                    raise IOError(
                        "corrupt interior WAL segment %s (valid to %d of "
                        "%d)" % (path, pos - seg_start, size))
                end = pos
            finally:
                os.close(fd)
        self.end = end
        self._open_active()
        return self.end

    def _open_active(self) -> None:
        if self._fd is not None:
            os.close(self._fd)
        self._active_start = self._segs[-1]
        self._fd = os.open(self._seg_path(self._active_start),
                           os.O_RDWR | os.O_CREAT, 0o600)

    def close(self) -> None:
        if self._fd is not None:
            self.flush_buffer()
            os.close(self._fd)
            self._fd = None

    # ------------------------------------------------------------ append
    def _maybe_roll(self) -> None:
        if self.end - self._active_start >= self.segment_bytes:
            self.flush_buffer()
            os.fsync(self._fd)
            self._segs.append(self.end)
            with open(self._seg_path(self.end), "wb"):
                pass
            self._open_active()

    def flush_buffer(self) -> None:
        """Write frames accumulated by ``append_buffered`` to the active
        segment in ONE pwrite.  Every path that observes file contents
        (read, fsync, roll, truncate, close) calls this first, so the
        buffer is never visible as a gap."""
        if self._buf:
            os.pwrite(self._fd, bytes(self._buf),
                      self._buf_at - self._active_start)
            self._buf.clear()

    def append_buffered(self, payload: bytes) -> int:
        """``append()`` without the per-record pwrite: frames accumulate
        in memory and hit the file at the next ``flush_buffer()``.  The
        server flushes before any socket write and before fsync, and
        replica reads flush implicitly (``read``), so observable
        durability is unchanged — an ack never rests on bytes that are
        only in this buffer."""
        self._maybe_roll()
        native = _codec()
        if native is not None:
            frame = native.encode_frame(payload)
        else:
            frame = _HDR.pack(len(payload), zlib.crc32(payload)) + payload
        if not self._buf:
            self._buf_at = self.end
        self._buf += frame
        self.end += len(frame)
        return self.end

    def append(self, payload: bytes) -> int:
        """Append one record; returns its commit LSN."""
        lsn = self.append_buffered(payload)
        self.flush_buffer()
        return lsn

    def append_raw(self, data: bytes, at: int) -> int:
        """Standby path: append raw replicated WAL bytes at offset ``at``
        (must equal current end)."""
        if at != self.end:
            raise ValueError("non-contiguous WAL append (%d != %d)"
                             % (at, self.end))
        self.flush_buffer()
        self._maybe_roll()
        os.pwrite(self._fd, data, self.end - self._active_start)
        self.end += len(data)
        return self.end

    def fsync(self) -> None:
        self.flush_buffer()
        os.fsync(self._fd)

    # -------------------------------------------------------- truncation
    def truncate_to(self, lsn: int) -> None:
        """Timeline fencing: discard everything above ``lsn``."""
        self.flush_buffer()
        keep = [s for s in self._segs if s <= lsn]
        drop = [s for s in self._segs if s > lsn]
        for s in drop:
            os.unlink(self._seg_path(s))
        if not keep:
            keep = [lsn]
            with open(self._seg_path(lsn), "wb"):
                pass
        self._segs = keep
        self._open_active()
        os.ftruncate(self._fd, max(0, lsn - self._active_start))
        self.end = lsn
        self.start = self._segs[0]

    def drop_below(self, lsn: int) -> int:
        """Checkpoint-driven recycling: delete whole segments whose every
        byte is < ``lsn``.  Returns the new ``start``."""
        while len(self._segs) > 1 and self._segs[1] <= lsn:
            os.unlink(self._seg_path(self._segs[0]))
            self._segs.pop(0)
        self.start = self._segs[0]
        return self.start

    # -------------------------------------------------------------- read
    def _seg_for(self, pos: int) -> Optional[int]:
        cand = None
        for s in self._segs:
            if s <= pos:
                cand = s
            else:
                break
        return cand

    def read(self, start: int, max_bytes: int = 1 << 20) -> bytes:
        """Raw WAL bytes [start, min(end, start+max_bytes)) for streaming.
        Raises ``WalGone`` if ``start`` predates the oldest segment."""
        if start < self.start:
            raise WalGone(start, self.start)
        self.flush_buffer()
        n = min(self.end - start, max_bytes)
        if n <= 0:
            return b""
        out = bytearray()
        pos = start
        while n > 0:
            seg_start = self._seg_for(pos)
            nxt = None
            for s in self._segs:
                if s > seg_start:
                    nxt = s
                    break
            seg_end = nxt if nxt is not None else self.end
            take = min(n, seg_end - pos)
            if take <= 0:
                break
            if seg_start == self._active_start:
                out += os.pread(self._fd, take, pos - seg_start)
            else:
                fd = os.open(self._seg_path(seg_start), os.O_RDONLY)
                try:
                    out += os.pread(fd, take, pos - seg_start)
                finally:
                    os.close(fd)
            pos += take
            n -= take
        return bytes(out)

    def read_aligned(self, start: int, max_bytes: int = 1 << 20) -> bytes:
        """Raw WAL bytes from ``start``, truncated to WHOLE records —
        the only safe unit for replication (a receiver applies chunk by
        chunk; a torn frame at a window boundary would be appended but
        never applied).  Grows the window if a single record exceeds
        ``max_bytes``; returns b"" at end-of-WAL."""
        while True:
            buf = self.read(start, max_bytes)
            if not buf:
                return b""
            valid, _records = _scan(buf)
            if valid > 0:
                return buf[:valid]
            if start + len(buf) >= self.end:
                # a torn tail at the very end of WAL cannot happen for
                # committed records; nothing streamable yet
                return b""
            max_bytes *= 2   # one record larger than the window

    def iter_records(self, start: int = 0
                     ) -> Iterator[Tuple[int, bytes]]:
        """Yield (commit_lsn, payload) from offset ``start`` (must be a
        record boundary)."""
        pos = max(start, self.start)
        while pos + _HDR.size <= self.end:
            hdr = self.read(pos, _HDR.size)
            length, crc = _HDR.unpack(hdr)
            payload = self.read(pos + _HDR.size, length)
            pos += _HDR.size + length
            yield pos, payload


class WalGone(Exception):
    """The requested LSN is below the oldest retained segment."""

    def __init__(self, wanted: int, have: int):
        super().__init__("WAL at %d already recycled (oldest is %d)"
                         % (wanted, have))
        self.wanted = wanted
        self.have = have


def _scan(buf: bytes) -> Tuple[int, List[Tuple[int, int]]]:
    """Validate a record-stream prefix; returns (valid_bytes,
    [(payload_offset, payload_len), ...]).  Native-accelerated."""
    native = _codec()
    if native is not None:
        valid, _count, offsets = native.scan_records(buf, True)
        return valid, offsets
    records: List[Tuple[int, int]] = []
    pos = 0
    size = len(buf)
    while pos + _HDR.size <= size:
        length, crc = _HDR.unpack_from(buf, pos)
        if length > 64 * 1024 * 1024 or pos + _HDR.size + length > size:
            break
        payload = buf[pos + _HDR.size:pos + _HDR.size + length]
        if zlib.crc32(payload) != crc:
            break
        records.append((pos + _HDR.size, length))
        pos += _HDR.size + length
    return pos, records


def parse_frames(data: bytes) -> Iterator[Tuple[int, bytes]]:
    """Split a raw replicated chunk into (frame_len, payload) records.
    The chunk always contains whole frames (senders send record-aligned).
    Native-accelerated when the codec extension is built."""
    native = _codec()
    if native is not None:
        return iter(native.parse_frames(data))
    return _parse_frames_py(data)


def _parse_frames_py(data: bytes) -> Iterator[Tuple[int, bytes]]:
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
