"""Native codec extension: build-if-needed, then parity-fuzz the C++
paths against the pure-Python reference implementations."""

import os
import random
import struct
import subprocess
import sys
import zlib

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope="module")
def codec():
    from manatee_amd.native import codec as c
    if c is None:
        r = subprocess.run(
            [sys.executable, "setup.py", "build_ext", "--inplace"],
            cwd=os.path.join(REPO, "manatee_amd", "native"),
            capture_output=True, text=True, timeout=600)
        assert r.returncode == 0, r.stdout + r.stderr
        import importlib
        import manatee_amd.native
        importlib.reload(manatee_amd.native)
        from manatee_amd.native import codec as c
    assert c is not None, "native codec failed to build/load"
    return c


def py_frame(payload: bytes) -> bytes:
    return struct.pack(">II", len(payload), zlib.crc32(payload)) + payload


def test_crc32_matches_zlib(codec):
    rng = random.Random(7)
    for n in (0, 1, 7, 8, 9, 63, 64, 65, 4096, 100001):
        d = bytes(rng.getrandbits(8) for _ in range(n))
        assert codec.crc32(d) == zlib.crc32(d), n


def test_encode_parse_roundtrip(codec):
    rng = random.Random(8)
    payloads = [bytes(rng.getrandbits(8) for _ in range(rng.randrange(200)))
                for _ in range(50)]
    chunk = b"".join(codec.encode_frame(p) for p in payloads)
    assert chunk == b"".join(py_frame(p) for p in payloads)
    frames = codec.parse_frames(chunk)
    assert [p for _, p in frames] == payloads
    assert [n for n, _ in frames] == [8 + len(p) for p in payloads]


def test_parse_rejects_corruption(codec):
    good = codec.encode_frame(b"abc") + codec.encode_frame(b"defg")
    with pytest.raises(ValueError):
        codec.parse_frames(good + b"trailing")
    flipped = bytearray(good)
    flipped[10] ^= 0xFF
    with pytest.raises(ValueError):
        codec.parse_frames(bytes(flipped))


def test_scan_records_parity_fuzz(codec):
    """scan_records must agree with the pure-Python _scan fallback on
    randomly-truncated and randomly-corrupted streams."""
    from manatee_amd.db.waldb import wal as walmod

    rng = random.Random(9)
    for case in range(60):
        payloads = [bytes(rng.getrandbits(8)
                          for _ in range(rng.randrange(1, 120)))
                    for _ in range(rng.randrange(1, 30))]
        stream = bytearray(b"".join(py_frame(p) for p in payloads))
        mode = case % 3
        if mode == 1 and len(stream) > 4:      # truncate
            del stream[rng.randrange(1, len(stream)):]
        elif mode == 2:                         # flip a byte
            stream[rng.randrange(len(stream))] ^= 0xFF
        buf = bytes(stream)
        n_valid, _count, offsets = codec.scan_records(buf, True)
        py_valid, py_records = walmod._scan.__wrapped__(buf) \
            if hasattr(walmod._scan, "__wrapped__") else _py_scan(buf)
        assert (n_valid, list(offsets)) == (py_valid, py_records), case


def _py_scan(buf):
    import zlib as z
    records = []
    pos = 0
    size = len(buf)
    while pos + 8 <= size:
        length, crc = struct.unpack_from(">II", buf, pos)
        if length > 64 * 1024 * 1024 or pos + 8 + length > size:
            break
        payload = buf[pos + 8:pos + 8 + length]
        if z.crc32(payload) != crc:
            break
        records.append((pos + 8, length))
        pos += 8 + length
    return pos, records


def test_wal_uses_native_when_built(codec):
    from manatee_amd.db.waldb import wal as walmod
    assert walmod._codec() is not None
