"""Parity fuzz for the native jute codec (manatee_amd/native/jutec.cpp)
against the pure-Python Writer/Reader in manatee_amd/coord/jute.py.

The native codec must produce byte-identical output and read back the
same values for arbitrary operation sequences — anything less corrupts
the ZooKeeper session wire stream.
"""

import random

import pytest

from manatee_amd.coord import jute

pytestmark = pytest.mark.skipif(
    jute.CODEC != "native",
    reason="native _jutec extension not built")

OPS = ("int32", "int64", "boolean", "buffer", "ustring")


def random_value(rng, op):
    if op == "int32":
        return rng.choice([0, 1, -1, 2**31 - 1, -2**31,
                           rng.randint(-2**31, 2**31 - 1)])
    if op == "int64":
        return rng.choice([0, 1, -1, 2**63 - 1, -2**63,
                           rng.randint(-2**63, 2**63 - 1)])
    if op == "boolean":
        return rng.random() < 0.5
    if op == "buffer":
        if rng.random() < 0.2:
            return None
        return bytes(rng.getrandbits(8)
                     for _ in range(rng.randint(0, 64)))
    if op == "ustring":
        if rng.random() < 0.2:
            return None
        alphabet = "abc/é☃ÿ0123-"
        return "".join(rng.choice(alphabet)
                       for _ in range(rng.randint(0, 32)))
    raise AssertionError(op)


def test_writer_byte_parity_fuzz():
    rng = random.Random(1234)
    for _ in range(200):
        script = [(op, random_value(rng, op))
                  for op in (rng.choice(OPS)
                             for _ in range(rng.randint(0, 30)))]
        nw, pw = jute.Writer(), jute.PyWriter()
        for op, val in script:
            getattr(nw, op)(val)
            getattr(pw, op)(val)
        assert nw.tobytes() == pw.tobytes(), script
        assert nw.framed() == pw.framed(), script


def test_reader_value_parity_fuzz():
    rng = random.Random(4321)
    for _ in range(200):
        script = [(op, random_value(rng, op))
                  for op in (rng.choice(OPS)
                             for _ in range(rng.randint(0, 30)))]
        pw = jute.PyWriter()
        for op, val in script:
            getattr(pw, op)(val)
        payload = pw.tobytes()
        nr, pr = jute.Reader(payload), jute.PyReader(payload)
        for op, _ in script:
            assert getattr(nr, op)() == getattr(pr, op)(), script
            assert nr.remaining() == pr.remaining()
        assert nr.remaining() == 0


def test_reader_short_buffer_raises():
    nr = jute.Reader(b"\x00\x00")
    with pytest.raises(ValueError):
        nr.int32()
    nr = jute.Reader(b"\x00\x00\x00\x05ab")   # buffer claims 5, has 2
    with pytest.raises(ValueError):
        nr.buffer()


def test_int32_range_enforced():
    w = jute.Writer()
    with pytest.raises((ValueError, OverflowError)):
        w.int32(2**31)


def test_records_over_native_codec():
    """The record layer (Stat, multi-op, connect handshake) must work
    unchanged over the native primitives."""
    st = jute.Stat(czxid=1, mzxid=2, ctime=3, mtime=4, version=5,
                   cversion=6, aversion=7, ephemeralOwner=8,
                   dataLength=9, numChildren=10, pzxid=11)
    w = jute.Writer()
    st.write(w)
    rt = jute.Stat.read(jute.Reader(w.tobytes()))
    assert rt.as_dict() == st.as_dict()

    ops = [jute.MultiOp.create("/a", b"x", jute.PERSISTENT_SEQUENTIAL),
           jute.MultiOp.set_data("/b", b"y", 3),
           jute.MultiOp.delete("/c", 4),
           jute.MultiOp.check("/d", 5)]
    w = jute.Writer()
    jute.write_multi_request(w, ops)
    back = jute.read_multi_request(jute.Reader(w.tobytes()))
    assert [(o.kind, o.path, o.data, o.version) for o in back] == \
        [(o.kind, o.path, o.data if o.kind in ("create", "setData")
          else None, o.version) for o in ops]
