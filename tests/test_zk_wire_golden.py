"""ZooKeeper wire-format golden vectors.

Our jute client and embedded server are a matched pair — a shared
encoding bug would be invisible to their round-trip tests.  These
vectors are assembled BYTE BY BYTE here from the published ZooKeeper
3.4 jute definitions (zookeeper.jute: ConnectRequest/Response,
RequestHeader/ReplyHeader, CreateRequest, GetDataRequest/Response,
SetDataRequest, GetChildren2Request, SetWatches, WatcherEvent,
MultiHeader framing) using struct only, never our own Writer, and
compared against what manatee_amd.coord.jute produces/consumes.  A
client that matches these vectors speaks the same protocol a real
ZooKeeper 3.4 ensemble expects (ref the reference's wire dependency,
joyent-zookeeper-client / lib/zookeeperMgr.js).
"""

import struct

from manatee_amd.coord import jute

be32 = lambda v: struct.pack(">i", v)       # noqa: E731
be64 = lambda v: struct.pack(">q", v)       # noqa: E731


def ustr(s: str) -> bytes:
    b = s.encode("utf-8")
    return be32(len(b)) + b


def buf(b) -> bytes:
    if b is None:
        return be32(-1)
    return be32(len(b)) + b


def frame(body: bytes) -> bytes:
    return be32(len(body)) + body


def test_connect_request_vector():
    # ConnectRequest: protocolVersion, lastZxidSeen, timeOut, sessionId,
    # passwd (16-byte buffer)
    passwd = bytes(range(16))
    want = frame(be32(0) + be64(0x1122334455667788) + be32(30000) +
                 be64(0x0102030405060708) + buf(passwd))
    got = jute.encode_connect_request(0x1122334455667788, 30000,
                                      0x0102030405060708, passwd)
    assert got == want
    # and the decoder reads the canonical bytes back
    last, timeout, session, pw = jute.decode_connect_request(want[4:])
    assert (last, timeout, session, pw) == \
        (0x1122334455667788, 30000, 0x0102030405060708, passwd)


def test_connect_response_vector():
    want = frame(be32(0) + be32(4000) + be64(0x77) + buf(b"\x00" * 16))
    got = jute.encode_connect_response(4000, 0x77, b"\x00" * 16)
    assert got == want
    timeout, session, pw = jute.decode_connect_response(want[4:])
    assert (timeout, session, pw) == (4000, 0x77, b"\x00" * 16)


def test_get_data_request_vector():
    # RequestHeader{xid, type=getData(4)} + GetDataRequest{path, watch}
    path = "/manatee/1.moray/state"
    want = be32(7) + be32(4) + ustr(path) + b"\x01"
    w = jute.encode_request_header(7, jute.OP_GETDATA)
    w.ustring(path).boolean(True)
    assert w.tobytes() == want


def test_create_request_vector_open_acl_unsafe():
    # CreateRequest{path, data, acl vector<ACL{perms,Id{scheme,id}}>,
    # flags=EPHEMERAL_SEQUENTIAL(3)}; OPEN_ACL_UNSAFE = perms ALL(0x1f),
    # world:anyone
    path = "/manatee/1.moray/election/x-"
    data = b'{"ip":"10.0.0.1"}'
    want = (be32(1) + be32(1) +                 # header xid=1, create=1
            ustr(path) + buf(data) +
            be32(1) + be32(0x1F) + ustr("world") + ustr("anyone") +
            be32(3))
    w = jute.encode_request_header(1, jute.OP_CREATE)
    w.ustring(path).buffer(data)
    jute.write_acls(w)
    w.int32(jute.EPHEMERAL_SEQUENTIAL)
    assert w.tobytes() == want


def test_set_data_request_vector():
    want = be32(9) + be32(5) + ustr("/a") + buf(b"v") + be32(12)
    w = jute.encode_request_header(9, jute.OP_SETDATA)
    w.ustring("/a").buffer(b"v").int32(12)
    assert w.tobytes() == want


def test_reply_header_and_stat_vector():
    # ReplyHeader{xid, zxid, err} + GetDataResponse{data, Stat} with 11
    # Stat fields in jute order: czxid, mzxid, ctime, mtime, version,
    # cversion, aversion, ephemeralOwner, dataLength, numChildren, pzxid
    stat_bytes = (be64(1) + be64(2) + be64(3) + be64(4) + be32(5) +
                  be32(6) + be32(7) + be64(8) + be32(9) + be32(10) +
                  be64(11))
    body = be32(5) + be64(0xABC) + be32(0) + buf(b"payload") + stat_bytes
    r = jute.Reader(body)
    xid, zxid, err = jute.decode_reply_header(r)
    assert (xid, zxid, err) == (5, 0xABC, 0)
    assert r.buffer() == b"payload"
    st = jute.Stat.read(r)
    assert [st.czxid, st.mzxid, st.ctime, st.mtime, st.version,
            st.cversion, st.aversion, st.ephemeralOwner, st.dataLength,
            st.numChildren, st.pzxid] == [1, 2, 3, 4, 5, 6, 7, 8, 9, 10,
                                          11]
    # and our encoder emits the identical Stat bytes
    w = jute.Writer()
    st.write(w)
    assert w.tobytes() == stat_bytes


def test_watcher_event_vector():
    # notification: ReplyHeader{xid=-1, zxid=0, err=0} +
    # WatcherEvent{type, state, path}
    want = frame(be32(-1) + be64(0) + be32(0) +
                 be32(jute.EVENT_NODE_DATA_CHANGED) + be32(3) +
                 ustr("/manatee/1.moray/state"))
    got = jute.encode_watcher_event(jute.EVENT_NODE_DATA_CHANGED, 3,
                                    "/manatee/1.moray/state")
    assert got == want
    r = jute.Reader(want[4:])
    xid, zxid, err = jute.decode_reply_header(r)
    assert xid == jute.XID_NOTIFICATION
    etype, state, path = jute.decode_watcher_event(r)
    assert (etype, state, path) == (3, 3, "/manatee/1.moray/state")


def test_set_watches_vector():
    # SetWatches{relativeZxid, dataWatches, existWatches, childWatches}
    # with xid=-8, type=101; vector<ustring> = count + strings
    want = (be32(-8) + be32(101) + be64(0x55) +
            be32(2) + ustr("/a") + ustr("/b") +
            be32(0) +
            be32(1) + ustr("/c"))
    w = jute.encode_request_header(jute.XID_SET_WATCHES,
                                   jute.OP_SETWATCHES)
    w.int64(0x55)
    for paths in (["/a", "/b"], [], ["/c"]):
        w.int32(len(paths))
        for p in paths:
            w.ustring(p)
    assert w.tobytes() == want


def test_multi_request_vector():
    # Multi framing: per op MultiHeader{type, done=false, err=-1} + op
    # body; terminated by MultiHeader{-1, true, -1}
    ops = [jute.MultiOp.create("/h/n-", b"s", jute.PERSISTENT_SEQUENTIAL),
           jute.MultiOp.set_data("/s", b"d", 7)]
    want = (be32(1) + b"\x00" + be32(-1) +          # create header
            ustr("/h/n-") + buf(b"s") +
            be32(1) + be32(0x1F) + ustr("world") + ustr("anyone") +
            be32(2) +                               # PERSISTENT_SEQUENTIAL
            be32(5) + b"\x00" + be32(-1) +          # setData header
            ustr("/s") + buf(b"d") + be32(7) +
            be32(-1) + b"\x01" + be32(-1))          # done header
    w = jute.Writer()
    jute.write_multi_request(w, ops)
    assert w.tobytes() == want
    # decoder reads the canonical bytes
    back = jute.read_multi_request(jute.Reader(want))
    assert [(o.kind, o.path) for o in back] == \
        [("create", "/h/n-"), ("setData", "/s")]


def test_multi_response_vector():
    # results: create → path; error → err twice (header err + body int)
    body = (be32(1) + b"\x00" + be32(0) + ustr("/h/n-0000000001") +
            be32(-1) + b"\x00" + be32(-110) + be32(-110) +
            be32(-1) + b"\x01" + be32(-1))
    out = jute.read_multi_response(jute.Reader(body))
    assert out == [("create", "/h/n-0000000001"), ("error", -110)]


def test_error_codes_match_zookeeper():
    # the numeric error space must be ZooKeeper's, not ours
    assert jute.ZNONODE == -101
    assert jute.ZNODEEXISTS == -110
    assert jute.ZBADVERSION == -103
    assert jute.ZSESSIONEXPIRED == -112
    assert jute.ZNOTEMPTY == -111
    assert jute.ZCONNECTIONLOSS == -4
    assert jute.ZAUTHFAILED == -115
    assert jute.EPHEMERAL_SEQUENTIAL == 3
    assert jute.PERSISTENT_SEQUENTIAL == 2
    assert jute.OP_CREATE == 1 and jute.OP_DELETE == 2
    assert jute.OP_GETDATA == 4 and jute.OP_SETDATA == 5
    assert jute.OP_GETCHILDREN2 == 12 and jute.OP_MULTI == 14
    assert jute.OP_PING == 11 and jute.OP_SETWATCHES == 101
    assert jute.OP_CLOSE_SESSION == -11
