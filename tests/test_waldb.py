"""waldb engine tests: replication semantics, crash recovery, divergence.

These spawn real waldb subprocesses and kill them with SIGKILL — the only
way this system ever stops a database (MANATEE-188)."""

import asyncio
import json
import os
import signal
import socket
import subprocess
import sys
import time

import pytest

from manatee_amd.common import confparser
from manatee_amd.db.waldb.client import WaldbClient, WaldbError
from manatee_amd.db.waldb.server import init_data_dir

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


class Node:
    def __init__(self, tmp_path, name, port=None):
        self.name = name
        self.data_dir = str(tmp_path / name)
        self.port = port or free_port()
        self.proc = None

    def init(self):
        init_data_dir(self.data_dir)

    def write_conf(self, role="primary", upstream=None, sync_name=None,
                   read_only=False, extra=None):
        conf = {
            "role": role,
            "listen_ip": "127.0.0.1",
            "port": str(self.port),
            "name": self.name,
            "default_transaction_read_only": "on" if read_only else "off",
        }
        if upstream:
            conf["primary_conninfo"] = "'%s'" % upstream
        if sync_name:
            conf["synchronous_standby_names"] = "'%s'" % sync_name
        conf.update(extra or {})
        confparser.write(os.path.join(self.data_dir, "waldb.conf"), conf)

    def start(self):
        env = dict(os.environ, PYTHONPATH=REPO)
        self.proc = subprocess.Popen(
            [sys.executable, "-m", "manatee_amd.db.waldb.server",
             "-D", self.data_dir],
            env=env, stdout=subprocess.DEVNULL, stderr=subprocess.PIPE)
        # wait for the pid file (readiness)
        pid_file = os.path.join(self.data_dir, "waldb.pid")
        deadline = time.time() + 10
        while not os.path.exists(pid_file) or \
                os.stat(pid_file).st_size == 0:
            if self.proc.poll() is not None:
                raise RuntimeError("waldb died: %s"
                                   % self.proc.stderr.read().decode())
            assert time.time() < deadline, "waldb did not start"
            time.sleep(0.02)

    def kill9(self):
        self.proc.send_signal(signal.SIGKILL)
        self.proc.wait()
        os.unlink(os.path.join(self.data_dir, "waldb.pid"))

    def sighup(self):
        self.proc.send_signal(signal.SIGHUP)

    def stop(self):
        if self.proc and self.proc.poll() is None:
            self.proc.kill()
            self.proc.wait()

    def client(self):
        return WaldbClient("127.0.0.1", self.port)

    def promote_trigger(self):
        open(os.path.join(self.data_dir, "promote"), "w").close()


async def wait_async(pred_coro, timeout=10.0, what="condition"):
    deadline = time.monotonic() + timeout
    while True:
        if await pred_coro():
            return
        assert time.monotonic() < deadline, "timeout: " + what
        await asyncio.sleep(0.05)


def run(coro):
    return asyncio.run(asyncio.wait_for(coro, 60))


def test_single_node_put_get_and_crash_recovery(tmp_path):
    n = Node(tmp_path, "n1")
    n.init()
    n.write_conf(role="primary")
    n.start()
    try:
        async def phase1():
            c = n.client()
            assert await c.ping()
            lsn1 = await c.put("a", 1)
            lsn2 = await c.put("b", {"x": [1, 2]})
            assert lsn2 > lsn1
            assert await c.get("a") == 1
            assert await c.get("b") == {"x": [1, 2]}
            assert await c.get("nope") is None
            st = await c.status()
            assert st["role"] == "primary"
            assert st["current_lsn"] == lsn2
            await c.close()
        run(phase1())
        n.kill9()
        # corrupt tail: simulate torn write from the dirty kill (the last
        # WAL segment is the only one that can be torn)
        segs = sorted(f for f in os.listdir(n.data_dir)
                      if f.startswith("wal-") and f.endswith(".seg"))
        with open(os.path.join(n.data_dir, segs[-1]), "ab") as f:
            f.write(b"\x00\x00\x00\x10partial")
        n.start()

        async def phase2():
            c = n.client()
            assert await c.get("a") == 1
            assert await c.get("b") == {"x": [1, 2]}
            await c.close()
        run(phase2())
    finally:
        n.stop()


def test_db_child_killable_during_boot_despite_inherited_sigign(tmp_path):
    """A db child spawned from a chain with SIGINT/SIGQUIT at SIG_IGN
    (POSIX backgrounding does this; dispositions survive exec) must
    still die promptly when a dirty stop lands in its BOOT window —
    before the event loop installs handlers.  Without the preexec
    reset, SIGINT and SIGQUIT were silently ignored for the whole boot
    and escalation burned 2 x ops_timeout to SIGKILL, stalling the
    serialized FSM ~60 s mid-failover (found by long chaos soaks)."""
    import subprocess

    from manatee_amd.db.manager import db_child_preexec

    data = str(tmp_path / "db")
    init_data_dir(data)
    confparser.write(os.path.join(data, "waldb.conf"), {
        "role": "primary", "listen_ip": "127.0.0.1", "port": "0",
        "name": "n1"})

    old_int = signal.signal(signal.SIGINT, signal.SIG_IGN)
    old_quit = signal.signal(signal.SIGQUIT, signal.SIG_IGN)
    try:
        env = dict(os.environ, PYTHONPATH=REPO)
        proc = subprocess.Popen(
            [sys.executable, "-m", "manatee_amd.db.waldb.server",
             "-D", data], env=env, start_new_session=True,
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
            preexec_fn=db_child_preexec)
        try:
            # SIGINT immediately — almost certainly inside the boot
            # window (python imports + recovery), before any handler
            time.sleep(0.05)
            os.killpg(proc.pid, signal.SIGINT)
            deadline = time.monotonic() + 8
            while proc.poll() is None and time.monotonic() < deadline:
                time.sleep(0.05)
            assert proc.poll() is not None, \
                "db child ignored SIGINT during boot"
        finally:
            if proc.poll() is None:
                os.killpg(proc.pid, signal.SIGKILL)
                proc.wait()
    finally:
        signal.signal(signal.SIGINT, old_int)
        signal.signal(signal.SIGQUIT, old_quit)


def test_dirty_exit_is_immediate_even_mid_checkpoint(tmp_path):
    """A stop signal must end the process in milliseconds regardless of
    what is in flight (MANATEE-188: durability comes from the WAL, not
    a clean shutdown).  Before the os._exit change, the graceful
    asyncio unwind WAITED for the default executor, where a whole-kv
    checkpoint write could run for tens of seconds under load — the
    dirty-stop escalation then burned 2 x ops_timeout to SIGKILL,
    the dominant cause of rare ~60 s failovers in long chaos soaks."""
    n = Node(tmp_path, "n1")
    n.init()
    # tiny checkpoint threshold: the server checkpoints continuously
    # under this load, so a SIGINT lands mid-checkpoint with high
    # probability
    n.write_conf(role="primary", extra={"checkpoint_wal_bytes": "65536"})
    n.start()
    try:
        async def load():
            cli = n.client()
            for i in range(40):
                await cli.put_many((("ck-%d-%d" % (i, j), "v" * 200)
                                    for j in range(500)))
            await cli.close()
        run(load())
        t0 = time.monotonic()
        os.kill(n.proc.pid, signal.SIGINT)
        while n.proc.poll() is None and time.monotonic() - t0 < 10:
            time.sleep(0.02)
        dt = time.monotonic() - t0
        assert n.proc.poll() is not None, "server survived SIGINT 10s"
        assert dt < 3.0, "dirty exit took %.2fs" % dt
    finally:
        n.stop()


def test_repl_handoff_consumes_pipelined_ack(tmp_path):
    """A standby's first ack arriving in the SAME TCP segment as its
    repl request must not be lost: the server's chunked request framing
    reads past the repl line, and the buffered leftover is handed to
    the ack consumer (server._BufferedReader).  Without that handoff
    the gated commit below would never be acknowledged."""
    n = Node(tmp_path, "prim")
    n.init()
    n.write_conf(role="primary", sync_name="fakesync")
    n.start()
    try:
        async def go():
            c = n.client()
            st = await c.status()
            ident = st["ident"]
            # this put appends immediately but blocks awaiting the sync
            put_task = asyncio.ensure_future(c.put("gated", 1,
                                                   timeout_s=20))
            await asyncio.sleep(0.3)
            assert not put_task.done()

            st2c = n.client()
            st2 = await st2c.status()
            lsn = int(st2["current_lsn"].split("/")[1], 16)
            await st2c.close()
            assert lsn > 0

            reader, writer = await asyncio.open_connection("127.0.0.1",
                                                           n.port)
            # repl request AND the ack for the appended record in ONE
            # write -> one TCP segment -> one server-side read()
            repl = json.dumps({"q": "repl", "name": "fakesync",
                               "ident": ident, "timeline": 1,
                               "start_lsn": 0})
            ack = json.dumps({"write_lsn": lsn, "flush_lsn": lsn,
                              "replay_lsn": lsn})
            writer.write((repl + "\n" + ack + "\n").encode())
            await writer.drain()
            line = await asyncio.wait_for(reader.readline(), 10)
            assert json.loads(line).get("ok") is True
            # the pipelined ack must release the gated commit
            await asyncio.wait_for(put_task, 10)
            writer.close()
            await c.close()
        run(go())
    finally:
        n.stop()


def test_reconnecting_replica_supersedes_stale_sender(tmp_path):
    """A replica reconnect under the same name must EVICT the wedged
    old connection immediately.  Before the fix, the stale row stayed
    first in pg_stat_replication until wal_sender_timeout (default
    60 s) reaped it, and the commit gate read its frozen ack LSNs —
    the rare ~60 s failover outliers seen in long chaos soaks."""
    n = Node(tmp_path, "prim")
    n.init()
    n.write_conf(role="primary", sync_name="fakesync")
    n.start()
    try:
        async def go():
            c = n.client()
            st = await c.status()
            ident = st["ident"]

            async def connect_sync(ack_lsn):
                reader, writer = await asyncio.open_connection(
                    "127.0.0.1", n.port)
                repl = json.dumps({"q": "repl", "name": "fakesync",
                                   "ident": ident, "timeline": 1,
                                   "start_lsn": 0})
                ack = json.dumps({"write_lsn": ack_lsn,
                                  "flush_lsn": ack_lsn,
                                  "replay_lsn": ack_lsn})
                writer.write((repl + "\n" + ack + "\n").encode())
                await writer.drain()
                line = await asyncio.wait_for(reader.readline(), 10)
                assert json.loads(line).get("ok") is True
                return reader, writer

            # wedged "old" connection: acks LSN 0 and then goes silent
            r1, w1 = await connect_sync(0)
            await asyncio.sleep(0.2)
            # a put now blocks on the silent sync
            put_task = asyncio.ensure_future(c.put("stuck", 1,
                                                   timeout_s=30))
            await asyncio.sleep(0.3)
            assert not put_task.done()
            st2c = n.client()
            lsn = int((await st2c.status())["current_lsn"].split("/")[1],
                      16)
            await st2c.close()

            # the replica reconnects (same name) and acks the record:
            # the commit must complete promptly — NOT after 60 s
            t0 = time.monotonic()
            r2, w2 = await connect_sync(lsn)
            await asyncio.wait_for(put_task, 10)
            assert time.monotonic() - t0 < 10
            # exactly one replication row remains
            st3c = n.client()
            rows = (await st3c.status())["replication"]
            assert len([r for r in rows
                        if r["application_name"] == "fakesync"]) == 1
            await st3c.close()
            for w in (w1, w2):
                try:
                    w.close()
                except Exception:
                    pass
            await c.close()
        run(go())
    finally:
        n.stop()


def test_sync_replication_gates_commit(tmp_path):
    prim = Node(tmp_path, "prim")
    sync = Node(tmp_path, "sync")
    prim.init()
    prim.write_conf(role="primary", sync_name="sync")
    prim.start()
    try:
        async def no_standby_blocks():
            c = n_client = prim.client()
            with pytest.raises(WaldbError):
                # no sync standby connected → commit must block
                await c.put("k", "v", timeout_s=1.0)
            await c.close()
        run(no_standby_blocks())

        # bootstrap the sync from the primary's data dir (they must share
        # the system identifier)
        import shutil
        shutil.copytree(prim.data_dir, sync.data_dir,
                        ignore=shutil.ignore_patterns("waldb.pid",
                                                      "waldb.conf"))
        sync.write_conf(role="standby",
                        upstream="127.0.0.1:%d" % prim.port)
        sync.start()

        async def with_standby():
            c = prim.client()
            sc = sync.client()

            async def sync_streaming():
                st = await sc.status()
                return st["upstream_status"] == "streaming"
            await wait_async(sync_streaming, what="sync streaming")
            lsn = await c.put("k", "v", timeout_s=5.0)

            async def replicated():
                return await sc.get("k") == "v"
            await wait_async(replicated, what="value on sync")
            st = await c.status()
            assert len(st["replication"]) == 1
            r = st["replication"][0]
            assert r["application_name"] == "sync"
            assert r["sync_state"] == "sync"
            assert r["write_lsn"] >= lsn
            # standby rejects writes
            with pytest.raises(WaldbError):
                await sc.put("x", 1)
            await c.close()
            await sc.close()
        run(with_standby())
    finally:
        prim.stop()
        sync.stop()


def test_cascading_replication_and_promote_divergence(tmp_path):
    import shutil
    prim = Node(tmp_path, "prim")
    sync = Node(tmp_path, "syncp")
    asy = Node(tmp_path, "asyncp")
    prim.init()
    prim.write_conf(role="primary", sync_name="syncp")
    prim.start()
    try:
        async def seed():
            c = prim.client()
            # ONWM-style: temporarily no sync gate
            prim.write_conf(role="primary")
            prim.sighup()
            await asyncio.sleep(0.2)
            for i in range(50):
                await c.put("seed%d" % i, i)
            await c.close()
        run(seed())

        for node in (sync, asy):
            shutil.copytree(prim.data_dir, node.data_dir,
                            ignore=shutil.ignore_patterns("waldb.pid",
                                                          "waldb.conf"))
        sync.write_conf(role="standby", upstream="127.0.0.1:%d" % prim.port)
        asy.write_conf(role="standby", upstream="127.0.0.1:%d" % sync.port)
        sync.start()
        asy.start()
        prim.write_conf(role="primary", sync_name="syncp")
        prim.sighup()

        async def chain():
            pc, sc, ac = prim.client(), sync.client(), asy.client()

            async def both_streaming():
                s1 = await sc.status()
                s2 = await ac.status()
                return (s1["upstream_status"] == "streaming"
                        and s2["upstream_status"] == "streaming")
            await wait_async(both_streaming, what="cascade streaming")
            await pc.put("cascade", "yes", timeout_s=5.0)

            async def reached_async():
                return await ac.get("cascade") == "yes"
            await wait_async(reached_async, what="value cascaded to async")
            # the async's row appears on the SYNC (cascading), not primary
            st = await sc.status()
            assert [r["application_name"]
                    for r in st["replication"]] == ["asyncp"]
            await pc.close()
            await sc.close()
            await ac.close()
        run(chain())

        # ---- failover: kill primary; promote the sync (timeline bump) ----
        prim.kill9()
        sync.kill9()
        sync.promote_trigger()
        sync.write_conf(role="primary")
        sync.start()
        asy.write_conf(role="standby", upstream="127.0.0.1:%d" % sync.port)
        asy.sighup()

        async def promoted():
            sc, ac = sync.client(), asy.client()
            st = await sc.status()
            assert st["role"] == "primary"
            assert st["timeline"] == 2
            await sc.put("after-failover", 1)

            async def follows():
                s = await ac.status()
                return (s["upstream_status"] == "streaming"
                        and s["timeline"] == 2)
            await wait_async(follows, what="async follows new timeline")

            async def got():
                return await ac.get("after-failover") == 1
            await wait_async(got, what="post-failover write cascaded")
            await sc.close()
            await ac.close()
        run(promoted())

        # ---- divergence: old primary writes past the switch point, then
        # tries to follow the new primary → must be refused ----
        prim.write_conf(role="primary")
        prim.start()

        async def diverge():
            pc = prim.client()
            await pc.put("diverged-write", 1)
            await pc.close()
        run(diverge())
        prim.kill9()
        prim.write_conf(role="standby", upstream="127.0.0.1:%d" % sync.port)
        prim.start()

        async def refused():
            pc = prim.client()

            async def diverged():
                st = await pc.status()
                return st["upstream_status"] == "diverged"
            await wait_async(diverged, what="divergence detected")
            await pc.close()
        run(refused())
    finally:
        prim.stop()
        sync.stop()
        asy.stop()


def test_acked_writes_survive_primary_kill9(tmp_path):
    """The north-star durability property: every acknowledged write is on
    the sync standby after kill -9 of the primary (BASELINE.json)."""
    import shutil
    prim = Node(tmp_path, "prim")
    sync = Node(tmp_path, "s1")
    prim.init()
    prim.write_conf(role="primary")
    prim.start()
    try:
        run(prim.client().ping())
        shutil.copytree(prim.data_dir, sync.data_dir,
                        ignore=shutil.ignore_patterns("waldb.pid",
                                                      "waldb.conf"))
        sync.write_conf(role="standby", upstream="127.0.0.1:%d" % prim.port)
        sync.start()
        prim.write_conf(role="primary", sync_name="s1")
        prim.sighup()

        acked = []

        async def load():
            c = prim.client()
            sc = sync.client()

            async def streaming():
                return (await sc.status())["upstream_status"] == "streaming"
            await wait_async(streaming, what="sync streaming")
            for i in range(200):
                lsn = await c.put("key%d" % i, i, timeout_s=5.0)
                acked.append(("key%d" % i, i, lsn))
            await c.close()
            await sc.close()
        run(load())
        prim.kill9()

        async def verify():
            sc = sync.client()
            for k, v, _lsn in acked:
                assert await sc.get(k) == v, \
                    "acknowledged write %s lost after primary kill -9" % k
            await sc.close()
        run(verify())
    finally:
        prim.stop()
        sync.stop()


# ----------------------------------------------------- segmented WAL + ckpt

def test_wal_class_segmentation_and_recycling(tmp_path):
    """Direct Wal unit test: rolling, cross-segment reads, drop_below,
    truncate_to, torn-tail recovery, WalGone."""
    from manatee_amd.db.waldb.wal import Wal, WalGone

    d = str(tmp_path / "wal")
    w = Wal(d, segment_bytes=256)
    w.open()
    lsns = []
    payloads = []
    for i in range(40):
        p = (b"rec-%03d-" % i) * 4
        payloads.append(p)
        lsns.append(w.append(p))
    w.fsync()
    assert len(w._segs) > 3, "expected several segments"

    # cross-segment read reassembles the stream
    whole = w.read(0, 1 << 20)
    assert len(whole) == w.end

    # reopen + full replay
    w.close()
    w2 = Wal(d, segment_bytes=256)
    seen = []
    w2.open(replay=lambda lsn, p: seen.append((lsn, p)))
    assert [p for _, p in seen] == payloads
    assert w2.end == lsns[-1]

    # checkpoint-style recycling drops whole segments
    mid = lsns[len(lsns) // 2]
    w2.drop_below(mid)
    assert 0 < w2.start <= mid
    with pytest.raises(WalGone):
        w2.read(0)
    # replay_from skips dropped history
    w2.close()
    w3 = Wal(d, segment_bytes=256)
    seen3 = []
    w3.open(replay=lambda lsn, p: seen3.append(lsn), replay_from=mid)
    assert all(lsn > mid for lsn in seen3)
    assert w3.end == lsns[-1]

    # timeline fencing truncation
    cut = lsns[-5]
    w3.truncate_to(cut)
    assert w3.end == cut
    w3.append(b"after-truncate")
    assert w3.end > cut

    # torn tail in the LAST segment only is repaired
    w3.fsync()
    last_seg = sorted(w3._segs)[-1]
    with open(os.path.join(d, "wal-%016x.seg" % last_seg), "ab") as f:
        f.write(b"\x00\x00\x00\x20torn")
    w3.close()
    w4 = Wal(d, segment_bytes=256)
    end = w4.open(replay_from=w4.start)
    assert end == cut + 8 + len(b"after-truncate")
    w4.close()


def test_wal_buffered_append_flush_and_crash_drop(tmp_path):
    """Direct Wal unit test for append_buffered: reads force a flush
    mid-stream (replica senders observe everything appended), segment
    rolls flush the old segment first, a crash before any flush drops
    ONLY the still-buffered (never-observed, hence never-acked) tail,
    and fsync makes buffered records fully durable."""
    from manatee_amd.db.waldb.wal import Wal, encode_op

    d = str(tmp_path / "wal")
    w = Wal(d, segment_bytes=4096)
    w.open()
    lsns = [w.append_buffered(encode_op({"op": "put", "k": "k%d" % i,
                                         "v": "x" * 100}))
            for i in range(200)]           # crosses several rolls
    assert len(w._segs) > 2
    # a read (the replica-sender path) flushes implicitly
    chunk = w.read_aligned(0)
    assert chunk
    for i in range(50):
        w.append_buffered(encode_op({"op": "put", "k": "m%d" % i, "v": 1}))
    end_before = w.end
    # simulate kill -9 with the tail still buffered
    w._buf.clear()
    os.close(w._fd)
    w._fd = None
    w2 = Wal(d, segment_bytes=4096)
    end = w2.open(replay=lambda lsn, p: None)
    assert end >= lsns[-1], "flushed records must survive"
    assert end <= end_before
    # buffered records become durable through fsync
    l = w2.append_buffered(encode_op({"op": "put", "k": "z", "v": 2}))
    w2.fsync()
    w2.close()
    w3 = Wal(d, segment_bytes=4096)
    seen = []
    assert w3.open(replay=lambda lsn, p: seen.append(lsn)) == l
    assert seen[-1] == l
    w3.close()


def test_checkpoint_bounds_recovery_and_wal_size(tmp_path):
    """A primary under write load checkpoints, recycles old WAL segments
    and recovers from checkpoint + tail after kill -9 with all data."""
    small = {"checkpoint_wal_bytes": "4096", "wal_keep_bytes": "4096",
             "wal_segment_bytes": "2048"}
    n = Node(tmp_path, "ckpt")
    n.init()
    n.write_conf(role="primary", extra=small)
    n.start()
    try:
        async def fill():
            c = n.client()
            for i in range(400):
                await c.put("k%d" % i, "v" * 50)

            async def ckpted():
                st = await c.status()
                return st["checkpoint_lsn"] != "0/00000000" and \
                    st["wal_start_lsn"] != "0/00000000"
            await wait_async(ckpted, timeout=15,
                             what="checkpoint + WAL recycling")
            st = await c.status()
            assert st["wal_retained_bytes"] < 64 * 1024
            assert await c.count(prefix="k") == 400
            await c.close()
        run(fill())
        segs = [f for f in os.listdir(n.data_dir)
                if f.startswith("wal-")]
        assert len(segs) < 20, "old segments were not recycled"
        assert os.path.exists(os.path.join(n.data_dir, "checkpoint.json"))

        n.kill9()
        n.start()

        async def verify():
            c = n.client()
            assert await c.count(prefix="k") == 400
            for i in (0, 199, 399):
                assert await c.get("k%d" % i) == "v" * 50
            await c.close()
        run(verify())
    finally:
        n.stop()


def test_follower_behind_recycled_wal_is_refused(tmp_path):
    """A standby whose position predates the primary's oldest retained
    segment must be refused with wal-gone and flag itself diverged (the
    manager then restores it from a snapshot — integ-tested elsewhere)."""
    small = {"checkpoint_wal_bytes": "2048", "wal_keep_bytes": "1024",
             "wal_segment_bytes": "1024"}
    prim = Node(tmp_path, "prim")
    prim.init()
    prim.write_conf(role="primary", extra=small)
    prim.start()
    try:
        async def fill():
            c = prim.client()
            for i in range(300):
                await c.put("w%d" % i, "x" * 40)

            async def recycled():
                st = await c.status()
                return st["wal_start_lsn"] != "0/00000000"
            await wait_async(recycled, timeout=15, what="WAL recycling")
            await c.close()
        run(fill())

        # fresh standby with an empty WAL (start position 0): the data it
        # needs is gone — streaming must be refused
        stand = Node(tmp_path, "stand")
        # copy the ident so only the WAL position is at issue
        os.makedirs(stand.data_dir, exist_ok=True)
        with open(os.path.join(prim.data_dir, "waldb_ident.json")) as f:
            ident = f.read()
        with open(os.path.join(stand.data_dir, "waldb_ident.json"),
                  "w") as f:
            f.write(ident)
        stand.write_conf(role="standby",
                         upstream="127.0.0.1:%d" % prim.port)
        stand.start()
        try:
            async def check():
                c = stand.client()

                async def diverged():
                    st = await c.status()
                    return st["upstream_status"] == "diverged"
                await wait_async(diverged, timeout=15,
                                 what="wal-gone -> diverged")
                await c.close()
            run(check())
        finally:
            stand.stop()
    finally:
        prim.stop()


def test_del_batch_scan_replicated_and_crash_safe(tmp_path):
    """del / atomic batch / prefix scan: WAL-logged, replicated to the
    standby, atomic across kill -9 recovery."""
    import shutil
    prim = Node(tmp_path, "prim")
    stby = Node(tmp_path, "bsync")
    prim.init()
    prim.write_conf(role="primary")
    prim.start()
    try:
        async def seed():
            c = prim.client()
            for i in range(10):
                await c.put("acct:%03d" % i, {"bal": i * 10})
            await c.close()
        run(seed())
        shutil.copytree(prim.data_dir, stby.data_dir,
                        ignore=shutil.ignore_patterns("waldb.pid",
                                                      "waldb.conf"))
        prim.write_conf(role="primary", sync_name="bsync")
        prim.sighup()
        stby.write_conf(role="standby",
                        upstream="127.0.0.1:%d" % prim.port)
        stby.start()

        async def exercise():
            c = prim.client()
            sc = stby.client()

            async def streaming():
                st = await sc.status()
                return st["upstream_status"] == "streaming"
            await wait_async(streaming, what="standby streaming")

            # delete
            await c.delete("acct:000")
            assert await c.get("acct:000") is None
            assert await c.count(prefix="acct:") == 9

            # atomic transfer: one batch record moves balance 001 -> 002
            await c.batch([
                {"op": "put", "k": "acct:001", "v": {"bal": 0}},
                {"op": "put", "k": "acct:002", "v": {"bal": 30}},
                {"op": "del", "k": "acct:003"},
            ])
            assert await c.get("acct:001") == {"bal": 0}
            assert await c.get("acct:002") == {"bal": 30}
            assert await c.get("acct:003") is None

            # scan with pagination, and on the STANDBY
            items = await c.scan(prefix="acct:", limit=3)
            assert [i["k"] for i in items] == \
                ["acct:001", "acct:002", "acct:004"]
            more = await c.scan(prefix="acct:", after=items[-1]["k"])
            assert more[0]["k"] == "acct:005"

            async def replicated():
                return await sc.get("acct:003") is None and \
                    (await sc.get("acct:002")) == {"bal": 30}
            await wait_async(replicated, what="batch on standby")
            sitems = await sc.scan(prefix="acct:")
            assert len(sitems) == 8
            # standby rejects writes of every kind
            with pytest.raises(WaldbError):
                await sc.delete("acct:004")
            with pytest.raises(WaldbError):
                await sc.batch([{"op": "del", "k": "acct:004"}])
            await c.close()
            await sc.close()
        run(exercise())

        # crash recovery preserves the batch atomically
        prim.kill9()
        prim.write_conf(role="primary")   # no sync gate for the check
        prim.start()

        async def recovered():
            c = prim.client()
            assert await c.get("acct:001") == {"bal": 0}
            assert await c.get("acct:002") == {"bal": 30}
            assert await c.get("acct:003") is None
            assert await c.count(prefix="acct:") == 8
            await c.close()
        run(recovered())
    finally:
        prim.stop()
        stby.stop()


def test_pipelined_puts(tmp_path):
    n = Node(tmp_path, "pipe")
    n.init()
    n.write_conf(role="primary")
    n.start()
    try:
        async def go():
            c = n.client()
            assert await c.put_many((("p%03d" % i, i)
                                     for i in range(500))) == 500
            assert await c.count(prefix="p") == 500
            assert await c.get("p499") == 499
            # a failed op in the pipeline raises, earlier ops committed
            n.write_conf(role="primary", read_only=True)
            n.sighup()
            await asyncio.sleep(0.2)
            with pytest.raises(WaldbError):
                await c.put_many([("q1", 1)])
            await c.close()
        run(go())
    finally:
        n.stop()


def test_standby_catches_up_from_far_behind(tmp_path):
    """Regression: the replication stream must be record-aligned.  A
    standby joining multiple MiB behind used to receive 1 MiB windows
    that cut records at the boundary — the bytes landed in its WAL (so
    its acks looked caught-up) but were never APPLIED, and a later
    promote served a stale kv.  The standby must apply everything."""
    import shutil
    prim = Node(tmp_path, "prim")
    late = Node(tmp_path, "late")
    prim.init()
    prim.write_conf(role="primary")
    prim.start()
    try:
        async def seed_small():
            c = prim.client()
            await c.put("seed", 0)
            await c.close()
        run(seed_small())
        # snapshot the dataset EARLY — the standby will start from here
        shutil.copytree(prim.data_dir, late.data_dir,
                        ignore=shutil.ignore_patterns("waldb.pid",
                                                      "db_child.pid",
                                                      "waldb.conf"))

        async def bulk():
            c = prim.client()
            payload = "x" * 120
            for start in range(0, 30000, 500):
                await c.put_many((("blk%05d" % i, payload)
                                  for i in range(start, start + 500)))
            st = await c.status()
            await c.close()
            return st["current_lsn"]
        end_lsn = run(bulk())
        # the backlog really is multiple read-windows deep
        assert int(end_lsn.split("/")[1], 16) > 3 * (1 << 20)

        late.write_conf(role="standby",
                        upstream="127.0.0.1:%d" % prim.port)
        late.start()

        async def verify():
            sc = late.client()

            async def caught_up():
                st = await sc.status()
                return st["upstream_status"] == "streaming" and \
                    st["current_lsn"] == end_lsn
            await wait_async(caught_up, timeout=60,
                             what="standby WAL catch-up")
            # THE critical check: the kv must have APPLIED everything,
            # not merely appended it to the WAL
            assert await sc.count(prefix="blk") == 30000
            assert await sc.get("blk29999") == "x" * 120
            assert await sc.get("blk00000") == "x" * 120
            await sc.close()
        run(verify())
    finally:
        prim.stop()
        late.stop()


def test_sighup_during_boot_is_survivable(tmp_path):
    """A reload signal delivered before the server installs its handler
    must not kill it (children spawn with SIGHUP ignored; the conf
    mtime watcher covers any missed reload)."""
    import subprocess as sp
    n = Node(tmp_path, "hup")
    n.init()
    n.write_conf(role="primary")
    env = dict(os.environ, PYTHONPATH=REPO)

    def child_init():
        signal.signal(signal.SIGHUP, signal.SIG_IGN)
    proc = sp.Popen(
        [sys.executable, "-m", "manatee_amd.db.waldb.server",
         "-D", n.data_dir],
        env=env, stdout=sp.DEVNULL, stderr=sp.DEVNULL,
        preexec_fn=child_init)
    try:
        # storm of reload signals through the boot window
        for _ in range(50):
            proc.send_signal(signal.SIGHUP)
            time.sleep(0.01)
        deadline = time.time() + 15
        pid_file = os.path.join(n.data_dir, "waldb.pid")
        while not os.path.exists(pid_file) or \
                os.stat(pid_file).st_size == 0:
            assert proc.poll() is None, \
                "server died from a boot-window SIGHUP"
            assert time.time() < deadline
            time.sleep(0.02)
        # and the mtime watcher still applies conf changes without signal
        n.write_conf(role="primary", read_only=True)

        async def becomes_ro():
            c = n.client()

            async def ro():
                st = await c.status()
                return st["read_only"]
            await wait_async(ro, timeout=10, what="mtime-watched reload")
            await c.close()
        run(becomes_ro())
    finally:
        proc.kill()
        proc.wait()


def test_foreign_conf_reload_refused(tmp_path):
    """A conf naming a DIFFERENT port (e.g. another peer's conf that
    rode in with a dataset restore) must never be adopted by a running
    server."""
    n = Node(tmp_path, "foreign")
    n.init()
    n.write_conf(role="primary")
    n.start()
    try:
        async def go():
            c = n.client()
            await c.put("mine", 1)
            # drop in a foreign conf: different port, standby role
            confparser.write(os.path.join(n.data_dir, "waldb.conf"), {
                "role": "standby", "listen_ip": "127.0.0.1",
                "port": str(n.port + 1), "name": "someone-else",
                "primary_conninfo": "'127.0.0.1:1'",
            })
            n.sighup()
            await asyncio.sleep(0.5)
            st = await c.status()
            assert st["role"] == "primary", "adopted a foreign conf!"
            await c.put("still-mine", 2)
            await c.close()
        run(go())
    finally:
        n.stop()


def test_post_restore_fixup_purges_peer_local_files(tmp_path):
    from manatee_amd.db.engine import WaldbEngine
    data = str(tmp_path / "store" / "live" / "data")
    os.makedirs(data)
    for name in ("waldb.pid", "db_child.pid", "waldb.conf", "promote"):
        open(os.path.join(data, name), "w").write("junk")
    open(os.path.join(tmp_path / "store" / "live", "waldb.log"),
         "w").write("foreign log")
    open(os.path.join(data, "waldb_ident.json"), "w").write("{}")
    eng = WaldbEngine(data, "127.0.0.1", 1, "p")
    eng.post_restore_fixup()
    left = sorted(os.listdir(data))
    assert left == ["waldb_ident.json"], left
    assert not os.path.exists(
        os.path.join(tmp_path / "store" / "live", "waldb.log"))


def test_sync_ack_pins_wal_retention_across_disconnect(tmp_path):
    """Regression (advisor finding): the NAMED sync standby's last acked
    LSN must pin the WAL retention floor across a brief disconnect —
    otherwise a sync that blips while the primary appends past the keep
    window gets 'wal-gone' on reconnect and is forced into a full
    restore (read-only storm under load)."""
    async def go():
        prim = Node(tmp_path, "prim")
        prim.init()
        # tiny windows so a burst of appends would normally recycle
        # everything the sync still needs
        extra = {"checkpoint_wal_bytes": "4096",
                 "wal_keep_bytes": "2048",
                 "wal_segment_bytes": "8192"}
        prim.write_conf(role="primary", sync_name="sync", extra=extra)
        prim.start()
        sync = Node(tmp_path, "sync")
        try:
            pcli = prim.client()
            # replicate + ack a first batch
            import shutil
            shutil.copytree(os.path.join(prim.data_dir),
                            sync.data_dir, dirs_exist_ok=True)
            for name in ("waldb.conf", "waldb.pid"):
                try:
                    os.unlink(os.path.join(sync.data_dir, name))
                except FileNotFoundError:
                    pass
            sync.write_conf(role="standby",
                            upstream="127.0.0.1:%d" % prim.port,
                            extra={"name": "'sync'"})
            sync.start()
            await pcli.put("base", "v")

            async def caught_up():
                st = await pcli.status()
                return any(r["application_name"] == "sync"
                           and r["write_lsn"] == st["current_lsn"]
                           for r in st["replication"])
            await wait_async(caught_up, what="sync caught up")
            st = await pcli.status()
            acked_at = st["current_lsn"]

            # sync blips; the primary keeps APPENDING (each gated put
            # appends before blocking on the ack, like a client retry
            # storm) far past the keep window, with checkpoints running
            sync.kill9()
            for i in range(40):
                try:
                    await pcli.put("burst%d" % i, "x" * 512,
                                   timeout_s=0.15)
                except WaldbError:
                    pass
            await asyncio.sleep(0.5)   # let checkpoints recycle

            st = await pcli.status()
            from manatee_amd.common.lsn import parse
            assert parse(st["wal_start_lsn"]) <= parse(acked_at), \
                "retention floor did not pin the disconnected sync's " \
                "acked LSN (start %s > acked %s)" \
                % (st["wal_start_lsn"], acked_at)

            # the sync reconnects and STREAMS (no wal-gone → no restore)
            sync.start()
            scli = sync.client()

            async def streaming():
                st2 = await scli.status()
                return st2["upstream_status"] == "streaming"
            await wait_async(streaming, what="sync resumed streaming")
            await scli.close()
            await pcli.close()
        finally:
            prim.stop()
            sync.stop()
    run(go())


def test_checkpoint_spacing_scales_with_image_size(tmp_path):
    """Checkpoint I/O must stay amortized O(1) per WAL byte: once an
    image has been written, the next checkpoint is due only after at
    least half that image's size in NEW WAL (never sooner than the
    configured checkpoint_wal_bytes).  Without this, a multi-hundred-MB
    store re-serialized its entire kv every 8 MB of appends — the I/O
    storm behind the shutdown-wait failover outliers."""
    async def go():
        from manatee_amd.common.logging import null_logger
        from manatee_amd.db.waldb.server import WaldbServer

        data = str(tmp_path / "n1")
        init_data_dir(data)
        confparser.write(os.path.join(data, "waldb.conf"), {
            "role": "primary", "listen_ip": "127.0.0.1", "port": "0",
            "name": "n1", "checkpoint_wal_bytes": "4096"})
        srv = WaldbServer(data, null_logger())
        await srv.start()
        try:
            for i in range(60):
                await srv._do_write({"op": "put", "k": "k%d" % i,
                                     "v": "x" * 100})
            await srv._checkpoint()
            first_lsn = srv.ckpt_lsn
            assert srv._last_ckpt_bytes > 4096, \
                "image should exceed the configured threshold"
            # append a bit more than checkpoint_wal_bytes but LESS than
            # half the image: the flusher must NOT schedule a new one
            for i in range(10):
                await srv._do_write({"op": "put", "k": "s%d" % i,
                                     "v": "y" * 100})
            due = max(srv.ckpt_wal_bytes, srv._last_ckpt_bytes // 2)
            assert srv.replay_lsn - srv.ckpt_lsn < due
            # ...but once past the adaptive threshold, it is due again
            while srv.replay_lsn - srv.ckpt_lsn < due:
                await srv._do_write({"op": "put", "k": "t%d" % srv.wal.end,
                                     "v": "z" * 200})
            await srv._checkpoint()
            assert srv.ckpt_lsn > first_lsn
        finally:
            if srv._server is not None:
                srv._server.close()
            if srv._flusher is not None:
                srv._flusher.cancel()
            srv.wal.close()
    run(go())


def test_checkpoint_fsyncs_wal_before_publishing(tmp_path):
    """Regression (advisor finding): the checkpoint must fsync the WAL
    BEFORE capturing/publishing its LSN — a checkpoint.lsn beyond the
    durable WAL end would let post-crash recovery reuse LSNs below
    already-streamed positions on the same timeline."""
    async def go():
        from manatee_amd.common.logging import null_logger
        from manatee_amd.db.waldb.server import WaldbServer, CKPT_NAME

        data = str(tmp_path / "n1")
        init_data_dir(data)
        confparser.write(os.path.join(data, "waldb.conf"), {
            "role": "primary", "listen_ip": "127.0.0.1", "port": "0",
            "name": "n1"})
        srv = WaldbServer(data, null_logger())
        await srv.start()
        try:
            for i in range(10):
                await srv._do_write({"op": "put", "k": "k%d" % i,
                                     "v": "v"})
            events = []
            orig_fsync = srv.wal.fsync

            def spying_fsync():
                events.append(srv.wal.end)
                return orig_fsync()

            srv.wal.fsync = spying_fsync
            await srv._checkpoint()
            with open(os.path.join(data, CKPT_NAME)) as f:
                ckpt = json.load(f)
            assert events, "checkpoint did not fsync the WAL first"
            assert events[0] >= ckpt["lsn"], \
                "checkpoint LSN published beyond the fsynced WAL end"
        finally:
            if srv._server is not None:
                srv._server.close()
            if srv._flusher is not None:
                srv._flusher.cancel()
    run(go())


def test_group_commit_pipeline_order_and_gate_break(tmp_path):
    """Group commit: a pipelined burst is answered in request order with
    every record acknowledged after ONE sync-ack wait; and when the
    commit gate breaks mid-burst (read-only flipped while commits are
    pending), the pending writes are answered with errors, in order,
    without wedging the connection."""
    prim = Node(tmp_path, "prim")
    sync = Node(tmp_path, "sync")
    prim.init()
    prim.write_conf(role="primary", sync_name="sync")
    prim.start()
    try:
        import shutil
        shutil.copytree(prim.data_dir, sync.data_dir,
                        ignore=shutil.ignore_patterns("waldb.pid",
                                                      "waldb.conf"))
        sync.write_conf(role="standby",
                        upstream="127.0.0.1:%d" % prim.port)
        sync.start()

        async def burst_ok():
            c = prim.client()
            sc = sync.client()

            async def streaming():
                st = await sc.status()
                return st["upstream_status"] == "streaming"
            await wait_async(streaming, what="sync streaming")
            resps = await c.pipeline(
                [{"q": "put", "k": "gc%d" % i, "v": i}
                 for i in range(60)], timeout_s=10.0)
            assert all(r["ok"] for r in resps)
            # responses are in request order: LSNs strictly increase
            from manatee_amd.common.lsn import parse
            lsns = [parse(r["lsn"]) for r in resps]
            assert lsns == sorted(lsns) and len(set(lsns)) == 60
            assert await c.count(prefix="gc") == 60
            await c.close()
            await sc.close()
        run(burst_ok())

        # break the gate mid-burst: kill the sync, start a pipelined
        # burst (appends immediately, commits pending), then flip
        # read-only via conf reload — the pending burst must be answered
        # with errors, not hang
        sync.kill9()

        async def burst_gate_break():
            c = prim.client()
            burst = asyncio.get_running_loop().create_task(
                c.pipeline([{"q": "put", "k": "rb%d" % i, "v": i}
                            for i in range(10)], timeout_s=20.0))
            await asyncio.sleep(0.5)    # burst is appended + gated
            assert not burst.done()
            prim.write_conf(role="primary", sync_name="sync",
                            read_only=True)
            prim.sighup()
            resps = await asyncio.wait_for(burst, 15.0)
            assert all(not r["ok"] for r in resps)
            assert all("read-only" in r["error"] for r in resps)
            # the connection still serves queries afterwards
            assert await c.count(prefix="gc") == 60
            await c.close()
        run(burst_gate_break())
    finally:
        prim.stop()
        sync.stop()


def test_promote_severs_repl_even_if_cancel_is_lost(tmp_path, monkeypatch):
    """Python 3.10's asyncio.wait_for can LOSE a task cancellation that
    races the inner read's completion (bpo-42130, fixed in 3.12) — and
    under streaming load the replication client lives inside
    wait_for(readexactly(...)).  A promote whose repl-task cancel was
    eaten left the promoted sync streaming from AND ACKING the deposed
    primary, whose sync-commit gate those acks kept open: it went on
    acknowledging writes the new timeline will never contain (split
    brain, observed ~1/6 under CPU load in the zk-partition tests).
    Promote must therefore sever the link without relying on the
    cancel: generation-bump + transport abort.  The WALDB_TEST_
    EAT_REPL_CANCEL seam simulates the lost cancel deterministically;
    on the pre-fix code this test fails (old primary keeps acking)."""
    import shutil
    prim = Node(tmp_path, "prim")
    sync = Node(tmp_path, "sync")
    prim.init()
    prim.write_conf(role="primary", sync_name="sync")
    prim.start()
    monkeypatch.setenv("WALDB_TEST_EAT_REPL_CANCEL", "1")
    try:
        shutil.copytree(prim.data_dir, sync.data_dir,
                        ignore=shutil.ignore_patterns("waldb.pid",
                                                      "waldb.conf"))
        sync.write_conf(role="standby",
                        upstream="127.0.0.1:%d" % prim.port)
        sync.start()

        async def go():
            c = prim.client()
            sc = sync.client()

            async def streaming():
                st = await sc.status()
                return st["upstream_status"] == "streaming"
            await wait_async(streaming, what="sync streaming")
            await c.put("before", 1, timeout_s=5.0)

            # promote the sync (conf role flip + trigger + SIGHUP),
            # with its repl-task cancel suppressed by the seam
            sync.write_conf(role="primary")
            sync.promote_trigger()
            sync.sighup()

            async def promoted():
                st = await sc.status()
                return st["role"] == "primary" and st["timeline"] == 2
            await wait_async(promoted, what="sync promoted")

            # the deposed primary's gate must close: within a short
            # window no further write may be acknowledged (the severed
            # link can carry no more acks)
            deadline = time.monotonic() + 6.0
            while True:
                try:
                    await c.put("after-%f" % time.monotonic(), 1,
                                timeout_s=1.0)
                    acked = True
                except WaldbError:
                    acked = False
                if not acked:
                    break
                assert time.monotonic() < deadline, \
                    "deposed primary still acknowledging writes after " \
                    "promote: the replication link survived"
            # the aborted socket surfaces at the sender's next keepalive
            # (≤5 s): the replica row must then disappear — proof the
            # link is dead at the TCP level, not merely quiet
            async def replica_gone():
                st = await c.status()
                return st["replication"] == []
            await wait_async(replica_gone, timeout=12.0,
                             what="deposed primary drops the dead sender")
            await c.close()
            await sc.close()
        run(go())
    finally:
        prim.stop()
        sync.stop()
