"""Tier-3 component tests: ZK client against the embedded server.

Mirrors the reference's test/zookeeperMgr.test.js coverage (election join,
CAS state writes, watches, session expiry) but runs hermetically against
our own wire-protocol server instead of requiring a live ensemble.
"""

import asyncio
import json

import pytest

from manatee_amd.coord import jute
from manatee_amd.coord.zkclient import ZkClient
from manatee_amd.coord.zkserver import ZkServer


def run(coro):
    return asyncio.run(coro)


async def _pair(**kw):
    srv = ZkServer(**kw)
    await srv.start()
    cli = ZkClient(srv.conn_str, session_timeout_ms=4000)
    await cli.connect()
    return srv, cli


def test_basic_crud():
    async def go():
        srv, cli = await _pair()
        try:
            await cli.mkdirp("/manatee/1.moray/election")
            path = await cli.create("/manatee/1.moray/state",
                                    b'{"generation":1}')
            assert path == "/manatee/1.moray/state"
            data, stat = await cli.get_data("/manatee/1.moray/state")
            assert data == b'{"generation":1}'
            assert stat.version == 0
            st2 = await cli.set_data("/manatee/1.moray/state", b'{"generation":2}',
                                     version=0)
            assert st2.version == 1
            with pytest.raises(jute.ZkError) as ei:
                await cli.set_data("/manatee/1.moray/state", b"x", version=0)
            assert ei.value.code == jute.ZBADVERSION
            assert (await cli.exists("/manatee/1.moray/state")) is not None
            assert (await cli.exists("/nope")) is None
            kids, _ = await cli.get_children("/manatee/1.moray")
            assert kids == ["election", "state"]
            await cli.delete("/manatee/1.moray/state")
            assert (await cli.exists("/manatee/1.moray/state")) is None
        finally:
            await cli.close()
            await srv.stop()
    run(go())


def test_ephemeral_sequential_election():
    async def go():
        srv, cli = await _pair()
        cli2 = ZkClient(srv.conn_str, session_timeout_ms=4000)
        await cli2.connect()
        try:
            await cli.mkdirp("/shard/election")
            p1 = await cli.create("/shard/election/10.0.0.1:5432:12345-",
                                  b'{"ip":"10.0.0.1"}',
                                  mode=jute.EPHEMERAL_SEQUENTIAL)
            p2 = await cli2.create("/shard/election/10.0.0.2:5432:12345-",
                                   b'{"ip":"10.0.0.2"}',
                                   mode=jute.EPHEMERAL_SEQUENTIAL)
            assert p1.endswith("-0000000000")
            assert p2.endswith("-0000000001")
            kids, _ = await cli.get_children("/shard/election")
            assert len(kids) == 2
            # closing cli2's session removes its ephemeral
            await cli2.close()
            await asyncio.sleep(0.1)
            kids, _ = await cli.get_children("/shard/election")
            assert kids == [p1.rsplit("/", 1)[1]]
        finally:
            await cli.close()
            await srv.stop()
    run(go())


def test_one_shot_watches():
    async def go():
        srv, cli = await _pair()
        watcher = ZkClient(srv.conn_str, session_timeout_ms=4000)
        await watcher.connect()
        events = []
        try:
            await cli.mkdirp("/shard")
            await cli.create("/shard/state", b"v0")

            loop = asyncio.get_running_loop()
            fired = loop.create_future()

            def on_data(etype, path):
                events.append((etype, path))
                if not fired.done():
                    fired.set_result(None)

            await watcher.get_data("/shard/state", watch=on_data)
            await cli.set_data("/shard/state", b"v1")
            await asyncio.wait_for(fired, 2)
            assert events == [(jute.EVENT_NODE_DATA_CHANGED, "/shard/state")]
            # one-shot: second change does not re-fire
            await cli.set_data("/shard/state", b"v2")
            await asyncio.sleep(0.2)
            assert len(events) == 1

            # child watch
            fired2 = loop.create_future()

            def on_child(etype, path):
                events.append((etype, path))
                if not fired2.done():
                    fired2.set_result(None)

            await watcher.get_children("/shard", watch=on_child)
            await cli.create("/shard/x", b"")
            await asyncio.wait_for(fired2, 2)
            assert events[-1] == (jute.EVENT_NODE_CHILDREN_CHANGED, "/shard")
        finally:
            await watcher.close()
            await cli.close()
            await srv.stop()
    run(go())


def test_multi_transaction_cas():
    """putClusterState shape: create history node + versioned setData on
    state, atomically (ref zookeeperMgr.js:605-630)."""
    async def go():
        srv, cli = await _pair()
        try:
            await cli.mkdirp("/shard/history")
            await cli.create("/shard/state", b'{"generation":1}')
            state2 = json.dumps({"generation": 2}).encode()
            res = await cli.multi([
                jute.MultiOp.create("/shard/history/2-", state2,
                                    jute.PERSISTENT_SEQUENTIAL),
                jute.MultiOp.set_data("/shard/state", state2, version=0),
            ])
            assert res[0][0] == "create"
            assert res[0][1].startswith("/shard/history/2-")
            assert res[1][0] == "setData"
            assert res[1][1].version == 1

            # stale version: whole transaction fails, nothing applied
            with pytest.raises(jute.ZkError):
                await cli.multi([
                    jute.MultiOp.create("/shard/history/3-", b"x",
                                        jute.PERSISTENT_SEQUENTIAL),
                    jute.MultiOp.set_data("/shard/state", b"x", version=0),
                ])
            kids, _ = await cli.get_children("/shard/history")
            assert len(kids) == 1
            data, _ = await cli.get_data("/shard/state")
            assert data == state2
        finally:
            await cli.close()
            await srv.stop()
    run(go())


def test_session_expiry_removes_ephemerals():
    async def go():
        srv = ZkServer(tick_ms=50, min_session_timeout_ms=200)
        await srv.start()
        cli = ZkClient(srv.conn_str, session_timeout_ms=300)
        await cli.connect()
        watcher = ZkClient(srv.conn_str, session_timeout_ms=4000)
        await watcher.connect()
        try:
            await watcher.mkdirp("/shard/election")
            path = await cli.create("/shard/election/a-", b"",
                                    mode=jute.EPHEMERAL_SEQUENTIAL)
            assert (await watcher.exists(path)) is not None
            # sever the TCP connection without closing the session, then
            # stop the client's reconnect machinery by cancelling its mgr
            cli._closing = True
            cli._writer.close()
            # wait past the session timeout: server must expire + clean up
            deadline = asyncio.get_running_loop().time() + 3
            while await watcher.exists(path) is not None:
                assert asyncio.get_running_loop().time() < deadline, \
                    "ephemeral not removed after session expiry"
                await asyncio.sleep(0.05)
            assert srv.stats["expired_sessions"] >= 1
        finally:
            await cli.close()
            await watcher.close()
            await srv.stop()
    run(go())


def test_reconnect_keeps_session():
    async def go():
        srv = ZkServer(tick_ms=50, min_session_timeout_ms=200)
        await srv.start()
        cli = ZkClient(srv.conn_str, session_timeout_ms=2000)
        await cli.connect()
        try:
            await cli.mkdirp("/shard")
            path = await cli.create("/shard/eph", b"",
                                    mode=jute.EPHEMERAL)
            sid = cli.session_id
            # drop the TCP connection; client should reconnect with same sid
            cli._writer.close()
            await asyncio.sleep(0.3)
            deadline = asyncio.get_running_loop().time() + 3
            while cli.state != "connected":
                assert asyncio.get_running_loop().time() < deadline
                await asyncio.sleep(0.05)
            assert cli.session_id == sid
            assert (await cli.exists(path)) is not None
        finally:
            await cli.close()
            await srv.stop()
    run(go())


def test_journal_restart(tmp_path):
    jp = str(tmp_path / "zk-journal.jsonl")

    async def phase1():
        srv = ZkServer(journal_path=jp)
        await srv.start()
        cli = ZkClient(srv.conn_str)
        await cli.connect()
        await cli.mkdirp("/shard/history")
        await cli.create("/shard/state", b'{"generation":3}')
        await cli.create("/shard/eph", b"", mode=jute.EPHEMERAL)
        await cli.close()
        await srv.stop()

    async def phase2():
        srv = ZkServer(journal_path=jp)
        await srv.start()
        cli = ZkClient(srv.conn_str)
        await cli.connect()
        data, _ = await cli.get_data("/shard/state")
        assert data == b'{"generation":3}'
        # ephemerals do not survive restart
        assert (await cli.exists("/shard/eph")) is None
        await cli.close()
        await srv.stop()

    run(phase1())
    run(phase2())


def test_journal_sequential_replay_and_compaction(tmp_path):
    """Persistent-sequential nodes must replay with their FINAL names
    (no re-sequencing) and the parent's counter must continue after
    them; the journal compacts to a tree snapshot once it outgrows the
    node count and the compacted journal replays identically."""
    from manatee_amd.coord import jute

    j = str(tmp_path / "zk.jsonl")

    async def go():
        srv = ZkServer(journal_path=j)
        await srv.start()
        cli = ZkClient(srv.conn_str)
        await cli.connect()
        await cli.mkdirp("/hist")
        await cli.create("/hist/gen-", b"a", jute.PERSISTENT_SEQUENTIAL)
        await cli.create("/hist/gen-", b"b", jute.PERSISTENT_SEQUENTIAL)
        await cli.close()
        await srv.stop()

        srv2 = ZkServer(host=srv.host, port=srv.port, journal_path=j)
        await srv2.start()
        cli2 = ZkClient(srv2.conn_str)
        await cli2.connect()
        ch, _ = await cli2.get_children("/hist")
        assert sorted(ch) == ["gen-0000000000", "gen-0000000001"], ch
        p3 = await cli2.create("/hist/gen-", b"c",
                               jute.PERSISTENT_SEQUENTIAL)
        assert p3 == "/hist/gen-0000000002"

        srv2.journal_compact_entries = 10
        for i in range(30):
            await cli2.set_data("/hist/gen-0000000000", b"x%d" % i)
        with open(j) as f:
            assert sum(1 for _ in f) < 30, "journal did not compact"
        await cli2.close()
        await srv2.stop()

        srv3 = ZkServer(host=srv.host, port=srv.port, journal_path=j)
        await srv3.start()
        cli3 = ZkClient(srv3.conn_str)
        await cli3.connect()
        data, _ = await cli3.get_data("/hist/gen-0000000000")
        assert data == b"x29"
        ch, _ = await cli3.get_children("/hist")
        assert len(ch) == 3
        await cli3.close()
        await srv3.stop()
    run(go())


def test_ensemble_conn_string_skips_dead_servers():
    """A multi-server connection string works when some addresses are
    down: the client rotates to a live server (the ensemble behavior
    the production configs rely on)."""
    async def go():
        srv = ZkServer()
        await srv.start()
        dead1 = "127.0.0.1:1"          # never listening
        dead2 = "127.0.0.1:2"
        conn = ",".join([dead1, dead2, srv.conn_str])
        cli = ZkClient(conn, session_timeout_ms=4000)
        await cli.connect(timeout_s=15)
        await cli.mkdirp("/ens")
        data, _ = await cli.get_data("/ens")
        assert data == b""
        await cli.close()
        await srv.stop()
    run(go())


def test_connection_manager_dies_on_bare_cancel():
    """asyncio.run's shutdown path cancels every leftover task ONCE and
    gathers them with no timeout.  The connection manager must
    therefore terminate on a bare cancel even mid-session (without
    close() having set _closing): it used to swallow the CancelledError
    raised through the io-task shield and loop back into reconnecting,
    surviving the cancel forever and wedging interpreter shutdown —
    observed as a whole pytest process hanging AFTER a partition test
    had already passed."""
    async def go():
        srv, cli = await _pair()
        try:
            assert cli.state == "connected"
            cli._mgr_task.cancel()
            done, pending = await asyncio.wait({cli._mgr_task}, timeout=5)
            assert not pending, \
                "connection manager survived a bare cancel"
        finally:
            await cli.close()
            await srv.stop()
    run(go())
