"""Backup service tier (ref lib/backupServer.js / backupQueue.js /
backupSender.js): REST job lifecycle and the sender's stall defense —
a receiver that stops reading must fail ITS job without wedging the
serial sender queue."""

import asyncio
import os

import pytest

from manatee_amd.backup.service import (BackupJob, BackupQueue,
                                        BackupSender, BackupServer)
from manatee_amd.common.httpd import http_request
from manatee_amd.common.logging import null_logger
from manatee_amd.storage.dirstore import DirStore


def run(coro, timeout=60):
    return asyncio.run(asyncio.wait_for(coro, timeout))


async def _store_with_data(tmp_path, nbytes=512 * 1024):
    st = DirStore(str(tmp_path / "store"), log=null_logger())
    await st.ensure()
    with open(os.path.join(st.mountpoint(), "blob.bin"), "wb") as f:
        f.write(os.urandom(nbytes))
    await st.snapshot("1000000000001")
    return st


def test_rest_job_lifecycle_and_stream(tmp_path):
    async def go():
        st = await _store_with_data(tmp_path)
        q = BackupQueue()
        sender = BackupSender(st, q, log=null_logger())
        srv = BackupServer("127.0.0.1", 0, q, log=null_logger())
        await srv.start()
        sender.start()

        received = bytearray()
        done = asyncio.Event()

        async def on_conn(reader, writer):
            while True:
                chunk = await reader.read(1 << 16)
                if not chunk:
                    break
                received.extend(chunk)
            writer.close()
            done.set()

        listener = await asyncio.start_server(on_conn, "127.0.0.1", 0)
        port = listener.sockets[0].getsockname()[1]
        try:
            status, resp = await http_request(
                "http://127.0.0.1:%d/backup" % srv.port, "POST",
                {"host": "127.0.0.1", "port": port})
            assert status == 200 and "jobid" in resp
            job_url = "http://127.0.0.1:%d%s" % (srv.port,
                                                 resp["jobPath"])
            deadline = asyncio.get_running_loop().time() + 20
            while True:
                jstatus, job = await http_request(job_url)
                assert jstatus == 200
                if job["done"]:
                    break
                assert asyncio.get_running_loop().time() < deadline
                await asyncio.sleep(0.05)
            assert not job["failed"]
            assert job["snapshot"] == "1000000000001"
            await asyncio.wait_for(done.wait(), 10)
            assert len(received) == job["completed"] > 0
            # unknown job → 404
            jstatus, _ = await http_request(
                "http://127.0.0.1:%d/backup/nope" % srv.port)
            assert jstatus == 404
        finally:
            listener.close()
            await sender.stop()
            await srv.stop()
    run(go())


def test_stalled_receiver_fails_job_without_wedging_queue(tmp_path):
    """The receiver accepts the stream and then never reads (the
    partitioned-mid-restore shape): its job must FAIL within the stall
    timeout, and the NEXT job must still be served — one dead receiver
    may not block every future bootstrap from this peer."""
    async def go():
        # the payload must overflow every OS socket buffer on the way,
        # or the 'stalled' receiver silently absorbs the whole stream
        st = await _store_with_data(tmp_path, nbytes=64 * 1024 * 1024)
        q = BackupQueue()
        sender = BackupSender(st, q, log=null_logger(),
                              stall_timeout_s=1.0)
        sender.start()

        async def black_hole(reader, writer):
            await asyncio.sleep(3600)   # accept, never read

        import socket as socketmod
        bh_sock = socketmod.socket()
        bh_sock.setsockopt(socketmod.SOL_SOCKET,
                           socketmod.SO_RCVBUF, 4096)
        bh_sock.bind(("127.0.0.1", 0))
        stalled = await asyncio.start_server(black_hole, sock=bh_sock)
        sport = stalled.sockets[0].getsockname()[1]

        good = bytearray()
        good_done = asyncio.Event()

        async def healthy(reader, writer):
            while True:
                chunk = await reader.read(1 << 16)
                if not chunk:
                    break
                good.extend(chunk)
            good_done.set()
            writer.close()

        ok_srv = await asyncio.start_server(healthy, "127.0.0.1", 0)
        gport = ok_srv.sockets[0].getsockname()[1]
        try:
            bad = BackupJob("127.0.0.1", sport)
            q.push(bad)
            ok = BackupJob("127.0.0.1", gport)
            q.push(ok)

            deadline = asyncio.get_running_loop().time() + 20
            while not (bad.done and ok.done):
                assert asyncio.get_running_loop().time() < deadline, \
                    (bad.as_dict(), ok.as_dict())
                await asyncio.sleep(0.1)
            assert bad.failed and "stalled" in (bad.error or "")
            assert not ok.failed
            await asyncio.wait_for(good_done.wait(), 10)
            assert len(good) > 0
        finally:
            stalled.close()
            ok_srv.close()
            await sender.stop()
    run(go())


def test_restore_fails_fast_when_source_freezes_mid_stream(tmp_path):
    """A restore whose SOURCE peer stops responding mid-stream (SIGSTOP,
    partition) must fail within ~one job-poll timeout so the FSM can
    re-evaluate — not hang for the full restore deadline.  This is the
    guard that keeps a takeover candidate from being wedged by a frozen
    restore source."""
    import json as _json
    import time as _time

    from manatee_amd.backup.restore import RestoreClient, RestoreError

    async def go():
        st = DirStore(str(tmp_path / "dst"), log=null_logger())

        # a minimal "backup source" that accepts the job, streams a few
        # bytes to the restore listener, then freezes: the job-status
        # poll connection is accepted but NEVER answered
        async def handle(reader, writer):
            req = await reader.readuntil(b"\r\n\r\n")
            line = req.split(b"\r\n", 1)[0].decode()
            if line.startswith("POST /backup"):
                hdrs = req.decode()
                n = int([h for h in hdrs.split("\r\n")
                         if h.lower().startswith("content-length")]
                        [0].split(":")[1])
                body = _json.loads((await reader.readexactly(n)).decode())
                resp = _json.dumps({"jobid": "j1",
                                    "jobPath": "/backup/j1"}).encode()
                writer.write(b"HTTP/1.1 200 OK\r\nContent-Type: "
                             b"application/json\r\nContent-Length: %d"
                             b"\r\n\r\n%s" % (len(resp), resp))
                await writer.drain()

                async def stream():
                    r2, w2 = await asyncio.open_connection(
                        body["host"], body["port"])
                    w2.write(b"partial-tar-bytes")
                    await w2.drain()
                    # ... then freeze: never close, never send more
                    await asyncio.sleep(3600)
                asyncio.ensure_future(stream())
            else:
                # job-status poll: accepted, never answered (frozen)
                await asyncio.sleep(3600)

        src = await asyncio.start_server(handle, "127.0.0.1", 0)
        src_port = src.sockets[0].getsockname()[1]
        rc = RestoreClient(st, "127.0.0.1", poll_interval_s=0.2,
                           log=null_logger())
        t0 = _time.monotonic()
        with pytest.raises(Exception):
            await rc.restore("http://127.0.0.1:%d" % src_port,
                             timeout_s=300.0, isolate=False)
        dt = _time.monotonic() - t0
        assert dt < 30, "restore hung %.1fs against a frozen source" % dt
        assert rc.restore_object.failed
        src.close()
    run(go(), timeout=90)
