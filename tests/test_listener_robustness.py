"""Malformed-input robustness for every TCP listener: random garbage,
truncated frames, absurd lengths and abrupt disconnects must never
crash a daemon or wedge its service for other clients."""

import asyncio
import json
import os
import random
import struct

import pytest

from manatee_amd.common.logging import null_logger
from manatee_amd.coord.zkclient import ZkClient
from manatee_amd.coord.zkserver import ZkServer


def run(coro, timeout=120):
    return asyncio.run(asyncio.wait_for(coro, timeout))


GARBAGE = [
    b"",
    b"\x00",
    b"GET / HTTP/1.1\r\n\r\n",
    b"\xff" * 64,
    struct.pack(">i", -5),                      # negative frame length
    struct.pack(">i", 2 ** 31 - 1) + b"x",      # absurd frame length
    b"{not json\n",
    b'{"q":"repl"}\n',                          # repl without fields
    b'{"q":"put"}\n' * 50,
    os.urandom(512),
]


async def _throw_garbage(host, port, rng):
    for payload in GARBAGE + [os.urandom(rng.randint(1, 2048))
                              for _ in range(10)]:
        try:
            reader, writer = await asyncio.open_connection(host, port)
        except OSError:
            raise AssertionError("listener died")
        try:
            writer.write(payload)
            await writer.drain()
            try:
                await asyncio.wait_for(reader.read(256), 0.2)
            except asyncio.TimeoutError:
                pass
        except (ConnectionError, OSError):
            pass           # server may reset us; it must not die
        finally:
            try:
                writer.close()
            except Exception:
                pass


def test_zkserver_survives_garbage():
    async def go():
        rng = random.Random(7)
        srv = ZkServer()
        await srv.start()
        try:
            host, port = srv.conn_str.split(":")
            await _throw_garbage(host, int(port), rng)
            # a REAL client still gets full service afterwards
            cli = ZkClient(srv.conn_str, session_timeout_ms=4000)
            await cli.connect(timeout_s=5)
            await cli.mkdirp("/garbage/after")
            await cli.create("/garbage/after/x", b"ok")
            data, _ = await cli.get_data("/garbage/after/x")
            assert data == b"ok"
            await cli.close()
        finally:
            await srv.stop()
    run(go())


def test_waldb_survives_garbage(tmp_path):
    async def go():
        from manatee_amd.common import confparser
        from manatee_amd.db.waldb.client import WaldbClient
        from manatee_amd.db.waldb.server import WaldbServer, init_data_dir

        rng = random.Random(8)
        data = str(tmp_path / "db")
        init_data_dir(data)
        confparser.write(os.path.join(data, "waldb.conf"), {
            "role": "primary", "listen_ip": "127.0.0.1", "port": "0",
            "name": "n1"})
        srv = WaldbServer(data, null_logger())
        await srv.start()
        try:
            with open(os.path.join(data, "waldb.pid")) as f:
                port = int(f.read().split()[1])
            await _throw_garbage("127.0.0.1", port, rng)
            cli = WaldbClient("127.0.0.1", port)
            await cli.put("after-garbage", 1)
            assert await cli.get("after-garbage") == 1
            await cli.close()
        finally:
            if srv._server is not None:
                srv._server.close()
            if srv._flusher is not None:
                srv._flusher.cancel()
    run(go())


def test_minipg_survives_garbage(tmp_path):
    async def go():
        from manatee_amd.common import confparser
        from manatee_amd.db.minipg.server import (MinipgServer,
                                                  init_data_dir)
        from manatee_amd.db.pgwire import PgClient

        rng = random.Random(9)
        data = str(tmp_path / "pg")
        init_data_dir(data, "12.0")
        confparser.write(os.path.join(data, "postgresql.conf"), {
            "listen_addresses": "'127.0.0.1'", "port": "0"})
        srv = MinipgServer(data, null_logger())
        await srv.start()
        try:
            with open(os.path.join(data, "postmaster.pid")) as f:
                port = int(f.read().split()[1])
            await _throw_garbage("127.0.0.1", port, rng)
            cli = PgClient("127.0.0.1", port, "postgres")
            await cli.connect()
            r = await cli.query("SELECT pg_is_in_recovery() as r;")
            assert r.rows[0][0] == "f"
            await cli.close()
        finally:
            if srv._server is not None:
                srv._server.close()
            if srv._flusher is not None:
                srv._flusher.cancel()
    run(go())


def test_httpd_survives_garbage():
    async def go():
        from manatee_amd.common.httpd import HttpServer, http_request

        rng = random.Random(10)
        srv = HttpServer("127.0.0.1", 0, log=null_logger())

        async def ping(*a, **k):
            return 200, {"ok": True}
        srv.route("GET", "ping", ping)
        await srv.start()
        try:
            await _throw_garbage("127.0.0.1", srv.port, rng)
            status, body = await http_request(
                "http://127.0.0.1:%d/ping" % srv.port)
            assert status == 200 and body["ok"]
        finally:
            await srv.stop()
    run(go())
