"""minipg server — PostgreSQL process conventions + libpq v3 frontend
over the waldb replication core.

See package docstring for the fidelity contract.  The process is
managed by db/postgres.py + db/manager.py exactly as the reference
manages a real ``postgres`` child (ref lib/postgresMgr.js:1653-1795):
spawned as ``postgres -D dataDir``, killed dirty, reloaded with SIGHUP,
promoted via trigger file, configured only through regenerated conf
files.

Peer replication is multiplexed on the listen port by first byte:
``{`` starts a waldb JSON/replication exchange (PostgreSQL likewise
multiplexes walsender connections on its port), anything else is
parsed as a libpq startup packet.
"""

from __future__ import annotations

import asyncio
import datetime
import os
import re
import struct
import sys
import time
from typing import Dict, List, Optional, Tuple

from ...common import confparser
from ...common import lsn as lsnmod
from ...common.logging import Logger, level_from_verbosity
from ...common.lsn import pg_strip_minor
from ..waldb.server import WaldbServer

_I32 = struct.Struct(">i")
_I16 = struct.Struct(">h")

SSL_REQUEST = 80877103
CANCEL_REQUEST = 80877102
GSSENC_REQUEST = 80877104
PROTOCOL_V3 = 196608

CONF_NAME = "postgresql.conf"
RECOVERY_NAME = "recovery.conf"
SIGNAL_NAME = "standby.signal"


def parse_conninfo(text: str) -> Dict[str, str]:
    """'host=H port=P user=U application_name=A ...' → dict."""
    out: Dict[str, str] = {}
    for part in text.strip().strip("'\"").split():
        k, _, v = part.partition("=")
        if k:
            out[k] = v
    return out


def sql_unquote(text: str) -> str:
    return text.replace("''", "'")


def split_statements(sql: str) -> List[str]:
    """Split a simple-Query string on ';' outside single quotes (the
    multi-statement form of the simple protocol)."""
    out, cur, inq = [], [], False
    for ch in sql:
        if ch == "'":
            inq = not inq
            cur.append(ch)
        elif ch == ";" and not inq:
            out.append("".join(cur))
            cur = []
        else:
            cur.append(ch)
    out.append("".join(cur))
    return [s.strip() for s in out if s.strip()]


class PgSqlError(Exception):
    def __init__(self, code: str, msg: str):
        self.code = code
        self.msg = msg
        super().__init__(msg)


class MinipgServer(WaldbServer):
    PID_FILE = "postmaster.pid"

    def __init__(self, data_dir: str, log: Logger):
        super().__init__(data_dir, log)
        self.log = log.child(component="minipg")
        self._trigger_file: Optional[str] = None
        try:
            with open(os.path.join(self.data_dir, "PG_VERSION")) as f:
                self.pg_major = f.read().strip() or "12"
        except OSError:
            self.pg_major = "12"
        self.lsn_word = "lsn" if float(self.pg_major) >= 10 else "location"

    # -------------------------------------------------- postgres conf model
    def conf_path(self) -> str:
        return os.path.join(self.data_dir, CONF_NAME)

    def trigger_path(self) -> str:
        if self._trigger_file:
            return self._trigger_file
        return os.path.join(os.path.dirname(self.data_dir), "promote")

    def load_conf(self) -> None:
        raw = confparser.read(self.conf_path())
        s = self._strip
        standby = False
        conninfo_s = None
        trigger = None
        rec_path = os.path.join(self.data_dir, RECOVERY_NAME)
        if os.path.exists(os.path.join(self.data_dir, SIGNAL_NAME)):
            # PG ≥12 style: standby.signal + settings in postgresql.conf
            standby = True
            conninfo_s = s(raw.get("primary_conninfo", ""))
            trigger = s(raw.get("promote_trigger_file", ""))
        elif os.path.exists(rec_path):
            # pre-12 style: recovery.conf with standby_mode=on
            rec = confparser.read(rec_path)
            if s(rec.get("standby_mode", "")) == "on":
                standby = True
                conninfo_s = s(rec.get("primary_conninfo", ""))
                trigger = s(rec.get("trigger_file", ""))
        ci = parse_conninfo(conninfo_s) if conninfo_s else {}

        listen = s(raw.get("listen_addresses", "127.0.0.1"))
        if listen in ("*", ""):
            listen = "0.0.0.0"
        self.conf = {
            "role": "standby" if standby else "primary",
            "listen_ip": listen,
            "port": s(raw.get("port", "5432")),
            "name": ci.get("application_name", "standby"),
        }
        self.role = self.conf["role"]
        self.read_only = s(raw.get("default_transaction_read_only",
                                   "off")) == "on"
        sync = s(raw.get("synchronous_standby_names", ""))
        self.sync_standby = sync or None
        self.upstream = None
        if standby and ci.get("host"):
            self.upstream = "%s:%s" % (ci["host"], ci.get("port", "5432"))
        self._trigger_file = trigger or None
        try:
            self.ckpt_wal_bytes = int(s(raw.get(
                "checkpoint_wal_bytes", str(self.ckpt_wal_bytes))))
            self.wal_keep_bytes = int(s(raw.get(
                "wal_keep_bytes", str(self.wal_keep_bytes))))
            self.wal.segment_bytes = int(s(raw.get(
                "wal_segment_bytes", str(self.wal.segment_bytes))))
            # PG expresses wal_sender_timeout in milliseconds
            self.wal_sender_timeout_s = float(s(raw.get(
                "wal_sender_timeout",
                str(int(self.wal_sender_timeout_s * 1000))))) / 1000.0
        except ValueError:
            pass

    # ------------------------------------------------- connection multiplex
    async def _handle_conn(self, reader: asyncio.StreamReader,
                           writer: asyncio.StreamWriter) -> None:
        try:
            first = await reader.readexactly(1)
        except (asyncio.IncompleteReadError, ConnectionError):
            writer.close()
            return
        if first == b"{":
            # waldb JSON exchange (peer replication / internal probes)
            try:
                line = first + await reader.readline()
            except (ConnectionError, asyncio.IncompleteReadError):
                writer.close()
                return
            await self._serve_json(reader, writer, first_line=line)
            return
        await self._serve_libpq(first, reader, writer)

    # ----------------------------------------------------- libpq v3 backend
    @staticmethod
    def _msg(t: bytes, payload: bytes = b"") -> bytes:
        return t + _I32.pack(len(payload) + 4) + payload

    def _row_desc(self, cols: List[str]) -> bytes:
        body = [_I16.pack(len(cols))]
        for c in cols:
            body.append(c.encode() + b"\x00")
            body.append(struct.pack(">ihihih", 0, 0, 25, -1, -1, 0))
        return self._msg(b"T", b"".join(body))

    def _data_row(self, vals: List[Optional[str]]) -> bytes:
        body = [_I16.pack(len(vals))]
        for v in vals:
            if v is None:
                body.append(_I32.pack(-1))
            else:
                b = v.encode("utf-8")
                body.append(_I32.pack(len(b)) + b)
        return self._msg(b"D", b"".join(body))

    def _complete(self, tag: str) -> bytes:
        return self._msg(b"C", tag.encode() + b"\x00")

    def _error(self, code: str, msg: str) -> bytes:
        fields = b"SERROR\x00" + b"C" + code.encode() + b"\x00" + \
            b"M" + msg.encode() + b"\x00" + b"\x00"
        return self._msg(b"E", fields)

    def _ready(self) -> bytes:
        return self._msg(b"Z", b"I")

    async def _serve_libpq(self, first: bytes,
                           reader: asyncio.StreamReader,
                           writer: asyncio.StreamWriter) -> None:
        MAX_MSG = 16 * 1024 * 1024
        try:
            while True:     # startup negotiation (SSL probe then startup)
                rest = await reader.readexactly(3)
                (length,) = struct.unpack(">I", first + rest)
                if length < 4 or length - 4 > MAX_MSG:
                    writer.close()
                    return
                payload = await reader.readexactly(length - 4)
                (code,) = struct.unpack_from(">i", payload)
                if code in (SSL_REQUEST, GSSENC_REQUEST):
                    writer.write(b"N")      # no TLS; client retries plain
                    await writer.drain()
                    first = await reader.readexactly(1)
                    continue
                if code == CANCEL_REQUEST:
                    writer.close()
                    return
                if code != PROTOCOL_V3:
                    writer.write(self._error(
                        "08P01", "unsupported protocol %d" % code))
                    await writer.drain()
                    writer.close()
                    return
                break
            # AuthenticationOk + parameters + ReadyForQuery
            out = [self._msg(b"R", _I32.pack(0))]
            for k, v in (("server_version", self.pg_major + ".0 (minipg)"),
                         ("client_encoding", "UTF8"),
                         ("server_encoding", "UTF8"),
                         ("integer_datetimes", "on")):
                out.append(self._msg(
                    b"S", k.encode() + b"\x00" + v.encode() + b"\x00"))
            out.append(self._msg(b"K", struct.pack(">ii", os.getpid(), 0)))
            out.append(self._ready())
            writer.write(b"".join(out))
            await writer.drain()

            while True:
                hdr = await reader.readexactly(5)
                t = hdr[:1]
                (ln,) = struct.unpack(">I", hdr[1:])
                if ln < 4 or ln - 4 > MAX_MSG:
                    writer.write(self._error(
                        "08P01", "invalid message length %d" % ln))
                    await writer.drain()
                    return
                payload = await reader.readexactly(ln - 4)
                if t == b"X":
                    return
                if t != b"Q":
                    writer.write(self._error(
                        "0A000", "only the simple query protocol is "
                        "supported"))
                    writer.write(self._ready())
                    await writer.drain()
                    continue
                sql = payload.rstrip(b"\x00").decode("utf-8")
                await self._run_query_cycle(sql, writer)
        except (asyncio.IncompleteReadError, ConnectionError,
                asyncio.CancelledError):
            pass
        except Exception as exc:
            self.log.error("libpq connection error", err=exc)
        finally:
            try:
                writer.close()
            except Exception:
                pass

    async def _run_query_cycle(self, sql: str,
                               writer: asyncio.StreamWriter) -> None:
        """One simple-Query cycle: possibly multiple statements, one
        ReadyForQuery at the end; an error aborts the rest (as the real
        backend does).  GROUP COMMIT: every write in the statement list
        is appended immediately and the whole cycle waits once for the
        highest LSN's sync ack, so a multi-statement INSERT batch pays
        one replication round trip, not one per row."""
        staged: List[bytes] = []
        max_lsn = None
        err: Optional[PgSqlError] = None
        for stmt in split_statements(sql) or [""]:
            try:
                chunks, lsn = await self._execute_staged(stmt)
            except PgSqlError as exc:
                err = exc
                break
            staged.extend(chunks)
            if lsn is not None:
                max_lsn = lsn
        if max_lsn is not None:
            res = await self._await_commit(max_lsn)
            if not res.get("ok"):
                # gate broke while waiting (e.g. read-only flipped):
                # nothing in this cycle may be acknowledged
                staged = []
                err = err or PgSqlError("25006", res.get("error",
                                                         "commit failed"))
        for chunk in staged:
            writer.write(chunk)
        if err is not None:
            writer.write(self._error(err.code, err.msg))
        writer.write(self._ready())
        await writer.drain()

    async def _execute_staged(self, stmt: str
                              ) -> Tuple[List[bytes], Optional[int]]:
        """Writes append-without-waiting (their acks are withheld until
        the cycle's single commit gate); everything else executes
        directly."""
        m = self._RE_INSERT.search(stmt)
        if m:
            lsn = self._append_sql(
                {"op": "put", "k": sql_unquote(m.group(1)),
                 "v": sql_unquote(m.group(2))}, "INSERT")
            return [self._complete("INSERT 0 1")], lsn
        m = self._RE_DELETE.search(stmt)
        if m:
            k = sql_unquote(m.group(1))
            if k not in self.kv:
                return [self._complete("DELETE 0")], None
            lsn = self._append_sql({"op": "del", "k": k}, "DELETE")
            return [self._complete("DELETE 1")], lsn
        return await self._execute(stmt), None

    def _append_sql(self, op: dict, verb: str) -> int:
        lsn, err = self._append_write(op)
        if err is not None:
            msg = err.get("error", "")
            if "read-only" in msg:
                raise PgSqlError(
                    "25006",
                    "cannot execute %s in a read-only transaction" % verb)
            raise PgSqlError("XX000", msg or "write failed")
        return lsn

    # ------------------------------------------------------------ SQL layer
    _RE_INSERT = re.compile(
        r"insert\s+into\s+kv\s*\(\s*k\s*,\s*v\s*\)\s*values\s*\(\s*"
        r"'((?:[^']|'')*)'\s*,\s*'((?:[^']|'')*)'\s*\)", re.I)
    _RE_SELECT_V = re.compile(
        r"select\s+v\s+from\s+kv\s+where\s+k\s*=\s*'((?:[^']|'')*)'", re.I)
    _RE_COUNT = re.compile(
        r"select\s+count\(\*\)\s+(?:as\s+\w+\s+)?from\s+kv"
        r"(?:\s+where\s+k\s+like\s+'((?:[^']|'')*)%')?", re.I)
    _RE_DELETE = re.compile(
        r"delete\s+from\s+kv\s+where\s+k\s*=\s*'((?:[^']|'')*)'", re.I)

    async def _execute(self, stmt: str) -> List[bytes]:
        low = " ".join(stmt.lower().split())
        if not low:
            return [self._msg(b"I")]        # EmptyQueryResponse

        if low.startswith("create table"):
            return [self._complete("CREATE TABLE")]

        if "current_time" in low and "from" not in low:
            now = datetime.datetime.now(datetime.timezone.utc)
            return [self._row_desc(["current_time"]),
                    self._data_row([now.strftime("%H:%M:%S.%f%z")]),
                    self._complete("SELECT 1")]

        if "pg_is_in_recovery" in low:
            return [self._row_desc(["r"]),
                    self._data_row(["t" if self.role == "standby"
                                    else "f"]),
                    self._complete("SELECT 1")]

        if "pg_current_wal_lsn" in low or "pg_current_xlog_location" in low:
            if self.role == "standby":
                raise PgSqlError("55000", "recovery is in progress")
            return [self._row_desc(["loc"]),
                    self._data_row([lsnmod.format_lsn(self.wal.end)]),
                    self._complete("SELECT 1")]

        if "pg_last_wal_replay_lsn" in low or \
                "pg_last_xlog_replay_location" in low:
            val = (lsnmod.format_lsn(self.replay_lsn)
                   if self.role == "standby" else None)
            return [self._row_desc(["loc"]), self._data_row([val]),
                    self._complete("SELECT 1")]

        if "pg_last_xact_replay_timestamp" in low:
            val = ("%f" % self.last_replay_time
                   if (self.role == "standby" and self.last_replay_time)
                   else None)
            return [self._row_desc(["t"]), self._data_row([val]),
                    self._complete("SELECT 1")]

        if "pg_stat_wal_receiver" in low:
            # the standby-side receiver status (real PG ≥9.6): one row
            # while this server is a standby.  'diverged' is a minipg
            # extension (real PG surfaces divergence only in its log);
            # the manager uses it to trigger the restore-on-divergence
            # path (ref standby-failure ⇒ restore :1339-1373).
            cols = ["status", "conninfo"]
            out = [self._row_desc(cols)]
            if self.role == "standby":
                status = {"streaming": "streaming",
                          "diverged": "diverged"}.get(
                              self.upstream_status, "stopped")
                out.append(self._data_row(
                    [status, "host=%s" % (self.upstream or "")]))
                out.append(self._complete("SELECT 1"))
            else:
                out.append(self._complete("SELECT 0"))
            return out

        if "pg_stat_replication" in low:
            return self._stat_replication()

        m = self._RE_SELECT_V.search(stmt)
        if m:
            k = sql_unquote(m.group(1))
            if k in self.kv:
                return [self._row_desc(["v"]),
                        self._data_row([str(self.kv[k])]),
                        self._complete("SELECT 1")]
            return [self._row_desc(["v"]), self._complete("SELECT 0")]
        m = self._RE_COUNT.search(stmt)
        if m:
            prefix = sql_unquote(m.group(1) or "")
            n = sum(1 for k in self.kv if k.startswith(prefix)) \
                if prefix else len(self.kv)
            return [self._row_desc(["count"]), self._data_row([str(n)]),
                    self._complete("SELECT 1")]
        raise PgSqlError("42601", 'syntax error at or near "%s"'
                         % stmt.split()[0][:40])

    def _stat_replication(self) -> List[bytes]:
        w = self.lsn_word
        cols = ["pid", "application_name", "client_addr", "state",
                "sent_" + w, "write_" + w, "flush_" + w, "replay_" + w,
                "sync_state"]
        out = [self._row_desc(cols)]
        n = 0
        for rep in self.replicas:
            addr = None
            try:
                peer = rep.writer.get_extra_info("peername")
                addr = peer[0] if peer else None
            except Exception:
                pass
            out.append(self._data_row([
                str(os.getpid()), rep.name, addr, "streaming",
                lsnmod.format_lsn(rep.sent_lsn),
                lsnmod.format_lsn(rep.write_lsn),
                lsnmod.format_lsn(rep.flush_lsn),
                lsnmod.format_lsn(rep.replay_lsn),
                "sync" if rep.name == self.sync_standby else "async",
            ]))
            n += 1
        out.append(self._complete("SELECT %d" % n))
        return out


# ---------------------------------------------------------------- binaries

def init_data_dir(data_dir: str, version: str) -> None:
    """The initdb analogue: PG_VERSION (major), system identity,
    timeline 1 (ref _prepareDatabase lib/postgresMgr.js:1806-1987)."""
    from ..waldb.server import init_data_dir as waldb_init
    os.makedirs(data_dir, exist_ok=True)
    waldb_init(data_dir)
    with open(os.path.join(data_dir, "PG_VERSION"), "w") as f:
        f.write(pg_strip_minor(version) + "\n")
    os.chmod(data_dir, 0o700)


def initdb_main(version: str, argv=None) -> int:
    import argparse
    ap = argparse.ArgumentParser(prog="initdb (minipg)")
    ap.add_argument("-D", "--pgdata", required=True)
    ap.add_argument("-E", "--encoding", default="UTF8")
    ap.add_argument("-U", "--username", default=None)
    ns, _ = ap.parse_known_args(argv)
    if os.path.exists(os.path.join(ns.pgdata, "PG_VERSION")):
        print("initdb: directory %s is not empty" % ns.pgdata,
              file=sys.stderr)
        return 1
    init_data_dir(ns.pgdata, version)
    print("Success. You can now start the database server.")
    return 0


def postgres_main(version: str, argv=None) -> int:
    import argparse
    import signal
    ap = argparse.ArgumentParser(prog="postgres (minipg)")
    ap.add_argument("-D", "--pgdata", required=True)
    ns, _ = ap.parse_known_args(argv)
    data_dir = os.path.abspath(ns.pgdata)
    log = Logger("minipg", level=level_from_verbosity(1),
                 path=os.path.join(os.path.dirname(data_dir),
                                   "minipg.log"))

    async def run():
        srv = MinipgServer(data_dir, log)
        await srv.start()
        stop = asyncio.Event()
        loop = asyncio.get_running_loop()
        # dirty-kill discipline: exit fast on any stop signal; the WAL
        # is the only durability story (MANATEE-188)
        for sig in (signal.SIGINT, signal.SIGTERM, signal.SIGQUIT):
            loop.add_signal_handler(sig, stop.set)
        await stop.wait()
        return 0

    return asyncio.run(run())


def main(argv=None) -> int:
    """module entry: ``python -m manatee_amd.db.minipg -D dir
    [--init] [--pg-version 12.0]``."""
    import argparse
    ap = argparse.ArgumentParser(prog="minipg")
    ap.add_argument("-D", "--pgdata", required=True)
    ap.add_argument("--init", action="store_true")
    ap.add_argument("--pg-version", default="12.0")
    ns = ap.parse_args(argv)
    if ns.init:
        return initdb_main(ns.pg_version, ["-D", ns.pgdata])
    return postgres_main(ns.pg_version, ["-D", ns.pgdata])


if __name__ == "__main__":
    sys.exit(main())
