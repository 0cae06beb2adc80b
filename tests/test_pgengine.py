"""PostgresEngine: version resolution, wal translations, tunables
layering, and conf generation across PG majors (ref lib/postgresMgr.js
resolveWalTranslations/getTunables/getVersionInfo/_updateUpstreamConf)
— all pure logic, no PostgreSQL binaries needed.  Plus the pgwire
client against a scripted fake backend."""

import asyncio
import json
import os
import struct

import pytest

from manatee_amd.common import confparser
from manatee_amd.db.postgres import (
    PostgresEngine, get_tunables, get_version_info,
    resolve_wal_translations)


def test_wal_translations():
    for major in ("9.2", "9.6"):
        t = resolve_wal_translations(major)
        assert t["lsn"] == "location" and t["wal"] == "xlog"
        assert "pg_current_xlog_location" in t["queries"]["current_lsn"]
        assert "pg_last_xlog_replay_location" in \
            t["queries"]["last_replay_lsn"]
    t = resolve_wal_translations("12")
    assert t["lsn"] == "lsn" and t["wal"] == "wal"
    assert "pg_current_wal_lsn" in t["queries"]["current_lsn"]


def test_tunables_layering():
    tun = {"common": {"shared_buffers": "1GB", "work_mem": "1MB"},
           "9.6": {"work_mem": "2MB"},
           "9.6.3": {"work_mem": "3MB", "extra": 7}}
    out = get_tunables(tun, "9.6.3", "9.6")
    assert out["shared_buffers"] == "1GB"
    assert out["work_mem"] == "3MB"            # full version wins
    assert out["extra"] == "7"
    assert out["synchronous_commit"] == "remote_write"
    out = get_tunables(tun, "9.6.9", "9.6")
    assert out["work_mem"] == "2MB"            # major layer only


def test_version_info_cases(tmp_path):
    data = str(tmp_path / "data")
    dconf = str(tmp_path / "manatee-config.json")
    versions = {"9.2": "9.2.4", "9.6": "9.6.3", "12": "12.0"}
    os.makedirs(data)

    # fresh dataset → default version
    vi = get_version_info(data, dconf, versions, "12")
    assert vi == {"initialized": "12.0", "current": "12.0"}

    # legacy dataset: PG_VERSION exists but no manatee-config → 9.2
    with open(os.path.join(data, "PG_VERSION"), "w") as f:
        f.write("9.2\n")
    vi = get_version_info(data, dconf, versions, "12")
    assert vi == {"initialized": "9.2.4", "current": "9.2.4"}

    # recorded version drives the choice
    with open(dconf, "w") as f:
        json.dump({"initialized": "9.2.4", "current": "9.6.3"}, f)
    with open(os.path.join(data, "PG_VERSION"), "w") as f:
        f.write("9.6\n")
    vi = get_version_info(data, dconf, versions, "12")
    assert vi["current"] == "9.6.3"

    # PG_VERSION / recorded mismatch is fatal
    with open(os.path.join(data, "PG_VERSION"), "w") as f:
        f.write("12\n")
    with pytest.raises(ValueError):
        get_version_info(data, dconf, versions, "12")


def test_version_info_torn_config_recovers(tmp_path):
    """A 0-byte or half-written manatee-config.json (kill -9 caught the
    pre-atomic rewrite, or a snapshot copied it mid-write) must be
    reconstructed from PG_VERSION — it previously raised
    JSONDecodeError on every transition, wedging the peer read-only
    forever (found by the 15-step engine=postgres bench)."""
    data = str(tmp_path / "data")
    dconf = str(tmp_path / "manatee-config.json")
    versions = {"9.6": "9.6.3", "12": "12.0"}
    os.makedirs(data)
    with open(os.path.join(data, "PG_VERSION"), "w") as f:
        f.write("12\n")

    for torn in ("", '{"initialized": "12.0", "cur'):
        with open(dconf, "w") as f:
            f.write(torn)
        vi = get_version_info(data, dconf, versions, "12")
        assert vi == {"initialized": "12.0", "current": "12.0"}

    # torn file, no PG_VERSION either → default version
    os.unlink(os.path.join(data, "PG_VERSION"))
    with open(dconf, "w") as f:
        f.write("")
    vi = get_version_info(data, dconf, versions, "12")
    assert vi == {"initialized": "12.0", "current": "12.0"}

    # torn file with a PG_VERSION we have no binaries for stays fatal
    with open(os.path.join(data, "PG_VERSION"), "w") as f:
        f.write("11\n")
    with pytest.raises(ValueError):
        get_version_info(data, dconf, versions, "12")


def test_resolve_versioned_paths_writes_config_atomically(tmp_path):
    """resolve_versioned_paths must never leave manatee-config.json
    observable in a truncated state: it writes tmp + fsync +
    os.replace, so concurrent snapshot copies and kill -9 see either
    the old or the new content."""
    eng = mk_engine(tmp_path, "12")
    eng.resolve_versioned_paths()
    dconf = os.path.join(str(tmp_path / "store"), "manatee-config.json")
    with open(dconf) as f:
        assert json.load(f) == {"initialized": "12.0", "current": "12.0"}
    assert not os.path.exists(dconf + ".tmp")


def mk_engine(tmp_path, major, versions=None, extra_cfg=None):
    versions = versions or {"9.6": "9.6.3", "12": "12.0"}
    data = str(tmp_path / "store" / "data")
    os.makedirs(data, exist_ok=True)
    cfg = {"versions": versions, "defaultVersion": major,
           "pgBaseDir": str(tmp_path / "pg")}
    cfg.update(extra_cfg or {})
    return PostgresEngine(
        data, "10.0.0.1", 5432, "10.0.0.1:5432:5434", cfg=cfg)


def test_conf_generation_pg96_standby_uses_recovery_conf(tmp_path):
    eng = mk_engine(tmp_path, "9.6")
    eng.write_conf("standby",
                   upstream_url="tcp://postgres@10.0.0.9:5432/postgres")
    conf = confparser.read(eng._conf_path())
    assert conf["synchronous_commit"] == "off"
    assert conf["port"] == "5432"
    assert "primary_conninfo" not in conf           # 9.6: recovery.conf
    rec = confparser.read(eng._recovery_path())
    assert rec["standby_mode"] == "on"
    assert "host=10.0.0.9" in rec["primary_conninfo"]
    assert "application_name=10.0.0.1:5432:5434" in \
        rec["primary_conninfo"]
    assert not os.path.exists(eng._signal_path())
    assert eng.current_conf_role() == "standby"
    # promotion to primary removes recovery.conf
    eng.write_conf("primary", sync_name="peer2", read_only=True)
    assert not os.path.exists(eng._recovery_path())
    conf = confparser.read(eng._conf_path())
    assert conf["synchronous_standby_names"] == "'peer2'"
    assert conf["default_transaction_read_only"] == "on"
    assert conf["synchronous_commit"] == "remote_write"
    assert eng.current_conf_role() == "primary"


def test_conf_generation_pg12_standby_uses_signal(tmp_path):
    eng = mk_engine(tmp_path, "12")
    eng.write_conf("standby",
                   upstream_url="tcp://postgres@10.0.0.9:5432/postgres")
    conf = confparser.read(eng._conf_path())
    assert "host=10.0.0.9" in conf["primary_conninfo"]   # 12: in main conf
    assert os.path.exists(eng._signal_path())
    assert not os.path.exists(eng._recovery_path())
    assert eng.current_conf_role() == "standby"
    eng.write_conf("primary", read_only=False)
    assert not os.path.exists(eng._signal_path())
    assert eng.current_conf_role() == "primary"
    # version metadata persisted on the dataset
    with open(eng.data_conf) as f:
        assert json.load(f)["current"] == "12.0"


def test_conf_regeneration_drops_custom_keys(tmp_path):
    """Conf files are regenerated from the template — hand edits are
    lost (ref lib/postgresMgr.js:2277-2281)."""
    eng = mk_engine(tmp_path, "12")
    eng.write_conf("primary")
    conf = confparser.read(eng._conf_path())
    conf["hand_edited"] = "yes"
    confparser.write(eng._conf_path(), conf)
    eng.write_conf("primary")
    assert "hand_edited" not in confparser.read(eng._conf_path())


# ------------------------------------------------------------------ pgwire

def _msg(t: bytes, payload: bytes) -> bytes:
    return t + struct.pack(">I", len(payload) + 4) + payload


async def fake_pg_backend(reader, writer):
    """Scripted v3 backend: trust auth, answers two canned queries."""
    ln = struct.unpack(">i", await reader.readexactly(4))[0]
    await reader.readexactly(ln - 4)              # startup params
    writer.write(_msg(b"R", struct.pack(">i", 0)))            # AuthOk
    writer.write(_msg(b"S", b"server_version\x0012.0\x00"))
    writer.write(_msg(b"Z", b"I"))                            # ready
    await writer.drain()
    while True:
        try:
            t = await reader.readexactly(1)
        except asyncio.IncompleteReadError:
            return
        ln = struct.unpack(">I", await reader.readexactly(4))[0]
        body = await reader.readexactly(ln - 4)
        if t == b"X":
            return
        sql = body.rstrip(b"\x00").decode()
        if "error" in sql:
            writer.write(_msg(b"E", b"SXERROR\x00C42601\x00"
                              b"Msyntax error\x00\x00"))
            writer.write(_msg(b"Z", b"I"))
        else:
            # one row, two text columns: a | b
            rd = struct.pack(">h", 2)
            for name in (b"a", b"b"):
                rd += name + b"\x00" + struct.pack(">ihihih", 0, 0, 25,
                                                   -1, -1, 0)
            writer.write(_msg(b"T", rd))
            row = struct.pack(">h", 2)
            row += struct.pack(">i", 2) + b"42"
            row += struct.pack(">i", -1)          # NULL
            writer.write(_msg(b"D", row))
            writer.write(_msg(b"C", b"SELECT 1\x00"))
            writer.write(_msg(b"Z", b"I"))
        await writer.drain()


def test_pgwire_client_roundtrip():
    from manatee_amd.db.pgwire import PgClient, PgError

    async def go():
        server = await asyncio.start_server(fake_pg_backend,
                                            "127.0.0.1", 0)
        port = server.sockets[0].getsockname()[1]
        cli = PgClient("127.0.0.1", port, "postgres")
        await cli.connect()
        assert cli.parameters["server_version"] == "12.0"
        res = await cli.query("SELECT 42 as a, NULL as b;")
        assert res.columns == ["a", "b"]
        assert res.rows == [("42", None)]
        assert res.dicts() == [{"a": "42", "b": None}]
        assert res.command == "SELECT 1"
        with pytest.raises(PgError) as exc:
            await cli.query("this is an error;")
        assert exc.value.code == "42601"
        # connection still usable after an error
        res = await cli.query("SELECT 42 as a, NULL as b;")
        assert res.rows == [("42", None)]
        await cli.close()
        server.close()
        await server.wait_closed()
    asyncio.run(asyncio.wait_for(go(), 30))


def test_full_page_writes_default_is_safe(tmp_path):
    """Regression (advisor finding): the reference's template turns
    full_page_writes OFF assuming ZFS (copy-on-write, no torn pages);
    on a plain filesystem that risks unrecoverable torn-page corruption
    after power loss.  The safe value must be the default, relaxed only
    when the configured store is copy-on-write."""
    eng = mk_engine(tmp_path, "12")
    eng.write_conf("primary")
    conf = confparser.read(eng._conf_path())
    assert conf["full_page_writes"] == "on"      # safe default (DirStore)

    eng2 = mk_engine(tmp_path, "12", extra_cfg={"storeIsCow": True})
    eng2.write_conf("primary")
    conf = confparser.read(eng2._conf_path())
    assert conf["full_page_writes"] == "off"     # ZFS: reference value
