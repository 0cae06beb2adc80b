"""Tier-1 unit tests for the foundations.

Mirrors the reference's pure unit tier:
test/tst.common.js (pgStripMinor table), test/confParser.test.js
(conf round-trips), plus LSN arithmetic and the schema validator.
"""

import io
import json
import os
import signal

import pytest

from manatee_amd.common import confparser, lsn, procutil, schema
from manatee_amd.common import logging as mlog


# ---------------------------------------------------------------- pgStripMinor
@pytest.mark.parametrize("version,expected", [
    ("9.2.4", "9.2"),
    ("9.6.3", "9.6"),
    ("9.6", "9.6"),
    ("12.0", "12"),
    ("12", "12"),
    ("10.1", "10"),
    ("13.4.1", "13"),
    ("", None),
    ("abc", None),
    ("9", None),
    ("9.x", None),
])
def test_pg_strip_minor(version, expected):
    assert lsn.pg_strip_minor(version) == expected


# ------------------------------------------------------------------------ LSN
def test_lsn_roundtrip():
    for text in ["0/00000000", "0/174A4D0", "16/B374D848", "FFFFFFFF/FFFFFFFF"]:
        assert lsn.format_lsn(lsn.parse(text)) == text.upper().replace(
            "0/174A4D0", "0/0174A4D0")


def test_lsn_compare():
    assert lsn.compare("0/174A4D0", "0/174A4D0") == 0
    assert lsn.compare("0/174A4D0", "0/174A4D1") == -1
    assert lsn.compare("1/00000000", "0/FFFFFFFF") == 1
    assert lsn.diff_bytes("1/00000000", "0/FFFFFFFF") == 1
    assert lsn.max_lsn("0/10", "0/20") == "0/20"
    assert lsn.is_lsn("0/174A4D0")
    assert not lsn.is_lsn("0-174A4D0")
    with pytest.raises(ValueError):
        lsn.parse("bogus")


# ----------------------------------------------------------------- confparser
def test_conf_roundtrip(tmp_path):
    p = str(tmp_path / "postgresql.conf")
    with open(p, "w") as f:
        f.write("# comment\n"
                "listen_addresses = '0.0.0.0'\n"
                "wal_level = hot_standby  # trailing comment\n"
                "synchronous_commit = remote_write\n"
                "port 5432\n")
    conf = confparser.read(p)
    assert conf["listen_addresses"] == "'0.0.0.0'"
    assert conf["wal_level"] == "hot_standby"
    assert conf["port"] == "5432"
    confparser.set_value(conf, "synchronous_standby_names", "'\"peer1\"'")
    confparser.set_value(conf, "default_transaction_read_only", "on")
    confparser.delete(conf, "port")
    confparser.write(p, conf)
    again = confparser.read(p)
    assert again["synchronous_standby_names"] == "'\"peer1\"'"
    assert again["default_transaction_read_only"] == "on"
    assert "port" not in again
    # regeneration drops nothing else
    assert again["wal_level"] == "hot_standby"


# --------------------------------------------------------------------- schema
def test_schema_validator():
    sch = {
        "type": "object",
        "properties": {
            "ip": {"type": "string", "required": True},
            "postgresPort": {"type": "integer", "required": True},
            "opts": {
                "type": "object",
                "properties": {"sessionTimeout": {"type": "number"}},
            },
            "versions": {"type": "array", "items": {"type": "string"}},
        },
    }
    schema.validate({"ip": "127.0.0.1", "postgresPort": 5432}, sch)
    schema.validate({"ip": "x", "postgresPort": 1,
                     "opts": {"sessionTimeout": 60000.0},
                     "versions": ["9.6", "12"]}, sch)
    with pytest.raises(schema.ValidationError):
        schema.validate({"ip": "127.0.0.1"}, sch)
    with pytest.raises(schema.ValidationError):
        schema.validate({"ip": 5, "postgresPort": 5432}, sch)
    with pytest.raises(schema.ValidationError):
        schema.validate({"ip": "x", "postgresPort": 1, "versions": [3]}, sch)
    assert schema.check({"ip": "x"}, sch) != []
    assert schema.check({"ip": "x", "postgresPort": 1}, sch) == []


# -------------------------------------------------------------------- logging
def test_bunyan_log_format():
    buf = io.StringIO()
    log = mlog.Logger("test-sitter", level="debug", stream=buf)
    child = log.child(component="zk", peer="10.0.0.1:5432:12345")
    child.info("state written", generation=3)
    child.trace("not emitted")
    lines = buf.getvalue().strip().splitlines()
    assert len(lines) == 1
    rec = json.loads(lines[0])
    assert rec["v"] == 0
    assert rec["name"] == "test-sitter"
    assert rec["level"] == 30
    assert rec["component"] == "zk"
    assert rec["generation"] == 3
    assert rec["msg"] == "state written"
    assert rec["time"].endswith("Z")


def test_verbosity_levels():
    assert mlog.level_from_verbosity(0) == mlog.INFO
    assert mlog.level_from_verbosity(1) == mlog.DEBUG
    assert mlog.level_from_verbosity(2) == mlog.TRACE
    assert mlog.level_from_verbosity(5) == mlog.TRACE


# ------------------------------------------------------------------- procutil
def test_run_basic():
    res = procutil.run(["/bin/echo", "hello"])
    assert res.returncode == 0
    assert res.stdout.strip() == "hello"


def test_run_failure():
    with pytest.raises(procutil.ExecError) as ei:
        procutil.run(["/bin/false"])
    assert ei.value.returncode == 1


def test_run_timeout():
    with pytest.raises(procutil.ExecError):
        procutil.run(["/bin/sleep", "5"], timeout=0.2)


def test_replace_file(tmp_path):
    p = str(tmp_path / "f")
    procutil.replace_file(p, "one")
    procutil.replace_file(p, "two", mode=0o600)
    assert open(p).read() == "two"
    assert oct(os.stat(p).st_mode & 0o777) == "0o600"


def test_kill_escalate():
    import asyncio
    import subprocess

    # a process that ignores SIGINT/SIGQUIT: escalation must reach SIGKILL
    proc = subprocess.Popen(
        ["/usr/bin/python3", "-c",
         "import signal,time,sys\n"
         "signal.signal(signal.SIGINT, signal.SIG_IGN)\n"
         "signal.signal(signal.SIGQUIT, signal.SIG_IGN)\n"
         "print('ready', flush=True)\n"
         "time.sleep(60)"], stdout=subprocess.PIPE)
    assert proc.stdout.readline().strip() == b"ready"
    try:
        sig = asyncio.run(procutil.kill_escalate(proc.pid, ops_timeout_s=0.5))
        assert sig == signal.SIGKILL
    finally:
        proc.wait(timeout=10)


def test_kill_escalate_wakes_stopped_target():
    """Escalating against a SIGSTOPped child must not burn a full
    per-step timeout waiting for queued SIGINT/SIGQUIT it can never
    run: the escalation CONTs a stopped target so it can die.  Without
    the CONT this took 2 x ops_timeout before SIGKILL — the rare ~60 s
    failover outliers when chaos froze a db under a live sitter."""
    import asyncio
    import subprocess
    import time as _time

    proc = subprocess.Popen(
        ["/usr/bin/python3", "-c",
         "import time; print('ready', flush=True); time.sleep(120)"],
        stdout=subprocess.PIPE)
    assert proc.stdout.readline().strip() == b"ready"
    os.kill(proc.pid, signal.SIGSTOP)
    # wait for the stop to be VISIBLE (state T) so the escalation's
    # stopped-target handling — not a send/stop race — is what's tested
    deadline = _time.monotonic() + 10
    while _time.monotonic() < deadline:
        with open("/proc/%d/stat" % proc.pid) as f:
            if f.read().rpartition(")")[2].split()[0] == "T":
                break
        _time.sleep(0.01)
    try:
        t0 = _time.monotonic()
        sig = asyncio.run(procutil.kill_escalate(proc.pid,
                                                 ops_timeout_s=20.0))
        dt = _time.monotonic() - t0
        # SIGINT + CONT kills a default-disposition python promptly
        assert sig == signal.SIGINT
        assert dt < 5, "stopped child burned %.1fs of escalation" % dt
    finally:
        proc.wait(timeout=10)


def test_run_async():
    import asyncio

    async def go():
        res = await procutil.run_async(["/bin/echo", "async"])
        return res

    res = asyncio.run(go())
    assert res.stdout.strip() == "async"
