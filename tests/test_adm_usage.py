"""Usage-text consistency tier (ref test/tst.manateeAdmUsage.js, 635
LoC): every manatee-adm subcommand's help must work, name the command,
and agree with the committed manual (docs/manatee-adm.md); the four
byte-compat commands' -h output must be the reference's cmdln text
verbatim."""

import os
import re
import subprocess
import sys

import pytest

from manatee_amd.adm.cli import COMMAND_HELP, _mk_parser

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
MANUAL = os.path.join(REPO, "docs", "manatee-adm.md")

ALL_COMMANDS = ["version", "status", "peers", "pg-status", "show",
                "verify", "zk-state", "zk-active", "history", "freeze",
                "unfreeze", "reap", "set-onwm", "state-backfill",
                "check-lock", "promote", "clear-promote", "rebuild"]

ALIASES = {"zk-state": "state", "zk-active": "active",
           "pg-status": "db-status"}


def run_cli(args):
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env.pop("MANATEE_ADM_TEST_STATE", None)
    return subprocess.run(
        [sys.executable, os.path.join(REPO, "bin", "manatee-adm")] + args,
        capture_output=True, text=True, timeout=30, env=env, cwd=REPO)


def test_parser_covers_exactly_the_documented_commands():
    """The CLI's subcommand set and the reference's 18-command surface
    (ref bin/manatee-adm:94-122) must match exactly — no drift in
    either direction."""
    parser = _mk_parser()
    sub = next(a for a in parser._actions
               if hasattr(a, "choices") and a.choices)
    names = set(sub.choices)
    for cmd in ALL_COMMANDS:
        assert cmd in names, "missing subcommand %r" % cmd
    for cmd, alias in ALIASES.items():
        assert alias in names, "missing alias %r for %r" % (alias, cmd)
    extras = names - set(ALL_COMMANDS) - set(ALIASES.values())
    assert not extras, "undocumented subcommands: %r" % extras


def test_every_command_documented_in_manual():
    """ref tst.manateeAdmUsage.js checks help/man consistency — every
    subcommand must have a section in docs/manatee-adm.md."""
    with open(MANUAL) as f:
        text = f.read()
    for cmd in ALL_COMMANDS:
        assert re.search(r"^### %s\b" % re.escape(cmd), text, re.M), \
            "docs/manatee-adm.md lacks a section for %r" % cmd


@pytest.mark.parametrize("cmd", sorted(COMMAND_HELP))
def test_committed_commands_help_is_reference_text(cmd):
    """-h on the committed commands prints the reference's cmdln help
    byte-for-byte (the same text its golden file captures on usage
    errors) and exits 0."""
    r = run_cli([cmd, "-h"])
    assert r.returncode == 0, r.stderr
    assert r.stdout == COMMAND_HELP[cmd]
    # structural sanity of the canned text itself
    assert "Usage:\n    manatee-adm %s" % cmd in r.stdout
    assert "-h, --help" in r.stdout


@pytest.mark.parametrize("cmd", [c for c in ALL_COMMANDS
                                 if c not in COMMAND_HELP])
def test_other_commands_have_working_help(cmd):
    r = run_cli([cmd, "-h"])
    assert r.returncode == 0, r.stderr
    out = r.stdout + r.stderr
    assert cmd in out
    assert "usage" in out.lower()


def test_usage_error_reprints_help():
    """A usage error on a committed command exits 2 and echoes the help
    after the one-line error (the cmdln usage() shape the reference's
    golden tests capture)."""
    r = run_cli(["peers", "-o", "nope"])
    assert r.returncode == 2
    assert r.stderr.startswith('manatee-adm: unsupported column: "nope"\n')
    assert r.stderr.endswith(COMMAND_HELP["peers"])
