"""Byte-compatibility of `manatee-adm` output with the REFERENCE's own
golden file.

Reconstructs the reference test suite's mock cluster states (the
MockState generator in /root/reference/test/tst.manateeAdm.js — same
uuids, ips, replication rows and lag values), feeds them to OUR CLI via
the MANATEE_ADM_TEST_STATE seam, formats the results in the reference's
catest capture format, and diffs the whole thing byte-for-byte against
/root/reference/test/tst.manateeAdm.js.out.

This is the strongest possible parity evidence for the committed
`manatee-adm` surface (BASELINE.json: "manatee-adm output ...
byte-compatible"): the reference's own expected output, rendered by this
build's renderer, with zero tolerance.
"""

import json
import math
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
REF_GOLDEN = "/root/reference/test/tst.manateeAdm.js.out"

# the reference's hardcoded peer uuids (test/tst.manateeAdm.js:180-191)
UUIDS = [
    "301e2d2c-cd09-11e4-837d-13ea7132a060",
    "30219700-cd09-11e4-a971-cf7f957afa2d",
    "3022acc6-cd09-11e4-9716-1ff89e748aff",
    "30250e3a-cd09-11e4-a9e0-cbb241418d62",
    "30265b32-cd09-11e4-b42b-d3e4ff184bb8",
    "3028df9c-cd09-11e4-baa6-23ac5bd2e03c",
    "302b6db6-cd09-11e4-af3b-5bda767e5008",
    "302d2bec-cd09-11e4-a023-a3ebc0c6fd36",
    "302f53b8-cd09-11e4-b38f-a35b1d876bb4",
    "30314268-cd09-11e4-bf7c-4f69c228740e",
]


class MockState:
    """Reference-shaped cluster-state fixture generator
    (ref test/tst.manateeAdm.js MockState/addPeer/finish)."""

    def __init__(self):
        self.d = {
            "pgs_peers": {},
            "pgs_generation": 3,
            "pgs_initwal": "3/12345678",
            "pgs_primary": None,
            "pgs_sync": None,
            "pgs_asyncs": [],
            "pgs_deposed": [],
            "pgs_frozen": False,
            "pgs_freeze_reason": None,
            "pgs_freeze_time": None,
            "pgs_singleton": False,
            "pgs_errors": [],
            "pgs_warnings": [],
        }
        self._n = 0
        self._uuids = list(UUIDS)

    def add_peer(self, role, peerstatus, lag=None):
        if peerstatus == "down":
            pgerr = {"message": "failed to connect to peer: ECONNREFUSED"}
            repl = None
        elif peerstatus == "no-repl":
            pgerr = None
            repl = None
        else:   # "sync" | "async"
            pgerr = None
            repl = {
                "state": "streaming",
                "sync_state": peerstatus,
                "client_addr": None,
                "sent_location": "0/12345678",
                "flush_location": "0/12345678",
                "write_location": "0/12345678",
                "replay_location": "0/12345678",
            }
        if role == "async":
            if lag is None:
                lag = 342
            lagobj = {"minutes": math.floor(lag / 60), "seconds": lag % 60}
        else:
            lagobj = None

        self._n += 1
        ip = "10.0.0.%d" % self._n
        zone_id = self._uuids.pop(0)
        pid = "%s:5432:12345-0000000001" % ip
        peer = {
            "pgp_label": zone_id[:8],
            "pgp_ident": {
                "id": pid,
                "ip": ip,
                "zoneId": zone_id,
                "pgUrl": "tcp://postgres@%s:5432/postgres" % ip,
                "backupUrl": "http://%s:12345" % ip,
            },
            "pgp_pgerr": pgerr,
            "pgp_lag": lagobj,
            "pgp_repl": repl,
        }
        self.d["pgs_peers"][pid] = peer
        if role == "primary":
            self.d["pgs_primary"] = pid
        elif role == "sync":
            self.d["pgs_sync"] = pid
        elif role == "async":
            self.d["pgs_asyncs"].append(pid)
        elif role == "deposed":
            self.d["pgs_deposed"].append(pid)

    def finish(self):
        chain = [self.d["pgs_peers"][self.d["pgs_primary"]]]
        if self.d["pgs_sync"] is not None:
            chain.append(self.d["pgs_peers"][self.d["pgs_sync"]])
            for a in self.d["pgs_asyncs"]:
                chain.append(self.d["pgs_peers"][a])
        for i in range(len(chain) - 1):
            repl = chain[i]["pgp_repl"]
            if repl is None or "client_addr" not in repl:
                continue
            repl["client_addr"] = chain[i + 1]["pgp_ident"]["ip"]

    def to_json(self):
        return json.dumps(self.d)


def make_singleton(primary=None):
    s = MockState()
    s.d["pgs_singleton"] = True
    s.add_peer("primary", primary or "no-repl", "none")
    s.d["pgs_frozen"] = True
    s.d["pgs_freeze_reason"] = "manatee setup: one node write mode"
    s.d["pgs_freeze_time"] = "2006-02-15T00:00:00.000Z"
    s.finish()
    return s


def make_normal(asyncs=1, deposed=0, primary=None, sync=None, lag=None,
                has_lag=False):
    s = MockState()
    s.add_peer("primary", primary or "sync")
    s.add_peer("sync", sync or ("no-repl" if asyncs == 0 else "async"))
    for i in range(asyncs):
        s.add_peer("async", "no-repl" if i == asyncs - 1 else "async",
                   lag if has_lag else None)
    for _ in range(deposed):
        s.add_peer("deposed", "down")
    s.finish()
    return s


# TEST CASE NAME, expected verify exit status, cluster state
CLUSTER_STATES = [
    ("singletonOk", 0, make_singleton()),
    ("singletonDown", 1, make_singleton(primary="down")),
    ("normalOk", 0, make_normal()),
    ("normal2Peers", 1, make_normal(asyncs=0)),
    ("normal5Peers", 0, make_normal(asyncs=3)),
    ("normalDeposed", 1, make_normal(deposed=1)),
    ("normal2Deposed", 1, make_normal(deposed=2)),
    ("normal5Peers2deposed", 1, make_normal(asyncs=3, deposed=2)),
    ("normalPdown", 1, make_normal(primary="down")),
    ("normalPnorepl", 1, make_normal(primary="no-repl")),
    ("normalPasync", 1, make_normal(primary="async")),
    ("normalSdown", 1, make_normal(sync="down")),
    ("normalSnorepl", 1, make_normal(sync="no-repl")),
    ("normalNoLag", 0, make_normal(lag=0, has_lag=True)),
    ("normalLargeLag", 0, make_normal(lag=86465, has_lag=True)),
]

NORMAL_OK_EXTRAS = [
    (0, ["peers", "-H"]),
    (0, ["peers", "-H", "-o", "role"]),
    (0, ["peers", "--omitHeader", "-o", "role", "-o", "peername"]),
    (0, ["peers", "--columns", "role", "-o", "peername,ip"]),
    (0, ["peers", "--role=primary"]),
    (0, ["peers", "-H", "-o", "ip", "-r", "sync"]),
    (0, ["pg-status", "-H"]),
    (0, ["pg-status", "-H", "-o", "role"]),
    (0, ["pg-status", "--omitHeader", "-o", "role", "-o", "peername"]),
    (0, ["pg-status", "--columns", "role", "-o", "peername,ip"]),
    (0, ["pg-status", "--role=primary"]),
    (0, ["pg-status", "-H", "-o", "ip", "-r", "sync"]),
    (2, ["peers", "-o", "pg-sent"]),
    (2, ["peers", "-o", "badcolumn"]),
    (2, ["pg-status", "-o", "badcolumn"]),
    (0, ["pg-status", "1", "2"]),
    (0, ["pg-status", "-w"]),
]


def run_cli(args, fixture_file):
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["MANATEE_ADM_TEST_STATE"] = fixture_file
    env["SHARD"] = "UNUSED"
    env["ZK_IPS"] = "UNUSED"
    return subprocess.run(
        [sys.executable, os.path.join(REPO, "bin", "manatee-adm")] + args,
        capture_output=True, text=True, timeout=60, env=env, cwd=REPO)


def generate_output(tmp_path):
    fixture_file = str(tmp_path / "state.json")
    chunks = []
    for name, verify_status, cs in CLUSTER_STATES:
        with open(fixture_file, "w") as f:
            f.write(cs.to_json())
        matrix = [
            (0, ["peers"]),
            (0, ["pg-status"]),
            (0, ["show"]),
            (0, ["show", "-v"]),
            (verify_status, ["verify"]),
            (verify_status, ["verify", "-v"]),
        ]
        if name == "normalOk":
            matrix += NORMAL_OK_EXTRAS
        for expected, args in matrix:
            r = run_cli(args, fixture_file)
            assert r.returncode == expected, \
                "%s: manatee-adm %s: exit %d != %d\nstdout:\n%s\n" \
                "stderr:\n%s" % (name, " ".join(args), r.returncode,
                                 expected, r.stdout, r.stderr)
            chunks.append('TEST CASE "%s": manatee-adm %s:\n'
                          % (name, " ".join(args)))
            chunks.append("--------- stdout ------------\n")
            chunks.append(r.stdout)
            chunks.append("--------- stderr ------------\n")
            chunks.append(r.stderr)
            chunks.append("-----------------------------\n\n")
    chunks.append("TEST PASSED\n")
    return "".join(chunks)


@pytest.mark.skipif(not os.path.exists(REF_GOLDEN),
                    reason="reference golden not available")
def test_reference_golden_byte_identical(tmp_path):
    got = generate_output(tmp_path)
    with open(REF_GOLDEN) as f:
        want = f.read()
    if got != want:
        # pinpoint the first differing line for a readable failure
        got_lines = got.splitlines(keepends=True)
        want_lines = want.splitlines(keepends=True)
        for i, (g, w) in enumerate(zip(got_lines, want_lines)):
            if g != w:
                raise AssertionError(
                    "first diff at line %d:\n  ours: %r\n  ref:  %r"
                    % (i + 1, g, w))
        raise AssertionError("length mismatch: ours %d lines, ref %d"
                             % (len(got_lines), len(want_lines)))
