"""Test harness: thin aliases over the in-package FSM simulator
(manatee_amd/fsm/sim.py) — an in-process shard of ManateePeer FSMs
against the embedded ZK server with scriptable mock databases, the
analogue of the reference's ``manatee-state-machine`` simulator
(SURVEY.md §2.2)."""

from manatee_amd.fsm.sim import (  # noqa: F401
    SHARD_PATH, MockDb, SimPeer as TestPeer, SimShard as Shard)


def pid(ip):
    return "%s:5432:12345" % ip
