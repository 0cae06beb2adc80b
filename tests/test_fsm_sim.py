"""Randomized FSM fuzzing: seeded fault schedules with invariant
checking (manatee_amd/fsm/sim.py).  The CPU tier runs a few seeds; the
dedicated-box tier sweeps more."""

import asyncio

import pytest

from manatee_amd.fsm.sim import Simulator


def run_sim(seed, steps):
    sim = Simulator(seed=seed, n_peers=3)
    return asyncio.run(asyncio.wait_for(sim.run(steps=steps), 300))


@pytest.mark.parametrize("seed", [0, 3, 7])
def test_random_fault_schedule_preserves_invariants(seed):
    res = run_sim(seed, steps=10)
    assert res["history_entries"] >= 1
    assert res["generations"] >= 1


@pytest.mark.gpu
@pytest.mark.parametrize("seed", list(range(10, 22)))
def test_random_fault_schedule_sweep(seed):
    res = run_sim(seed, steps=25)
    assert res["history_entries"] >= 1
