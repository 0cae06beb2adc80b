"""Distributed bench contract: the driver launches bench.py under
``torch.distributed.run`` with one rank per GPU (gloo aggregation on
CPU here) — cover that path with world_size 2 so the round-end scaling
run is correct by construction."""

import json
import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_bench_world2_aggregates_worst_rank():
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run",
         "--nnodes=1", "--nproc-per-node", "2",
         "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()),
         os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "1", "--warmup", "0"],
        capture_output=True, text=True, timeout=600, env=env, cwd=REPO)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, "exactly one JSON line from rank 0"
    out = json.loads(lines[0])
    assert out["n_gpus"] == 2
    assert out["config"]["shards"] == 2
    assert out["config"]["acked_writes_lost"] == 0
    assert out["value"] > 0
