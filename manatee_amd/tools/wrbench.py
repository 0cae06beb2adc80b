"""Write/read throughput micro-benchmark for the replicated engine.

    python -m manatee_amd.tools.wrbench [--seconds 10] [--writers 1,8,32]
        [--engine waldb|postgres]

Spins a local 3-peer shard (primary -> sync -> async) and measures,
for each writer count, the rate of *synchronously acknowledged* puts
(every put waits for the sync standby's write-ack, the shard's
durability contract) plus standby read throughput.  One JSON line on
stdout."""

from __future__ import annotations

import argparse
import asyncio
import json
import shutil
import sys
import tempfile
import time

from .devcluster import DevCluster


async def writer_task(peer, label, stop, counter, pipeline=0):
    cli = peer.db_client()
    i = 0
    try:
        while not stop.is_set():
            if pipeline > 1:
                n = await cli.put_many(
                    (("w-%s-%d" % (label, i + j), i + j)
                     for j in range(pipeline)))
                counter[0] += n
                i += pipeline
            else:
                await cli.put("w-%s-%d" % (label, i), i)
                counter[0] += 1
                i += 1
    finally:
        await cli.close()


async def reader_task(peer, stop, counter):
    cli = peer.db_client()
    i = 0
    try:
        while not stop.is_set():
            await cli.get("w-0-%d" % (i % 1000))
            counter[0] += 1
            i += 1
    finally:
        await cli.close()


async def measure(c: DevCluster, n_writers: int, seconds: float,
                  pipeline: int = 0) -> dict:
    s = await c.cluster_state()
    prim = c.peer_by_id(s["primary"]["id"])
    sync = c.peer_by_id(s["sync"]["id"])
    stop = asyncio.Event()
    wcount = [0]
    rcount = [0]
    tasks = [asyncio.ensure_future(
        writer_task(prim, "%d.%d" % (pipeline, w),
                    stop, wcount, pipeline=pipeline))
        for w in range(n_writers)]
    tasks += [asyncio.ensure_future(
        reader_task(sync, stop, rcount))
        for _ in range(4)]
    t0 = time.monotonic()
    await asyncio.sleep(seconds)
    stop.set()
    await asyncio.gather(*tasks, return_exceptions=True)
    dt = time.monotonic() - t0
    return {"writers": n_writers, "pipeline": pipeline,
            "acked_puts_per_s": round(wcount[0] / dt, 1),
            "standby_reads_per_s": round(rcount[0] / dt, 1)}


async def run(writers, seconds, workdir, engine="waldb") -> dict:
    c = DevCluster(workdir, n_peers=3, shard_name="1.wrbench",
                   engine=engine)
    try:
        await c.start()
        await c.wait_cluster(
            lambda s: s.get("sync") and len(s.get("async", [])) == 1,
            timeout_s=120, what="formation")
        await c.wait_writable(timeout_s=120)
        results = []
        for w in writers:
            r = await measure(c, w, seconds)
            results.append(r)
            print("# writers=%d: %.0f acked puts/s, %.0f standby reads/s"
                  % (w, r["acked_puts_per_s"], r["standby_reads_per_s"]),
                  file=sys.stderr)
        # pipelined bulk mode: batches of 50 per round trip
        r = await measure(c, 4, seconds, pipeline=50)
        results.append(r)
        print("# writers=4 pipeline=50: %.0f acked puts/s"
              % r["acked_puts_per_s"], file=sys.stderr)
        # durability sanity: everything acked must be present
        s = await c.cluster_state()
        cli = c.peer_by_id(s["primary"]["id"]).db_client()
        total = await cli.count(prefix="w-")
        await cli.close()
        return {"engine": engine, "results": results, "total_keys": total}
    finally:
        c.stop()


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(prog="manatee-wrbench")
    ap.add_argument("--seconds", type=float, default=10.0)
    ap.add_argument("--engine", choices=("waldb", "postgres"),
                    default="waldb")
    ap.add_argument("--writers", default="1,8,32")
    ap.add_argument("-d", "--dir", default=None)
    ns = ap.parse_args(argv)
    writers = [int(w) for w in ns.writers.split(",")]
    workdir = ns.dir or tempfile.mkdtemp(prefix="manatee-wrbench-")
    try:
        out = asyncio.run(run(writers, ns.seconds, workdir,
                              engine=ns.engine))
    finally:
        if ns.dir is None:
            shutil.rmtree(workdir, ignore_errors=True)
    print(json.dumps(out))
    return 0


if __name__ == "__main__":
    sys.exit(main())
