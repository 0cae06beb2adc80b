"""DevCluster — build and drive a multi-peer shard on one host.

The equivalent of ``tools/mkdevsitters`` + ``test/testManatee.js``: creates
per-peer directories and configs (ports stepped +10 per peer like the
reference, ref docs/working-on-manatee.md:179-197), spawns the embedded ZK
server, and one sitter + backupserver (+ optional snapshotter) pair per
peer as real subprocesses.  Used by the integration tests, ``bench.py``
and ``tools/mkdevsitters``.
"""

from __future__ import annotations

import asyncio
import json
import os
import signal
import socket
import subprocess
import sys
import time
from typing import Dict, List, Optional

from ..common.httpd import http_request
from ..db.waldb.client import WaldbClient

REPO_ROOT = os.path.dirname(os.path.dirname(
    os.path.dirname(os.path.abspath(__file__))))


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _ephemeral_low() -> int:
    """Lower bound of the kernel's ephemeral port range — peer ports must
    stay BELOW it or random client sockets will collide with them."""
    try:
        with open("/proc/sys/net/ipv4/ip_local_port_range") as f:
            return int(f.read().split()[0])
    except (OSError, ValueError, IndexError):
        return 32768


_EPHEMERAL_LOW = _ephemeral_low()


class DevPeer:
    def __init__(self, cluster: "DevCluster", index: int, base_port: int):
        self.cluster = cluster
        self.index = index
        self.ip = cluster.ip
        self.pg_port = base_port
        self.status_port = base_port + 1
        self.backup_port = base_port + 2
        self.id = "%s:%d:%d" % (self.ip, self.pg_port, self.backup_port)
        self.dir = os.path.join(cluster.base_dir, "peer%d" % index)
        self.store_dir = os.path.join(self.dir, "store")
        self.sitter_proc: Optional[subprocess.Popen] = None
        self.backup_proc: Optional[subprocess.Popen] = None
        self.snap_proc: Optional[subprocess.Popen] = None

    # ------------------------------------------------------------- configs
    def storage_cfg(self) -> dict:
        c = self.cluster
        if c.storage_provider == "zfs":
            # one fakezfs pool per peer (each peer = its own host in the
            # reference's deployment model); the shim bakes in the state
            # root since zfs invocations run env-scrubbed
            from .fakezfs import install_fakezfs
            zfs_path = install_fakezfs(os.path.join(self.dir, "zfsbin"),
                                       os.path.join(self.dir, "zfspool"))
            for parent in ("tank", "tank/manatee"):
                subprocess.run([zfs_path, "create", "-o", "canmount=off",
                                parent], capture_output=True)
            return {"provider": "zfs",
                    "dataset": "tank/manatee/data",
                    "mountpoint": os.path.join(self.store_dir, "live"),
                    "zfsPath": zfs_path}
        return {"provider": "dir", "mountpoint": self.store_dir}

    def sitter_config(self) -> dict:
        c = self.cluster
        out = {
            "ip": self.ip,
            "postgresPort": self.pg_port,
            "backupPort": self.backup_port,
            "shardPath": c.shard_path,
            "zoneId": "peer%d" % self.index,
            "zkCfg": {
                "connStr": c.zk_conn_str,
                "opts": {"sessionTimeout": c.session_timeout_ms},
            },
            "postgresMgrCfg": {
                "engine": c.engine,
                "storageCfg": self.storage_cfg(),
                "healthChkInterval": c.health_interval_ms,
                "healthChkTimeout": c.health_timeout_ms,
                "opsTimeout": c.ops_timeout_ms,
                "replicationTimeout": c.replication_timeout_ms,
                "tickInterval": c.tick_interval_ms,
                "oneNodeWriteMode": c.singleton,
            },
        }
        if c.engine == "postgres":
            # versioned binary dirs under pgBaseDir (minipg
            # initdb/postgres shims, ref resolveVersionedPaths
            # lib/postgresMgr.js:569-634)
            cfg = out["postgresMgrCfg"]
            cfg["versions"] = {"12": "12.0", "9.6": "9.6.3"}
            cfg["defaultVersion"] = c.pg_version
            cfg["pgBaseDir"] = c.pg_base_dir
            cfg["dbUser"] = "postgres"
        return out

    def backupserver_config(self) -> dict:
        return {
            "ip": self.ip,
            "backupServerCfg": {"port": self.backup_port},
            "backupSenderCfg": {
                "storageCfg": self.storage_cfg(),
            },
        }

    def snapshotter_config(self) -> dict:
        c = self.cluster
        return {
            "storageCfg": self.storage_cfg(),
            "pollInterval": c.snapshot_interval_ms,
            "snapshotNumber": c.snapshot_number,
            "statusUrl": "http://%s:%d/ping" % (self.ip,
                                                self.status_port),
        }

    def write_configs(self) -> None:
        os.makedirs(self.dir, exist_ok=True)
        with open(os.path.join(self.dir, "sitter.json"), "w") as f:
            json.dump(self.sitter_config(), f, indent=2)
        with open(os.path.join(self.dir, "backupserver.json"), "w") as f:
            json.dump(self.backupserver_config(), f, indent=2)
        with open(os.path.join(self.dir, "snapshotter.json"), "w") as f:
            json.dump(self.snapshotter_config(), f, indent=2)

    # ------------------------------------------------------------- control
    def _spawn(self, module: str, config: str, logname: str
               ) -> subprocess.Popen:
        env = dict(os.environ)
        env["PYTHONPATH"] = REPO_ROOT + os.pathsep + \
            env.get("PYTHONPATH", "")
        if self.cluster.proxied:
            # every outbound connection this peer (and its db child)
            # makes is rewritten through per-directed-link proxies so
            # tests can induce asymmetric network partitions
            env["MANATEE_DIAL_MAP"] = os.path.join(self.dir,
                                                   "dialmap.json")
        logf = open(os.path.join(self.dir, logname), "a")

        def _reset_signals():
            # a backgrounded test runner (nohup … &) has SIGINT/SIGQUIT
            # set to SIG_IGN, which survives exec — the daemons (and
            # anything they spawn) must get default dispositions or
            # dirty-stop escalation cannot touch them during boot
            for s in (signal.SIGINT, signal.SIGQUIT, signal.SIGTERM):
                signal.signal(s, signal.SIG_DFL)

        return subprocess.Popen(
            [sys.executable, "-m", module, "-f",
             os.path.join(self.dir, config), "-v",
             "--log-file", os.path.join(self.dir, logname + ".json")],
            env=env, stdout=logf, stderr=logf, start_new_session=True,
            preexec_fn=_reset_signals)

    def start(self) -> None:
        self.write_configs()
        self.backup_proc = self._spawn("manatee_amd.daemons.backupserver",
                                       "backupserver.json", "backupserver.log")
        self.sitter_proc = self._spawn("manatee_amd.daemons.sitter",
                                       "sitter.json", "sitter.log")
        if self.cluster.run_snapshotter:
            self.snap_proc = self._spawn("manatee_amd.daemons.snapshotter",
                                         "snapshotter.json",
                                         "snapshotter.log")

    def db_pids(self) -> List[int]:
        """Candidate db pids: the manager-written db_child.pid (written at
        spawn time, so it can never lag the child) plus the db's own pid
        file (covers children of sitters from earlier incarnations).
        Every pid is verified against /proc/<pid>/cmdline (must be a db
        server running on THIS peer's data dir) — pid files go stale and
        pids get recycled, and SIGKILLing a recycled pid would murder an
        innocent process."""
        data = os.path.join(self.store_dir, "live", "data")
        pids = []
        candidates = [os.path.join(data, "db_child.pid"),
                      os.path.join(data, "waldb.pid"),
                      os.path.join(data, "postmaster.pid"),
                      # survives dataset replacement by restores
                      os.path.join(self.store_dir, "db_child.pid")]
        for path in candidates:
            try:
                with open(path) as f:
                    pid = int(f.read().split()[0])
            except (OSError, ValueError, IndexError):
                continue
            try:
                with open("/proc/%d/cmdline" % pid, "rb") as f:
                    cmdline = f.read().replace(b"\x00", b" ").decode(
                        "utf-8", "replace")
            except OSError:
                continue        # no such process
            if ("waldb" in cmdline or "postgres" in cmdline) \
                    and data in cmdline and pid not in pids:
                pids.append(pid)
        return pids

    def db_pid(self) -> Optional[int]:
        pids = self.db_pids()
        return pids[0] if pids else None

    def kill9(self) -> None:
        """SIGKILL the whole peer: sitter process group + db child +
        backupserver — the integ-test failure mode
        (ref test/integ.test.js primaryDeath et al)."""
        for proc in (self.sitter_proc, self.backup_proc, self.snap_proc):
            if proc is not None and proc.poll() is None:
                try:
                    os.killpg(proc.pid, signal.SIGKILL)
                except ProcessLookupError:
                    pass
                proc.wait()
        # read the db pids only AFTER the sitter is dead: the sitter can
        # respawn the db at any moment, so a pid captured earlier can go
        # stale and the fresh child would survive holding the port
        for db_pid in self.db_pids():
            try:
                os.killpg(db_pid, signal.SIGKILL)
            except (ProcessLookupError, PermissionError):
                try:
                    os.kill(db_pid, signal.SIGKILL)
                except ProcessLookupError:
                    pass
        self.sitter_proc = None
        self.backup_proc = None
        self.snap_proc = None

    @staticmethod
    def _signal_pids(pids, sig) -> None:
        for pid in pids:
            try:
                os.killpg(pid, sig)
            except (ProcessLookupError, PermissionError):
                try:
                    os.kill(pid, sig)
                except ProcessLookupError:
                    pass

    @staticmethod
    def _proc_state(pid: int) -> str:
        try:
            with open("/proc/%d/stat" % pid) as f:
                return f.read().rpartition(")")[2].split()[0]
        except (OSError, IndexError):
            return "?"

    def pause(self) -> None:
        """SIGSTOP the whole peer (sitter pg + db pg) — the network-
        partition analogue on one host: processes stay alive but stop
        responding, the ZK session expires, and on resume() the peer
        discovers the cluster moved on without it.

        The sitter is frozen FIRST and its stop confirmed before the
        db pids are read: a live sitter can respawn the db at any
        moment, and a half-frozen peer (db stopped, sitter running)
        produces misleading chaos results.  The stopped pids are
        recorded so resume() wakes exactly what was frozen even if pid
        files changed meanwhile."""
        daemons = [p.pid for p in (self.sitter_proc, self.backup_proc,
                                   self.snap_proc)
                   if p is not None and p.poll() is None]
        self._signal_pids(daemons, signal.SIGSTOP)
        deadline = time.monotonic() + 2.0
        while time.monotonic() < deadline:
            if all(self._proc_state(pid) in ("T", "Z", "?")
                   for pid in daemons):
                break
            time.sleep(0.01)
        dbs = self.db_pids()
        self._signal_pids(dbs, signal.SIGSTOP)
        self._paused_pids = daemons + dbs

    def resume(self) -> None:
        pids = list(getattr(self, "_paused_pids", []))
        self._paused_pids = []
        for p in (self.sitter_proc, self.backup_proc, self.snap_proc):
            if p is not None and p.poll() is None and p.pid not in pids:
                pids.append(p.pid)
        pids += [pid for pid in self.db_pids() if pid not in pids]
        self._signal_pids(pids, signal.SIGCONT)

    def kill_db_only(self) -> None:
        """SIGKILL only the database child (the sitter must notice and
        restart it)."""
        db_pid = self.db_pid()
        if db_pid is not None:
            try:
                os.kill(db_pid, signal.SIGKILL)
            except ProcessLookupError:
                pass

    def stop(self) -> None:
        self.kill9()

    def alive(self) -> bool:
        return self.sitter_proc is not None and \
            self.sitter_proc.poll() is None

    # -------------------------------------------------------------- clients
    def db_client(self):
        if self.cluster.engine == "postgres":
            from ..db.pgkv import PgKvClient
            return PgKvClient(self.ip, self.pg_port)
        return WaldbClient(self.ip, self.pg_port)

    async def http_status(self, path: str = "/state"):
        status, body = await http_request(
            "http://%s:%d%s" % (self.ip, self.status_port, path),
            timeout_s=5)
        return status, body


class DevCluster:
    def __init__(self, base_dir: str, n_peers: int = 3,
                 ip: str = "127.0.0.1",
                 engine: str = "waldb",
                 shard_name: str = "1.dev",
                 session_timeout_ms: int = 2000,
                 health_interval_ms: int = 500,
                 health_timeout_ms: int = 3000,
                 ops_timeout_ms: int = 30000,
                 replication_timeout_ms: int = 30000,
                 tick_interval_ms: int = 250,
                 singleton: bool = False,
                 base_port: Optional[int] = None,
                 run_snapshotter: bool = True,
                 snapshot_interval_ms: int = 30000,
                 snapshot_number: int = 5,
                 proxied: bool = False,
                 storage_provider: str = "dir",
                 pg_version: str = "12"):
        self.base_dir = os.path.abspath(base_dir)
        self.ip = ip
        self.engine = engine
        self.shard_path = "/manatee/" + shard_name
        self.session_timeout_ms = session_timeout_ms
        self.health_interval_ms = health_interval_ms
        self.health_timeout_ms = health_timeout_ms
        self.ops_timeout_ms = ops_timeout_ms
        self.replication_timeout_ms = replication_timeout_ms
        self.tick_interval_ms = tick_interval_ms
        self.singleton = singleton
        self.run_snapshotter = run_snapshotter
        self.snapshot_interval_ms = snapshot_interval_ms
        self.snapshot_number = snapshot_number
        self.proxied = proxied
        self.proxies: Dict[tuple, object] = {}
        self.storage_provider = storage_provider
        self.pg_version = pg_version
        self.pg_base_dir = ""
        if engine == "postgres":
            self.pg_base_dir = self._write_minipg_binaries()
        self.zk_port = 0  # assigned below the ephemeral range in __init__
        self.zk_conn_str = ""
        self.zk_proc: Optional[subprocess.Popen] = None
        self.peers: List[DevPeer] = []
        # pick a base below the ephemeral range so client sockets can never
        # collide with peer listen ports; partition the space by RANK so
        # concurrent torchrun ranks (one shard per rank) can never race
        # each other into the same block
        lo = max(10000, _EPHEMERAL_LOW - 22000)
        rank = int(os.environ.get("RANK", "0"))
        world = max(1, int(os.environ.get("WORLD_SIZE", "1")))
        span = max(400, 18000 // world)
        slot = lo + (rank % world) * span
        self._next_base_port = base_port or \
            (slot + (os.getpid() * 131) % max(1, span - 60)) // 10 * 10
        zk_peer = self.add_peer_config()   # reserve a port block for ZK
        self.peers.clear()
        self.zk_port = zk_peer.pg_port
        self.zk_conn_str = "%s:%d" % (ip, self.zk_port)
        for _ in range(n_peers):
            self.add_peer_config()

    def add_peer_config(self) -> DevPeer:
        # ports stepped +10 per peer (ref mkdevsitters port scheme)
        while True:
            base = self._next_base_port
            self._next_base_port += 10
            if base + 10 >= _EPHEMERAL_LOW:
                base = self._next_base_port = 10000
                self._next_base_port += 10
            try:
                for off in (0, 1, 2):
                    probe = socket.socket()
                    probe.setsockopt(socket.SOL_SOCKET,
                                     socket.SO_REUSEADDR, 1)
                    probe.bind((self.ip, base + off))
                    probe.close()
                break
            except OSError:
                continue
        peer = DevPeer(self, len(self.peers), base)
        self.peers.append(peer)
        return peer

    def _write_minipg_binaries(self) -> str:
        """Install minipg's initdb/postgres shims under the versioned
        layout the engine expects: <pgBaseDir>/<version>/bin/{initdb,
        postgres} (ref resolveVersionedPaths lib/postgresMgr.js:569-634;
        the reference's mkdevsitters builds real PG from source the same
        shape)."""
        base = os.path.join(self.base_dir, "pgbase")
        for version in ("12.0", "9.6.3"):
            bindir = os.path.join(base, version, "bin")
            os.makedirs(bindir, exist_ok=True)
            for name, fn in (("initdb", "initdb_main"),
                             ("postgres", "postgres_main")):
                path = os.path.join(bindir, name)
                with open(path, "w") as f:
                    f.write(
                        "#!%s\nimport sys\nsys.path.insert(0, %r)\n"
                        "from manatee_amd.db.minipg.server import %s\n"
                        "sys.exit(%s(%r, sys.argv[1:]))\n"
                        % (sys.executable, REPO_ROOT, fn, fn, version))
                os.chmod(path, 0o755)
        return base

    # -------------------------------------------------------------- control
    def start_zk(self) -> None:
        env = dict(os.environ)
        env["PYTHONPATH"] = REPO_ROOT + os.pathsep + \
            env.get("PYTHONPATH", "")
        os.makedirs(self.base_dir, exist_ok=True)
        logf = open(os.path.join(self.base_dir, "zk.log"), "a")
        self.zk_proc = subprocess.Popen(
            [sys.executable, "-m", "manatee_amd.coord.zkserver",
             "-H", self.ip, "-p", str(self.zk_port),
             "-j", os.path.join(self.base_dir, "zk-journal.jsonl")],
            env=env, stdout=logf, stderr=logf, start_new_session=True)

    def kill_zk(self) -> None:
        """SIGKILL the coordination server (full-ZK outage tier of the
        reference's chaos plan, docs/test-plan.md)."""
        if self.zk_proc is not None and self.zk_proc.poll() is None:
            try:
                os.killpg(self.zk_proc.pid, signal.SIGKILL)
            except ProcessLookupError:
                pass
            self.zk_proc.wait()
        self.zk_proc = None

    async def wait_zk(self, timeout_s: float = 15.0) -> None:
        deadline = time.monotonic() + timeout_s
        while True:
            try:
                r, w = await asyncio.open_connection(self.ip, self.zk_port)
                w.close()
                return
            except OSError:
                if time.monotonic() > deadline:
                    raise RuntimeError("zk server did not start")
                await asyncio.sleep(0.05)

    async def start(self, peers: Optional[List[int]] = None) -> None:
        if self.proxied:
            await self._start_proxies()
        self.start_zk()
        await self.wait_zk()
        for i, peer in enumerate(self.peers):
            if peers is None or i in peers:
                peer.start()

    # ---------------------------------------------------- network partitions
    async def _start_proxies(self) -> None:
        """One LinkProxy per directed (src peer → dst) link: dst is
        another peer's db/backup port or the ZK server.  Each peer's
        dial map (MANATEE_DIAL_MAP) routes its outbound connections
        through its own proxies, so tests can drop bytes per directed
        link — the ipdadm-network-partition analogue
        (ref docs/test-plan.md:24-113)."""
        from .netproxy import LinkProxy
        for src in self.peers:
            pz = LinkProxy(self.ip, self.zk_port,
                           name="peer%d->zk" % src.index)
            await pz.start()
            self.proxies[(src.index, "zk")] = pz
            for dst in self.peers:
                if dst.index == src.index:
                    continue
                for kind, port in (("pg", dst.pg_port),
                                   ("backup", dst.backup_port)):
                    p = LinkProxy(self.ip, port,
                                  name="peer%d->peer%d:%s"
                                  % (src.index, dst.index, kind))
                    await p.start()
                    self.proxies[(src.index, dst.index, kind)] = p
            # the dial map must exist before the peer spawns
            os.makedirs(src.dir, exist_ok=True)
            dmap = {"%s:%d" % (self.ip, self.zk_port): pz.addr}
            for dst in self.peers:
                if dst.index == src.index:
                    continue
                dmap["%s:%d" % (self.ip, dst.pg_port)] = \
                    self.proxies[(src.index, dst.index, "pg")].addr
                dmap["%s:%d" % (self.ip, dst.backup_port)] = \
                    self.proxies[(src.index, dst.index, "backup")].addr
            with open(os.path.join(src.dir, "dialmap.json"), "w") as f:
                json.dump(dmap, f, indent=2)

    def set_link(self, a: DevPeer, b: DevPeer,
                 drop_a2b: Optional[bool] = None,
                 drop_b2a: Optional[bool] = None) -> None:
        """Drop bytes traveling a→b and/or b→a, on BOTH carriers
        (connections a initiated to b AND connections b initiated to a).
        ``drop_a2b=True`` alone is a one-way partition: a's packets
        never reach b, but b's still reach a."""
        for kind in ("pg", "backup"):
            pab = self.proxies.get((a.index, b.index, kind))
            pba = self.proxies.get((b.index, a.index, kind))
            if pab is not None:
                pab.set_drops(to_server=drop_a2b, to_client=drop_b2a)
            if pba is not None:
                pba.set_drops(to_server=drop_b2a, to_client=drop_a2b)

    def partition(self, a: DevPeer, b: DevPeer) -> None:
        self.set_link(a, b, drop_a2b=True, drop_b2a=True)

    def heal_link(self, a: DevPeer, b: DevPeer) -> None:
        for kind in ("pg", "backup"):
            for key in ((a.index, b.index, kind), (b.index, a.index, kind)):
                p = self.proxies.get(key)
                if p is not None:
                    p.heal()

    def partition_zk(self, a: DevPeer) -> None:
        """Cut peer a off from the coordination server (its session
        expires) while leaving every peer↔peer link intact."""
        p = self.proxies.get((a.index, "zk"))
        if p is not None:
            p.set_drops(to_server=True, to_client=True)

    def heal_zk(self, a: DevPeer) -> None:
        p = self.proxies.get((a.index, "zk"))
        if p is not None:
            p.heal()

    def isolate(self, a: DevPeer) -> None:
        """Full network isolation of one peer (ZK + every peer link);
        clients (the test) can still reach it directly."""
        self.partition_zk(a)
        for b in self.peers:
            if b.index != a.index:
                self.partition(a, b)

    def heal_all(self) -> None:
        for p in self.proxies.values():
            p.heal()

    def stop(self) -> None:
        for peer in self.peers:
            peer.stop()
        for p in self.proxies.values():
            if p._server is not None:
                p._server.close()
            p.kill_connections()
        self.proxies.clear()
        if self.zk_proc is not None and self.zk_proc.poll() is None:
            try:
                os.killpg(self.zk_proc.pid, signal.SIGKILL)
            except ProcessLookupError:
                pass
            self.zk_proc.wait()
            self.zk_proc = None

    # ------------------------------------------------------------ inspection
    async def cluster_state(self) -> Optional[dict]:
        from ..coord.zkclient import ZkClient
        from ..coord import jute
        cli = ZkClient(self.zk_conn_str, session_timeout_ms=5000)
        try:
            await cli.connect(timeout_s=5)
            data, _ = await cli.get_data(self.shard_path + "/state")
            return json.loads(data)
        except (jute.ZkError, asyncio.TimeoutError, OSError):
            return None
        finally:
            await cli.close()

    async def wait_cluster(self, pred, timeout_s: float = 30.0,
                           what: str = "cluster state"):
        deadline = time.monotonic() + timeout_s
        while True:
            s = await self.cluster_state()
            if s is not None and pred(s):
                return s
            if time.monotonic() > deadline:
                raise AssertionError("timeout waiting for %s; last=%r"
                                     % (what, s))
            await asyncio.sleep(0.1)

    def peer_by_id(self, peer_id: str) -> DevPeer:
        for p in self.peers:
            if p.id == peer_id:
                return p
        raise KeyError(peer_id)

    async def rebuild_peer(self, peer: DevPeer,
                           timeout_s: float = 120.0) -> None:
        """The ``manatee-adm rebuild`` flow for a dead/deposed peer
        (ref lib/adm.js:1319-1684): stop it, destroy its dataset (it is
        deposed — its WAL may have diverged), remove it from the deposed
        list, restart it; it restores from the primary and rejoins as an
        async."""
        import shutil
        from ..adm import core as adm
        from ..storage import open_store
        peer.kill9()
        # destroy through the provider (for zfs the pool state lives
        # outside store_dir; ref deposed ⇒ destroyDataset lib/adm.js:1479)
        try:
            await open_store(peer.storage_cfg(), log=None).destroy()
        except Exception:
            pass
        shutil.rmtree(peer.store_dir, ignore_errors=True)
        zk = await adm.create_zk_client(self.zk_conn_str)
        try:
            state, version = await adm.get_state(zk, self.shard_path)
            if state and any(d["id"] == peer.id
                             for d in state.get("deposed", [])):
                await adm.reap(zk, self.shard_path, peer_id=peer.id)
        finally:
            await zk.close()
        peer.start()
        await self.wait_cluster(
            lambda s: any(a["id"] == peer.id for a in s.get("async", [])),
            timeout_s=timeout_s, what="rebuilt peer rejoining as async")

    async def wait_writable(self, timeout_s: float = 60.0) -> DevPeer:
        """Wait until the cluster primary accepts a write; returns it."""
        deadline = time.monotonic() + timeout_s
        last_err = None
        while time.monotonic() < deadline:
            s = await self.cluster_state()
            if s is not None:
                prim = self.peer_by_id(s["primary"]["id"])
                cli = prim.db_client()
                try:
                    await cli.put("__writable_probe__", time.time(),
                                  timeout_s=1.0)
                    await cli.close()
                    return prim
                except Exception as exc:
                    last_err = exc
                    await cli.close()
            await asyncio.sleep(0.1)
        raise AssertionError("cluster never became writable: %r" % last_err)


# ------------------------------------------------------------------ CLI
def main(argv=None) -> int:
    """Dev-cluster launcher — the mkdevsitters + "start all sitters"
    analogue (ref tools/mkdevsitters, docs/working-on-manatee.md):
    builds N peers with stepped ports under a directory, starts the
    embedded ZK + all daemons, prints how to reach them, and runs until
    interrupted."""
    import argparse
    import signal as _signal

    ap = argparse.ArgumentParser(
        prog="manatee-devcluster",
        description="Run a local N-peer manatee shard for development")
    ap.add_argument("-d", "--dir", required=True,
                    help="base directory for peer state/logs")
    ap.add_argument("-n", "--peers", type=int, default=3)
    ap.add_argument("-s", "--shard", default="1.dev")
    ap.add_argument("--singleton", action="store_true",
                    help="one peer in one-node-write mode")
    ap.add_argument("--session-timeout-ms", type=int, default=10000)
    ns = ap.parse_args(argv)

    async def run():
        c = DevCluster(ns.dir, n_peers=ns.peers, shard_name=ns.shard,
                       singleton=ns.singleton or ns.peers == 1,
                       session_timeout_ms=ns.session_timeout_ms)
        await c.start()
        print("zk:     %s" % c.zk_conn_str)
        print("shard:  %s" % c.shard_path)
        for p in c.peers:
            print("peer%d:  db %s:%d  status http://%s:%d  backup "
                  "http://%s:%d  dir %s"
                  % (p.index, p.ip, p.pg_port, p.ip, p.status_port,
                     p.ip, p.backup_port, p.dir))
        print("adm:    ZK_IPS=%s SHARD=%s bin/manatee-adm pg-status"
              % (c.zk_conn_str, c.shard_path))
        try:
            await c.wait_cluster(lambda s: s.get("primary"),
                                 timeout_s=120, what="formation")
            print("cluster formed; Ctrl-C to stop")
        except AssertionError as exc:
            print("WARNING: %s" % exc)
        stop = asyncio.Event()
        loop = asyncio.get_running_loop()
        for sig in (_signal.SIGINT, _signal.SIGTERM):
            loop.add_signal_handler(sig, stop.set)
        await stop.wait()
        c.stop()
        return 0

    return asyncio.run(run())


if __name__ == "__main__":
    import sys as _sys
    _sys.exit(main())
