"""Test harness: an in-process shard of ManateePeer FSMs against the
embedded ZK server with scriptable mock databases.

The FSM analogue of the reference's ``manatee-state-machine`` simulator
(SURVEY.md §2.2): explores topology transitions without real databases.
"""

import asyncio
from typing import Dict, List, Optional

from manatee_amd.common import lsn as lsnmod
from manatee_amd.coord.zkmgr import ZkMgr
from manatee_amd.coord.zkserver import ZkServer
from manatee_amd.fsm import state as st
from manatee_amd.fsm.peer import ManateePeer

SHARD_PATH = "/manatee/testshard"


class MockDb:
    """Scriptable database manager implementing the FSM-facing contract
    (ref lib/postgresMgr.js events :401-421, reconfigure :758-867)."""

    def __init__(self, xlog: str = lsnmod.ZERO):
        self._listeners: Dict[str, list] = {}
        self.configs: List[dict] = []
        self.current: Optional[dict] = None
        self.xlog = xlog
        self.fail_reconfigure = False

    def on(self, event, cb):
        self._listeners.setdefault(event, []).append(cb)

    def emit(self, event, *args):
        for cb in self._listeners.get(event, []):
            cb(*args)

    def fire_init(self, setup=True, online=False):
        self.emit("init", {"setup": setup, "online": online})

    async def reconfigure(self, cfg):
        if self.fail_reconfigure:
            raise RuntimeError("injected reconfigure failure")
        self.configs.append(cfg)
        self.current = cfg

    async def get_xlog_location(self):
        return self.xlog

    async def stop(self):
        pass

    @property
    def role(self):
        return self.current["role"] if self.current else None


class TestPeer:
    def __init__(self, ip: str, srv: ZkServer, singleton=False,
                 session_timeout_ms=1000, xlog=lsnmod.ZERO):
        self.id = "%s:5432:12345" % ip
        self.ident = st.make_ident(self.id, zone_id="zone-" + ip)
        self.zk = ZkMgr(id=self.id, data={k: v for k, v in self.ident.items()
                                          if k != "id"},
                        path=SHARD_PATH, conn_str=srv.conn_str,
                        session_timeout_ms=session_timeout_ms)
        self.db = MockDb(xlog=xlog)
        self.fsm = ManateePeer(zk=self.zk, db=self.db, self_ident=self.ident,
                               singleton=singleton, tick_interval_s=0.1)

    async def start(self):
        self.fsm.start()
        await self.zk.init()
        self.db.fire_init()

    async def kill(self):
        """SIGKILL analogue: sever the ZK session without clean close."""
        await self.fsm.close()
        self.zk._closed = True
        cli = self.zk._zk
        if cli is not None:
            cli._closing = True
            if cli._writer is not None:
                cli._writer.close()
            for t in (cli._mgr_task, cli._io_task, cli._ping_task):
                if t is not None:
                    t.cancel()

    async def close(self):
        await self.fsm.close()
        await self.zk.close()


class Shard:
    def __init__(self, session_timeout_ms=1000):
        self.srv: Optional[ZkServer] = None
        self.peers: Dict[str, TestPeer] = {}
        self.session_timeout_ms = session_timeout_ms

    async def start(self, n_peers=3, singleton=False):
        self.srv = ZkServer(tick_ms=50, min_session_timeout_ms=300)
        await self.srv.start()
        for i in range(n_peers):
            await self.add_peer("10.0.0.%d" % (i + 1), singleton=singleton)
            # small pause so election sequence order is deterministic
            await asyncio.sleep(0.05)
        return self

    async def add_peer(self, ip, singleton=False, xlog=lsnmod.ZERO):
        p = TestPeer(ip, self.srv, singleton=singleton,
                     session_timeout_ms=self.session_timeout_ms, xlog=xlog)
        self.peers[p.id] = p
        await p.start()
        return p

    def peer(self, i) -> TestPeer:
        return list(self.peers.values())[i]

    async def state(self) -> Optional[dict]:
        import json
        # read authoritatively from the server's own tree
        node = self.srv.nodes.get(SHARD_PATH + "/state")
        return json.loads(node.data) if node else None

    async def wait_state(self, pred, timeout=10.0, what="cluster state"):
        deadline = asyncio.get_running_loop().time() + timeout
        while True:
            s = await self.state()
            if s is not None and pred(s):
                return s
            if asyncio.get_running_loop().time() > deadline:
                raise AssertionError("timeout waiting for %s; last=%r"
                                     % (what, s))
            await asyncio.sleep(0.05)

    async def stop(self):
        for p in self.peers.values():
            try:
                await p.close()
            except Exception:
                pass
        if self.srv:
            await self.srv.stop()


def pid(ip):
    return "%s:5432:12345" % ip
