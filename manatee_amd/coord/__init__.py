"""Coordination layer: ZooKeeper wire protocol, embedded server, client and
the shard coordination manager (ref: lib/zookeeperMgr.js)."""
