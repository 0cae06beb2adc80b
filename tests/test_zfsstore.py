"""ZfsStore against the behavioral fakezfs emulation (no ZFS kernel or
userland exists in this image): real dataset semantics — hierarchical
create with strict parents, mount/canmount, point-in-time snapshots,
``send | recv`` that recreates the snapshot on the receiver, ``rename
-p`` isolation — through the exact fork-exec grammar the store emits
(ref lib/common.js:148-451, lib/zfsClient.js, lib/backupSender.js)."""

import asyncio
import os
import subprocess

import pytest

from manatee_amd.storage.zfsstore import ZfsStore
from manatee_amd.tools.fakezfs import install_fakezfs


def run(coro, timeout=120):
    return asyncio.run(asyncio.wait_for(coro, timeout))


@pytest.fixture
def zfs_path(tmp_path):
    path = install_fakezfs(str(tmp_path / "bin"), str(tmp_path / "pool"))
    for parent in ("tank", "tank/manatee"):
        subprocess.run([path, "create", "-o", "canmount=off", parent],
                       capture_output=True)
    return path


@pytest.fixture
def store(tmp_path, zfs_path):
    return ZfsStore("tank/manatee/data", str(tmp_path / "mnt" / "live"),
                    zfs_path=zfs_path)


def test_ensure_mount_and_write_through_mountpoint(store):
    async def go():
        assert not await store.exists()
        await store.ensure()
        assert await store.exists()
        # contents written via the mountpoint are the dataset's contents
        with open(os.path.join(store.mountpoint(), "f.txt"), "w") as f:
            f.write("hello")
        # ensure() again is a no-op on a mounted dataset
        await store.ensure()
        with open(os.path.join(store.mountpoint(), "f.txt")) as f:
            assert f.read() == "hello"
    run(go())


def test_create_requires_parent(tmp_path, zfs_path):
    async def go():
        s = ZfsStore("tank/nosuch/parent/data", str(tmp_path / "m2"),
                     zfs_path=zfs_path)
        with pytest.raises(Exception):
            await s.ensure()
    run(go())


def test_snapshots_are_point_in_time(store):
    async def go():
        await store.ensure()
        p = os.path.join(store.mountpoint(), "data.bin")
        with open(p, "w") as f:
            f.write("v1")
        name = await store.snapshot("1000000000001")
        assert name == "1000000000001"
        with open(p, "w") as f:
            f.write("v2-after-snapshot")
        # duplicate snapshot name fails like real zfs
        with pytest.raises(Exception):
            await store.snapshot("1000000000001")
        assert await store.list_snapshots() == ["1000000000001"]
        assert await store.send_size("1000000000001") > 0
        await store.destroy_snapshot("1000000000001")
        assert await store.list_snapshots() == []
    run(go())


def test_send_recv_bootstrap_recreates_snapshot(tmp_path, zfs_path):
    """The bootstrap pipeline (ref zfs send | zfs recv over TCP,
    lib/backupSender.js:154-242 → lib/zfsClient.js:765-886): the
    receiver ends up with the sender's contents AND the snapshot."""
    async def go():
        src = ZfsStore("tank/manatee/data", str(tmp_path / "m1" / "live"),
                       zfs_path=zfs_path)
        await src.ensure()
        for i in range(5):
            with open(os.path.join(src.mountpoint(), "f%d" % i),
                      "w") as f:
                f.write("payload-%d" % i)
        os.makedirs(os.path.join(src.mountpoint(), "sub"))
        with open(os.path.join(src.mountpoint(), "sub", "deep"),
                  "w") as f:
            f.write("nested")
        snap = await src.snapshot("1000000000777")

        # a second "host": its own pool
        zfs2 = install_fakezfs(str(tmp_path / "bin2"),
                               str(tmp_path / "pool2"))
        for parent in ("tank", "tank/manatee"):
            subprocess.run([zfs2, "create", "-o", "canmount=off", parent],
                           capture_output=True)
        dst = ZfsStore("tank/manatee/data", str(tmp_path / "m2" / "live"),
                       zfs_path=zfs2)
        await dst.recv(await src.send(snap))

        for i in range(5):
            with open(os.path.join(dst.mountpoint(), "f%d" % i)) as f:
                assert f.read() == "payload-%d" % i
        with open(os.path.join(dst.mountpoint(), "sub", "deep")) as f:
            assert f.read() == "nested"
        # recv recreated the snapshot (the receiver can re-serve it)
        assert await dst.list_snapshots() == [snap]
    run(go())


def test_isolate_preserves_data_and_frees_the_name(store, zfs_path):
    async def go():
        await store.ensure()
        with open(os.path.join(store.mountpoint(), "keep.me"), "w") as f:
            f.write("precious")
        target = await store.isolate("autorebuild")
        assert target.startswith("tank/manatee/isolated/autorebuild-")
        assert not await store.exists()
        # the data still exists under the isolated dataset (never deleted
        # on rebuild, ref isolateDataset lib/zfsClient.js:514-624)
        r = subprocess.run([zfs_path, "list", target],
                           capture_output=True, text=True)
        assert r.returncode == 0
        # and a fresh dataset can take the name again
        await store.ensure()
        assert await store.exists()
        assert os.listdir(store.mountpoint()) == []
    run(go())


def test_destroy(store):
    async def go():
        await store.ensure()
        await store.snapshot("1000000000002")
        await store.destroy()
        assert not await store.exists()
        assert await store.list_snapshots() == []
    run(go())
