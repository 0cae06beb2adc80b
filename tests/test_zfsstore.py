"""ZfsStore against a scripted fake ``zfs`` binary: verifies the exact
command grammar (the reference's zfs usage, lib/common.js:148-451,
lib/zfsClient.js, lib/backupSender.js) and the send/recv byte path,
without needing real ZFS in the image."""

import asyncio
import json
import os
import stat

import pytest

from manatee_amd.storage.zfsstore import ZfsStore

FAKE_ZFS = r'''#!/bin/bash
# scripted zfs: state lives in $FAKE_ZFS_DIR
D="$FAKE_ZFS_DIR"
echo "$@" >> "$D/calls.log"
cmd="$1"; shift
case "$cmd" in
  list)
    if [ "$1" = "-t" ]; then           # list -t snapshot -H -o name -r DS
      ds="${@: -1}"
      touch "$D/snaps"
      while read -r s; do echo "$ds@$s"; done < "$D/snaps"
      exit 0
    fi
    [ -e "$D/exists" ] && exit 0 || exit 1 ;;
  create) touch "$D/exists"; exit 0 ;;
  get) echo "yes"; exit 0 ;;
  set|inherit|mount|rename) exit 0 ;;
  snapshot) echo "${1#*@}" >> "$D/snaps"; exit 0 ;;
  destroy)
    if [[ "$1" == "-r" ]]; then rm -f "$D/exists"; exit 0; fi
    snap="${1#*@}"
    grep -v "^$snap$" "$D/snaps" > "$D/snaps.t" 2>/dev/null || true
    mv "$D/snaps.t" "$D/snaps"; exit 0 ;;
  send)
    if [ "$1" = "-nvP" ]; then echo "size 12345"; exit 0; fi
    cat "$D/payload"; exit 0 ;;
  recv) cat > "$D/received"; exit 0 ;;
  *) echo "unknown: $cmd" >&2; exit 2 ;;
esac
'''


@pytest.fixture
def store(tmp_path, monkeypatch):
    d = tmp_path / "fakezfs"
    d.mkdir()
    zfs = tmp_path / "zfs"
    zfs.write_text(FAKE_ZFS)
    zfs.chmod(zfs.stat().st_mode | stat.S_IEXEC)
    monkeypatch.setenv("FAKE_ZFS_DIR", str(d))
    # the store scrubs env; propagate the state dir through a wrapper
    wrap = tmp_path / "zfswrap"
    wrap.write_text("#!/bin/bash\nFAKE_ZFS_DIR=%s exec %s \"$@\"\n"
                    % (d, zfs))
    wrap.chmod(wrap.stat().st_mode | stat.S_IEXEC)
    s = ZfsStore("tank/manatee/data", str(tmp_path / "mnt"),
                 zfs_path=str(wrap))
    s._state_dir = str(d)
    return s


def calls(store):
    try:
        with open(os.path.join(store._state_dir, "calls.log")) as f:
            return [l.strip() for l in f]
    except FileNotFoundError:
        return []


def run(coro):
    return asyncio.run(asyncio.wait_for(coro, 30))


def test_ensure_create_and_mount_grammar(store):
    async def go():
        assert not await store.exists()
        await store.ensure()
        assert await store.exists()
        await store.ensure()     # second time: mounted check only
    run(go())
    log = calls(store)
    assert "create -o mountpoint=%s tank/manatee/data" \
        % store.mountpoint() in log
    assert "get -H -o value mounted tank/manatee/data" in log


def test_snapshot_lifecycle(store):
    async def go():
        await store.ensure()
        await store.snapshot("1700000000000")
        await store.snapshot("1700000000001")
        assert await store.list_snapshots() == ["1700000000000",
                                                "1700000000001"]
        await store.destroy_snapshot("1700000000000")
        assert await store.list_snapshots() == ["1700000000001"]
        assert await store.send_size("1700000000001") == 12345
    run(go())
    assert "snapshot tank/manatee/data@1700000000000" in calls(store)


def test_send_recv_byte_fidelity(store):
    payload = os.urandom(3 << 20)
    with open(os.path.join(store._state_dir, "payload"), "wb") as f:
        f.write(payload)

    async def go():
        await store.ensure()
        chunks = await store.send("1700000000000")
        buf = b""
        async for c in chunks:
            buf += c
        assert buf == payload

        async def gen():
            for i in range(0, len(payload), 1 << 18):
                yield payload[i:i + (1 << 18)]
        await store.recv(gen())
    run(go())
    with open(os.path.join(store._state_dir, "received"), "rb") as f:
        assert f.read() == payload
    log = calls(store)
    assert "recv -u -F tank/manatee/data" in log
    # post-receive fixups (ref lib/zfsClient.js:152-183)
    assert "set canmount=noauto tank/manatee/data" in log
    assert "inherit snapdir tank/manatee/data" in log


def test_isolate_renames_not_deletes(store):
    async def go():
        await store.ensure()
        target = await store.isolate("autorebuild")
        assert target.startswith("tank/manatee/isolated/autorebuild-")
    run(go())
    log = calls(store)
    assert "set canmount=off tank/manatee/data" in log
    assert "inherit mountpoint tank/manatee/data" in log
    assert any(l.startswith("rename -p tank/manatee/data "
                            "tank/manatee/isolated/autorebuild-")
               for l in log)
    assert not any(l.startswith("destroy") for l in log)


def test_destroy_recursive(store):
    async def go():
        await store.ensure()
        await store.destroy()
        assert not await store.exists()
    run(go())
    assert "destroy -r tank/manatee/data" in calls(store)
