"""fakezfs — a behavioral ZFS CLI emulation for hosts without ZFS.

No ZFS kernel module or userland exists in this environment, so the
ZfsStore fork-exec path (storage/zfsstore.py, mirroring the reference's
zfs usage in lib/common.js:148-451 / lib/zfsClient.js /
lib/backupSender.js) is exercised against this emulation instead of a
scripted per-test stub: real hierarchical datasets with properties and
strict-parent creation, point-in-time snapshot copies, mount/canmount
semantics (a mounted dataset's contents appear at its effective
mountpoint via symlink), ``rename -p`` isolation that carries children
and snapshots along, and ``send | recv`` streams that recreate the
snapshot on the receiver — the full grammar ZfsStore emits, with
ZFS's error behaviors (missing parent, busy mountpoint, duplicate
snapshot) rather than blanket success.

State lives under a root directory; the generated shim bakes the root
in (the store scrubs the environment exactly like the reference, so no
env var can carry it).  Mutations take an flock so concurrent daemons
(sitter + backupserver) behave.

    from manatee_amd.tools.fakezfs import install_fakezfs
    zfs_path = install_fakezfs(bindir, state_root)
"""

from __future__ import annotations

import fcntl
import json
import os
import shutil
import subprocess
import sys
from typing import List, Optional


class ZfsError(Exception):
    pass


class FakeZfs:
    def __init__(self, root: str):
        self.root = os.path.abspath(root)
        self.data = os.path.join(self.root, "data")
        self.snap = os.path.join(self.root, "snap")
        self.props = os.path.join(self.root, "props")
        for d in (self.data, self.snap, self.props):
            os.makedirs(d, exist_ok=True)

    # ------------------------------------------------------------- helpers
    def _lock(self):
        f = open(os.path.join(self.root, ".lock"), "w")
        fcntl.flock(f, fcntl.LOCK_EX)
        return f

    def _data_dir(self, ds: str) -> str:
        return os.path.join(self.data, ds)

    def _props_path(self, ds: str) -> str:
        return os.path.join(self.props, ds.replace("/", "%") + ".json")

    def _snap_dir(self, ds: str, name: str) -> str:
        return os.path.join(self.snap, "%s@%s" % (ds.replace("/", "%"),
                                                  name))

    def exists(self, ds: str) -> bool:
        return os.path.isdir(self._data_dir(ds))

    def get_props(self, ds: str) -> dict:
        try:
            with open(self._props_path(ds)) as f:
                return json.load(f)
        except (OSError, ValueError):
            return {}

    def set_props(self, ds: str, props: dict) -> None:
        with open(self._props_path(ds), "w") as f:
            json.dump(props, f)

    def effective_mountpoint(self, ds: str) -> Optional[str]:
        """Nearest ancestor's local mountpoint + relative path (the ZFS
        inheritance rule); None when no ancestor sets one."""
        parts = ds.split("/")
        for i in range(len(parts), 0, -1):
            anc = "/".join(parts[:i])
            mp = self.get_props(anc).get("mountpoint")
            if mp:
                rel = "/".join(parts[i:])
                return os.path.join(mp, rel) if rel else mp
        return None

    def is_mounted(self, ds: str) -> bool:
        mp = self.effective_mountpoint(ds)
        return bool(mp) and os.path.islink(mp) and \
            os.path.realpath(mp) == os.path.realpath(self._data_dir(ds))

    def children(self, ds: str) -> List[str]:
        out = []
        base = self._data_dir(ds)
        for cur, dirs, _files in os.walk(base):
            rel = os.path.relpath(cur, base)
            if rel == ".":
                continue
            # only dirs that are registered datasets count
            cand = ds + "/" + rel.replace(os.sep, "/")
            if os.path.exists(self._props_path(cand)):
                out.append(cand)
        return sorted(out)

    def snapshots_of(self, ds: str) -> List[str]:
        pref = ds.replace("/", "%") + "@"
        out = []
        for name in os.listdir(self.snap):
            if name.startswith(pref):
                out.append(name[len(pref):])
        return sorted(out)

    # ------------------------------------------------------------ commands
    def cmd_create(self, args: List[str]) -> None:
        props = {}
        while args and args[0] == "-o":
            k, _, v = args[1].partition("=")
            props[k] = v
            args = args[2:]
        if not args:
            raise ZfsError("missing dataset argument")
        ds = args[0]
        if self.exists(ds):
            raise ZfsError("cannot create '%s': dataset already exists"
                           % ds)
        parent = ds.rsplit("/", 1)[0] if "/" in ds else None
        if parent and not self.exists(parent):
            raise ZfsError("cannot create '%s': parent does not exist"
                           % ds)
        os.makedirs(self._data_dir(ds))
        self.set_props(ds, props)
        if props.get("canmount", "on") != "off":
            self.cmd_mount([ds], strict=False)

    def cmd_list(self, args: List[str]) -> str:
        if args[:2] == ["-t", "snapshot"]:
            ds = args[-1]
            if not self.exists(ds):
                raise ZfsError("cannot open '%s': dataset does not exist"
                               % ds)
            return "".join("%s@%s\n" % (ds, s)
                           for s in self.snapshots_of(ds))
        ds = args[-1]
        if not self.exists(ds):
            raise ZfsError("cannot open '%s': dataset does not exist" % ds)
        return ds + "\n"

    def cmd_get(self, args: List[str]) -> str:
        # get -H -o value PROP DS
        args = [a for a in args if a not in ("-H",)]
        if args[:2] == ["-o", "value"]:
            args = args[2:]
        prop, ds = args[0], args[1]
        if not self.exists(ds):
            raise ZfsError("cannot open '%s': dataset does not exist" % ds)
        if prop == "mounted":
            return ("yes" if self.is_mounted(ds) else "no") + "\n"
        if prop == "mountpoint":
            return (self.effective_mountpoint(ds) or "none") + "\n"
        return str(self.get_props(ds).get(prop, "-")) + "\n"

    def cmd_set(self, args: List[str]) -> None:
        kv, ds = args[0], args[1]
        k, _, v = kv.partition("=")
        if not self.exists(ds):
            raise ZfsError("cannot open '%s': dataset does not exist" % ds)
        props = self.get_props(ds)
        was_mounted = self.is_mounted(ds)
        old_mp = self.effective_mountpoint(ds)
        props[k] = v
        self.set_props(ds, props)
        if k == "canmount" and v == "off" and was_mounted and old_mp:
            os.unlink(old_mp)
        if k == "mountpoint" and was_mounted and old_mp and old_mp != v:
            os.unlink(old_mp)   # remount happens on explicit `zfs mount`

    def cmd_inherit(self, args: List[str]) -> None:
        prop, ds = args[0], args[1]
        if not self.exists(ds):
            raise ZfsError("cannot open '%s': dataset does not exist" % ds)
        props = self.get_props(ds)
        was_mounted = self.is_mounted(ds)
        old_mp = self.effective_mountpoint(ds)
        props.pop(prop, None)
        self.set_props(ds, props)
        if prop == "mountpoint" and was_mounted and old_mp:
            os.unlink(old_mp)

    def cmd_mount(self, args: List[str], strict: bool = True) -> None:
        ds = args[-1]
        if not self.exists(ds):
            raise ZfsError("cannot open '%s': dataset does not exist" % ds)
        if self.get_props(ds).get("canmount", "on") == "off":
            if strict:
                raise ZfsError("cannot mount '%s': canmount=off" % ds)
            return
        mp = self.effective_mountpoint(ds)
        if not mp:
            if strict:
                raise ZfsError("cannot mount '%s': no mountpoint" % ds)
            return
        if self.is_mounted(ds):
            return
        if os.path.islink(mp):
            os.unlink(mp)
        elif os.path.isdir(mp):
            if os.listdir(mp):
                raise ZfsError("cannot mount '%s': directory is not empty"
                               % ds)
            os.rmdir(mp)
        os.makedirs(os.path.dirname(mp), exist_ok=True)
        os.symlink(self._data_dir(ds), mp)

    def cmd_snapshot(self, args: List[str]) -> None:
        full = args[-1]
        ds, _, name = full.partition("@")
        if not self.exists(ds):
            raise ZfsError("cannot open '%s': dataset does not exist" % ds)
        tgt = self._snap_dir(ds, name)
        if os.path.exists(tgt):
            raise ZfsError("cannot create snapshot '%s': dataset already "
                           "exists" % full)
        shutil.copytree(self._data_dir(ds), tgt, symlinks=True)

    def cmd_destroy(self, args: List[str]) -> None:
        recursive = False
        if args and args[0] == "-r":
            recursive = True
            args = args[1:]
        full = args[0]
        if "@" in full:
            ds, _, name = full.partition("@")
            tgt = self._snap_dir(ds, name)
            if not os.path.exists(tgt):
                raise ZfsError("could not find any snapshots to destroy")
            shutil.rmtree(tgt)
            return
        ds = full
        if not self.exists(ds):
            raise ZfsError("cannot open '%s': dataset does not exist" % ds)
        kids = self.children(ds)
        if kids and not recursive:
            raise ZfsError("cannot destroy '%s': filesystem has children"
                           % ds)
        for d in [ds] + kids:
            if self.is_mounted(d):
                os.unlink(self.effective_mountpoint(d))
            for s in self.snapshots_of(d):
                shutil.rmtree(self._snap_dir(d, s))
            try:
                os.unlink(self._props_path(d))
            except OSError:
                pass
        shutil.rmtree(self._data_dir(ds))

    def cmd_rename(self, args: List[str]) -> None:
        create_parents = False
        if args and args[0] == "-p":
            create_parents = True
            args = args[1:]
        src, dst = args[0], args[1]
        if not self.exists(src):
            raise ZfsError("cannot open '%s': dataset does not exist"
                           % src)
        if self.exists(dst):
            raise ZfsError("cannot rename '%s': dataset already exists"
                           % dst)
        parent = dst.rsplit("/", 1)[0] if "/" in dst else None
        if parent and not self.exists(parent):
            if not create_parents:
                raise ZfsError("cannot rename '%s': parent of target "
                               "does not exist" % src)
            # create intermediates top-down (the -p behavior)
            parts = parent.split("/")
            for i in range(1, len(parts) + 1):
                anc = "/".join(parts[:i])
                if not self.exists(anc):
                    os.makedirs(self._data_dir(anc))
                    self.set_props(anc, {"canmount": "off"})
        if self.is_mounted(src):
            os.unlink(self.effective_mountpoint(src))
        moves = [(src, dst)] + [(c, dst + c[len(src):])
                                for c in self.children(src)]
        # snapshots and props move with their datasets
        for s, d in sorted(moves, key=lambda m: m[0], reverse=True):
            for snap in self.snapshots_of(s):
                os.rename(self._snap_dir(s, snap), self._snap_dir(d, snap))
            if os.path.exists(self._props_path(s)):
                os.rename(self._props_path(s), self._props_path(d))
        os.makedirs(os.path.dirname(self._data_dir(dst)), exist_ok=True)
        os.rename(self._data_dir(src), self._data_dir(dst))

    def cmd_send(self, args: List[str]) -> Optional[str]:
        dry = False
        if args and args[0] in ("-nvP", "-nv", "-vP"):
            dry = "n" in args[0]
            args = args[1:]
        full = args[-1]
        ds, _, name = full.partition("@")
        tgt = self._snap_dir(ds, name)
        if not os.path.exists(tgt):
            raise ZfsError("cannot open '%s': snapshot does not exist"
                           % full)
        size = sum(os.path.getsize(os.path.join(cur, f))
                   for cur, _d, files in os.walk(tgt) for f in files)
        if dry:
            return "size %d\n" % size
        # stream: header naming the snapshot (recv recreates it), then tar
        sys.stdout.buffer.write(b"FAKEZFS1 %s\n" % name.encode())
        sys.stdout.buffer.flush()
        subprocess.run(["tar", "-cf", "-", "-C", tgt, "."],
                       stdout=sys.stdout.buffer, check=True)
        return None

    def cmd_recv(self, args: List[str]) -> None:
        args = [a for a in args if a not in ("-u", "-F", "-v")]
        ds = args[-1]
        parent = ds.rsplit("/", 1)[0] if "/" in ds else None
        if parent and not self.exists(parent):
            raise ZfsError("cannot receive: parent '%s' does not exist"
                           % parent)
        # read the header byte-by-byte from the RAW fd: a buffered
        # readline() would read ahead past the newline and swallow the
        # start of the tar stream before tar inherits the fd
        fd = sys.stdin.buffer.fileno()
        header = bytearray()
        while not header.endswith(b"\n"):
            b = os.read(fd, 1)
            if not b:
                break
            header += b
        if not header.startswith(b"FAKEZFS1 "):
            raise ZfsError("invalid stream (bad magic)")
        snap_name = header.split(b" ", 1)[1].strip().decode()
        ddir = self._data_dir(ds)
        if os.path.isdir(ddir):
            shutil.rmtree(ddir)     # -F: discard existing contents
        os.makedirs(ddir)
        if not os.path.exists(self._props_path(ds)):
            self.set_props(ds, {"canmount": "noauto"})
        subprocess.run(["tar", "-xf", "-", "-C", ddir],
                       stdin=sys.stdin.buffer, check=True)
        # recv recreates the sent snapshot on the receiver
        tgt = self._snap_dir(ds, snap_name)
        if os.path.exists(tgt):
            shutil.rmtree(tgt)
        shutil.copytree(ddir, tgt, symlinks=True)


def main(root: str, argv: List[str]) -> int:
    z = FakeZfs(root)
    if not argv:
        print("usage: zfs <command> ...", file=sys.stderr)
        return 2
    cmd, args = argv[0], argv[1:]
    fn = getattr(z, "cmd_" + cmd, None)
    if fn is None:
        print("unrecognized command '%s'" % cmd, file=sys.stderr)
        return 2
    lock = z._lock()
    try:
        out = fn(args)
        if isinstance(out, str):
            sys.stdout.write(out)
        return 0
    except ZfsError as exc:
        print("cannot %s: %s" % (cmd, exc), file=sys.stderr)
        return 1
    finally:
        lock.close()


SHIM = """#!%(python)s
import sys
sys.path.insert(0, %(repo)r)
from manatee_amd.tools.fakezfs import main
sys.exit(main(%(root)r, sys.argv[1:]))
"""


def install_fakezfs(bindir: str, state_root: str) -> str:
    """Write an executable ``zfs`` shim bound to a state root; returns
    its path (usable as ZfsStore's zfs_path / storageCfg.zfsPath)."""
    repo = os.path.dirname(os.path.dirname(
        os.path.dirname(os.path.abspath(__file__))))
    os.makedirs(bindir, exist_ok=True)
    os.makedirs(state_root, exist_ok=True)
    path = os.path.join(bindir, "zfs")
    with open(path, "w") as f:
        f.write(SHIM % {"python": sys.executable, "repo": repo,
                        "root": os.path.abspath(state_root)})
    os.chmod(path, 0o755)
    return path


if __name__ == "__main__":
    root = os.environ.get("FAKEZFS_ROOT", "/var/tmp/fakezfs")
    sys.exit(main(root, sys.argv[1:]))
