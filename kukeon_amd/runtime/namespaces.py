"""Linux namespace primitives for process cells (pure ctypes, no
external binaries).

Pod model (reference internal/ctr/spec.go:38 JoinContainerNamespaces +
runner/start.go:794-915): the cell's root pause process OWNS the net,
UTS and IPC namespaces; every other container of the cell JOINS them via
setns on /proc/<rootpid>/ns/*, and additionally gets a PRIVATE mount
namespace where /etc/hosts and /etc/hostname are bind-mounted from
runner-rendered files (reference runner/cell_etc_files.go) and an
optional image rootfs is entered by pivot/chroot.

Everything degrades: hosts that refuse unshare/setns (no CAP_SYS_ADMIN)
fall back to plain host-namespace processes and the cell status records
the degradation.
"""
from __future__ import annotations

import ctypes
import ctypes.util
import os
from typing import Dict, List, Optional

CLONE_NEWNS = 0x00020000
CLONE_NEWUTS = 0x04000000
CLONE_NEWIPC = 0x08000000
CLONE_NEWNET = 0x40000000
CLONE_NEWPID = 0x20000000

MS_BIND = 4096
MS_REC = 16384
MS_RDONLY = 1
MS_REMOUNT = 32
MS_PRIVATE = 1 << 18

_libc = ctypes.CDLL(ctypes.util.find_library("c") or "libc.so.6",
                    use_errno=True)

NS_FLAGS = {"mnt": CLONE_NEWNS, "uts": CLONE_NEWUTS, "ipc": CLONE_NEWIPC,
            "net": CLONE_NEWNET, "pid": CLONE_NEWPID}


def _check(rc: int, what: str) -> None:
    if rc != 0:
        e = ctypes.get_errno()
        raise OSError(e, f"{what}: {os.strerror(e)}")


def unshare(flags: int) -> None:
    _check(_libc.unshare(flags), "unshare")


def setns(fd: int, flags: int = 0) -> None:
    _check(_libc.setns(fd, flags), "setns")


def sethostname(name: str) -> None:
    b = name.encode()
    _check(_libc.sethostname(b, len(b)), "sethostname")


def mount(src: str, target: str, fstype: str = "", flags: int = 0,
          data: str = "") -> None:
    _check(_libc.mount(src.encode() if src else None, target.encode(),
                       fstype.encode() if fstype else None,
                       ctypes.c_ulong(flags),
                       data.encode() if data else None), f"mount {target}")


def bind_mount(src: str, target: str) -> None:
    mount(src, target, "", MS_BIND)


def make_mounts_private() -> None:
    """After unshare(CLONE_NEWNS), stop mount events propagating back to
    the host (mount --make-rprivate /)."""
    mount("none", "/", "", MS_REC | MS_PRIVATE)


def join(root_pid: int, kinds: List[str]) -> List[str]:
    """setns into the root container's namespaces; returns the kinds
    actually joined (missing /proc entries or EPERM are skipped)."""
    joined = []
    for kind in kinds:
        path = f"/proc/{root_pid}/ns/{kind}"
        try:
            fd = os.open(path, os.O_RDONLY)
        except OSError:
            continue
        try:
            setns(fd, NS_FLAGS.get(kind, 0))
            joined.append(kind)
        except OSError:
            pass
        finally:
            os.close(fd)
    return joined


def can_unshare(flags: int = CLONE_NEWUTS) -> bool:
    """Probe namespace capability without changing this process: fork a
    child that tries unshare and report its success."""
    pid = os.fork()
    if pid == 0:
        try:
            unshare(flags)
            os._exit(0)
        except OSError:
            os._exit(1)
    _, status = os.waitpid(pid, 0)
    return os.waitstatus_to_exitcode(status) == 0


def setup_etc(hostname: str, hosts_entries: Dict[str, str],
              workdir: str) -> None:
    """Inside a private mount namespace: render /etc/hostname and
    /etc/hosts under workdir and bind-mount them over the host files
    (reference runner/cell_etc_files.go)."""
    os.makedirs(workdir, exist_ok=True)
    hn = os.path.join(workdir, "hostname")
    with open(hn, "w") as f:
        f.write(hostname + "\n")
    ho = os.path.join(workdir, "hosts")
    with open(ho, "w") as f:
        f.write("127.0.0.1\tlocalhost\n")
        f.write(f"127.0.1.1\t{hostname}\n")
        for name, ip in sorted(hosts_entries.items()):
            f.write(f"{ip}\t{name}\n")
    bind_mount(hn, "/etc/hostname")
    bind_mount(ho, "/etc/hosts")


def enter_rootfs(rootfs: str) -> None:
    """chroot into a prepared image rootfs (kukebuild-analog artifacts).
    The caller is already in a private mount namespace."""
    os.chroot(rootfs)
    os.chdir("/")
