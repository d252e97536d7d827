"""kukeon system user/group provisioning (reference internal/sysuser:
EnsureUserGroup + ownership fixups during `kuke init`).

The daemon socket is group-owned by `kukeon` so non-root users in that
group can drive the CLI; `kuke init` provisions the group (and, when
asked, a system user) and chowns the run tree. Hosts without useradd or
without root degrade gracefully — single-user operation keeps working.
"""
from __future__ import annotations

import grp
import logging
import os
import pwd
import shutil
import subprocess
from pathlib import Path
from typing import Optional

log = logging.getLogger("kukeon.sysuser")

GROUP = "kukeon"
USER = "kukeon"


def lookup_group(name: str = GROUP) -> Optional[int]:
    try:
        return grp.getgrnam(name).gr_gid
    except KeyError:
        return None


def lookup_user(name: str = USER) -> Optional[int]:
    try:
        return pwd.getpwnam(name).pw_uid
    except KeyError:
        return None


def ensure_group(name: str = GROUP) -> Optional[int]:
    """Create the kukeon group if missing; returns its gid (None when
    the host refuses — non-root, or no groupadd)."""
    gid = lookup_group(name)
    if gid is not None:
        return gid
    tool = shutil.which("groupadd")
    if tool is None or os.geteuid() != 0:
        return None
    r = subprocess.run([tool, "--system", name], capture_output=True,
                       text=True)
    if r.returncode != 0:
        log.warning("groupadd %s failed: %s", name, r.stderr.strip())
        return None
    return lookup_group(name)


def ensure_user(name: str = USER, group: str = GROUP) -> Optional[int]:
    """Create the kukeon system user (no login shell) if missing."""
    uid = lookup_user(name)
    if uid is not None:
        return uid
    if ensure_group(group) is None:
        return None
    tool = shutil.which("useradd")
    if tool is None or os.geteuid() != 0:
        return None
    r = subprocess.run([tool, "--system", "--gid", group,
                        "--shell", "/usr/sbin/nologin",
                        "--no-create-home", name],
                       capture_output=True, text=True)
    if r.returncode != 0:
        log.warning("useradd %s failed: %s", name, r.stderr.strip())
        return None
    return lookup_user(name)


def chown_tree(root: Path, gid: int, skip_suffixes=(".lock",)) -> int:
    """Group-own the run tree (reference ChownTreeAndChmodSkip): dirs
    g+rwxs, files g+rw; lock tombstones skipped. Returns entries
    changed."""
    changed = 0
    for p in [root, *root.rglob("*")]:
        if any(str(p).endswith(sfx) for sfx in skip_suffixes):
            continue
        try:
            st = p.stat()
            if st.st_gid != gid:
                os.chown(p, -1, gid)
                changed += 1
            if p.is_dir():
                os.chmod(p, (st.st_mode & 0o7777) | 0o2070)
        except OSError:
            continue
    return changed


def apply_socket_group(sock_path: str, gid: Optional[int]) -> None:
    """chown the daemon socket to the kukeon group, mode 0660
    (reference daemon/server.go:129-146)."""
    if gid is None:
        return
    try:
        os.chown(sock_path, -1, gid)
        os.chmod(sock_path, 0o660)
    except OSError as e:
        log.warning("socket group ownership: %s", e)
