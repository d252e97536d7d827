"""cgroup management for process cells: kukeon/<realm>/<space>/<stack>/<cell>.

Adaptive v2/v1: on a unified (v2) host, creates the subtree under the
mounted cgroup2 root with subtree_control delegation (cpu/memory/io/pids —
the reference's resource subset) and applies memory.max / cpu.weight; on a
hybrid/v1 host it uses the memory and devices hierarchies; with no writable
cgroupfs it degrades to a no-op (every operation is best-effort and
recorded, so unit tests assert against an injected fs root).
"""
from __future__ import annotations

import contextlib
import os
from pathlib import Path
from typing import List, Optional

RESOURCE_CONTROLLERS = ["cpu", "memory", "io", "pids"]


class CgroupManager:
    def __init__(self, root: Optional[str] = None, enabled: bool = True):
        self.enabled = enabled
        self.v2_root: Optional[Path] = None
        self.v1_roots: dict = {}
        if not enabled:
            return
        if root is not None:
            self.v2_root = Path(root)
            return
        if self._is_cgroup2("/sys/fs/cgroup"):
            self.v2_root = Path("/sys/fs/cgroup")
        else:
            for ctrl in ("memory", "devices", "cpu", "pids"):
                p = Path("/sys/fs/cgroup") / ctrl
                if p.is_dir() and os.access(p, os.W_OK):
                    self.v1_roots[ctrl] = p
            uni = Path("/sys/fs/cgroup/unified")
            if self._is_cgroup2(str(uni)):
                self.v2_root = uni

    @staticmethod
    def _is_cgroup2(path: str) -> bool:
        return os.path.exists(os.path.join(path, "cgroup.controllers"))

    @property
    def mode(self) -> str:
        if not self.enabled:
            return "disabled"
        if self.v2_root is not None:
            return "v2"
        if self.v1_roots:
            return "v1"
        return "none"

    def available_controllers(self) -> List[str]:
        if self.v2_root is not None:
            try:
                return (self.v2_root / "cgroup.controllers").read_text().split()
            except OSError:
                return []
        return list(self.v1_roots)

    # ------------------------------------------------------------------
    def _v2_dir(self, rel: str) -> Optional[Path]:
        return self.v2_root / "kukeon" / rel if self.v2_root else None

    def create(self, rel: str, nested_full_delegation: bool = False) -> List[str]:
        """Create the cgroup chain; returns created paths (for status)."""
        created = []
        if self.v2_root is not None:
            target = self._v2_dir(rel)
            try:
                target.mkdir(parents=True, exist_ok=True)
                created.append(str(target))
                self._delegate_chain(target, nested_full_delegation)
            except OSError:
                pass
        for ctrl, root in self.v1_roots.items():
            p = root / "kukeon" / rel
            with contextlib.suppress(OSError):
                p.mkdir(parents=True, exist_ok=True)
                created.append(str(p))
        return created

    def _delegate_chain(self, leaf: Path, full: bool) -> None:
        """Enable subtree controllers on every ancestor up to the v2 root
        (reference: subtree-controller delegation up the ancestor chain)."""
        try:
            avail = set((self.v2_root / "cgroup.controllers"
                         ).read_text().split())
        except OSError:
            return
        want = (sorted(avail) if full
                else [c for c in RESOURCE_CONTROLLERS if c in avail])
        node = leaf.parent
        chain = []
        while node != self.v2_root.parent and node != node.parent:
            chain.append(node)
            if node == self.v2_root:
                break
            node = node.parent
        for node in reversed(chain):
            ctl = node / "cgroup.subtree_control"
            with contextlib.suppress(OSError):
                ctl.write_text(" ".join(f"+{c}" for c in want))

    def attach(self, rel: str, pid: int) -> bool:
        ok = False
        if self.v2_root is not None:
            with contextlib.suppress(OSError):
                (self._v2_dir(rel) / "cgroup.procs").write_text(str(pid))
                ok = True
        for ctrl, root in self.v1_roots.items():
            with contextlib.suppress(OSError):
                (root / "kukeon" / rel / "cgroup.procs").write_text(str(pid))
                ok = True
        return ok

    def set_memory_limit(self, rel: str, limit_bytes: int) -> None:
        if limit_bytes <= 0:
            return
        if self.v2_root is not None:
            with contextlib.suppress(OSError):
                (self._v2_dir(rel) / "memory.max").write_text(str(limit_bytes))
        if "memory" in self.v1_roots:
            with contextlib.suppress(OSError):
                (self.v1_roots["memory"] / "kukeon" / rel /
                 "memory.limit_in_bytes").write_text(str(limit_bytes))

    def set_cpu_shares(self, rel: str, shares: int) -> None:
        if shares <= 0:
            return
        if self.v2_root is not None:
            # map docker-style shares (2..262144, default 1024) to cpu.weight
            weight = max(1, min(10000, int(1 + (shares - 2) * 9999 / 262142)))
            with contextlib.suppress(OSError):
                (self._v2_dir(rel) / "cpu.weight").write_text(str(weight))
        if "cpu" in self.v1_roots:
            with contextlib.suppress(OSError):
                (self.v1_roots["cpu"] / "kukeon" / rel /
                 "cpu.shares").write_text(str(shares))

    def allow_devices(self, rel: str, device_paths: List[str]) -> None:
        """v1 devices-cgroup pinning: deny-all then allow the listed nodes
        (+ default pseudo devices). The amdgpu pinning hook: /dev/kfd and
        /dev/dri/renderD<n> for the session's GPUs."""
        if "devices" not in self.v1_roots or not device_paths:
            return
        base = self.v1_roots["devices"] / "kukeon" / rel
        rules = []
        for p in device_paths:
            try:
                st = os.stat(p)
            except OSError:
                continue
            if not (os.path.exists(p)):
                continue
            major, minor = os.major(st.st_rdev), os.minor(st.st_rdev)
            kind = "c"
            rules.append(f"{kind} {major}:{minor} rwm")
        if not rules:
            return
        with contextlib.suppress(OSError):
            (base / "devices.deny").write_text("a")
            # re-allow standard pseudo devices
            for std in ("c 1:3 rwm", "c 1:5 rwm", "c 1:7 rwm", "c 1:8 rwm",
                        "c 1:9 rwm", "c 5:0 rwm", "c 5:2 rwm",
                        "c 136:* rwm"):
                (base / "devices.allow").write_text(std)
            for r in rules:
                (base / "devices.allow").write_text(r)

    def kill_all(self, rel: str) -> None:
        if self.v2_root is not None:
            with contextlib.suppress(OSError):
                (self._v2_dir(rel) / "cgroup.kill").write_text("1")

    def procs(self, rel: str) -> List[int]:
        if self.v2_root is not None:
            with contextlib.suppress(OSError):
                txt = (self._v2_dir(rel) / "cgroup.procs").read_text()
                return [int(x) for x in txt.split()]
        return []

    def delete(self, rel: str) -> None:
        if self.v2_root is not None:
            self._rmdir_tree(self._v2_dir(rel))
        for ctrl, root in self.v1_roots.items():
            self._rmdir_tree(root / "kukeon" / rel)

    @staticmethod
    def _rmdir_tree(p: Optional[Path]) -> None:
        if p is None or not p.is_dir():
            return
        for sub in sorted(p.rglob("*"), reverse=True):
            if sub.is_dir():
                with contextlib.suppress(OSError):
                    sub.rmdir()
        with contextlib.suppress(OSError):
            p.rmdir()
