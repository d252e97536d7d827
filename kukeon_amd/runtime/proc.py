"""PID liveness utilities with start-time pid-reuse guards."""
from __future__ import annotations

import contextlib
import os
import signal
import time
from typing import Optional


def _stat_fields(pid: int) -> Optional[list]:
    try:
        with open(f"/proc/{pid}/stat", "rb") as f:
            data = f.read().decode("utf-8", "replace")
        # comm may contain spaces/parens: split after the last ')'
        return data.rsplit(")", 1)[1].split()
    except (FileNotFoundError, ProcessLookupError, IndexError, ValueError):
        return None


def proc_starttime(pid: int) -> Optional[int]:
    """Kernel start time (clock ticks) of pid; None if gone or zombie."""
    rest = _stat_fields(pid)
    if rest is None or rest[0] == "Z":  # zombies are dead for our purposes
        return None
    try:
        return int(rest[19])
    except (IndexError, ValueError):
        return None


def metrics(pid: int) -> Optional[dict]:
    """Per-process resource metrics (reference parity: ctr TaskMetrics,
    ctr/client.go:123 — there from containerd's cgroup sampling; here from
    /proc since cells are host processes). None if the pid is gone."""
    rest = _stat_fields(pid)
    if rest is None or rest[0] == "Z":
        return None
    try:
        tck = os.sysconf("SC_CLK_TCK")
        page = os.sysconf("SC_PAGE_SIZE")
        # stat fields after comm/state: index 11/12 are utime/stime
        # (fields 14/15 1-based), 17 is num_threads (field 20)
        cpu_s = (int(rest[11]) + int(rest[12])) / tck
        threads = int(rest[17])
        with open(f"/proc/{pid}/statm") as f:
            rss_pages = int(f.read().split()[1])
        return {"pid": pid, "cpuSeconds": round(cpu_s, 3),
                "rssBytes": rss_pages * page, "threads": threads}
    except (OSError, ValueError, IndexError):
        return None


def alive(pid: int, starttime: Optional[int] = None) -> bool:
    if pid <= 0:
        return False
    st = proc_starttime(pid)
    if st is None:
        return False
    if starttime is not None and st != starttime:
        return False  # pid reused by another process
    return True


def terminate(pid: int, starttime: Optional[int], grace_seconds: float = 10.0,
              use_group: bool = True) -> None:
    """SIGTERM, wait up to grace, then SIGKILL (reference: 10s escalation)."""
    if not alive(pid, starttime):
        return
    _signal(pid, signal.SIGTERM, use_group)
    deadline = time.monotonic() + grace_seconds
    while time.monotonic() < deadline:
        if not alive(pid, starttime):
            return
        time.sleep(0.05)
    kill(pid, starttime, use_group)


def kill(pid: int, starttime: Optional[int], use_group: bool = True) -> None:
    if alive(pid, starttime):
        _signal(pid, signal.SIGKILL, use_group)
        for _ in range(100):
            if not alive(pid, starttime):
                return
            time.sleep(0.02)


def _signal(pid: int, sig: int, use_group: bool) -> None:
    with contextlib.suppress(ProcessLookupError, PermissionError):
        if use_group:
            try:
                os.killpg(pid, sig)
                return
            except (ProcessLookupError, PermissionError, OSError):
                pass
        os.kill(pid, sig)
