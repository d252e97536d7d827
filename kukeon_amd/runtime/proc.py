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
