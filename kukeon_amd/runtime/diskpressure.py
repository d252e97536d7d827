"""Data-volume disk-pressure guard (reference: internal/util/diskpressure).

Samples the filesystem backing the run path's data tree; the controller
warns above warn_percent (rate-limited to one WARN per 5 minutes, the
reference's re-emit interval) and blocks new cell creation above
block_percent unless the request sets ignore_disk_pressure.
"""
from __future__ import annotations

import logging
import os
import time
from dataclasses import dataclass

log = logging.getLogger("kukeon.diskpressure")

WARN_REEMIT_SECONDS = 300.0


@dataclass
class Sample:
    total_bytes: int
    used_bytes: int

    @property
    def used_percent(self) -> float:
        if self.total_bytes <= 0:
            return 0.0
        return 100.0 * self.used_bytes / self.total_bytes


def sample(path: str) -> Sample:
    st = os.statvfs(path)
    total = st.f_blocks * st.f_frsize
    free = st.f_bavail * st.f_frsize
    return Sample(total_bytes=total, used_bytes=total - free)


class Guard:
    def __init__(self, path: str, warn_percent: float = 85.0,
                 block_percent: float = 95.0, sampler=sample,
                 now_fn=time.monotonic):
        self.path = path
        self.warn_percent = warn_percent
        self.block_percent = block_percent
        self.sampler = sampler
        self.now = now_fn
        self._last_warn = 0.0

    def check(self, ignore: bool = False) -> None:
        """Raises DiskPressure above block_percent (unless ignored);
        warns (rate-limited) above warn_percent."""
        from kukeon_amd.api import errors
        try:
            s = self.sampler(self.path)
        except OSError:
            return
        pct = s.used_percent
        if pct >= self.block_percent and not ignore:
            raise errors.DiskPressure(
                f"data volume at {pct:.1f}% (block threshold "
                f"{self.block_percent:.0f}%); pass ignoreDiskPressure to "
                "override")
        if pct >= self.warn_percent:
            now = self.now()
            if now - self._last_warn >= WARN_REEMIT_SECONDS:
                self._last_warn = now
                log.warning("data volume at %.1f%% (warn threshold %.0f%%)",
                            pct, self.warn_percent)
