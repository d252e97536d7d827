"""amdgpu device discovery + per-cell GPU pinning.

Each MI355X shows up as /dev/dri/renderD<128+i> plus the shared /dev/kfd
compute node. Pinning a cell to GPUs = (a) ROCR_VISIBLE_DEVICES env so the
ROCm runtime only enumerates the granted devices, (b) device-cgroup allow
rules for /dev/kfd + the granted renderD nodes (CgroupManager.allow_devices).

The GPUAllocator is the daemon-side scheduler state: Sessions reserve whole
GPUs; allocations persist in <run_path>/gpus.json so a daemon restart
re-loads the assignment.
"""
from __future__ import annotations

import glob
import json
import os
import re
from pathlib import Path
from typing import Dict, List, Optional

from kukeon_amd.api import errors


def discover_gpus() -> List[int]:
    """Render-node indices of the amdgpu devices on this host."""
    out = []
    for p in sorted(glob.glob("/dev/dri/renderD*")):
        m = re.search(r"renderD(\d+)$", p)
        if m:
            out.append(int(m.group(1)) - 128)
    return out


def device_paths_for(gpu_ids: List[int]) -> List[str]:
    if not gpu_ids:
        return []
    return ["/dev/kfd"] + [f"/dev/dri/renderD{128 + i}" for i in gpu_ids]


def visible_devices_env(gpu_ids: List[int]) -> List[str]:
    ids = ",".join(str(i) for i in gpu_ids)
    return [f"ROCR_VISIBLE_DEVICES={ids}", f"HIP_VISIBLE_DEVICES={ids}"]


class GPUAllocator:
    """Whole-GPU reservations keyed by owner (session/cell id), persisted."""

    def __init__(self, state_path: str, devices: Optional[List[int]] = None):
        self.path = Path(state_path)
        self.devices = devices if devices is not None else discover_gpus()
        self.assignments: Dict[str, List[int]] = {}
        self._load()

    def _load(self) -> None:
        try:
            data = json.loads(self.path.read_text())
            self.assignments = {k: list(v) for k, v in
                                data.get("assignments", {}).items()}
        except (OSError, ValueError):
            self.assignments = {}

    def _save(self) -> None:
        self.path.parent.mkdir(parents=True, exist_ok=True)
        tmp = self.path.with_suffix(".tmp")
        tmp.write_text(json.dumps({"devices": self.devices,
                                   "assignments": self.assignments}))
        os.replace(tmp, self.path)

    @property
    def free(self) -> List[int]:
        used = {g for v in self.assignments.values() for g in v}
        return [d for d in self.devices if d not in used]

    def allocate(self, owner: str, count: int) -> List[int]:
        have = self.assignments.get(owner, [])
        if have and len(have) >= count:
            return have
        # top up an existing (shorter) assignment after a spec change
        # requesting more GPUs — never return a stale short list, which
        # would make start_cell silently pin fewer GPUs than spec
        free = self.free
        need = count - len(have)
        if need > len(free):
            raise errors.GPUUnavailable(
                f"want {count} GPUs ({need} more), {len(free)} free of "
                f"{len(self.devices)}")
        got = have + free[:need]
        self.assignments[owner] = got
        self._save()
        return got

    def release(self, owner: str) -> None:
        if owner in self.assignments:
            del self.assignments[owner]
            self._save()
