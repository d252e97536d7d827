"""Desired-vs-actual diff with change classification.

Reference contract (internal/controller/apply/diff.go): ChangeType in
{None, Additive, Compatible, Breaking}; Breaking changes force a cell
recreate; annotations and provenance are deliberately not compared.
"""
from __future__ import annotations

from enum import Enum
from typing import List, Tuple

from kukeon_amd.api import v1beta1 as api


class ChangeType(Enum):
    NONE = "None"
    ADDITIVE = "Additive"
    COMPATIBLE = "Compatible"
    BREAKING = "Breaking"


_SEVERITY = {ChangeType.NONE: 0, ChangeType.ADDITIVE: 1,
             ChangeType.COMPATIBLE: 2, ChangeType.BREAKING: 3}


class DiffResult:
    def __init__(self):
        self.changes: List[Tuple[str, ChangeType]] = []

    def add(self, path: str, ct: ChangeType):
        self.changes.append((path, ct))

    @property
    def change_type(self) -> ChangeType:
        worst = ChangeType.NONE
        for _, ct in self.changes:
            if _SEVERITY[ct] > _SEVERITY[worst]:
                worst = ct
        return worst

    @property
    def paths(self) -> List[str]:
        return [p for p, _ in self.changes]


# container fields whose change requires recreating the process
_BREAKING_CONTAINER_FIELDS = [
    "image", "command", "args", "working_dir", "env", "user", "privileged",
    "host_network", "host_pid", "devices", "gpus", "attachable", "volumes",
    "secrets", "repos", "git", "tty",
]
_COMPATIBLE_CONTAINER_FIELDS = [
    "restart_policy", "restart_backoff_seconds", "restart_max_retries",
    "resources",
]


def diff_cell(desired: api.CellDoc, actual: api.CellDoc) -> DiffResult:
    r = DiffResult()
    if desired.spec.auto_delete != actual.spec.auto_delete:
        r.add("spec.autoDelete", ChangeType.COMPATIBLE)
    des = {_cname(c): c for c in desired.spec.containers}
    act = {_cname(c): c for c in actual.spec.containers}
    for name, c in des.items():
        if name not in act:
            r.add(f"spec.containers[{name}]", ChangeType.ADDITIVE)
            continue
        a = act[name]
        for fname in _BREAKING_CONTAINER_FIELDS:
            if getattr(c, fname) != getattr(a, fname):
                r.add(f"spec.containers[{name}].{fname}", ChangeType.BREAKING)
        for fname in _COMPATIBLE_CONTAINER_FIELDS:
            if getattr(c, fname) != getattr(a, fname):
                r.add(f"spec.containers[{name}].{fname}",
                      ChangeType.COMPATIBLE)
    for name in act:
        if name not in des:
            r.add(f"spec.containers[{name}]", ChangeType.BREAKING)
    return r


def diff_space(desired: api.SpaceDoc, actual: api.SpaceDoc) -> DiffResult:
    r = DiffResult()
    dn = desired.spec.network.to_dict() if desired.spec.network else {}
    an = actual.spec.network.to_dict() if actual.spec.network else {}
    if dn != an:
        r.add("spec.network", ChangeType.COMPATIBLE)
    dd = desired.spec.defaults.to_dict() if desired.spec.defaults else {}
    ad = actual.spec.defaults.to_dict() if actual.spec.defaults else {}
    if dd != ad:
        r.add("spec.defaults", ChangeType.COMPATIBLE)
    return r


def _cname(c: api.ContainerSpec) -> str:
    return c.id or "main"
