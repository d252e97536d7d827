"""Deterministic naming + validation (reference: internal/util/naming)."""
from __future__ import annotations

import re
import secrets

from kukeon_amd.api import errors

NAME_RE = re.compile(r"^[a-z0-9]([a-z0-9-]{0,61}[a-z0-9])?$")

DEFAULT_REALM = "default"
DEFAULT_SPACE = "default"
DEFAULT_STACK = "default"
SYSTEM_REALM = "kuke-system"
SYSTEM_SPACE = "kukeon"
SYSTEM_STACK = "kukeon"


def validate_name(name: str, what: str = "name") -> str:
    if not NAME_RE.match(name or ""):
        raise errors.ValidationError(
            f"invalid {what} {name!r}: must match {NAME_RE.pattern}")
    return name


def root_container_id(space: str, stack: str, cell: str) -> str:
    return f"{space}-{stack}-{cell}"


def container_id(space: str, stack: str, cell: str, container: str) -> str:
    return f"{space}-{stack}-{cell}-{container}"


def generate_cell_name(prefix: str, taken) -> str:
    """<prefix>-<6hex> with collision retry (reference cellname.go)."""
    prefix = prefix or "cell"
    for _ in range(64):
        name = f"{prefix}-{secrets.token_hex(3)}"
        if name not in taken:
            return name
    raise errors.KukeonError("could not allocate a unique cell name")
