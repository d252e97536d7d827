"""Multi-document YAML parse + per-kind validation for `kuke apply`
(reference: internal/apply/parser)."""
from __future__ import annotations

from typing import Any, Dict, List

import yaml

from kukeon_amd.api import errors
from kukeon_amd.api import v1beta1 as api
from kukeon_amd.controller import naming


def parse_documents(text: str, validate: bool = True) -> List[Any]:
    """Split a multi-doc YAML stream into typed docs, kind-detected."""
    docs = []
    for raw in yaml.safe_load_all(text):
        if raw is None:
            continue
        if not isinstance(raw, dict):
            raise errors.ValidationError("document is not a mapping")
        docs.append(detect_and_convert(raw, validate=validate))
    return docs


def detect_and_convert(raw: Dict[str, Any], validate: bool = True):
    kind = raw.get("kind")
    if not kind:
        raise errors.ValidationError("document missing kind")
    if api.DOC_TYPES.get(kind) is None:
        raise errors.ValidationError(f"unknown kind {kind!r}")
    # version seam: detect + up-convert + default (api/scheme.py — the
    # apischeme analog); the controller only ever sees internal-shape
    # v1beta1 documents
    from kukeon_amd.api import scheme
    doc = scheme.normalize_doc(raw)
    if validate:
        validate_document(doc)
    return doc


def validate_document(doc) -> None:
    kind = doc.kind
    naming.validate_name(doc.metadata.name, f"{kind} name")
    if kind == api.KIND_CELL:
        _validate_cell(doc)
    elif kind == api.KIND_SPACE:
        if not doc.spec.realm_id:
            raise errors.ValidationError("Space.spec.realmId is required")
        if doc.spec.network and doc.spec.network.egress:
            eg = doc.spec.network.egress
            if eg.default not in ("allow", "deny"):
                raise errors.ValidationError(
                    f"egress default must be allow|deny, got {eg.default!r}")
            for r in eg.allow:
                if bool(r.host) == bool(r.cidr):
                    raise errors.ValidationError(
                        "egress allow rule needs exactly one of host|cidr")
    elif kind == api.KIND_STACK:
        for f in ("realm_id", "space_id"):
            if not getattr(doc.spec, f):
                raise errors.ValidationError(f"Stack.spec.{f} is required")
    elif kind == api.KIND_SESSION:
        if not doc.spec.stack_id:
            raise errors.ValidationError("Session.spec.stackId is required")
        if doc.spec.lifetime:
            for fld in ("wall_clock", "idle_timeout"):
                v = getattr(doc.spec.lifetime, fld)
                if v:
                    parse_duration(v)
    elif kind == api.KIND_CELL_CONFIG:
        if not doc.spec.blueprint:
            raise errors.ValidationError("CellConfig.spec.blueprint required")


def _validate_cell(doc: api.CellDoc) -> None:
    for f in ("realm_id", "space_id", "stack_id"):
        if not getattr(doc.spec, f):
            raise errors.ValidationError(f"Cell.spec.{f} is required")
    if not doc.spec.containers:
        raise errors.ValidationError("Cell.spec.containers must be non-empty")
    seen = set()
    attachables = 0
    for c in doc.spec.containers:
        name = c.id or "main"
        naming.validate_name(name, "container id")
        if name in seen:
            raise errors.ValidationError(f"duplicate container id {name!r}")
        seen.add(name)
        if c.attachable:
            attachables += 1
        if c.tty and not c.attachable:
            raise errors.ValidationError(
                f"container {name!r}: tty requires attachable: true")
        if c.restart_policy not in (api.RESTART_NEVER, api.RESTART_ALWAYS,
                                    api.RESTART_ON_FAILURE):
            raise errors.ValidationError(
                f"container {name!r}: invalid restartPolicy "
                f"{c.restart_policy!r}")
        if c.restart_backoff_seconds is not None:
            if c.restart_policy == api.RESTART_NEVER:
                raise errors.ValidationError(
                    f"container {name!r}: restartBackoffSeconds needs a "
                    "restarting policy")
            if c.restart_backoff_seconds < 0:
                raise errors.ValidationError(
                    f"container {name!r}: negative restartBackoffSeconds")
        if c.restart_max_retries is not None:
            if c.restart_policy != api.RESTART_ON_FAILURE:
                raise errors.ValidationError(
                    f"container {name!r}: restartMaxRetries only valid with "
                    "on-failure")
            if c.restart_max_retries < 1:
                raise errors.ValidationError(
                    f"container {name!r}: restartMaxRetries must be >= 1")
        if c.gpus < 0:
            raise errors.ValidationError(f"container {name!r}: negative gpus")
    if attachables > 1:
        raise errors.ValidationError("at most one attachable container")


_DUR_UNITS = {"s": 1, "m": 60, "h": 3600, "d": 86400}


def parse_duration(text: str) -> float:
    """'30m', '1h30m', '45s' -> seconds."""
    import re
    if not text:
        return 0.0
    m = re.fullmatch(r"(?:(\d+)h)?(?:(\d+)m)?(?:(\d+)s)?", text)
    if not m or not any(m.groups()):
        raise errors.ValidationError(f"invalid duration {text!r}")
    h, mi, s = (int(g) if g else 0 for g in m.groups())
    return h * 3600 + mi * 60 + s
