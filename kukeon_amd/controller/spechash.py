"""Spawn-spec hashing for idempotent container reuse.

`kuke start` on a cell with live containers must reuse a process whose
spawn spec is unchanged and recreate one whose spec drifted (reference:
runner/start.go:867+ spec-hash compare, runner/spec_hash.go). The hash
covers the full ContainerSpec except fields that do not feed the spawn:
scope ids (implied by the cell's path) and the restart-policy knobs (a
restart-knob change must not bounce a healthy process).
"""
from __future__ import annotations

import hashlib
import json

from kukeon_amd.api import v1beta1 as api

SPEC_HASH_LABEL = "kukeon.io/spec-hash"

_NON_SPAWN_FIELDS = (
    "realmId", "spaceId", "stackId", "cellId",
    "restartPolicy", "restartBackoffSeconds", "restartMaxRetries",
)


def spec_hash(c: api.ContainerSpec) -> str:
    payload = c.to_dict()
    for k in _NON_SPAWN_FIELDS:
        payload.pop(k, None)
    blob = json.dumps(payload, sort_keys=True).encode()
    return hashlib.sha256(blob).hexdigest()[:16]
