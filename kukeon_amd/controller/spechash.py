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


def spec_hash(c: api.ContainerSpec, image_layers=None) -> str:
    payload = c.to_dict()
    for k in _NON_SPAWN_FIELDS:
        payload.pop(k, None)
    if image_layers:
        # image CONTENT drift (reference ctr chainID drift detection,
        # client.go:142-161): a rebuilt tag changes its layer ids, so a
        # live container on the old rootfs respawns at the next start
        payload["_imageLayers"] = list(image_layers)
    blob = json.dumps(payload, sort_keys=True).encode()
    return hashlib.sha256(blob).hexdigest()[:16]
