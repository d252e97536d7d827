"""Version translation seam: external wire documents ↔ the internal
model (reference internal/apischeme, ~6.4k LoC of Normalize*/Convert*).

The internal model is the v1beta1 dataclass set (api/v1beta1.py); every
document entering the controller passes through `normalize()`, which

* detects the apiVersion (bare "v1beta1", the "kukeon.io/v1beta1" group
  form, and the legacy "v1alpha1" wire format),
* up-converts v1alpha1 documents field-by-field (the alpha wire used
  `realm`/`space`/`stack` scope keys, `autoRemove`, a flat single-
  container `spec.image/command/args` shorthand, and `egress.mode`),
* applies cross-version defaulting (scope fall-backs to the default
  hierarchy, container ids, restart policy normalization),

and `to_wire()` re-exports an internal doc at a requested version so old
clients keep working (v1alpha1 export is lossy where beta-only features
are present; the lost fields are reported).
"""
from __future__ import annotations

import copy
from typing import Any, Dict, List, Tuple

from kukeon_amd.api import errors
from kukeon_amd.api import v1beta1 as api

V1ALPHA1 = "v1alpha1"
V1BETA1 = api.API_VERSION
SUPPORTED = (V1ALPHA1, V1BETA1)

_DEFAULT_SCOPE = {"realmId": "default", "spaceId": "default",
                  "stackId": "default"}


def detect_version(raw: Dict[str, Any]) -> str:
    av = raw.get("apiVersion", V1BETA1)
    if "/" in av:  # group form kukeon.io/<version>
        av = av.rsplit("/", 1)[1]
    if av not in SUPPORTED:
        raise errors.ValidationError(f"unsupported apiVersion {av!r}")
    return av


def normalize(raw: Dict[str, Any]) -> Dict[str, Any]:
    """External document (any supported version) -> internal-shape dict
    (v1beta1). The input is not mutated."""
    version = detect_version(raw)
    doc = copy.deepcopy(raw)
    if version == V1ALPHA1:
        doc = _up_convert_alpha(doc)
    doc["apiVersion"] = V1BETA1
    _default(doc)
    return doc


def normalize_doc(raw: Dict[str, Any]):
    """normalize + typed conversion to the internal dataclass."""
    d = normalize(raw)
    cls = api.DOC_TYPES.get(d.get("kind", ""))
    if cls is None:
        raise errors.ValidationError(f"unknown kind {d.get('kind')!r}")
    return cls.from_dict(d)


# ---------------------------------------------------------------------------
# v1alpha1 -> v1beta1
# ---------------------------------------------------------------------------
_ALPHA_SCOPE = {"realm": "realmId", "space": "spaceId", "stack": "stackId"}


def _up_convert_alpha(doc: Dict[str, Any]) -> Dict[str, Any]:
    kind = doc.get("kind", "")
    spec = doc.setdefault("spec", {})
    for old, new in _ALPHA_SCOPE.items():
        if old in spec:
            spec[new] = spec.pop(old)
    if kind == api.KIND_CELL:
        if "autoRemove" in spec:
            spec["autoDelete"] = spec.pop("autoRemove")
        # alpha flat single-container shorthand:
        #   spec: {image, command, args, env} with no containers list
        if "containers" not in spec and ("image" in spec or
                                         "command" in spec):
            c = {"id": "main"}
            for k in ("image", "command", "args", "env", "workingDir",
                      "attachable", "tty", "gpus"):
                if k in spec:
                    c[k] = spec.pop(k)
            spec["containers"] = [c]
        for c in spec.get("containers", []):
            rp = c.get("restartPolicy")
            if rp == "onFailure":  # alpha camelCase enum
                c["restartPolicy"] = "on-failure"
    if kind == api.KIND_SPACE:
        net = spec.get("network")
        if net and "egress" in net and isinstance(net["egress"], dict):
            eg = net["egress"]
            if "mode" in eg:  # alpha called the default action "mode"
                eg["default"] = eg.pop("mode")
    return doc


# ---------------------------------------------------------------------------
# defaulting (shared by every inbound path)
# ---------------------------------------------------------------------------
def _default(doc: Dict[str, Any]) -> None:
    kind = doc.get("kind", "")
    spec = doc.setdefault("spec", {})
    scoped = {
        api.KIND_CELL: ("realmId", "spaceId", "stackId"),
        api.KIND_SESSION: ("realmId", "spaceId"),
        api.KIND_STACK: ("realmId", "spaceId"),
        api.KIND_SPACE: ("realmId",),
        api.KIND_SECRET: ("realmId", "spaceId"),
        api.KIND_VOLUME: ("realmId", "spaceId"),
        api.KIND_CELL_BLUEPRINT: ("realmId", "spaceId"),
        api.KIND_CELL_CONFIG: ("realmId", "spaceId"),
    }
    for key in scoped.get(kind, ()):
        spec.setdefault(key, _DEFAULT_SCOPE[key])
    if kind == api.KIND_CELL:
        for i, c in enumerate(spec.get("containers", [])):
            c.setdefault("id", "main" if i == 0 else f"c{i}")


# ---------------------------------------------------------------------------
# internal -> wire (downgrade support)
# ---------------------------------------------------------------------------
def to_wire(doc, version: str = V1BETA1) -> Tuple[Dict[str, Any],
                                                  List[str]]:
    """Export an internal doc at `version`. Returns (wire_doc,
    lost_fields): downgrading to v1alpha1 drops beta-only features and
    names them instead of silently losing data."""
    d = doc.to_dict() if hasattr(doc, "to_dict") else copy.deepcopy(doc)
    if version == V1BETA1:
        return d, []
    if version != V1ALPHA1:
        raise errors.ValidationError(f"unsupported export version "
                                     f"{version!r}")
    lost: List[str] = []
    d = copy.deepcopy(d)
    d["apiVersion"] = V1ALPHA1
    spec = d.get("spec", {})
    for new, old in (("realmId", "realm"), ("spaceId", "space"),
                     ("stackId", "stack")):
        if new in spec:
            spec[old] = spec.pop(new)
    if d.get("kind") == api.KIND_CELL:
        if "autoDelete" in spec:
            spec["autoRemove"] = spec.pop("autoDelete")
        for c in spec.get("containers", []):
            for beta_only in ("repos", "git", "secrets", "devices",
                              "resources"):
                if c.get(beta_only):
                    lost.append(f"containers[{c.get('id')}].{beta_only}")
                    c.pop(beta_only, None)
            if c.get("restartPolicy") == "on-failure":
                c["restartPolicy"] = "onFailure"
    if d.get("kind") == api.KIND_SPACE:
        net = spec.get("network") or {}
        eg = net.get("egress")
        if eg and "default" in eg:
            eg["mode"] = eg.pop("default")
    if d.get("kind") == api.KIND_SESSION:
        lost.append("kind Session (no v1alpha1 representation)")
    return d, lost
