"""CellBlueprint / CellConfig materialization.

`kuke run -b <blueprint>`: ${KEY} scalar substitution into the blueprint's
cell template with resolution order cliParams > config values > param
default > environment (reference: internal/cellblueprint + cellconfig);
stamps lineage (kukeon.io/blueprint label + spec.provenance) so the
reconciler's OutOfSync pass can re-materialize and diff.
"""
from __future__ import annotations

import os
import re
from typing import Dict, List, Optional

from kukeon_amd.api import errors
from kukeon_amd.api import v1beta1 as api

_VAR_RE = re.compile(r"\$\{([A-Za-z_][A-Za-z0-9_]*)\}")


def resolve_params(bp: api.CellBlueprintDoc,
                   cli_params: Dict[str, str],
                   config_values: Optional[Dict[str, str]] = None,
                   env: Optional[Dict[str, str]] = None) -> Dict[str, str]:
    env = env if env is not None else dict(os.environ)
    out: Dict[str, str] = {}
    declared = {p.name for p in bp.spec.params}
    for p in bp.spec.params:
        if p.name in cli_params:
            out[p.name] = cli_params[p.name]
        elif config_values and p.name in config_values:
            out[p.name] = config_values[p.name]
        elif p.default:
            out[p.name] = p.default
        elif p.name in env:
            out[p.name] = env[p.name]
        elif p.required:
            raise errors.ValidationError(
                f"blueprint param {p.name!r} is required and unset")
        else:
            out[p.name] = ""
    for k in cli_params:
        if k not in declared:
            raise errors.ValidationError(f"unknown blueprint param {k!r}")
    return out


def _subst(value, params: Dict[str, str]):
    if isinstance(value, str):
        return _VAR_RE.sub(lambda m: params.get(m.group(1), m.group(0)), value)
    if isinstance(value, list):
        return [_subst(v, params) for v in value]
    if isinstance(value, dict):
        return {k: _subst(v, params) for k, v in value.items()}
    return value


def materialize(bp: api.CellBlueprintDoc, name: str,
                cli_params: Dict[str, str],
                binding_kind: str = api.KIND_CELL_BLUEPRINT,
                binding_ref: str = "",
                config_values: Optional[Dict[str, str]] = None,
                env_overlay: Optional[List[str]] = None) -> api.CellDoc:
    params = resolve_params(bp, cli_params, config_values)
    tmpl = _subst(bp.spec.template, params)
    tmpl.setdefault("apiVersion", api.API_VERSION)
    tmpl.setdefault("kind", api.KIND_CELL)
    cell = api.CellDoc.from_dict(tmpl)
    cell.metadata.name = name
    cell.metadata.labels.setdefault(api.LABEL_BLUEPRINT, bp.metadata.name)
    cell.spec.provenance = api.CellProvenance(
        binding_kind=("config" if binding_kind == api.KIND_CELL_CONFIG
                      else "blueprint"),
        binding_ref=binding_ref or bp.metadata.name,
        params=dict(params),
        env=list(env_overlay or []),
    )
    if env_overlay:
        for c in cell.spec.containers:
            if c.attachable or len(cell.spec.containers) == 1:
                merged = {kv.split("=", 1)[0]: kv for kv in c.env}
                for kv in env_overlay:
                    merged[kv.split("=", 1)[0]] = kv
                c.env = list(merged.values())
    return cell


def materialize_from_config(cfg: api.CellConfigDoc,
                            bp: api.CellBlueprintDoc, name: str,
                            cli_params: Dict[str, str]) -> api.CellDoc:
    return materialize(bp, name, cli_params,
                       binding_kind=api.KIND_CELL_CONFIG,
                       binding_ref=cfg.metadata.name,
                       config_values=dict(cfg.spec.values),
                       env_overlay=list(cfg.spec.env))
