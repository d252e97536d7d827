"""Team distribution: the `kuke team init` subsystem (reference §2.7).

kuketeams.io/v1 document model + parser/validator (kuketeams analog).
Pipeline (pipeline.py): load kuketeam.yaml -> resolve the agents source
(source.py) -> render roles x harnesses into CellBlueprints/CellConfigs
(render.py) -> compose layered secrets (secrets.py) -> register catalog
images -> apply with per-team prune.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List

import yaml

from kukeon_amd.api import errors

API_VERSION = "kuketeams.io/v1"


@dataclass
class Metadata:
    name: str = ""


@dataclass
class RoleNeeds:
    image: List[str] = field(default_factory=list)


@dataclass
class RoleRef:
    ref: str = ""
    needs: RoleNeeds = field(default_factory=RoleNeeds)
    harnesses: List[str] = field(default_factory=list)
    # per-role template variables (merged over team vars at render time)
    vars: dict = field(default_factory=dict)


@dataclass
class TeamSource:
    repo: str = ""
    branch: str = ""
    tag: str = ""
    commit: str = ""
    path: str = ""       # local checkout (offline environments)

    @property
    def pinned(self) -> bool:
        return bool(self.tag or self.commit)


@dataclass
class ProjectTeam:
    metadata: Metadata = field(default_factory=Metadata)
    source: TeamSource = field(default_factory=TeamSource)
    default_harnesses: List[str] = field(default_factory=list)
    default_needs: RoleNeeds = field(default_factory=RoleNeeds)
    roles: List[RoleRef] = field(default_factory=list)
    vars: dict = field(default_factory=dict)


@dataclass
class Role:
    metadata: Metadata = field(default_factory=Metadata)
    description: str = ""
    prompt: str = ""
    env: List[str] = field(default_factory=list)


@dataclass
class HarnessSeed:
    path: str = ""
    mode: int = 0o644
    content: str = ""


@dataclass
class Harness:
    metadata: Metadata = field(default_factory=Metadata)
    base_image: str = ""
    skill_path: str = ""
    template: str = ""
    seeds: List[HarnessSeed] = field(default_factory=list)


@dataclass
class CatalogImage:
    ref: str = ""
    harness: str = ""
    image: str = ""
    capabilities: List[str] = field(default_factory=list)
    build_context: str = ""
    build_dockerfile: str = ""
    base: str = ""           # FROM-graph parent ref (build ordering)


@dataclass
class ImageCatalog:
    images: List[CatalogImage] = field(default_factory=list)


def parse_team_doc(raw: Dict[str, Any]):
    av = raw.get("apiVersion")
    if av != API_VERSION:
        raise errors.ValidationError(
            f"team doc apiVersion must be {API_VERSION}, got {av!r}")
    kind = raw.get("kind")
    md = Metadata(name=(raw.get("metadata") or {}).get("name", ""))
    spec = raw.get("spec") or {}
    if kind == "ProjectTeam":
        src = spec.get("source") or {}
        roles = []
        for r in spec.get("roles", []):
            needs = r.get("needs") or {}
            roles.append(RoleRef(ref=r.get("ref", ""),
                                 needs=RoleNeeds(
                                     image=list(needs.get("image", []))),
                                 harnesses=list(r.get("harnesses", [])),
                                 vars=dict(r.get("vars", {}))))
        defaults = spec.get("defaults") or {}
        dneeds = defaults.get("needs") or {}
        doc = ProjectTeam(
            metadata=md,
            source=TeamSource(repo=src.get("repo", ""),
                              branch=src.get("branch", ""),
                              tag=src.get("tag", ""),
                              commit=src.get("commit", ""),
                              path=src.get("path", "")),
            default_harnesses=list(defaults.get("harnesses", [])),
            default_needs=RoleNeeds(image=list(dneeds.get("image", []))),
            roles=roles,
            vars=dict(spec.get("vars", {})))
        if not doc.metadata.name:
            raise errors.ValidationError("ProjectTeam needs metadata.name")
        if not doc.roles:
            raise errors.ValidationError("ProjectTeam needs at least one role")
        return doc
    if kind == "Role":
        return Role(metadata=md, description=spec.get("description", ""),
                    prompt=spec.get("prompt", ""),
                    env=list(spec.get("env", [])))
    if kind == "Harness":
        seeds = [HarnessSeed(path=s.get("path", ""),
                             mode=s.get("mode", 0o644),
                             content=s.get("content", ""))
                 for s in spec.get("seeds", [])]
        h = Harness(metadata=md, base_image=spec.get("baseImage", ""),
                    skill_path=spec.get("skillPath", ""),
                    template=spec.get("template", ""), seeds=seeds)
        if not h.template:
            raise errors.ValidationError(
                f"Harness {md.name}: spec.template is required")
        return h
    if kind == "ImageCatalog":
        imgs = []
        for e in spec.get("images", []):
            b = e.get("build") or {}
            imgs.append(CatalogImage(
                ref=e.get("ref", ""), harness=e.get("harness", ""),
                image=e.get("image", ""),
                capabilities=list(e.get("capabilities", [])),
                build_context=b.get("context", ""),
                build_dockerfile=b.get("dockerfile", ""),
                base=e.get("base", "")))
        return ImageCatalog(images=imgs)
    raise errors.ValidationError(f"unknown kuketeams kind {kind!r}")


def parse_team_file(text: str) -> List[Any]:
    docs = []
    for raw in yaml.safe_load_all(text):
        if raw:
            docs.append(parse_team_doc(raw))
    return docs
