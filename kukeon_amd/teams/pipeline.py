"""The `kuke team init` pipeline (reference cmd/kuke/team/init.go flow):

load kuketeam.yaml -> resolve the agents source -> render roles x harnesses
-> compose secrets -> register catalog images -> apply with per-team prune.
"""
from __future__ import annotations

import contextlib
import os
import subprocess
from pathlib import Path
from string import Template
from typing import Dict, List, Optional, Tuple

import yaml

from kukeon_amd.api import errors
from kukeon_amd.api import v1beta1 as api
from kukeon_amd.teams import (Harness, ImageCatalog, ProjectTeam, Role,
                              parse_team_file)


# ---------------------------------------------------------------------------
# teamsource: materialize the agents repo (git clone/fetch, or a local path)
# ---------------------------------------------------------------------------
def resolve_source(team: ProjectTeam, cache_root: Path) -> Path:
    src = team.source
    if src.path:
        p = Path(os.path.expanduser(src.path))
        if not p.is_dir():
            raise errors.ValidationError(f"agents source path {p} not found")
        return p
    if not src.repo:
        raise errors.ValidationError("team source needs repo: or path:")
    dest = cache_root / team.metadata.name
    url = src.repo
    if not url.startswith(("http://", "https://", "git@", "ssh://", "/")):
        url = f"git@{url.split('/', 1)[0]}:{url.split('/', 1)[1]}.git" \
            if "/" in url else url
    if not (dest / ".git").is_dir():
        dest.parent.mkdir(parents=True, exist_ok=True)
        _git(["clone", url, str(dest)])
    elif not src.pinned:
        _git(["-C", str(dest), "fetch", "--all", "--prune"])
    ref = src.commit or src.tag or (f"origin/{src.branch}" if src.branch
                                    else "origin/HEAD")
    _git(["-C", str(dest), "checkout", "--detach", ref])
    return dest


def _git(args: List[str]) -> None:
    proc = subprocess.run(["git"] + args, capture_output=True, text=True)
    if proc.returncode != 0:
        raise errors.KukeonError(f"git {' '.join(args)}: {proc.stderr[-400:]}")


# ---------------------------------------------------------------------------
# teamhost: ~/.kuke/teams layout + harness seed files (never overwrite)
# ---------------------------------------------------------------------------
def provision_host(team: ProjectTeam, harnesses: Dict[str, Harness],
                   teams_root: Path) -> Path:
    team_root = teams_root / team.metadata.name
    team_root.mkdir(parents=True, exist_ok=True)
    for name in ("state", "workspaces"):
        (team_root / name).mkdir(exist_ok=True)
    # host-wide + per-team secrets scaffolds (0600)
    for p in (teams_root / "secrets.env", team_root / "secrets.env"):
        if not p.exists():
            p.touch()
            p.chmod(0o600)
    for h in harnesses.values():
        for seed in h.seeds:
            path = Template(seed.path).safe_substitute(
                TEAM_ROOT=str(team_root), HARNESS=h.metadata.name)
            sp = Path(path)
            if not sp.is_absolute():
                sp = team_root / sp
            if sp.exists():
                continue  # hand-edited files are never overwritten
            sp.parent.mkdir(parents=True, exist_ok=True)
            sp.write_text(seed.content)
            sp.chmod(seed.mode or 0o644)
    return team_root


# ---------------------------------------------------------------------------
# teamsecrets: layered secrets.env -> Secret docs
# ---------------------------------------------------------------------------
def compose_secrets(team: ProjectTeam, teams_root: Path, realm: str,
                    space: str) -> List[api.SecretDoc]:
    data: Dict[str, str] = {}
    for p in (teams_root / "secrets.env",
              teams_root / team.metadata.name / "secrets.env"):
        if not p.exists():
            continue
        for line in p.read_text().splitlines():
            line = line.strip()
            if not line or line.startswith("#") or "=" not in line:
                continue
            k, _, v = line.partition("=")
            data[k.strip()] = v.strip()
    if not data:
        return []
    doc = api.SecretDoc(
        metadata=api.Metadata(name=f"team-{team.metadata.name}",
                              labels={api.LABEL_TEAM: team.metadata.name}),
        spec=api.SecretSpec(realm_id=realm, space_id=space, data=data))
    return [doc]


# ---------------------------------------------------------------------------
# teamrender: needs-merge -> image-select -> template render
# ---------------------------------------------------------------------------
def select_image(catalog: Optional[ImageCatalog], harness: str,
                 needs: List[str]) -> str:
    if catalog is None:
        return ""
    for entry in catalog.images:
        if entry.harness != harness:
            continue
        if all(cap in entry.capabilities for cap in needs):
            return entry.image or f"kukeon.internal/{entry.ref}"
    if needs:
        raise errors.ValidationError(
            f"no catalog image for harness {harness!r} with capabilities "
            f"{needs}")
    return ""


def render_team(team: ProjectTeam, roles: Dict[str, Role],
                harnesses: Dict[str, Harness],
                catalog: Optional[ImageCatalog], source_dir: Path,
                team_root: Path, realm: str, space: str) -> List[dict]:
    """-> list of v1beta1 doc dicts (CellBlueprint + CellConfig per
    role x harness), ready for apply_documents."""
    out: List[dict] = []
    for rr in team.roles:
        role = roles.get(rr.ref)
        if role is None:
            raise errors.ValidationError(f"role {rr.ref!r} not found in the "
                                         "agents source")
        hlist = rr.harnesses or team.default_harnesses
        if not hlist:
            raise errors.ValidationError(
                f"role {rr.ref!r}: no harnesses (set defaults.harnesses)")
        for hname in hlist:
            h = harnesses.get(hname)
            if h is None:
                raise errors.ValidationError(f"harness {hname!r} not found")
            # needs-merge (reference teamrender): team defaults union
            # role requirements before image selection
            from kukeon_amd.teams.template import TemplateEngine, merge_needs
            needs = merge_needs(team.default_needs.image, rr.needs.image)
            image = select_image(catalog, hname, needs)
            tmpl_path = source_dir / h.template
            if not tmpl_path.exists():
                raise errors.ValidationError(
                    f"harness {hname}: template {tmpl_path} missing")
            ctx = {
                "TEAM": team.metadata.name,
                "ROLE": role.metadata.name,
                "HARNESS": hname,
                "IMAGE": image or h.base_image,
                "TEAM_ROOT": str(team_root),
                "SKILL_PATH": h.skill_path,
                "ROLE_PROMPT": role.prompt,
                "NEEDS": needs,
                "ROLE_ENV": list(role.env),
            }
            ctx.update(team.vars)
            ctx.update(rr.vars)
            engine = TemplateEngine(partials_dir=source_dir / "partials")
            rendered = engine.render(tmpl_path.read_text(), ctx)
            for raw in yaml.safe_load_all(rendered):
                if not raw:
                    continue
                raw.setdefault("apiVersion", api.API_VERSION)
                md = raw.setdefault("metadata", {})
                md.setdefault("name", f"{team.metadata.name}-{rr.ref}-{hname}")
                md.setdefault("labels", {})[api.LABEL_TEAM] = \
                    team.metadata.name
                spec = raw.setdefault("spec", {})
                spec.setdefault("realmId", realm)
                spec.setdefault("spaceId", space)
                out.append(raw)
    return out


# ---------------------------------------------------------------------------
# teambuild: topo-ordered (FROM-graph) image registration
# ---------------------------------------------------------------------------
def build_order(catalog: ImageCatalog) -> List[str]:
    done: List[str] = []
    seen = set()

    def visit(ref: str, path: Tuple[str, ...] = ()):
        if ref in seen:
            return
        if ref in path:
            raise errors.ValidationError(f"image FROM-cycle at {ref!r}")
        entry = next((e for e in catalog.images if e.ref == ref), None)
        if entry is None:
            return
        if entry.base:
            visit(entry.base, path + (ref,))
        seen.add(ref)
        done.append(ref)

    for e in catalog.images:
        visit(e.ref)
    return done


# ---------------------------------------------------------------------------
# the pipeline
# ---------------------------------------------------------------------------
def team_init(controller, team_file: str, realm: str = "default",
              space: str = "default", teams_root: Optional[str] = None,
              build_images: bool = True) -> Dict[str, object]:
    teams_root_p = Path(teams_root or
                        os.path.expanduser("~/.kuke/teams"))
    docs = parse_team_file(Path(team_file).read_text())
    team = next((d for d in docs if isinstance(d, ProjectTeam)), None)
    if team is None:
        raise errors.ValidationError(f"{team_file} has no ProjectTeam doc")
    source_dir = resolve_source(team, teams_root_p / ".sources")

    # load roles/harnesses/catalog from the agents source tree
    roles: Dict[str, Role] = {}
    harnesses: Dict[str, Harness] = {}
    catalog: Optional[ImageCatalog] = None
    for sub in ("roles", "harnesses", "."):
        d = source_dir / sub
        if not d.is_dir():
            continue
        for yml in sorted(list(d.glob("*.yaml")) + list(d.glob("*.yml"))):
            if ".tmpl." in yml.name:
                continue  # blueprint templates are rendered, never parsed raw
            with contextlib.suppress(errors.ValidationError, yaml.YAMLError):
                for doc in parse_team_file(yml.read_text()):
                    if isinstance(doc, Role):
                        roles[doc.metadata.name] = doc
                    elif isinstance(doc, Harness):
                        harnesses[doc.metadata.name] = doc
                    elif isinstance(doc, ImageCatalog):
                        catalog = doc
    team_root = provision_host(team, harnesses, teams_root_p)

    built: List[str] = []
    if build_images and catalog:
        from kukeon_amd.images import (Builder, BuildError, ImageStore,
                                       overlay_supported)
        can_build = overlay_supported()
        store = ImageStore(str(controller.run_path)) if can_build else None
        for ref in build_order(catalog):
            entry = next(e for e in catalog.images if e.ref == ref)
            tag = f"kukeon.internal/{ref}"
            kf = (source_dir / entry.build_dockerfile
                  if entry.build_dockerfile else None)
            ctx = (source_dir / entry.build_context
                   if entry.build_context else source_dir)
            if can_build and kf is not None and kf.is_file():
                # real layered build (kukebuild analog): catalog entries
                # with a Kukefile become rootfs images cells can pivot
                # into; FROM kukeon.internal/<base> layers on the parent
                # built earlier in the topo order. The built manifest IS
                # the registration (same catalog path) — registering on
                # top would clobber its layer list.
                try:
                    man = Builder(store).build(ctx, kf.read_text(), tag)
                except BuildError as e:
                    raise errors.ValidationError(
                        f"catalog image {ref!r} build failed: {e}")
                man["labels"].update({"harness": entry.harness,
                                      "capabilities": entry.capabilities,
                                      "team-source": str(source_dir)})
                store.put_manifest(tag, man["layers"], man["config"],
                                   man["labels"])
            else:
                controller.register_image(
                    name=tag,
                    spec={"harness": entry.harness,
                          "capabilities": entry.capabilities,
                          "buildContext": entry.build_context,
                          "dockerfile": entry.build_dockerfile,
                          "source": str(source_dir)})
            built.append(ref)

    rendered = render_team(team, roles, harnesses, catalog, source_dir,
                           team_root, realm, space)
    secret_docs = compose_secrets(team, teams_root_p, realm, space)
    applied = []
    for sd in secret_docs:
        controller.put_secret(sd)
        applied.append(("Secret", sd.metadata.name, "updated"))
    results = controller.apply_documents(
        yaml.safe_dump_all(rendered), team=team.metadata.name)
    applied += [(r.kind, r.name, r.action) for r in results]
    pruned = controller.prune_team(
        team.metadata.name,
        keep={(r["kind"], r["metadata"]["name"]) for r in rendered} |
        {("Secret", s.metadata.name) for s in secret_docs})
    return {"team": team.metadata.name, "source": str(source_dir),
            "applied": applied, "built": built, "pruned": pruned}
