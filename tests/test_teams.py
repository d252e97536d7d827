"""Team distribution pipeline tests: local agents source -> render -> apply
with per-team prune, seeds, layered secrets, image registration."""

import pytest

from kukeon_amd.api import errors
from kukeon_amd.api import v1beta1 as api
from kukeon_amd.controller.core import Controller
from kukeon_amd.runtime.process import FakeRuntime
from kukeon_amd.teams import parse_team_file
from kukeon_amd.teams.pipeline import build_order, team_init


@pytest.fixture
def agents_src(tmp_path):
    src = tmp_path / "agents"
    (src / "roles").mkdir(parents=True)
    (src / "harnesses").mkdir()
    (src / "roles" / "dev.yaml").write_text("""
apiVersion: kuketeams.io/v1
kind: Role
metadata: {name: dev}
spec:
  description: developer agent
  prompt: "You are the dev."
""")
    (src / "roles" / "pm.yaml").write_text("""
apiVersion: kuketeams.io/v1
kind: Role
metadata: {name: pm}
spec: {prompt: "You are the PM."}
""")
    (src / "harnesses" / "claude.yaml").write_text("""
apiVersion: kuketeams.io/v1
kind: Harness
metadata: {name: claude}
spec:
  skillPath: /skills
  makeTarget: claude-image
  template: harnesses/claude.tmpl.yaml
  seeds:
    - path: state/${HARNESS}/settings.json
      mode: 384
      content: '{"seeded": true}'
""")
    (src / "harnesses" / "claude.tmpl.yaml").write_text("""
kind: CellBlueprint
metadata: {name: ${TEAM}-${ROLE}-${HARNESS}}
spec:
  namePrefix: ${ROLE}
  template:
    spec:
      realmId: default
      spaceId: default
      stackId: default
      containers:
        - id: agent
          image: "${IMAGE}"
          command: /bin/sh
          args: ["-c", "echo ${ROLE_PROMPT}"]
""")
    (src / "catalog.yaml").write_text("""
apiVersion: kuketeams.io/v1
kind: ImageCatalog
spec:
  images:
    - ref: base
      harness: claude
      image: kukeon.internal/base
      capabilities: [root]
      build: {context: images/base, dockerfile: Dockerfile}
    - ref: go
      harness: claude
      base: base
      image: kukeon.internal/go
      capabilities: [root, go]
      build: {context: images/go, dockerfile: Dockerfile}
""")
    return src


def write_team_file(tmp_path, src, roles="[{ref: dev, needs: {image: [go]}}, {ref: pm}]"):
    tf = tmp_path / "kuketeam.yaml"
    tf.write_text(f"""
apiVersion: kuketeams.io/v1
kind: ProjectTeam
metadata: {{name: myteam}}
spec:
  source: {{path: {src}}}
  defaults: {{harnesses: [claude]}}
  roles: {roles}
""")
    return tf


def test_team_init_end_to_end(tmp_path, agents_src):
    ctl = Controller(str(tmp_path / "run"), runtime=FakeRuntime(),
                     gpu_devices=[])
    ctl.bootstrap()
    teams_root = tmp_path / "teams"
    (teams_root).mkdir()
    (teams_root / "secrets.env").write_text("GLOBAL_KEY=g1\nSHARED=host\n")
    (teams_root / "myteam").mkdir()
    (teams_root / "myteam" / "secrets.env").write_text("SHARED=team\n")
    tf = write_team_file(tmp_path, agents_src)

    res = team_init(ctl, str(tf), teams_root=str(teams_root))
    assert res["team"] == "myteam"
    # blueprints rendered per role x harness
    bps = ctl.list_blueprints("default", "default")
    assert sorted(bps) == ["myteam-dev-claude", "myteam-pm-claude"]
    bp = ctl.get_blueprint("default", "default", "myteam-dev-claude")
    assert bp.metadata.labels[api.LABEL_TEAM] == "myteam"
    # image capability selection: dev needs [go] -> kukeon.internal/go
    img = bp.spec.template["spec"]["containers"][0]["image"]
    assert img == "kukeon.internal/go"
    # catalog images registered in FROM order
    assert res["built"] == ["base", "go"]
    assert ctl.get_image("kukeon.internal/go")["spec"]["harness"] == "claude"
    # layered secrets: per-team overrides host-wide
    sec = ctl.get_secret("default", "default", "team-myteam")
    assert sec.spec.data == {"GLOBAL_KEY": "g1", "SHARED": "team"}
    # harness seeds written once, not overwritten
    seeded = teams_root / "myteam" / "state" / "claude" / "settings.json"
    assert seeded.read_text() == '{"seeded": true}'
    seeded.write_text("edited")
    team_init(ctl, str(tf), teams_root=str(teams_root))
    assert seeded.read_text() == "edited"


def test_team_prune_removes_dropped_roles(tmp_path, agents_src):
    ctl = Controller(str(tmp_path / "run"), runtime=FakeRuntime(),
                     gpu_devices=[])
    ctl.bootstrap()
    teams_root = tmp_path / "teams"
    tf = write_team_file(tmp_path, agents_src)
    team_init(ctl, str(tf), teams_root=str(teams_root))
    assert len(ctl.list_blueprints("default", "default")) == 2
    # drop the pm role -> its blueprint is pruned on re-init
    tf2 = write_team_file(tmp_path, agents_src,
                          roles="[{ref: dev, needs: {image: [go]}}]")
    res = team_init(ctl, str(tf2), teams_root=str(teams_root))
    assert "CellBlueprint/myteam-pm-claude" in res["pruned"]
    assert ctl.list_blueprints("default", "default") == ["myteam-dev-claude"]


def test_capability_mismatch_fails(tmp_path, agents_src):
    ctl = Controller(str(tmp_path / "run"), runtime=FakeRuntime(),
                     gpu_devices=[])
    ctl.bootstrap()
    tf = write_team_file(tmp_path, agents_src,
                         roles="[{ref: dev, needs: {image: [cuda]}}]")
    with pytest.raises(errors.ValidationError):
        team_init(ctl, str(tf), teams_root=str(tmp_path / "teams"))


def test_build_order_cycle_detection():
    from kukeon_amd.teams import CatalogImage, ImageCatalog
    cat = ImageCatalog(images=[CatalogImage(ref="a", base="b"),
                               CatalogImage(ref="b", base="a")])
    with pytest.raises(errors.ValidationError):
        build_order(cat)


def test_parse_rejects_wrong_apiversion():
    with pytest.raises(errors.ValidationError):
        parse_team_file("apiVersion: v1beta1\nkind: Role\n"
                        "metadata: {name: x}\n")


def test_template_engine_partials_conditionals_each(tmp_path):
    """teamrender engine: partials, {{#if}}/{{else}}, {{#each}} and the
    ${VAR} layer compose (reference internal/teamrender partials
    pipeline)."""
    from kukeon_amd.teams.template import TemplateEngine, merge_needs
    from kukeon_amd.api import errors as kerrors

    pd = tmp_path / "partials"
    pd.mkdir()
    (pd / "resources.tmpl").write_text(
        "{{#if GPU}}gpus: ${GPU}{{else}}gpus: 0{{/if}}\n")
    (pd / "env.tmpl").write_text(
        "env:\n{{#each ROLE_ENV}}  - ${ITEM}\n{{/each}}")
    eng = TemplateEngine(partials_dir=pd)
    text = ("name: ${ROLE}\n"
            "{{> resources}}"
            "{{> env}}")
    out = eng.render(text, {"ROLE": "builder", "GPU": "2",
                            "ROLE_ENV": ["A=1", "B=2"]})
    assert "name: builder" in out
    assert "gpus: 2" in out
    assert "  - A=1" in out and "  - B=2" in out
    # falsy branch
    out = eng.render("{{#if GPU}}y{{else}}n{{/if}}", {"GPU": ""})
    assert out == "n"
    # nested if inside each
    out = eng.render(
        "{{#each XS}}{{#if LOUD}}${ITEM}!{{else}}${ITEM}{{/if}},{{/each}}",
        {"XS": ["a", "b"], "LOUD": "yes"})
    assert out == "a!,b!,"
    # missing partial is a validation error
    import pytest as _pytest
    with _pytest.raises(kerrors.ValidationError):
        eng.render("{{> nope}}", {})
    # needs-merge unions in order without dupes
    assert merge_needs(["git"], ["git", "gpu"], None) == ["git", "gpu"]


def test_render_team_with_partials_and_needs_merge(tmp_path):
    """Full pipeline: a harness template using partials + conditionals
    renders per role with team/role vars and the merged needs drive
    image selection."""
    from kukeon_amd.teams import parse_team_doc
    from kukeon_amd.teams.pipeline import render_team
    import yaml as _yaml

    src = tmp_path / "agents"
    (src / "partials").mkdir(parents=True)
    (src / "partials" / "cellspec.tmpl").write_text(
        "containers:\n"
        "  - id: main\n"
        "    image: ${IMAGE}\n"
        "    command: sleep\n"
        "    args: [\"60\"]\n"
        "{{#if GPU}}    gpus: ${GPU}\n{{/if}}")
    (src / "agent.tmpl").write_text(
        "apiVersion: v1beta1\n"
        "kind: CellBlueprint\n"
        "metadata: {name: ${TEAM}-${ROLE}-${HARNESS}}\n"
        "spec:\n"
        "  namePrefix: ${ROLE}\n"
        "  template:\n"
        "    spec:\n"
        "      stackId: default\n"
        "      {{> cellspec}}\n")
    team = parse_team_doc(_yaml.safe_load("""
apiVersion: kuketeams.io/v1
kind: ProjectTeam
metadata: {name: squad}
spec:
  source: {path: .}
  defaults:
    harnesses: [cc]
    needs: {image: [git]}
  vars: {GPU: "1"}
  roles:
    - ref: builder
      needs: {image: [gpu]}
    - ref: scout
      vars: {GPU: ""}
"""))
    roles = {
        "builder": parse_team_doc(_yaml.safe_load(
            "apiVersion: kuketeams.io/v1\nkind: Role\n"
            "metadata: {name: builder}\nspec: {prompt: build}")),
        "scout": parse_team_doc(_yaml.safe_load(
            "apiVersion: kuketeams.io/v1\nkind: Role\n"
            "metadata: {name: scout}\nspec: {prompt: scout}")),
    }
    harness = parse_team_doc(_yaml.safe_load(
        "apiVersion: kuketeams.io/v1\nkind: Harness\n"
        "metadata: {name: cc}\nspec: {template: agent.tmpl}"))
    catalog = parse_team_doc(_yaml.safe_load("""
apiVersion: kuketeams.io/v1
kind: ImageCatalog
metadata: {name: cat}
spec:
  images:
    - ref: base
      harness: cc
      capabilities: [git]
      image: reg/base
    - ref: gpu
      harness: cc
      capabilities: [git, gpu]
      image: reg/gpu
"""))
    docs = render_team(team, roles, {"cc": harness}, catalog, src,
                       tmp_path, "default", "default")
    by_name = {d["metadata"]["name"]: d for d in docs}
    b = by_name["squad-builder-cc"]["spec"]["template"]["spec"]
    s = by_name["squad-scout-cc"]["spec"]["template"]["spec"]
    # merged needs [git, gpu] -> the gpu-capable image for builder
    assert b["containers"][0]["image"] == "reg/gpu"
    assert s["containers"][0]["image"] == "reg/base"
    # conditional: builder (GPU=1 team var) gets gpus; scout overrode
    assert b["containers"][0].get("gpus") == 1
    assert "gpus" not in s["containers"][0]
